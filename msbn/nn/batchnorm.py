"""msbn SyncBatchNorm module layer.

API parity with ``torch.nn.SyncBatchNorm`` as used by the reference recipe
(/root/reference/README.md:44-60; behavior map SURVEY.md §2.2
"batchnorm.py:614-902"):

  * ``SyncBatchNorm(num_features, eps, momentum, affine, track_running_stats,
    process_group)``
  * ``SyncBatchNorm.convert_sync_batchnorm(module, process_group=None)``
    converts every BatchNorm1d/2d/3d (torch's or msbn's) in a module tree.

Differences from stock, by design (documented, not accidental):
  * The sync path also runs on CPU tensors when a process group is active
    (gloo), so the CPU/gloo plumbing config exercises the real distributed
    algorithm instead of silently falling back to per-process stats.
  * Cross-replica combination masks zero-count ranks inside the gather kernel:
    no GPU->CPU sync per layer and safe under hipGraph capture.
"""

from typing import Optional

import torch
import torch.distributed as dist
from torch import Tensor
from torch.nn import Module, Parameter, init

from msbn.nn.functions import SyncBatchNormFunction


class _NormBase(Module):
    """Common machinery: affine parameters, running stats, state-dict versioning.

    Mirrors the buffer layout of the stock ``_NormBase`` (running_mean,
    running_var, num_batches_tracked — SURVEY.md §2.2) so state_dicts are
    interchangeable with torch BatchNorm / SyncBatchNorm checkpoints.
    """

    _version = 2
    num_features: int
    eps: float
    momentum: Optional[float]
    affine: bool
    track_running_stats: bool

    def __init__(
        self,
        num_features: int,
        eps: float = 1e-5,
        momentum: Optional[float] = 0.1,
        affine: bool = True,
        track_running_stats: bool = True,
        device=None,
        dtype=None,
    ) -> None:
        factory_kwargs = {"device": device, "dtype": dtype}
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.affine = affine
        self.track_running_stats = track_running_stats
        if self.affine:
            self.weight = Parameter(torch.empty(num_features, **factory_kwargs))
            self.bias = Parameter(torch.empty(num_features, **factory_kwargs))
        else:
            self.register_parameter("weight", None)
            self.register_parameter("bias", None)
        if self.track_running_stats:
            self.register_buffer(
                "running_mean", torch.zeros(num_features, **factory_kwargs)
            )
            self.register_buffer(
                "running_var", torch.ones(num_features, **factory_kwargs)
            )
            self.register_buffer(
                "num_batches_tracked",
                torch.tensor(
                    0, dtype=torch.long,
                    **{k: v for k, v in factory_kwargs.items() if k == "device"},
                ),
            )
        else:
            self.register_buffer("running_mean", None)
            self.register_buffer("running_var", None)
            self.register_buffer("num_batches_tracked", None)
        self.reset_parameters()

    def reset_running_stats(self) -> None:
        if self.track_running_stats:
            self.running_mean.zero_()
            self.running_var.fill_(1)
            self.num_batches_tracked.zero_()

    def reset_parameters(self) -> None:
        self.reset_running_stats()
        if self.affine:
            init.ones_(self.weight)
            init.zeros_(self.bias)

    def _check_input_dim(self, input):
        raise NotImplementedError

    def extra_repr(self):
        return (
            "{num_features}, eps={eps}, momentum={momentum}, affine={affine}, "
            "track_running_stats={track_running_stats}".format(**self.__dict__)
        )

    def _load_from_state_dict(
        self, state_dict, prefix, local_metadata, strict,
        missing_keys, unexpected_keys, error_msgs,
    ):
        version = local_metadata.get("version", None)
        if (version is None or version < 2) and self.track_running_stats:
            # version 1 -> 2 migration: num_batches_tracked added
            # (stock behavior, batchnorm.py:109-141)
            key = prefix + "num_batches_tracked"
            if key not in state_dict:
                state_dict[key] = (
                    self.num_batches_tracked
                    if self.num_batches_tracked is not None
                    else torch.tensor(0, dtype=torch.long)
                )
        super()._load_from_state_dict(
            state_dict, prefix, local_metadata, strict,
            missing_keys, unexpected_keys, error_msgs,
        )


class _BatchNorm(_NormBase):
    """Non-sync N-D BatchNorm (the single-process fallback family).

    On GPU the training path runs the same hand-written gfx950 kernels as the
    sync path (world_size == 1 skips the collectives); eval normalizes with
    running stats.
    """

    def forward(self, input: Tensor) -> Tensor:
        self._check_input_dim(input)
        return _batch_norm_forward(self, input, sync=False)


def _momentum_factor(module) -> float:
    """exponential_average_factor incl. cumulative moving average when
    momentum is None (stock behavior, batchnorm.py:754-765)."""
    if module.momentum is None:
        if module.num_batches_tracked is not None:
            # .item() is a host sync — illegal inside hipGraph capture.  The
            # CMA factor changes every step, so this config is fundamentally
            # incompatible with whole-step capture; fail loudly instead of
            # corrupting the capture.
            if (
                module.num_batches_tracked.is_cuda
                and torch.cuda.is_current_stream_capturing()
            ):
                raise RuntimeError(
                    "SyncBatchNorm with momentum=None (cumulative moving "
                    "average) reads num_batches_tracked on the host each "
                    "step and cannot be captured in a hipGraph; use a fixed "
                    "momentum for graphed steps"
                )
            return 1.0 / float(module.num_batches_tracked.item())
        return 0.0
    return module.momentum


def _batch_norm_forward(module, input: Tensor, sync: bool) -> Tensor:
    if input.dim() < 2:
        raise ValueError(f"expected at least 2D input (got {input.dim()}D input)")

    bn_training = module.training or (
        module.running_mean is None and module.running_var is None
    )

    if module.training and module.track_running_stats:
        if module.num_batches_tracked is not None:
            module.num_batches_tracked.add_(1)
    factor = _momentum_factor(module) if module.training else (
        module.momentum if module.momentum is not None else 0.0
    )

    if not bn_training:
        # Eval fast path: fused elemt kernel when no autograd is needed
        # (inference serving); composed differentiable expression otherwise.
        if input.is_cuda and not torch.is_grad_enabled():
            from msbn import ops as _ops
            from msbn.nn.functions import _contig

            input = _contig(input)
            rm = module.running_mean.to(torch.float32)
            invstd = torch.rsqrt(module.running_var.to(torch.float32) + module.eps)
            return _ops.batch_norm_elemt_act(
                input, None, module.weight, module.bias, rm, invstd, False
            )
        rm = module.running_mean.to(torch.float32)
        rv = module.running_var.to(torch.float32)
        invstd = torch.rsqrt(rv + module.eps)
        scale = invstd
        shift = -rm * invstd
        if module.weight is not None:
            wf = module.weight.to(torch.float32)
            scale = scale * wf
            shift = shift * wf
        if module.bias is not None:
            shift = shift + module.bias.to(torch.float32)
        shape = [1] * input.dim()
        shape[1] = input.shape[1]
        out = input.to(torch.float32) * scale.reshape(shape) + shift.reshape(shape)
        return out.to(input.dtype)

    process_group = None
    world_size = 1
    if sync and dist.is_available() and dist.is_initialized():
        process_group = getattr(module, "process_group", None) or dist.group.WORLD
        world_size = dist.get_world_size(process_group)

    running_mean = module.running_mean if module.track_running_stats else None
    running_var = module.running_var if module.track_running_stats else None
    return SyncBatchNormFunction.apply(
        input,
        module.weight,
        module.bias,
        running_mean,
        running_var,
        module.eps,
        factor,
        process_group,
        world_size,
    )


class BatchNorm1d(_BatchNorm):
    def _check_input_dim(self, input):
        if input.dim() not in (2, 3):
            raise ValueError(f"expected 2D or 3D input (got {input.dim()}D input)")


class BatchNorm2d(_BatchNorm):
    def _check_input_dim(self, input):
        if input.dim() != 4:
            raise ValueError(f"expected 4D input (got {input.dim()}D input)")


class BatchNorm3d(_BatchNorm):
    def _check_input_dim(self, input):
        if input.dim() != 5:
            raise ValueError(f"expected 5D input (got {input.dim()}D input)")


class SyncBatchNorm(_NormBase):
    """N-D BatchNorm with cross-replica statistics over RCCL/xGMI.

    Drop-in for ``torch.nn.SyncBatchNorm`` (README.md:44-48): per-GPU moments
    are computed by gfx950 HIP kernels, the (2C+1)-float packed moment vector
    is all-gathered across the process group, and the combined mean/invstd
    normalize the local activations — activations never leave their GPU.
    """

    def __init__(
        self,
        num_features: int,
        eps: float = 1e-5,
        momentum: Optional[float] = 0.1,
        affine: bool = True,
        track_running_stats: bool = True,
        process_group=None,
        device=None,
        dtype=None,
    ) -> None:
        super().__init__(
            num_features, eps, momentum, affine, track_running_stats, device, dtype
        )
        self.process_group = process_group

    def _check_input_dim(self, input):
        if input.dim() < 2:
            raise ValueError(f"expected at least 2D input (got {input.dim()}D input)")

    def _check_non_zero_input_channels(self, input):
        if input.size(1) == 0:
            raise ValueError(
                "SyncBatchNorm number of input channels should be non-zero"
            )

    def forward(self, input: Tensor) -> Tensor:
        self._check_input_dim(input)
        self._check_non_zero_input_channels(input)
        return _batch_norm_forward(self, input, sync=True)

    @classmethod
    def convert_sync_batchnorm(cls, module: Module, process_group=None) -> Module:
        """Recursively replace every BatchNorm child (torch's BatchNorm1d/2d/3d,
        torch.nn.SyncBatchNorm, or msbn's _BatchNorm family) with
        ``msbn.nn.SyncBatchNorm``, preserving parameters, running stats,
        ``num_batches_tracked``, training flag and qconfig
        (stock behavior, batchnorm.py:842-902).
        """
        module_output = module
        targets = (torch.nn.modules.batchnorm._BatchNorm, _NormBase)
        if isinstance(module, targets) and not isinstance(module, SyncBatchNorm):
            module_output = SyncBatchNorm(
                module.num_features,
                module.eps,
                module.momentum,
                module.affine,
                module.track_running_stats,
                process_group,
            )
            if module.affine:
                with torch.no_grad():
                    module_output.weight = module.weight
                    module_output.bias = module.bias
            module_output.running_mean = module.running_mean
            module_output.running_var = module.running_var
            module_output.num_batches_tracked = module.num_batches_tracked
            module_output.training = module.training
            if hasattr(module, "qconfig"):
                module_output.qconfig = module.qconfig
        for name, child in module.named_children():
            module_output.add_module(
                name, cls.convert_sync_batchnorm(child, process_group)
            )
        del module
        return module_output


convert_sync_batchnorm = SyncBatchNorm.convert_sync_batchnorm
