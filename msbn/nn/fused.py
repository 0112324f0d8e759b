"""Fused SyncBatchNorm(+residual add)(+ReLU) — the msbn epilogue-fusion path.

Mathematically identical to ``relu(bn(x) + residual)`` composed from separate
modules, but executed as ONE elementwise kernel forward and mask-recomputing
kernels backward (no clamp / add / threshold_backward kernels, no stored mask,
~3x fewer full-tensor round trips on the BN backward — see
profiles/r01_single_gpu.md for the measured eager-kernel cost this removes).

``SyncBatchNormAct2d`` is used directly by the fused model variants
(msbn.models.resnet{18,50}(fused=True)); ``convert_sync_batchnorm`` leaves it
alone (it already IS a SyncBatchNorm subclass).
"""

from typing import Optional

import torch
import torch.distributed as dist

from msbn import ops
from msbn.nn.batchnorm import SyncBatchNorm, _momentum_factor
from msbn.nn.functions import (_combined_view, _contig, _force_sync,
                               _match_layout, compute_sync_stats)


class SyncBatchNormActFunction(torch.autograd.Function):
    @staticmethod
    def forward(
        ctx,
        input: torch.Tensor,
        residual: Optional[torch.Tensor],
        weight: Optional[torch.Tensor],
        bias: Optional[torch.Tensor],
        running_mean: Optional[torch.Tensor],
        running_var: Optional[torch.Tensor],
        eps: float,
        momentum: float,
        process_group,
        world_size: int,
        relu: bool,
    ):
        input = _contig(input)
        if residual is not None:
            residual = _match_layout(residual, input)
        if weight is not None:
            weight = weight.contiguous()
        if bias is not None:
            bias = bias.contiguous()

        use_sync = world_size > 1 or (
            process_group is not None and _force_sync()
        )
        if (
            not use_sync
            and input.numel() > 0
            and (residual is None or residual.is_contiguous())
            and ops.bn_fused_local_eligible(
                input, weight, bias, running_mean, running_var
            )
        ):
            # single-launch small-plane path: stats + running update +
            # normalize(+res)(+relu) in ONE kernel (K10-family)
            y, mean, invstd, count_sum, coefs = ops.batch_norm_fwd_fused_local(
                input, residual, weight, bias, eps, momentum,
                running_mean, running_var, relu,
            )
            ctx.save_for_backward(input, residual, weight, bias, mean,
                                  invstd, count_sum, coefs)
            ctx.has_coefs = True
            ctx.local_fused = True
            ctx.relu = relu
            ctx.process_group = process_group
            ctx.world_size = world_size
            return y
        ctx.local_fused = False

        # coefs ([scale|shift]) are emitted by the finalize/gather kernel in
        # the same launch and reused by forward elemt AND the two
        # mask-recomputing backward kernels.
        mean, invstd, count_sum, coefs = compute_sync_stats(
            input, eps, momentum, running_mean, running_var,
            process_group, world_size, weight, bias, want_coefs=True,
        )
        if coefs is None:
            ctx.save_for_backward(input, residual, weight, bias, mean, invstd,
                                  count_sum)
            ctx.has_coefs = False
        else:
            ctx.save_for_backward(input, residual, weight, bias, mean, invstd,
                                  count_sum, coefs)
            ctx.has_coefs = True
        ctx.relu = relu
        ctx.process_group = process_group
        ctx.world_size = world_size
        ctx.use_sync = use_sync
        if input.numel() == 0:
            return torch.empty_like(input)
        return ops.batch_norm_elemt_act(
            input, residual, weight, bias, mean, invstd, relu, coefs
        )

    @staticmethod
    def backward(ctx, grad_output: torch.Tensor):
        if ctx.has_coefs:
            (input, residual, weight, bias, mean, invstd, count_sum,
             coefs) = ctx.saved_tensors
        else:
            input, residual, weight, bias, mean, invstd, count_sum = (
                ctx.saved_tensors
            )
            coefs = None
        grad_output = _match_layout(grad_output, input)
        relu = ctx.relu
        process_group = ctx.process_group
        world_size = ctx.world_size
        need_input_g = ctx.needs_input_grad[0]
        need_res_g = residual is not None and ctx.needs_input_grad[1]
        need_weight_g = weight is not None and ctx.needs_input_grad[2]
        need_bias_g = bias is not None and ctx.needs_input_grad[3]

        C = int(input.shape[1])

        if getattr(ctx, "local_fused", False) and input.numel() > 0:
            # single-launch backward (mask recompute + reduce + coefs + dx
            # (+dres) in one kernel)
            grad_input, grad_weight, grad_bias, grad_res = (
                ops.batch_norm_bwd_fused_local(
                    grad_output, input, residual, mean, invstd, weight,
                    coefs, relu, need_res_g, need_weight_g, need_bias_g,
                )
            )
            return (
                grad_input if need_input_g else None,
                grad_res if need_res_g else None,
                grad_weight if need_weight_g else None,
                grad_bias if need_bias_g else None,
                None, None, None, None, None, None, None,
            )

        # Masked-grad materialization: when the residual branch needs its
        # gradient, the reduce pass writes gm = dy*1[z>0] once; the elemt
        # pass then reads gm instead of (dy, residual) and gm itself IS the
        # residual gradient -> one fewer full-tensor read+write (8A -> 7A).
        use_gm = (
            relu and residual is not None and need_res_g and input.is_cuda
            and input.numel() > 0
        )
        use_sync = getattr(ctx, "use_sync", world_size > 1)
        if input.numel() == 0:
            # empty-input rank (join): contribute zeros, keep peers unblocked
            if use_sync and (need_input_g or need_res_g):
                combined = torch.zeros(2 * C, dtype=torch.float32,
                                       device=grad_output.device)
                from msbn.utils import debug as _dbg
                if _dbg.enabled():  # peers verify; skipping would hang them
                    _dbg.verify_collective("syncbn.bwd.all_reduce", combined,
                                           process_group)
                dist.all_reduce(combined, dist.ReduceOp.SUM,
                                group=process_group)
            zg = torch.empty_like(grad_output)
            return (
                zg if need_input_g else None,
                torch.empty_like(grad_output) if need_res_g else None,
                torch.zeros_like(weight) if need_weight_g else None,
                torch.zeros_like(bias) if need_bias_g else None,
                None, None, None, None, None, None, None,
            )
        gm = torch.empty_like(grad_output) if use_gm else None
        sum_dy, sum_dy_xmu, grad_weight, grad_bias = (
            ops.batch_norm_backward_reduce_act(
                grad_output, input, residual, mean, invstd, weight, bias,
                relu, need_input_g, need_weight_g, need_bias_g, coefs, gm,
            )
        )
        grad_input = grad_res = None
        if need_input_g or need_res_g:
            if use_sync:
                combined, copied = _combined_view(sum_dy, sum_dy_xmu, C)
                from msbn.utils import debug as _dbg
                if _dbg.enabled():
                    _dbg.verify_collective("syncbn.bwd.all_reduce",
                                           combined, process_group)
                dist.all_reduce(combined, dist.ReduceOp.SUM, group=process_group)
                from msbn.utils.logging import comm_log
                comm_log.record("all_reduce", combined.numel() * 4,
                                f"syncbn bwd C={C}")
                if copied:
                    sum_dy, sum_dy_xmu = combined[:C], combined[C:]
            if use_gm:
                grad_input, _ = ops.batch_norm_backward_elemt_act(
                    gm, input, None, mean, invstd, weight, bias,
                    sum_dy, sum_dy_xmu, count_sum, False, False, coefs,
                )
                grad_res = gm
            else:
                grad_input, grad_res = ops.batch_norm_backward_elemt_act(
                    grad_output, input, residual, mean, invstd, weight, bias,
                    sum_dy, sum_dy_xmu, count_sum, relu, need_res_g, coefs,
                )
        return (
            grad_input if need_input_g else None,
            grad_res,
            grad_weight if need_weight_g else None,
            grad_bias if need_bias_g else None,
            None, None, None, None, None, None, None,
        )


class SyncBatchNormAct2d(SyncBatchNorm):
    """SyncBatchNorm with a fused (optional residual-add +) ReLU epilogue.

    forward(x, residual=None) == relu(bn(x) + residual) with identical math.
    """

    def __init__(self, num_features, eps=1e-5, momentum=0.1, affine=True,
                 track_running_stats=True, process_group=None, device=None,
                 dtype=None, relu: bool = True):
        super().__init__(num_features, eps, momentum, affine,
                         track_running_stats, process_group, device, dtype)
        self.relu = relu

    def forward(self, input: torch.Tensor,
                residual: Optional[torch.Tensor] = None) -> torch.Tensor:
        self._check_input_dim(input)
        self._check_non_zero_input_channels(input)

        bn_training = self.training or (
            self.running_mean is None and self.running_var is None
        )
        if self.training and self.track_running_stats:
            if self.num_batches_tracked is not None:
                self.num_batches_tracked.add_(1)
        factor = _momentum_factor(self) if self.training else (
            self.momentum if self.momentum is not None else 0.0
        )

        if not bn_training:
            # eval fast path: one fused kernel under no_grad (serving)
            if input.is_cuda and not torch.is_grad_enabled():
                from msbn import ops as _ops

                input = _contig(input)
                if residual is not None:
                    residual = _match_layout(residual, input)
                rm = self.running_mean.to(torch.float32)
                invstd = torch.rsqrt(
                    self.running_var.to(torch.float32) + self.eps
                )
                return _ops.batch_norm_elemt_act(
                    input, residual, self.weight, self.bias, rm, invstd,
                    self.relu,
                )
            # composed (differentiable) path
            rm = self.running_mean.to(torch.float32)
            rv = self.running_var.to(torch.float32)
            invstd = torch.rsqrt(rv + self.eps)
            scale = invstd
            shift = -rm * invstd
            if self.weight is not None:
                wf = self.weight.to(torch.float32)
                scale = scale * wf
                shift = shift * wf
            if self.bias is not None:
                shift = shift + self.bias.to(torch.float32)
            shape = [1] * input.dim()
            shape[1] = input.shape[1]
            out = input.to(torch.float32) * scale.reshape(shape) + \
                shift.reshape(shape)
            if residual is not None:
                out = out + residual.to(torch.float32)
            if self.relu:
                out = torch.relu(out)
            return out.to(input.dtype)

        process_group = None
        world_size = 1
        if dist.is_available() and dist.is_initialized():
            process_group = self.process_group or dist.group.WORLD
            world_size = dist.get_world_size(process_group)

        running_mean = self.running_mean if self.track_running_stats else None
        running_var = self.running_var if self.track_running_stats else None
        return SyncBatchNormActFunction.apply(
            input, residual, self.weight, self.bias, running_mean, running_var,
            self.eps, factor, process_group, world_size, self.relu,
        )
