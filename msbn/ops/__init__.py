"""msbn.ops — the BatchNorm math op layer.

Dispatch policy:
  * CUDA (= HIP/MI355X) tensors -> the hand-written gfx950 kernels in the
    in-tree extension ``msbn._C`` (built by setup.py / __graft_entry__.build()).
    If the extension is missing on a GPU machine the ops raise — there is no
    silent eager fallback on the GPU path.
  * CPU tensors -> the pure-PyTorch reference implementations (used by the
    CPU/gloo plumbing config and as the numerics oracle in tests).

Op semantics: SURVEY.md §2.3 (the five stock SyncBatchNorm ATen ops).
"""

from typing import Optional

import torch

from msbn.ops import _reference as _ref

_C = None
_C_IMPORT_ERROR: Optional[BaseException] = None
try:  # built in-tree: msbn/_C*.so
    import os as _os

    if _os.environ.get("MSBN_AB_NONT", "0") == "1":
        # A/B measurement knob: route every op through the cached-access
        # build (no nontemporal bits).  Needs MSBN_BUILD_NONT=1 at setup.
        from msbn import _C_nont as _C  # type: ignore
    else:
        from msbn import _C as _C  # type: ignore
except Exception as e:  # pragma: no cover - exercised only when ext missing
    _C_IMPORT_ERROR = e


def hip_available() -> bool:
    """True when the gfx950 extension is importable."""
    return _C is not None


def _require_hip():
    if _C is None:
        raise RuntimeError(
            "msbn._C (the gfx950 HIP kernel extension) is not built, but a CUDA "
            "tensor reached msbn.ops. Build it with `python setup.py "
            "build_ext --inplace` (or __graft_entry__.build()). "
            f"Import error was: {_C_IMPORT_ERROR!r}"
        )
    return _C


def batch_norm_stats(input: torch.Tensor, eps: float):
    """Per-channel (mean, invstd) in fp32; invstd = 1/sqrt(biased_var + eps)."""
    if input.is_cuda:
        return _require_hip().batch_norm_stats(input, eps)
    return _ref.batch_norm_stats(input, eps)


def batch_norm_stats_packed(input: torch.Tensor, eps: float, out: torch.Tensor):
    """Fused variant: write [mean(C) | invstd(C) | count(1)] into ``out`` (2C+1 fp32).

    Saves the cat() + extra kernels of the stock path (SURVEY.md §2.2
    _functions.py:41-49) — the packed buffer is what the all_gather ships.
    Falls back to stats + copies on CPU.
    """
    C = input.shape[1]
    if input.is_cuda:
        _require_hip().batch_norm_stats_packed(input, eps, out)
        return out
    mean, invstd = _ref.batch_norm_stats(input, eps)
    out[:C] = mean
    out[C : 2 * C] = invstd
    n = input.numel() // C if C > 0 else 0
    out[2 * C] = float(n)
    return out


def batch_norm_gather_stats_with_counts(
    input: torch.Tensor,
    mean_all: torch.Tensor,
    invstd_all: torch.Tensor,
    running_mean: Optional[torch.Tensor],
    running_var: Optional[torch.Tensor],
    momentum: float,
    eps: float,
    counts: torch.Tensor,
):
    if input.is_cuda:
        return _require_hip().batch_norm_gather_stats_with_counts(
            mean_all, invstd_all, running_mean, running_var, momentum, eps, counts
        )
    return _ref.batch_norm_gather_stats_with_counts(
        input, mean_all, invstd_all, running_mean, running_var, momentum, eps, counts
    )


def batch_norm_gather_stats_packed(
    input: torch.Tensor,
    packed_all: torch.Tensor,  # [W, 2C+1] fp32: per-rank [mean | invstd | count]
    running_mean: Optional[torch.Tensor],
    running_var: Optional[torch.Tensor],
    momentum: float,
    eps: float,
):
    """Combine the all-gathered packed per-rank stats; returns (mean, invstd, count_sum).

    Zero-count ranks are masked INSIDE the kernel — no GPU->CPU sync (the stock
    path's mask at _functions.py:88-100 forces one per BN layer).
    count_sum is returned as a 1-element fp32 tensor on device.
    """
    W = packed_all.shape[0]
    C = (packed_all.shape[1] - 1) // 2
    if input.is_cuda:
        return _require_hip().batch_norm_gather_stats_packed(
            packed_all, running_mean, running_var, momentum, eps
        )
    mean_all = packed_all[:, :C]
    invstd_all = packed_all[:, C : 2 * C]
    counts = packed_all[:, 2 * C]
    mean, invstd = _ref.batch_norm_gather_stats_with_counts(
        input, mean_all, invstd_all, running_mean, running_var, momentum, eps, counts
    )
    return mean, invstd, counts.sum().reshape(1)


def batch_norm_elemt(
    input: torch.Tensor,
    weight: Optional[torch.Tensor],
    bias: Optional[torch.Tensor],
    mean: torch.Tensor,
    invstd: torch.Tensor,
    eps: float,
):
    if input.is_cuda:
        return _require_hip().batch_norm_elemt(input, weight, bias, mean, invstd, eps)
    return _ref.batch_norm_elemt(input, weight, bias, mean, invstd, eps)


def batch_norm_backward_reduce(
    grad_out: torch.Tensor,
    input: torch.Tensor,
    mean: torch.Tensor,
    invstd: torch.Tensor,
    weight: Optional[torch.Tensor],
    input_g: bool,
    weight_g: bool,
    bias_g: bool,
):
    if input.is_cuda:
        return _require_hip().batch_norm_backward_reduce(
            grad_out, input, mean, invstd, weight, input_g, weight_g, bias_g
        )
    return _ref.batch_norm_backward_reduce(
        grad_out, input, mean, invstd, weight, input_g, weight_g, bias_g
    )


def batch_norm_backward_elemt(
    grad_out: torch.Tensor,
    input: torch.Tensor,
    mean: torch.Tensor,
    invstd: torch.Tensor,
    weight: Optional[torch.Tensor],
    sum_dy: torch.Tensor,
    sum_dy_xmu: torch.Tensor,
    count: torch.Tensor,
):
    if input.is_cuda:
        return _require_hip().batch_norm_backward_elemt(
            grad_out, input, mean, invstd, weight, sum_dy, sum_dy_xmu, count
        )
    return _ref.batch_norm_backward_elemt(
        grad_out, input, mean, invstd, weight, sum_dy, sum_dy_xmu, count
    )


def bn_fused_local_eligible(input, weight, bias, running_mean, running_var):
    """True when the single-launch small-plane world-1 kernels apply
    (NCHW, plane <= 256K elems, C >= 64, fp32 stats/affine)."""
    if not input.is_cuda or _C is None:
        return False
    return _C.bn_fused_local_eligible(
        input, weight, bias, running_mean, running_var
    )


def batch_norm_fwd_fused_local(input, residual, weight, bias, eps, momentum,
                               running_mean, running_var, relu):
    """ONE kernel: local stats + running update + normalize(+res)(+relu).
    Returns (y, mean, invstd, count[1], coefs[2C])."""
    return _require_hip().batch_norm_fwd_fused_local(
        input, residual, weight, bias, eps, momentum, running_mean,
        running_var, relu,
    )


def batch_norm_bwd_fused_local(grad_out, input, residual, mean, invstd,
                               weight, coefs, relu_mask, want_res_grad,
                               weight_g, bias_g):
    """ONE kernel: whole local BN backward (reduce + coefs + dx(+dres))."""
    return _require_hip().batch_norm_bwd_fused_local(
        grad_out, input, residual, mean, invstd, weight, coefs, relu_mask,
        want_res_grad, weight_g, bias_g,
    )


def bn_make_coefs(mean, invstd, weight, bias):
    """Packed [scale | shift] fp32 per-channel affine (GPU); None on CPU."""
    if mean.is_cuda:
        return _require_hip().bn_make_coefs(mean, invstd, weight, bias)
    return None


def batch_norm_elemt_act(input, residual, weight, bias, mean, invstd,
                         relu: bool, coefs=None):
    if input.is_cuda:
        return _require_hip().batch_norm_elemt_act(
            input, residual, weight, bias, mean, invstd, relu, coefs
        )
    return _ref.batch_norm_elemt_act(
        input, residual, weight, bias, mean, invstd, relu
    )


def batch_norm_backward_reduce_act(grad_out, input, residual, mean, invstd,
                                   weight, bias, relu_mask, input_g, weight_g,
                                   bias_g, coefs=None, gm_out=None):
    if input.is_cuda:
        return _require_hip().batch_norm_backward_reduce_act(
            grad_out, input, residual, mean, invstd, weight, bias, relu_mask,
            input_g, weight_g, bias_g, coefs, gm_out,
        )
    return _ref.batch_norm_backward_reduce_act(
        grad_out, input, residual, mean, invstd, weight, bias, relu_mask,
        input_g, weight_g, bias_g, gm_out,
    )


def batch_norm_backward_elemt_act(grad_out, input, residual, mean, invstd,
                                  weight, bias, sum_dy, sum_dy_xmu, count,
                                  relu_mask, want_res_grad, coefs=None):
    if input.is_cuda:
        dx, dres = _require_hip().batch_norm_backward_elemt_act(
            grad_out, input, residual, mean, invstd, weight, bias, sum_dy,
            sum_dy_xmu, count, relu_mask, want_res_grad, coefs,
        )
        return dx, (dres if want_res_grad else None)
    return _ref.batch_norm_backward_elemt_act(
        grad_out, input, residual, mean, invstd, weight, bias, sum_dy,
        sum_dy_xmu, count, relu_mask, want_res_grad,
    )


__all__ = [
    "hip_available",
    "batch_norm_elemt_act",
    "bn_make_coefs",
    "batch_norm_backward_reduce_act",
    "batch_norm_backward_elemt_act",
    "batch_norm_stats",
    "batch_norm_stats_packed",
    "batch_norm_gather_stats_with_counts",
    "batch_norm_gather_stats_packed",
    "batch_norm_elemt",
    "batch_norm_backward_reduce",
    "batch_norm_backward_elemt",
]
