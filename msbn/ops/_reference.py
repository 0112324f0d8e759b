"""Pure-PyTorch reference implementations of the msbn BatchNorm op set.

These are the CPU execution path and the numerics oracle for the HIP kernels.
Semantics mirror the five ATen ops the stock SyncBatchNorm path calls
(signatures verified in SURVEY.md §2.3 against
/usr/local/lib/python3.10/dist-packages/torch/include/ATen/ops/batch_norm_*.h):

  batch_norm_stats                       -> per-channel (mean, invstd), biased var
  batch_norm_gather_stats_with_counts    -> combine per-rank moments (Chan merge),
                                            update running stats in-place (unbiased)
  batch_norm_elemt                       -> y = (x - mean) * invstd * w + b
  batch_norm_backward_reduce             -> (sum_dy, sum_dy_xmu, grad_weight, grad_bias)
  batch_norm_backward_elemt              -> grad_input with global count

All statistics accumulate in float32 (float64 internally for the reductions, to
serve as oracle), matching the fp32-accumulator rule of the stock kernels for
half/bf16 inputs (SURVEY.md §2.3 mixed-dtype rules).
"""

from typing import Optional, Tuple

import torch


def _flatten_to_ncs(input: torch.Tensor) -> torch.Tensor:
    """View (N, C, *spatial) as (N, C, S). 2-D input (N, C) becomes (N, C, 1)."""
    if input.dim() == 2:
        return input.unsqueeze(-1)
    return input.reshape(input.shape[0], input.shape[1], -1)


def batch_norm_stats(input: torch.Tensor, eps: float) -> Tuple[torch.Tensor, torch.Tensor]:
    """Per-channel mean and inverse std (1/sqrt(biased_var + eps)) over N x spatial.

    Returns float32 tensors of shape [C].  Empty input (0 elements per channel)
    returns zeros for mean and invstd (the caller masks by count, matching the
    zero-count handling of the stock sync path).
    """
    x = _flatten_to_ncs(input)
    C = x.shape[1]
    n = x.shape[0] * x.shape[2]
    if n == 0:
        z = torch.zeros(C, dtype=torch.float32, device=input.device)
        return z, z.clone()
    xd = x.transpose(0, 1).reshape(C, -1).to(torch.float64)
    mean = xd.mean(dim=1)
    var = xd.var(dim=1, unbiased=False)
    invstd = torch.rsqrt(var + eps)
    return mean.to(torch.float32), invstd.to(torch.float32)


def batch_norm_gather_stats_with_counts(
    input: torch.Tensor,
    mean_all: torch.Tensor,
    invstd_all: torch.Tensor,
    running_mean: Optional[torch.Tensor],
    running_var: Optional[torch.Tensor],
    momentum: float,
    eps: float,
    counts: torch.Tensor,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Combine W rows of per-rank (mean, invstd, count) into global (mean, invstd).

    mean_all/invstd_all: [W, C] float32; counts: [W] (any float/int dtype).
    Zero-count rows are ignored.  Updates running_mean/running_var in-place with
    momentum and the UNBIASED global variance (n-1 divisor), matching
    batch_norm_gather_stats_with_counts.
    ``input`` is only consulted for dtype/device (as in ATen).
    """
    cnt = counts.to(torch.float64)
    m = mean_all.to(torch.float64)
    # var_r reconstructed from invstd: var = 1/invstd^2 - eps.
    # Guard zero-count ranks (invstd == 0) BEFORE the reciprocal.
    istd = invstd_all.to(torch.float64)
    safe = torch.where(istd == 0, torch.ones_like(istd), istd)
    var = torch.where(istd == 0, torch.zeros_like(istd), safe.pow(-2) - eps)
    mask = (cnt > 0).to(torch.float64)  # [W]
    n_tot = (cnt * mask).sum()
    if n_tot.item() == 0:
        C = mean_all.shape[1]
        z = torch.zeros(C, dtype=torch.float32, device=mean_all.device)
        return z, z.clone()
    w = (cnt * mask).unsqueeze(1)  # [W,1]
    mean_g = (m * w).sum(dim=0) / n_tot
    # Chan combine: E[x^2] based merge (biased variance over the union)
    ex2 = ((var + m * m) * w).sum(dim=0) / n_tot
    var_g = ex2 - mean_g * mean_g
    invstd_g = torch.rsqrt(var_g + eps)

    if running_mean is not None:
        running_mean.mul_(1 - momentum).add_(
            mean_g.to(running_mean.dtype), alpha=momentum
        )
    if running_var is not None:
        unbiased = var_g * (n_tot / (n_tot - 1.0)) if n_tot.item() > 1 else var_g
        running_var.mul_(1 - momentum).add_(
            unbiased.to(running_var.dtype), alpha=momentum
        )
    return mean_g.to(torch.float32), invstd_g.to(torch.float32)


def _chan_view(t: torch.Tensor, ref: torch.Tensor) -> torch.Tensor:
    """Reshape per-channel [C] tensor for broadcast against (N, C, *spatial) ref."""
    shape = [1] * ref.dim()
    shape[1] = ref.shape[1]
    return t.reshape(shape)


def batch_norm_elemt(
    input: torch.Tensor,
    weight: Optional[torch.Tensor],
    bias: Optional[torch.Tensor],
    mean: torch.Tensor,
    invstd: torch.Tensor,
    eps: float,
) -> torch.Tensor:
    """y = (x - mean) * invstd * weight + bias, elementwise per channel.

    Output dtype == input dtype; math in float32.
    """
    xf = input.to(torch.float32)
    scale = invstd.to(torch.float32)
    if weight is not None:
        scale = scale * weight.to(torch.float32)
    shift = -mean.to(torch.float32) * scale
    if bias is not None:
        shift = shift + bias.to(torch.float32)
    y = xf * _chan_view(scale, input) + _chan_view(shift, input)
    return y.to(input.dtype)


def _act_coefs(mean, invstd, weight, bias):
    scale = invstd.to(torch.float32)
    if weight is not None:
        scale = scale * weight.to(torch.float32)
    shift = -mean.to(torch.float32) * scale
    if bias is not None:
        shift = shift + bias.to(torch.float32)
    return scale, shift


def batch_norm_elemt_act(
    input, residual, weight, bias, mean, invstd, relu: bool
):
    """y = relu?(x*scale + shift [+ residual]) — fused epilogue reference."""
    scale, shift = _act_coefs(mean, invstd, weight, bias)
    z = input.to(torch.float32) * _chan_view(scale, input) + _chan_view(
        shift, input
    )
    if residual is not None:
        z = z + residual.to(torch.float32)
    if relu:
        z = torch.relu(z)
    return z.to(input.dtype)


def _masked_grad(grad_out, input, residual, mean, invstd, weight, bias,
                 relu_mask: bool):
    g = grad_out.to(torch.float32)
    if relu_mask:
        scale, shift = _act_coefs(mean, invstd, weight, bias)
        z = input.to(torch.float32) * _chan_view(scale, input) + _chan_view(
            shift, input
        )
        if residual is not None:
            z = z + residual.to(torch.float32)
        g = torch.where(z > 0, g, torch.zeros_like(g))
    return g


def batch_norm_backward_reduce_act(
    grad_out, input, residual, mean, invstd, weight, bias, relu_mask,
    input_g, weight_g, bias_g, gm_out=None,
):
    g = _masked_grad(grad_out, input, residual, mean, invstd, weight, bias,
                     relu_mask)
    if gm_out is not None:
        gm_out.copy_(g.to(grad_out.dtype))
    return batch_norm_backward_reduce(
        g, input, mean, invstd, weight, input_g, weight_g, bias_g
    )


def batch_norm_backward_elemt_act(
    grad_out, input, residual, mean, invstd, weight, bias, sum_dy,
    sum_dy_xmu, count, relu_mask, want_res_grad,
):
    g = _masked_grad(grad_out, input, residual, mean, invstd, weight, bias,
                     relu_mask)
    dx = batch_norm_backward_elemt(
        g, input, mean, invstd, weight, sum_dy, sum_dy_xmu, count
    )
    dres = g.to(grad_out.dtype) if want_res_grad else None
    return dx, dres


def batch_norm_backward_reduce(
    grad_out: torch.Tensor,
    input: torch.Tensor,
    mean: torch.Tensor,
    invstd: torch.Tensor,
    weight: Optional[torch.Tensor],
    input_g: bool,
    weight_g: bool,
    bias_g: bool,
) -> Tuple[Optional[torch.Tensor], Optional[torch.Tensor], Optional[torch.Tensor], Optional[torch.Tensor]]:
    """Per-channel reductions of the BN backward:
      sum_dy      = sum(dy)                      [float32, C]  (if input_g)
      sum_dy_xmu  = sum(dy * (x - mean))         [float32, C]  (if input_g)
      grad_weight = sum(dy * (x - mean)) * invstd  (dtype of weight, if weight_g)
      grad_bias   = sum(dy)                        (dtype of weight, if bias_g)
    """
    dy = _flatten_to_ncs(grad_out).to(torch.float64)
    x = _flatten_to_ncs(input).to(torch.float64)
    m = mean.to(torch.float64).reshape(1, -1, 1)
    s_dy = dy.sum(dim=(0, 2))
    s_dy_xmu = (dy * (x - m)).sum(dim=(0, 2))
    wdtype = weight.dtype if weight is not None else torch.float32
    sum_dy = s_dy.to(torch.float32) if input_g else None
    sum_dy_xmu = s_dy_xmu.to(torch.float32) if input_g else None
    grad_weight = (s_dy_xmu * invstd.to(torch.float64)).to(wdtype) if weight_g else None
    grad_bias = s_dy.to(wdtype) if bias_g else None
    return sum_dy, sum_dy_xmu, grad_weight, grad_bias


def batch_norm_backward_elemt(
    grad_out: torch.Tensor,
    input: torch.Tensor,
    mean: torch.Tensor,
    invstd: torch.Tensor,
    weight: Optional[torch.Tensor],
    sum_dy: torch.Tensor,
    sum_dy_xmu: torch.Tensor,
    count: torch.Tensor,
) -> torch.Tensor:
    """grad_input = (dy - sum_dy/n - (x-mean)*invstd^2*sum_dy_xmu/n) * invstd * w

    with n = count.sum() (global element count across ranks).  Output dtype ==
    grad_out dtype; math in float32.
    """
    n = count.sum().to(torch.float32)
    dy = grad_out.to(torch.float32)
    x = input.to(torch.float32)
    istd = invstd.to(torch.float32)
    f_scale = istd * (weight.to(torch.float32) if weight is not None else 1.0)
    mean_dy = sum_dy.to(torch.float32) / n
    proj = istd * istd * sum_dy_xmu.to(torch.float32) / n
    dx = (
        dy
        - _chan_view(mean_dy, input)
        - (x - _chan_view(mean.to(torch.float32), input)) * _chan_view(proj, input)
    ) * _chan_view(f_scale, input)
    return dx.to(grad_out.dtype)
