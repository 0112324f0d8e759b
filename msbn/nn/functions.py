"""The distributed SyncBatchNorm autograd function.

Algorithm (behavioral parity with the stock sync path, SURVEY.md §2.2
"_functions.py:7-209", re-designed MI355X-first):

forward:
  1. local per-channel moments via ONE fused kernel writing the packed
     [mean | invstd | count] (2C+1 fp32) buffer directly (no cat()).
  2. all_gather of the packed buffer over RCCL/xGMI -> [W, 2C+1].
  3. fused gather kernel: zero-count masking IN-KERNEL (no GPU->CPU sync,
     graph-capture safe), Chan variance merge with heterogeneous counts,
     running-stats update (unbiased, momentum) in the same launch.
  4. fused normalize-scale-shift (elemt kernel).

backward:
  1. fused per-channel reduction -> packed [sum_dy | sum_dy_xmu] (2C fp32)
     plus grad_weight/grad_bias.
  2. ONE all_reduce(SUM) of the packed 2C buffer.
  3. fused grad_input kernel with the global element count.
  grad_weight / grad_bias are NOT synced here — the DDP reducer's gradient
  buckets carry them (same division of labor as stock, _functions.py:180-181).

Empty-input ranks still post both collectives so peers are never blocked.
"""

import os
from typing import Optional

import torch
import torch.distributed as dist

from msbn import ops


def _force_sync() -> bool:
    """MSBN_FORCE_SYNC=1 runs the full distributed stat-sync path (packed
    stats kernel -> all_gather -> gather kernel, and the backward all_reduce)
    even at world_size == 1.  Debug/validation knob: a 1-GPU nccl process
    group then exercises the exact RCCL code path the 8-GPU job runs, and the
    result must be bit-identical to the local path."""
    return os.environ.get("MSBN_FORCE_SYNC", "0") == "1"


def _combined_view(sum_dy: torch.Tensor, sum_dy_xmu: torch.Tensor, C: int):
    """The HIP backward-reduce op returns sum_dy / sum_dy_xmu as views of ONE
    contiguous [2C] buffer; all_reduce that buffer directly instead of
    cat-copying.  Falls back to cat when the two are not adjacent (CPU ref)."""
    if (
        sum_dy.untyped_storage().data_ptr()
        == sum_dy_xmu.untyped_storage().data_ptr()
        and sum_dy.data_ptr() + sum_dy.numel() * sum_dy.element_size()
        == sum_dy_xmu.data_ptr()
        and sum_dy.is_contiguous() and sum_dy_xmu.is_contiguous()
    ):
        base = sum_dy.new_empty(0)
        base.set_(
            sum_dy.untyped_storage(), sum_dy.storage_offset(), (2 * C,), (1,)
        )
        return base, False
    return torch.cat([sum_dy, sum_dy_xmu]), True


def _is_nccl_like(process_group) -> bool:
    try:
        return dist.get_backend(process_group) in ("nccl", "hccl")
    except Exception:
        return False


def _contig(t: torch.Tensor) -> torch.Tensor:
    if t.is_contiguous(memory_format=torch.channels_last) or t.is_contiguous(
        memory_format=torch.channels_last_3d
    ):
        return t
    return t.contiguous()


def _match_layout(t: torch.Tensor, ref: torch.Tensor) -> torch.Tensor:
    """Coerce t to ref's memory format (the elementwise/reduce kernels index
    both tensors with ONE layout)."""
    if ref.dim() == 4 and ref.is_contiguous(memory_format=torch.channels_last):
        return t.contiguous(memory_format=torch.channels_last)
    if ref.dim() == 5 and ref.is_contiguous(
        memory_format=torch.channels_last_3d
    ):
        return t.contiguous(memory_format=torch.channels_last_3d)
    return t.contiguous()


def compute_sync_stats(
    input: torch.Tensor,
    eps: float,
    momentum: float,
    running_mean: Optional[torch.Tensor],
    running_var: Optional[torch.Tensor],
    process_group,
    world_size: int,
    weight: Optional[torch.Tensor] = None,
    bias: Optional[torch.Tensor] = None,
    want_coefs: bool = False,
):
    """Local packed moments -> all_gather over RCCL/xGMI (S5) -> counts-weighted
    combine with in-kernel zero-count masking + running-stats update.  The
    combine/finalize kernel also emits the per-channel [scale|shift] coefs
    when requested (one launch for the whole per-layer channel math).
    Returns (mean, invstd, count_sum[1], coefs-or-None)."""
    C = int(input.shape[1])
    local_count = input.numel() // C if C > 0 else 0
    on_gpu = input.is_cuda

    if world_size > 1 or (process_group is not None and _force_sync()):
        packed = torch.empty(2 * C + 1, dtype=torch.float32, device=input.device)
        if local_count > 0:
            ops.batch_norm_stats_packed(input, eps, packed)
        else:
            packed.zero_()

        packed_all = torch.empty(
            (world_size, 2 * C + 1), dtype=torch.float32, device=input.device
        )
        from msbn.utils.logging import comm_log
        from msbn.utils import debug as _dbg

        if _dbg.enabled():
            _dbg.verify_collective("syncbn.fwd.all_gather", packed,
                                   process_group)
        if _is_nccl_like(process_group):
            dist.all_gather_into_tensor(packed_all, packed, group=process_group)
        else:
            chunks = list(packed_all.unbind(0))
            dist.all_gather(chunks, packed, group=process_group)
            packed_all = torch.stack(chunks, dim=0)
        comm_log.record("all_gather", packed.numel() * 4, f"syncbn C={C}")

        if on_gpu and want_coefs:
            from msbn.ops import _require_hip

            mean, invstd, count_sum, coefs = (
                _require_hip().batch_norm_gather_stats_packed_coefs(
                    packed_all, running_mean, running_var, momentum, eps,
                    weight, bias, True,
                )
            )
            return mean, invstd, count_sum, coefs
        mean, invstd, count_sum = ops.batch_norm_gather_stats_packed(
            input, packed_all, running_mean, running_var, momentum, eps
        )
        return mean, invstd, count_sum, None

    if on_gpu and local_count > 0:
        from msbn.ops import _require_hip

        mean, invstd, count_sum, coefs = _require_hip().batch_norm_stats_local(
            input, eps, running_mean, running_var, momentum, weight, bias,
            want_coefs,
        )
        return mean, invstd, count_sum, (coefs if want_coefs else None)

    mean, invstd = ops.batch_norm_stats(input, eps)
    count_sum = torch.full(
        (1,), float(local_count), dtype=torch.float32, device=input.device
    )
    if running_mean is not None and local_count > 0:
        with torch.no_grad():
            var = invstd.to(torch.float32).pow(-2) - eps
            unbiased = (
                var * (local_count / (local_count - 1.0))
                if local_count > 1
                else var
            )
            running_mean.mul_(1 - momentum).add_(
                mean.to(running_mean.dtype), alpha=momentum
            )
            running_var.mul_(1 - momentum).add_(
                unbiased.to(running_var.dtype), alpha=momentum
            )
    return mean, invstd, count_sum, None


class SyncBatchNormFunction(torch.autograd.Function):
    @staticmethod
    def forward(
        ctx,
        input: torch.Tensor,
        weight: Optional[torch.Tensor],
        bias: Optional[torch.Tensor],
        running_mean: Optional[torch.Tensor],
        running_var: Optional[torch.Tensor],
        eps: float,
        momentum: float,
        process_group,
        world_size: int,
    ):
        input = _contig(input)
        if weight is not None:
            weight = weight.contiguous()

        C = int(input.shape[1])
        local_count = input.numel() // C if C > 0 else 0
        use_sync = world_size > 1 or (
            process_group is not None and _force_sync()
        )

        if (
            not use_sync
            and local_count > 0
            and ops.bn_fused_local_eligible(
                input, weight, bias, running_mean, running_var
            )
        ):
            # single-launch small-plane path: stats + running update +
            # normalize in ONE kernel (K10-family; GAN/small-batch regime)
            y, mean, invstd, count_sum, _coefs = ops.batch_norm_fwd_fused_local(
                input, None, weight, bias, eps, momentum,
                running_mean, running_var, False,
            )
            ctx.save_for_backward(input, weight, mean, invstd, count_sum)
            ctx.process_group = process_group
            ctx.world_size = world_size
            ctx.use_sync = False
            ctx.local_fused = True
            return y

        mean, invstd, count_sum, coefs = compute_sync_stats(
            input, eps, momentum, running_mean, running_var,
            process_group, world_size, weight, bias, want_coefs=True,
        )

        ctx.save_for_backward(input, weight, mean, invstd, count_sum)
        ctx.process_group = process_group
        ctx.world_size = world_size
        ctx.use_sync = use_sync
        ctx.local_fused = False

        if local_count == 0:
            return torch.empty_like(input)
        return ops.batch_norm_elemt_act(
            input, None, weight, bias, mean, invstd, False, coefs
        )

    @staticmethod
    def backward(ctx, grad_output: torch.Tensor):
        input, weight, mean, invstd, count_sum = ctx.saved_tensors
        grad_output = _match_layout(grad_output, input)
        process_group = ctx.process_group
        world_size = ctx.world_size
        need_input_g, need_weight_g, need_bias_g = ctx.needs_input_grad[0:3]

        C = int(input.shape[1])
        local_count = input.numel() // C if C > 0 else 0
        grad_input = grad_weight = grad_bias = None

        if getattr(ctx, "local_fused", False) and local_count > 0:
            # single-launch backward (reduce + coefs + dx in one kernel)
            grad_input, grad_weight, grad_bias, _ = (
                ops.batch_norm_bwd_fused_local(
                    grad_output, input, None, mean, invstd, weight, None,
                    False, False, need_weight_g, need_bias_g,
                )
            )
            return (
                grad_input if need_input_g else None,
                grad_weight if need_weight_g else None,
                grad_bias if need_bias_g else None,
                None, None, None, None, None, None,
            )

        if local_count > 0:
            sum_dy, sum_dy_xmu, grad_weight, grad_bias = ops.batch_norm_backward_reduce(
                grad_output, input, mean, invstd, weight,
                need_input_g, need_weight_g, need_bias_g,
            )
            if need_input_g:
                if getattr(ctx, "use_sync", world_size > 1):
                    combined, copied = _combined_view(sum_dy, sum_dy_xmu, C)
                    from msbn.utils import debug as _dbg
                    if _dbg.enabled():
                        _dbg.verify_collective("syncbn.bwd.all_reduce",
                                               combined, process_group)
                    dist.all_reduce(
                        combined, dist.ReduceOp.SUM, group=process_group
                    )
                    from msbn.utils.logging import comm_log
                    comm_log.record("all_reduce", combined.numel() * 4,
                                    f"syncbn bwd C={C}")
                    if copied:
                        sum_dy, sum_dy_xmu = combined[:C], combined[C:]
                grad_input = ops.batch_norm_backward_elemt(
                    grad_output, input, mean, invstd, weight,
                    sum_dy, sum_dy_xmu, count_sum,
                )
        elif getattr(ctx, "use_sync", world_size > 1) and need_input_g:
            # Empty-input rank: still contribute zeros to unblock peers
            # (stock behavior, _functions.py:187-204).
            combined = torch.zeros(
                2 * C, dtype=torch.float32, device=grad_output.device
            )
            from msbn.utils import debug as _dbg
            if _dbg.enabled():  # peers verify; an absent call would hang them
                _dbg.verify_collective("syncbn.bwd.all_reduce", combined,
                                       process_group)
            dist.all_reduce(combined, dist.ReduceOp.SUM, group=process_group)
            grad_input = torch.empty_like(grad_output)

        return (
            grad_input,
            grad_weight if need_weight_g else None,
            grad_bias if need_bias_g else None,
            None, None, None, None, None, None,
        )
