"""Collective-agreement verification (stock ProcessGroupWrapper under
TORCH_DISTRIBUTED_DEBUG=DETAIL — SURVEY.md §5.2).

With ``MSBN_DEBUG_COLLECTIVES=1`` (or ``TORCH_DISTRIBUTED_DEBUG=DETAIL``),
every msbn-issued collective first verifies on a SIDE gloo group that all
ranks are about to post the same operation with the same message shape and
dtype.  A desync (different BN layer order across ranks, mismatched channel
counts, a rank skipping a collective) then raises a clean descriptive error
on every rank instead of hanging or silently corrupting data.

Debug-only: costs one extra gloo all_gather per collective.
"""

import hashlib
import os
from typing import Optional

import torch
import torch.distributed as dist

_side_group = None


def enabled() -> bool:
    return (
        os.environ.get("MSBN_DEBUG_COLLECTIVES", "0") == "1"
        or os.environ.get("TORCH_DISTRIBUTED_DEBUG", "").upper() == "DETAIL"
    )


def _side(process_group) -> Optional[object]:
    """Lazy gloo side group.  All ranks reach their first verified
    collective together, so the collective new_group call is symmetric."""
    global _side_group
    if _side_group is None:
        _side_group = dist.new_group(backend="gloo")
    return _side_group


def verify_collective(op: str, tensor: torch.Tensor, process_group) -> None:
    """Raise (on every rank) if any rank disagrees on (op, numel, dtype)."""
    if not enabled() or not dist.is_initialized():
        return
    group = _side(process_group)
    world = dist.get_world_size(group)
    if world <= 1:
        return
    tag = int.from_bytes(
        hashlib.sha1(op.encode()).digest()[:6], "big"
    )
    dt = str(tensor.dtype)
    dtag = int.from_bytes(hashlib.sha1(dt.encode()).digest()[:6], "big")
    mine = torch.tensor([tag, tensor.numel(), dtag], dtype=torch.int64)
    allv = [torch.zeros_like(mine) for _ in range(world)]
    dist.all_gather(allv, mine, group=group)
    for r, v in enumerate(allv):
        if not torch.equal(v, mine):
            raise RuntimeError(
                f"msbn collective-agreement check failed: this rank is about "
                f"to post {op} (numel={tensor.numel()}, dtype={dt}) but rank "
                f"{r} posted a different collective "
                f"(tag/numel/dtype codes {v.tolist()} vs {mine.tolist()}). "
                "Ranks have desynchronized — check for per-rank model "
                "differences or conditional collectives."
            )
