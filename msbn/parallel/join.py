"""Uneven-input training support (the stock ``DDP.join()`` capability,
SURVEY.md §5.3 "uneven-input tolerance").

Design: msbn SyncBatchNorm tolerates empty inputs (zero-count stats are
masked in-kernel and empty ranks still post their collectives), and the
reducer all-reduces zero gradients for a shadow step.  So a rank that runs
out of data keeps the collective schedule matched by stepping on an EMPTY
batch until every rank is done — same observable semantics as stock join
with divide_by_initial_world_size=True (zero contributions keep the /world
averaging).

    for batch in run_with_join(ddp_model, loader, make_empty_batch=mk):
        ...  # batch may be the empty batch on exhausted ranks

or drive explicitly with a step function:

    run_with_join(ddp_model, loader, step_fn=my_step, make_empty_batch=mk)
"""

from typing import Callable, Iterable, Optional

import torch
import torch.distributed as dist


def _any_rank_has_data(has: bool, process_group, device) -> int:
    t = torch.tensor([1 if has else 0], dtype=torch.int64, device=device)
    dist.all_reduce(t, op=dist.ReduceOp.SUM, group=process_group)
    return int(t.item())


def run_with_join(
    ddp_model,
    data_iter: Iterable,
    step_fn: Optional[Callable] = None,
    make_empty_batch: Optional[Callable] = None,
):
    """Iterate data with uneven per-rank lengths under DDP.

    Each round, ranks agree (one tiny all_reduce) whether anyone still has
    data.  Exhausted ranks step on ``make_empty_batch()`` so SyncBN stats
    syncs and gradient-bucket all-reduces stay matched across ranks.

    With ``step_fn``: calls ``step_fn(batch)`` for real and empty batches and
    returns the number of real steps this rank took.  Without ``step_fn``:
    a generator yielding batches (real or empty) to the caller's loop.
    """
    if make_empty_batch is None:
        raise ValueError("run_with_join needs make_empty_batch")
    pg = ddp_model.process_group
    # the flag all_reduce runs on gloo/CPU tensors for gloo, GPU for nccl
    backend = dist.get_backend(pg)
    device = (
        torch.device("cuda", torch.cuda.current_device())
        if backend == "nccl"
        else torch.device("cpu")
    )

    if step_fn is None:
        def gen():
            it = iter(data_iter)
            while True:
                try:
                    batch = next(it)
                    has = True
                except StopIteration:
                    batch = None
                    has = False
                if _any_rank_has_data(has, pg, device) == 0:
                    return
                yield batch if has else make_empty_batch()

        return gen()

    it = iter(data_iter)
    real_steps = 0
    while True:
        try:
            batch = next(it)
            has = True
        except StopIteration:
            batch = None
            has = False
        if _any_rank_has_data(has, pg, device) == 0:
            return real_steps
        if has:
            real_steps += 1
            step_fn(batch)
        else:
            step_fn(make_empty_batch())
