"""Gradient-communication hook support (stock ``register_comm_hook``,
SURVEY.md §2.2 comm utils: the GradBucket carrier + hook execution).

Two tiers:

* The three c10d builtin hooks (``allreduce_hook``, ``fp16_compress_hook``,
  ``bf16_compress_hook``), matched BY IDENTITY, map onto the C++ reducer's
  fast path (wire-dtype cast + in-order async all-reduce overlapped with
  backward).
* Any other callable runs as a real Python hook: the C++ reducer hands it
  ``(state, GradBucket)`` per bucket at FINALIZE time (all hooks launched in
  bucket order, futures awaited in order).  Documented caveat: Python hooks
  do not overlap with backward compute — they replace the all-reduce AND the
  world-size division, exactly like stock hooks do.
"""

from typing import List

import torch


class GradBucket:
    """Carrier the reducer passes to Python comm hooks (stock
    dist.GradBucket surface: buffer / gradients / index / is_last /
    set_buffer)."""

    __slots__ = ("_flat", "_views", "_index", "_is_last")

    def __init__(self, flat: torch.Tensor, views: List[torch.Tensor],
                 index: int, is_last: bool):
        self._flat = flat
        self._views = views
        self._index = index
        self._is_last = is_last

    def buffer(self) -> torch.Tensor:
        """The flat gradient tensor for this bucket (pre-division)."""
        return self._flat

    def set_buffer(self, tensor: torch.Tensor) -> None:
        self._flat.copy_(tensor.reshape(-1))

    def gradients(self) -> List[torch.Tensor]:
        """Per-parameter views into the flat buffer."""
        return list(self._views)

    def index(self) -> int:
        return self._index

    def is_last(self) -> bool:
        return self._is_last


def allreduce_via_pg(process_group):
    """Reference Python hook: what the default C++ path does, expressed as a
    comm hook (useful as a template for custom hooks)."""
    import torch.distributed as dist

    world = dist.get_world_size(process_group)

    def hook(state, bucket: GradBucket):
        t = bucket.buffer()
        t.div_(world)
        work = dist.all_reduce(t, group=process_group, async_op=True)
        fut = work.get_future() if hasattr(work, "get_future") else None
        if fut is not None:
            return fut.then(lambda f: f.value()[0])
        work.wait()
        ret = torch.futures.Future()
        ret.set_result(t)
        return ret

    return hook
