"""DDP construction + runtime observability (stock c10d::Logger equivalent,
SURVEY.md §5.5) and the rank-0-prints convention helper (README.md:9)."""

import os
import sys
import time
from collections import deque
from typing import Optional

import torch.distributed as dist


def is_master() -> bool:
    if dist.is_available() and dist.is_initialized():
        return dist.get_rank() == 0
    return int(os.environ.get("RANK", "0")) == 0


def master_print(*args, **kwargs):
    """Print only on rank 0 (the README.md:9 convention)."""
    if is_master():
        print(*args, **kwargs)


class DDPLogger:
    """Construction data + runtime counters for a msbn DDP instance.

    Retrieved via ``model._get_ddp_logging_data()`` (dict of strs/ints, shaped
    like stock DDPLoggingData — SURVEY.md §5.5)."""

    def __init__(self, ddp):
        self._strs = {}
        self._ints = {}
        m = ddp.module
        self._strs["module_name"] = type(m).__name__
        self._strs["device"] = str(ddp._device)
        self._ints["world_size"] = (
            dist.get_world_size(ddp.process_group)
            if dist.is_initialized()
            else 1
        )
        self._ints["broadcast_buffers"] = int(ddp.broadcast_buffers)
        self._ints["find_unused_parameters"] = int(ddp.find_unused_parameters)
        self._ints["gradient_as_bucket_view"] = int(ddp.gradient_as_bucket_view)
        self._ints["bucket_cap_bytes"] = ddp.bucket_bytes_cap
        self._ints["has_sync_bn"] = int(
            any(type(mm).__name__ == "SyncBatchNorm" for mm in m.modules())
        )
        self._ints["num_parameter_tensors"] = len(ddp._params)
        self._ints["total_parameter_size_bytes"] = sum(
            p.numel() * p.element_size() for p in ddp._params
        )
        self._ints["num_buckets"] = len(ddp.reducer.get_bucket_indices())
        self._ints["forward_count"] = 0
        self._ints["rebuilt_bucket_count"] = 0
        self._fwd_times = deque(maxlen=64)
        self._last_fwd: Optional[float] = None

    def note_forward(self):
        now = time.perf_counter()
        if self._last_fwd is not None:
            self._fwd_times.append(now - self._last_fwd)
        self._last_fwd = now
        self._ints["forward_count"] += 1

    def note_rebuilt(self, bucket_indices):
        self._ints["rebuilt_bucket_count"] += 1
        self._ints["num_buckets"] = len(bucket_indices)

    def data(self):
        out = {"strs_map": dict(self._strs), "ints_map": dict(self._ints)}
        if self._fwd_times:
            out["ints_map"]["avg_iter_time_us"] = int(
                1e6 * sum(self._fwd_times) / len(self._fwd_times)
            )
        return out


class CommLog:
    """Flight-recorder-style ring buffer of recent collectives (SURVEY.md §5.1:
    the TORCH_NCCL_TRACE_BUFFER idea).  Enabled with MSBN_COMM_LOG=1."""

    def __init__(self, capacity: int = 256):
        self.buf = deque(maxlen=capacity)
        self.enabled = os.environ.get("MSBN_COMM_LOG", "0") == "1"

    def record(self, op: str, bytes_: int, note: str = ""):
        if self.enabled:
            self.buf.append((time.time(), op, bytes_, note))

    def dump(self, file=sys.stderr):
        for t, op, b, note in self.buf:
            print(f"[msbn-comm] {t:.6f} {op} {b}B {note}", file=file)


comm_log = CommLog()
