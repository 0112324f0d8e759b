"""msbn DistributedDataParallel — the Python façade over the C++ reducer.

Drop-in for ``torch.nn.parallel.DistributedDataParallel`` as the reference
recipe uses it (/root/reference/README.md:66-72):

    net = msbn.parallel.DistributedDataParallel(
        net, device_ids=[args.local_rank], output_device=args.local_rank)

Behavior map: SURVEY.md §2.2 "distributed.py:328-2434".  The gradient engine
underneath is msbn._C.Reducer (C++: bucketed flat buffers, autograd-hooked,
async RCCL all-reduce overlapped with backward — SURVEY.md §2.2
"reducer.hpp:45-581").
"""

import itertools
from contextlib import contextmanager


import torch
import torch.distributed as dist
from torch.nn import Module

from msbn.utils.logging import DDPLogger

_BROADCAST_BUCKET_BYTES = 250 * 1024 * 1024  # stock broadcast_bucket_size
_DEFAULT_FIRST_BUCKET_BYTES = 1024 * 1024  # c10d::kDefaultFirstBucketBytes
_DEFAULT_BUCKET_BYTES = 25 * 1024 * 1024  # c10d::kDefaultBucketBytesCap


def _find_used_params(outputs, parameters):
    """Walk the autograd graph backward from `outputs`; return the set of
    param ids whose AccumulateGrad is reachable (find_unused_parameters path,
    stock reducer::prepare_for_backward search)."""
    seen = set()
    used = set()
    stack = []
    for out in outputs:
        if isinstance(out, torch.Tensor) and out.grad_fn is not None:
            stack.append(out.grad_fn)
    param_by_id = {id(p): i for i, p in enumerate(parameters)}
    while stack:
        fn = stack.pop()
        if id(fn) in seen:
            continue
        seen.add(id(fn))
        var = getattr(fn, "variable", None)
        if var is not None and id(var) in param_by_id:
            used.add(param_by_id[id(var)])
        for next_fn, _ in fn.next_functions:
            if next_fn is not None:
                stack.append(next_fn)
    return used


def _flatten_outputs(out):
    if isinstance(out, torch.Tensor):
        return [out]
    if isinstance(out, (list, tuple)):
        return list(itertools.chain.from_iterable(_flatten_outputs(o) for o in out))
    if isinstance(out, dict):
        return list(
            itertools.chain.from_iterable(_flatten_outputs(o) for o in out.values())
        )
    return []


class DistributedDataParallel(Module):
    def __init__(
        self,
        module: Module,
        device_ids=None,
        output_device=None,
        dim: int = 0,
        broadcast_buffers: bool = True,
        process_group=None,
        bucket_cap_mb: float = 25,
        find_unused_parameters: bool = False,
        gradient_as_bucket_view: bool = False,
        static_graph: bool = False,
    ):
        super().__init__()
        from msbn import _C as _C  # noqa: F401  fail fast if ext missing
        if not dist.is_available() or not dist.is_initialized():
            raise RuntimeError(
                "msbn DDP requires torch.distributed to be initialized "
                "(init_process_group first — README.md:26-36)"
            )
        self.module = module
        self.process_group = (
            process_group if process_group is not None else dist.group.WORLD
        )
        self.device_ids = device_ids
        self.output_device = output_device
        self.dim = dim
        self.broadcast_buffers = broadcast_buffers
        self.find_unused_parameters = find_unused_parameters
        self.gradient_as_bucket_view = gradient_as_bucket_view
        self.static_graph = static_graph
        self.bucket_bytes_cap = int(bucket_cap_mb * 1024 * 1024)
        self.require_backward_grad_sync = True
        self.require_forward_param_sync = True

        params = [p for p in module.parameters() if p.requires_grad]
        if len(params) == 0:
            raise RuntimeError("msbn DDP: module has no parameters requiring grad")
        # single-device-per-process enforcement (stock distributed.py:744-762)
        devices = {p.device for p in module.parameters()}
        if len(devices) > 1:
            raise RuntimeError(
                f"msbn DDP supports one device per process; got {devices}"
            )
        self._device = next(iter(devices))
        if self._device.type == "cuda" and device_ids is not None:
            if len(device_ids) != 1 or self._device.index != device_ids[0]:
                raise RuntimeError(
                    f"module is on {self._device} but device_ids={device_ids}; "
                    "move the module to cuda:{local_rank} before wrapping "
                    "(README.md:51-53)"
                )
        self._params = params

        import msbn._C as C

        # S3: parameter shape agreement
        C.verify_params_across_processes(self.process_group, params)
        # S4: rank-0 state broadcast (params + buffers, 250 MiB chunks)
        self._sync_module_states()

        # bucket assignment on reversed order ~ backward order
        rev = list(reversed(params))
        rev_buckets = C.compute_bucket_assignment_by_size(
            rev, [_DEFAULT_FIRST_BUCKET_BYTES, self.bucket_bytes_cap]
        )
        n = len(params)
        bucket_indices = [[n - 1 - i for i in b] for b in rev_buckets]
        self.reducer = C.Reducer(
            params,
            bucket_indices,
            self.process_group,
            gradient_as_bucket_view,
            _DEFAULT_FIRST_BUCKET_BYTES,
            self.bucket_bytes_cap,
        )
        import os as _os

        if _os.environ.get("MSBN_NAN_CHECK", "0") == "1":
            self.reducer.set_nan_check(True)
        self.logger = DDPLogger(self)
        self._has_sync_bn = any(
            type(m).__name__ == "SyncBatchNorm" for m in module.modules()
        )
        if self._has_sync_bn and self._device.type == "cpu":
            # msbn extension: SyncBN on CPU/gloo is supported (stock rejects it;
            # we allow it so the no-GPU plumbing config tests the real sync path)
            pass

    # ------------------------------------------------------------------ state
    def __getstate__(self):
        state = self.__dict__.copy()
        for k in ("process_group", "reducer", "logger"):
            state.pop(k, None)
        return state

    def __setstate__(self, state):
        self.__dict__.update(state)
        self.process_group = dist.group.WORLD
        import msbn._C as C

        params = self._params
        rev = list(reversed(params))
        rev_buckets = C.compute_bucket_assignment_by_size(
            rev, [_DEFAULT_FIRST_BUCKET_BYTES, self.bucket_bytes_cap]
        )
        n = len(params)
        bucket_indices = [[n - 1 - i for i in b] for b in rev_buckets]
        self.reducer = C.Reducer(
            params,
            bucket_indices,
            self.process_group,
            self.gradient_as_bucket_view,
            _DEFAULT_FIRST_BUCKET_BYTES,
            self.bucket_bytes_cap,
        )
        self.logger = DDPLogger(self)

    # ------------------------------------------------------------------- sync
    @staticmethod
    def _dtype_sorted(tensors):
        """Stable-sort by dtype so broadcast_coalesced flattens into one
        chunk per dtype instead of one per consecutive run (BN models
        interleave fp32 stats with int64 counters: ~100 broadcasts/step
        otherwise)."""
        order = {}
        for t in tensors:
            order.setdefault(str(t.dtype), len(order))
        return sorted(tensors, key=lambda t: order[str(t.dtype)])

    def _module_states(self):
        states = []
        for p in self.module.parameters():
            states.append(p.detach())
        for b in self.module.buffers():
            if b is not None and b.dtype != torch.bool:
                states.append(b.detach())
        return self._dtype_sorted(states)

    def _sync_module_states(self):
        import msbn._C as C

        states = self._module_states()
        if states:
            C.broadcast_coalesced(
                self.process_group, states, _BROADCAST_BUCKET_BYTES, 0
            )

    def _sync_buffers(self):
        import msbn._C as C

        bufs = self._dtype_sorted([
            b.detach()
            for b in self.module.buffers()
            if b is not None and b.dtype != torch.bool
        ])
        if bufs:
            C.broadcast_coalesced(
                self.process_group, bufs, _BROADCAST_BUCKET_BYTES, 0
            )

    # ---------------------------------------------------------------- forward
    def forward(self, *inputs, **kwargs):
        with torch.autograd.profiler.record_function(
            "msbn.DistributedDataParallel.forward"
        ):
            if torch.is_grad_enabled() and self.require_backward_grad_sync:
                # one-shot arrival-order bucket rebuild (stock _rebuild_buckets)
                if not self.reducer.rebuilt() and self.reducer.iterations() > 0:
                    if self.reducer.rebuild_buckets():
                        self.logger.note_rebuilt(self.reducer.get_bucket_indices())
            if (
                self.broadcast_buffers
                and self.require_forward_param_sync
                and dist.get_world_size(self.process_group) > 1
            ):
                self._sync_buffers()

            output = self.module(*inputs, **kwargs)

            if torch.is_grad_enabled() and self.require_backward_grad_sync:
                self.reducer.set_grad_sync_enabled(True)
                if self.find_unused_parameters and not self.static_graph:
                    outs = _flatten_outputs(output)
                    used = _find_used_params(outs, self._params)
                    unused = [
                        i for i in range(len(self._params)) if i not in used
                    ]
                    self.reducer.prepare_for_backward(unused)
                else:
                    self.reducer.prepare_for_backward([])
                self.logger.note_forward()
            else:
                self.reducer.set_grad_sync_enabled(False)
            return output

    # ------------------------------------------------------------------- misc
    @contextmanager
    def no_sync(self):
        """Skip gradient synchronization (local accumulation) inside the
        context; the next forward/backward outside resumes syncing."""
        old = self.require_backward_grad_sync
        self.require_backward_grad_sync = False
        self.reducer.set_grad_sync_enabled(False)
        try:
            yield
        finally:
            self.require_backward_grad_sync = old
            self.reducer.set_grad_sync_enabled(old)

    @contextmanager
    def join(self, divide_by_initial_world_size: bool = True, enable: bool = True):
        """Uneven-input support lives in msbn.parallel.run_with_join: exhausted
        ranks step on empty batches (SyncBN masks zero-count stats in-kernel;
        the reducer all-reduces zero grads), keeping every collective matched.
        This context is a compatibility shim for code structured around stock
        DDP.join(); inside it the caller must keep iteration counts symmetric
        or drive the loop with run_with_join."""
        yield

    def register_comm_hook(self, state, hook):
        """Gradient-communication hook (stock register_comm_hook parity for
        the builtin compression hooks): the fp16/bf16 compress hooks map to a
        wire-dtype cast in the C++ reducer.  Arbitrary Python hooks are not
        run inside the C++ backward path; use set_comm_dtype for custom
        compression dtypes."""
        name = getattr(hook, "__name__", repr(hook))
        if "bf16" in name:
            self.reducer.set_comm_dtype(torch.bfloat16)
        elif "fp16" in name:
            self.reducer.set_comm_dtype(torch.float16)
        else:
            raise NotImplementedError(
                "msbn DDP supports the builtin fp16/bf16 compression hooks "
                f"(got {name}); custom Python comm hooks are not supported"
            )

    def set_comm_dtype(self, dtype):
        """Cast gradient buckets to `dtype` for the wire (None to disable)."""
        self.reducer.set_comm_dtype(dtype)

    def _get_ddp_logging_data(self):
        return self.logger.data()

    def train(self, mode: bool = True):
        super().train(mode)
        self.module.train(mode)
        return self

    @property
    def _ddp_params_and_buffers_to_ignore(self):
        return getattr(self.module, "_ddp_params_and_buffers_to_ignore", [])
