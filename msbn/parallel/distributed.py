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
from torch.utils._pytree import tree_map

from msbn.utils.logging import DDPLogger

_BROADCAST_BUCKET_BYTES = 250 * 1024 * 1024  # stock broadcast_bucket_size
_DEFAULT_FIRST_BUCKET_BYTES = 1024 * 1024  # c10d::kDefaultFirstBucketBytes
_DEFAULT_BUCKET_BYTES = 25 * 1024 * 1024  # c10d::kDefaultBucketBytesCap


def _find_used_params(outputs, parameters):
    """Walk the autograd graph backward from `outputs`; return the set of
    param ids whose AccumulateGrad is reachable.  Pure-Python reference for
    Reducer::find_unused (the C++ walk DDP actually uses); kept as the test
    oracle.  NOTE: `seen` must PIN the visited grad_fn wrappers (dict, not an
    id set) — the wrappers are transient Python objects and freed ids get
    recycled, silently truncating the walk."""
    seen = {}
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
        seen[id(fn)] = fn
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
        # join() state (uneven-input support, stock distributed.py:1765+)
        self._join_config = None
        self._join_input_spec = None
        # static_graph: the unused-parameter set is computed once (iteration 1)
        # and reused — the per-step graph walk is skipped (stock static_graph
        # one-shot assumption, distributed.py:79)
        self._static_unused = None

    # ------------------------------------------------------------------ state
    def __getstate__(self):
        state = self.__dict__.copy()
        for k in ("process_group", "reducer", "logger", "_h2d_stream"):
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

    @staticmethod
    def _commable(b):
        """Buffer as a broadcast-able tensor; bool buffers go over the wire as
        uint8 views of the same storage (gloo/RCCL have no bool type; stock
        syncs every buffer, so skipping bool would silently diverge)."""
        return b.detach().view(torch.uint8) if b.dtype == torch.bool else b.detach()

    def _module_states(self):
        states = []
        for p in self.module.parameters():
            states.append(p.detach())
        for b in self.module.buffers():
            if b is not None:
                states.append(self._commable(b))
        return self._dtype_sorted(states)

    def _sync_module_states(self, src: int = 0):
        import msbn._C as C

        states = self._module_states()
        if states:
            C.broadcast_coalesced(
                self.process_group, states, _BROADCAST_BUCKET_BYTES, src
            )

    def _sync_buffers(self):
        import msbn._C as C

        bufs = self._dtype_sorted([
            self._commable(b)
            for b in self.module.buffers()
            if b is not None
        ])
        if bufs:
            C.broadcast_coalesced(
                self.process_group, bufs, _BROADCAST_BUCKET_BYTES, 0
            )

    # ---------------------------------------------------------------- forward
    def _compute_unused(self, output):
        """Unused-parameter indices for this iteration's graph.  The walk runs
        in C++ (Reducer::find_unused — the stock reducer's prepare_for_backward
        search); with static_graph the result from iteration 1 is cached and
        the per-step walk is skipped entirely."""
        if self.static_graph and self._static_unused is not None:
            return self._static_unused
        outs = [o for o in _flatten_outputs(output) if o.requires_grad]
        unused = self.reducer.find_unused(outs)
        if self.static_graph:
            self._static_unused = unused
        return unused

    def _notify_join_context(self):
        """First collective of a joined-training iteration (stock
        Join.notify_join_context + _check_global_requires_backward_grad_sync,
        S11): ONE 2-element all_reduce carrying [I-have-data,
        I-will-sync-grads].  Shadowing ranks contribute [0, 0]; trainers
        under no_sync() contribute [1, 0] so shadows skip their bucket
        all-reduces too.  Mixed sync settings among trainers raise on every
        rank.  Returns the active-trainer count, or None outside join()."""
        cfg = self._join_config
        if cfg is None or cfg.get("shadowing"):
            return None
        t = torch.tensor(
            [1.0, 1.0 if self.require_backward_grad_sync else 0.0],
            dtype=torch.float32, device=self._comm_device(),
        )
        dist.all_reduce(t, op=dist.ReduceOp.SUM, group=self.process_group)
        active = int(t[0].item())
        wants_sync = int(t[1].item())
        if 0 < wants_sync < active:
            raise RuntimeError(
                "msbn DDP.join(): require_backward_grad_sync differs across "
                f"active ranks ({wants_sync}/{active} syncing) — no_sync() "
                "must be entered by every rank on the same iterations"
            )
        world = dist.get_world_size(self.process_group)
        if cfg["throw_on_early_termination"] and active < world:
            raise RuntimeError(
                "Detected at least one rank that exhausted inputs. "
                "Throwing across all ranks (join(throw_on_early_termination=True))."
            )
        if not cfg["divide_by_initial_world_size"]:
            self.reducer.set_div_factor(float(max(active, 1)))
        cfg["real_iters"] = cfg.get("real_iters", 0) + 1
        return active

    def _comm_device(self):
        if self._device.type == "cuda":
            return self._device
        return torch.device("cpu")

    def _move_inputs_to_device(self, inputs, kwargs):
        """Move CPU tensor inputs to the module's GPU on a side stream
        overlapped with the buffer sync (stock _pre_forward's input H2D move,
        distributed.py:1564-1571).  No-op when the module is on CPU or every
        tensor is already on-device (the common bench/serving case)."""
        if self._device.type != "cuda":
            return inputs, kwargs
        needs_move = any(
            isinstance(t, torch.Tensor) and t.device != self._device
            for t in itertools.chain(
                _flatten_outputs(list(inputs)), _flatten_outputs(kwargs)
            )
        )
        if not needs_move:
            return inputs, kwargs
        if getattr(self, "_h2d_stream", None) is None:
            self._h2d_stream = torch.cuda.Stream(device=self._device)
        stream = self._h2d_stream
        current = torch.cuda.current_stream(self._device)
        moved = []
        with torch.cuda.stream(stream):
            def mv(t):
                if isinstance(t, torch.Tensor) and t.device != self._device:
                    out = t.to(self._device, non_blocking=True)
                    moved.append(out)
                    return out
                return t

            inputs = tree_map(mv, inputs)
            kwargs = tree_map(mv, kwargs)
        current.wait_stream(stream)
        for t in moved:
            t.record_stream(current)
        return inputs, kwargs

    def _record_join_input_spec(self, inputs, kwargs):
        """Remember the structure of real inputs so joined (exhausted) ranks
        can fabricate batch-size-0 shadows of them: every tensor leaf is
        replaced by an empty tensor with batch dimension 0 (non-tensor leaves
        are kept by reference)."""
        def to_empty(x):
            if isinstance(x, torch.Tensor):
                shape = ((0,) + tuple(x.shape[1:])) if x.dim() >= 1 else ()
                return torch.empty(shape, dtype=x.dtype, device=x.device)
            return x

        self._join_input_spec = (
            tree_map(to_empty, inputs), tree_map(to_empty, kwargs)
        )

    def forward(self, *inputs, **kwargs):
        with torch.autograd.profiler.record_function(
            "msbn.DistributedDataParallel.forward"
        ):
            if torch.is_grad_enabled() and self._join_config is not None:
                # notify runs even under no_sync: the agreement flag tells
                # shadowing ranks whether bucket all-reduces happen this
                # iteration (stock S11 semantics)
                self._notify_join_context()
                if not self._join_config.get("shadowing"):
                    self._record_join_input_spec(inputs, kwargs)
            if torch.is_grad_enabled() and self.require_backward_grad_sync:
                # one-shot arrival-order bucket rebuild (stock _rebuild_buckets).
                # Gated off under find_unused without static_graph: per-rank
                # arrival completeness can differ across ranks there, and a
                # rank entering the rebuild broadcast while another skips it
                # deadlocks (stock behavior).
                if not self.find_unused_parameters or self.static_graph:
                    if not self.reducer.rebuilt() and self.reducer.iterations() > 0:
                        if self.reducer.rebuild_buckets():
                            self.logger.note_rebuilt(
                                self.reducer.get_bucket_indices()
                            )
            if (
                self.broadcast_buffers
                and self.require_forward_param_sync
                and dist.get_world_size(self.process_group) > 1
            ):
                self._sync_buffers()

            inputs, kwargs = self._move_inputs_to_device(inputs, kwargs)
            output = self.module(*inputs, **kwargs)

            if torch.is_grad_enabled() and self.require_backward_grad_sync:
                self.reducer.set_grad_sync_enabled(True)
                if self.find_unused_parameters or self.static_graph:
                    self.reducer.prepare_for_backward(
                        self._compute_unused(output)
                    )
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
    def join(
        self,
        divide_by_initial_world_size: bool = True,
        enable: bool = True,
        throw_on_early_termination: bool = False,
    ):
        """Train with uneven per-rank input counts (stock ``DDP.join()``,
        distributed.py:1765+).

        Mechanism (msbn-native): each iteration inside the context opens with
        a one-scalar all_reduce counting ranks that still have data (S11).
        A rank that exits its training loop blocks in the context exit and,
        while peers are still training, *shadow-steps the real model on
        batch-size-0 inputs* (fabricated from the recorded shapes of its last
        real batch).  SyncBN contributes zero-count stats (masked in-kernel),
        the reducer all-reduces zero gradient buckets, and the buffer
        broadcast runs as usual — every collective stays matched across ranks
        with no bespoke shadow-collective bookkeeping.

        ``divide_by_initial_world_size=False`` divides gradients by the
        per-iteration count of active ranks instead of the world size.
        ``throw_on_early_termination=True`` raises on ALL ranks as soon as
        one rank runs out of data (for re-sharding loops).
        """
        world = dist.get_world_size(self.process_group)
        if not enable or world == 1:
            yield
            return
        self._join_config = {
            "divide_by_initial_world_size": divide_by_initial_world_size,
            "throw_on_early_termination": throw_on_early_termination,
            "shadowing": False,
        }
        try:
            yield
            # this rank is out of data: shadow-step until everyone is done
            self._join_shadow_until_all_done()
        finally:
            self._join_config = None
            self._join_input_spec = None
            self.reducer.set_div_factor(float(world))

    def _join_shadow_until_all_done(self):
        cfg = self._join_config
        device = self._comm_device()
        while True:
            t = torch.zeros(2, dtype=torch.float32, device=device)
            dist.all_reduce(t, op=dist.ReduceOp.SUM, group=self.process_group)
            active = int(t[0].item())
            wants_sync = int(t[1].item())
            if active == 0:
                # All ranks are done.  Joined ranks never ran the optimizer
                # during their shadow steps, so their parameters are stale by
                # however many iterations they shadowed; broadcast the final
                # model from the authoritative rank — the one that processed
                # the most real batches (stock Join post-hook semantics).
                world = dist.get_world_size(self.process_group)
                counts = torch.zeros(world, dtype=torch.float32, device=device)
                counts[dist.get_rank(self.process_group)] = float(
                    cfg.get("real_iters", 0)
                )
                dist.all_reduce(
                    counts, op=dist.ReduceOp.SUM, group=self.process_group
                )
                auth = int(torch.argmax(counts).item())
                self._sync_module_states(src=auth)
                return
            if cfg["throw_on_early_termination"]:
                raise RuntimeError(
                    "Detected at least one rank that exhausted inputs. "
                    "Throwing across all ranks "
                    "(join(throw_on_early_termination=True))."
                )
            if self._join_input_spec is None:
                raise RuntimeError(
                    "msbn DDP.join(): this rank never ran a real forward "
                    "inside the join() context, so it cannot fabricate an "
                    "empty shadow batch. Feed at least one batch per rank or "
                    "use msbn.parallel.run_with_join."
                )
            if not cfg["divide_by_initial_world_size"]:
                self.reducer.set_div_factor(float(max(active, 1)))
            # Stale .grad would ACCUMULATE under the 0-batch backward and be
            # copied into the bucket — the shadow contribution must be zeros.
            for p in self._params:
                p.grad = None
            inputs, kwargs = self._join_input_spec
            cfg["shadowing"] = True
            # mirror the trainers' no_sync state: when they skip gradient
            # sync this iteration (wants_sync == 0), the shadow must not
            # post bucket all-reduces or enter the rebuild broadcast either
            old_sync = self.require_backward_grad_sync
            self.require_backward_grad_sync = wants_sync > 0
            try:
                output = self.forward(*inputs, **kwargs)
                outs = [
                    o for o in _flatten_outputs(output) if o.requires_grad
                ]
                if outs:
                    torch.autograd.backward([o.sum() for o in outs])
            finally:
                cfg["shadowing"] = False
                self.require_backward_grad_sync = old_sync

    def register_comm_hook(self, state, hook):
        """Gradient-communication hook (stock register_comm_hook).

        The three builtin c10d hooks are recognized BY FUNCTION IDENTITY
        (not name sniffing) and map onto the C++ reducer's fast path:
        allreduce_hook is the default, fp16/bf16 compress hooks become a
        wire-dtype cast — full backward/comm overlap is preserved.

        Any OTHER callable runs as a real Python hook: the reducer calls
        ``hook(state, GradBucket)`` per bucket at finalize (launched in
        bucket order, futures awaited in order).  The hook replaces the
        all-reduce AND the world-size division, matching stock semantics.
        Caveat (documented): Python hooks run after backward compute — no
        overlap; prefer the builtins for production."""
        try:
            from torch.distributed.algorithms.ddp_comm_hooks import (
                default_hooks as _dh,
            )
        except Exception:  # pragma: no cover - torch always ships these
            _dh = None
        if _dh is not None:
            if hook is _dh.allreduce_hook:
                self.reducer.set_comm_dtype(None)
                self.reducer.set_python_comm_hook(None, None, None)
                return
            if hook is _dh.fp16_compress_hook:
                self.reducer.set_comm_dtype(torch.float16)
                self.reducer.set_python_comm_hook(None, None, None)
                return
            if hook is _dh.bf16_compress_hook:
                self.reducer.set_comm_dtype(torch.bfloat16)
                self.reducer.set_python_comm_hook(None, None, None)
                return
        if not callable(hook):
            raise TypeError(f"comm hook must be callable, got {hook!r}")
        from msbn.parallel.comm_hooks import GradBucket

        self.reducer.set_python_comm_hook(state, hook, GradBucket)

    def set_comm_dtype(self, dtype):
        """Cast gradient buckets to `dtype` for the wire (None to disable)."""
        self.reducer.set_comm_dtype(dtype)

    def _get_ddp_logging_data(self):
        data = self.logger.data()
        # per-param grad-ready times (us since first hook) from the last
        # backward — the stock reducer's backward_stats_
        stats = self.reducer.get_backward_stats()
        if any(s >= 0 for s in stats):
            data["ints_map"]["last_backward_span_us"] = int(max(stats))
        data["backward_grad_ready_us"] = stats
        return data

    def train(self, mode: bool = True):
        super().train(mode)
        self.module.train(mode)
        return self

    @property
    def _ddp_params_and_buffers_to_ignore(self):
        return getattr(self.module, "_ddp_params_and_buffers_to_ignore", [])
