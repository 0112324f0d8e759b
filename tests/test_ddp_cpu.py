"""Multi-process (gloo, world_size=2) integration tests on CPU.

Pattern follows the stock test strategy: golden-model
equivalence — a single-process model trained on the GLOBAL batch must match
DDP+SyncBN ranks trained on per-rank SLICES, parameters compared after several
optimizer steps (_test_DistributedDataParallel_SyncBatchNorm ≈
distributed_test.py:5575-5634).
"""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

WORLD = 2


def _worker(rank, fn_name, tmpdir, q):
    try:
        dist.init_process_group(
            "gloo",
            init_method=f"file://{tmpdir}/pg",
            rank=rank,
            world_size=WORLD,
        )
        fn = globals()[fn_name]
        fn(rank)
        q.put((rank, None))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, traceback.format_exc()))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def _spawn(fn_name, tmp_path):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    ps = [
        ctx.Process(target=_worker, args=(r, fn_name, str(tmp_path), q))
        for r in range(WORLD)
    ]
    for p in ps:
        p.start()
    errs = []
    for _ in range(WORLD):
        rank, err = q.get()
        if err:
            errs.append((rank, err))
    for p in ps:
        p.join(timeout=60)
        if p.is_alive():
            p.terminate()
            errs.append((p.pid, "timeout"))
    assert not errs, "\n".join(f"rank {r}:\n{e}" for r, e in errs)


# --------------------------------------------------------------------------
def _golden_body(rank):
    import msbn

    torch.manual_seed(42)  # SAME init everywhere
    global_bs = 8
    local_bs = global_bs // WORLD

    ddp_net = msbn.convert_sync_batchnorm(msbn.models.SimpleCNN(width=8))
    ddp_net = msbn.parallel.DistributedDataParallel(ddp_net)
    opt = torch.optim.SGD(ddp_net.parameters(), lr=0.05, momentum=0.9)

    torch.manual_seed(42)
    gold = msbn.models.SimpleCNN(width=8)  # plain BN on the global batch
    gold_opt = torch.optim.SGD(gold.parameters(), lr=0.05, momentum=0.9)

    loss_fn = torch.nn.CrossEntropyLoss()
    for it in range(5):
        g = torch.Generator().manual_seed(100 + it)
        x = torch.randn(global_bs, 3, 16, 16, generator=g)
        y = torch.randint(0, 10, (global_bs,), generator=g)
        xs = x[rank * local_bs : (rank + 1) * local_bs]
        ys = y[rank * local_bs : (rank + 1) * local_bs]

        opt.zero_grad(set_to_none=True)
        loss = loss_fn(ddp_net(xs), ys)
        loss.backward()
        opt.step()

        gold_opt.zero_grad(set_to_none=True)
        loss_fn(gold(x), y).backward()
        gold_opt.step()

    for (n1, p1), (n2, p2) in zip(
        ddp_net.module.named_parameters(), gold.named_parameters()
    ):
        assert torch.allclose(p1, p2, atol=1e-5), f"{n1} diverged from golden"
    # running stats must equal global-batch stats too
    for (n1, b1), (n2, b2) in zip(
        ddp_net.module.named_buffers(), gold.named_buffers()
    ):
        assert torch.allclose(b1.float(), b2.float(), atol=1e-5), n1


def test_golden_model_equivalence(tmp_path):
    _spawn("_golden_body", tmp_path)


# --------------------------------------------------------------------------
def _uneven_running_stats_body(rank):
    """Different per-rank batch sizes: running stats must equal the analytic
    stats of the concatenated global data (≈ distributed_test.py:6065-6102)."""
    import msbn
    from msbn.nn import SyncBatchNorm

    bn = SyncBatchNorm(4, momentum=None)  # cumulative average
    bn.train()
    n_local = 2 if rank == 0 else 6
    steps = 20
    datas = []
    for it in range(steps):
        full = []
        for r in range(WORLD):
            g = torch.Generator().manual_seed(1000 + it * WORLD + r)
            full.append(torch.randn(2 if r == 0 else 6, 4, 3, 3, generator=g))
        datas.append(full)
        bn(full[rank])
    allx = torch.cat([torch.cat(f, dim=0) for f in datas], dim=0).double()
    ref_mean = allx.mean(dim=(0, 2, 3))
    # cumulative running var averages per-step unbiased global vars
    per_step_vars = [
        torch.cat(f, dim=0).double().var(dim=(0, 2, 3), unbiased=True)
        for f in datas
    ]
    ref_var = torch.stack(per_step_vars).mean(0)
    assert torch.allclose(bn.running_mean.double(), ref_mean, atol=1e-3)
    assert torch.allclose(bn.running_var.double(), ref_var, atol=2e-2)


def test_uneven_input_sizes_running_stats(tmp_path):
    _spawn("_uneven_running_stats_body", tmp_path)


# --------------------------------------------------------------------------
def _uneven_grad_body(rank):
    """Counts-weighted backward with different per-rank batches: grads of
    SyncBN on shards == grads of plain BN on the concatenation
    (≈ distributed_test.py:6109-6126)."""
    import msbn
    from msbn.nn import SyncBatchNorm

    torch.manual_seed(5)
    bs = rank + 2
    xs = [torch.randn(r + 2, 3, 4, 4, dtype=torch.float64).float()
          for r in range(WORLD)]
    x_local = xs[rank].clone().requires_grad_(True)

    bn = SyncBatchNorm(3)
    y = bn(x_local)
    y.sum().backward()

    # oracle: plain BN over the concatenation
    x_all = torch.cat(xs, dim=0).clone().requires_grad_(True)
    tbn = torch.nn.BatchNorm2d(3)
    with torch.no_grad():
        tbn.weight.copy_(bn.weight)
        tbn.bias.copy_(bn.bias)
    ty = tbn(x_all)
    ty.sum().backward()
    off = sum(r + 2 for r in range(rank))
    ref_slice = x_all.grad[off : off + bs]
    assert torch.allclose(x_local.grad, ref_slice, atol=1e-4)


def test_uneven_input_sizes_gradient(tmp_path):
    _spawn("_uneven_grad_body", tmp_path)


# --------------------------------------------------------------------------
def _no_sync_body(rank):
    import msbn

    torch.manual_seed(0)
    net = msbn.parallel.DistributedDataParallel(torch.nn.Linear(4, 2))
    x = torch.full((2, 4), float(rank + 1))
    with net.no_sync():
        net(x).sum().backward()  # local grads only, accumulated
    net(x).sum().backward()  # synced step: grads averaged INCL. accumulation
    # grad of w wrt sum over batch = sum of x rows; two backwards accumulate
    local = 2 * x.sum(0)  # per-rank accumulated (2 backwards)
    expect = (2 * torch.full((4,), 1.0 * 2) + 2 * torch.full((4,), 2.0 * 2)) / 2
    assert torch.allclose(net.module.weight.grad[0], expect), (
        net.module.weight.grad[0],
        expect,
        local,
    )


def test_no_sync_accumulation(tmp_path):
    _spawn("_no_sync_body", tmp_path)


# --------------------------------------------------------------------------
def _empty_input_body(rank):
    """Rank 1 feeds an empty batch; SyncBN must not deadlock and stats must
    come from rank 0 only (stock zero-count semantics, _functions.py:50-57)."""
    import msbn
    from msbn.nn import SyncBatchNorm

    bn = SyncBatchNorm(3)
    bn.train()
    if rank == 0:
        x = torch.randn(4, 3, 2, 2, requires_grad=True)
    else:
        x = torch.randn(0, 3, 2, 2, requires_grad=True)
    y = bn(x)
    y.sum().backward()
    if rank == 0:
        ref = torch.nn.functional.batch_norm(
            x.detach(), None, None, bn.weight.detach(), bn.bias.detach(),
            training=True, eps=bn.eps,
        )
        assert torch.allclose(y, ref, atol=1e-5)
    else:
        assert y.shape == x.shape


def test_empty_input_rank(tmp_path):
    _spawn("_empty_input_body", tmp_path)


# --------------------------------------------------------------------------
def _checkpoint_body(rank):
    import msbn

    torch.manual_seed(1)
    net = msbn.convert_sync_batchnorm(msbn.models.SimpleCNN(width=8))
    net = msbn.parallel.DistributedDataParallel(net)
    x = torch.randn(2, 3, 8, 8)
    net(x).sum().backward()
    path = os.environ.get("MSBN_TEST_CKPT", "/tmp/msbn_ckpt_test.pt")
    msbn.utils.save_checkpoint(path, net, epoch=3)
    net2 = msbn.convert_sync_batchnorm(msbn.models.SimpleCNN(width=8))
    net2 = msbn.parallel.DistributedDataParallel(net2)
    epoch = msbn.utils.load_checkpoint(path, net2)
    assert epoch == 3
    for p1, p2 in zip(net.module.parameters(), net2.module.parameters()):
        assert torch.equal(p1, p2)


def test_checkpoint_roundtrip(tmp_path):
    os.environ["MSBN_TEST_CKPT"] = str(tmp_path / "ckpt.pt")
    _spawn("_checkpoint_body", tmp_path)


# --------------------------------------------------------------------------
def _mixed_dtype_body(rank):
    """bf16 + fp32 params split into per-dtype buckets; grads still averaged."""
    import msbn

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.a = torch.nn.Linear(4, 4).to(torch.bfloat16)
            self.b = torch.nn.Linear(4, 4)

        def forward(self, x):
            return self.b(self.a(x.to(torch.bfloat16)).float())

    torch.manual_seed(0)
    net = msbn.parallel.DistributedDataParallel(M())
    x = torch.full((2, 4), float(rank + 1))
    net(x).sum().backward()
    for p in net.module.parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all()
    # grads identical across ranks (averaged)
    g = torch.cat([p.grad.float().flatten() for p in net.module.parameters()])
    g0 = g.clone()
    dist.broadcast(g0, src=0)
    assert torch.allclose(g, g0, atol=1e-2)


def test_mixed_dtype_buckets(tmp_path):
    _spawn("_mixed_dtype_body", tmp_path)


# --------------------------------------------------------------------------
def _broadcast_buffers_false_body(rank):
    """broadcast_buffers=False: per-rank buffers stay local."""
    import msbn

    torch.manual_seed(0)
    net = msbn.convert_sync_batchnorm(msbn.models.SimpleCNN(width=8))
    net = msbn.parallel.DistributedDataParallel(net, broadcast_buffers=False)
    # desync a buffer intentionally AFTER init sync
    with torch.no_grad():
        list(net.module.buffers())[0].fill_(float(rank))
    net(torch.randn(2, 3, 8, 8)).sum().backward()
    # forward must NOT have re-synced it to rank 0's value... (SyncBN updates
    # running stats from GLOBAL batch stats though, so check num_batches only)
    nbt = [b for n, b in net.module.named_buffers()
           if "num_batches_tracked" in n][0]
    assert nbt.item() == 1


def test_broadcast_buffers_false(tmp_path):
    _spawn("_broadcast_buffers_false_body", tmp_path)


# --------------------------------------------------------------------------
def _shared_param_body(rank):
    """A parameter referenced twice in the graph fires its hook once per
    accumulation; the reducer must not double-count."""
    import msbn

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.lin = torch.nn.Linear(4, 4)

        def forward(self, x):
            return self.lin(self.lin(x))  # shared use

    torch.manual_seed(0)
    net = msbn.parallel.DistributedDataParallel(M())
    opt = torch.optim.SGD(net.parameters(), lr=0.01)
    for _ in range(2):
        opt.zero_grad(set_to_none=True)
        net(torch.randn(2, 4)).sum().backward()
        opt.step()
    assert net.reducer.iterations() == 2


def test_shared_parameter(tmp_path):
    _spawn("_shared_param_body", tmp_path)


# --------------------------------------------------------------------------
def _comm_dtype_body(rank):
    """bf16 wire compression: grads still averaged (to bf16 precision)."""
    import msbn

    torch.manual_seed(0)
    net = msbn.parallel.DistributedDataParallel(torch.nn.Linear(8, 4))
    net.set_comm_dtype(torch.bfloat16)
    x = torch.full((2, 8), float(rank + 1))
    net(x).sum().backward()
    expect = (x.new_full((8,), 1.0 * 2) + x.new_full((8,), 2.0 * 2)) / 2
    assert torch.allclose(net.module.weight.grad[0], expect, atol=0.05), (
        net.module.weight.grad[0], expect)


def test_comm_dtype_compression(tmp_path):
    _spawn("_comm_dtype_body", tmp_path)


# --------------------------------------------------------------------------
def _join_uneven_body(rank):
    """run_with_join: rank 0 has 4 batches, rank 1 has 2; both must finish
    with identical (averaged) parameters and no hang."""
    import msbn
    from msbn.parallel import run_with_join

    torch.manual_seed(3)
    net = msbn.convert_sync_batchnorm(msbn.models.SimpleCNN(width=8))
    net = msbn.parallel.DistributedDataParallel(net)
    opt = torch.optim.SGD(net.parameters(), lr=0.05)

    n_batches = 4 if rank == 0 else 2
    data = [torch.randn(2, 3, 8, 8, generator=torch.Generator().manual_seed(
        rank * 100 + i)) for i in range(n_batches)]

    def step(x):
        opt.zero_grad(set_to_none=True)
        out = net(x)
        loss = out.float().pow(2).sum() * (1.0 / max(1, x.shape[0]))
        loss.backward()
        opt.step()

    real = run_with_join(
        net, data, step_fn=step,
        make_empty_batch=lambda: torch.randn(0, 3, 8, 8),
    )
    assert real == n_batches
    # parameters identical across ranks afterwards
    flat = torch.cat([p.detach().flatten() for p in net.module.parameters()])
    flat0 = flat.clone()
    dist.broadcast(flat0, src=0)
    assert torch.allclose(flat, flat0, atol=1e-6)


def test_join_uneven_inputs(tmp_path):
    _spawn("_join_uneven_body", tmp_path)


# --------------------------------------------------------------------------
def _rebuild_buckets_body(rank):
    import msbn

    torch.manual_seed(2)
    net = msbn.parallel.DistributedDataParallel(
        torch.nn.Sequential(torch.nn.Linear(8, 32), torch.nn.ReLU(),
                            torch.nn.Linear(32, 4))
    )
    opt = torch.optim.SGD(net.parameters(), lr=0.01)
    for it in range(3):
        opt.zero_grad(set_to_none=True)
        net(torch.randn(4, 8)).sum().backward()
        opt.step()
    assert net.reducer.rebuilt()
    # all ranks hold identical bucket binning after the broadcast
    import msbn._C  # noqa

    idx = net.reducer.get_bucket_indices()
    t = torch.tensor([i for b in idx for i in b], dtype=torch.long)
    t0 = t.clone()
    dist.broadcast(t0, src=0)
    assert torch.equal(t, t0)


def test_rebuild_buckets_consistent(tmp_path):
    _spawn("_rebuild_buckets_body", tmp_path)


# --------------------------------------------------------------------------
def _ddp_pickle_roundtrip_body(rank):
    """The DDP module itself survives torch.save/torch.load (stock test
    round-trip, distributed_test.py:5603-5612): reducer/process_group are
    rebuilt by __setstate__."""
    import io

    import msbn

    torch.manual_seed(0)
    net = msbn.convert_sync_batchnorm(msbn.models.SimpleCNN(width=8))
    net = msbn.parallel.DistributedDataParallel(net)
    net(torch.randn(2, 3, 8, 8)).sum().backward()
    buf = io.BytesIO()
    torch.save(net, buf)
    buf.seek(0)
    net2 = torch.load(buf, weights_only=False)
    # restored wrapper keeps training: forward+backward+reducer work
    opt = torch.optim.SGD(net2.parameters(), lr=0.01)
    opt.zero_grad(set_to_none=True)
    net2(torch.randn(2, 3, 8, 8)).sum().backward()
    opt.step()
    for p1, p2 in zip(net.module.parameters(), net2.module.parameters()):
        assert p1.shape == p2.shape


def test_ddp_pickle_roundtrip(tmp_path):
    _spawn("_ddp_pickle_roundtrip_body", tmp_path)


# --------------------------------------------------------------------------
def _single_sample_body(rank):
    """One sample per rank (the small-per-GPU-batch regime SyncBN exists
    for): global stats over 2 samples; backward finishes."""
    import msbn
    from msbn.nn import SyncBatchNorm

    bn = SyncBatchNorm(4)
    bn.train()
    x = torch.randn(1, 4, 3, 3, requires_grad=True)
    y = bn(x)
    y.sum().backward()
    assert x.grad is not None
    # biased var over the GLOBAL batch (2 samples x 9) is nonzero
    assert torch.all(bn.running_var > 0)


def test_single_sample_per_rank(tmp_path):
    _spawn("_single_sample_body", tmp_path)


# --------------------------------------------------------------------------
def _find_unused_body(rank):
    """find_unused_parameters=True: a branch that never runs still gets
    zero-contribution sync (DDP-level autograd-graph walk)."""
    import msbn

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.used = torch.nn.Linear(4, 4)
            self.unused = torch.nn.Linear(4, 4)

        def forward(self, x):
            return self.used(x)

    torch.manual_seed(0)
    net = msbn.parallel.DistributedDataParallel(
        M(), find_unused_parameters=True
    )
    opt = torch.optim.SGD(net.parameters(), lr=0.01)
    for _ in range(3):
        opt.zero_grad(set_to_none=True)
        net(torch.full((2, 4), float(rank + 1))).sum().backward()
        opt.step()
    assert net.reducer.iterations() == 3
    # used-branch grads averaged across ranks
    g = net.module.used.weight.grad[0].clone()
    g0 = g.clone()
    dist.broadcast(g0, src=0)
    assert torch.allclose(g, g0, atol=1e-6)
    # unused branch never moved
    assert net.module.unused.weight.grad is None or \
        torch.all(net.module.unused.weight.grad == 0)


def test_find_unused_parameters(tmp_path):
    _spawn("_find_unused_body", tmp_path)
