"""Multi-rank GPU tests on a single device: 2 processes share cuda:0 over the
gloo backend (gloo supports CUDA tensors), exercising the full
DDP + SyncBN GPU code path — HIP kernels, packed stat sync, C++ reducer
buckets — with real world_size=2 semantics.  (RCCL forbids two ranks on one
device; the 8-GPU RCCL run is the driver's scale bench.)"""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu

WORLD = 2


def _worker(rank, fn_name, tmpdir, q):
    try:
        dist.init_process_group(
            "gloo", init_method=f"file://{tmpdir}/pg", rank=rank,
            world_size=WORLD,
        )
        torch.cuda.set_device(0)
        globals()[fn_name](rank)
        q.put((rank, None))
    except Exception:  # pragma: no cover
        import traceback

        q.put((rank, traceback.format_exc()))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def _spawn(fn_name, tmp_path):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    ps = [ctx.Process(target=_worker, args=(r, fn_name, str(tmp_path), q))
          for r in range(WORLD)]
    for p in ps:
        p.start()
    errs = []
    for _ in range(WORLD):
        rank, err = q.get()
        if err:
            errs.append((rank, err))
    for p in ps:
        p.join(timeout=120)
        if p.is_alive():
            p.terminate()
            errs.append((p.pid, "timeout"))
    assert not errs, "\n".join(f"rank {r}:\n{e}" for r, e in errs)


def _golden_gpu_body(rank):
    import msbn

    dev = torch.device("cuda:0")
    torch.manual_seed(42)
    global_bs = 8
    local_bs = global_bs // WORLD

    net = msbn.convert_sync_batchnorm(msbn.models.SimpleCNN(width=8)).to(dev)
    net = msbn.parallel.DistributedDataParallel(net)
    opt = torch.optim.SGD(net.parameters(), lr=0.05, momentum=0.9)

    torch.manual_seed(42)
    gold = msbn.models.SimpleCNN(width=8).to(dev)
    gopt = torch.optim.SGD(gold.parameters(), lr=0.05, momentum=0.9)

    loss_fn = torch.nn.CrossEntropyLoss()
    for it in range(4):
        g = torch.Generator().manual_seed(100 + it)
        x = torch.randn(global_bs, 3, 16, 16, generator=g).to(dev)
        y = torch.randint(0, 10, (global_bs,), generator=g).to(dev)
        xs = x[rank * local_bs:(rank + 1) * local_bs]
        ys = y[rank * local_bs:(rank + 1) * local_bs]
        opt.zero_grad(set_to_none=True)
        loss_fn(net(xs), ys).backward()
        opt.step()
        gopt.zero_grad(set_to_none=True)
        loss_fn(gold(x), y).backward()
        gopt.step()

    for (n1, p1), (n2, p2) in zip(net.module.named_parameters(),
                                  gold.named_parameters()):
        assert torch.allclose(p1, p2, atol=1e-4), f"{n1} diverged"
    for (n1, b1), (n2, b2) in zip(net.module.named_buffers(),
                                  gold.named_buffers()):
        assert torch.allclose(b1.float(), b2.float(), atol=1e-4), n1


def test_golden_model_world2_gpu(tmp_path):
    _spawn("_golden_gpu_body", tmp_path)


def _fused_world2_body(rank):
    """Fused SyncBatchNormAct2d across 2 ranks == single-process composed."""
    import msbn
    from msbn.nn.fused import SyncBatchNormAct2d

    dev = torch.device("cuda:0")
    torch.manual_seed(7)
    C = 16
    bn = SyncBatchNormAct2d(C, relu=True).to(dev)
    xs = [torch.randn(3, C, 6, 6) for _ in range(WORLD)]
    res = [torch.randn(3, C, 6, 6) for _ in range(WORLD)]
    x_local = xs[rank].to(dev).requires_grad_(True)
    r_local = res[rank].to(dev).requires_grad_(True)
    y = bn(x_local, r_local)
    y.sum().backward()

    # oracle: plain BN over concatenation + add + relu (CPU fp32)
    x_all = torch.cat(xs).requires_grad_(True)
    r_all = torch.cat(res).requires_grad_(True)
    tbn = torch.nn.BatchNorm2d(C)
    ty = torch.relu(tbn(x_all) + r_all)
    ty.sum().backward()
    off = rank * 3
    assert torch.allclose(y.detach().cpu(), ty[off:off + 3].detach(),
                          atol=1e-4)
    assert torch.allclose(x_local.grad.cpu(), x_all.grad[off:off + 3],
                          atol=1e-4)
    assert torch.allclose(r_local.grad.cpu(), r_all.grad[off:off + 3],
                          atol=1e-5)
    assert torch.allclose(bn.running_mean.cpu(), tbn.running_mean, atol=1e-5)


def test_fused_syncbn_world2_gpu(tmp_path):
    _spawn("_fused_world2_body", tmp_path)


def _join_ctx_gpu_body(rank):
    """Stock-style `with ddp.join():` on GPU tensors (2 gloo ranks on one
    device): shadow steps run the 0-batch HIP path (zero-count stats masked
    in-kernel), final model broadcast from the most-iterated rank."""
    import msbn

    dev = torch.device("cuda:0")
    torch.manual_seed(17)
    net = msbn.convert_sync_batchnorm(msbn.models.SimpleCNN(width=8)).to(dev)
    net = msbn.parallel.DistributedDataParallel(net)
    opt = torch.optim.SGD(net.parameters(), lr=0.05)
    n_batches = 3 if rank == 0 else 1
    with net.join():
        for i in range(n_batches):
            x = torch.randn(
                2, 3, 8, 8,
                generator=torch.Generator().manual_seed(rank * 31 + i),
            ).to(dev)
            opt.zero_grad(set_to_none=True)
            net(x).float().pow(2).mean().backward()
            opt.step()
    flat = torch.cat([p.detach().flatten() for p in net.module.parameters()])
    flat0 = flat.clone()
    dist.broadcast(flat0, src=0)
    assert torch.allclose(flat, flat0, atol=1e-5)


def test_join_context_world2_gpu(tmp_path):
    _spawn("_join_ctx_gpu_body", tmp_path)
