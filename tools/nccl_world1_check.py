#!/usr/bin/env python3
"""RCCL world-1 validation.

A backend="nccl" process group with world_size=1 on one MI355X, with
MSBN_FORCE_SYNC=1, executes the EXACT code path an 8-GPU job runs:
`dist.all_gather_into_tensor` of the packed BN stats over RCCL
(msbn/nn/functions.py), the backward stat `all_reduce`, and the C++
reducer's bucket all-reduce on ProcessGroupNCCL's dedicated comm stream.

Cases (argv[1]):
  ddp    — 3 full DDP+SyncBN train steps over RCCL; then forced-sync
           forward/backward must match the local (no-collective) path.
  graph  — whole-step hipGraph capture (fwd+bwd+opt) WITH the RCCL
           collectives inside the graph; replays must track an eager clone.

Run standalone (a known ROCm issue segfaults hipGraph capture inside the
pytest host process — see tests/test_gpu_fused.py::test_graphed_step_gpu).
"""

import os
import sys

os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29611")
os.environ["MSBN_FORCE_SYNC"] = "1"

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402


def make_model(seed=7):
    import msbn

    torch.manual_seed(seed)
    m = msbn.convert_sync_batchnorm(msbn.models.SimpleCNN(width=16))
    return m.cuda()


def case_ddp():
    import msbn

    model = make_model()
    ddp = msbn.parallel.DistributedDataParallel(model, device_ids=[0],
                                                output_device=0)
    opt = torch.optim.SGD(ddp.parameters(), lr=0.05, momentum=0.9)
    loss_fn = torch.nn.CrossEntropyLoss()
    for it in range(3):
        g = torch.Generator().manual_seed(it)
        x = torch.randn(8, 3, 8, 8, generator=g).cuda()
        y = torch.randint(0, 10, (8,), generator=g).cuda()
        opt.zero_grad(set_to_none=True)
        loss = loss_fn(ddp(x), y)
        loss.backward()
        opt.step()
        assert torch.isfinite(loss).item(), f"non-finite loss at iter {it}"

    # forced-sync (RCCL all_gather path) == local path, same module state
    bn1 = msbn.nn.SyncBatchNorm(16).cuda()
    bn2 = msbn.nn.SyncBatchNorm(16).cuda()
    bn2.load_state_dict(bn1.state_dict())
    bn1.train(), bn2.train()
    x = torch.randn(4, 16, 6, 6, device="cuda")

    x1 = x.clone().requires_grad_(True)
    os.environ["MSBN_FORCE_SYNC"] = "1"
    y1 = bn1(x1)
    y1.pow(2).sum().backward()

    x2 = x.clone().requires_grad_(True)
    os.environ["MSBN_FORCE_SYNC"] = "0"
    y2 = bn2(x2)
    y2.pow(2).sum().backward()
    os.environ["MSBN_FORCE_SYNC"] = "1"

    torch.testing.assert_close(y1, y2, atol=2e-5, rtol=2e-5)
    torch.testing.assert_close(x1.grad, x2.grad, atol=2e-5, rtol=2e-5)
    torch.testing.assert_close(bn1.running_mean, bn2.running_mean,
                               atol=1e-6, rtol=1e-6)
    torch.testing.assert_close(bn1.running_var, bn2.running_var,
                               atol=1e-6, rtol=1e-6)
    print("CASE ddp OK")


def case_graph():
    import msbn

    steps = 4

    def train(graphed):
        # DDP construction, warmup AND capture all happen on ONE side
        # stream: the Reducer's autograd hooks hang off AccumulateGrad nodes
        # created at DDP-construction time, and the engine runs those nodes
        # on their creation-time stream — if that is the default stream, the
        # hook's bucket copy + RCCL all-reduce would be launched OUTSIDE the
        # capture and segfault it (the torch input_buffer.cpp stream-mismatch
        # warning).  Same recipe as stock DDP+CUDA-graph usage.
        side = torch.cuda.Stream() if graphed else None
        import contextlib

        stream_ctx = (
            torch.cuda.stream(side) if graphed else contextlib.nullcontext()
        )
        with stream_ctx:
            model = make_model(seed=21)
            model = msbn.parallel.DistributedDataParallel(
                model, device_ids=[0], output_device=0
            )
            opt = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9)
            loss_fn = torch.nn.CrossEntropyLoss()
            gen = torch.Generator().manual_seed(123)
            x = torch.randn(8, 3, 8, 8, generator=gen).cuda()
            y = torch.randint(0, 10, (8,), generator=gen).cuda()

            def step():
                out = model(x)
                loss = loss_fn(out, y)
                loss.backward()
                opt.step()
                torch._foreach_zero_(
                    [p.grad for p in model.parameters() if p.grad is not None]
                )
                return loss

            # warmup (allocator steady state; primes RCCL communicators)
            for _ in range(3):
                opt.zero_grad(set_to_none=False)
                step()
        torch.cuda.synchronize()
        if graphed:
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g, stream=side):
                step()
            for _ in range(steps):
                g.replay()
        else:
            for _ in range(steps):
                step()
        torch.cuda.synchronize()
        return torch.cat([p.detach().flatten().float()
                          for p in model.parameters()])

    p_eager = train(graphed=False)
    p_graph = train(graphed=True)

    diff = (p_eager - p_graph).abs().max().item()
    assert torch.isfinite(p_graph).all().item(), "non-finite params after replay"
    assert diff < 1e-3, f"graphed step diverged from eager: max diff {diff}"
    print("CASE graph OK")


def case_h2d():
    """CPU inputs are moved to the module's GPU on a side stream inside
    DDP.forward (stock _pre_forward input move, distributed.py:1564-1571)."""
    import msbn

    model = make_model(seed=5)
    ddp = msbn.parallel.DistributedDataParallel(model, device_ids=[0],
                                                output_device=0)
    x_cpu = torch.randn(4, 3, 8, 8)
    out = ddp(x_cpu)  # would raise device-mismatch without the move
    assert out.is_cuda and out.shape[0] == 4
    out.sum().backward()
    stats = ddp._get_ddp_logging_data()
    assert "backward_grad_ready_us" in stats
    ready = [s for s in stats["backward_grad_ready_us"] if s >= 0]
    assert len(ready) == len(list(p for p in model.parameters()
                                  if p.requires_grad))
    print("CASE h2d OK")


def case_stress():
    """Allocator-safety of the async bucket path under churn: every
    iteration allocates/frees random-size temporaries while the reducer's
    bucket all-reduces are in flight on PGNCCL's comm stream.  30 DDP steps
    must track a plain (no-DDP) clone bit-for-bit at world 1 — any
    allocator reuse of a bucket block mid-flight corrupts grads and
    diverges the models (SURVEY.md §5.2 recordStream/stash problem)."""
    import msbn

    torch.manual_seed(31)
    gen = torch.Generator().manual_seed(77)
    model = make_model(seed=31)
    ddp = msbn.parallel.DistributedDataParallel(model, device_ids=[0],
                                                output_device=0)
    ref = make_model(seed=31)  # identical init
    opt = torch.optim.SGD(ddp.parameters(), lr=0.03, momentum=0.9)
    ropt = torch.optim.SGD(ref.parameters(), lr=0.03, momentum=0.9)
    loss_fn = torch.nn.CrossEntropyLoss()
    junk = []
    for it in range(30):
        x = torch.randn(8, 3, 8, 8, generator=gen).cuda()
        y = torch.randint(0, 10, (8,), generator=gen).cuda()
        opt.zero_grad(set_to_none=True)
        loss_fn(ddp(x), y).backward()
        # allocator churn while comm-stream work may still be in flight
        junk.clear()
        for _ in range(6):
            n = int(torch.randint(1, 4 << 20, (1,), generator=gen))
            junk.append(torch.empty(n, device="cuda").fill_(float(it)))
        opt.step()
        ropt.zero_grad(set_to_none=True)
        loss_fn(ref(x), y).backward()
        ropt.step()
    torch.cuda.synchronize()
    for (n, p), (_, rp) in zip(ddp.module.named_parameters(),
                               ref.named_parameters()):
        torch.testing.assert_close(p, rp, atol=1e-5, rtol=1e-5, msg=n)
    print("CASE stress OK")


def main():
    case = sys.argv[1] if len(sys.argv) > 1 else "ddp"
    torch.cuda.set_device(0)
    dist.init_process_group("nccl", init_method="env://", world_size=1, rank=0)
    try:
        if case == "ddp":
            case_ddp()
        elif case == "graph":
            case_graph()
        elif case == "h2d":
            case_h2d()
        elif case == "stress":
            case_stress()
        else:
            raise SystemExit(f"unknown case {case}")
    finally:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
