#!/usr/bin/env python3
"""DCGAN G+D with SyncBN (BASELINE.json config 4: small-per-GPU-batch GAN).

    python benchmarks/bench_dcgan.py [--steps 20 --warmup 5 --batch-size 64]
    torchrun --nproc-per-node 4 benchmarks/bench_dcgan.py ...

Rank 0 prints one JSON line (images/sec whole job, G+D step)."""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

import msbn


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--batch-size", type=int, default=64)
    p.add_argument("--image-size", type=int, default=64)
    p.add_argument("--stock", action="store_true",
                   help="stock torch SyncBatchNorm+DDP comparison line "
                        "(identical architecture and init)")
    p.add_argument("--benchmark", action="store_true",
                   help="MIOpen conv autotune (uses/extends the in-tree "
                        "miopen_db find-db)")
    p.add_argument("--graph", action="store_true",
                   help="capture the G+D step in one hipGraph (msbn only; "
                        "the GAN regime is launch-bound)")
    p.add_argument("--local_rank", "--local-rank", type=int,
                   default=int(os.environ.get("LOCAL_RANK", 0)),
                   dest="local_rank")
    args = p.parse_args()

    _db = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "miopen_db")
    if os.path.isdir(_db) and os.listdir(_db):
        os.environ.setdefault("MIOPEN_USER_DB_PATH", _db)
        # the shipped find-db covers this bench's default shapes: tuned
        # solver lookups are free, so autotune defaults ON
        if os.environ.get("MSBN_NO_AUTOTUNE", "0") != "1":
            args.benchmark = True
    if args.benchmark:
        torch.backends.cudnn.benchmark = True

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    use_cuda = torch.cuda.is_available()
    device = torch.device(f"cuda:{args.local_rank}" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(device)
    if world > 1:
        dist.init_process_group("nccl" if use_cuda else "gloo",
                                init_method="env://", world_size=world,
                                rank=rank)

    torch.manual_seed(7)
    import contextlib

    use_graph = args.graph and use_cuda and not args.stock
    side = torch.cuda.Stream() if use_graph else None
    stream_ctx = torch.cuda.stream(side) if use_graph else contextlib.nullcontext()
    with stream_ctx:
        if args.stock:
            from msbn.models import convert_to_torch_batchnorm

            G = convert_to_torch_batchnorm(msbn.models.Generator())
            D = convert_to_torch_batchnorm(msbn.models.Discriminator())
            if world > 1:
                G = torch.nn.SyncBatchNorm.convert_sync_batchnorm(G)
                D = torch.nn.SyncBatchNorm.convert_sync_batchnorm(D)
            G, D = G.to(device), D.to(device)
            if world > 1:
                G = torch.nn.parallel.DistributedDataParallel(
                    G, device_ids=[args.local_rank] if use_cuda else None)
                D = torch.nn.parallel.DistributedDataParallel(
                    D, device_ids=[args.local_rank] if use_cuda else None)
        else:
            from msbn.nn import fuse_bn_act

            G = fuse_bn_act(msbn.convert_sync_batchnorm(msbn.models.Generator())).to(device)
            D = msbn.convert_sync_batchnorm(msbn.models.Discriminator()).to(device)
            if world > 1:
                G = msbn.parallel.DistributedDataParallel(
                    G, device_ids=[args.local_rank] if use_cuda else None)
                D = msbn.parallel.DistributedDataParallel(
                    D, device_ids=[args.local_rank] if use_cuda else None)
        # capturable Adam keeps the step/lr state on-device so the optimizer
        # records into the hipGraph (required for --graph; tiny host cost
        # otherwise, so only enabled when graphing)
        adam_kw = {"capturable": True} if use_graph else {}
        optG = torch.optim.Adam(G.parameters(), lr=2e-4, betas=(0.5, 0.999),
                                **adam_kw)
        optD = torch.optim.Adam(D.parameters(), lr=2e-4, betas=(0.5, 0.999),
                                **adam_kw)
        bce = torch.nn.BCEWithLogitsLoss()

        bs = args.batch_size
        real = torch.randn(bs, 3, args.image_size, args.image_size, device=device)
        ones = torch.ones(bs, 1, device=device)
        zeros = torch.zeros(bs, 1, device=device)

        def step():
            z = torch.randn(bs, 100, 1, 1, device=device)
            # D step
            optD.zero_grad(set_to_none=True)
            fake = G(z)
            d_loss = bce(D(real), ones) + bce(D(fake.detach()), zeros)
            d_loss.backward()
            optD.step()
            # G step
            optG.zero_grad(set_to_none=True)
            g_loss = bce(D(fake), ones)
            g_loss.backward()
            optG.step()
            return d_loss, g_loss

        # warmup (on the side stream when graphing)
        for _ in range(args.warmup):
            step()

    graph = None
    if use_graph:
        # one hipGraph for the WHOLE G+D step (both backwards + both Adam
        # steps + in-graph RNG for z): the GAN regime is launch-bound, this
        # collapses ~hundreds of launches into one.  Same side-stream recipe
        # as bench.py; eager fallback on capture failure.
        torch.cuda.synchronize()
        try:
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph, stream=side):
                step()
            for _ in range(max(args.warmup, 3)):
                graph.replay()
        except Exception as e:
            print(f"[bench_dcgan] hipGraph capture failed ({e!r}); eager",
                  file=sys.stderr)
            graph = None
            torch.cuda.synchronize()

    if world > 1:
        dist.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    if graph is not None:
        for _ in range(args.steps):
            graph.replay()
    else:
        for _ in range(args.steps):
            step()
    if use_cuda:
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    el = time.perf_counter() - t0
    if world > 1:
        t = torch.tensor([el], device=device if use_cuda else "cpu",
                         dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        el = t.item()

    if rank == 0:
        print(json.dumps({
            "metric": "images/sec (whole job) DCGAN G+D SyncBN",
            "value": round(bs * world * args.steps / el, 2),
            "unit": "images/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(1000 * el / args.steps, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "dtype": "fp32",
            "data": "synthetic",
            "config": {"model": "dcgan64", "per_gpu_batch": bs,
                       "parallelism": f"dp{world}",
                       "hip_graph": graph is not None,
                       "impl": "stock" if args.stock else "msbn"},
        }))
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
