#!/usr/bin/env python3
"""All-reduce / all-gather latency microbenchmark (the second half of the
BASELINE.json headline metric): message sizes spanning the SyncBN stat
messages (2C+1 floats, 0.5-16 KB) and the DDP gradient buckets (1/25 MiB).

    torchrun --nproc-per-node N benchmarks/bench_allreduce.py
    python benchmarks/bench_allreduce.py            # world=1 smoke

Rank 0 prints one JSON line per (op, size).
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=50)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--local_rank", "--local-rank", type=int,
                   default=int(os.environ.get("LOCAL_RANK", 0)),
                   dest="local_rank")
    args = p.parse_args()

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

    # bytes: SyncBN stat messages for C=64..2048 (2C+1 fp32), then buckets
    sizes = [4 * (2 * c + 1) for c in (64, 256, 512, 2048)] + [
        64 * 1024, 1 << 20, 25 << 20,
    ]
    results = []
    for op in ("all_reduce", "all_gather"):
        for nbytes in sizes:
            n = nbytes // 4
            t = torch.ones(n, dtype=torch.float32, device=device)
            out = (
                torch.empty(world * n, dtype=torch.float32, device=device)
                if op == "all_gather"
                else None
            )

            def call():
                if world > 1:
                    if op == "all_reduce":
                        dist.all_reduce(t)
                    else:
                        dist.all_gather_into_tensor(out, t) if use_cuda else \
                            dist.all_gather(list(out.chunk(world)), t)

            for _ in range(args.warmup):
                call()
            if use_cuda:
                torch.cuda.synchronize()
            if world > 1:
                dist.barrier()
            t0 = time.perf_counter()
            for _ in range(args.iters):
                call()
            if use_cuda:
                torch.cuda.synchronize()
            el = time.perf_counter() - t0
            us = 1e6 * el / args.iters
            if world > 1:
                mt = torch.tensor([us], dtype=torch.float64,
                                  device=device if use_cuda else "cpu")
                dist.all_reduce(mt, op=dist.ReduceOp.MAX)
                us = mt.item()
            results.append({"op": op, "bytes": nbytes, "latency_us":
                            round(us, 2), "n_gpus": world})

    if rank == 0:
        for r in results:
            print(json.dumps(r))
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
