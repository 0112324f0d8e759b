#!/usr/bin/env python3
"""Uneven per-rank input counts with stock-style ``DDP.join()``.

Each rank gets a different number of batches; exhausted ranks shadow-step
automatically inside the context (SyncBN posts zero-count stats, the
reducer all-reduces zero buckets), so nothing hangs and the final model is
identical on every rank.

    python -m msbn.launch --nproc_per_node=2 examples/uneven_inputs.py
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

import msbn


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--local_rank", "--local-rank", type=int,
                   default=int(os.environ.get("LOCAL_RANK", 0)),
                   dest="local_rank")
    args = p.parse_args()
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(args.local_rank)
    dist.init_process_group("nccl" if use_cuda else "gloo",
                            init_method="env://")
    rank = dist.get_rank()
    device = torch.device(f"cuda:{args.local_rank}" if use_cuda else "cpu")

    torch.manual_seed(0)
    net = msbn.convert_sync_batchnorm(msbn.models.SimpleCNN(width=8)).to(device)
    net = msbn.parallel.DistributedDataParallel(
        net, device_ids=[args.local_rank] if use_cuda else None)
    opt = torch.optim.SGD(net.parameters(), lr=0.05)

    n_batches = 4 - rank  # rank 0 trains longer than the others
    with net.join():
        for i in range(n_batches):
            x = torch.randn(4, 3, 8, 8, device=device)
            opt.zero_grad(set_to_none=True)
            net(x).float().pow(2).mean().backward()
            opt.step()

    flat = torch.cat([p.detach().flatten() for p in net.module.parameters()])
    ref = flat.clone()
    dist.broadcast(ref, src=0)
    assert torch.allclose(flat, ref, atol=1e-5)
    msbn.utils.master_print(f"JOIN_OK (ranks finished {n_batches}.. batches, "
                            "params identical)")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
