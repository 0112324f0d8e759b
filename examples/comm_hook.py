#!/usr/bin/env python3
"""Gradient-communication hooks with msbn DDP.

Two tiers (msbn/parallel/comm_hooks.py):
  * builtin c10d hooks, matched by identity -> C++ overlapped fast path
  * arbitrary Python hooks with a GradBucket carrier (run at finalize)

    python -m msbn.launch --nproc_per_node=2 examples/comm_hook.py
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist
from torch.distributed.algorithms.ddp_comm_hooks import default_hooks as dh

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
    device = torch.device(f"cuda:{args.local_rank}" if use_cuda else "cpu")
    world = dist.get_world_size()

    torch.manual_seed(0)
    net = msbn.parallel.DistributedDataParallel(
        torch.nn.Linear(16, 16).to(device),
        device_ids=[args.local_rank] if use_cuda else None)

    # 1) builtin compression hook: wire-dtype cast inside the C++ reducer,
    #    full backward/comm overlap kept
    net.register_comm_hook(None, dh.bf16_compress_hook)
    net(torch.randn(4, 16, device=device)).sum().backward()

    # 2) custom Python hook: receives (state, GradBucket), returns a Future
    #    resolving to the reduced flat tensor; replaces allreduce + division
    log = []

    def logging_allreduce(state, bucket):
        log.append((bucket.index(), bucket.buffer().numel()))
        t = bucket.buffer()
        t.div_(world)
        work = dist.all_reduce(t, async_op=True)
        return work.get_future().then(lambda f: f.value()[0])

    net.register_comm_hook(None, logging_allreduce)
    for p_ in net.parameters():
        p_.grad = None
    net(torch.randn(4, 16, device=device)).sum().backward()
    assert log, "python hook did not run"
    msbn.utils.master_print(f"HOOKS_OK buckets={log}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
