"""Env-contract helpers: read the launcher-provided variables
and set up the per-process device, mirroring README.md steps 1-2."""

import argparse
import os

import torch
import torch.distributed as dist


def get_local_rank(args=None) -> int:
    """--local_rank argv (legacy launch) wins, then LOCAL_RANK env, then 0."""
    if args is not None and getattr(args, "local_rank", None) is not None:
        return int(args.local_rank)
    return int(os.environ.get("LOCAL_RANK", "0"))


def add_local_rank_arg(parser: argparse.ArgumentParser):
    """The README.md:15-19 contract: --local_rank, type=int, default=0
    (argparse also accepts --local-rank for the same dest)."""
    parser.add_argument("--local_rank", "--local-rank", type=int, default=0,
                        dest="local_rank")
    return parser


def init_distributed(local_rank: int, world_size: int = None, backend: str = None):
    """README.md steps 2: set device then init_process_group('nccl', env://).
    Falls back to gloo when no GPU is present (CPU plumbing config)."""
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(local_rank)
    if backend is None:
        backend = "nccl" if use_cuda else "gloo"
    if world_size is None:
        world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", str(local_rank)))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    dist.init_process_group(
        backend=backend, init_method="env://", world_size=world_size, rank=rank
    )
    return rank, world_size
