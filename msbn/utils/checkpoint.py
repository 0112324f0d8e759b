"""Checkpoint / resume conventions for msbn DDP training:
rank 0 saves ``model.module.state_dict()`` (BN running stats ride the buffers),
all ranks load behind a barrier.  DDP modules themselves are picklable
(__getstate__/__setstate__ rebuild the reducer)."""

import os
from typing import Optional

import torch
import torch.distributed as dist


def _unwrap(model):
    return model.module if hasattr(model, "module") else model


def save_checkpoint(path: str, model, optimizer=None, epoch: int = 0,
                    extra: Optional[dict] = None):
    """Rank 0 writes; all ranks barrier afterwards so a following load is safe."""
    if not dist.is_initialized() or dist.get_rank() == 0:
        state = {
            "model": _unwrap(model).state_dict(),
            "epoch": epoch,
        }
        if optimizer is not None:
            state["optimizer"] = optimizer.state_dict()
        if extra:
            state["extra"] = extra
        tmp = path + ".tmp"
        torch.save(state, tmp)
        os.replace(tmp, path)
    if dist.is_initialized():
        dist.barrier()


def load_checkpoint(path: str, model, optimizer=None, map_location="cpu"):
    state = torch.load(path, map_location=map_location, weights_only=True)
    _unwrap(model).load_state_dict(state["model"])
    if optimizer is not None and "optimizer" in state:
        optimizer.load_state_dict(state["optimizer"])
    if dist.is_initialized():
        dist.barrier()
    return state.get("epoch", 0)
