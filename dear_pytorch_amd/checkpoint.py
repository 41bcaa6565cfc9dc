"""Checkpoint / resume helpers.

The reference has no in-training checkpointing — only the
startup consistency broadcasts.  This module adds rank-0 save / all-rank
resume on top of them, completing the capability for production training:

    dear.checkpoint.save(path, model, optimizer, step=1234)
    step = dear.checkpoint.load(path, model, optimizer)   # then broadcasts
"""
from __future__ import annotations

import os
from typing import Optional

import torch

import dear_pytorch_amd as dear

__all__ = ["save", "load"]


def save(path: str, model: torch.nn.Module, optimizer=None,
         step: int = 0, extra: Optional[dict] = None):
    """Rank-0 writes {model, optimizer, step, extra} atomically."""
    if dear.rank() != 0:
        if dear.size() > 1:
            import torch.distributed as dist
            dist.barrier()
        return
    from .parallel.dear import DearOptimizer
    opt_sd = None
    if optimizer is not None:
        if isinstance(optimizer, DearOptimizer):
            optimizer.synchronize()  # fold in the pending lazy update
        opt_sd = optimizer.state_dict()
    tmp = path + ".tmp"
    torch.save({"model": model.state_dict(), "optimizer": opt_sd,
                "step": step, "extra": extra or {}}, tmp)
    os.replace(tmp, path)
    if dear.size() > 1:
        import torch.distributed as dist
        dist.barrier()


def load(path: str, model: torch.nn.Module, optimizer=None,
         map_location="cpu") -> int:
    """Every rank loads; parameters/optimizer state are then broadcast from
    rank 0 so all ranks are bit-identical. Returns the saved step."""
    ckpt = torch.load(path, map_location=map_location, weights_only=False)
    device = next(model.parameters()).device
    model.load_state_dict({k: v.to(device) if torch.is_tensor(v) else v
                           for k, v in ckpt["model"].items()})
    if optimizer is not None and ckpt.get("optimizer") is not None:
        sd = ckpt["optimizer"]
        for st in sd.get("state", {}).values():
            for k, v in st.items():
                if torch.is_tensor(v) and v.dim() > 0:
                    st[k] = v.to(device)
        optimizer.load_state_dict(sd)
    if dear.size() > 1:
        dear.broadcast_parameters(model.state_dict(), root_rank=0)
        if optimizer is not None:
            dear.broadcast_optimizer_state(optimizer, root_rank=0)
    return int(ckpt.get("step", 0))
