"""Small device-aware torch.distributed helpers."""
from __future__ import annotations

from typing import List

import torch
import torch.distributed as dist

__all__ = ["bcast_floats"]


def bcast_floats(vals: List[float], src: int = 0) -> List[float]:
    """Broadcast a small float list over the default PG, device-matched:
    nccl(=RCCL) requires device tensors; gloo wants CPU."""
    t = torch.tensor(vals, dtype=torch.float64)
    if dist.get_backend() == "nccl" and torch.cuda.is_available():
        t = t.cuda()
    dist.broadcast(t, src=src)
    return t.cpu().tolist()
