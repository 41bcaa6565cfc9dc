"""DeAR without tensor fusion: per-TENSOR reduce-scatter/all-gather pipelining
(reference dear/dopt_rsag_naive.py — the "DeAR w/o TF" ablation demonstrating
why fusion matters: heavy per-collective startup costs)."""
from __future__ import annotations

from typing import List

from .dear import DearOptimizer
from .fusion import BucketGroup, ParamSlot, _module_param_order

__all__ = ["NaiveDearOptimizer"]


class NaiveDearOptimizer(DearOptimizer):
    def __init__(self, optimizer, model, **kw):
        kw.pop("threshold_bytes", None)
        super().__init__(optimizer, model, threshold_bytes=None, **kw)

    def _build(self, threshold_bytes):
        # one group per parameter tensor
        groups: List[BucketGroup] = []
        for m, ps in _module_param_order(self.model):
            for name, p in ps:
                g = BucketGroup(index=len(groups), modules=[m],
                                slots=[ParamSlot(name, p, 0, p.numel())])
                g.numel = p.numel()
                groups.append(g)
        self.groups = groups
        for g in self.groups:
            g.allocate(self.size, self._device)
        self._slot_of = {g.slots[0].param: (g, g.slots[0]) for g in self.groups}
        n = len(self.groups)
        from ..comm.backend import NULL_HANDLE
        self._ready_count = [0] * n
        self._rs_handle = [NULL_HANDLE] * n
        self._ag_handle = [NULL_HANDLE] * n
        self._updated = [True] * n
        self._prev_iter_done = NULL_HANDLE
