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

    def _build(self, threshold_bytes, fusion_flags=None):
        # one group per parameter tensor (fusion_flags are meaningless without
        # fusion; accepted so regroup()'s call signature matches)
        groups: List[BucketGroup] = []
        for m, ps in _module_param_order(self.model):
            for name, p in ps:
                g = BucketGroup(index=len(groups), modules=[m],
                                slots=[ParamSlot(name, p, 0, p.numel())])
                g.numel = p.numel()
                groups.append(g)
        self.groups = groups
        for g in self.groups:
            g.allocate(self.size, self._device,
                       comm_dtype=self.comm_dtype if self.size > 1 else None,
                       attach_grads=not self.pack_grads)
        self._slot_of = {g.slots[0].param: (g, g.slots[0]) for g in self.groups}
        self._init_sched_state()
