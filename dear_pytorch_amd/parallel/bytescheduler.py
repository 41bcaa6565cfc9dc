"""ByteScheduler-style baseline: priority-scheduled, partitioned gradient sync.

Reference capability: the bytescheduler/ benchmark path (C23 in SURVEY.md —
`bsc.ScheduledOptimizer(model, optimizer, max_steps)` over Horovod).  The
external bytescheduler library negotiates op order through Horovod's
coordinator; RCCL instead requires every rank to issue collectives in the
same order, so this implementation uses a DETERMINISTIC priority drain:

* each parameter tensor is partitioned into <= partition_bytes chunks
  (ByteScheduler's tensor partitioning — small high-priority pieces are not
  stuck behind a large low-priority transfer),
* ready chunks enter a priority queue keyed by FORWARD order (layers needed
  earliest next iteration sync first — ByteScheduler's priority rule),
* the queue is drained in priority order at every hook/step boundary; since
  the backward-ready sequence is identical on every rank, the issue order is
  identical too (RCCL-safe by construction).

step() waits for all chunks, averages, and applies the wrapped optimizer.
"""
from __future__ import annotations

import heapq
from typing import List, Optional

import torch

from ..comm.backend import CommBackend, create_backend
from .fusion import BucketGroup, build_groups

__all__ = ["ByteSchedulerOptimizer"]


class ByteSchedulerOptimizer(torch.optim.Optimizer):
    def __init__(self, optimizer, model, partition_bytes: int = 4 * 1024 * 1024,
                 backend: Optional[CommBackend] = None, **kw):
        self.optim = optimizer
        self.model = model
        self.backend = backend or create_backend("bsc")
        self.rank, self.size = self.backend.rank, self.backend.size
        self._device = next(model.parameters()).device
        self.partition = max(partition_bytes // 4, 1024)  # elements
        # one group per module; priority = forward position
        self.groups: List[BucketGroup] = build_groups(model, None)
        for g in self.groups:
            g.allocate(self.size, self._device)
        self._slot_of = {}
        for g in self.groups:
            for s in g.slots:
                self._slot_of[s.param] = (g, s)
        self._ready = [0] * len(self.groups)
        self._heap = []  # (priority, seq, chunk_tensor)
        self._seq = 0
        self._handles = []
        self._grad_accs = []
        for g in self.groups:
            for s in g.slots:
                p = s.param
                acc = p.expand_as(p).grad_fn.next_functions[0][0]
                acc.register_hook(self._make_hook(p))
                self._grad_accs.append(acc)

    def _make_hook(self, p):
        def hook(*_):
            group, slot = self._slot_of[p]
            b = group.bucket[slot.offset: slot.offset + slot.numel]
            if p.grad is not None and p.grad.data_ptr() != b.data_ptr():
                from .fusion import grad_view
                gv = grad_view(b, p)
                gv.add_(p.grad)
                p.grad = gv
            self._ready[group.index] += 1
            if self._ready[group.index] == len(group.slots):
                self._enqueue(group)
                self._drain()
        return hook

    def _enqueue(self, group: BucketGroup):
        n = group.bucket.numel()
        off = 0
        while off < n:
            end = min(off + self.partition, n)
            heapq.heappush(self._heap,
                           (group.index, self._seq, group.bucket[off:end]))
            self._seq += 1
            off = end

    def _drain(self):
        while self._heap:
            _, _, chunk = heapq.heappop(self._heap)
            if self.size > 1:
                self._handles.append(self.backend.all_reduce(chunk))

    def zero_grad(self, set_to_none: bool = False):
        pass

    def step(self, closure=None):
        loss = closure() if closure is not None else None
        self._drain()
        for h in self._handles:
            h.wait_compute()
        self._handles.clear()
        if self.size > 1:
            for g in self.groups:
                g.bucket.mul_(1.0 / self.size)
        self.optim.step()
        for g in self.groups:
            g.bucket.zero_()
        self._ready = [0] * len(self.groups)
        return loss

    def synchronize(self):
        if torch.cuda.is_available() and self._device.type == "cuda":
            torch.cuda.synchronize()

    @property
    def param_groups(self):
        return self.optim.param_groups

    @property
    def state(self):
        return self.optim.state

    def state_dict(self):
        return self.optim.state_dict()

    def load_state_dict(self, sd):
        self.optim.load_state_dict(sd)
