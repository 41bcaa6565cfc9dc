"""ByteScheduler-style baseline: priority-scheduled, partitioned gradient sync.

Reference capability: the bytescheduler/ benchmark path (C23 in SURVEY.md —
`bsc.ScheduledOptimizer(model, optimizer, max_steps)` over Horovod).  The
external bytescheduler library negotiates op order through Horovod's
coordinator; RCCL instead requires every rank to issue collectives in the
same order, so this implementation uses a DETERMINISTIC credit scheduler:

* each parameter tensor is partitioned into <= partition_bytes chunks
  (ByteScheduler's tensor partitioning — small high-priority pieces are not
  stuck behind a large low-priority transfer),
* ready chunks enter a priority queue keyed by FORWARD order (layers needed
  earliest next iteration sync first — ByteScheduler's priority rule),
* at every drain opportunity (each backward hook and step()) the queue is
  drained in priority order, but only while the in-flight byte budget
  (``credit_bytes``) lasts; chunks that do not fit stay QUEUED across hooks,
  so a high-priority chunk arriving later (backward visits last layers
  first) preempts queued low-priority chunks at the next opportunity,
* credit is replenished by retiring the oldest in-flight transfer; since
  retirement order, the backward-ready sequence and every scheduling
  decision are functions of rank-identical state, all ranks issue the same
  collective order (RCCL-safe by construction, no coordinator needed).

step() drains with blocking credit waits, waits for all chunks, averages,
and applies the wrapped optimizer.
"""
from __future__ import annotations

import heapq
from collections import deque
from typing import List, Optional

import torch

from ..comm.backend import CommBackend, create_backend
from .fusion import BucketGroup, build_groups

__all__ = ["ByteSchedulerOptimizer"]


class ByteSchedulerOptimizer(torch.optim.Optimizer):
    def __init__(self, optimizer, model, partition_bytes: int = 4 * 1024 * 1024,
                 credit_bytes: Optional[int] = None,
                 backend: Optional[CommBackend] = None, **kw):
        self.optim = optimizer
        self.model = model
        self.backend = backend or create_backend("bsc")
        self.rank, self.size = self.backend.rank, self.backend.size
        self._device = next(model.parameters()).device
        self.partition = max(partition_bytes // 4, 1024)  # elements
        # in-flight budget: ByteScheduler's credit. Default 4 partitions —
        # enough to keep the channel busy, small enough that priority matters.
        self.credit_bytes = credit_bytes if credit_bytes is not None \
            else 4 * self.partition * 4
        # one group per module; priority = forward position
        self.groups: List[BucketGroup] = build_groups(model, None)
        for g in self.groups:
            g.allocate(self.size, self._device)
        self._slot_of = {}
        for g in self.groups:
            for s in g.slots:
                self._slot_of[s.param] = (g, s)
        self._ready = [0] * len(self.groups)
        self._heap = []  # (priority=fwd index, seq, chunk_tensor)
        self._seq = 0
        self._inflight = deque()  # (handle, nbytes), oldest first
        self._outstanding = 0     # bytes issued, not yet retired
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
                self._drain(block=False)
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

    def _issue(self, chunk, nbytes):
        if self.size > 1:
            h = self.backend.all_reduce(chunk)
            self._inflight.append((h, nbytes))
            self._outstanding += nbytes

    def _retire_oldest(self):
        h, nb = self._inflight.popleft()
        h.host_wait()
        self._outstanding -= nb

    def _drain(self, block: bool):
        """Issue queued chunks in priority order while credit lasts.

        block=False (backward hooks): stop when the budget is exhausted —
        the remaining chunks stay queued so later, higher-priority arrivals
        can overtake them.  block=True (step()): recover credit by retiring
        the oldest in-flight transfer and keep going.
        """
        while self._heap:
            nbytes = self._heap[0][2].numel() * self._heap[0][2].element_size()
            while self._outstanding + nbytes > self.credit_bytes \
                    and self._inflight:
                if not block:
                    return
                self._retire_oldest()
            # a chunk larger than the whole budget issues alone
            _, _, chunk = heapq.heappop(self._heap)
            self._issue(chunk, nbytes)

    def zero_grad(self, set_to_none: bool = False):
        pass

    def step(self, closure=None):
        loss = closure() if closure is not None else None
        self._drain(block=True)
        while self._inflight:
            self._retire_oldest()
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
