"""Wait-time adaptive fusion (reference capability: dear/dopt_rsag_wt.py).

Instead of a byte threshold, derive fusion boundaries from MEASURED
wait-in-buffer times: start with everything merged, record how long each
module's gradients sit in the bucket before the group completes during
backward, then split groups so no gradient waits longer than a cycle budget
(reference CYCLE_TIME = 5 ms, dopt_rsag_wt.py:40; EMA alpha = 0.9; flags
broadcast from rank 0 at a fixed warmup step for cross-rank consistency).

On GPU the waits are DEVICE times: each push and each group completion
records a hipEvent on the compute stream and the gaps are read back with
``elapsed_time`` after the iteration's sync (host
``perf_counter`` in backward hooks measures kernel-launch spacing, not how
long gradients actually sit in the bucket — hooks run far ahead of the
device).  The reference paid a full ``torch.cuda.synchronize()`` per hook
instead (profiling.py:47); events are free at hook time and only one sync
per measured iteration is needed.
"""
from __future__ import annotations

import time
from typing import Dict, List

import torch

__all__ = ["WaitTimeAdaptiveFusion"]


class WaitTimeAdaptiveFusion:
    def __init__(self, opt, cycle_time_s: float = 5e-3, ema: float = 0.9,
                 regroup_at_step: int = 5, verbose: bool = True):
        self.opt = opt
        self.cycle = cycle_time_s
        self.ema = ema
        self.regroup_at = regroup_at_step
        self.verbose = verbose and opt.rank == 0
        self._step = 0
        self._push_t: Dict[int, float] = {}
        self._push_ev: Dict[int, torch.cuda.Event] = {}
        self._pending = []                     # (group, done_event) per iter
        self._wait: Dict[int, float] = {}      # param-id -> EMA wait seconds
        self._on_gpu = (torch.cuda.is_available()
                        and opt._device.type == "cuda")
        self._orig_hook = opt._make_bw_hook
        self._done = False
        self._install()

    def _install(self):
        """Wrap the optimizer's backward hooks to timestamp pushes and group
        completions (capability of _update_wait_times, dopt_rsag_wt.py:376)."""
        opt = self.opt
        outer = self

        def make_hook(p):
            inner = outer._orig_hook(p)

            def hook(*a):
                if outer._on_gpu:
                    ev = torch.cuda.Event(enable_timing=True)
                    ev.record()
                    outer._push_ev[id(p)] = ev
                else:
                    outer._push_t[id(p)] = time.perf_counter()
                g, s = opt._slot_of[p]
                pre = opt._ready_count[g.index]
                inner(*a)
                # the inner hook resets ready_count to 0 on completion
                if pre == len(g.slots) - 1:
                    if outer._on_gpu:
                        done_ev = torch.cuda.Event(enable_timing=True)
                        done_ev.record()
                        outer._pending.append((g, done_ev))
                    else:
                        done = time.perf_counter()
                        for slot in g.slots:
                            t0 = outer._push_t.get(id(slot.param))
                            if t0 is not None:
                                outer._ema_update(id(slot.param), done - t0)
            return hook

        opt._make_bw_hook = make_hook
        # re-register so wrapped hooks are live (groups unchanged); the old
        # backward hooks MUST be removed or readiness double-counts
        opt._remove_hooks()
        opt._register_hooks()

    def _ema_update(self, pid: int, w: float):
        prev = self._wait.get(pid, w)
        self._wait[pid] = self.ema * prev + (1 - self.ema) * w

    def _harvest_events(self):
        """Resolve this iteration's device-time waits (one sync, GPU only)."""
        if not self._pending:
            return
        torch.cuda.synchronize()
        for g, done_ev in self._pending:
            for slot in g.slots:
                ev = self._push_ev.get(id(slot.param))
                if ev is None:
                    continue
                self._ema_update(id(slot.param),
                                 ev.elapsed_time(done_ev) * 1e-3)
        self._pending.clear()
        self._push_ev.clear()

    def step_end(self):
        """Call once per training iteration, after opt.step()."""
        self._step += 1
        if self._done:
            return
        if self._on_gpu:
            self._harvest_events()
        if self._step < self.regroup_at:
            return
        flags = self._flags_from_waits()
        flags = self._sync_flags(flags)
        self.opt._make_bw_hook = self._orig_hook  # stop timestamping
        self.opt.regroup(fusion_flags=flags)
        self._done = True
        if self.verbose:
            print(f"[dear-wt] regrouped into {len(self.opt.groups)} groups "
                  f"at step {self._step}", flush=True)

    def _flags_from_waits(self) -> List[bool]:
        """Split where accumulated wait exceeds the cycle budget
        (capability of _update_groups_with_wait_times,
        dopt_rsag_wt.py:152-192)."""
        from .fusion import _module_param_order
        mods = _module_param_order(self.opt.model)
        flags = [False] * len(mods)
        if flags:
            flags[0] = True
        acc = 0.0
        for i, (m, ps) in enumerate(mods):
            w = max((self._wait.get(id(p), 0.0) for _, p in ps), default=0.0)
            acc += w
            if i > 0 and acc > self.cycle:
                flags[i] = True
                acc = 0.0
        return flags

    def _sync_flags(self, flags: List[bool]) -> List[bool]:
        import torch.distributed as dist
        if dist.is_initialized() and dist.get_world_size() > 1:
            from ..utils.dist_helpers import bcast_floats
            vals = bcast_floats([1.0 if f else 0.0 for f in flags])
            return [bool(v) for v in vals]
        return flags
