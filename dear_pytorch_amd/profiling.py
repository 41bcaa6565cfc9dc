"""Profiling subsystems (reference */profiling.py + chrome_profiler.py).

* ``Profiling`` — per-layer backward timing via parameter hooks; feeds the
  MG-WFBP planner (reference Profiling/benchmark(), profiling.py:11-129).
  On ROCm, per-hook host timestamps bracket ``torch.cuda.synchronize()``
  exactly like the reference; ``estimate_backward_times`` gives a cheap
  FLOP-proportional estimate when measuring is not worth it.
* ``CommunicationProfiler`` — message-size sweep over the live RCCL channel
  to fit the xGMI alpha-beta constants (reference profiling.py:132-165; the
  reference's Ethernet tables are replaced by this measurement).
* ``ChromeTracer`` — chrome://tracing event writer (reference
  chrome_profiler.py), enabled via the DEAR_TIMELINE env var.
"""
from __future__ import annotations

import json
import os
import threading
import time
from typing import Dict, List, Optional

import torch

__all__ = ["Profiling", "CommunicationProfiler", "ChromeTracer", "tracer"]


class Profiling:
    """Measure per-module backward time of one model by hooking parameters."""

    def __init__(self, model: torch.nn.Module):
        self.model = model
        self._t = {}
        self._times: Dict[int, List[float]] = {}
        self._order: List[int] = []
        self._handles = []
        from .parallel.fusion import _module_param_order
        self._mods = _module_param_order(model)
        for m, ps in self._mods:
            last_p = ps[-1][1]
            self._handles.append(
                last_p.register_hook(self._make_hook(id(m))))

    def _make_hook(self, mid):
        def hook(grad):
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            now = time.perf_counter()
            prev = self._t.get("last", None)
            if prev is not None:
                self._times.setdefault(mid, []).append(now - prev)
                if mid not in self._order:
                    self._order.append(mid)
            self._t["last"] = now
            return grad
        return hook

    def start_backward(self):
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        self._t["last"] = time.perf_counter()

    def layerwise_times(self) -> Dict[int, float]:
        """median seconds per module id (backward order discovery)."""
        import statistics
        return {mid: statistics.median(v) for mid, v in self._times.items()}

    def remove(self):
        for h in self._handles:
            h.remove()

    @staticmethod
    def estimate_backward_times(model: torch.nn.Module) -> Dict[int, float]:
        """Parameter-count-proportional backward-time estimate (no run)."""
        from .parallel.fusion import _module_param_order
        out = {}
        for m, ps in _module_param_order(model):
            n = sum(p.numel() for _, p in ps)
            out[id(m)] = max(n * 2e-11, 2e-6)  # ~50 GFLOP/s/param heuristic
        return out

    @staticmethod
    def benchmark(model, make_batch, loss_fn, warmup=5, iters=20):
        """Run warmup+iters backward passes and return
        (module ids in backward order, median per-layer times, sizes)."""
        prof = Profiling(model)
        for i in range(warmup + iters):
            if i == warmup:
                prof._times.clear()
                prof._order.clear()
            x, y = make_batch()
            model.zero_grad(set_to_none=True)
            prof.start_backward()
            loss_fn(model(x), y).backward()
        times = prof.layerwise_times()
        prof.remove()
        sizes = {id(m): sum(p.numel() for _, p in ps) * 4
                 for m, ps in prof._mods}
        return prof._order, times, sizes


class CommunicationProfiler:
    """Sweep all-reduce sizes over a CommBackend and fit alpha-beta."""

    def __init__(self, backend, sizes_bytes: Optional[List[int]] = None,
                 iters: int = 10):
        self.backend = backend
        # xGMI-relevant band: 64 KB .. 64 MB (the reference swept 8-512 KB
        # for Ethernet; xGMI startup amortizes later)
        self.sizes = sizes_bytes or [1 << s for s in range(16, 27)]
        self.iters = iters

    def benchmark(self):
        device = torch.device("cuda", torch.cuda.current_device()) \
            if torch.cuda.is_available() else torch.device("cpu")
        xs, ts = [], []
        for nbytes in self.sizes:
            n = nbytes // 4
            t = torch.ones(n, device=device)
            # warmup
            for _ in range(2):
                self.backend.all_reduce(t).host_wait()
            if device.type == "cuda":
                torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(self.iters):
                self.backend.all_reduce(t).host_wait()
            if device.type == "cuda":
                torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / self.iters
            xs.append(nbytes)
            ts.append(dt)
        return xs, ts

    def fit(self):
        from .utils.perf_model import fit_alpha_beta
        xs, ts = self.benchmark()
        return fit_alpha_beta(xs, ts)


class ChromeTracer:
    """Chrome trace-event JSON writer with a background flush thread
    (capability of reference chrome_profiler.py; backend-agnostic)."""

    def __init__(self, path: str):
        self.path = path
        self._events = []
        self._lock = threading.Lock()
        self._t0 = time.perf_counter()
        self._pid = os.getpid()

    def begin(self, name: str, cat: str = "op"):
        self._put(name, cat, "B")

    def end(self, name: str, cat: str = "op"):
        self._put(name, cat, "E")

    def instant(self, name: str, cat: str = "op"):
        self._put(name, cat, "i")

    def _put(self, name, cat, ph):
        ev = {"name": name, "cat": cat, "ph": ph,
              "ts": (time.perf_counter() - self._t0) * 1e6,
              "pid": self._pid, "tid": threading.get_ident() & 0xFFFF}
        with self._lock:
            self._events.append(ev)

    def save(self):
        with self._lock:
            with open(self.path, "w") as f:
                json.dump({"traceEvents": self._events}, f)

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.save()


def torch_profile(step_fn, path: str, steps: int = 3, warmup: int = 2):
    """Profile `steps` calls of step_fn with torch.profiler (CPU + HIP
    activities on ROCm) and export a chrome trace to `path` (the MI355X
    equivalent of the reference's nvprof scripts, horovod/prof.sh)."""
    from torch.profiler import profile, schedule, ProfilerActivity
    acts = [ProfilerActivity.CPU]
    if torch.cuda.is_available():
        acts.append(ProfilerActivity.CUDA)
    with profile(activities=acts,
                 schedule=schedule(wait=0, warmup=warmup, active=steps),
                 record_shapes=False) as prof:
        for _ in range(warmup + steps):
            step_fn()
            prof.step()
    prof.export_chrome_trace(path)
    return prof


_tracer: Optional[ChromeTracer] = None


def tracer() -> Optional[ChromeTracer]:
    """Process-wide tracer, enabled by DEAR_TIMELINE=<path> (the reference
    gates on WFSGD_TIMELINE via horovod_mpi_cj.sh)."""
    global _tracer
    if _tracer is None:
        path = os.environ.get("DEAR_TIMELINE", "")
        if path:
            rank = os.environ.get("RANK", "0")
            _tracer = ChromeTracer(path.replace("%r", rank))
    return _tracer
