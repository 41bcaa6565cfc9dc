"""Baseline gradient-sync methods on the same RCCL path as DeAR.

Reference parity targets: WFBP (wfbp/dopt.py), MG-WFBP
(mgwfbp/dopt.py + the alpha-beta merge planner), naive per-tensor RS/AG
(dear/dopt_rsag_naive.py) and the reduce+broadcast decomposition
(dear/dopt_rb.py).  All share the fusion layer and comm backends.
"""
from __future__ import annotations

from .wfbp import WfbpOptimizer
from .naive import NaiveDearOptimizer
from .rb import ReduceBcastOptimizer
from .bytescheduler import ByteSchedulerOptimizer

__all__ = ["make"]


def make(method: str, optimizer, model, threshold_bytes=None, **kw):
    method = method.lower()
    if method == "wfbp":
        # classic WFBP: per-layer (no fusion) all-reduce from backward hooks
        return WfbpOptimizer(optimizer, model, threshold_bytes=None, **kw)
    if method == "wfbp-fused":
        return WfbpOptimizer(optimizer, model, threshold_bytes=threshold_bytes,
                             **kw)
    if method == "mgwfbp":
        return WfbpOptimizer(optimizer, model, threshold_bytes=threshold_bytes,
                             mode="mgwfbp", **kw)
    if method == "asc":
        # merge only when the comm channel would idle (reference
        # _generate_groups_asc, hv_distributed_optimizer.py:353-428)
        return WfbpOptimizer(optimizer, model, threshold_bytes=threshold_bytes,
                             mode="asc", **kw)
    if method == "mgs":
        # merged gradient sparsification: top-k + sparse allgather planning
        # (reference _generate_groups_mgs, hv_distributed_optimizer.py:430-508)
        kw.setdefault("compressor", "topk")
        kw.setdefault("density", 0.01)
        return WfbpOptimizer(optimizer, model, threshold_bytes=threshold_bytes,
                             mode="mgs", **kw)
    if method == "naive":
        return NaiveDearOptimizer(optimizer, model, **kw)
    if method == "bytescheduler":
        return ByteSchedulerOptimizer(optimizer, model, **kw)
    if method == "rb":
        return ReduceBcastOptimizer(optimizer, model,
                                    threshold_bytes=threshold_bytes, **kw)
    raise KeyError(f"unknown method {method}")
