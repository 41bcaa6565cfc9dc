"""Analytic communication-cost models + the MG-WFBP merge planner.

Reference capability: */utils.py (alpha-beta tables + closed-form predictors)
and the MG-WFBP planner (_generate_groups_mgwfbp, wfbp/dopt.py:380-486).
The reference hard-codes alpha-beta constants for 10GbE/56Gb-IB Ethernet
clusters (utils.py:62-88) — meaningless on xGMI.  Here the constants are
MEASURED on the actual fabric by CommunicationProfiler (profiling.py) and fit
by least squares; defaults below are xGMI-scale placeholders used only when
no measurement is available.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence

import torch

# xGMI-scale fallbacks (seconds, seconds/byte): ~15 us startup, ~100 GB/s
# effective ring bandwidth per link — replaced by measured fits at runtime.
DEFAULT_ALPHA = 15e-6
DEFAULT_BETA = 1.0 / (100e9)

__all__ = ["AlphaBeta", "fit_alpha_beta", "predict_allreduce_time",
           "plan_mgwfbp_flags", "topk_perf_model", "allgather_perf_model",
           "predict_density_with_size_and_computation",
           "gen_threshold_from_normal_distribution"]


class AlphaBeta:
    def __init__(self, alpha: float = DEFAULT_ALPHA, beta: float = DEFAULT_BETA):
        self.alpha = alpha
        self.beta = beta

    def allreduce_time(self, nbytes: int, world: int) -> float:
        # ring allreduce: 2(P-1)/P * bytes moved per link + startup
        if world <= 1:
            return 0.0
        return self.alpha + self.beta * (2.0 * (world - 1) / world) * nbytes


def fit_alpha_beta(sizes: Sequence[int], times: Sequence[float]) -> AlphaBeta:
    """Least-squares fit t = alpha + beta * size (reference fits with sklearn
    LinearRegression, wfbp/dopt.py:260-285; plain numpy here)."""
    import numpy as np
    A = np.vstack([np.ones(len(sizes)), np.asarray(sizes, dtype=float)]).T
    coef, *_ = np.linalg.lstsq(A, np.asarray(times, dtype=float), rcond=None)
    alpha = max(float(coef[0]), 1e-7)
    beta = max(float(coef[1]), 1e-13)
    return AlphaBeta(alpha, beta)


def predict_allreduce_time(nbytes: int, world: int,
                           ab: Optional[AlphaBeta] = None) -> float:
    return (ab or AlphaBeta()).allreduce_time(nbytes, world)


def plan_mgwfbp_flags(model: torch.nn.Module, backend=None,
                      layerwise_times: Optional[Dict[str, float]] = None,
                      ab: Optional[AlphaBeta] = None) -> List[bool]:
    """MG-WFBP merge plan → per-module 'start new group' flags (forward order).

    Algorithm (capability of reference _generate_groups_mgwfbp): walk modules
    in BACKWARD order with their measured backward-compute times tau_b; a
    layer's collective can start when its gradients are ready; merge layer l
    into the following group when the extra wait it causes is smaller than the
    saved startup alpha.  Flags are broadcast from rank 0 so every rank builds
    identical buckets.
    """
    from ..parallel.fusion import _module_param_order
    mods = _module_param_order(model)
    n = len(mods)
    if n == 0:
        return []
    ab = ab or AlphaBeta()
    world = backend.size if backend is not None else 1
    if layerwise_times is None:
        from ..profiling import Profiling
        layerwise_times = Profiling.estimate_backward_times(model)
    # backward order = reverse forward order
    sizes = [sum(p.numel() for _, p in ps) * 4 for _, ps in mods]
    taus = [layerwise_times.get(id(m), 1e-4) if isinstance(layerwise_times, dict)
            else 1e-4 for m, _ in mods]
    flags = [False] * n  # True = start new group at module i (forward order)
    flags[0] = True
    # Greedy backward-order merge: accumulate bytes; a new group starts (in
    # backward order) when the accumulated communication would overlap worse
    # than paying a fresh startup — i.e. when comm time for the merged group
    # exceeds the backward compute time remaining to hide it by more than
    # alpha.
    acc_bytes = 0
    remaining_tau = 0.0
    for i in range(n - 1, 0, -1):  # backward order, boundary decided at i
        acc_bytes += sizes[i]
        remaining_tau += taus[i]
        t_merged = ab.allreduce_time(acc_bytes + sizes[i - 1], world)
        t_split = ab.allreduce_time(acc_bytes, world) + \
            ab.allreduce_time(sizes[i - 1], world)
        # force-merge tiny tensors (reference merges <8192 elems, :467)
        if sizes[i - 1] < 8192 * 4:
            continue
        if t_merged > t_split - ab.alpha + remaining_tau:
            flags[i] = True
            acc_bytes = 0
            remaining_tau = 0.0
    if backend is not None and world > 1:
        from .dist_helpers import bcast_floats
        flags = [bool(v) for v in bcast_floats([1.0 if f else 0.0
                                                for f in flags])]
    return flags


# ---- closed-form models carried over for the sparsification planner --------
def topk_perf_model(k: int, world: int, ab: Optional[AlphaBeta] = None) -> float:
    ab = ab or AlphaBeta()
    if world <= 1:
        return 0.0
    import math
    rounds = math.ceil(math.log2(max(world, 2)))
    return rounds * (ab.alpha + ab.beta * k * 12)  # values+indices per round


def allgather_perf_model(nbytes: int, world: int,
                         ab: Optional[AlphaBeta] = None) -> float:
    ab = ab or AlphaBeta()
    if world <= 1:
        return 0.0
    return ab.alpha + ab.beta * (world - 1) * nbytes


def predict_density_with_size_and_computation(nbytes: int, tau_b: float,
                                              world: int,
                                              ab: Optional[AlphaBeta] = None
                                              ) -> float:
    """Density at which sparse allgather time ≈ hidden backward time."""
    ab = ab or AlphaBeta()
    if world <= 1:
        return 1.0
    budget = max(tau_b - ab.alpha, 1e-6)
    k_bytes = budget / (ab.beta * (world - 1) * 3)  # idx+val overhead ~3x
    return float(min(max(k_bytes / max(nbytes, 1), 1e-4), 1.0))


def gen_threshold_from_normal_distribution(p_value: float, mu: float,
                                           sigma: float) -> float:
    """|x| threshold keeping fraction p_value of a N(mu, sigma) population."""
    from scipy import stats
    left = stats.norm.ppf((1 - p_value) / 2, mu, sigma)
    right = stats.norm.ppf(1 - (1 - p_value) / 2, mu, sigma)
    return max(abs(left), abs(right))
