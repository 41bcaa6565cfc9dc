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
           "plan_mgwfbp_flags", "mgwfbp_merge_plan", "mgwfbp_schedule_time",
           "plan_asc_flags", "asc_merge_plan",
           "plan_mgs_flags", "mgs_merge_plan", "topk_compute_time",
           "topk_perf_model", "allgather_perf_model",
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


# Tensors below this many bytes always merge forward — a collective this
# small is pure startup cost (reference force-merges <8192 elements).
_TINY_BYTES = 8192 * 4


def mgwfbp_merge_plan(sizes_bytes: Sequence[int], tb: Sequence[float],
                      world: int, ab: Optional[AlphaBeta] = None) -> List[int]:
    """Core MG-WFBP decision (pure function, backward execution order).

    Inputs are per-layer payload bytes and backward-compute times, index 0 =
    the FIRST layer to finish backward.  Returns a group id per layer (group
    ids increase along backward order; layers in one group communicate as one
    merged collective when the group's last layer is ready).

    Model (reference __calculate_comm_start recurrence, wfbp/dopt.py:409-470):
    all collectives share one comm channel and issue in backward order, so
    comm of layer i starts at  start[i] = max(start[i-1] + t_c[i-1], ready[i]).
    Merging layer i into i+1 removes one startup ``alpha`` but delays i's
    payload until ready[i+1]; merge when the channel would otherwise sit idle
    (start[i] >= ready[i+1]) or the idle gap it closes is < alpha.
    """
    ab = ab or AlphaBeta()
    L = len(sizes_bytes)
    if L == 0:
        return []
    p = [float(s) for s in sizes_bytes]
    tc = [ab.allreduce_time(s, world) for s in p]
    ready = [0.0] * L
    acc = 0.0
    for i in range(L):
        acc += tb[i]
        ready[i] = acc

    def comm_starts():
        start = [0.0] * L
        start[0] = ready[0]
        for i in range(1, L):
            start[i] = max(start[i - 1] + tc[i - 1], ready[i])
        return start

    merged_next = [False] * L  # layer i joins layer i+1's group
    for i in range(L - 1):
        start = comm_starts()
        nxt_ready = ready[i + 1]
        do_merge = False
        if nxt_ready < start[i] + tc[i]:
            # i's comm would still be in flight when i+1 becomes ready
            if start[i] >= nxt_ready:
                do_merge = True       # comm hasn't even started: free merge
            elif (nxt_ready - start[i]) < ab.alpha:
                do_merge = True       # idle gap closed is cheaper than alpha
        if not do_merge and p[i] < _TINY_BYTES:
            do_merge = True           # startup-dominated tiny payload
        if do_merge:
            p[i + 1] += p[i]
            p[i] = 0.0
            tc[i] = 0.0
            tc[i + 1] = ab.allreduce_time(p[i + 1], world)
            merged_next[i] = True
    gids = [0] * L
    for i in range(1, L):
        gids[i] = gids[i - 1] + (0 if merged_next[i - 1] else 1)
    return gids


def mgwfbp_schedule_time(sizes_bytes: Sequence[int], tb: Sequence[float],
                         gids: Sequence[int], world: int,
                         ab: Optional[AlphaBeta] = None) -> float:
    """Predicted iteration span (backward start → last collective done) for a
    given grouping, same serialized-channel model as mgwfbp_merge_plan.  Used
    by the planner tests to check plans against exhaustive small-case optima.
    """
    ab = ab or AlphaBeta()
    L = len(sizes_bytes)
    if L == 0:
        return 0.0
    ready_layer = []
    acc = 0.0
    for i in range(L):
        acc += tb[i]
        ready_layer.append(acc)
    # group payloads and ready times (a group is ready at its LAST layer)
    n_groups = max(gids) + 1
    g_bytes = [0.0] * n_groups
    g_ready = [0.0] * n_groups
    for i in range(L):
        g_bytes[gids[i]] += sizes_bytes[i]
        g_ready[gids[i]] = max(g_ready[gids[i]], ready_layer[i])
    t = 0.0
    for g in range(n_groups):
        t = max(t, g_ready[g]) + ab.allreduce_time(g_bytes[g], world)
    return t


def asc_merge_plan(sizes_bytes: Sequence[int], tb: Sequence[float],
                   world: int, ab: Optional[AlphaBeta] = None) -> List[int]:
    """ASC plan (reference _generate_groups_asc, hv_distributed_optimizer.py
    in mgwfbp/): same serialized-channel model as MG-WFBP but merge layer i
    into i+1 ONLY when i's collective could not have started before i+1 is
    ready anyway (a strictly free merge) — no alpha-saving merges and no
    tiny-tensor force merge.  Backward execution order, index 0 first.
    """
    ab = ab or AlphaBeta()
    L = len(sizes_bytes)
    if L == 0:
        return []
    p = [float(s) for s in sizes_bytes]
    tc = [ab.allreduce_time(s, world) for s in p]
    ready = [0.0] * L
    acc = 0.0
    for i in range(L):
        acc += tb[i]
        ready[i] = acc
    merged_next = [False] * L
    for i in range(L - 1):
        start = [0.0] * L
        start[0] = ready[0]
        for j in range(1, L):
            start[j] = max(start[j - 1] + tc[j - 1], ready[j])
        if start[i] > ready[i + 1]:   # channel idle past i+1's readiness
            p[i + 1] += p[i]
            p[i] = 0.0
            tc[i] = 0.0
            tc[i + 1] = ab.allreduce_time(p[i + 1], world)
            merged_next[i] = True
    gids = [0] * L
    for i in range(1, L):
        gids[i] = gids[i - 1] + (0 if merged_next[i - 1] else 1)
    return gids


# fitted top-k kernel constant: t = TOPK_S * n * log2(n) (reference utils.py
# topk_perf_model; refit on MI355X via CommunicationProfiler if needed)
TOPK_S = 2.5e-10


def topk_compute_time(n_elems: float, s: float = TOPK_S) -> float:
    import math
    if n_elems <= 1:
        return 0.0
    return s * n_elems * math.log2(n_elems)


def mgs_merge_plan(sizes_elems: Sequence[int], tb: Sequence[float],
                   world: int, density: float,
                   ab: Optional[AlphaBeta] = None,
                   topk_s: float = TOPK_S) -> List[int]:
    """MGS-SGD plan (reference _generate_groups_mgs): sparsified sync where
    each group is top-k compressed on the compute stream right after its
    backward, then its (values, indices) are all-gathered on the comm channel.
    Merge layer i into i+1 when the extra wait tw (deferred backward + the
    merged top-k's superlinear cost − the channel idle gap that exists anyway)
    is smaller than the allgather startup saved tsave.  Backward order,
    index 0 first; 12 bytes/selected element (fp32 value + int64 index).
    """
    ab = ab or AlphaBeta()
    L = len(sizes_elems)
    if L == 0:
        return []
    p = [float(s) for s in sizes_elems]
    tbl = [float(t) for t in tb]

    def tk(n):
        return topk_compute_time(n, topk_s)

    def ag(n):
        if n <= 0:
            return 0.0
        return allgather_perf_model(int(max(n * density, 1)) * 12, world, ab)

    merged_next = [False] * L
    for i in range(L - 1):
        # compute-stream schedule: backward(i) then topk(i), serialized
        sparse_done = [0.0] * L
        t = 0.0
        for j in range(L):
            t += tbl[j] + tk(p[j])
            sparse_done[j] = t
        # comm channel start recurrence over allgathers
        start = [0.0] * L
        tc = [ag(p[j]) for j in range(L)]
        start[0] = sparse_done[0]
        for j in range(1, L):
            start[j] = max(start[j - 1] + tc[j - 1], sparse_done[j])
        idle_gap = start[i] - sparse_done[i]
        tw = tbl[i + 1] + tk(p[i] + p[i + 1]) - tk(p[i]) - tk(p[i + 1]) \
            - idle_gap
        tsave = ag(p[i]) + ag(p[i + 1]) - ag(p[i] + p[i + 1])
        if tw < tsave:
            p[i + 1] += p[i]
            p[i] = 0.0
            tbl[i + 1] += tbl[i]
            tbl[i] = 0.0
            merged_next[i] = True
    gids = [0] * L
    for i in range(1, L):
        gids[i] = gids[i - 1] + (0 if merged_next[i - 1] else 1)
    return gids


def _module_plan(model, backend, layerwise_times, planner):
    """Shared flag plumbing: run `planner(sizes_bw, taus_bw, world)` over the
    model's modules in backward order, map the group ids back to forward-order
    'start new group' flags, broadcast from rank 0."""
    from ..parallel.fusion import _module_param_order
    mods = _module_param_order(model)
    n = len(mods)
    if n == 0:
        return []
    world = backend.size if backend is not None else 1
    if layerwise_times is None:
        from ..profiling import Profiling
        layerwise_times = Profiling.estimate_backward_times(model)
    sizes_bw = [sum(p.numel() for _, p in ps) for _, ps in reversed(mods)]
    taus_bw = [layerwise_times.get(id(m), 1e-4)
               if isinstance(layerwise_times, dict) else 1e-4
               for m, _ in reversed(mods)]
    gids_bw = planner(sizes_bw, taus_bw, world)
    gid_fw = list(reversed(gids_bw))
    flags = [True] + [gid_fw[i] != gid_fw[i - 1] for i in range(1, n)]
    import torch.distributed as dist
    if backend is not None and world > 1 and dist.is_initialized():
        from .dist_helpers import bcast_floats
        flags = [bool(v) for v in bcast_floats([1.0 if f else 0.0
                                                for f in flags])]
    return flags


def plan_asc_flags(model: torch.nn.Module, backend=None,
                   layerwise_times: Optional[Dict[str, float]] = None,
                   ab: Optional[AlphaBeta] = None) -> List[bool]:
    """ASC merge plan → per-module 'start new group' flags (forward order)."""
    return _module_plan(model, backend, layerwise_times,
                        lambda s, t, w: asc_merge_plan([x * 4 for x in s],
                                                       t, w, ab))


def plan_mgs_flags(model: torch.nn.Module, backend=None,
                   layerwise_times: Optional[Dict[str, float]] = None,
                   density: float = 0.01,
                   ab: Optional[AlphaBeta] = None) -> List[bool]:
    """MGS merge plan (sparse allgather) → forward-order flags."""
    return _module_plan(model, backend, layerwise_times,
                        lambda s, t, w: mgs_merge_plan(s, t, w, density, ab))


def plan_mgwfbp_flags(model: torch.nn.Module, backend=None,
                      layerwise_times: Optional[Dict[str, float]] = None,
                      ab: Optional[AlphaBeta] = None) -> List[bool]:
    """MG-WFBP merge plan → per-module 'start new group' flags (forward order).

    Wraps mgwfbp_merge_plan (the reference's start-time recurrence,
    wfbp/dopt.py:409-470) over the model's modules; flags are broadcast from
    rank 0 so every rank builds identical buckets.
    """
    return _module_plan(model, backend, layerwise_times,
                        lambda s, t, w: mgwfbp_merge_plan([x * 4 for x in s],
                                                          t, w, ab))


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
