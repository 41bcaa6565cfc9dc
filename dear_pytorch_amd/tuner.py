"""Online Bayesian optimization of the fusion-buffer threshold.

Reference capability: dear/tuner.py (bayes_opt BayesianOptimization + EI
acquisition, 5-iteration timing windows, 10 trials then lock best) and
dear/dopt_rsag_bo.py (regroup with the new threshold between step() and the
next forward).  `bayes_opt` is not in this environment; the GP + expected
improvement loop is implemented on sklearn's GaussianProcessRegressor.

Usage:
    opt = dear.DistributedOptimizer(inner, model=model)
    tuned = ThresholdTuner(opt, bounds_mb=(1, 256))
    ... per iteration: tuned.step_begin(); train step; tuned.step_end()
"""
from __future__ import annotations

import time
from typing import Optional, Tuple

import numpy as np

__all__ = ["BayesOpt", "ThresholdTuner"]


class BayesOpt:
    """Minimal GP-EI Bayesian optimizer over a 1-D box."""

    def __init__(self, bounds: Tuple[float, float], seed: int = 17,
                 xi: float = 0.01):
        from sklearn.gaussian_process import GaussianProcessRegressor
        from sklearn.gaussian_process.kernels import Matern, WhiteKernel
        self.bounds = bounds
        self.xi = xi
        self.rng = np.random.RandomState(seed)
        self.gp = GaussianProcessRegressor(
            kernel=Matern(nu=2.5, length_scale=(bounds[1] - bounds[0]) / 4)
            + WhiteKernel(1e-6), normalize_y=True, alpha=1e-6,
            n_restarts_optimizer=2, random_state=seed)
        self.X: list[float] = []
        self.Y: list[float] = []

    def register(self, x: float, y: float):
        self.X.append(float(x))
        self.Y.append(float(y))

    def suggest(self) -> float:
        lo, hi = self.bounds
        if len(self.X) < 3:  # bootstrap: spread over the box
            probes = [lo + (hi - lo) * f for f in (0.1, 0.5, 0.9)]
            return probes[len(self.X)]
        X = np.array(self.X)[:, None]
        self.gp.fit(X, np.array(self.Y))
        cand = self.rng.uniform(lo, hi, 256)[:, None]
        mu, sd = self.gp.predict(cand, return_std=True)
        best = max(self.Y)
        with np.errstate(divide="ignore", invalid="ignore"):
            z = (mu - best - self.xi) / np.maximum(sd, 1e-9)
            from scipy.stats import norm
            ei = (mu - best - self.xi) * norm.cdf(z) + sd * norm.pdf(z)
        return float(cand[int(np.argmax(ei)), 0])

    def best(self) -> Tuple[float, float]:
        i = int(np.argmax(self.Y))
        return self.X[i], self.Y[i]


class ThresholdTuner:
    """Drives BayesOpt over the DeAR fusion threshold (MB), measuring mean
    iteration time over `window`-step windows (reference: 5-step windows,
    tuner.py:9; >= `warmup` discarded first; `trials` then lock)."""

    def __init__(self, opt, bounds_mb=(1.0, 256.0), window: int = 5,
                 warmup: int = 10, trials: int = 10, verbose: bool = True):
        self.opt = opt
        self.bo = BayesOpt(bounds_mb)
        self.window = window
        self.warmup = warmup
        self.trials = trials
        self.verbose = verbose and opt.rank == 0
        self._iter = 0
        self._win_t0: Optional[float] = None
        self._win_times: list[float] = []
        self._cur_mb = (opt.threshold_bytes or 25 << 20) / (1 << 20)
        self._done = False
        self._trial = 0
        self._pending_mb: Optional[float] = None

    @property
    def locked(self) -> bool:
        return self._done

    def step_begin(self):
        # regroup window: between step() and the next forward (reference
        # dopt_rsag_bo.py:148-171) — i.e. before this iteration's forward.
        if self._pending_mb is not None:
            mb = self._sync_threshold(self._pending_mb)
            self.opt.regroup(int(mb * (1 << 20)))
            self._cur_mb = mb
            self._pending_mb = None
            if self.verbose:
                print(f"[dear-bo] trial {self._trial}: threshold {mb:.1f} MB",
                      flush=True)
        self._t0 = time.perf_counter()

    def step_end(self):
        self._iter += 1
        if self._done or self._iter <= self.warmup:
            return
        self._win_times.append(time.perf_counter() - self._t0)
        if len(self._win_times) < self.window:
            return
        mean_t = float(np.mean(self._win_times[1:]))  # drop window head
        self._win_times.clear()
        self.bo.register(self._cur_mb, -mean_t)
        self._trial += 1
        if self._trial >= self.trials:
            best_mb, best_y = self.bo.best()
            self._pending_mb = best_mb
            self._done = True
            if self.verbose:
                print(f"[dear-bo] locked threshold {best_mb:.1f} MB "
                      f"({-best_y * 1e3:.2f} ms/iter)", flush=True)
        else:
            self._pending_mb = self.bo.suggest()

    def _sync_threshold(self, mb: float) -> float:
        """Rank-0's threshold wins (reference bcasts via MPI,
        dopt_rsag_bo.py:153)."""
        import torch.distributed as dist
        if dist.is_initialized() and dist.get_world_size() > 1:
            from .utils.dist_helpers import bcast_floats
            return float(bcast_floats([mb])[0])
        return mb
