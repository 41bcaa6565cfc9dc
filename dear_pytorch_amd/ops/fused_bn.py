"""Fused NHWC BatchNorm(+ReLU)(+residual) training op (CDNA4 kernels).

``FusedBNAct2d`` subclasses nn.BatchNorm2d (state-dict compatible) and fuses
the activation and residual add into the normalization pass, replacing the
MIOpen BN kernel stack + separate relu/add elementwise kernels in the
ResNet/DenseNet hot path (csrc/bn_kernels.hip; motivation in
profiles/README.md).

Fast path requires: CUDA + fp32 + channels_last.  Anything else falls back to
a torch composition with IDENTICAL numerics (so models stay CPU-testable and
eval/fp16 paths just work).
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

__all__ = ["FusedBNAct2d"]


def _nparts(rows: int, C: int) -> int:
    import dear_pytorch_amd._kernels as K
    return int(K.bn_nparts(rows, C))


class _Workspace:
    """Per-(module, input-shape) scratch reused every iteration: stats
    partials + mean/invstd.  Gradients are NOT cached (autograd may adopt the
    returned tensors as .grad)."""

    __slots__ = ("mean", "invstd", "psum", "psumsq", "nparts", "rows")

    def __init__(self, rows, C, device):
        self.rows = rows
        self.nparts = _nparts(rows, C)
        self.mean = torch.empty(C, device=device, dtype=torch.float32)
        self.invstd = torch.empty_like(self.mean)
        self.psum = torch.empty(self.nparts, C, device=device,
                                dtype=torch.float32)
        self.psumsq = torch.empty_like(self.psum)


class _FusedBNFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var, training,
                momentum, eps, relu, residual, ws):
        import dear_pytorch_amd._kernels as K
        N, C, H, W = x.shape
        rows = N * H * W
        y = torch.empty_like(x)
        if training:
            mean, invstd = ws.mean, ws.invstd
        else:
            mean = running_mean
            invstd = torch.rsqrt(running_var + eps)
        K.bn_fwd(x, residual, y, weight, bias, mean, invstd,
                 running_mean, running_var, ws.psum, ws.psumsq, rows, C,
                 eps, momentum, training, relu)
        # save CLONES of the per-channel stats (C floats, negligible): the
        # workspace tensors are overwritten by the next forward, which would
        # silently corrupt a backward that runs after it (multi-forward
        # patterns like gradient accumulation)
        ctx.save_for_backward(x, y, weight, bias, mean.clone(), invstd.clone())
        ctx.relu = relu
        ctx.has_res = residual is not None
        ctx.dims = (rows, C)
        ctx.ws = ws
        return y

    @staticmethod
    def backward(ctx, dy):
        import dear_pytorch_amd._kernels as K
        x, y, weight, bias, mean, invstd = ctx.saved_tensors
        rows, C = ctx.dims
        ws = ctx.ws
        dy = dy.contiguous(memory_format=torch.channels_last)
        dx = torch.empty_like(x)
        dgamma = torch.empty(C, device=x.device, dtype=torch.float32)
        dbeta = torch.empty_like(dgamma)
        dy_eff = torch.empty_like(x) if ctx.has_res else None
        # partial buffers reused from forward (forward's reduce consumed them)
        K.bn_bwd(x, dy, y, dy_eff, weight, bias, mean, invstd, ws.psum,
                 ws.psumsq, dbeta, dgamma, dx, rows, C, ctx.relu, ctx.has_res)
        d_res = dy_eff if ctx.has_res else None
        return (dx, dgamma, dbeta, None, None, None, None, None, None, d_res,
                None)


class FusedBNAct2d(nn.BatchNorm2d):
    """BatchNorm2d with optional fused ReLU and fused residual add.

    forward(x, residual=None) computes
        act(bn(x) + residual)   with act = ReLU if relu else identity
    """

    # (C, H*W) shapes dispatched to the MIOpen composition instead of the
    # fused kernels.  Empty since the two-stage partial reduce + recomputed
    # ReLU mask made the fused path win on every ResNet-50 shape
    # (profiles/README.md r2 history: with the r1 kernels C512@28^2 and
    # C1024@14^2 lost in backward and were skipped here).  Override with
    # DEAR_BN_SKIP="C:HW,C:HW" for empirical per-shape dispatch.
    _skip_shapes = frozenset()

    def __init__(self, num_features, relu=False, **kw):
        super().__init__(num_features, **kw)
        self.relu = relu
        self._ws = {}  # rows -> _Workspace (variable batch sizes alternate)
        import os
        env = os.environ.get("DEAR_BN_SKIP")
        if env is not None:
            self._skip_shapes = frozenset(
                () if env in ("", "none") else
                (tuple(map(int, p.split(":"))) for p in env.split(",")))

    def _fast_ok(self, x, residual):
        return (x.is_cuda and x.dtype == torch.float32
                and x.shape[1] % 4 == 0
                and (x.shape[1], x.shape[2] * x.shape[3])
                not in self._skip_shapes
                and self.affine and self.track_running_stats
                and x.is_contiguous(memory_format=torch.channels_last)
                and (residual is None or
                     residual.is_contiguous(memory_format=torch.channels_last)))

    def forward(self, x, residual: Optional[torch.Tensor] = None):
        if self._fast_ok(x, residual):
            if self.training and self.num_batches_tracked is not None:
                self.num_batches_tracked += 1
            mom = self.momentum if self.momentum is not None else 0.1
            N, C, H, W = x.shape
            rows = N * H * W
            ws = self._ws.get(rows)
            if ws is None or ws.mean.device != x.device:
                ws = _Workspace(rows, C, x.device)
                self._ws[rows] = ws
            return _FusedBNFn.apply(x, self.weight, self.bias,
                                    self.running_mean, self.running_var,
                                    self.training, mom, self.eps, self.relu,
                                    residual, ws)
        # reference-numerics fallback (CPU, fp16, NCHW, no-affine...)
        y = super().forward(x)
        if residual is not None:
            y = y + residual
        return F.relu(y) if self.relu else y
