"""Gradient compression codecs + the gTopK recursive sparse all-reduce.

Reference capability: */compression.py (NoneCompressor, TopKCompressor,
EFTopKCompressor, SignCompressor, EFSignCompressor, GaussianCompressor with
error feedback) and wfbp/dopt.py:50-106 (gtopk_sparse_recursive_allreduce,
which needed the missing native `tcmm.f_topk`; our device top-k lives in
csrc/kernels.hip via ops/topk).

All compressors operate on flat fp32 tensors and keep per-name residuals for
error feedback.
"""
from __future__ import annotations

from typing import Dict, Tuple

import torch

__all__ = ["compressors", "NoneCompressor", "TopKCompressor",
           "EFTopKCompressor", "GaussianCompressor", "SignCompressor",
           "EFSignCompressor", "gtopk_sparse_allreduce"]


def _topk_abs(t: torch.Tensor, k: int) -> Tuple[torch.Tensor, torch.Tensor]:
    """values+indices of the k largest |t|; native select kernel on GPU."""
    k = max(1, min(k, t.numel()))
    if t.is_cuda:
        from .ops.topk import topk_abs_native
        return topk_abs_native(t, k)
    _, idx = torch.topk(t.abs(), k, sorted=False)
    return t[idx], idx


class NoneCompressor:
    residuals: Dict[str, torch.Tensor] = {}

    @staticmethod
    def compress(tensor, name=None, ratio=1.0):
        return tensor, None

    @staticmethod
    def decompress(tensor, ctx, name=None):
        return tensor


class TopKCompressor:
    """Keep top-k |g| with residual error feedback accumulated BEFORE
    selection (reference TopKCompressor.compress, compression.py:56)."""

    def __init__(self):
        self.residuals: Dict[str, torch.Tensor] = {}
        self.zc = None

    def clear(self):
        self.residuals.clear()

    def compress(self, tensor: torch.Tensor, name: str = "g", ratio=0.01):
        with torch.no_grad():
            flat = tensor.view(-1)
            r = self.residuals.get(name)
            if r is not None:
                flat.add_(r)
            k = max(int(flat.numel() * ratio), 1)
            vals, idx = _topk_abs(flat, k)
            resid = flat.clone()
            resid[idx] = 0.0
            self.residuals[name] = resid
            out = torch.zeros_like(flat)
            out[idx] = vals
            tensor.view(-1).copy_(out)
            return tensor, (vals, idx)

    @staticmethod
    def decompress(tensor, ctx, name=None):
        return tensor


class EFTopKCompressor(TopKCompressor):
    """Error-feedback top-k: residual = what was NOT sent (same as TopK here;
    the reference's EF variant accumulates before compress, ours already
    does)."""


class GaussianCompressor:
    """Threshold-search top-k: model |g| ~ N(mu, sigma), pick the threshold
    keeping ~ratio of elements, refine by counting (reference
    GaussianCompressor, compression.py:210).  Counting uses the native
    count_ge kernel on GPU — no sort."""

    def __init__(self):
        self.residuals: Dict[str, torch.Tensor] = {}

    def clear(self):
        self.residuals.clear()

    def compress(self, tensor, name="g", ratio=0.01):
        with torch.no_grad():
            flat = tensor.view(-1)
            r = self.residuals.get(name)
            if r is not None:
                flat.add_(r)
            k = max(int(flat.numel() * ratio), 1)
            from .utils.perf_model import gen_threshold_from_normal_distribution
            mu, sigma = float(flat.mean()), float(flat.std())
            thr = gen_threshold_from_normal_distribution(
                1 - ratio, mu, max(sigma, 1e-12))
            for _ in range(3):  # refine: halve/expand threshold to hit ~k
                n_ge = int((flat.abs() >= thr).sum())
                if n_ge <= k * 2 and n_ge >= max(k // 2, 1):
                    break
                thr *= 1.3 if n_ge > 2 * k else 0.7
            mask = flat.abs() >= thr
            idx = mask.nonzero(as_tuple=False).view(-1)[:k]
            vals = flat[idx]
            resid = flat.clone()
            resid[idx] = 0.0
            self.residuals[name] = resid
            out = torch.zeros_like(flat)
            out[idx] = vals
            tensor.view(-1).copy_(out)
            return tensor, (vals, idx)

    @staticmethod
    def decompress(tensor, ctx, name=None):
        return tensor


class SignCompressor:
    """1-bit sign quantization with scale = mean |g| (reference SignCompressor
    bit-packing, compression.py:112)."""

    def __init__(self):
        self.residuals: Dict[str, torch.Tensor] = {}

    def compress(self, tensor, name="g", ratio=None):
        with torch.no_grad():
            flat = tensor.view(-1)
            scale = flat.abs().mean()
            signs = torch.sign(flat)
            signs[signs == 0] = 1.0
            tensor.view(-1).copy_(signs * scale)
            return tensor, scale

    @staticmethod
    def decompress(tensor, ctx, name=None):
        return tensor

    @staticmethod
    def pack_bits(signs: torch.Tensor) -> torch.Tensor:
        """sign vector (+-1) -> packed uint8 bitmap (wire format)."""
        bits = (signs > 0).to(torch.uint8)
        pad = (-bits.numel()) % 8
        if pad:
            bits = torch.cat([bits, bits.new_zeros(pad)])
        b = bits.view(-1, 8)
        weights = torch.tensor([1, 2, 4, 8, 16, 32, 64, 128],
                               dtype=torch.uint8, device=bits.device)
        return (b * weights).sum(1, dtype=torch.int64).to(torch.uint8)

    @staticmethod
    def unpack_bits(packed: torch.Tensor, n: int) -> torch.Tensor:
        weights = torch.tensor([1, 2, 4, 8, 16, 32, 64, 128],
                               dtype=torch.uint8, device=packed.device)
        bits = (packed[:, None] & weights) > 0
        signs = bits.view(-1)[:n].to(torch.float32) * 2 - 1
        return signs


class EFSignCompressor(SignCompressor):
    def __init__(self):
        super().__init__()

    def compress(self, tensor, name="g", ratio=None):
        with torch.no_grad():
            flat = tensor.view(-1)
            r = self.residuals.get(name)
            if r is not None:
                flat.add_(r)
            scale = flat.abs().mean()
            signs = torch.sign(flat)
            signs[signs == 0] = 1.0
            q = signs * scale
            self.residuals[name] = flat - q
            tensor.view(-1).copy_(q)
            return tensor, scale


compressors = {
    "none": NoneCompressor,
    None: NoneCompressor,
    "topk": TopKCompressor,
    "eftopk": EFTopKCompressor,
    "gaussian": GaussianCompressor,
    "sign": SignCompressor,
    "efsign": EFSignCompressor,
}


def gtopk_sparse_allreduce(backend, flat: torch.Tensor, k: int,
                           name: str = "g"):
    """Global top-k sparse all-reduce by recursive halving (reference
    gtopk_sparse_recursive_allreduce, wfbp/dopt.py:50-106): log2(P) rounds of
    value/index sendrecv, scatter-add, re-top-k.  Requires power-of-two world
    size (as the reference does).  Returns (dense result, values, indices)."""
    P, rank = backend.size, backend.rank
    if P == 1:
        vals, idx = _topk_abs(flat.view(-1), k)
        return flat, vals, idx
    assert P & (P - 1) == 0, "gtopk needs power-of-two world size"
    vals, idx = _topk_abs(flat.view(-1), k)
    vals = vals.clone()
    idx = idx.clone()
    dense = torch.zeros_like(flat.view(-1))
    dense[idx] = vals
    span = 1
    while span < P:
        peer = rank ^ span
        rv = torch.empty_like(vals)
        ri = torch.empty_like(idx)
        backend.send_recv(vals, rv, peer).host_wait()
        backend.send_recv(idx, ri, peer).host_wait()
        dense.scatter_add_(0, ri, rv)
        vals, idx = _topk_abs(dense, k)
        vals = vals.clone()
        idx = idx.clone()
        keep = torch.zeros_like(dense)
        keep[idx] = vals
        dense = keep
        span <<= 1
    return dense.view_as(flat), vals, idx
