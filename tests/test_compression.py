"""Compression codecs + gTopK sparse allreduce (CPU; gloo for the collective)."""
import pytest
import torch

from dear_pytorch_amd import compression as C
from utils_dist import run_dist


def test_registry():
    for k in ("none", "topk", "eftopk", "gaussian", "sign", "efsign"):
        assert k in C.compressors


def test_topk_selects_largest_and_keeps_residual():
    comp = C.TopKCompressor()
    t = torch.tensor([0.1, -5.0, 0.2, 3.0, -0.05, 0.0])
    out, (vals, idx) = comp.compress(t.clone(), name="a", ratio=0.34)
    assert set(idx.tolist()) == {1, 3}
    assert out[1] == -5.0 and out[3] == 3.0
    assert out[0] == 0.0
    r = comp.residuals["a"]
    assert r[1] == 0.0 and abs(r[0] - 0.1) < 1e-7


def test_topk_error_feedback_accumulates():
    comp = C.TopKCompressor()
    t1 = torch.tensor([1.0, 0.4, 0.0, 0.0])
    comp.compress(t1.clone(), name="a", ratio=0.25)
    # residual 0.4 at pos 1; next round small grads + residual should win
    t2 = torch.tensor([0.0, 0.4, 0.1, 0.0])
    out, (vals, idx) = comp.compress(t2.clone(), name="a", ratio=0.25)
    assert idx.tolist() == [1]
    assert abs(out[1] - 0.8) < 1e-6


def test_sign_compressor_scale():
    comp = C.SignCompressor()
    t = torch.tensor([2.0, -1.0, 0.5, -0.5])
    out, scale = comp.compress(t.clone())
    assert abs(scale - 1.0) < 1e-6
    assert torch.allclose(out.abs(), torch.full((4,), 1.0))
    assert torch.equal(torch.sign(out), torch.sign(t))


def test_sign_bit_pack_roundtrip():
    g = torch.Generator().manual_seed(0)
    signs = torch.sign(torch.randn(100, generator=g))
    signs[signs == 0] = 1
    packed = C.SignCompressor.pack_bits(signs)
    assert packed.numel() == 13
    back = C.SignCompressor.unpack_bits(packed, 100)
    assert torch.equal(back, signs)


def test_efsign_residual():
    comp = C.EFSignCompressor()
    t = torch.tensor([2.0, -0.1])
    out, scale = comp.compress(t.clone(), name="x")
    r = comp.residuals["x"]
    assert torch.allclose(out + r, t, atol=1e-6)


def test_gaussian_compressor_approx_k():
    comp = C.GaussianCompressor()
    g = torch.Generator().manual_seed(1)
    t = torch.randn(10000, generator=g)
    out, (vals, idx) = comp.compress(t.clone(), name="g", ratio=0.01)
    assert 1 <= idx.numel() <= 200  # ~100 +- search tolerance


def _rank_gtopk(rank, world):
    import dear_pytorch_amd as dear
    dear.init(backend="gloo")
    from dear_pytorch_amd.comm.backend import create_backend
    be = create_backend("gtopk")
    torch.manual_seed(100 + rank)
    flat = torch.randn(64)
    dense, vals, idx = C.gtopk_sparse_allreduce(be, flat, k=8)
    # result must be identical on all ranks and have <= 8 nonzeros
    n_nonzero = int((dense != 0).sum())
    out = (dense.clone(), n_nonzero)
    dear.shutdown()
    return out


@pytest.mark.timeout(300)
def test_gtopk_ws2():
    outs = run_dist(_rank_gtopk, world_size=2)
    (d0, n0), (d1, n1) = outs
    assert torch.equal(d0, d1)
    assert 1 <= n0 <= 8 and n0 == n1


def test_topk_impl_logic_cpu_stub():
    """The native top-k's search/trim logic vs torch.topk, kernels stubbed."""
    from dear_pytorch_amd.ops.topk import _topk_abs_impl, _TorchKernels
    g = torch.Generator().manual_seed(4)
    for n, k in [(1000, 10), (1000, 1), (64, 64), (64, 63), (5000, 500)]:
        x = torch.randn(n, generator=g)
        vals, idx = _topk_abs_impl(x, k, _TorchKernels)
        assert vals.numel() == k and idx.numel() == k
        assert torch.equal(x[idx], vals)
        ref_vals, _ = torch.topk(x.abs(), k, sorted=True)
        got = vals.abs().sort(descending=True).values
        assert torch.allclose(got, ref_vals), (n, k)
    # all-zero input
    z = torch.zeros(100)
    vals, idx = _topk_abs_impl(z, 5, _TorchKernels)
    assert vals.numel() == 5
    # duplicates / ties
    t = torch.ones(50)
    vals, idx = _topk_abs_impl(t, 7, _TorchKernels)
    assert vals.numel() == 7 and (vals == 1).all()


def test_error_feedback_conserves_signal():
    """Property: for the EF codecs, (sent values) + (stored residual) ==
    (input + previous residual) — nothing is lost, only deferred."""
    torch.manual_seed(3)
    from dear_pytorch_amd.compression import TopKCompressor, EFSignCompressor
    c = TopKCompressor()
    prev_resid = torch.zeros(1000)
    for step in range(4):
        g = torch.randn(1000)
        expect_total = g + prev_resid
        work = g.clone()
        _, (vals, idx) = c.compress(work, name="p", ratio=0.1)
        resid = c.residuals["p"]
        total = resid.clone()
        total[idx] += vals
        assert torch.allclose(total, expect_total, atol=1e-6)
        prev_resid = resid.clone()
    s = EFSignCompressor()
    prev = torch.zeros(500)
    for step in range(3):
        g = torch.randn(500)
        expect = g + prev
        work = g.clone()
        _, scale = s.compress(work, name="q")
        # quantized output + residual == accumulated input
        assert torch.allclose(work.view(-1) + s.residuals["q"], expect,
                              atol=1e-6)
        prev = s.residuals["q"].clone()


def test_factory_accepts_reference_kwargs():
    """Horovod-shaped factory signature parity: named_parameters/compression
    accepted (reference dopt_rsag.py:377-394)."""
    import torch.nn as nn
    import dear_pytorch_amd as dear
    m = nn.Linear(8, 4)
    opt = dear.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.1), model=m,
        named_parameters=m.named_parameters(), compression=None)
    x = torch.randn(2, 8)
    nn.functional.mse_loss(m(x), torch.zeros(2, 4)).backward()
    opt.step()
    opt.synchronize()
