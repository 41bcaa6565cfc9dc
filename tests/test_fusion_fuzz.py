"""Property-based tests: bucket planning invariants over random module trees
and DeAR-vs-serial equivalence over random tiny models."""
import hypothesis
from hypothesis import given, settings, strategies as st

import torch
import torch.nn as nn

import dear_pytorch_amd as dear
from dear_pytorch_amd.parallel.fusion import build_groups, ALIGN_ELEMS


def _random_model(widths, seed):
    torch.manual_seed(seed)
    layers = []
    prev = widths[0]
    for w in widths[1:]:
        layers.append(nn.Linear(prev, w))
        layers.append(nn.ReLU())
        prev = w
    return nn.Sequential(*layers), widths[0], prev


@settings(max_examples=25, deadline=None)
@given(widths=st.lists(st.integers(1, 40), min_size=2, max_size=8),
       threshold=st.one_of(st.none(), st.integers(1, 1 << 16)),
       seed=st.integers(0, 1000))
def test_plan_invariants(widths, threshold, seed):
    m, _, _ = _random_model(widths, seed)
    groups = build_groups(m, threshold_bytes=threshold)
    # every trainable param exactly once, forward order preserved
    planned = [s.param for g in groups for s in g.slots]
    expect = [p for p in m.parameters() if p.requires_grad]
    assert len(planned) == len(expect)
    assert all(a is b for a, b in zip(planned, expect))
    # offsets aligned and non-overlapping within each group
    for g in groups:
        end = 0
        for s in g.slots:
            assert s.offset % ALIGN_ELEMS == 0
            assert s.offset >= end
            end = s.offset + s.numel
        assert g.numel == end
    # allocation: padded buckets divide evenly for any world size
    for P in (1, 2, 3, 8):
        gs = build_groups(m, threshold_bytes=threshold)
        for g in gs:
            g.allocate(P, torch.device("cpu"))
            assert g.padded % P == 0
            assert g.shard.numel() * P == g.padded
            g.free()


@settings(max_examples=10, deadline=None)
@given(widths=st.lists(st.integers(2, 24), min_size=2, max_size=5),
       threshold=st.one_of(st.none(), st.integers(64, 1 << 14)),
       seed=st.integers(0, 100))
def test_dear_equals_serial_random_models(widths, threshold, seed):
    ma, cin, cout = _random_model(widths, seed)
    mb, _, _ = _random_model(widths, seed)
    g = torch.Generator().manual_seed(seed + 1)
    data = [(torch.randn(4, cin, generator=g),
             torch.randn(4, cout, generator=g)) for _ in range(3)]
    oa = torch.optim.SGD(ma.parameters(), lr=0.05, momentum=0.9)
    for x, y in data:
        oa.zero_grad()
        nn.functional.mse_loss(ma(x), y).backward()
        oa.step()
    ob = dear.DistributedOptimizer(
        torch.optim.SGD(mb.parameters(), lr=0.05, momentum=0.9), model=mb,
        threshold_bytes=threshold)
    for x, y in data:
        ob.zero_grad()
        nn.functional.mse_loss(mb(x), y).backward()
        ob.step()
    ob.synchronize()
    for (n, pa), (_, pb) in zip(ma.named_parameters(), mb.named_parameters()):
        assert torch.allclose(pa, pb, atol=1e-6), n
