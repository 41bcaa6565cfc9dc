"""Bucket-plan and grad-as-bucket-view unit tests (CPU)."""
import torch
import torch.nn as nn

from dear_pytorch_amd.parallel.fusion import build_groups, ALIGN_ELEMS


def _mlp(widths=(32, 64, 64, 16)):
    layers = []
    for a, b in zip(widths[:-1], widths[1:]):
        layers += [nn.Linear(a, b), nn.ReLU()]
    return nn.Sequential(*layers)


def test_threshold_grouping_covers_all_params():
    m = _mlp()
    groups = build_groups(m, threshold_bytes=64 * 4 * 10)  # tiny threshold
    names = [s.name for g in groups for s in g.slots]
    assert len(names) == len(set(names))
    assert set(names) == {n for n, p in m.named_parameters() if p.requires_grad}
    # forward order preserved
    flat = [s.param for g in groups for s in g.slots]
    expect = [p for p in m.parameters() if p.requires_grad]
    assert all(a is b for a, b in zip(flat, expect))


def test_no_fusion_one_module_per_group():
    m = _mlp()
    groups = build_groups(m, threshold_bytes=None)
    assert len(groups) == 3  # 3 Linear modules with params
    for g in groups:
        assert len(g.modules) == 1


def test_num_groups_override():
    m = _mlp()
    groups = build_groups(m, threshold_bytes=123, num_groups=2)
    assert 1 <= len(groups) <= 3


def test_shared_params_deduped():
    lin = nn.Linear(8, 8)

    class Tied(nn.Module):
        def __init__(self):
            super().__init__()
            self.a = lin
            self.b = lin  # same module object twice

        def forward(self, x):
            return self.b(self.a(x))

    groups = build_groups(Tied(), threshold_bytes=None)
    slots = [s for g in groups for s in g.slots]
    assert len(slots) == 2  # weight + bias once


def test_allocate_grad_views_and_padding():
    m = _mlp()
    P = 4
    groups = build_groups(m, threshold_bytes=None)
    for g in groups:
        g.allocate(P, torch.device("cpu"))
        assert g.padded % (P * ALIGN_ELEMS) == 0
        assert g.shard.numel() * P == g.padded
        for s in g.slots:
            assert s.param.grad is not None
            assert s.param.grad.data_ptr() == \
                g.bucket[s.offset:].data_ptr()
    # autograd accumulates straight into the bucket
    x = torch.randn(5, 32)
    m(x).sum().backward()
    for g in groups:
        for s in g.slots:
            assert s.param.grad.data_ptr() == \
                g.bucket[s.offset: s.offset + s.numel].data_ptr()
            assert torch.isfinite(s.param.grad).all()
    g0 = groups[0]
    s0 = g0.slots[0]
    assert torch.equal(
        g0.bucket[s0.offset: s0.offset + s0.numel].view(s0.param.shape),
        s0.param.grad)


def test_channels_last_grad_views_accumulate_in_place():
    import dear_pytorch_amd as dear
    torch.manual_seed(0)
    m = nn.Sequential(nn.Conv2d(3, 8, 3, padding=1), nn.ReLU(),
                      nn.Conv2d(8, 4, 3, padding=1))
    m = m.to(memory_format=torch.channels_last)
    opt = dear.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9), model=m)
    x = torch.randn(2, 3, 8, 8).to(memory_format=torch.channels_last)
    for _ in range(3):
        opt.zero_grad()
        m(x).sum().backward()
        opt.step()
    opt.synchronize()
    assert opt._grad_view_fixups == 0, \
        f"{opt._grad_view_fixups} out-of-place grad accumulations"
    # numerics vs serial torch on the same layout
    torch.manual_seed(0)
    m2 = nn.Sequential(nn.Conv2d(3, 8, 3, padding=1), nn.ReLU(),
                       nn.Conv2d(8, 4, 3, padding=1))
    m2 = m2.to(memory_format=torch.channels_last)
    o2 = torch.optim.SGD(m2.parameters(), lr=0.05, momentum=0.9)
    for _ in range(3):
        o2.zero_grad()
        m2(x).sum().backward()
        o2.step()
    for (na, pa), (_, pb) in zip(m2.named_parameters(), m.named_parameters()):
        assert torch.allclose(pa, pb, atol=1e-6), na


def test_nearby_layers_grouping():
    m = _mlp()
    groups = build_groups(m, threshold_bytes=None, nearby_layers=2)
    assert len(groups) == 2  # 3 param modules in runs of 2
    assert len(groups[0].modules) == 2 and len(groups[1].modules) == 1


def test_padding_aligned_for_all_world_sizes():
    """Bucket padding invariants for every DP degree we target: padded is a
    multiple of world*64 elements (256-B-aligned shards per rank) and every
    slot offset is 64-element aligned."""
    import torch.nn as nn
    from dear_pytorch_amd.parallel.fusion import build_groups
    torch.manual_seed(0)
    m = nn.Sequential(nn.Linear(17, 33), nn.ReLU(), nn.Linear(33, 7),
                      nn.Linear(7, 129))  # deliberately odd sizes
    for world in (1, 2, 4, 8):
        groups = build_groups(m, threshold_bytes=1 << 10)
        for g in groups:
            g.allocate(world, torch.device("cpu"))
            assert g.padded % (64 * world) == 0, (world, g.padded)
            assert g.shard.numel() * world == g.padded
            for s in g.slots:
                assert s.offset % 64 == 0
            g.free()


def test_summary_and_repr():
    import torch.nn as nn
    import dear_pytorch_amd as dear
    m = nn.Sequential(nn.Linear(8, 8), nn.Linear(8, 4))
    opt = dear.DistributedOptimizer(torch.optim.SGD(m.parameters(), lr=0.1),
                                    model=m)
    s = opt.summary()
    assert "DeAR plan" in s and "tensors" in s
    assert "DearOptimizer" in repr(opt)
