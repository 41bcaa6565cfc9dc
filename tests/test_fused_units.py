"""Unit tests for ops/fused.py internals (CPU)."""
import torch
import torch.nn as nn

from dear_pytorch_amd.ops import fused as F
from dear_pytorch_amd.parallel.fusion import build_groups


def _group(model):
    gs = build_groups(model, threshold_bytes=None)
    for g in gs:
        g.allocate(1, torch.device("cpu"))
    return gs


def test_kind_detection():
    ps = [nn.Parameter(torch.zeros(2))]
    assert F._kind(torch.optim.SGD(ps, lr=0.1)) == "sgd"
    assert F._kind(torch.optim.Adam(ps, lr=0.1)) == "adam"
    assert F._kind(torch.optim.AdamW(ps, lr=0.1)) == "adamw"
    assert F._kind(torch.optim.RMSprop(ps, lr=0.1)) == "other"


def test_group_hypers_uniform_and_hetero():
    m = nn.Sequential(nn.Linear(4, 4))
    g = _group(m)[0]
    opt = torch.optim.SGD(m.parameters(), lr=0.1, momentum=0.9)
    hyp, uniform = F._group_hypers(opt, g)
    assert uniform and hyp["lr"] == 0.1 and hyp["momentum"] == 0.9
    # two param groups with different lr covering one bucket group
    opt2 = torch.optim.SGD([
        {"params": [m[0].weight], "lr": 0.1},
        {"params": [m[0].bias], "lr": 0.2}], momentum=0.0)
    hyp2, uniform2 = F._group_hypers(opt2, g)
    assert not uniform2
    # param not in the optimizer at all
    opt3 = torch.optim.SGD([m[0].weight], lr=0.1)
    _, uniform3 = F._group_hypers(opt3, g)
    assert not uniform3


def test_flat_param_storage_order():
    p = torch.arange(24.0).reshape(2, 3, 2, 2)
    cl = p.to(memory_format=torch.channels_last)
    flat = F._flat_param(nn.Parameter(cl))
    # storage order of channels_last == NHWC traversal
    assert torch.equal(flat, cl.permute(0, 2, 3, 1).reshape(-1))
    cont = F._flat_param(nn.Parameter(p.clone()))
    assert torch.equal(cont, p.reshape(-1))


def test_build_desc_chunks():
    m = nn.Sequential(nn.Linear(64, 300))  # weight 19200 elems > CHUNK? no
    g = _group(m)[0]
    desc = F._build_desc(g)
    assert desc.shape[1] == 3
    # covers every param element exactly once
    total = int(desc[:, 2].sum())
    assert total == sum(s.numel for s in g.slots)
    assert int(desc[:, 2].max()) <= F.CHUNK


def test_shadow_step_scale_and_zero():
    m = nn.Sequential(nn.Linear(4, 4))
    g = _group(m)[0]
    opt = torch.optim.RMSprop(m.parameters(), lr=0.0)  # lr=0: params frozen
    g.bucket[: g.numel] = 2.0
    F._shadow_step(opt, g, scale=0.5)
    assert torch.equal(g.bucket, torch.zeros_like(g.bucket))  # re-zeroed
    assert "shadow_opt" in g.extra
