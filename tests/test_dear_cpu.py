"""DeAR-vs-serial numeric equivalence (the reference's implicit correctness
contract, SURVEY.md §4: decoupled lazy update == SGD up to one-iteration-late
semantics). CPU, world_size 1 here; world_size 2 in test_dear_dist.py."""
import copy

import pytest
import torch
import torch.nn as nn

import dear_pytorch_amd as dear


def _model(seed=0):
    torch.manual_seed(seed)
    return nn.Sequential(
        nn.Linear(16, 32), nn.ReLU(),
        nn.Linear(32, 32), nn.Tanh(),
        nn.Linear(32, 4),
    )


def _data(T=6, bs=8, seed=42):
    g = torch.Generator().manual_seed(seed)
    return [(torch.randn(bs, 16, generator=g), torch.randn(bs, 4, generator=g))
            for _ in range(T)]


def _train_serial(model, opt_fn, data):
    opt = opt_fn(model.parameters())
    for x, y in data:
        opt.zero_grad()
        nn.functional.mse_loss(model(x), y).backward()
        opt.step()
    return model


def _train_dear(model, opt_fn, data, **kw):
    opt = dear.DistributedOptimizer(opt_fn(model.parameters()), model=model, **kw)
    for x, y in data:
        opt.zero_grad()
        nn.functional.mse_loss(model(x), y).backward()
        opt.step()
    opt.synchronize()  # apply the last pending lazy update
    return model


OPTS = {
    "sgd": lambda ps: torch.optim.SGD(ps, lr=0.05),
    "sgd_mom": lambda ps: torch.optim.SGD(ps, lr=0.05, momentum=0.9,
                                          weight_decay=1e-4),
    "sgd_nesterov": lambda ps: torch.optim.SGD(ps, lr=0.05, momentum=0.9,
                                               nesterov=True),
    "sgd_dampening": lambda ps: torch.optim.SGD(ps, lr=0.05, momentum=0.9,
                                                dampening=0.3),
    "adam": lambda ps: torch.optim.Adam(ps, lr=1e-3, weight_decay=1e-4),
    "adamw": lambda ps: torch.optim.AdamW(ps, lr=1e-3, weight_decay=1e-2),
    "rmsprop": lambda ps: torch.optim.RMSprop(ps, lr=1e-3),  # shadow path
}


@pytest.mark.parametrize("name", list(OPTS))
@pytest.mark.parametrize("threshold", [None, 1 << 12, 1 << 30])
def test_dear_matches_serial(name, threshold):
    data = _data()
    a = _train_serial(_model(), OPTS[name], data)
    b = _train_dear(_model(), OPTS[name], data, threshold_bytes=threshold)
    for (na, pa), (nb, pb) in zip(a.named_parameters(), b.named_parameters()):
        assert torch.allclose(pa, pb, atol=1e-6), \
            f"{name} thr={threshold}: {na} max diff " \
            f"{(pa - pb).abs().max().item():.3e}"


def test_lazy_update_is_one_iteration_late():
    data = _data(T=3)
    m = _model()
    before = copy.deepcopy(m.state_dict())
    opt = dear.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.1), model=m)
    x, y = data[0]
    nn.functional.mse_loss(m(x), y).backward()
    opt.step()
    # weights unchanged until the NEXT forward touches each group
    for k, v in m.state_dict().items():
        assert torch.equal(v, before[k])
    m(x)  # forward applies the pending update group by group
    changed = any(not torch.equal(v, before[k])
                  for k, v in m.state_dict().items())
    assert changed


def test_exclude_parts_flags_run():
    data = _data(T=3)
    for part in ("reducescatter", "allgather"):
        _train_dear(_model(), OPTS["sgd"], data, exclude_parts=part)


def test_regroup_between_iterations():
    data = _data(T=6)
    a = _train_serial(_model(), OPTS["sgd_mom"], data)
    m = _model()
    opt = dear.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9,
                        weight_decay=1e-4), model=m)
    for i, (x, y) in enumerate(data):
        nn.functional.mse_loss(m(x), y).backward()
        opt.step()
        if i == 2:
            opt.regroup(1 << 12)  # BO-tuner style mid-training rebucket
    opt.synchronize()
    for (na, pa), (nb, pb) in zip(a.named_parameters(), m.named_parameters()):
        assert torch.allclose(pa, pb, atol=1e-6), na


def test_regroup_preserves_momentum_state():
    m = _model()
    opt = dear.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9), model=m)
    data = _data(T=4)
    for x, y in data[:2]:
        nn.functional.mse_loss(m(x), y).backward()
        opt.step()
    opt.synchronize()
    mom_before = {id(p): opt.optim.state[p]["momentum_buffer"].clone()
                  for g in opt.groups for s in g.slots for p in [s.param]}
    opt.regroup(1 << 14)
    for g in opt.groups:
        for s in g.slots:
            assert torch.equal(opt.optim.state[s.param]["momentum_buffer"],
                               mom_before[id(s.param)])


def test_state_dict_roundtrip():
    m = _model()
    opt = dear.DistributedOptimizer(
        torch.optim.Adam(m.parameters(), lr=1e-3), model=m)
    data = _data(T=3)
    for x, y in data:
        nn.functional.mse_loss(m(x), y).backward()
        opt.step()
    opt.synchronize()
    sd = opt.state_dict()
    assert sd["state"], "optimizer state should be populated"
    opt.load_state_dict(sd)


@pytest.mark.parametrize("name", ["sgd", "sgd_mom", "sgd_nesterov", "adam",
                                  "adamw", "rmsprop"])
def test_pack_mode_matches_serial(name):
    """Packed-grad mode (autograd assigns, one pack_add per group) must be
    numerically identical to the grad-as-bucket-view mode and serial."""
    data = _data()
    a = _train_serial(_model(), OPTS[name], data)
    b = _train_dear(_model(), OPTS[name], data, threshold_bytes=1 << 12,
                    pack_grads=True)
    for (na, pa), (_, pb) in zip(a.named_parameters(), b.named_parameters()):
        assert torch.allclose(pa, pb, atol=1e-6), \
            f"{name}: {na} {(pa - pb).abs().max().item():.3e}"


def test_pack_mode_gradient_accumulation():
    """accum_steps=2 in pack mode: grads from both micro-batches must sum in
    the bucket before the RS fires."""
    T = 4
    g = torch.Generator().manual_seed(11)
    halves = [(torch.randn(8, 16, generator=g), torch.randn(8, 4, generator=g))
              for _ in range(2 * T)]
    ref = _model()
    opt_r = torch.optim.SGD(ref.parameters(), lr=0.05)
    for t in range(T):
        opt_r.zero_grad()
        for x, y in halves[2 * t: 2 * t + 2]:
            (nn.functional.mse_loss(ref(x), y) / 2).backward()
        opt_r.step()
    m = _model()
    opt = dear.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05), model=m,
        accum_steps=2, pack_grads=True)
    for t in range(T):
        for x, y in halves[2 * t: 2 * t + 2]:
            (nn.functional.mse_loss(m(x), y) / 2).backward()
        opt.step()
    opt.synchronize()
    for (na, pa), (_, pb) in zip(ref.named_parameters(), m.named_parameters()):
        assert torch.allclose(pa, pb, atol=1e-6), na


@pytest.mark.parametrize("name", ["sgd_mom", "adam", "rmsprop"])
def test_midtraining_checkpoint_resume_matches(name):
    """Save after 3 steps, load into a FRESH wrapper, continue 3 more steps:
    the resumed run must track the uninterrupted one exactly (the
    shadow/generic path lost state across state_dict/load_state_dict, and
    load after the first step was ignored by the fused slabs)."""
    data = _data(T=6)
    # uninterrupted run
    m_ref = _model()
    opt_ref = dear.DistributedOptimizer(OPTS[name](m_ref.parameters()),
                                        model=m_ref)
    for x, y in data:
        nn.functional.mse_loss(m_ref(x), y).backward()
        opt_ref.step()
    opt_ref.synchronize()
    # interrupted at step 3
    m1 = _model()
    opt1 = dear.DistributedOptimizer(OPTS[name](m1.parameters()), model=m1)
    for x, y in data[:3]:
        nn.functional.mse_loss(m1(x), y).backward()
        opt1.step()
    opt1.synchronize()
    osd = copy.deepcopy(opt1.state_dict())
    msd = copy.deepcopy(m1.state_dict())
    assert osd["state"], f"{name}: state_dict must not be empty"
    # resume in a brand-new process-like wrapper
    m2 = _model()
    m2.load_state_dict(msd)
    opt2 = dear.DistributedOptimizer(OPTS[name](m2.parameters()), model=m2)
    # take one throwaway-free path: load AFTER construction (slabs may exist)
    nn.functional.mse_loss(m2(data[0][0]), data[0][1]).backward()
    opt2.step()
    opt2.synchronize()
    m2.load_state_dict(msd)
    opt2.load_state_dict(osd)
    for x, y in data[3:]:
        nn.functional.mse_loss(m2(x), y).backward()
        opt2.step()
    opt2.synchronize()
    for (na, pa), (_, pb) in zip(m_ref.named_parameters(),
                                 m2.named_parameters()):
        assert torch.allclose(pa, pb, atol=1e-6), \
            f"{name} {na}: {(pa - pb).abs().max().item():.3e}"


def test_generic_optimizer_state_dict_not_empty():
    """RMSprop runs through the shadow path; its square_avg state must be
    visible in state_dict() after steps."""
    m = _model()
    opt = dear.DistributedOptimizer(
        torch.optim.RMSprop(m.parameters(), lr=1e-3), model=m)
    data = _data(T=2)
    for x, y in data:
        nn.functional.mse_loss(m(x), y).backward()
        opt.step()
    opt.synchronize()
    sd = opt.state_dict()
    assert sd["state"]
    any_sq = any("square_avg" in st for st in sd["state"].values())
    assert any_sq, "RMSprop square_avg missing from checkpoint"


def test_pack_mode_regroup_between_iterations():
    """BO-style mid-training regroup must work in packed-grad mode (buckets
    freed and reallocated without grad views to restore)."""
    data = _data(T=6)
    a = _train_serial(_model(), OPTS["sgd_mom"], data)
    m = _model()
    opt = dear.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9,
                        weight_decay=1e-4), model=m, pack_grads=True)
    for i, (x, y) in enumerate(data):
        nn.functional.mse_loss(m(x), y).backward()
        opt.step()
        if i == 2:
            opt.regroup(1 << 12)
    opt.synchronize()
    for (na, pa), (_, pb) in zip(a.named_parameters(), m.named_parameters()):
        assert torch.allclose(pa, pb, atol=1e-6), na


def test_frozen_params_excluded_from_buckets():
    """requires_grad=False params must not enter any bucket and training the
    rest must match a serial run with the same freeze."""
    def frozen_model():
        m = _model()
        for p in m[0].parameters():  # freeze first Linear
            p.requires_grad_(False)
        return m
    data = _data(T=4)
    a = frozen_model()
    opt_a = torch.optim.SGD([p for p in a.parameters() if p.requires_grad],
                            lr=0.05, momentum=0.9)
    for x, y in data:
        opt_a.zero_grad()
        nn.functional.mse_loss(a(x), y).backward()
        opt_a.step()
    b = frozen_model()
    opt_b = dear.DistributedOptimizer(
        torch.optim.SGD([p for p in b.parameters() if p.requires_grad],
                        lr=0.05, momentum=0.9), model=b)
    frozen = {id(p) for p in b[0].parameters()}
    for g in opt_b.groups:
        for s in g.slots:
            assert id(s.param) not in frozen
    for x, y in data:
        opt_b.zero_grad()
        nn.functional.mse_loss(b(x), y).backward()
        opt_b.step()
    opt_b.synchronize()
    for (na, pa), (_, pb) in zip(a.named_parameters(), b.named_parameters()):
        assert torch.allclose(pa, pb, atol=1e-6), na


def test_pack_mode_with_comm_dtype_single_rank():
    """pack mode + bf16 wire buffers coexist (wire cast only kicks in at
    world>1; here just exercise the allocation path end-to-end)."""
    data = _data(T=3)
    m = _train_dear(_model(), OPTS["sgd_mom"], data, threshold_bytes=1 << 12,
                    pack_grads=True, comm_dtype=torch.bfloat16)
    assert all(torch.isfinite(p).all() for p in m.parameters())


def test_gradient_accumulation_matches_serial_big_batch():
    """accum_steps=2 with half batches == one step on the full batch (loss
    scaled by 1/accum so gradients average)."""
    T, bs = 4, 8
    data = _data(T=T, bs=2 * bs, seed=9)
    # serial: full batch per step
    a = _model()
    oa = torch.optim.SGD(a.parameters(), lr=0.05, momentum=0.9)
    for x, y in data:
        oa.zero_grad()
        nn.functional.mse_loss(a(x), y).backward()
        oa.step()
    # DeAR: two half-batch backwards per step, accum_steps=2
    b = _model()
    ob = dear.DistributedOptimizer(
        torch.optim.SGD(b.parameters(), lr=0.05, momentum=0.9), model=b,
        threshold_bytes=1 << 12, accum_steps=2)
    for x, y in data:
        ob.zero_grad()
        for k in range(2):
            xs, ys = x[k * bs:(k + 1) * bs], y[k * bs:(k + 1) * bs]
            (nn.functional.mse_loss(b(xs), ys) / 2).backward()
        ob.step()
    ob.synchronize()
    for (na, pa), (_, pb) in zip(a.named_parameters(), b.named_parameters()):
        assert torch.allclose(pa, pb, atol=1e-5), \
            f"{na}: {(pa - pb).abs().max():.3e}"


def test_dear_with_autocast_cpu():
    """--amp path: autocast compute, fp32 grads into buckets."""
    data = _data(T=3)
    m = _model()
    opt = dear.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05), model=m)
    for x, y in data:
        opt.zero_grad()
        with torch.autocast("cpu", dtype=torch.bfloat16):
            loss = nn.functional.mse_loss(m(x), y)
        loss.backward()
        opt.step()
    opt.synchronize()
    assert all(torch.isfinite(p).all() for p in m.parameters())
    assert all(p.grad.dtype == torch.float32 for p in m.parameters())
