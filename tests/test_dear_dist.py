"""Distributed plumbing tests: world_size=2 over gloo on CPU (BASELINE.json
configs[0]).  Verifies the full DeAR choreography — backward-hook RS, step()
AG enqueue, next-forward lazy fused update — against single-process training
on the combined batch."""
import torch
import torch.nn as nn

import pytest

from utils_dist import run_dist


def _model(seed=0):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(16, 32), nn.ReLU(), nn.Linear(32, 4))


def _full_data(T=5, bs=8, seed=7):
    g = torch.Generator().manual_seed(seed)
    return [(torch.randn(2 * bs, 16, generator=g),
             torch.randn(2 * bs, 4, generator=g)) for _ in range(T)]


def _serial_reference(T=5, bs=8):
    m = _model()
    opt = torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9)
    for x, y in _full_data(T, bs):
        opt.zero_grad()
        nn.functional.mse_loss(m(x), y).backward()
        opt.step()
    return {k: v.clone() for k, v in m.state_dict().items()}


def _rank_train(rank, world, T, bs):
    import dear_pytorch_amd as dear
    dear.init(backend="gloo")
    m = _model()
    dear.broadcast_parameters(m.state_dict(), root_rank=0)
    opt = dear.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9), model=m,
        threshold_bytes=1 << 12)
    for x, y in _full_data(T, bs):
        xs, ys = x[rank * bs:(rank + 1) * bs], y[rank * bs:(rank + 1) * bs]
        opt.zero_grad()
        nn.functional.mse_loss(m(xs), ys).backward()
        opt.step()
    opt.synchronize()
    out = {k: v.clone() for k, v in m.state_dict().items()}
    dear.shutdown()
    return out


@pytest.mark.timeout(300)
def test_dear_ws2_matches_serial_full_batch():
    T, bs = 5, 8
    ref = _serial_reference(T, bs)
    outs = run_dist(_rank_train, world_size=2, args=(T, bs))
    for r, sd in enumerate(outs):
        for k in ref:
            assert torch.allclose(ref[k], sd[k], atol=1e-5), \
                f"rank {r} {k}: {(ref[k] - sd[k]).abs().max():.3e}"
    # both ranks identical
    for k in ref:
        assert torch.equal(outs[0][k], outs[1][k])


def _rank_train_w(rank, world, T, bs):
    # world-size-agnostic DeAR training on 1/world of the batch
    import dear_pytorch_amd as dear
    dear.init(backend="gloo")
    m = _model()
    dear.broadcast_parameters(m.state_dict(), root_rank=0)
    opt = dear.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9), model=m,
        threshold_bytes=1 << 12)
    per = (2 * bs) // world
    for x, y in _full_data(T, bs):
        xs, ys = x[rank * per:(rank + 1) * per], y[rank * per:(rank + 1) * per]
        opt.zero_grad()
        nn.functional.mse_loss(m(xs), ys).backward()
        opt.step()
    opt.synchronize()
    out = {k: v.clone() for k, v in m.state_dict().items()}
    dear.shutdown()
    return out


@pytest.mark.timeout(600)
@pytest.mark.parametrize("world", [4, 8])
def test_dear_wsN_matches_serial_full_batch(world):
    """world_size 4 and 8 (padding/shard boundaries differ from ws2): still
    must equal serial SGD on the combined batch — first line of defense for
    the 8-GPU day-one run."""
    T, bs = 4, 8
    ref = _serial_reference(T, bs)
    outs = run_dist(_rank_train_w, world_size=world, args=(T, bs))
    for r, sd in enumerate(outs):
        for k in ref:
            assert torch.allclose(ref[k], sd[k], atol=1e-5), f"rank {r} {k}"
    for k in ref:
        for r in range(1, world):
            assert torch.equal(outs[0][k], outs[r][k])


def _rank_train_pack(rank, world, T, bs):
    import dear_pytorch_amd as dear
    dear.init(backend="gloo")
    m = _model()
    dear.broadcast_parameters(m.state_dict(), root_rank=0)
    opt = dear.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9), model=m,
        threshold_bytes=1 << 12, pack_grads=True)
    for x, y in _full_data(T, bs):
        xs, ys = x[rank * bs:(rank + 1) * bs], y[rank * bs:(rank + 1) * bs]
        opt.zero_grad()
        nn.functional.mse_loss(m(xs), ys).backward()
        opt.step()
    opt.synchronize()
    out = {k: v.clone() for k, v in m.state_dict().items()}
    dear.shutdown()
    return out


@pytest.mark.timeout(300)
def test_pack_mode_ws2_matches_serial():
    """Packed-grad mode with real collectives (gloo ws2) == serial SGD."""
    T, bs = 5, 8
    ref = _serial_reference(T, bs)
    outs = run_dist(_rank_train_pack, world_size=2, args=(T, bs))
    for r, sd in enumerate(outs):
        for k in ref:
            assert torch.allclose(ref[k], sd[k], atol=1e-5), f"rank {r} {k}"
    for k in ref:
        assert torch.equal(outs[0][k], outs[1][k])


def _rank_train_rb(rank, world, T, bs):
    import dear_pytorch_amd as dear
    from dear_pytorch_amd.parallel.rb import ReduceBcastOptimizer
    dear.init(backend="gloo")
    m = _model()
    dear.broadcast_parameters(m.state_dict(), root_rank=0)
    opt = ReduceBcastOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9), m,
        threshold_bytes=1 << 12)
    for x, y in _full_data(T, bs):
        xs, ys = x[rank * bs:(rank + 1) * bs], y[rank * bs:(rank + 1) * bs]
        opt.zero_grad()
        nn.functional.mse_loss(m(xs), ys).backward()
        opt.step()
    opt.synchronize()
    out = {k: v.clone() for k, v in m.state_dict().items()}
    dear.shutdown()
    return out


@pytest.mark.timeout(300)
def test_rb_ws2_matches_serial():
    """reduce+broadcast decomposition (reference dopt_rb.py) must equal
    serial SGD on the combined batch."""
    T, bs = 5, 8
    ref = _serial_reference(T, bs)
    outs = run_dist(_rank_train_rb, world_size=2, args=(T, bs))
    for k in ref:
        assert torch.allclose(ref[k], outs[0][k], atol=1e-5), k
        assert torch.equal(outs[0][k], outs[1][k])


def _rank_backend_agreement(rank, world):
    """Rank 0 claims native capability, rank 1 does not: BOTH must agree on
    torch-dist (a per-rank decision would leave rank 0 in ncclCommInitRank
    while rank 1 sits in dist.new_group — ADVICE r1)."""
    import dear_pytorch_amd as dear
    from dear_pytorch_amd.comm import backend as B
    dear.init(backend="gloo")
    B._native_precheck = lambda: rank == 0  # simulate split capability
    be = B.create_backend("agreement_test")
    kind = type(be).__name__
    prov = B.backend_provenance().get("agreement_test")
    dear.shutdown()
    return kind, prov


@pytest.mark.timeout(300)
def test_backend_choice_is_collective():
    outs = run_dist(_rank_backend_agreement, world_size=2)
    for kind, prov in outs:
        assert kind == "TorchDistBackend", outs
        assert prov == "torch-dist", outs


def _rank_train_naive(rank, world, T, bs):
    import dear_pytorch_amd as dear
    from dear_pytorch_amd.parallel.naive import NaiveDearOptimizer
    dear.init(backend="gloo")
    m = _model()
    dear.broadcast_parameters(m.state_dict(), root_rank=0)
    opt = NaiveDearOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9), model=m)
    for x, y in _full_data(T, bs):
        xs, ys = x[rank * bs:(rank + 1) * bs], y[rank * bs:(rank + 1) * bs]
        opt.zero_grad()
        nn.functional.mse_loss(m(xs), ys).backward()
        opt.step()
    opt.synchronize()
    out = {k: v.clone() for k, v in m.state_dict().items()}
    dear.shutdown()
    return out


@pytest.mark.timeout(300)
def test_naive_ws2_matches_serial_full_batch():
    """DeAR w/o TF ablation (reference dopt_rsag_naive.py): per-tensor RS/AG
    must still match serial SGD on the combined batch."""
    T, bs = 5, 8
    ref = _serial_reference(T, bs)
    outs = run_dist(_rank_train_naive, world_size=2, args=(T, bs))
    for r, sd in enumerate(outs):
        for k in ref:
            assert torch.allclose(ref[k], sd[k], atol=1e-5), \
                f"rank {r} {k}: {(ref[k] - sd[k]).abs().max():.3e}"
    for k in ref:
        assert torch.equal(outs[0][k], outs[1][k])


def _rank_naive_regroup(rank, world):
    # regroup() on the naive optimizer must keep its per-tensor plan and the
    # fusion_flags kwarg must be accepted (signature mismatch crash)
    import dear_pytorch_amd as dear
    from dear_pytorch_amd.parallel.naive import NaiveDearOptimizer
    dear.init(backend="gloo")
    m = _model()
    dear.broadcast_parameters(m.state_dict(), root_rank=0)
    opt = NaiveDearOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05), model=m)
    n_groups = len(opt.groups)
    for i, (x, y) in enumerate(_full_data(4, 4)):
        xs, ys = x[rank * 4:(rank + 1) * 4], y[rank * 4:(rank + 1) * 4]
        opt.zero_grad()
        nn.functional.mse_loss(m(xs), ys).backward()
        opt.step()
        if i == 1:
            opt.regroup(fusion_flags=None)
    opt.synchronize()
    assert len(opt.groups) == n_groups
    out = {k: v.clone() for k, v in m.state_dict().items()}
    dear.shutdown()
    return out


@pytest.mark.timeout(300)
def test_naive_regroup_ws2_consistent():
    outs = run_dist(_rank_naive_regroup, world_size=2)
    for k in outs[0]:
        assert torch.equal(outs[0][k], outs[1][k])


def _rank_collectives(rank, world):
    import dear_pytorch_amd as dear
    import torch.distributed as dist
    dear.init(backend="gloo")
    from dear_pytorch_amd.comm.backend import create_backend
    be = create_backend("t")
    # odd size exercises shard padding upstream; here plain collectives
    t = torch.full((17,), float(rank + 1))
    h = be.all_reduce(t)
    h.host_wait()
    assert torch.allclose(t, torch.full((17,), 3.0))

    P = world
    n = 8
    bucket = torch.arange(P * n, dtype=torch.float32) + rank
    shard = torch.empty(n)
    be.reduce_scatter(bucket.clone(), shard).host_wait()
    expect = sum(torch.arange(P * n, dtype=torch.float32) + r
                 for r in range(P))[rank * n:(rank + 1) * n]
    assert torch.allclose(shard, expect)

    out = torch.empty(P * n)
    be.all_gather(shard, out).host_wait()
    full = sum(torch.arange(P * n, dtype=torch.float32) + r for r in range(P))
    assert torch.allclose(out, full)

    b = torch.full((5,), float(rank))
    be.broadcast(b, 0).host_wait()
    assert torch.allclose(b, torch.zeros(5))

    sr = torch.full((4,), float(rank))
    rv = torch.empty(4)
    be.send_recv(sr, rv, 1 - rank).host_wait()
    assert torch.allclose(rv, torch.full((4,), float(1 - rank)))
    m = dear.allreduce(torch.tensor([float(rank)]), average=True)
    assert abs(m.item() - 0.5) < 1e-6
    dear.shutdown()
    return True


@pytest.mark.timeout(300)
def test_backend_collectives_ws2():
    assert all(run_dist(_rank_collectives, world_size=2))


def _rank_broadcast_opt_state(rank, world):
    import dear_pytorch_amd as dear
    dear.init(backend="gloo")
    m = _model(seed=rank)  # deliberately different
    dear.broadcast_parameters(m.state_dict(), root_rank=0)
    opt = torch.optim.Adam(m.parameters(), lr=1e-3)
    if rank == 0:  # run a couple of steps only on root → state diverges
        for _ in range(2):
            opt.zero_grad()
            nn.functional.mse_loss(m(torch.randn(4, 16)),
                                   torch.randn(4, 4)).backward()
            opt.step()
    dear.broadcast_parameters(m.state_dict(), root_rank=0)
    dear.broadcast_optimizer_state(opt, root_rank=0)
    sd = opt.state_dict()
    vals = [(k, v) for pid, st in sd["state"].items()
            for k, v in st.items() if torch.is_tensor(v)]
    out = {f"{i}": v.sum().item() for i, (k, v) in enumerate(vals)}
    dear.shutdown()
    return out


@pytest.mark.timeout(300)
def test_broadcast_optimizer_state_ws2():
    a, b = run_dist(_rank_broadcast_opt_state, world_size=2)
    assert a.keys() == b.keys() and len(a) > 0
    for k in a:
        assert abs(a[k] - b[k]) < 1e-6, k


def _rank_bcast_wrapper_state(rank, world):
    """broadcast_optimizer_state on the DearOptimizer WRAPPER must re-adopt
    the fused slabs on receiving ranks (stale slabs would silently ignore
    the broadcast and the ranks would diverge)."""
    import dear_pytorch_amd as dear
    dear.init(backend="gloo")
    m = _model()
    dear.broadcast_parameters(m.state_dict(), root_rank=0)
    opt = dear.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9), model=m,
        threshold_bytes=1 << 12)
    data = _full_data(6, 8)
    for x, y in data[:2]:
        xs, ys = x[rank * 8:(rank + 1) * 8], y[rank * 8:(rank + 1) * 8]
        nn.functional.mse_loss(m(xs), ys).backward()
        opt.step()
    opt.synchronize()
    if rank == 0:  # perturb root's momentum so the broadcast has to matter
        for st in opt.optim.state.values():
            if "momentum_buffer" in st:
                st["momentum_buffer"].add_(0.5)
    dear.broadcast_optimizer_state(opt, root_rank=0)
    for x, y in data[2:]:
        xs, ys = x[rank * 8:(rank + 1) * 8], y[rank * 8:(rank + 1) * 8]
        nn.functional.mse_loss(m(xs), ys).backward()
        opt.step()
    opt.synchronize()
    out = {k: v.clone() for k, v in m.state_dict().items()}
    dear.shutdown()
    return out


@pytest.mark.timeout(300)
def test_broadcast_wrapper_state_readopts_slabs_ws2():
    outs = run_dist(_rank_bcast_wrapper_state, world_size=2)
    for k in outs[0]:
        assert torch.equal(outs[0][k], outs[1][k]), k


def _rank_train_bf16comm(rank, world, T, bs):
    import torch
    import dear_pytorch_amd as dear
    dear.init(backend="gloo")
    m = _model()
    dear.broadcast_parameters(m.state_dict(), root_rank=0)
    opt = dear.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9), model=m,
        threshold_bytes=1 << 12, comm_dtype=torch.float16)
    for x, y in _full_data(T, bs):
        xs, ys = x[rank * bs:(rank + 1) * bs], y[rank * bs:(rank + 1) * bs]
        opt.zero_grad()
        nn.functional.mse_loss(m(xs), ys).backward()
        opt.step()
    opt.synchronize()
    out = {k: v.clone() for k, v in m.state_dict().items()}
    dear.shutdown()
    return out


@pytest.mark.timeout(300)
def test_dear_ws2_fp16_comm_close_to_serial():
    T, bs = 5, 8
    ref = _serial_reference(T, bs)
    outs = run_dist(_rank_train_bf16comm, world_size=2, args=(T, bs))
    for k in ref:
        # reduced-precision wire: looser tolerance, but must track
        assert torch.allclose(ref[k], outs[0][k], atol=5e-3), \
            f"{k}: {(ref[k] - outs[0][k]).abs().max():.3e}"
        assert torch.equal(outs[0][k], outs[1][k])


def _rank_wfbp_sparse(rank, world):
    import torch
    import dear_pytorch_amd as dear
    dear.init(backend="gloo")
    from dear_pytorch_amd.parallel.wfbp import WfbpOptimizer
    m = _model()
    dear.broadcast_parameters(m.state_dict(), root_rank=0)
    opt = WfbpOptimizer(torch.optim.SGD(m.parameters(), lr=0.05), m,
                        threshold_bytes=1 << 12, compressor="topk",
                        density=0.25)
    for x, y in _full_data(4, 8):
        xs, ys = x[rank * 8:(rank + 1) * 8], y[rank * 8:(rank + 1) * 8]
        opt.zero_grad()
        nn.functional.mse_loss(m(xs), ys).backward()
        opt.step()
    out = {k: v.clone() for k, v in m.state_dict().items()}
    dear.shutdown()
    return out


@pytest.mark.timeout(300)
def test_wfbp_sparse_topk_ws2_ranks_agree():
    outs = run_dist(_rank_wfbp_sparse, world_size=2)
    for k in outs[0]:
        assert torch.equal(outs[0][k], outs[1][k]), k
        assert torch.isfinite(outs[0][k]).all()


def _rank_wfbp_dense(rank, world):
    import torch
    import dear_pytorch_amd as dear
    dear.init(backend="gloo")
    from dear_pytorch_amd.parallel.wfbp import WfbpOptimizer
    m = _model()
    dear.broadcast_parameters(m.state_dict(), root_rank=0)
    opt = WfbpOptimizer(torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9),
                        m, threshold_bytes=None)
    for x, y in _full_data(5, 8):
        xs, ys = x[rank * 8:(rank + 1) * 8], y[rank * 8:(rank + 1) * 8]
        opt.zero_grad()
        nn.functional.mse_loss(m(xs), ys).backward()
        opt.step()
    out = {k: v.clone() for k, v in m.state_dict().items()}
    dear.shutdown()
    return out


@pytest.mark.timeout(300)
def test_wfbp_dense_ws2_matches_serial():
    ref = _serial_reference(5, 8)
    outs = run_dist(_rank_wfbp_dense, world_size=2)
    for k in ref:
        assert torch.allclose(ref[k], outs[0][k], atol=1e-5), k
        assert torch.equal(outs[0][k], outs[1][k])


def _rank_wfbp_mode(rank, world, mode):
    import torch
    import dear_pytorch_amd as dear
    dear.init(backend="gloo")
    from dear_pytorch_amd.parallel import baselines
    m = _model()
    dear.broadcast_parameters(m.state_dict(), root_rank=0)
    kw = {}
    if mode == "mgs":
        kw = dict(density=0.5)  # sparse sync: ranks must still agree
    opt = baselines.make(mode, torch.optim.SGD(m.parameters(), lr=0.05,
                                               momentum=0.9), m, **kw)
    for x, y in _full_data(5, 8):
        xs, ys = x[rank * 8:(rank + 1) * 8], y[rank * 8:(rank + 1) * 8]
        opt.zero_grad()
        nn.functional.mse_loss(m(xs), ys).backward()
        opt.step()
    out = {k: v.clone() for k, v in m.state_dict().items()}
    dear.shutdown()
    return out


@pytest.mark.timeout(300)
def test_asc_ws2_matches_serial():
    """ASC planning changes the fusion plan, not the numerics: dense
    all-reduce must still match serial SGD on the combined batch."""
    ref = _serial_reference(5, 8)
    outs = run_dist(_rank_wfbp_mode, world_size=2, args=("asc",))
    for k in ref:
        assert torch.allclose(ref[k], outs[0][k], atol=1e-5), k
        assert torch.equal(outs[0][k], outs[1][k])


@pytest.mark.timeout(300)
def test_mgs_ws2_ranks_agree():
    """MGS: sparse sync is lossy (top-k) so only cross-rank consistency and
    finiteness are checked (same standard as the wfbp sparse test)."""
    outs = run_dist(_rank_wfbp_mode, world_size=2, args=("mgs",))
    for k in outs[0]:
        assert torch.equal(outs[0][k], outs[1][k]), k
        assert torch.isfinite(outs[0][k]).all()


def _rank_bytescheduler(rank, world):
    import torch
    import dear_pytorch_amd as dear
    dear.init(backend="gloo")
    from dear_pytorch_amd.parallel.bytescheduler import ByteSchedulerOptimizer
    m = _model()
    dear.broadcast_parameters(m.state_dict(), root_rank=0)
    opt = ByteSchedulerOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9), m,
        partition_bytes=256)
    for x, y in _full_data(5, 8):
        xs, ys = x[rank * 8:(rank + 1) * 8], y[rank * 8:(rank + 1) * 8]
        opt.zero_grad()
        nn.functional.mse_loss(m(xs), ys).backward()
        opt.step()
    out = {k: v.clone() for k, v in m.state_dict().items()}
    dear.shutdown()
    return out


@pytest.mark.timeout(300)
def test_bytescheduler_ws2_matches_serial():
    ref = _serial_reference(5, 8)
    outs = run_dist(_rank_bytescheduler, world_size=2)
    for k in ref:
        assert torch.allclose(ref[k], outs[0][k], atol=1e-5), k
        assert torch.equal(outs[0][k], outs[1][k])


def _rank_bert_tied(rank, world, pack=False):
    import torch
    import dear_pytorch_amd as dear
    from dear_pytorch_amd import models
    dear.init(backend="gloo")
    torch.manual_seed(0)
    cfg = models.bert_base()
    cfg.num_hidden_layers = 1
    cfg.hidden_size = 32
    cfg.num_attention_heads = 2
    cfg.intermediate_size = 64
    cfg.vocab_size = 128
    m = models.BertForPreTraining(cfg)
    dear.broadcast_parameters(m.state_dict(), root_rank=0)
    crit = models.BertPretrainingCriterion(cfg.vocab_size)
    opt = dear.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=1e-3), model=m,
        threshold_bytes=1 << 12, pack_grads=pack)
    # tied decoder/embedding weight must occupy exactly one bucket slot
    n_slots = sum(len(g.slots) for g in opt.groups)
    n_params = len({id(p) for p in m.parameters()})
    assert n_slots == n_params
    g = torch.Generator().manual_seed(5 + rank)
    for _ in range(3):
        ids = torch.randint(0, cfg.vocab_size, (2, 12), generator=g)
        tt = torch.zeros(2, 12, dtype=torch.long)
        mlm = torch.full((2, 12), -1)
        mlm[:, 2] = ids[:, 2]
        nsp = torch.randint(0, 2, (2,), generator=g)
        opt.zero_grad()
        scores, rel = m(ids, tt)
        crit(scores, rel, mlm, nsp).backward()
        opt.step()
    opt.synchronize()
    out = {k: v.clone() for k, v in m.state_dict().items()}
    dear.shutdown()
    return out


@pytest.mark.timeout(300)
@pytest.mark.parametrize("pack", [False, True])
def test_bert_tied_weights_dear_ws2(pack):
    outs = run_dist(_rank_bert_tied, world_size=2, args=(pack,))
    for k in outs[0]:
        assert torch.equal(outs[0][k], outs[1][k]), k
        assert torch.isfinite(outs[0][k]).all(), k


def _rank_hetero_param_groups(rank, world):
    import torch
    import dear_pytorch_amd as dear
    dear.init(backend="gloo")
    m = _model()
    dear.broadcast_parameters(m.state_dict(), root_rank=0)
    # two param groups with different lr -> shadow-optimizer path
    pgs = [{"params": [p for n, p in m.named_parameters() if "weight" in n],
            "lr": 0.05},
           {"params": [p for n, p in m.named_parameters() if "bias" in n],
            "lr": 0.01}]
    opt = dear.DistributedOptimizer(torch.optim.SGD(pgs, momentum=0.9),
                                    model=m, threshold_bytes=1 << 12)
    for x, y in _full_data(4, 8):
        xs, ys = x[rank * 8:(rank + 1) * 8], y[rank * 8:(rank + 1) * 8]
        opt.zero_grad()
        nn.functional.mse_loss(m(xs), ys).backward()
        opt.step()
    opt.synchronize()
    out = {k: v.clone() for k, v in m.state_dict().items()}
    dear.shutdown()
    return out


@pytest.mark.timeout(300)
def test_hetero_param_groups_ws2():
    # serial reference with the same two param groups
    m = _model()
    pgs = [{"params": [p for n, p in m.named_parameters() if "weight" in n],
            "lr": 0.05},
           {"params": [p for n, p in m.named_parameters() if "bias" in n],
            "lr": 0.01}]
    opt = torch.optim.SGD(pgs, momentum=0.9)
    for x, y in _full_data(4, 8):
        opt.zero_grad()
        nn.functional.mse_loss(m(x), y).backward()
        opt.step()
    ref = {k: v.clone() for k, v in m.state_dict().items()}
    outs = run_dist(_rank_hetero_param_groups, world_size=2)
    for k in ref:
        assert torch.allclose(ref[k], outs[0][k], atol=1e-5), k
        assert torch.equal(outs[0][k], outs[1][k])


def _rank_regroup_ws2(rank, world):
    import torch
    import dear_pytorch_amd as dear
    dear.init(backend="gloo")
    m = _model()
    dear.broadcast_parameters(m.state_dict(), root_rank=0)
    opt = dear.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9), model=m,
        threshold_bytes=1 << 12)
    data = _full_data(6, 8)
    for i, (x, y) in enumerate(data):
        xs, ys = x[rank * 8:(rank + 1) * 8], y[rank * 8:(rank + 1) * 8]
        opt.zero_grad()
        nn.functional.mse_loss(m(xs), ys).backward()
        opt.step()
        if i == 2:
            # regression: regroup used to leave the old backward hooks
            # registered -> readiness double-counted -> RS fired with HALF a
            # group's gradients on multi-rank runs
            opt.regroup(1 << 14)
    opt.synchronize()
    # exactly one backward hook per param must be live
    assert len(opt._bw_hook_handles) == \
        len({id(p) for p in m.parameters() if p.requires_grad})
    out = {k: v.clone() for k, v in m.state_dict().items()}
    dear.shutdown()
    return out


@pytest.mark.timeout(300)
def test_regroup_midtraining_ws2_matches_serial():
    ref = _serial_reference(6, 8)
    outs = run_dist(_rank_regroup_ws2, world_size=2)
    for k in ref:
        assert torch.allclose(ref[k], outs[0][k], atol=1e-5), \
            f"{k}: {(ref[k] - outs[0][k]).abs().max():.3e}"
        assert torch.equal(outs[0][k], outs[1][k])


def _rank_checkpoint(rank, world, path):
    import torch
    import dear_pytorch_amd as dear
    from dear_pytorch_amd import checkpoint
    dear.init(backend="gloo")
    m = _model()
    dear.broadcast_parameters(m.state_dict(), root_rank=0)
    opt = dear.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9), model=m)
    for x, y in _full_data(3, 8):
        xs, ys = x[rank * 8:(rank + 1) * 8], y[rank * 8:(rank + 1) * 8]
        opt.zero_grad()
        nn.functional.mse_loss(m(xs), ys).backward()
        opt.step()
    checkpoint.save(path, m, opt, step=3)
    # resume into fresh replicas (deliberately divergent inits)
    m2 = _model(seed=17 + rank)
    opt2 = dear.DistributedOptimizer(
        torch.optim.SGD(m2.parameters(), lr=0.05, momentum=0.9), model=m2)
    step = checkpoint.load(path, m2, opt2.optim)
    assert step == 3
    out = {k: v.clone() for k, v in m2.state_dict().items()}
    dear.shutdown()
    return out


@pytest.mark.timeout(300)
def test_checkpoint_save_resume_ws2(tmp_path):
    path = str(tmp_path / "ck.pt")
    outs = run_dist(_rank_checkpoint, world_size=2, args=(path,))
    for k in outs[0]:
        assert torch.equal(outs[0][k], outs[1][k]), k


def _rank_cnn_bn(rank, world):
    import torch
    import dear_pytorch_amd as dear
    dear.init(backend="gloo")
    torch.manual_seed(0)

    def make():
        torch.manual_seed(0)
        return nn.Sequential(
            nn.Conv2d(3, 8, 3, padding=1), nn.BatchNorm2d(8), nn.ReLU(),
            nn.Conv2d(8, 8, 3, padding=1), nn.BatchNorm2d(8), nn.ReLU(),
            nn.AdaptiveAvgPool2d(1), nn.Flatten(), nn.Linear(8, 4))
    m = make()
    dear.broadcast_parameters(m.state_dict(), root_rank=0)
    opt = dear.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9), model=m,
        threshold_bytes=256)
    g = torch.Generator().manual_seed(3)
    data = [(torch.randn(2 * 4, 3, 8, 8, generator=g),
             torch.randn(2 * 4, 4, generator=g)) for _ in range(4)]
    for x, y in data:
        xs, ys = x[rank * 4:(rank + 1) * 4], y[rank * 4:(rank + 1) * 4]
        opt.zero_grad()
        nn.functional.mse_loss(m(xs), ys).backward()
        opt.step()
    opt.synchronize()
    out = {k: v.clone() for k, v in m.state_dict().items()}
    dear.shutdown()
    return out


@pytest.mark.timeout(300)
def test_cnn_batchnorm_dear_ws2_ranks_agree():
    """Conv+BN model under DeAR ws2: parameters identical across ranks;
    BN running stats differ per rank (local batches) as with DDP defaults."""
    outs = run_dist(_rank_cnn_bn, world_size=2)
    for k in outs[0]:
        if "running" in k or "num_batches" in k:
            continue
        assert torch.equal(outs[0][k], outs[1][k]), k
        assert torch.isfinite(outs[0][k]).all(), k


def _rank_bo_stress(rank, world, pack=False):
    import torch
    import dear_pytorch_amd as dear
    from dear_pytorch_amd.tuner import ThresholdTuner
    import torch.distributed as dist
    dear.init(backend="gloo")
    torch.manual_seed(0)
    m = nn.Sequential(nn.Linear(64, 128), nn.ReLU(), nn.Linear(128, 128),
                      nn.ReLU(), nn.Linear(128, 8))
    dear.broadcast_parameters(m.state_dict(), root_rank=0)
    opt = dear.DistributedOptimizer(
        torch.optim.SGD(m.parameters(), lr=0.01, momentum=0.9), model=m,
        threshold_bytes=1 << 14, pack_grads=pack)
    tuner = ThresholdTuner(opt, bounds_mb=(0.005, 0.5), window=2, warmup=2,
                           trials=8, verbose=False)
    g = torch.Generator().manual_seed(1)
    x = torch.randn(8, 64, generator=g)
    y = torch.randn(8, 8, generator=g)
    for _ in range(60):
        tuner.step_begin()
        opt.zero_grad()
        nn.functional.mse_loss(m(x), y).backward()
        opt.step()
        tuner.step_end()
    opt.synchronize()
    assert tuner.locked
    for p in m.parameters():
        c = p.detach().clone()
        dist.broadcast(c, src=0)
        assert torch.equal(c, p.detach()), "rank divergence under regrouping"
    out = all(torch.isfinite(p).all() for p in m.parameters())
    dear.shutdown()
    return out


@pytest.mark.timeout(600)
@pytest.mark.parametrize("pack", [False, True])
def test_bo_tuning_stress_ws2_stays_consistent(pack):
    """60 iterations with ~8 live regroups (the riskiest path: buffer
    teardown + hook re-registration mid-training) must keep ranks
    bit-identical — in both grad-view and packed-grad modes."""
    assert all(run_dist(_rank_bo_stress, world_size=2, args=(pack,)))
