"""dear_pytorch_amd — MI355X-native decoupled-all-reduce (DeAR) DP training engine.

Horovod-shaped public API with capability parity to the reference
lzhangbv/dear_pytorch (dear/__init__.py + dear/dopt_rsag.py L3/L4 surface):

    import dear_pytorch_amd as dear
    dear.init()                                   # torchrun / env:// rendezvous
    opt = dear.DistributedOptimizer(opt, model=model)
    dear.broadcast_parameters(model.state_dict(), root_rank=0)
    dear.broadcast_optimizer_state(opt, root_rank=0)

Process model: one process per GPU, ``torch.distributed`` over RCCL/xGMI
(backend "nccl" IS RCCL on ROCm).  The reference bootstraps with MPI at import
time (dopt_rsag.py:32); here ``init()`` performs the rendezvous explicitly.
"""
from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist

from .parallel.dear import DistributedOptimizer, DearOptimizer  # noqa: F401
from .parallel.fusion import build_groups, BucketGroup  # noqa: F401
from .comm.backend import create_backend, CommBackend  # noqa: F401

__version__ = "0.1.0"

_generic_backend: CommBackend | None = None


def init(backend: str | None = None, timeout_s: int = 1800):
    """Initialize the distributed world from torchrun env vars.

    backend: None picks "nccl" (RCCL) when a GPU is visible, else "gloo".
    Safe to call when WORLD_SIZE is absent (single-process mode).
    """
    global _generic_backend
    if dist.is_initialized():
        return
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1 and "MASTER_ADDR" not in os.environ:
        _generic_backend = create_backend()
        return
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    # Single-node runs rendezvous on loopback (the driver/torchrun use
    # 127.0.0.1).  RCCL's own TCP bootstrap (inside ncclCommInitRank) excludes
    # loopback by default and can HANG on boxes whose other interfaces don't
    # route between local processes — pin it to lo for loopback rendezvous.
    # Data still moves over xGMI/P2P; the socket is bootstrap-only.
    addr = os.environ.get("MASTER_ADDR", "127.0.0.1")
    if addr in ("127.0.0.1", "localhost", "::1"):
        os.environ.setdefault("NCCL_SOCKET_IFNAME", "lo")
    if torch.cuda.is_available():
        # modulo pinning (reference imagenet_benchmark.py:65 rank%4): lets
        # world>device_count test configs share a GPU instead of crashing
        lr = int(os.environ.get("LOCAL_RANK", "0"))
        torch.cuda.set_device(lr % torch.cuda.device_count())
    dist.init_process_group(backend=backend,
                            timeout=datetime.timedelta(seconds=timeout_s))
    _generic_backend = None  # created lazily after init


def _generic():
    global _generic_backend
    if _generic_backend is None:
        _generic_backend = create_backend("generic")
    return _generic_backend


def is_initialized() -> bool:
    return dist.is_initialized()


def rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", "0"))


def local_device() -> torch.device:
    """This rank's pinned device (LOCAL_RANK modulo visible GPUs)."""
    if torch.cuda.is_available():
        return torch.device("cuda", local_rank() % torch.cuda.device_count())
    return torch.device("cpu")


def barrier():
    """Host barrier across the world (reference comm_core g_barriar /
    Communicator.barrier)."""
    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.barrier()


def shutdown():
    global _generic_backend
    _generic_backend = None
    if dist.is_initialized():
        dist.destroy_process_group()


def allreduce(tensor: torch.Tensor, average: bool = True, name: str | None = None):
    """Blocking mean/sum all-reduce of a (typically small metric) tensor —
    reference dear/dopt_rsag.py:543."""
    t = tensor.detach().clone()
    src_device = t.device
    if (size() > 1 and torch.cuda.is_available()
            and dist.get_backend() == "nccl" and not t.is_cuda):
        t = t.cuda()  # native RCCL channel needs device tensors
    _generic().all_reduce(t).host_wait()
    if average and size() > 1:
        t /= size()
    return t.to(src_device)


def broadcast_parameters(params, root_rank: int = 0):
    """Broadcast model parameters (a state_dict or named-parameter iterable)
    from root so every rank starts identical — reference dopt_rsag.py:396-421."""
    if size() == 1:
        return
    if isinstance(params, dict):
        items = sorted(params.items())
    else:
        items = sorted(dict(params).items())
    be = _generic()
    tensors = [v.data for _, v in items if torch.is_tensor(v)]
    from .comm.backend import RcclBackend
    if isinstance(be, RcclBackend):
        # one grouped RCCL call (ncclGroupStart/End) instead of N round trips;
        # order behind the compute stream that initialized the params
        be.record_compute()
        be.comm.wait_op_host(be.comm.broadcast_many(tensors, root_rank))
        return
    handles = [be.broadcast(t, root_rank) for t in tensors]
    for h in handles:
        h.host_wait()


def broadcast_optimizer_state(optimizer, root_rank: int = 0):
    """Broadcast optimizer state (incl. python scalars via tensor wrapping) —
    reference dopt_rsag.py:424-540."""
    if size() == 1:
        return
    # keep the wrapper so its load_state_dict can re-adopt the fused state
    # slabs on receiving ranks (plain inner load would leave them stale)
    wrapper = optimizer if isinstance(optimizer, DearOptimizer) else None
    inner = optimizer.optim if wrapper is not None else optimizer
    # one-shot startup consistency: ship root's full state (scalars wrapped
    # with the tensors, mirroring the reference's tensor-wrap callbacks)
    obj = [inner.state_dict() if rank() == root_rank else None]
    dist.broadcast_object_list(obj, src=root_rank)
    if rank() != root_rank:
        cpu_state = obj[0]
        dev = next(iter(
            p.device for g in inner.param_groups for p in g["params"]))
        for st in cpu_state["state"].values():
            for k, v in st.items():
                if torch.is_tensor(v) and v.dim() > 0:
                    st[k] = v.to(dev)
        (wrapper or inner).load_state_dict(cpu_state)


# convenience namespace parity with the reference benchmark drivers
def broadcast_object(obj, root_rank: int = 0):
    lst = [obj]
    if size() > 1:
        dist.broadcast_object_list(lst, src=root_rank)
    return lst[0]
