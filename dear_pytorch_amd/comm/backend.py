"""Communication backends for the DeAR engine.

The reference framework (lzhangbv/dear_pytorch, common/comm_core/src/communicator.cpp)
bootstraps with MPI and drives NCCL on CUDA side streams with integer stream-index
handles.  This MI355X-native rebuild keeps the same *capability* — independent
reduce-scatter / all-gather communicators that overlap with compute — but is designed
for one-process-per-GPU ``torch.distributed`` over RCCL/xGMI:

* rendezvous via torchrun / env:// (no MPI dependency),
* the native path (``RcclBackend``) uses the in-tree C++/HIP extension
  ``dear_pytorch_amd._comm_core``: one RCCL communicator per logical channel
  (RS / AG / generic), each on its own HIP side stream taken from the ATen pool,
  with hipEvent-based handles so compute→comm and comm→compute dependencies are
  expressed **device-side** (the reference's handles are stream indices and all
  its syncs are host-blocking; its compute→comm edge is a placebo self-wait,
  tensorfusion.py:304),
* a pure-torch backend (``TorchDistBackend``) provides the same interface on
  gloo/CPU for world_size>1 plumbing tests and as the generic fallback, and a
  ``LocalBackend`` services world_size==1 without any communicator.

Handles returned by collective calls are opaque; ``wait_compute(handle)`` makes the
*current compute stream* wait (device-side on GPU, host-side otherwise) and
``host_wait(handle)`` blocks the host.
"""
from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist

__all__ = [
    "CommBackend",
    "LocalBackend",
    "TorchDistBackend",
    "RcclBackend",
    "create_backend",
    "backend_provenance",
]


class _Handle:
    """Opaque completion handle. Subclasses carry an event or a dist.Work."""

    def wait_compute(self):  # pragma: no cover - interface
        raise NotImplementedError

    def host_wait(self):  # pragma: no cover - interface
        raise NotImplementedError


class _NullHandle(_Handle):
    def wait_compute(self):
        pass

    def host_wait(self):
        pass


NULL_HANDLE = _NullHandle()


class _WorkHandle(_Handle):
    __slots__ = ("work", "_post")

    def __init__(self, work, post=None):
        self.work = work
        self._post = post

    def wait_compute(self):
        if self.work is not None:
            self.work.wait()
            self.work = None
        if self._post is not None:
            self._post()
            self._post = None

    def host_wait(self):
        self.wait_compute()


class CommBackend:
    """Interface: flat fp32/bf16 bucket collectives with async handles."""

    rank: int = 0
    size: int = 1

    # -- collective ops on flat contiguous tensors -------------------------
    def reduce_scatter(self, bucket: torch.Tensor, shard: torch.Tensor) -> _Handle:
        """sum-reduce-scatter bucket (numel = P * shard.numel()) into shard."""
        raise NotImplementedError

    def all_gather(self, shard: torch.Tensor, bucket: torch.Tensor,
                   after: Optional[_Handle] = None) -> _Handle:
        """all-gather shard into bucket; if `after` given, order behind it device-side."""
        raise NotImplementedError

    def all_reduce(self, t: torch.Tensor) -> _Handle:
        raise NotImplementedError

    def reduce(self, t: torch.Tensor, root: int) -> _Handle:
        raise NotImplementedError

    def broadcast(self, t: torch.Tensor, root: int) -> _Handle:
        raise NotImplementedError

    def send_recv(self, send: torch.Tensor, recv: torch.Tensor, peer: int) -> _Handle:
        raise NotImplementedError

    # -- ordering ----------------------------------------------------------
    def record_compute(self):
        """Record the current compute-stream position; collectives launched next
        will wait on it device-side (native backend). Returns a token."""
        return None

    def synchronize(self):
        """Host-block until every outstanding collective issued here completed."""
        pass

    def barrier(self):
        if dist.is_initialized():
            dist.barrier()


class LocalBackend(CommBackend):
    """world_size == 1: collectives degenerate to copies / no-ops."""

    def __init__(self):
        self.rank, self.size = 0, 1

    def reduce_scatter(self, bucket, shard):
        if shard.data_ptr() != bucket.data_ptr():
            shard.copy_(bucket[: shard.numel()])
        return NULL_HANDLE

    def all_gather(self, shard, bucket, after=None):
        if shard.data_ptr() != bucket.data_ptr():
            bucket[: shard.numel()].copy_(shard)
        return NULL_HANDLE

    def all_reduce(self, t):
        return NULL_HANDLE

    def reduce(self, t, root=0):
        return NULL_HANDLE

    def broadcast(self, t, root=0):
        return NULL_HANDLE

    def send_recv(self, send, recv, peer):
        raise RuntimeError("send_recv with world_size == 1")

    def barrier(self):
        pass


class TorchDistBackend(CommBackend):
    """torch.distributed-backed channel (gloo on CPU; also usable over nccl).

    gloo has no reduce_scatter_tensor, so RS is emulated with all_reduce +
    own-shard copy (numerically identical; used only for CPU plumbing tests).
    Each instance gets its own process group so RS and AG traffic are
    independent channels like the native backend's communicators.
    """

    def __init__(self, group=None):
        assert dist.is_initialized(), "call dear.init() first"
        self.group = group
        self.rank = dist.get_rank()
        self.size = dist.get_world_size()
        self._native_rs = dist.get_backend() == "nccl"
        self._gloo = dist.get_backend() == "gloo"

    def _stage(self, t):
        """gloo + CUDA tensor: stage through the host (gloo's device support
        is build-dependent; host staging makes multi-rank-on-one-GPU tests
        deterministic). Returns (op_tensor, copy_back_or_None)."""
        if self._gloo and t.is_cuda:
            h = t.detach().cpu()
            return h, (lambda: t.copy_(h))
        return t, None

    def reduce_scatter(self, bucket, shard):
        if self._native_rs:
            work = dist.reduce_scatter_tensor(shard, bucket, op=dist.ReduceOp.SUM,
                                              group=self.group, async_op=True)
            return _WorkHandle(work)
        n = shard.numel()
        b, _ = self._stage(bucket)
        work = dist.all_reduce(b, op=dist.ReduceOp.SUM, group=self.group,
                               async_op=True)
        r = self.rank

        def post():
            if b is not bucket:
                bucket.copy_(b)
            shard.copy_(bucket[r * n:(r + 1) * n])

        return _WorkHandle(work, post)

    def all_gather(self, shard, bucket, after=None):
        if after is not None:
            after.wait_compute()
        s, _ = self._stage(shard)
        b = torch.empty_like(bucket, device=s.device) if s is not shard \
            else bucket
        work = dist.all_gather_into_tensor(b, s, group=self.group,
                                           async_op=True)
        post = (lambda: bucket.copy_(b)) if b is not bucket else None
        return _WorkHandle(work, post)

    def all_reduce(self, t):
        h, back = self._stage(t)
        work = dist.all_reduce(h, group=self.group, async_op=True)
        return _WorkHandle(work, back)

    def reduce(self, t, root):
        h, back = self._stage(t)
        work = dist.reduce(h, root, group=self.group, async_op=True)
        return _WorkHandle(work, back)

    def broadcast(self, t, root):
        h, back = self._stage(t)
        work = dist.broadcast(h, root, group=self.group, async_op=True)
        return _WorkHandle(work, back)

    def send_recv(self, send, recv, peer):
        s, _ = self._stage(send)
        r, back = self._stage(recv)
        reqs = dist.batch_isend_irecv([
            dist.P2POp(dist.isend, s, peer, group=self.group),
            dist.P2POp(dist.irecv, r, peer, group=self.group),
        ])

        class _H(_Handle):
            def wait_compute(self):
                for req in reqs:
                    req.wait()
                if back is not None:
                    back()

            host_wait = wait_compute

        return _H()


class _EventHandle(_Handle):
    """Handle over the native extension: integer op id + communicator ref."""

    __slots__ = ("comm", "opid")

    def __init__(self, comm, opid):
        self.comm = comm
        self.opid = opid

    def wait_compute(self):
        # device-side: current torch stream waits on the op's hipEvent
        self.comm.wait_op_stream(self.opid, torch.cuda.current_stream().cuda_stream)

    def host_wait(self):
        self.comm.wait_op_host(self.opid)


class RcclBackend(CommBackend):
    """Native RCCL-over-xGMI channel via the in-tree C++/HIP extension.

    One RCCL communicator on one dedicated HIP side stream per instance;
    DeAR creates three (generic / reduce-scatter / all-gather) so RS and AG
    rings use xGMI link capacity concurrently (7 p2p links/GPU).
    """

    _instances = 0  # SPMD-ordered creation => identical count on every rank

    def __init__(self, device: torch.device, tag: str = "generic"):
        import dear_pytorch_amd._comm_core as comm_core  # loud ImportError on GPU box

        assert dist.is_initialized()
        self.rank = dist.get_rank()
        self.size = dist.get_world_size()
        self.device = device
        # RCCL unique-id exchange over the torch.distributed store (replaces
        # the reference's MPI_Bcast, communicator.cpp:43-57).  The instance
        # counter keeps keys unique when several backends reuse a tag.
        RcclBackend._instances += 1
        uid_key = f"dear_rccl_uid/{tag}/{RcclBackend._instances}"
        store = dist.distributed_c10d._get_default_store()
        if self.rank == 0:
            uid = comm_core.get_unique_id()
            store.set(uid_key, uid)
        else:
            uid = store.get(uid_key)
        with torch.cuda.device(device):
            self.comm = comm_core.Communicator(self.rank, self.size, uid)

    def record_compute(self):
        self.comm.wait_stream(torch.cuda.current_stream().cuda_stream)

    def reduce_scatter(self, bucket, shard):
        self.record_compute()
        return _EventHandle(self.comm, self.comm.reduce_scatter(bucket, shard))

    def all_gather(self, shard, bucket, after=None):
        if isinstance(after, _EventHandle):
            import dear_pytorch_amd._comm_core as comm_core
            comm_core.wait_op_across(self.comm, after.comm, after.opid)
        elif after is not None:
            after.wait_compute()
        self.record_compute()
        return _EventHandle(self.comm, self.comm.all_gather(shard, bucket))

    def all_reduce(self, t):
        self.record_compute()
        return _EventHandle(self.comm, self.comm.all_reduce(t))

    def reduce(self, t, root):
        self.record_compute()
        return _EventHandle(self.comm, self.comm.reduce(t, root))

    def broadcast(self, t, root):
        self.record_compute()
        return _EventHandle(self.comm, self.comm.broadcast(t, root))

    def send_recv(self, send, recv, peer):
        self.record_compute()
        return _EventHandle(self.comm, self.comm.send_recv(send, recv, peer))

    def synchronize(self):
        self.comm.synchronize()


# Which backend kind actually serves each channel tag, for honest benchmark
# provenance (a silent native→torch fallback must not be able to
# launder bench numbers).  bench.py embeds this in its JSON line.
PROVENANCE: dict = {}


def backend_provenance() -> dict:
    """tag -> 'rccl-native' | 'torch-dist' | 'local' for every channel created."""
    return dict(PROVENANCE)


def _native_precheck() -> bool:
    """Per-rank, non-collective conditions for the native RCCL path."""
    if not torch.cuda.is_available() or dist.get_backend() == "gloo":
        return False
    if os.environ.get("DEAR_FORCE_TORCH_COMM", "0") == "1":
        return False
    try:
        import dear_pytorch_amd._comm_core  # noqa: F401
    except Exception:  # noqa: BLE001
        return False
    return True


def create_backend(tag: str = "generic") -> CommBackend:
    """Pick the backend for this process: native RCCL on ROCm GPUs, torch.distributed
    (gloo) on CPU, LocalBackend when not distributed.

    The native/torch decision is COLLECTIVE: every rank all-reduces its
    precheck result so all take the same branch (a per-rank fallback would
    leave some ranks in dist.new_group() and others in ncclCommInitRank —
    desync/hang).  Once all ranks agree to go native, an init
    failure raises on every rank instead of falling back, because a partial
    native init cannot be unwound collectively.
    """
    if not dist.is_initialized() or dist.get_world_size() == 1:
        PROVENANCE[tag] = "local"
        return LocalBackend()
    can_native = _native_precheck()
    flag = torch.tensor([1 if can_native else 0], dtype=torch.int64)
    if dist.get_backend() != "gloo" and torch.cuda.is_available():
        flag = flag.to(torch.device("cuda", torch.cuda.current_device()))
    dist.all_reduce(flag, op=dist.ReduceOp.MIN)
    all_native = bool(flag.item())
    if can_native and not all_native:
        import sys
        print(f"[dear] WARNING: channel '{tag}': native comm_core available "
              f"here but not on every rank; all ranks using torch.distributed",
              file=sys.stderr, flush=True)
    if all_native:
        backend = RcclBackend(torch.device("cuda",
                                           torch.cuda.current_device()), tag)
        PROVENANCE[tag] = "rccl-native"
        return backend
    if torch.cuda.is_available() and dist.get_backend() != "gloo" and \
            os.environ.get("DEAR_STRICT_NATIVE_COMM", "0") == "1":
        raise RuntimeError(
            f"DEAR_STRICT_NATIVE_COMM=1 but native comm_core unavailable for "
            f"channel '{tag}' on at least one rank")
    # independent PG per channel so RS/AG are separate traffic streams
    group = dist.new_group(backend=dist.get_backend())
    PROVENANCE[tag] = "torch-dist"
    return TorchDistBackend(group)
