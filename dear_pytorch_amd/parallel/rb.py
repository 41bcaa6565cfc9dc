"""Reduce+broadcast decomposition of the decoupled all-reduce (reference
dear/dopt_rb.py): backward hooks fire reduce-to-root, step() enqueues the
broadcast back, the lazy update applies in the next forward.  Kept for
comparison — bandwidth-suboptimal vs RS+AG on xGMI (root's links bottleneck),
as in the reference."""
from __future__ import annotations

from .dear import DearOptimizer
from .fusion import BucketGroup
from ..comm.backend import NULL_HANDLE

__all__ = ["ReduceBcastOptimizer"]


class ReduceBcastOptimizer(DearOptimizer):
    ROOT = 0

    def __init__(self, *a, **kw):
        if kw.get("comm_dtype") is not None:
            raise ValueError("reduce+broadcast ablation supports fp32 wire "
                             "format only")
        super().__init__(*a, **kw)

    def _launch_rs(self, group: BucketGroup):
        if self.size > 1 and self._do_rs:
            self._rs_handle[group.index] = self.comm_rs.reduce(
                group.bucket, self.ROOT)
        else:
            self._rs_handle[group.index] = NULL_HANDLE

    def _enqueue_gather(self, g: BucketGroup):
        if self.size > 1 and self._do_ag and self._do_rs:
            return self._bcast_after(g, self._rs_handle[g.index])
        return self._rs_handle[g.index]

    def _bcast_after(self, g, h):
        # order the broadcast behind the reduce: device-side when native,
        # host wait otherwise (gloo/CPU path)
        from ..comm.backend import RcclBackend, _EventHandle
        if isinstance(self.comm_ag, RcclBackend) and isinstance(h, _EventHandle):
            import dear_pytorch_amd._comm_core as comm_core
            comm_core.wait_op_across(self.comm_ag.comm, h.comm, h.opid)
        else:
            h.host_wait()
        return self.comm_ag.broadcast(g.bucket, self.ROOT)
