"""Unit tests for the comm backend layer (single-process paths)."""
import torch

from dear_pytorch_amd.comm.backend import (LocalBackend, NULL_HANDLE,
                                           create_backend)


def test_local_backend_rs_ag_roundtrip():
    be = LocalBackend()
    assert (be.rank, be.size) == (0, 1)
    bucket = torch.arange(8, dtype=torch.float32)
    shard = torch.empty(8)
    be.reduce_scatter(bucket, shard).host_wait()
    assert torch.equal(shard, bucket)
    out = torch.empty(8)
    be.all_gather(shard, out).host_wait()
    assert torch.equal(out, bucket)
    t = torch.ones(4)
    be.all_reduce(t).host_wait()
    assert torch.equal(t, torch.ones(4))  # sum over world of 1


def test_null_handle_idempotent():
    NULL_HANDLE.wait_compute()
    NULL_HANDLE.host_wait()
    NULL_HANDLE.wait_compute()


def test_create_backend_no_dist_is_local():
    import torch.distributed as dist
    assert not dist.is_initialized()
    be = create_backend("x")
    assert isinstance(be, LocalBackend)


def test_local_send_recv_raises():
    be = LocalBackend()
    try:
        be.send_recv(torch.ones(1), torch.ones(1), 0)
        raise AssertionError("expected RuntimeError")
    except RuntimeError:
        pass


def test_public_api_single_process():
    import dear_pytorch_amd as dear
    dear.init()  # no WORLD_SIZE: single-process mode
    assert dear.rank() == 0 and dear.size() == 1
    assert dear.local_rank() == 0
    assert not dear.is_initialized()  # no PG created in single-process mode
    dear.barrier()  # no-op
    t = dear.allreduce(torch.tensor([2.0]), average=True)
    assert t.item() == 2.0
    assert dear.broadcast_object({"a": 1}) == {"a": 1}
    dear.broadcast_parameters({"w": torch.ones(2)})  # no-op at ws 1
    dear.shutdown()
