"""Probe: full native RcclBackend 2-rank choreography with both ranks on ONE
GPU (bootstrap over gloo, collectives over RCCL).  Not a pytest test — run via
torchrun on a GPU box; prints OK or the RCCL error (NCCL traditionally rejects
duplicate devices in one communicator; if RCCL allows it we can regression-test
the real multi-rank path on 1-GPU boxes)."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    import torch.distributed as dist
    import dear_pytorch_amd as dear
    dear.init(backend="gloo")
    rank, world = dear.rank(), dear.size()
    torch.cuda.set_device(0)
    from dear_pytorch_amd.comm.backend import RcclBackend
    be = RcclBackend(torch.device("cuda", 0), tag="probe")
    n = 1024
    bucket = (torch.arange(world * n, dtype=torch.float32) + rank).cuda()
    shard = torch.empty(n, device="cuda")
    be.reduce_scatter(bucket, shard).host_wait()
    expect = sum(torch.arange(world * n) + r
                 for r in range(world)).float().cuda()[rank * n:(rank + 1) * n]
    assert torch.allclose(shard, expect), "RS mismatch"
    out = torch.empty(world * n, device="cuda")
    be.all_gather(shard, out).host_wait()
    full = sum(torch.arange(world * n) + r for r in range(world)).float().cuda()
    assert torch.allclose(out, full), "AG mismatch"
    t = torch.full((17,), float(rank + 1), device="cuda")
    be.all_reduce(t).host_wait()
    assert torch.allclose(t, torch.full((17,), float(sum(range(1, world + 1))),
                                        device="cuda"))
    if rank == 0:
        print("TWO-RANK-ONE-GPU RCCL OK", flush=True)
    dist.barrier()
    dear.shutdown()


if __name__ == "__main__":
    main()
