#!/usr/bin/env python
"""xGMI collective microbenchmark: per-size all-reduce / reduce-scatter /
all-gather bandwidth over the native RCCL channels + alpha-beta fit.

    torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 8 \
        tools/comm_bench.py --sizes-mb 1,4,16,25,64,128

Rank 0 prints a table and the fitted alpha/beta (feeds
utils/perf_model.AlphaBeta and the MG-WFBP planner).
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def bench_op(be, op, nbytes, device, world, iters=10):
    n = max(nbytes // 4, world)
    n -= n % world
    full = torch.ones(n, device=device)
    shard = torch.empty(n // world, device=device)
    if op == "all_reduce":
        fn = lambda: be.all_reduce(full)
    elif op == "reduce_scatter":
        fn = lambda: be.reduce_scatter(full, shard)
    else:
        fn = lambda: be.all_gather(shard, full)
    for _ in range(3):
        fn().host_wait()
    if device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn().host_wait()
    if device.type == "cuda":
        torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    # algorithmic bus bandwidth (ring): allreduce moves 2(P-1)/P * N bytes
    factor = {"all_reduce": 2.0 * (world - 1) / world,
              "reduce_scatter": (world - 1) / world,
              "all_gather": (world - 1) / world}[op]
    return dt, factor * nbytes / dt / 1e9


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--sizes-mb", default="1,4,16,25,64,128")
    p.add_argument("--iters", type=int, default=10)
    args = p.parse_args()

    import dear_pytorch_amd as dear
    from dear_pytorch_amd.comm.backend import create_backend
    from dear_pytorch_amd.utils.perf_model import fit_alpha_beta

    dear.init()
    rank, world = dear.rank(), dear.size()
    device = dear.local_device()
    if device.type == "cuda":
        torch.cuda.set_device(device)
    be = create_backend("commbench")
    sizes = [int(float(s) * (1 << 20)) for s in args.sizes_mb.split(",")]

    results = {}
    for op in ("all_reduce", "reduce_scatter", "all_gather"):
        rows = []
        for nb in sizes:
            dt, bw = bench_op(be, op, nb, device, max(world, 2), args.iters)
            rows.append((nb, dt, bw))
        results[op] = rows
    if rank == 0:
        for op, rows in results.items():
            print(f"\n== {op} (world={world})")
            for nb, dt, bw in rows:
                print(f"  {nb / (1 << 20):8.1f} MB  {dt * 1e6:10.1f} us  "
                      f"{bw:7.1f} GB/s bus")
        ab = fit_alpha_beta([r[0] for r in results["all_reduce"]],
                            [r[1] for r in results["all_reduce"]])
        print(f"\nalpha = {ab.alpha * 1e6:.1f} us, "
              f"beta = 1/{1.0 / ab.beta / 1e9:.1f} GB/s "
              f"(allreduce, fits utils.perf_model.AlphaBeta)")
    dear.shutdown()


if __name__ == "__main__":
    main()
