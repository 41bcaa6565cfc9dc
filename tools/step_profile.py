#!/usr/bin/env python
"""Steady-state per-step kernel breakdown of the bench workload.

rocprofv3 --stats aggregates the whole process, so MIOpen's find-phase
benchmarking (tens of seconds of candidate kernels under
torch.backends.cudnn.benchmark) swamps the steady-state signal.  This tool
warms up first, then profiles exactly N steps with torch.profiler (HIP
activities) and prints per-step kernel-time totals grouped by kernel name.

    python tools/step_profile.py [--steps 10] [bench.py args...]
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=8)
    ap.add_argument("--top", type=int, default=30)
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--method", default="dear")
    ap.add_argument("--batch-size", type=int, default=None)
    args = ap.parse_args()

    import bench
    sys.argv = ["bench.py", "--model", args.model, "--method", args.method]
    if args.batch_size:
        sys.argv += ["--batch-size", str(args.batch_size)]
    bargs = bench.parse_args()
    import dear_pytorch_amd as dear
    dear.init()
    device = dear.local_device()
    if device.type == "cuda":
        torch.cuda.set_device(device)
        torch.backends.cudnn.benchmark = True
    model, opt_fn, step_fn, bs, unit, metric = bench.build_workload(bargs,
                                                                    device)
    model, opt = bench.wrap_method(bargs, model, opt_fn)
    model.train()
    for _ in range(args.warmup):
        step_fn(model, opt)
    torch.cuda.synchronize()

    from torch.profiler import profile, ProfilerActivity
    with profile(activities=[ProfilerActivity.CUDA]) as prof:
        for _ in range(args.steps):
            step_fn(model, opt)
        torch.cuda.synchronize()

    agg = {}
    total = 0.0
    for ev in prof.key_averages():
        t = ev.self_device_time_total
        if t and ev.device_type != torch.autograd.DeviceType.CPU:
            agg[ev.key] = (t, ev.count)
            total += t
    rows = sorted(agg.items(), key=lambda kv: -kv[1][0])
    print(f"total device time/step: {total / args.steps / 1e3:.3f} ms "
          f"({args.steps} steps)")
    print(f"{'ms/step':>8}  {'%':>5}  {'calls/step':>10}  name")
    for name, (t, c) in rows[:args.top]:
        print(f"{t / args.steps / 1e3:8.3f}  {100 * t / total:5.1f}  "
              f"{c / args.steps:10.1f}  {name[:100]}")


if __name__ == "__main__":
    main()
