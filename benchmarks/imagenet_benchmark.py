#!/usr/bin/env python
"""Synthetic-ImageNet CNN throughput benchmark (reference parity:
dear/imagenet_benchmark.py and its per-method copies).

Same CLI axes (--model/--batch-size/--num-iters/--method/--threshold/
--compressor/--density/--exclude-parts/--no-fusion) and the same log
contract the harness parses: ``Total img/sec on N GPU(s): X +-Y``
(benchmarks.py:119-128).  Launch: torchrun --nproc-per-node N
benchmarks/imagenet_benchmark.py ...
"""
import argparse
import os
import sys
import timeit

os.environ.setdefault("MIOPEN_FIND_MODE", "FAST")

import numpy as np  # noqa: E402
import torch  # noqa: E402

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="resnet50")
    p.add_argument("--batch-size", type=int, default=64)
    p.add_argument("--num-warmup-batches", type=int, default=10)
    p.add_argument("--num-batches-per-iter", type=int, default=10)
    p.add_argument("--num-iters", type=int, default=5)
    p.add_argument("--method", default="dear",
                   choices=["dear", "dear-bo", "dear-wt", "ddp", "wfbp", "mgwfbp",
                            "asc", "mgs", "naive", "rb", "bytescheduler"])
    p.add_argument("--asc", action="store_true",
                   help="alias for --method asc (reference --asc flag)")
    p.add_argument("--mgs", action="store_true",
                   help="alias for --method mgs (reference MGS-SGD planning)")
    p.add_argument("--threshold", type=int, default=25 * 1024 * 1024,
                   help="fusion threshold bytes")
    p.add_argument("--no-fusion", action="store_true")
    p.add_argument("--exclude-parts", default="")
    p.add_argument("--compressor", default="none")
    p.add_argument("--density", type=float, default=1.0)
    p.add_argument("--timeline", default="")
    p.add_argument("--amp", action="store_true",
                   help="bf16 autocast compute (reference --fp16 axis; "
                        "gradients/optimizer stay fp32)")
    args = p.parse_args()
    if args.asc:
        args.method = "asc"
    if args.mgs:
        args.method = "mgs"
    if args.timeline:
        os.environ["DEAR_TIMELINE"] = args.timeline  # chrome-trace of the
        # DeAR queue events (reference WFSGD_TIMELINE via horovod_mpi_cj.sh)

    import dear_pytorch_amd as dear
    from dear_pytorch_amd import models

    dear.init()
    rank, world = dear.rank(), dear.size()
    on_gpu = torch.cuda.is_available()
    device = dear.local_device()
    if on_gpu:
        torch.cuda.set_device(device)
        torch.backends.cudnn.benchmark = True

    def log(msg):
        if rank == 0:
            print(msg, flush=True)

    fused_bn = on_gpu and args.model.startswith(("resnet", "densenet",
                                                 "inception"))
    model = models.get_cnn(args.model, fused_bn=fused_bn).to(device)
    res = 299 if args.model == "inceptionv4" else 224
    g = torch.Generator().manual_seed(55 + rank)
    data = torch.randn(args.batch_size, 3, res, res, generator=g).to(device)
    target = torch.randint(0, 1000, (args.batch_size,), generator=g).to(device)
    lossf = torch.nn.CrossEntropyLoss().to(device)
    if on_gpu:  # NHWC: MIOpen's fast igemm path on CDNA (profiles/README.md)
        model = model.to(memory_format=torch.channels_last)
        data = data.to(memory_format=torch.channels_last)

    if world > 1:
        dear.broadcast_parameters(model.state_dict(), root_rank=0)
    base_opt = torch.optim.SGD(model.parameters(), lr=0.01, momentum=0.9)
    threshold = None if args.no_fusion else args.threshold
    tuner = None
    if args.method == "ddp":
        if world > 1:
            model = torch.nn.parallel.DistributedDataParallel(
                model, gradient_as_bucket_view=True)
        opt = base_opt
    elif args.method in ("dear", "dear-bo", "dear-wt"):
        opt = dear.DistributedOptimizer(base_opt, model=model,
                                        threshold_bytes=threshold,
                                        exclude_parts=args.exclude_parts)
        if args.method == "dear-bo":
            from dear_pytorch_amd.tuner import ThresholdTuner
            tuner = ThresholdTuner(opt)
        elif args.method == "dear-wt":
            from dear_pytorch_amd.parallel.waittime import \
                WaitTimeAdaptiveFusion
            wt = WaitTimeAdaptiveFusion(opt)

            class _WtTuner:  # same step_begin/step_end surface as the BO tuner
                def step_begin(self):
                    pass

                def step_end(self):
                    wt.step_end()
            tuner = _WtTuner()
    else:
        from dear_pytorch_amd.parallel import baselines
        kw = {}
        if args.method in ("wfbp", "mgwfbp", "asc", "mgs") and \
                args.compressor != "none" and args.density < 1.0:
            kw = dict(compressor=args.compressor, density=args.density)
        if args.method in ("mgwfbp", "asc", "mgs"):
            # reference protocol: measure per-layer backward times first and
            # hand them to the planner (mgwfbp/imagenet_benchmark.py:98-100)
            from dear_pytorch_amd.profiling import Profiling
            _, times, _ = Profiling.benchmark(
                model, lambda: (data, target),
                lambda out, tgt: lossf(out, tgt), warmup=2, iters=5)
            kw["layerwise_times"] = times
        opt = baselines.make(args.method, base_opt, model,
                             threshold_bytes=threshold, **kw)

    model.train()

    amp_ctx = (lambda: torch.autocast("cuda", dtype=torch.bfloat16)) \
        if (args.amp and on_gpu) else torch.enable_grad

    def benchmark_step():
        if tuner:
            tuner.step_begin()
        opt.zero_grad()
        with amp_ctx():
            loss = lossf(model(data), target)
        loss.backward()
        opt.step()
        if tuner:
            tuner.step_end()

    log(f"Model: {args.model}, bs {args.batch_size}/GPU, method {args.method}, "
        f"{world} GPU(s)")
    timeit.timeit(benchmark_step, number=args.num_warmup_batches)
    img_secs = []
    import time as _time
    for _ in range(args.num_iters):
        # sync-bracketed window: DeAR's step() is async, so honest timing
        # needs device completion inside the window (unlike host timeit)
        if on_gpu:
            torch.cuda.synchronize()
        t0 = _time.perf_counter()
        for _b in range(args.num_batches_per_iter):
            benchmark_step()
        if on_gpu:
            torch.cuda.synchronize()
        t = _time.perf_counter() - t0
        img_secs.append(args.batch_size * args.num_batches_per_iter / t)
    img_sec_mean = np.mean(img_secs)
    img_sec_conf = 1.96 * np.std(img_secs)
    log(f"Img/sec per GPU: {img_sec_mean:.1f} +-{img_sec_conf:.1f}")
    log(f"Total img/sec on {world} GPU(s): {world * img_sec_mean:.1f} "
        f"+-{world * img_sec_conf:.1f}")
    if args.timeline:
        from dear_pytorch_amd.profiling import tracer
        t = tracer()
        if t:
            t.save()
            log(f"timeline written to {t.path}")
    dear.shutdown()


if __name__ == "__main__":
    main()
