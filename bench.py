#!/usr/bin/env python
"""Flagship benchmark: DeAR data-parallel training throughput on MI355X.

Driver contract:
    python bench.py --gpus N --steps K --warmup W
launched (N>1) via torch.distributed.run with one rank per GPU over RCCL.
Rank 0 prints ONE JSON line: whole-job samples/sec on the BASELINE.json
config (ResNet-50 bs64 fake ImageNet, fp32 — the reference's headline
`Total img/sec on N GPU(s)` protocol, dear/imagenet_benchmark.py:159-172;
--model bert_large for the BERT-Large seq128 headline).
"""
import argparse
import json
import os
import time

# MIOpen's exhaustive conv search costs minutes per fresh box; FAST find uses
# heuristics immediately (override with DEAR_MIOPEN_FIND=... if tuning runs
# are wanted).
os.environ.setdefault("MIOPEN_FIND_MODE", os.environ.get("DEAR_MIOPEN_FIND",
                                                         "FAST"))
# a committed tuned perf-DB (tools/miopen_tune.sh) is picked up automatically
_udb = os.path.join(os.path.dirname(os.path.abspath(__file__)), "miopen_udb")
if os.path.isdir(_udb):
    os.environ.setdefault("MIOPEN_USER_DB_PATH", _udb)
    os.environ.setdefault("MIOPEN_CUSTOM_CACHE_DIR", _udb)

import torch  # noqa: E402


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--model", default="resnet50",
                   help="resnet50|vgg16|densenet201|inceptionv4|bert_base|bert_large")
    p.add_argument("--batch-size", type=int, default=None,
                   help="per-GPU batch (default: 64 CNN / 32 BERT)")
    p.add_argument("--seq-len", type=int, default=128)
    p.add_argument("--method", default="dear",
                   choices=["dear", "dear-bo", "dear-wt", "ddp", "wfbp",
                            "mgwfbp", "asc", "mgs", "naive", "rb",
                            "bytescheduler"],
                   help="gradient-sync method (dear is the product)")
    p.add_argument("--threshold-mb", type=float, default=25.0)
    p.add_argument("--compressor", default="none",
                   choices=["none", "topk", "eftopk", "gaussian", "sign",
                            "efsign"],
                   help="gradient compression codec (wfbp-family methods)")
    p.add_argument("--density", type=float, default=1.0,
                   help="kept gradient fraction for sparse codecs")
    p.add_argument("--exclude-parts", default="")
    p.add_argument("--no-fusion", action="store_true")
    p.add_argument("--comm-dtype", default="fp32",
                   choices=["fp32", "bf16", "fp16"],
                   help="gradient wire format (bf16/fp16 halve xGMI bytes)")
    p.add_argument("--hipgraph", action="store_true",
                   help="capture the whole training step in a hipGraph and "
                        "replay it (removes per-kernel launch gaps)")
    p.add_argument("--fused-bn", action="store_true", default=None,
                   help="fused NHWC BN+ReLU+residual kernels for ResNets "
                        "(default on for GPU ResNets)")
    p.add_argument("--no-fused-bn", dest="fused_bn", action="store_false")
    p.add_argument("--channels-last", action="store_true", default=None,
                   help="NHWC layout for CNNs (MIOpen igemm fast path; default on)")
    p.add_argument("--no-channels-last", dest="channels_last",
                   action="store_false")
    return p.parse_args()


def build_workload(args, device):
    from dear_pytorch_amd import models
    is_bert = args.model.startswith("bert")
    if is_bert:
        bs = args.batch_size or 32
        cfg = models.bert_large() if args.model == "bert_large" else \
            models.bert_base()
        model = models.BertForPreTraining(cfg).to(device)
        crit = models.BertPretrainingCriterion(cfg.vocab_size).to(device)
        S = args.seq_len
        g = torch.Generator().manual_seed(1234)
        ids = torch.randint(0, cfg.vocab_size, (bs, S), generator=g).to(device)
        tt = torch.zeros(bs, S, dtype=torch.long, device=device)
        mask = torch.ones(bs, S, dtype=torch.long, device=device)
        mlm = torch.full((bs, S), -1, dtype=torch.long)
        sel = torch.rand(bs, S, generator=g) < 0.15
        mlm[sel] = torch.randint(0, cfg.vocab_size, (int(sel.sum()),),
                                 generator=g)
        mlm = mlm.to(device)
        nsp = torch.randint(0, 2, (bs,), generator=g).to(device)

        def step_fn(model, opt):
            opt.zero_grad()
            scores, seq_rel = model(ids, tt, mask)
            loss = crit(scores, seq_rel, mlm, nsp)
            loss.backward()
            opt.step()

        opt_fn = lambda ps: torch.optim.SGD(ps, lr=2e-5)  # reference :122
        unit, metric = "sen/sec", "sen/sec"
    else:
        bs = args.batch_size or 64
        fused_bn = args.fused_bn
        if fused_bn is None:
            fused_bn = device.type == "cuda" and args.model.startswith(
                ("resnet", "densenet", "inception"))
        model = models.get_cnn(args.model, fused_bn=fused_bn).to(device)
        g = torch.Generator().manual_seed(1234)
        res = 299 if args.model == "inceptionv4" else 224
        data = torch.randn(bs, 3, res, res, generator=g).to(device)
        target = torch.randint(0, 1000, (bs,), generator=g).to(device)
        if args.channels_last or (args.channels_last is None and
                                  device.type == "cuda"):
            model = model.to(memory_format=torch.channels_last)
            data = data.to(memory_format=torch.channels_last)
        lossf = torch.nn.CrossEntropyLoss().to(device)

        def step_fn(model, opt):
            opt.zero_grad()
            loss = lossf(model(data), target)
            loss.backward()
            opt.step()

        opt_fn = lambda ps: torch.optim.SGD(ps, lr=0.01, momentum=0.9)
        unit, metric = "img/sec", "img/sec"
    return model, opt_fn, step_fn, bs, unit, metric


def wrap_method(args, model, opt_fn):
    import dear_pytorch_amd as dear
    threshold = None if args.no_fusion else int(args.threshold_mb * 1024 * 1024)
    if args.method == "ddp":
        if dear.size() > 1:
            model = torch.nn.parallel.DistributedDataParallel(
                model, bucket_cap_mb=args.threshold_mb,
                gradient_as_bucket_view=True)
        opt = opt_fn(model.parameters())
        return model, opt
    if args.method in ("dear", "dear-bo", "dear-wt"):
        cdt = {"fp32": None, "bf16": torch.bfloat16,
               "fp16": torch.float16}[args.comm_dtype]
        opt = dear.DistributedOptimizer(opt_fn(model.parameters()),
                                        model=model, threshold_bytes=threshold,
                                        exclude_parts=args.exclude_parts,
                                        comm_dtype=cdt)
        if args.method == "dear-bo":
            from dear_pytorch_amd.tuner import ThresholdTuner
            tuner = ThresholdTuner(opt, window=3,
                                   warmup=min(args.warmup, 5), trials=8,
                                   verbose=False)
            inner = opt

            class _TunedStep:
                """step() facade driving the BO tuner around each iteration."""

                def __getattr__(self, k):
                    return getattr(inner, k)

                def zero_grad(self, *a, **k):
                    tuner.step_begin()
                    inner.zero_grad()

                def step(self, *a, **k):
                    r = inner.step()
                    tuner.step_end()
                    return r
            return model, _TunedStep()
        if args.method == "dear-wt":
            from dear_pytorch_amd.parallel.waittime import \
                WaitTimeAdaptiveFusion
            wt = WaitTimeAdaptiveFusion(opt, verbose=False)
            inner_wt = opt

            class _WtStep:
                def __getattr__(self, k):
                    return getattr(inner_wt, k)

                def step(self, *a, **k):
                    r = inner_wt.step()
                    wt.step_end()
                    return r
            return model, _WtStep()
        return model, opt
    from dear_pytorch_amd.parallel import baselines
    kw = {}
    if args.method in ("wfbp", "mgwfbp", "asc", "mgs") and \
            args.compressor != "none" and args.density < 1.0:
        kw = dict(compressor=args.compressor, density=args.density)
    opt = baselines.make(args.method, opt_fn(model.parameters()), model,
                         threshold_bytes=threshold, **kw)
    return model, opt


def _provenance_str(args):
    """Which comm backend actually served each channel (honest-numbers
    contract: 'rccl-native', 'torch-dist' or 'local').
    DDP runs on torch's own ProcessGroupNCCL."""
    if args.method == "ddp":
        import torch.distributed as dist
        return "ddp/torch-" + (dist.get_backend() if dist.is_initialized()
                               else "none")
    from dear_pytorch_amd.comm.backend import backend_provenance
    prov = backend_provenance()
    if not prov:
        return "none"
    kinds = sorted(set(prov.values()))
    return kinds[0] if len(kinds) == 1 else \
        ",".join(f"{k}={v}" for k, v in sorted(prov.items()))


def main():
    args = parse_args()
    import dear_pytorch_amd as dear
    dear.init()
    rank, world = dear.rank(), dear.size()
    on_gpu = torch.cuda.is_available()
    device = dear.local_device()
    if on_gpu:
        torch.cuda.set_device(device)
        torch.backends.cudnn.benchmark = True

    # on GPU the in-house methods must run their native RCCL channels — a
    # silent fallback to torch-dist would invalidate the measured number
    # (VERDICT r1); explicit DEAR_STRICT_NATIVE_COMM=0 opts out.
    if on_gpu and world > 1 and args.method != "ddp":
        os.environ.setdefault("DEAR_STRICT_NATIVE_COMM", "1")

    if args.gpus != world and rank == 0:
        import sys
        print(f"[bench] note: --gpus {args.gpus} but WORLD_SIZE={world}; "
              f"using the launched world", file=sys.stderr, flush=True)
    model, opt_fn, step_fn, bs, unit, metric = build_workload(args, device)
    if world > 1:
        dear.broadcast_parameters(model.state_dict(), root_rank=0)
    # hipGraph mode: hook/AccumulateGrad creation, warmup and capture must all
    # live on ONE non-default stream or backward breaks the capture
    graph_stream = torch.cuda.Stream() if (args.hipgraph and on_gpu) else None
    if graph_stream is not None:
        graph_stream.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(graph_stream):
            model, opt = wrap_method(args, model, opt_fn)
    else:
        model, opt = wrap_method(args, model, opt_fn)
    model.train()

    def sync():
        if world > 1:
            torch.distributed.barrier()
        if on_gpu:
            torch.cuda.synchronize()

    run_step = lambda: step_fn(model, opt)
    if graph_stream is not None:
        # warm up allocator/find, then capture one full iteration (fwd + bwd
        # + DeAR update enqueues — all device-static: fused buckets, grad
        # views and synthetic inputs never reallocate), all on graph_stream
        with torch.cuda.stream(graph_stream):
            for _ in range(3):
                step_fn(model, opt)
        torch.cuda.current_stream().wait_stream(graph_stream)
        torch.cuda.synchronize()
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph, stream=graph_stream):
            step_fn(model, opt)
        run_step = graph.replay

    for _ in range(args.warmup):
        run_step()
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        run_step()
    sync()
    elapsed = time.perf_counter() - t0
    # MAX over ranks
    t = torch.tensor([elapsed], dtype=torch.float64)
    if world > 1:
        t = t.to(device) if torch.distributed.get_backend() == "nccl" else t
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
    elapsed = float(t.item())

    value = world * bs * args.steps / elapsed
    if rank == 0:
        out = {
            "metric": metric,
            "value": round(value, 2),
            "unit": unit,
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": world * bs,
                "seq_len": args.seq_len if args.model.startswith("bert") else None,
                "parallelism": f"dp{world}",
                "method": args.method,
                "threshold_mb": None if args.no_fusion else args.threshold_mb,
                "compressor": None if args.compressor == "none"
                else f"{args.compressor}@{args.density}",
                "comm_backend": _provenance_str(args),
            },
        }
        print(json.dumps(out), flush=True)
    dear.shutdown()


if __name__ == "__main__":
    main()
