#!/usr/bin/env python
"""Synthetic BERT pretraining throughput benchmark (reference parity:
dear/bert_benchmark.py).  Log contract: ``Total sen/sec on N GPU(s): X``."""
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
    p.add_argument("--model", default="bert_large",
                   choices=["bert_base", "bert_large"])
    p.add_argument("--batch-size", type=int, default=32)
    p.add_argument("--sentence-len", type=int, default=128)
    p.add_argument("--num-warmup-batches", type=int, default=10)
    p.add_argument("--num-batches-per-iter", type=int, default=10)
    p.add_argument("--num-iters", type=int, default=5)
    p.add_argument("--method", default="dear",
                   choices=["dear", "dear-bo", "ddp", "wfbp", "mgwfbp",
                            "asc", "mgs", "naive", "rb", "bytescheduler"])
    p.add_argument("--compressor", default="none")
    p.add_argument("--density", type=float, default=1.0)
    p.add_argument("--threshold", type=int, default=25 * 1024 * 1024)
    p.add_argument("--no-fusion", action="store_true")
    p.add_argument("--exclude-parts", default="")
    p.add_argument("--optimizer", default="sgd", choices=["sgd", "adamw"])
    p.add_argument("--amp", action="store_true",
                   help="bf16 autocast compute (reference --fp16 axis; "
                        "gradients/optimizer stay fp32)")
    args = p.parse_args()

    import dear_pytorch_amd as dear
    from dear_pytorch_amd import models

    dear.init()
    rank, world = dear.rank(), dear.size()
    on_gpu = torch.cuda.is_available()
    device = dear.local_device()
    if on_gpu:
        torch.cuda.set_device(device)

    def log(msg):
        if rank == 0:
            print(msg, flush=True)

    cfg = models.bert_large() if args.model == "bert_large" else \
        models.bert_base()
    model = models.BertForPreTraining(cfg).to(device)
    crit = models.BertPretrainingCriterion(cfg.vocab_size).to(device)

    bs, S = args.batch_size, args.sentence_len
    g = torch.Generator().manual_seed(77 + rank)
    ids = torch.randint(0, cfg.vocab_size, (bs, S), generator=g).to(device)
    tt = torch.zeros(bs, S, dtype=torch.long, device=device)
    mask = torch.ones(bs, S, dtype=torch.long, device=device)
    mlm = torch.full((bs, S), -1, dtype=torch.long)
    sel = torch.rand(bs, S, generator=g) < 0.15
    mlm[sel] = torch.randint(0, cfg.vocab_size, (int(sel.sum()),), generator=g)
    mlm = mlm.to(device)
    nsp = torch.randint(0, 2, (bs,), generator=g).to(device)

    if world > 1:
        dear.broadcast_parameters(model.state_dict(), root_rank=0)
    base_opt = torch.optim.SGD(model.parameters(), lr=2e-5) \
        if args.optimizer == "sgd" else \
        torch.optim.AdamW(model.parameters(), lr=1e-4)
    threshold = None if args.no_fusion else args.threshold
    tuner = None
    if args.method == "ddp":
        if world > 1:
            model = torch.nn.parallel.DistributedDataParallel(
                model, gradient_as_bucket_view=True)
        opt = base_opt
    elif args.method in ("dear", "dear-bo"):
        opt = dear.DistributedOptimizer(base_opt, model=model,
                                        threshold_bytes=threshold,
                                        exclude_parts=args.exclude_parts)
        if args.method == "dear-bo":
            from dear_pytorch_amd.tuner import ThresholdTuner
            tuner = ThresholdTuner(opt)
    else:
        from dear_pytorch_amd.parallel import baselines
        kw = {}
        if args.method in ("wfbp", "mgwfbp", "asc", "mgs") and \
                args.compressor != "none" and args.density < 1.0:
            kw = dict(compressor=args.compressor, density=args.density)
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
            scores, seq_rel = model(ids, tt, mask)
            loss = crit(scores, seq_rel, mlm, nsp)
        loss.backward()
        opt.step()
        if tuner:
            tuner.step_end()

    log(f"Model: {args.model} seq{S}, bs {bs}/GPU, method {args.method}, "
        f"{world} GPU(s)")
    timeit.timeit(benchmark_step, number=args.num_warmup_batches)
    sen_secs = []
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
        sen_secs.append(bs * args.num_batches_per_iter / t)
    mean = np.mean(sen_secs)
    conf = 1.96 * np.std(sen_secs)
    log(f"Sen/sec per GPU: {mean:.1f} +-{conf:.1f}")
    log(f"Total sen/sec on {world} GPU(s): {world * mean:.1f} "
        f"+-{world * conf:.1f}")
    dear.shutdown()


if __name__ == "__main__":
    main()
