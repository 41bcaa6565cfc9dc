#!/usr/bin/env python
"""Batch experiment harness (reference parity: benchmarks.py).

Runs the method x model matrix via torchrun (one process per GPU over RCCL),
parses the ``Total ... GPU(s): X`` contract lines, resumes via exp.log, and
writes reports.json.

    python benchmarks/run_matrix.py --gpus 8 --set tf
"""
import argparse
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

# (driver, model, per-GPU batch) — reference task list (benchmarks.py:16-27)
TASKS = [
    ("imagenet", "resnet50", 64),
    ("imagenet", "densenet201", 32),
    ("imagenet", "inceptionv4", 64),
    ("bert", "bert_base", 64),
    ("bert", "bert_large", 32),
]
METHOD_SETS = {
    # with tensor fusion vs without (reference tf/notf sets)
    "tf": ["ddp", "mgwfbp", "dear"],
    "notf": ["wfbp", "naive", "dear-notf", "bytescheduler"],
    "all": ["ddp", "wfbp", "mgwfbp", "asc", "mgs", "naive", "rb",
            "bytescheduler", "dear", "dear-bo", "dear-wt"],
    # merge-planner comparison (mgwfbp vs asc vs mgs, reference hv_* modes)
    "planners": ["mgwfbp", "asc", "mgs", "dear"],
    # time-breakdown ablations (reference dear/batch.sh exclude_parts sweeps)
    "breakdown": ["dear", "dear-nors", "dear-noag"],
}


def gen_cmd(driver, model, bs, method, gpus, iters):
    script = os.path.join(REPO, "benchmarks",
                          "imagenet_benchmark.py" if driver == "imagenet"
                          else "bert_benchmark.py")
    cmd = [sys.executable, "-m", "torch.distributed.run", "--standalone",
           "--local-addr", "127.0.0.1", f"--nproc-per-node={gpus}", script,
           "--model", model, "--batch-size", str(bs),
           "--num-iters", str(iters)]
    if method == "dear-notf":
        cmd += ["--method", "dear", "--no-fusion"]
    elif method == "dear-nors":
        cmd += ["--method", "dear", "--exclude-parts", "reducescatter"]
    elif method == "dear-noag":
        cmd += ["--method", "dear", "--exclude-parts", "allgather"]
    elif method == "wfbp":
        cmd += ["--method", "wfbp", "--no-fusion"]
    else:
        cmd += ["--method", method]
    return cmd


def extract_total(text):
    for line in text.splitlines():
        if "Total" in line and "GPU(s)" in line:
            try:
                return float(line.rsplit(": ", 1)[1].split()[0])
            except (IndexError, ValueError):
                continue
    return None


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=8)
    p.add_argument("--set", default="tf", choices=list(METHOD_SETS))
    p.add_argument("--num-iters", type=int, default=5)
    p.add_argument("--log", default="exp.log")
    p.add_argument("--out", default="reports.json")
    p.add_argument("--timeout", type=int, default=1800)
    p.add_argument("--models", default="",
                   help="comma filter, e.g. resnet50,bert_large")
    args = p.parse_args()

    done = set()
    if os.path.exists(args.log):
        done = {l.strip() for l in open(args.log)}
    reports = {}
    if os.path.exists(args.out):
        reports = json.load(open(args.out))

    want = {m for m in args.models.split(",") if m}
    for driver, model, bs in TASKS:
        if want and model not in want:
            continue
        for method in METHOD_SETS[args.set]:
            key = f"{driver}/{model}/bs{bs}/{method}/g{args.gpus}"
            if key in done:
                print(f"skip {key} (exp.log)")
                continue
            cmd = gen_cmd(driver, model, bs, method, args.gpus,
                          args.num_iters)
            print("run:", " ".join(cmd), flush=True)
            try:
                r = subprocess.run(cmd, capture_output=True, text=True,
                                   timeout=args.timeout, cwd=REPO)
                total = extract_total(r.stdout)
            except subprocess.TimeoutExpired:
                total = None
            reports[key] = total
            print(f"  -> {total}")
            with open(args.out, "w") as f:
                json.dump(reports, f, indent=2)
            with open(args.log, "a") as f:
                f.write(key + "\n")
    print(json.dumps(reports, indent=2))


if __name__ == "__main__":
    main()
