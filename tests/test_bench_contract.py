"""bench.py driver-contract test: one JSON line on stdout with the required
fields (the round driver parses this)."""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"}


@pytest.mark.timeout(300)
def test_bench_json_contract_cpu():
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--model", "resnet18",
         "--batch-size", "2", "--steps", "2", "--warmup", "1",
         "--no-channels-last"],
        capture_output=True, text=True, timeout=280, cwd=REPO)
    assert r.returncode == 0, r.stderr[-1500:]
    lines = [l for l in r.stdout.strip().splitlines() if l.strip()]
    assert len(lines) == 1, f"stdout must be exactly one JSON line: {lines}"
    d = json.loads(lines[0])
    assert REQUIRED <= set(d), REQUIRED - set(d)
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["dtype"] == "fp32" and d["data"] == "synthetic"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    cfg = d["config"]
    assert cfg["model"] == "resnet18" and cfg["parallelism"] == "dp1"
    assert cfg["global_batch"] == 2


def test_matrix_runner_contract_parsing():
    sys.path.insert(0, os.path.join(REPO, "benchmarks"))
    import run_matrix
    txt = "noise\nImg/sec per GPU: 10 +-1\nTotal img/sec on 8 GPU(s): 12345.6 +-7.8\n"
    assert run_matrix.extract_total(txt) == 12345.6
    assert run_matrix.extract_total("nothing here") is None
    cmd = run_matrix.gen_cmd("imagenet", "resnet50", 64, "dear-notf", 8, 5)
    assert "--no-fusion" in cmd and "--method" in cmd
    cmd = run_matrix.gen_cmd("bert", "bert_large", 32, "mgwfbp", 4, 5)
    assert "bert_benchmark.py" in " ".join(cmd)
    assert "--nproc-per-node=4" in " ".join(cmd)
    cmd = run_matrix.gen_cmd("imagenet", "resnet50", 64, "dear-nors", 8, 5)
    assert "reducescatter" in cmd


@pytest.mark.timeout(600)
def test_bench_ws2_feature_flags_cpu():
    """bench CLI over torchrun ws=2 (gloo) with reduced-precision wire and
    the BO-tuned method — the flag paths a scale run may use."""
    base = [sys.executable, "-m", "torch.distributed.run", "--standalone",
            "--local-addr", "127.0.0.1", "--nproc-per-node", "2",
            os.path.join(REPO, "bench.py"), "--model", "resnet18",
            "--batch-size", "2", "--steps", "2", "--warmup", "1",
            "--no-channels-last"]
    for extra in (["--comm-dtype", "fp16"], ["--method", "dear-bo"]):
        r = subprocess.run(base + extra, capture_output=True, text=True,
                           timeout=280, cwd=REPO)
        assert r.returncode == 0, (extra, r.stderr[-1500:])
        line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
        d = json.loads(line)
        assert d["n_gpus"] == 2 and d["value"] > 0
