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
