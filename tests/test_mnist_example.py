"""MNIST convergence integration test — single process and gloo ws=2
(reference's de-facto end-to-end test, examples/mnist/pytorch_mnist.py)."""
import subprocess
import sys
import os

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(600)
def test_mnist_converges_single_process():
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "examples", "mnist.py"),
         "--epochs", "2", "--train-size", "2000", "--test-size", "500"],
        capture_output=True, text=True, timeout=540, cwd=REPO)
    assert r.returncode == 0, r.stderr[-2000:]
    accs = [float(l.rsplit("accuracy ", 1)[1].rstrip("%\n"))
            for l in r.stdout.splitlines() if "accuracy" in l]
    assert accs and accs[-1] > 80.0, r.stdout[-2000:]


@pytest.mark.timeout(600)
def test_mnist_converges_ws2_gloo():
    env = dict(os.environ)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--standalone",
         "--local-addr", "127.0.0.1", "--nproc-per-node", "2",
         os.path.join(REPO, "examples", "mnist.py"),
         "--epochs", "2", "--train-size", "2000", "--test-size", "500"],
        capture_output=True, text=True, timeout=540, cwd=REPO, env=env)
    assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-2000:])
    accs = [float(l.rsplit("accuracy ", 1)[1].rstrip("%\n"))
            for l in r.stdout.splitlines() if "accuracy" in l]
    assert accs and accs[-1] > 80.0, r.stdout[-2000:]
