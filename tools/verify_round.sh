#!/bin/bash
# One-command verification of everything checkable on this machine.
# (GPU tiers — pytest -m gpu, smoke, bench — run on an MI355X via gpurun.)
set -e
cd "$(dirname "$0")/.."
echo "== build (gfx950 cross-compile)"
python __graft_entry__.py
echo "== CPU test suite (incl. gloo world_size=2)"
python -m pytest tests/ -q -m "not gpu"
echo "== bench contract (CPU dry-run)"
python bench.py --model resnet18 --batch-size 2 --steps 2 --warmup 1 --no-channels-last
echo "ALL LOCAL CHECKS PASSED"
