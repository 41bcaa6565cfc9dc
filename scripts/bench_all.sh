#!/bin/bash
# Full method x model matrix on N local GPUs (reference batch experiments).
N=${1:-8}
SET=${2:-tf}
exec python "$(dirname "$0")/../benchmarks/run_matrix.py" --gpus "$N" --set "$SET"
