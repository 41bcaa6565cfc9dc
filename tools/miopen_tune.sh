#!/bin/bash
# One-off MIOpen exhaustive tuning for the benchmark conv shapes; writes the
# tuned perf DB into miopen_udb/ (commit it — bench.py picks it up via
# MIOPEN_USER_DB_PATH). Run on an MI355X box; takes many minutes.
set -e
REPO="$(cd "$(dirname "$0")/.." && pwd)"
mkdir -p "$REPO/miopen_udb"
export MIOPEN_USER_DB_PATH="$REPO/miopen_udb"
export MIOPEN_CUSTOM_CACHE_DIR="$REPO/miopen_udb"
export MIOPEN_FIND_MODE=NORMAL
export MIOPEN_FIND_ENFORCE=SEARCH
for model in resnet50 vgg16 densenet201 inceptionv4; do
  echo "== tuning $model"
  python "$REPO/bench.py" --model "$model" --steps 3 --warmup 2 || true
done
echo "tuned DB in $REPO/miopen_udb — commit it"
