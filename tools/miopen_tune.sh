#!/bin/bash
# One-off MIOpen exhaustive tuning for the benchmark conv shapes; writes the
# tuned perf DB into gpurun_out/miopen_udb (merged back by gpurun) — copy it
# to miopen_udb/ and commit; bench.py picks it up via MIOPEN_USER_DB_PATH.
# Run on an MI355X box; takes many minutes per model.
#   bash tools/miopen_tune.sh [model ...]   (default: resnet50)
set -e
REPO="$(cd "$(dirname "$0")/.." && pwd)"
DB="$REPO/gpurun_out/miopen_udb"
mkdir -p "$DB"
# seed with any committed DB so tuning is incremental
[ -d "$REPO/miopen_udb" ] && cp -r "$REPO/miopen_udb/." "$DB/" 2>/dev/null || true
export MIOPEN_USER_DB_PATH="$DB"
export MIOPEN_CUSTOM_CACHE_DIR="$DB"
export DEAR_MIOPEN_FIND=NORMAL
export MIOPEN_FIND_MODE=NORMAL
export MIOPEN_FIND_ENFORCE=SEARCH
MODELS=${@:-resnet50}
for model in $MODELS; do
  echo "== tuning $model"
  python "$REPO/bench.py" --model "$model" --steps 3 --warmup 2 || true
done
echo "== tuned DB contents:"
ls -la "$DB"
echo "== re-benching with tuned DB (FIND_ENFORCE off):"
unset MIOPEN_FIND_ENFORCE
export MIOPEN_FIND_MODE=FAST
for model in $MODELS; do
  python "$REPO/bench.py" --model "$model" --steps 15 --warmup 5 || true
done
