#!/bin/bash
# 8-GPU day-one probe: the moment a multi-GPU MI355X node is
# available, this script measures — in minutes — everything round 1 could not:
#   1. xGMI alpha-beta fit over the native RCCL channels (comm_bench)
#   2. RCCL env sweep: NCCL_MIN_NCHANNELS spreading rings over the 7 xGMI
#      links, and the ring-vs-tree algo axis
#   3. DeAR vs DDP scaling at 2/4/8 GPUs (headline ResNet-50 bs64 config)
#   4. BO threshold re-tune at 8 GPUs (the 25 MB default is an Ethernet-era
#      value; xGMI may want bigger buckets)
# Outputs land under gpurun_out/scale_probe/ — copy the summaries you keep
# into profiles/.
#
# Usage: bash scripts/scale_probe.sh [MAX_GPUS]
set -u
cd "$(dirname "$0")/.."
MAXG=${1:-8}
OUT=gpurun_out/scale_probe
mkdir -p "$OUT"
STEPS=${STEPS:-15}
WARM=${WARM:-5}
TR="python -m torch.distributed.run --nnodes=1 --master-addr 127.0.0.1"
PORT=29540

run() { # name, cmd...
  local name=$1; shift
  echo "=== $name: $*" | tee -a "$OUT/probe.log"
  timeout 600 "$@" >"$OUT/$name.log" 2>&1
  local rc=$?
  echo "=== $name rc=$rc" | tee -a "$OUT/probe.log"
  tail -n 5 "$OUT/$name.log" | tee -a "$OUT/probe.log"
}

# 1. alpha-beta fit on the real fabric
run comm_bench_${MAXG}g $TR --master-port $((PORT++)) --nproc-per-node "$MAXG" \
    tools/comm_bench.py --sizes-mb 1,4,16,25,64,128

# 2. RCCL env sweep at MAXG GPUs (bucket-sized all-reduce bandwidth)
for ch in "" 4 7 14; do
  for algo in "" Ring Tree; do
    tag="env_ch${ch:-def}_algo${algo:-def}"
    env_args=()
    [ -n "$ch" ] && export NCCL_MIN_NCHANNELS=$ch || unset NCCL_MIN_NCHANNELS
    [ -n "$algo" ] && export NCCL_ALGO=$algo || unset NCCL_ALGO
    run "$tag" $TR --master-port $((PORT++)) --nproc-per-node "$MAXG" \
        tools/comm_bench.py --sizes-mb 25 --ops all_reduce
  done
done
unset NCCL_MIN_NCHANNELS NCCL_ALGO 2>/dev/null

# 3. DeAR vs DDP scaling curve
for n in 1 2 4 8; do
  [ "$n" -gt "$MAXG" ] && break
  for m in dear ddp; do
    if [ "$n" = 1 ]; then
      run "bench_${m}_n1" python bench.py --gpus 1 --steps $STEPS --warmup $WARM --method $m
    else
      run "bench_${m}_n$n" $TR --master-port $((PORT++)) --nproc-per-node "$n" \
          bench.py --gpus "$n" --steps $STEPS --warmup $WARM --method $m
    fi
  done
done

# 4. BO threshold re-tune at MAXG GPUs (longer run so the tuner locks)
run bo_tune_${MAXG}g $TR --master-port $((PORT++)) --nproc-per-node "$MAXG" \
    bench.py --gpus "$MAXG" --steps 60 --warmup 10 --method dear-bo

echo "scale probe done; results in $OUT" | tee -a "$OUT/probe.log"
grep -h '"metric"' "$OUT"/bench_*.log > "$OUT/scaling_summary.jsonl" 2>/dev/null || true
cat "$OUT/scaling_summary.jsonl"
