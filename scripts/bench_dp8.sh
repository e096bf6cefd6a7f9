#!/usr/bin/env bash
# Multi-GPU scaling bench: runs the flagship bench.py AND bench_suite
# configs 3-5 (partitionBy all-to-all, inference all-reduce, gzip ByteArray)
# at N = 1, 2, 4, 8 ranks (capped at the visible GPU count), one JSON line
# per (config, N) appended to $OUT_DIR/dpN_<config>.json.
#
# Usage: scripts/bench_dp8.sh [rows] [reps]   (defaults: 1000000 3)
set -u
cd "$(dirname "$0")/.."

ROWS="${1:-1000000}"
REPS="${2:-3}"
OUT_DIR="${OUT_DIR:-gpurun_out/dp_scale}"
mkdir -p "$OUT_DIR"

NGPU=$(python -c 'import torch; print(torch.cuda.device_count())')
echo "# visible GPUs: $NGPU"
export HSA_ENABLE_IPC_MODE_LEGACY=${HSA_ENABLE_IPC_MODE_LEGACY:-0}

run_n () {  # run_n <nproc> <cmd...>
  local n=$1; shift
  if [ "$n" -eq 1 ]; then
    python "$@"
  else
    python -m torch.distributed.run --nnodes=1 --nproc-per-node "$n" \
      --master-addr 127.0.0.1 --master-port $((29510 + n)) "$@"
  fi
}

for N in 1 2 4 8; do
  [ "$N" -gt "$NGPU" ] && break
  echo "=== dp$N flagship (bench.py) ==="
  run_n "$N" bench.py --gpus "$N" --steps 10 --warmup 3 --rows "$ROWS" \
    | tee "$OUT_DIR/dp${N}_flagship.json"
  for CFG in partitionby infer gzip_bytearray; do
    echo "=== dp$N $CFG ==="
    run_n "$N" bench_suite.py "$CFG" --rows "$ROWS" --reps "$REPS" \
      | tee "$OUT_DIR/dp${N}_${CFG}.json"
  done
done
echo "done; per-N JSON under $OUT_DIR/"
