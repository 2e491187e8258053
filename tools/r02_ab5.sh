#!/bin/bash
# Round-2 sweep #5: gloo-L2 rig retest (host-staged), k16 G8 A/B, fused
# grid/chunk fine-tune, k-sweep datapoints on the round-2 defaults.
set -u
cd "$(dirname "$0")/.."
OUT=gpurun_out/r02_ab5.log
mkdir -p gpurun_out
: > "$OUT"

run1() {
  local label="$1"; shift
  echo "### $label" >> "$OUT"
  # shellcheck disable=SC2086
  ARROW_TRAFFIC_PROBE=0 timeout 300 env "$@" \
    python bench.py --no-cpu-baseline $EXTRA >> "$OUT" 2>&1 \
    || echo "FAILED rc=$?" >> "$OUT"
}

EXTRA="--rows 20000000 --features 16 --steps 10 --warmup 3"
run1 k16_g8            ARROW_K16_G8=1
run1 k16_g8_q0         ARROW_K16_G8=1 ARROW_QUEUE=0

EXTRA="--rows 100000000 --steps 8 --warmup 2"
run1 tune_qb2560_c4    ARROW_Q_BLOCKS=2560
run1 tune_qb3072_c3    ARROW_Q_BLOCKS=3072 ARROW_Q_CHUNK=3
run1 tune_qb2048_c5    ARROW_Q_CHUNK=5
run1 tune_qb1536_c6    ARROW_Q_BLOCKS=1536 ARROW_Q_CHUNK=6

EXTRA="--rows 100000000 --features 64 --steps 8 --warmup 2"
run1 k64_100M          ARROW_DUMMY=0
EXTRA="--rows 100000000 --features 32 --steps 8 --warmup 2"
run1 k32_100M          ARROW_DUMMY=0

w2() {
  local label="$1"; shift
  echo "### $label" >> "$OUT"
  # shellcheck disable=SC2086
  ARROW_TRAFFIC_PROBE=0 timeout 240 env ARROW_BENCH_BACKEND=gloo "$@" \
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
      --master-addr 127.0.0.1 --master-port 29518 \
      bench.py --gpus 2 --rows 16000000 --steps 4 --warmup 1 \
      --no-cpu-baseline $EXTRA >> "$OUT" 2>&1 || echo "FAILED rc=$?" >> "$OUT"
}
EXTRA="--graph off --parts 2"
w2 w2_gloo_L2_staged   ARROW_DUMMY=0

echo DONE >> "$OUT"
