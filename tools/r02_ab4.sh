#!/bin/bash
# Round-2 sweep #4: gloo-backend world-2 rig on 1 GPU (RCCL 2.26 refuses two
# ranks per device — see DESIGN.md §round-2), k16 wave-chunk regression
# check, and a cfg5-style 50-iteration stability run.
set -u
cd "$(dirname "$0")/.."
OUT=gpurun_out/r02_ab4.log
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
run1 k16_wavechunk2        ARROW_DUMMY=0

# cfg5: 50-iteration hipGraph-captured loop on the fused default
EXTRA="--rows 100000000 --steps 50 --warmup 3"
run1 cfg5_50it_fused       ARROW_DUMMY=0

# --- world-2 on ONE GPU via gloo backend (plumbing/correctness; collective
# costs are NOT representative — the real RCCL path runs on the driver's
# 8-GPU node)
w2() {
  local label="$1"; shift
  echo "### $label" >> "$OUT"
  # shellcheck disable=SC2086
  ARROW_TRAFFIC_PROBE=0 timeout 300 env ARROW_BENCH_BACKEND=gloo "$@" \
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
      --master-addr 127.0.0.1 --master-port 29517 \
      bench.py --gpus 2 --rows 16000000 --steps 6 --warmup 2 \
      --no-cpu-baseline $EXTRA >> "$OUT" 2>&1 || echo "FAILED rc=$?" >> "$OUT"
}
EXTRA="--graph off"
w2 w2_gloo_base       ARROW_DUMMY=0
w2 w2_gloo_chunks8    ARROW_ROW0_CHUNKS=8
EXTRA="--graph off --parts 2"
w2 w2_gloo_L2         ARROW_DUMMY=0

echo DONE >> "$OUT"
