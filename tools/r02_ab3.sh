#!/bin/bash
# Round-2 sweep #3: real-file pipeline on GPU, cfg3 captured loop, and the
# world>1 validation rig (2 ranks oversubscribed on 1 GPU): hipGraph capture
# with RCCL + ARROW_ROW0_CHUNKS sweep.
set -u
cd "$(dirname "$0")/.."
OUT=gpurun_out/r02_ab3.log
mkdir -p gpurun_out
: > "$OUT"

echo "### mtx_e2e" >> "$OUT"
ARROW_TRAFFIC_PROBE=0 timeout 600 python tools/run_mtx_e2e.py >> "$OUT" 2>&1 \
  || echo "FAILED rc=$?" >> "$OUT"

echo "### cfg3_captured" >> "$OUT"
ARROW_TRAFFIC_PROBE=0 timeout 600 python tools/run_cfg3.py >> "$OUT" 2>&1 \
  || echo "FAILED rc=$?" >> "$OUT"

# --- 2 ranks oversubscribed on ONE GPU: correctness/plumbing validation of
# the world>1 path (absolute times are not meaningful; hangs/errors are)
w2() {
  local label="$1"; shift
  echo "### $label" >> "$OUT"
  # shellcheck disable=SC2086
  ARROW_TRAFFIC_PROBE=0 timeout 300 env "$@" \
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
      --master-addr 127.0.0.1 --master-port 29517 \
      bench.py --gpus 2 --rows 16000000 --steps 6 --warmup 2 \
      --no-cpu-baseline $EXTRA >> "$OUT" 2>&1 || echo "FAILED rc=$?" >> "$OUT"
}
EXTRA="--graph on"
w2 w2_graph_on        ARROW_DUMMY=0
EXTRA="--graph off"
w2 w2_chunks1         ARROW_ROW0_CHUNKS=1
w2 w2_chunks4         ARROW_ROW0_CHUNKS=4
w2 w2_chunks8         ARROW_ROW0_CHUNKS=8
EXTRA="--graph off --parts 2"
w2 w2_L2_overlap      ARROW_DUMMY=0

# --- single-GPU regression checks on the new defaults ---------------------
run1() {
  local label="$1"; shift
  echo "### $label" >> "$OUT"
  # shellcheck disable=SC2086
  ARROW_TRAFFIC_PROBE=0 timeout 300 env "$@" \
    python bench.py --no-cpu-baseline $EXTRA >> "$OUT" 2>&1 \
    || echo "FAILED rc=$?" >> "$OUT"
}
EXTRA="--rows 20000000 --steps 10 --warmup 3"
run1 chk_20M_chunk4        ARROW_DUMMY=0
EXTRA="--rows 20000000 --features 16 --steps 10 --warmup 3"
run1 chk_k16_new_default   ARROW_DUMMY=0
run1 chk_k16_twolaunch     ARROW_FUSE_ALL=0
EXTRA="--rows 100000000 --steps 8 --warmup 2"
run1 chk_100M_final        ARROW_DUMMY=0

# --- PMC traffic on the fused kernel (DBs now under /tmp) -----------------
echo "### traffic_fused_100M" >> "$OUT"
timeout 500 python tools/measure_traffic.py --rows 100000000 --steps 2 --warmup 1 \
  --out gpurun_out/traffic_fused_100M.json >> "$OUT" 2>&1 || echo "FAILED" >> "$OUT"
cat gpurun_out/traffic_fused_100M.json >> "$OUT" 2>/dev/null

echo DONE >> "$OUT"
