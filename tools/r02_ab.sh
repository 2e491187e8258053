#!/bin/bash
# Round-2 kernel A/B sweep (runs on the GPU box via gpurun).
# Each line: one bench.py invocation under a named env config; results are
# appended as "### <label>" + the bench JSON line to gpurun_out/r02_ab.log.
set -u
cd "$(dirname "$0")/.."
OUT=gpurun_out/r02_ab.log
mkdir -p gpurun_out
: > "$OUT"

run() {
  local label="$1"; shift
  echo "### $label" >> "$OUT"
  # shellcheck disable=SC2086
  ARROW_TRAFFIC_PROBE=0 timeout 240 env "$@" \
    python bench.py --no-cpu-baseline --steps 10 --warmup 3 $EXTRA \
    2>> "$OUT" >> "$OUT" || echo "FAILED rc=$?" >> "$OUT"
}

# --- broad sweep at 20M rows, k=128 ---------------------------------------
EXTRA="--rows 20000000"
run base_20M            ARROW_DUMMY=0
run fuse_all_20M        ARROW_FUSE_ALL=1
run par_row0_512_20M    ARROW_PAR_ROW0=1 ARROW_Q_BLOCKS_ROW0=512 ARROW_Q_BLOCKS_REST=1536
run par_row0_1024_20M   ARROW_PAR_ROW0=1 ARROW_Q_BLOCKS_ROW0=1024 ARROW_Q_BLOCKS_REST=1024
run colsort2_20M        ARROW_ROW0_COLSORT=2
run qwave_20M           ARROW_QWAVE=1
run qwave_c8_20M        ARROW_QWAVE=1 ARROW_Q_CHUNK=8
run qblocks1024_20M     ARROW_Q_BLOCKS=1024
run qblocks4096_20M     ARROW_Q_BLOCKS=4096

# --- k=16 (the reference default feature width) ---------------------------
EXTRA="--rows 20000000 --features 16"
run k16_base_20M        ARROW_DUMMY=0
run k16_qwave_20M       ARROW_QUEUE=1 ARROW_QWAVE=1
run k16_qwave_c8_20M    ARROW_QUEUE=1 ARROW_QWAVE=1 ARROW_Q_CHUNK=8

# --- top candidates at the headline 100M ----------------------------------
EXTRA="--rows 100000000 --steps 8 --warmup 2"
run base_100M           ARROW_DUMMY=0
run fuse_all_100M       ARROW_FUSE_ALL=1
run par_row0_512_100M   ARROW_PAR_ROW0=1 ARROW_Q_BLOCKS_ROW0=512 ARROW_Q_BLOCKS_REST=1536
run colsort2_100M       ARROW_ROW0_COLSORT=2

echo DONE >> "$OUT"
