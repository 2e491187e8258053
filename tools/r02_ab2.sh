#!/bin/bash
# Round-2 sweep #2: tune the fused single-launch default; fold L=2; PMC.
set -u
cd "$(dirname "$0")/.."
OUT=gpurun_out/r02_ab2.log
mkdir -p gpurun_out
: > "$OUT"

run() {
  local label="$1"; shift
  echo "### $label" >> "$OUT"
  # shellcheck disable=SC2086
  ARROW_TRAFFIC_PROBE=0 timeout 300 env "$@" \
    python bench.py --no-cpu-baseline $EXTRA \
    2>> "$OUT" >> "$OUT" || echo "FAILED rc=$?" >> "$OUT"
}

# --- fused default tuning at 100M -----------------------------------------
EXTRA="--rows 100000000 --steps 8 --warmup 2"
run fused_default       ARROW_DUMMY=0
run fused_chunk1        ARROW_Q_CHUNK=1
run fused_chunk4        ARROW_Q_CHUNK=4
run fused_qb1024        ARROW_Q_BLOCKS=1024
run fused_qb4096        ARROW_Q_BLOCKS=4096
run fused_nt0           ARROW_SPMM_NT=0
run splitcol_queue      ARROW_FUSE_ALL=0 ARROW_SPLIT_COL=1
run splitcol_q_colsort  ARROW_FUSE_ALL=0 ARROW_SPLIT_COL=1 ARROW_ROW0_COLSORT=1

# --- k=16 on the fused default --------------------------------------------
EXTRA="--rows 20000000 --features 16"
run k16_fused           ARROW_DUMMY=0
run k16_fused_q1        ARROW_QUEUE=1
run k16_fused_q1_qw     ARROW_QUEUE=1 ARROW_QWAVE=1

# --- L=2 folded exchange (round-1: 2629 GF/s sequential) ------------------
EXTRA="--rows 20000000 --parts 2 --steps 10 --warmup 3"
run L2_fold             ARROW_DUMMY=0
run L2_seq              ARROW_FOLD=0
EXTRA="--rows 100000000 --parts 2 --steps 6 --warmup 2"
run L2_fold_100M        ARROW_DUMMY=0

# --- PMC: fused traffic at 100M + L2 hit rate ------------------------------
echo "### traffic_fused_100M" >> "$OUT"
timeout 500 python tools/measure_traffic.py --rows 100000000 --steps 2 --warmup 1 \
  --out gpurun_out/traffic_fused_100M.json >> "$OUT" 2>&1 || echo "FAILED" >> "$OUT"
echo "### tcc_hit_fused" >> "$OUT"
( cd /tmp && export TMPDIR=/tmp && \
  timeout 400 rocprofv3 --pmc TCC_HIT_sum TCC_MISS_sum \
    --kernel-include-regex spmm_kernel -d /root/repo/gpurun_out/tcc_fused -- \
    python /root/repo/bench.py --rows 100000000 --steps 2 --warmup 1 \
      --no-cpu-baseline --graph off ) >> "$OUT" 2>&1 \
  && python - >> "$OUT" 2>&1 << 'PYEOF'
import glob, sqlite3
db = (glob.glob('/root/repo/gpurun_out/tcc_fused/**/*_results.db', recursive=True))
con = sqlite3.connect(db[0]); cur = con.cursor()
ts = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
t_disp = [t for t in ts if 'kernel_dispatch' in t][0]
t_sym = [t for t in ts if 'info_kernel_symbol' in t][0]
t_pmc = [t for t in ts if t.startswith('rocpd_pmc_event')][0]
t_ipmc = [t for t in ts if 'info_pmc' in t][0]
for row in cur.execute(f"""
    SELECT i.name, d.dispatch_id, SUM(p.value) FROM {t_pmc} p
    JOIN {t_disp} d ON p.event_id=d.event_id
    JOIN {t_sym} s ON d.kernel_id=s.id
    JOIN {t_ipmc} i ON p.pmc_id=i.id
    WHERE s.display_name LIKE '%spmm_kernel%' GROUP BY 1,2"""):
    print(row)
PYEOF

echo DONE >> "$OUT"
