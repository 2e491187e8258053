#!/usr/bin/env python3
"""Measure per-launch HBM-side traffic of the spmm kernel with rocprofv3 PMC
counters, corrected per MI355X_MICROARCH.md §HBM:

  - FETCH_SIZE and WRITE_SIZE cannot share a TCC pass (3+2 of 4 slots), so
    two separate runs are made;
  - on gfx950 FETCH_SIZE reports exactly HALF of wide coalesced streaming
    reads (16 B/lane) — the dominant X/pair reads here are 16 B/lane float4
    and 8 B/lane int2, so the correction factor 2 is applied to reads and
    stated; WRITE_SIZE is uncalibrated on gfx950 and is reported raw with
    that caveat;
  - counter units are KB.

Writes gpurun_out/traffic.json: {"per_launch": [...], "avg_read_bytes": R,
"avg_write_bytes_raw": W, ...}. bench.py embeds it into roofline.traffic
when ARROW_TRAFFIC_JSON points at the file.

Usage (on the GPU box):
  python tools/measure_traffic.py --rows 20000000 --steps 2 --warmup 1
"""
import argparse
import glob
import json
import os
import sqlite3
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_pass(counters, outdir, bench_args):
    # ARROW_TRAFFIC_CHILD stops the bench child from launching its own
    # traffic probe (bench.py runs this script automatically at N=1)
    env = dict(os.environ, TMPDIR='/tmp', ARROW_TRAFFIC_CHILD='1')
    cmd = ['rocprofv3', '--pmc', *counters, '--kernel-include-regex',
           'spmm_kernel', '-d', outdir, '--',
           sys.executable, os.path.join(REPO, 'bench.py'), *bench_args,
           '--no-cpu-baseline', '--graph', 'off']
    os.makedirs(outdir, exist_ok=True)
    log = os.path.join(outdir, 'pass.log')
    try:
        with open(log, 'w') as lf:
            subprocess.run(cmd, check=True, cwd='/tmp', env=env,
                           stdout=lf, stderr=lf)
    except subprocess.CalledProcessError:
        sys.stderr.write("---- rocprofv3 pass log tail ----\n")
        with open(log) as lf:
            sys.stderr.write(''.join(lf.readlines()[-25:]))
        raise
    db = glob.glob(f'{outdir}/runc/*_results.db') + glob.glob(f'{outdir}/*_results.db')
    con = sqlite3.connect(db[0])
    cur = con.cursor()
    ts = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    t_disp = [t for t in ts if 'kernel_dispatch' in t][0]
    t_sym = [t for t in ts if 'info_kernel_symbol' in t][0]
    t_pmc = [t for t in ts if t.startswith('rocpd_pmc_event')][0]
    t_ipmc = [t for t in ts if 'info_pmc' in t][0]
    out = {}
    for name, disp, val in cur.execute(f"""
        SELECT i.name, d.dispatch_id, SUM(p.value) FROM {t_pmc} p
        JOIN {t_disp} d ON p.event_id=d.event_id
        JOIN {t_sym} s ON d.kernel_id=s.id
        JOIN {t_ipmc} i ON p.pmc_id=i.id
        WHERE s.display_name LIKE '%spmm_kernel%'
        GROUP BY 1, 2"""):
        out.setdefault(name, {})[disp] = val
    return out


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--rows', default='20000000')
    ap.add_argument('--features', default='128')
    ap.add_argument('--band', default='1024')
    ap.add_argument('--steps', default='2')
    ap.add_argument('--warmup', default='1')
    ap.add_argument('--out', default=os.path.join(REPO, 'gpurun_out', 'traffic.json'))
    a = ap.parse_args()
    bench_args = ['--rows', a.rows, '--features', a.features, '--band', a.band,
                  '--steps', a.steps, '--warmup', a.warmup]
    # rocprof DBs are large — keep them OUT of gpurun_out (its merge-back
    # is capped at 64 MiB); only the small summary json goes there
    base = os.path.join('/tmp', 'arrow_traffic')
    fetch = run_pass(['FETCH_SIZE'], os.path.join(base, 'tr_fetch'), bench_args)
    write = run_pass(['WRITE_SIZE'], os.path.join(base, 'tr_write'), bench_args)
    fvals = sorted(fetch.get('FETCH_SIZE', {}).items())
    wvals = sorted(write.get('WRITE_SIZE', {}).items())
    # KB -> bytes; gfx950 FETCH_SIZE = 1/2 of wide coalesced reads
    reads = [v * 1024 * 2 for _, v in fvals]
    writes = [v * 1024 for _, v in wvals]
    result = {
        "workload": {"rows": int(a.rows), "steps": int(a.steps)},
        "read_bytes_per_launch": reads,
        "write_bytes_per_launch_raw": writes,
        "avg_read_bytes": sum(reads) / max(len(reads), 1),
        "avg_write_bytes_raw": sum(writes) / max(len(writes), 1),
        "notes": "reads corrected x2 (gfx950 FETCH_SIZE undercount for wide "
                 "coalesced loads); WRITE_SIZE uncalibrated on gfx950 "
                 "(MI355X_MICROARCH.md §HBM); L3 hits are counted (memory-"
                 "side L2 counters)",
    }
    os.makedirs(os.path.dirname(a.out), exist_ok=True)
    with open(a.out, 'w') as f:
        json.dump(result, f, indent=1)
    print(json.dumps({k: result[k] for k in
                      ('avg_read_bytes', 'avg_write_bytes_raw')}))
    print("wrote", a.out)


if __name__ == '__main__':
    main()
