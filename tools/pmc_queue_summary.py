#!/usr/bin/env python3
"""Summarize SQ/TCC PMC counters for the spmm kernel from a rocprofv3 -d dir
(queue-scheduler evidence run; compare with profiles/r01_pmc_sq_tcc_20M.txt)."""
import glob
import sqlite3
import sys

d = sys.argv[1] if len(sys.argv) > 1 else 'gpurun_out/q3_pmc'
dbs = glob.glob(f'{d}/**/*_results.db', recursive=True)
print('dbs:', dbs)
for db in dbs:
    con = sqlite3.connect(db)
    cur = con.cursor()
    ts = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    t_disp = [t for t in ts if 'kernel_dispatch' in t][0]
    t_sym = [t for t in ts if 'info_kernel_symbol' in t][0]
    t_pmc = [t for t in ts if t.startswith('rocpd_pmc_event')][0]
    t_ipmc = [t for t in ts if 'info_pmc' in t][0]
    q = (f"SELECT i.name, COUNT(DISTINCT d.dispatch_id), SUM(p.value) "
         f"FROM {t_pmc} p JOIN {t_disp} d ON p.event_id=d.event_id "
         f"JOIN {t_sym} s ON d.kernel_id=s.id "
         f"JOIN {t_ipmc} i ON p.pmc_id=i.id "
         f"WHERE s.display_name LIKE '%spmm_kernel%' GROUP BY 1")
    vals = {}
    for name, nd, val in cur.execute(q):
        vals[name] = (nd, val)
        print(f'{name}: dispatches={nd} total={val:.6g} '
              f'per_dispatch={val / nd:.6g}')
    if 'TCC_HIT' in vals and 'TCC_MISS' in vals:
        h, m = vals['TCC_HIT'][1], vals['TCC_MISS'][1]
        print(f'L2 hit rate: {h / (h + m):.3f}')
    if 'SQ_WAVE_CYCLES' in vals and 'SQ_WAIT_ANY' in vals:
        print(f'wait frac: {vals["SQ_WAIT_ANY"][1] / vals["SQ_WAVE_CYCLES"][1]:.3f}')
