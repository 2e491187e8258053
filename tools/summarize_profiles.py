#!/usr/bin/env python3
"""Summarise rocprofv3 result databases (gpurun_out/<dir>/runc/*_results.db)
into the text summaries committed under profiles/.

Usage: python tools/summarize_profiles.py gpurun_out/prof [out.txt]
"""
import glob
import sqlite3
import sys


def summarize(db_dir, out=None):
    dbs = glob.glob(f'{db_dir}/runc/*_results.db') or glob.glob(f'{db_dir}/*_results.db')
    if not dbs:
        raise SystemExit(f"no results.db under {db_dir}")
    con = sqlite3.connect(dbs[0])
    cur = con.cursor()
    ts = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    t_disp = [t for t in ts if 'kernel_dispatch' in t][0]
    t_sym = [t for t in ts if 'info_kernel_symbol' in t][0]
    lines = [f"summary of {dbs[0]}", "",
             "kernel time (total ms | calls | avg ms | vgpr | name):"]
    for r in cur.execute(f"""SELECT s.display_name, COUNT(*),
            SUM(d.end-d.start)/1e6, AVG(d.end-d.start)/1e6, s.arch_vgpr_count
        FROM {t_disp} d JOIN {t_sym} s ON d.kernel_id=s.id
        GROUP BY 1 ORDER BY 3 DESC LIMIT 15"""):
        lines.append(f"{r[2]:10.3f} | {r[1]:5d} | {r[3]:8.4f} | {r[4]:3d} | {r[0][:90]}")
    t_pmc = [t for t in ts if t.startswith('rocpd_pmc_event')]
    if t_pmc:
        t_ipmc = [t for t in ts if 'info_pmc' in t][0]
        rows = list(cur.execute(f"""SELECT i.name, s.display_name, COUNT(*), AVG(p.value), SUM(p.value)
            FROM {t_pmc[0]} p JOIN {t_disp} d ON p.event_id=d.event_id
            JOIN {t_sym} s ON d.kernel_id=s.id JOIN {t_ipmc} i ON p.pmc_id=i.id
            GROUP BY 1,2 ORDER BY 2,1"""))
        if rows:
            lines += ["", "PMC (counter | avg/instance | n | kernel).",
                      "gfx950 notes (MI355X_MICROARCH.md): FETCH_SIZE reports 1/2 of wide",
                      "coalesced reads and counts Infinity-Cache hits; SQ_* are quad-cycles."]
            for r in rows:
                lines.append(f"{r[0]:22s} | {r[3]:.4e} | {r[2]:5d} | {r[1][:70]}")
    text = "\n".join(lines) + "\n"
    if out:
        open(out, 'w').write(text)
    else:
        print(text)


if __name__ == '__main__':
    summarize(sys.argv[1], sys.argv[2] if len(sys.argv) > 2 else None)
