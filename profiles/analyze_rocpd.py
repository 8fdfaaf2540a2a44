#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd SQLite database: total time per kernel.

Usage: python profiles/analyze_rocpd.py gpurun_out/prof1/**/*.db [out.md]
"""
import glob
import re
import sqlite3
import sys


def summarize(db_path: str):
    con = sqlite3.connect(db_path)
    cur = con.cursor()
    tbl = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table' "
        "AND name LIKE 'rocpd_kernel_dispatch%'")][0]
    uuid = tbl[len("rocpd_kernel_dispatch_"):]
    q = f"""
    SELECT s.display_name, COUNT(*), SUM(d.end - d.start),
           AVG(d.end - d.start), s.arch_vgpr_count, s.accum_vgpr_count,
           s.group_segment_size
    FROM rocpd_kernel_dispatch_{uuid} d
    JOIN rocpd_info_kernel_symbol_{uuid} s ON d.kernel_id = s.id
    GROUP BY s.display_name ORDER BY SUM(d.end - d.start) DESC
    """
    rows = cur.execute(q).fetchall()
    total = sum(r[2] for r in rows)
    out = []
    out.append(f"total GPU kernel time: {total/1e9:.3f} s "
               f"({sum(r[1] for r in rows)} dispatches)")
    out.append(f"{'%':>5} {'time_ms':>9} {'calls':>6} {'avg_us':>8} "
               f"{'vgpr':>4} {'agpr':>4} {'lds':>6}  kernel")
    for name, calls, t, avg, vgpr, agpr, lds in rows[:40]:
        short = re.sub(r"\(.*", "", name)[:90]
        out.append(f"{100*t/total:5.1f} {t/1e6:9.2f} {calls:6d} "
                   f"{avg/1e3:8.1f} {vgpr or 0:4d} {agpr or 0:4d} "
                   f"{lds or 0:6d}  {short}")
    return "\n".join(out)


if __name__ == "__main__":
    paths = []
    for a in sys.argv[1:]:
        paths += glob.glob(a, recursive=True)
    if not paths:
        paths = glob.glob("gpurun_out/**/*.db", recursive=True)
    for p in paths:
        if p.endswith(".db"):
            print(f"==== {p} ====")
            print(summarize(p))
