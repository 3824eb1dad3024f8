#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd SQLite results.db: per-kernel total time,
count, mean, grid — the per-kernel view the optimization loop works from."""
import glob
import re
import sqlite3
import sys


def summarize(db_path, top=40):
    con = sqlite3.connect(db_path)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    rows = cur.execute(f"""
        SELECT s.display_name, COUNT(*), SUM(d.end - d.start),
               AVG(d.end - d.start), MAX(d.grid_size_x)
        FROM {disp} d JOIN {sym} s ON d.kernel_id = s.id
        GROUP BY s.display_name ORDER BY SUM(d.end - d.start) DESC
    """).fetchall()
    total = sum(r[2] for r in rows)
    print(f"== {db_path}: total GPU time {total/1e6:.1f} ms over "
          f"{sum(r[1] for r in rows)} dispatches ==")
    print(f"{'%':>6} {'total_ms':>9} {'n':>6} {'mean_us':>9} {'maxgrid':>8}  kernel")
    for name, n, tot, avg, grid in rows[:top]:
        short = re.sub(r"<[^>]*>", "<>", name)[:95]
        print(f"{100*tot/total:6.2f} {tot/1e6:9.2f} {n:6d} {avg/1e3:9.1f} "
              f"{grid:8d}  {short}")


if __name__ == "__main__":
    paths = sys.argv[1:] or glob.glob("gpurun_out/prof*/**/*_results.db", recursive=True)
    for p in paths:
        summarize(p)
