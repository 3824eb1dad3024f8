#!/usr/bin/env python3
"""Summarize rocprofv3 PMC counters per kernel from a rocpd results.db.
Prints per-kernel aggregated counter values plus derived MFMA utilization
(SQ_VALU_MFMA_BUSY_CYCLES / SQ_WAVE_CYCLES per-wave proxy)."""
import glob
import re
import sqlite3
import sys


def summarize(db_path, top=25):
    con = sqlite3.connect(db_path)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]

    def tab(prefix):
        return next((t for t in tables if t.startswith(prefix)), None)

    pmc_ev = tab("rocpd_pmc_event")
    pmc_info = tab("rocpd_info_pmc")
    disp = tab("rocpd_kernel_dispatch")
    sym = tab("rocpd_info_kernel_symbol")
    if pmc_ev is None:
        print(f"{db_path}: no pmc tables")
        return
    cols_ev = [r[1] for r in cur.execute(f"PRAGMA table_info({pmc_ev})")]
    cols_info = [r[1] for r in cur.execute(f"PRAGMA table_info({pmc_info})")]
    print(f"# pmc_event cols: {cols_ev}")
    print(f"# info_pmc cols: {cols_info}")
    # typical schema: pmc_event(pmc_id, event_id, value); dispatch has event_id
    try:
        rows = cur.execute(f"""
            SELECT s.display_name, i.name, SUM(e.value), COUNT(DISTINCT d.id)
            FROM {pmc_ev} e
            JOIN {pmc_info} i ON e.pmc_id = i.id
            JOIN {disp} d ON e.event_id = d.event_id
            JOIN {sym} s ON d.kernel_id = s.id
            GROUP BY s.display_name, i.name
        """).fetchall()
    except Exception as ex:
        print("join failed:", ex)
        for t in (pmc_ev, pmc_info):
            print(t, cur.execute(f"SELECT * FROM {t} LIMIT 3").fetchall())
        return
    agg = {}
    for name, counter, val, n in rows:
        short = re.sub(r"<[^>]*>", "<>", name)
        short = short.split("(")[0][:70]
        agg.setdefault(short, {})[counter.upper()] = (val, n)
    print(f"{'kernel':70} {'mfma_busy':>12} {'wave_cyc':>12} {'mfma%':>6} "
          f"{'bank_conf':>10}")
    items = sorted(agg.items(),
                   key=lambda kv: -kv[1].get("SQ_WAVE_CYCLES", (0, 0))[0])
    for name, c in items[:top]:
        mb = c.get("SQ_VALU_MFMA_BUSY_CYCLES", (0, 0))[0]
        wc = c.get("SQ_WAVE_CYCLES", (0, 0))[0]
        bc = c.get("SQ_LDS_BANK_CONFLICT", (0, 0))[0]
        # WAVE_CYCLES counts quad-cycles; MFMA_BUSY counts cycles
        pct = 100.0 * mb / (wc * 4) if wc else 0.0
        print(f"{name:70} {mb:12.3e} {wc:12.3e} {pct:6.1f} {bc:10.3e}")


if __name__ == "__main__":
    paths = sys.argv[1:] or glob.glob("gpurun_out/prof_pmc/**/*_results.db",
                                      recursive=True)
    for p in paths:
        summarize(p)
