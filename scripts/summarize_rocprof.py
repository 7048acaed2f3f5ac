#!/usr/bin/env python3
"""Summarize a rocprofv3 SQLite results DB into a per-kernel time table."""
import glob
import re
import sqlite3
import sys


def summarize(db_path):
    con = sqlite3.connect(db_path)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    rows = cur.execute(
        f"""SELECT s.display_name, COUNT(*), SUM(k.end-k.start)/1e6, AVG(k.end-k.start)/1e6
            FROM {disp} k JOIN {sym} s ON k.kernel_id = s.id
            GROUP BY s.display_name ORDER BY SUM(k.end-k.start) DESC"""
    ).fetchall()
    total = sum(r[2] for r in rows)
    out = [f"{'kernel':64s} {'count':>7s} {'total_ms':>10s} {'avg_ms':>9s} {'%':>6s}"]
    for name, cnt, tot, avg in rows:
        name = re.sub(r"\(.*", "", name)[:64]
        out.append(f"{name:64s} {cnt:7d} {tot:10.1f} {avg:9.4f} {100*tot/total:6.2f}")
    out.append(f"TOTAL kernel ms: {total:.1f}")
    return "\n".join(out)


if __name__ == "__main__":
    import os

    pattern = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/**/*.db"
    if os.path.isdir(pattern):
        pattern = os.path.join(pattern, "**", "*.db")
    for db in glob.glob(pattern, recursive=True):
        print(f"== {db} ==")
        print(summarize(db))
