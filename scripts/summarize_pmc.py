#!/usr/bin/env python3
"""Aggregate rocprofv3 --pmc counter values per kernel from the results DB."""
import glob
import re
import sqlite3
import sys


def summarize(db_path):
    con = sqlite3.connect(db_path)
    cur = con.cursor()
    tabs = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]

    def first(prefix):
        return next((t for t in tabs if t.startswith(prefix)), None)

    info = first("rocpd_info_pmc")
    ev = first("rocpd_pmc_event")
    disp = first("rocpd_kernel_dispatch")
    sym = first("rocpd_info_kernel_symbol")
    for t in (info, ev, disp, sym):
        if t is None:
            print("missing table; have:", tabs)
            return
    icols = [r[1] for r in cur.execute(f"PRAGMA table_info({info})")]
    ecols = [r[1] for r in cur.execute(f"PRAGMA table_info({ev})")]
    print("# info cols:", icols)
    print("# event cols:", ecols)

    # common layout: info(id, name, ...); event(pmc_id->info.id, event_id ->
    # dispatch correlation, value)
    name_col = "name" if "name" in icols else icols[1]
    pmc_ref = next((c for c in ecols if "pmc" in c), None)
    val_col = next((c for c in ecols if "value" in c), None)
    link_col = next((c for c in ecols if c not in (pmc_ref, val_col)), None)
    dcols = [r[1] for r in cur.execute(f"PRAGMA table_info({disp})")]
    print("# dispatch cols:", dcols)
    dlink = next((c for c in dcols if c == link_col or c.rstrip("_id") in link_col), None)
    if dlink is None:
        dlink = "id" if "id" in dcols else dcols[0]
    q = f"""SELECT s.display_name, i.{name_col}, SUM(e.{val_col}), COUNT(*)
            FROM {ev} e
            JOIN {info} i ON e.{pmc_ref} = i.id
            JOIN {disp} k ON e.{link_col} = k.{dlink}
            JOIN {sym} s ON k.kernel_id = s.id
            GROUP BY s.display_name, i.{name_col}"""
    try:
        rows = cur.execute(q).fetchall()
    except Exception as exc:
        print("join failed:", exc)
        print("sample event rows:", cur.execute(f"SELECT * FROM {ev} LIMIT 3").fetchall())
        print("sample info rows:", cur.execute(f"SELECT * FROM {info} LIMIT 6").fetchall())
        return
    agg = {}
    for kname, cname, tot, cnt in rows:
        kname = re.sub(r"\(.*", "", kname)[:48]
        agg.setdefault(kname, {})[cname] = (tot, cnt)
    for kname in sorted(agg, key=lambda k: -max(v[0] for v in agg[k].values())):
        print(f"== {kname}")
        for cname, (tot, cnt) in sorted(agg[kname].items()):
            print(f"   {cname:24s} sum={tot:>18.0f}  n={cnt}")


if __name__ == "__main__":
    import os

    pattern = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/**/*.db"
    if os.path.isdir(pattern):
        pattern = os.path.join(pattern, "**", "*.db")
    for db in glob.glob(pattern, recursive=True):
        print(f"== {db} ==")
        summarize(db)
