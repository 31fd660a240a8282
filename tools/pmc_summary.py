#!/usr/bin/env python3
"""Summarize rocprofv3 --pmc results: per-kernel counter totals + duration.

Usage: python tools/pmc_summary.py results.db [...]"""
import sqlite3
import sys
from collections import defaultdict


def summarize(path):
    db = sqlite3.connect(path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sfx = disp[len("rocpd_kernel_dispatch_"):]
    rows = cur.execute(f"""
        SELECT ks.display_name, pd.name, SUM(pe.value), COUNT(*),
               SUM(k.end - k.start)
        FROM rocpd_pmc_event_{sfx} pe
        JOIN rocpd_kernel_dispatch_{sfx} k ON k.id = pe.event_id
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON ks.id = k.kernel_id
        JOIN rocpd_info_pmc_{sfx} pd ON pd.id = pe.pmc_id
        GROUP BY ks.display_name, pd.name ORDER BY 5 DESC
    """).fetchall()
    print(f"## {path}")
    print("| kernel | counter | total | calls | total_ns |")
    print("|---|---|---|---|---|")
    for name, cname, val, n, ns in rows:
        short = name.split("(")[0][:60]
        print(f"| {short} | {cname} | {val:.4g} | {n} | {ns} |")


for p in sys.argv[1:]:
    try:
        summarize(p)
    except Exception as e:  # noqa: BLE001
        print(f"{p}: FAILED {e}")
