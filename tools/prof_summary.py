#!/usr/bin/env python3
"""Summarize a rocprofv3 results .db: per-kernel total/avg time, sorted.

Usage: python tools/prof_summary.py gpurun_out/prof/bench1_results.db [out.md]
"""

from __future__ import annotations

import sqlite3
import sys


def summarize(db_path: str):
    if not db_path.endswith(".db"):
        print(__doc__.strip())
        raise SystemExit(0)
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sfx = disp[len("rocpd_kernel_dispatch_"):]
    rows = cur.execute(f"""
        SELECT ks.display_name, COUNT(*), SUM(k.end-k.start),
               AVG(k.end-k.start),
               AVG(k.grid_size_x*k.grid_size_y*k.grid_size_z),
               AVG(ks.group_segment_size)
        FROM rocpd_kernel_dispatch_{sfx} k
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON ks.id = k.kernel_id
        GROUP BY ks.display_name ORDER BY SUM(k.end-k.start) DESC
    """).fetchall()
    total = sum(r[2] for r in rows)
    out = []
    out.append(f"| kernel | calls | total ms | avg us | % | grid | lds B |")
    out.append("|---|---|---|---|---|---|---|")
    for name, calls, tot, avg, grid, lds in rows:
        short = name.split("(")[0]
        if len(short) > 70:
            short = short[:67] + "..."
        out.append(
            f"| {short} | {calls} | {tot/1e6:.3f} | {avg/1e3:.1f} | "
            f"{100*tot/total:.1f}% | {grid:.0f} | {lds:.0f} |"
        )
    out.append(f"\nTotal kernel time: {total/1e6:.3f} ms")
    # memory copies
    try:
        mc = cur.execute(f"""
            SELECT COUNT(*), SUM(end-start) FROM rocpd_memory_copy_{sfx}
        """).fetchone()
        if mc and mc[0]:
            out.append(f"Memory copies: {mc[0]} totaling {mc[1]/1e6:.3f} ms")
    except Exception:
        pass
    return "\n".join(out)


if __name__ == "__main__":
    text = summarize(sys.argv[1])
    print(text)
    if len(sys.argv) > 2:
        with open(sys.argv[2], "w") as f:
            f.write(text + "\n")
