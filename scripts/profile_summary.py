#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd SQLite database into a kernel-time table.

Usage: python scripts/profile_summary.py gpurun_out/prof/bench_results.db [N]

Prints total wall span, GPU-busy time, dispatch count, and the top-N
kernels by total time — the evidence tables committed under profiles/.
"""
from __future__ import annotations

import sqlite3
import sys


def summarize(db_path: str, top: int = 40) -> str:
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    wall, busy, n = cur.execute(
        f"SELECT (MAX(end)-MIN(start))/1e6, SUM(end-start)/1e6, COUNT(*) FROM {disp}"
    ).fetchone()
    out = [
        f"db: {db_path}",
        f"wall span: {wall:.1f} ms   gpu busy: {busy:.1f} ms "
        f"({100*busy/wall:.1f}%)   dispatches: {n}",
        "",
        f"{'total_ms':>10} {'calls':>7} {'avg_us':>9}  kernel",
        "-" * 100,
    ]
    rows = cur.execute(
        f"""SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6,
                   AVG(kd.end-kd.start)/1e3
            FROM {disp} kd JOIN {sym} ks ON kd.kernel_id = ks.id
            GROUP BY ks.display_name ORDER BY 3 DESC LIMIT {int(top)}"""
    ).fetchall()
    for name, calls, ms, avg in rows:
        out.append(f"{ms:10.2f} {calls:7d} {avg:9.1f}  {name[:90]}")
    return "\n".join(out)


if __name__ == "__main__":
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 40
    print(summarize(sys.argv[1], top))
