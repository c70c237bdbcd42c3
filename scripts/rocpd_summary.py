#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd SQLite database (ROCm 7.2 default output)
into a per-kernel table: calls, total/avg µs, % of GPU time, VGPR/SGPR,
LDS bytes, workgroup and grid sizes, and a derived occupancy bound
(waves/SIMD limited by VGPR allocation granule and LDS per workgroup,
MI355X: 512 VGPRs/SIMD in granules of 8, 160 KiB LDS/CU, 32 waves/CU).

Usage: python scripts/rocpd_summary.py <results.db> [--top 30] [--md]
"""
from __future__ import annotations

import argparse
import sqlite3


def waves_per_simd_vgpr(vgpr: int) -> int:
    alloc = max(8, (vgpr + 7) // 8 * 8)
    return max(1, min(8, 512 // alloc))


def wg_per_cu_lds(lds_bytes: int) -> int:
    if lds_bytes <= 0:
        return 32
    return max(1, (160 * 1024) // lds_bytes)


def summarize(path: str, top: int, md: bool, args=None):
    db = sqlite3.connect(path)
    tabs = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    sym = next(t for t in tabs if t.startswith("rocpd_info_kernel_symbol"))
    dis = next(t for t in tabs if t.startswith("rocpd_kernel_dispatch"))
    rows = db.execute(f"""
        SELECT s.display_name, COUNT(*), SUM(d.end - d.start),
               AVG(d.end - d.start), MAX(s.arch_vgpr_count),
               MAX(s.accum_vgpr_count), MAX(s.sgpr_count),
               MAX(d.group_segment_size),
               MAX(d.workgroup_size_x * d.workgroup_size_y * d.workgroup_size_z),
               MAX(d.grid_size_x * d.grid_size_y * d.grid_size_z)
        FROM {dis} d JOIN {sym} s ON d.kernel_id = s.id
        GROUP BY s.display_name ORDER BY SUM(d.end - d.start) DESC
    """).fetchall()
    total = sum(r[2] for r in rows) or 1  # pct is of ALL kernels
    flt = getattr(args, "like", "")
    if flt:  # substring filter in python (SQL LIKE treats _ as wildcard)
        rows = [r for r in rows if flt in r[0]]
    hdr = ("kernel", "calls", "total_us", "avg_us", "pct", "vgpr", "agpr",
           "sgpr", "lds_B", "wg", "grid_wgs", "waves/simd", "wg/cu_lds")
    out = []
    for r in rows[:top]:
        name, calls, tot, avg, vgpr, agpr, sgpr, lds, wg, grid = r
        waves = waves_per_simd_vgpr((vgpr or 0) + (agpr or 0))
        wgcu = wg_per_cu_lds(lds or 0)
        nm = name.split("(")[0][:52]
        out.append((nm, calls, round(tot / 1e3, 1), round(avg / 1e3, 2),
                    f"{100*tot/total:.1f}%", vgpr, agpr, sgpr, lds, wg,
                    grid // max(wg, 1), waves, wgcu))
    if md:
        print("| " + " | ".join(hdr) + " |")
        print("|" + "---|" * len(hdr))
        for row in out:
            print("| " + " | ".join(str(x) for x in row) + " |")
    else:
        w = [max(len(str(row[i])) for row in out + [hdr])
             for i in range(len(hdr))]
        print("  ".join(h.ljust(w[i]) for i, h in enumerate(hdr)))
        for row in out:
            print("  ".join(str(x).ljust(w[i]) for i, x in enumerate(row)))
    print(f"\ntotal GPU kernel time: {total/1e6:.2f} ms over "
          f"{sum(r[1] for r in rows)} dispatches, {len(rows)} distinct kernels")


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("db")
    ap.add_argument("--top", type=int, default=30)
    ap.add_argument("--like", default="")
    ap.add_argument("--md", action="store_true")
    args = ap.parse_args()
    summarize(args.db, args.top, args.md, args)
