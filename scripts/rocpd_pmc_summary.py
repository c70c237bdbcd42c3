#!/usr/bin/env python3
"""Summarize PMC counters from a rocprofv3 rocpd database into a small
per-kernel text table (meant to run ON the GPU box so only text comes
back, not the multi-MB db).

Handles the ROCm 7.x rocpd schema defensively: finds the counter tables
by name pattern, prints the schema if the expected shape is missing.

Usage: python scripts/rocpd_pmc_summary.py <db> [--like bt_] [--out f.txt]
"""
from __future__ import annotations

import argparse
import sqlite3
import sys


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("db")
    ap.add_argument("--like", default="")
    ap.add_argument("--out", default=None)
    args = ap.parse_args()
    out = open(args.out, "w") if args.out else sys.stdout
    db = sqlite3.connect(args.db)
    tabs = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]

    def find(prefix):
        m = [t for t in tabs if t.startswith(prefix)]
        return m[0] if m else None

    sym = find("rocpd_info_kernel_symbol")
    dis = find("rocpd_kernel_dispatch")
    cnt = find("rocpd_counter") or find("rocpd_pmc")
    if cnt is None:
        cand = [t for t in tabs if "counter" in t.lower() or "pmc" in t.lower()]
        cnt = cand[0] if cand else None
    if cnt is None:
        print("no counter table; tables are:", file=out)
        for t in tabs:
            print(" ", t, file=out)
        return
    cols = [r[1] for r in db.execute(f"PRAGMA table_info({cnt})")]
    print(f"counter table: {cnt}; cols: {cols}", file=out)
    # counter-name lookup: any pmc/counter info table with (id, name)-ish
    names = {}
    for t in tabs:
        tl = t.lower()
        if ("pmc" in tl or "counter" in tl) and t != cnt:
            tcols = [r[1] for r in db.execute(f"PRAGMA table_info({t})")]
            idc = next((c for c in ("id", "pmc_id") if c in tcols), None)
            nmc = next((c for c in ("name", "symbol", "counter_name")
                        if c in tcols), None)
            if idc and nmc:
                names.update(db.execute(
                    f"SELECT {idc}, {nmc} FROM {t}").fetchall())
                print(f"names from {t} ({len(names)})", file=out)
    # ROCm 7.2 rocpd shape: rocpd_pmc_event(event_id, pmc_id, value)
    # joined to the dispatch via its event_id
    key = next((c for c in ("pmc_id", "counter_id") if c in cols), None)
    val = "value" if "value" in cols else None
    if "event_id" in cols:
        join = f"JOIN {dis} d ON c.event_id = d.event_id"
    elif "dispatch_id" in cols:
        join = f"JOIN {dis} d ON c.dispatch_id = d.dispatch_id"
    else:
        join = None
    if not (key and val and join and sym and dis):
        print("unexpected schema; dumping 5 sample rows:", file=out)
        for r in db.execute(f"SELECT * FROM {cnt} LIMIT 5"):
            print(" ", r, file=out)
        return
    like = f"AND s.display_name LIKE '%{args.like}%'" if args.like else ""
    q = f"""
        SELECT s.display_name, c.{key}, SUM(c.{val}), COUNT(*)
        FROM {cnt} c
        {join}
        JOIN {sym} s ON d.kernel_id = s.id
        WHERE 1=1 {like}
        GROUP BY s.display_name, c.{key}
        ORDER BY s.display_name
    """
    try:
        rows = db.execute(q).fetchall()
    except Exception as e:  # noqa: BLE001
        print("join failed:", e, file=out)
        for r in db.execute(f"SELECT * FROM {cnt} LIMIT 5"):
            print(" ", r, file=out)
        return
    cur = None
    for name, cid, total, n in rows:
        nm = name.split("(")[0][:60]
        if nm != cur:
            print(f"\n{nm}  (dispatch-rows {n})", file=out)
            cur = nm
        print(f"  {names.get(cid, cid):>24}: {int(total):,}", file=out)


if __name__ == "__main__":
    main()
