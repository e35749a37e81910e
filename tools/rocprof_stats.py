#!/usr/bin/env python3
"""Summarise a rocprofv3 rocpd sqlite database: per-kernel total/mean time.

Usage: python tools/rocprof_stats.py results.db [topk]
Works against the rocpd schema (rocpd_kernel_dispatch + kernel info via
rocpd_string); falls back to dumping the schema when tables move.
"""

from __future__ import annotations

import sqlite3
import sys


def main() -> int:
    path = sys.argv[1]
    topk = int(sys.argv[2]) if len(sys.argv) > 2 else 25
    db = sqlite3.connect(path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next((t for t in tables if "kernel_dispatch" in t), None)
    if kd is None:
        print("tables:", tables)
        return 1
    cols = [r[1] for r in cur.execute(f"PRAGMA table_info({kd})")]
    # locate the kernel-name join: rocpd keeps a kernel-info table with a
    # display-name string id
    ki = next((t for t in tables if "kernel" in t and "info" in t), None)
    if ki is None:
        ki = next((t for t in tables if t.endswith("kernel_symbol")), None)
    kic = [r[1] for r in cur.execute(f"PRAGMA table_info({ki})")] if ki else []
    strt = next((t for t in tables if t.endswith("_string")), None)
    name_col = next((c for c in ("display_name", "kernel_name", "name")
                     if c in kic), None)
    kid_col = next((c for c in ("kernel_id", "id") if c in kic), None)
    dur = ("(d.end - d.start)" if "end" in cols and "start" in cols
           else "d.duration")
    q = f"""
      SELECT s.string AS nm, COUNT(*) AS calls,
             SUM({dur}) AS tot, AVG({dur}) AS mean
      FROM {kd} d JOIN {ki} k ON d.kernel_id = k.{kid_col}
      JOIN {strt} s ON k.{name_col} = s.id
      GROUP BY nm ORDER BY tot DESC LIMIT {topk}
    """
    try:
        rows = list(cur.execute(q))
    except Exception as e:
        print("query failed:", e)
        print("kd cols:", cols)
        print("ki:", ki, kic)
        # maybe name_col is already a string column, not a string id
        q2 = f"""
          SELECT k.{name_col} AS nm, COUNT(*) AS calls,
                 SUM({dur}) AS tot, AVG({dur}) AS mean
          FROM {kd} d JOIN {ki} k ON d.kernel_id = k.{kid_col}
          GROUP BY nm ORDER BY tot DESC LIMIT {topk}
        """
        rows = list(cur.execute(q2))
    total = sum(r[2] for r in rows)
    print(f"{'total ms':>10} {'calls':>7} {'us/call':>9}  kernel")
    for nm, calls, tot, mean in rows:
        print(f"{tot / 1e6:10.2f} {calls:7d} {mean / 1e3:9.1f}  {str(nm)[:100]}")
    print(f"# sum of listed: {total / 1e6:.2f} ms")
    return 0


if __name__ == "__main__":
    sys.exit(main())
