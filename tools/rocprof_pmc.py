#!/usr/bin/env python3
"""Aggregate rocprofv3 --pmc counters per kernel from a rocpd sqlite db.

Usage: python tools/rocprof_pmc.py results.db
"""

from __future__ import annotations

import sqlite3
import sys
from collections import defaultdict


def main() -> int:
    db = sqlite3.connect(sys.argv[1])
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]

    def tab(sub):
        return next((t for t in tables if sub in t), None)

    pmc_ev = tab("pmc_event")
    kd = tab("kernel_dispatch")
    ki = tab("info_kernel_symbol")
    strt = tab("_string")
    pmc_info = tab("info_pmc")
    # pmc_id -> counter name
    pmc_names = {}
    if pmc_info:
        cols = [r[1] for r in cur.execute(f"PRAGMA table_info({pmc_info})")]
        namecol = next((c for c in ("name", "symbol") if c in cols), cols[1])
        idcol = "id" if "id" in cols else cols[0]
        for pid, nm in cur.execute(f"SELECT {idcol}, {namecol} FROM {pmc_info}"):
            pmc_names[pid] = str(nm)
    # event_id -> kernel name (display_name holds the string directly in
    # pmc-mode DBs; string-table indirection only exists in trace DBs)
    try:
        q = f"""SELECT d.event_id, s.string FROM {kd} d
                JOIN {ki} k ON d.kernel_id = k.id
                JOIN {strt} s ON k.display_name = s.id"""
        ev2k = dict(cur.execute(q))
        if not ev2k or all(isinstance(v, int) for v in ev2k.values()):
            raise ValueError
    except Exception:
        q = f"""SELECT d.event_id, k.display_name FROM {kd} d
                JOIN {ki} k ON d.kernel_id = k.id"""
        ev2k = dict(cur.execute(q))
    agg = defaultdict(float)
    cnt = defaultdict(int)
    for eid, pid, val in cur.execute(
            f"SELECT event_id, pmc_id, value FROM {pmc_ev}"):
        knm = str(ev2k.get(eid, "?"))[:70]
        pnm = pmc_names.get(pid, str(pid))
        agg[(knm, pnm)] += float(val)
        cnt[(knm, pnm)] += 1
    rows = sorted(agg.items(), key=lambda kv: -kv[1])
    byk = defaultdict(dict)
    for (knm, pnm), v in rows:
        byk[knm][pnm] = v
    for knm, d in sorted(byk.items(),
                         key=lambda kv: -max(kv[1].values())):
        parts = "  ".join(f"{p}={v:.3e}" for p, v in sorted(d.items()))
        ncalls = max(cnt[(knm, p)] for p in d)
        print(f"{knm}\n    calls={ncalls}  {parts}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
