#!/usr/bin/env python3
"""Per-kernel stats / PMC extraction from rocprofv3 .db (rocpd) outputs.

rocprofv3 on this image writes SQLite databases; this tool reproduces the
`--stats` summary (name, calls, total/avg/min/max duration) and, for PMC
runs, per-kernel average counter values — the inputs profiles/ and
tools/make_pmc_manifest.py need.

Usage:
  python tools/rocpd_stats.py stats <db> [out.csv]
  python tools/rocpd_stats.py pmc   <db> [out.csv]
"""

import csv
import sqlite3
import sys
from collections import defaultdict


def _uuid(cur):
    row = cur.execute("SELECT name FROM sqlite_master WHERE type='table' "
                      "AND name LIKE 'rocpd_kernel_dispatch_%'").fetchone()
    assert row, "no kernel dispatch table"
    return row[0][len("rocpd_kernel_dispatch_"):]


def kernel_rows(db):
    con = sqlite3.connect(db)
    cur = con.cursor()
    u = _uuid(cur)
    q = f"""
      SELECT k.display_name, d.start, d.end, d.id
      FROM rocpd_kernel_dispatch_{u} d
      JOIN rocpd_info_kernel_symbol_{u} k ON d.kernel_id = k.id
    """
    return cur, u, list(cur.execute(q))


def stats(db, out=None):
    _, _, rows = kernel_rows(db)
    agg = defaultdict(lambda: [0, 0.0, float("inf"), 0.0])
    for (name, start, end, _id) in rows:
        dur = (end - start)
        a = agg[name]
        a[0] += 1
        a[1] += dur
        a[2] = min(a[2], dur)
        a[3] = max(a[3], dur)
    total = sum(a[1] for a in agg.values()) or 1
    lines = [("Name", "Calls", "TotalDurationNs", "AverageNs",
              "Percentage", "MinNs", "MaxNs")]
    for name, (n, tot, mn, mx) in sorted(agg.items(),
                                         key=lambda kv: -kv[1][1]):
        lines.append((name, n, int(tot), tot / n, 100.0 * tot / total,
                      int(mn), int(mx)))
    _emit(lines, out)


def pmc(db, out=None):
    con = sqlite3.connect(db)
    cur = con.cursor()
    u = _uuid(cur)
    q = f"""
      SELECT k.display_name, c.name, e.value
      FROM rocpd_pmc_event_{u} e
      JOIN rocpd_kernel_dispatch_{u} d ON e.event_id = d.event_id
      JOIN rocpd_info_kernel_symbol_{u} k ON d.kernel_id = k.id
      JOIN rocpd_info_pmc_{u} c ON e.pmc_id = c.id
    """
    rows = list(cur.execute(q))
    agg = defaultdict(lambda: [0, 0.0])
    for (kname, ctr, val) in rows:
        a = agg[(kname, ctr)]
        a[0] += 1
        a[1] += val
    lines = [("Kernel_Name", "Counter_Name", "Launches",
              "Avg_Counter_Value")]
    for (kname, ctr), (n, tot) in sorted(agg.items(),
                                         key=lambda kv: -kv[1][1]):
        lines.append((kname, ctr, n, tot / n))
    _emit(lines, out)


def _emit(lines, out):
    if out:
        with open(out, "w", newline="") as f:
            csv.writer(f).writerows(lines)
    for ln in lines[:40]:
        print(",".join(str(x) for x in ln))


if __name__ == "__main__":
    mode, db = sys.argv[1], sys.argv[2]
    out = sys.argv[3] if len(sys.argv) > 3 else None
    (stats if mode == "stats" else pmc)(db, out)
