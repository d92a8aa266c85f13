#!/usr/bin/env python3
"""Kernel stats / PMC extraction from a rocprofv3 rocpd sqlite DB.

rocprofv3 in this image writes `<out>/<run>/NNN_results.db` (rocpd schema)
instead of the older CSV files. This tool reproduces the
`--stats`-style kernel table (and per-kernel PMC sums when counters were
collected) so summaries can be committed under profiles/.

  python scripts/rocpd_stats.py gpurun_out/prof/runc/123_results.db \
      [--csv out.csv] [--like attn] [--pmc]
"""
from __future__ import annotations

import argparse
import csv
import sqlite3
import sys


def table_suffix(db, base: str) -> str:
    rows = db.execute(
        "SELECT name FROM sqlite_master WHERE type='table' AND name LIKE ?",
        (base + "%",),
    ).fetchall()
    if not rows:
        raise SystemExit(f"no {base}* table in this DB")
    return rows[0][0][len(base):]


def kernel_stats(db, like: str | None):
    sfx = table_suffix(db, "rocpd_kernel_dispatch_")
    where = "WHERE ks.display_name LIKE ?" if like else ""
    q = f"""
    SELECT ks.display_name, COUNT(*), SUM(kd.end - kd.start),
           AVG(kd.end - kd.start), MIN(kd.end - kd.start),
           MAX(kd.end - kd.start)
    FROM rocpd_kernel_dispatch_{sfx} kd
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON ks.id = kd.kernel_id
    {where}
    GROUP BY ks.display_name ORDER BY SUM(kd.end - kd.start) DESC
    """
    args = (f"%{like}%",) if like else ()
    rows = db.execute(q, args).fetchall()
    total = sum(r[2] for r in rows) or 1
    return [
        (name, calls, tot, 100.0 * tot / total, avg, mn, mx)
        for name, calls, tot, avg, mn, mx in rows
    ]


def pmc_sums(db, like: str | None):
    sfx = table_suffix(db, "rocpd_pmc_event_")
    where = "WHERE ks.display_name LIKE ?" if like else ""
    q = f"""
    SELECT ks.display_name, pi.name, SUM(pe.value)
    FROM rocpd_pmc_event_{sfx} pe
    JOIN rocpd_info_pmc_{sfx} pi ON pi.id = pe.pmc_id
    JOIN rocpd_kernel_dispatch_{sfx} kd ON kd.event_id = pe.event_id
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON ks.id = kd.kernel_id
    {where}
    GROUP BY ks.display_name, pi.name
    """
    args = (f"%{like}%",) if like else ()
    return db.execute(q, args).fetchall()


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("db")
    ap.add_argument("--csv", default=None, help="also write a CSV")
    ap.add_argument("--like", default=None, help="kernel-name substring")
    ap.add_argument("--pmc", action="store_true",
                    help="dump per-kernel PMC sums instead of timings")
    ap.add_argument("--top", type=int, default=20)
    args = ap.parse_args(argv)
    db = sqlite3.connect(args.db)
    if args.pmc:
        for name, ctr, val in pmc_sums(db, args.like):
            print(f"{val:16.4e}  {ctr:28s}  {name[:70]}")
        return
    rows = kernel_stats(db, args.like)
    for name, calls, tot, pct, avg, mn, mx in rows[: args.top]:
        print(f"{pct:6.2f}%  {calls:6d}  avg {avg/1e3:9.1f} us  {name[:70]}")
    if args.csv:
        with open(args.csv, "w", newline="") as f:
            w = csv.writer(f)
            w.writerow(["Name", "Calls", "TotalDurationNs", "Pct",
                        "AvgNs", "MinNs", "MaxNs"])
            for name, calls, tot, pct, avg, mn, mx in rows:
                w.writerow([name[:100], calls, tot, round(pct, 2),
                            round(avg, 1), mn, mx])
        print(f"wrote {args.csv}", file=sys.stderr)


if __name__ == "__main__":
    main()
