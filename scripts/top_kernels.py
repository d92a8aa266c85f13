#!/usr/bin/env python3
"""Summarize a rocprofv3 kernel_stats.csv (args: glob pattern).

NOTE: newer rocprofv3 builds in this image emit rocpd sqlite DBs instead
of CSVs — use scripts/rocpd_stats.py for those."""
import csv, glob, sys

pat = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/prof5/*kernel_stats.csv"
files = glob.glob(pat)
if not files:
    print("no kernel_stats csv at", pat)
    sys.exit(0)
rows = list(csv.DictReader(open(files[0])))
rows.sort(key=lambda r: -float(r["TotalDurationNs"]))
tot = sum(float(r["TotalDurationNs"]) for r in rows)
print("total kernel time: %.2f s" % (tot / 1e9))
for r in rows[:15]:
    print("%9.1f ms %5.2f%% %5s  %s" % (
        float(r["TotalDurationNs"]) / 1e6, float(r["Percentage"]),
        r["Calls"], r["Name"][:90]))
