"""Summarize a rocprofv3 kernel_stats.csv: top kernels by total time.

Usage: python tools/kstats.py <kernel_stats.csv> [steps]
"""
import csv
import sys

rows = list(csv.DictReader(open(sys.argv[1])))
steps = float(sys.argv[2]) if len(sys.argv) > 2 else 1.0
rows.sort(key=lambda r: -float(r["TotalDurationNs"]))
tot = sum(float(r["TotalDurationNs"]) for r in rows)
print("total %.1f ms over %g steps -> %.2f ms/step GPU busy"
      % (tot / 1e6, steps, tot / steps / 1e6))
for r in rows[:26]:
    print("%-78s %5d %8.2f %7.1f %5.1f%%"
          % (r["Name"][:78], int(r["Calls"]),
             float(r["TotalDurationNs"]) / 1e6,
             float(r["AverageNs"]) / 1e3,
             100 * float(r["TotalDurationNs"]) / tot))
