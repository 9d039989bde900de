"""Aggregate a rocprofv3 --pmc counter_collection.csv by kernel name.

Usage: python tools/pmcagg.py <counter_collection.csv> [top_n]
Prints per-kernel counter sums (M units) for every counter column.
"""
import csv
import sys
from collections import defaultdict

path = sys.argv[1]
topn = int(sys.argv[2]) if len(sys.argv) > 2 else 14
agg = defaultdict(lambda: defaultdict(float))
calls = defaultdict(int)
with open(path) as f:
    for row in csv.DictReader(f):
        name = row.get("Kernel_Name", row.get("Kernel-Name", ""))[:60]
        ctr = row.get("Counter_Name", row.get("Counter-Name", ""))
        val = float(row.get("Counter_Value", row.get("Counter-Value", 0)))
        agg[name][ctr] += val
        calls[name] += 1
ctrs = sorted({c for v in agg.values() for c in v})
order = sorted(agg, key=lambda k: -sum(agg[k].values()))
print("%-60s %s" % ("kernel", " ".join("%14s" % c[:14] for c in ctrs)))
for k in order[:topn]:
    print("%-60s %s" % (k, " ".join("%14.1f" % (agg[k][c] / 1e6)
                                    for c in ctrs)))
