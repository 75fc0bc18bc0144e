#!/usr/bin/env python3
"""Print the top-N kernels of a rocprofv3 kernel_stats.csv."""
import csv
import sys

rows = list(csv.DictReader(open(sys.argv[1])))
n = int(sys.argv[2]) if len(sys.argv) > 2 else 20
rows.sort(key=lambda r: -float(r["TotalDurationNs"]))
total = sum(float(r["TotalDurationNs"]) for r in rows)
print(f"# total kernel time: {total/1e9:.3f} s")
for r in rows[:n]:
    calls = int(r["Calls"])
    avg_us = float(r["AverageNs"]) / 1e3
    pct = 100 * float(r["TotalDurationNs"]) / total
    name = r["Name"][:100]
    print(f"{pct:7.3f}% {calls:6d}x {avg_us:10.2f}us  {name}")
