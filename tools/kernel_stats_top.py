#!/usr/bin/env python3
"""Print top-N kernels from a rocprofv3 kernel_stats.csv."""
import csv
import sys

rows = list(csv.DictReader(open(sys.argv[1])))
rows.sort(key=lambda r: -float(r["TotalDurationNs"]))
tot = sum(float(r["TotalDurationNs"]) for r in rows)
for r in rows[: int(sys.argv[2]) if len(sys.argv) > 2 else 15]:
  pct = float(r["TotalDurationNs"]) / tot * 100
  print(f'{pct:5.1f}%  {int(r["Calls"]):6d}  {float(r["AverageNs"])/1e3:9.1f}us  {r["Name"][:85]}')
