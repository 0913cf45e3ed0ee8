#!/usr/bin/env python3
"""Summarize a rocprofv3 kernel_stats CSV: top kernels by total time."""
import csv
import sys

rows = list(csv.DictReader(open(sys.argv[1])))
rows.sort(key=lambda r: -float(r["TotalDurationNs"]))
tot = sum(float(r["TotalDurationNs"]) for r in rows)
print(f"total kernel time: {tot/1e6:.1f} ms")
for r in rows[: int(sys.argv[2]) if len(sys.argv) > 2 else 14]:
    print(
        f'{float(r["TotalDurationNs"])/1e6:8.2f} ms {int(r["Calls"]):5d}x '
        f'{float(r["AverageNs"])/1e3:8.1f} us  {r["Name"][:90]}'
    )
