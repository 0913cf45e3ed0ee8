#!/usr/bin/env python3
"""Inter-kernel gap analysis for the fetch benchmark from a rocprofv3
kernel-trace CSV: are back-to-back gather launches GPU-bound or host-bound?"""
import csv
import sys

rows = [r for r in csv.DictReader(open(sys.argv[1])) if "ddstore" in r["Kernel_Name"]]
rows.sort(key=lambda r: int(r["Start_Timestamp"]))
spans = [(int(r["Start_Timestamp"]), int(r["End_Timestamp"])) for r in rows]
durs = [e - s for s, e in spans]
gaps = [spans[i + 1][0] - spans[i][1] for i in range(len(spans) - 1)]
tail = gaps[len(gaps) // 4 :]  # steady state
print(f"{len(spans)} gather dispatches")
print(f"kernel duration: mean {sum(durs)/len(durs)/1e3:.1f} us")
print(f"inter-kernel gap (steady state): mean {sum(tail)/len(tail)/1e3:.2f} us, "
      f"max {max(tail)/1e3:.1f} us, min {min(tail)/1e3:.2f} us")
