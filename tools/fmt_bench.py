#!/usr/bin/env python3
"""Read a bench.py JSON line from stdin, print one compact row."""
import json
import sys

d = json.load(sys.stdin)
print(f'B={d["config"]["global_batch"]:7d}: {d["value"]/1e9:6.2f} G samples/s  '
      f'{d["ms_per_step"]*1e3:7.1f} us/step')
