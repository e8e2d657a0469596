#!/usr/bin/env python3
"""Summarize a rocprofv3 kernel_stats CSV: top-N kernels by total time."""
import csv
import sys

path = sys.argv[1]
topn = int(sys.argv[2]) if len(sys.argv) > 2 else 20
rows = list(csv.DictReader(open(path)))
tot = sum(float(r["TotalDurationNs"]) for r in rows)
calls = sum(int(r["Calls"]) for r in rows)
print(f"total kernel time: {tot/1e6:.1f} ms over {calls} dispatches")
for r in sorted(rows, key=lambda r: -float(r["TotalDurationNs"]))[:topn]:
    short = r["Name"].split("(")[0]
    if short.startswith("void "):
        short = short[5:]
    print(f'{float(r["Percentage"]):5.2f}%  {int(r["Calls"]):6d}x  {float(r["AverageNs"])/1e3:8.1f}us  {short[:78]}')
