#!/usr/bin/env python3
"""Summarize a rocprofv3 kernel-trace CSV (or rocpd db -> csv) into
per-kernel totals. Usage: kernel_stats.py <csv> [topN] [t0_ns]"""
import csv as csvmod, re, sys

path, top = sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 25
t0 = int(sys.argv[3]) if len(sys.argv) > 3 else 0
agg = {}
with open(path) as f:
    for row in csvmod.DictReader(f):
        st, en = int(row["Start_Timestamp"]), int(row["End_Timestamp"])
        if st < t0:
            continue
        name = row["Kernel_Name"]
        nm = re.sub(r"[<(].*", "", name).strip()
        a = agg.setdefault(nm, [0, 0])
        a[0] += 1
        a[1] += en - st
rows = sorted(agg.items(), key=lambda kv: -kv[1][1])
total = sum(v[1] for _, v in rows)
print(f"{'kernel':<66} {'calls':>7} {'total_ms':>10} {'avg_us':>9} {'%':>6}")
for nm, (calls, tot) in rows[:top]:
    print(f"{nm[:66]:<66} {calls:>7} {tot/1e6:>10.2f} {tot/1e3/calls:>9.1f} {100*tot/total:>5.1f}")
print(f"{'TOTAL':<66} {sum(v[0] for _,v in rows):>7} {total/1e6:>10.2f}")
