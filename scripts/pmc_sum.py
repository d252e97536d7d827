"""Sum rocprofv3 PMC csv per kernel (counter_collection csv)."""
import csv, re, sys
from collections import defaultdict
agg = defaultdict(lambda: defaultdict(float))
with open(sys.argv[1]) as f:
    for row in csv.DictReader(f):
        nm = re.sub(r"[<(].*", "", row["Kernel_Name"]).strip()
        agg[nm][row["Counter_Name"]] += float(row["Counter_Value"])
for nm, cs in agg.items():
    wc = cs.get("SQ_WAVE_CYCLES", 0) or 1
    print(f"{nm[:58]:<58}")
    for c, v in sorted(cs.items()):
        extra = f"  ({100*v/wc:5.1f}% of wave cycles)" if c.startswith("SQ_WAIT") or c == "SQ_ACTIVE_INST_ANY" else ""
        print(f"    {c:<24} {v:16.0f}{extra}")
