"""Aggregate a rocprofv3 --pmc CSV by kernel: sums per counter."""
import sys, csv, glob
from collections import defaultdict
f = glob.glob(sys.argv[1] + '/**/*counter_collection.csv', recursive=True)
if not f:
    print("no counter csv under", sys.argv[1]); sys.exit(1)
agg = defaultdict(lambda: defaultdict(float))
calls = defaultdict(set)
with open(f[0]) as fh:
    for row in csv.DictReader(fh):
        k = row.get("Kernel_Name", row.get("Kernel-Name", ""))[:60]
        c = row.get("Counter_Name", row.get("Counter-Name", ""))
        v = float(row.get("Counter_Value", row.get("Counter-Value", 0)))
        agg[k][c] += v
        calls[k].add(row.get("Dispatch_Id", row.get("Correlation_Id", "")))
for k, cs in sorted(agg.items(), key=lambda kv: -max(kv[1].values())):
    if "memset" in k or not k.strip():
        continue
    print(f"== {k}  (n={len(calls[k])})")
    for c, v in sorted(cs.items()):
        print(f"   {c:24s} {v:,.0f}")
