"""Summarize rocprofv3 counter_collection CSVs: per kernel-name counter sums.

usage: python pmc_summarize.py <dir> [name-filter]
"""

import csv, glob, sys
from collections import defaultdict

d = sys.argv[1]
filt = sys.argv[2] if len(sys.argv) > 2 else ""
agg = defaultdict(lambda: defaultdict(float))
ndisp = defaultdict(int)
for f in glob.glob(f"{d}/**/*counter_collection.csv", recursive=True):
    with open(f) as fh:
        for row in csv.DictReader(fh):
            kn = row.get("Kernel_Name", "")[:60]
            if filt and filt not in kn:
                continue
            agg[kn][row["Counter_Name"]] += float(row["Counter_Value"])
            ndisp[kn] += 1
for kn, cs in sorted(agg.items(), key=lambda kv: -max(kv[1].values())):
    print(kn)
    for c, v in sorted(cs.items()):
        print(f"    {c:32s} {v:.3e}")
