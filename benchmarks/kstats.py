"""Print top kernels from rocprofv3 kernel_stats csv dirs: kstats.py <dir> [n]"""

import csv, glob, sys

d = sys.argv[1]
n = int(sys.argv[2]) if len(sys.argv) > 2 else 18
rows = []
for f in glob.glob(f"{d}/**/*kernel_stats.csv", recursive=True):
    with open(f) as fh:
        for r in csv.DictReader(fh):
            rows.append(r)


def dur(r):
    for k in ("TotalDurationNs", "DurationNs", "TOTAL_DURATION_Ns"):
        if k in r:
            return float(r[k])
    return 0.0


rows.sort(key=lambda r: -dur(r))
tot = sum(dur(r) for r in rows)
print(f"total kernel ns: {tot / 1e6:.1f} ms")
for r in rows[:n]:
    name = r.get("Name") or r.get("Kernel_Name", "?")
    calls = r.get("Calls") or r.get("TotalCalls", "?")
    print(f"{dur(r) / 1e6:9.2f} ms {calls:>6}x  {name[:84]}")
