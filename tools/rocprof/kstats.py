"""Summarize a rocprofv3 *_kernel_stats.csv: top kernels by total time."""
import csv
import sys


def dur(r):
    for k in ("TotalDurationNs", "TOTAL_DURATION_NS", "DurationNs"):
        if k in r:
            return float(r[k])
    return 0.0


rows = list(csv.DictReader(open(sys.argv[1])))
rows.sort(key=lambda r: -dur(r))
tot = sum(dur(r) for r in rows)
print(f"total GPU ns: {tot:.3e}")
for r in rows[: int(sys.argv[2]) if len(sys.argv) > 2 else 15]:
    name = (r.get("Name") or r.get("NAME", "?"))[:72]
    calls = r.get("Calls", r.get("CALLS", "?"))
    print(f"{dur(r)/tot*100:5.1f}%  {dur(r)/1e6:8.2f} ms  n={calls:>5}  {name}")
