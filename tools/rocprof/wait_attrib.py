"""Summarize a rocprofv3 counter_collection.csv for the GEMM kernels:
wave-cycle attribution (WAIT_ANY / WAIT_INST_ANY / ACTIVE_INST_ANY) and
MFMA duty. SQ_* counters are quad-cycles; SQ_VALU_MFMA_BUSY_CYCLES is
cycles (MI355X_MICROARCH.md rocprof notes)."""
import collections
import csv
import sys

agg = collections.defaultdict(float)
cnt = collections.defaultdict(int)
kname = None
with open(sys.argv[1]) as f:
    for row in csv.DictReader(f):
        name = row.get("Kernel_Name", "")
        if "gemm" not in name:
            continue
        if ("8phase" in name or "nt_k" in name or "pers" in name
                or "pingpong" in name):
            kname = name.split("(")[0]
            c = row["Counter_Name"]
            v = float(row["Counter_Value"])
            agg[c] += v
            cnt[c] += 1
print("kernel:", kname)
for k in sorted(agg):
    print(f"  {k:32s} {agg[k]:.4e}  (n={cnt[k]})")
w = agg.get("SQ_WAVE_CYCLES", 0) or 1
for k in ["SQ_WAIT_ANY", "SQ_WAIT_INST_ANY", "SQ_ACTIVE_INST_ANY"]:
    print(f"  {k}/WAVE_CYCLES = {agg.get(k, 0) / w:.3f}")
print(f"  MFMA_BUSY/(4*WAVE_CYCLES) = "
      f"{agg.get('SQ_VALU_MFMA_BUSY_CYCLES', 0) / 4 / w:.3f}")
