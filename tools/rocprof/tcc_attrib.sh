#!/bin/bash
# L2 (TCC) hit/miss + fetch attribution for the NT GEMM variants.
# Usage: bash tools/rocprof/tcc_attrib.sh "<env pairs>" M N K
set -e
cd /tmp && export TMPDIR=/tmp
REPO=${REPO:-/root/repo}
OUT=${OUT:-$REPO/gpurun_out/rocprof_tcc}
M=${1:-4096} N=${2:-4096} K=${3:-4096}
mkdir -p "$OUT"
for CFG in "2 0" "2 8" "6 8"; do
  set -- $CFG; V=$1; G=$2
  TAG="v${V}g${G}"
  NEZHA_GEMM_VARIANT=$V NEZHA_GEMM_GROUP_M=$G rocprofv3 \
    --pmc TCC_HIT_sum TCC_MISS_sum \
    --output-format csv -d "$OUT/$TAG" -o "$TAG" \
    -- "$REPO/build/nezha-bench" --op=shape --m=$M --n=$N --k=$K --tb=1 \
       --iters=4 > "$OUT/$TAG.bench.log" 2>&1
  echo "=== $TAG (M=$M N=$N K=$K) ==="
  tail -1 "$OUT/$TAG.bench.log"
  CSV=$(ls "$OUT/$TAG"/*counter_collection.csv | head -1)
  python3 - "$CSV" <<'EOF'
import sys, csv, collections
agg = collections.defaultdict(float)
with open(sys.argv[1]) as f:
    for row in csv.DictReader(f):
        if "8phase" not in row.get("Kernel_Name", ""):
            continue
        agg[row["Counter_Name"]] += float(row["Counter_Value"])
h, m = agg.get("TCC_HIT_sum", 0), agg.get("TCC_MISS_sum", 0)
print(f"  TCC_HIT={h:.3e} TCC_MISS={m:.3e} hit_rate={h/(h+m+1e-9):.3f} "
      f"miss_bytes~={m*128/6:.3e}/launch")
EOF
done
