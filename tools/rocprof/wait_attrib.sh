#!/bin/bash
# Wait-attribution PMC pass for the NT GEMM variants (VERDICT r01 #3 loop):
# SQ_WAIT_ANY (parked at barrier/waitcnt), SQ_WAIT_INST_ANY (issue-stall),
# SQ_ACTIVE_INST_ANY, MFMA busy, + GRBM for effective clock.
# Usage (on the GPU box): bash tools/rocprof/wait_attrib.sh "2 6" 4096 4096 4096
set -e
cd /tmp && export TMPDIR=/tmp
REPO=${REPO:-/root/repo}
OUT=${OUT:-$REPO/gpurun_out/rocprof_r2}
VARIANTS=${1:-"2 6"}
M=${2:-4096} N=${3:-4096} K=${4:-4096}
mkdir -p "$OUT"
for V in $VARIANTS; do
  NEZHA_GEMM_VARIANT=$V rocprofv3 \
    --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY \
          SQ_VALU_MFMA_BUSY_CYCLES GRBM_GUI_ACTIVE \
    --output-format csv -d "$OUT/v$V" -o "v$V" \
    -- "$REPO/build/nezha-bench" --op=shape --m=$M --n=$N --k=$K --tb=1 \
       --iters=4 > "$OUT/v$V.bench.log" 2>&1
  echo "=== variant $V (M=$M N=$N K=$K) ==="
  tail -1 "$OUT/v$V.bench.log"
  CSV=$(ls "$OUT/v$V"/*counter_collection.csv | head -1)
  python3 "$REPO/tools/rocprof/wait_attrib.py" "$CSV"
done
