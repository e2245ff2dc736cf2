#!/bin/bash
# Per-kernel time breakdown of the flagship training step.
# --stats/--kernel-trace only — never combined with --pmc (gpurun refuses).
set -ex
cd /tmp && export TMPDIR=/tmp
REPO=${REPO:-/root/repo}
OUT=${OUT:-$REPO/gpurun_out/rocprof}
mkdir -p "$OUT"
rocprofv3 --stats -d "$OUT/stats" -o step \
  -- python "$REPO/bench.py" --steps 10 --warmup 3 2>&1 | tail -40
echo "stats in $OUT/stats"
