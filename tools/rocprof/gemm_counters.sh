#!/bin/bash
# C4 evidence: rocprofv3 counter capture for the GEMM (K1)
# and LayerNorm (K3) kernels — MFMA utilization, LDS bank conflicts, HBM
# bytes. PMC counters must run WITHOUT any trace domain (gpurun refuses the
# combination); kernel timing comes from the separate --stats run
# (kernel_stats.sh). Run on the GPU box:
#   cd /tmp && export TMPDIR=/tmp && bash /root/repo/tools/rocprof/gemm_counters.sh
set -ex
cd /tmp && export TMPDIR=/tmp
REPO=${REPO:-/root/repo}
OUT=${OUT:-$REPO/gpurun_out/rocprof}
mkdir -p "$OUT"

# Pass 1: MFMA + VALU activity (SQ block, <=8 slots)
rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_BUSY_CYCLES SQ_WAVES \
  SQ_INSTS_VALU SQ_INSTS_MFMA -d "$OUT/pmc_mfma" -o mfma \
  -- "$REPO/build/nezha-bench" --op=gemm --iters=3 2>&1 | tail -5

# Pass 2: LDS bank conflicts + LDS activity
rocprofv3 --pmc SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE SQ_WAIT_INST_LDS \
  -d "$OUT/pmc_lds" -o lds \
  -- "$REPO/build/nezha-bench" --op=gemm --iters=3 2>&1 | tail -5

# Pass 3: HBM traffic (TCC block; FETCH_SIZE reads ~1/2 actual on gfx950 —
# calibrate before trusting absolutes, MI355X_MICROARCH.md §HBM)
rocprofv3 --pmc FETCH_SIZE -d "$OUT/pmc_hbm" -o hbm \
  -- "$REPO/build/nezha-bench" --op=gemm --iters=3 2>&1 | tail -5

echo "counter CSVs in $OUT"
