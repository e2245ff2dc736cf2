#!/bin/bash
# T3/T5/T6 in ONE gpurun call (gpurun calls are budgeted —
# never one call per test). Writes machine-readable verdicts to gpurun_out/.
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 600 python -m pytest tests -m gpu -q > gpurun_out/run_all_pytest.log 2>&1
echo "pytest rc=$?" >> gpurun_out/run_all_pytest.log
timeout 120 ./build/nezha-info --comm-check > gpurun_out/run_all_comm.log 2>&1
timeout 240 python tools/race_screen.py > gpurun_out/run_all_race.log 2>&1
grep -E "passed|failed|OK" gpurun_out/run_all_*.log
