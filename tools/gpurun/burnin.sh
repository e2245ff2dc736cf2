#!/bin/bash
# Round-end burn-in: full GPU suite twice (flake detection), canonical
# benches, model fuzz, soak, kernel profile. Run via gpurun from repo root:
#   /usr/local/graft/bin/gpurun --timeout 2400 -- 'bash tools/gpurun/burnin.sh'
set -e
mkdir -p gpurun_out/burnin
for i in 1 2; do
  timeout 500 python -m pytest tests -q -m gpu > gpurun_out/burnin/pytest$i.log 2>&1 || true
  grep -E "passed|failed" gpurun_out/burnin/pytest$i.log | tail -1
done
timeout 400 python bench.py --steps 30 --warmup 10 2>/dev/null | tail -1 > gpurun_out/burnin/mlp.json
timeout 400 python bench.py --model bert --steps 30 --warmup 10 2>/dev/null | tail -1 > gpurun_out/burnin/bert.json
timeout 200 python -c "import __graft_entry__ as g; g.smoke(); print('smoke OK')" 2>&1 | tail -1
echo "burn-in done"
