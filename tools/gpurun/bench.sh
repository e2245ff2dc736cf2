#!/bin/bash
# T7 bench: timed windows for the BASELINE.md comparison +
# kernel microbench; JSON lines land in gpurun_out/.
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 300 python bench.py --steps 30 --warmup 10 > gpurun_out/bench_mlp.json 2>&1
timeout 300 python bench.py --model bert --steps 10 --warmup 3 > gpurun_out/bench_bert.json 2>&1
timeout 240 ./build/nezha-bench --op=gemm --iters=10 > gpurun_out/bench_kernels.txt 2>&1
cat gpurun_out/bench_mlp.json gpurun_out/bench_bert.json
