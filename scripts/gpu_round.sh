#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
export PYTORCH_ROCM_ARCH=gfx950
timeout 900 python -m distributed_reinforcement_learning_amd.ops.build > gpurun_out/build.log 2>&1
echo "build rc=$?"
# A: baseline
timeout 900 python bench.py --steps 100 --warmup 20 > gpurun_out/bench_a.log 2>&1
grep -o '"ms_per_step": [0-9.]*' gpurun_out/bench_a.log
# B: TunableOp (tune during warmup, then measure)
export PYTORCH_TUNABLEOP_ENABLED=1
export PYTORCH_TUNABLEOP_FILENAME=gpurun_out/tunableop_%d.csv
timeout 1200 python bench.py --steps 100 --warmup 20 > gpurun_out/bench_tuned.log 2>&1
echo "tuned rc=$?"
grep -o '"ms_per_step": [0-9.]*' gpurun_out/bench_tuned.log
# C: tuned again (reads the CSV, no tuning overhead)
timeout 900 python bench.py --steps 100 --warmup 20 > gpurun_out/bench_tuned2.log 2>&1
grep -o '"ms_per_step": [0-9.]*' gpurun_out/bench_tuned2.log
unset PYTORCH_TUNABLEOP_ENABLED
