#!/bin/bash
# First-line GPU validation on a fresh MI355X box (run via gpurun).
set -x
cd /root/repo
mkdir -p gpurun_out
export PYTORCH_ROCM_ARCH=gfx950

# 1. extension import + smoke (fwd+bwd+fused optimizer on cuda:0)
timeout 600 python __graft_entry__.py smoke > gpurun_out/smoke.log 2>&1
echo "smoke rc=$?"
tail -3 gpurun_out/smoke.log

# 2. GPU kernel parity tests
timeout 900 python -m pytest tests/ -q -m gpu > gpurun_out/gpu_tests.log 2>&1
echo "gpu tests rc=$?"
tail -5 gpurun_out/gpu_tests.log

# 3. bench: default driver contract shape
timeout 900 python bench.py --steps 30 --warmup 10 > gpurun_out/bench_n1.log 2>&1
echo "bench rc=$?"
cat gpurun_out/bench_n1.log

# 4. rocprof kernel stats on a short bench run
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 900 rocprofv3 --kernel-trace --stats -d gpurun_out/prof -o bench \
  -- python bench.py --steps 10 --warmup 5 > gpurun_out/rocprof.log 2>&1
echo "rocprof rc=$?"
ls gpurun_out/prof 2>/dev/null | head
grep -m1 '"metric"' gpurun_out/rocprof.log
