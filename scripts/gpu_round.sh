#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
export PYTORCH_ROCM_ARCH=gfx950
export TMPDIR=/tmp
timeout 900 python -m distributed_reinforcement_learning_amd.ops.build > gpurun_out/build.log 2>&1
echo "build rc=$?"
timeout 1500 python -m pytest tests/ -q -m gpu > gpurun_out/gpu_tests.log 2>&1
echo "gpu tests rc=$?"; tail -3 gpurun_out/gpu_tests.log
timeout 900 python -m pytest tests/test_end_to_end.py -q -x > gpurun_out/e2e.log 2>&1
echo "e2e rc=$?"; tail -3 gpurun_out/e2e.log
timeout 900 python bench.py --steps 150 --warmup 30 > gpurun_out/bench_graph.log 2>&1
echo "bench rc=$?"; grep -o '"value": [0-9.]*' gpurun_out/bench_graph.log; grep -o '"ms_per_step": [0-9.]*' gpurun_out/bench_graph.log
timeout 900 python bench.py --steps 60 --warmup 15 --model resnet > gpurun_out/bench_resnet.log 2>&1
echo "resnet rc=$?"; grep -o '"value": [0-9.]*' gpurun_out/bench_resnet.log; grep -o '"ms_per_step": [0-9.]*' gpurun_out/bench_resnet.log
timeout 600 python -c "import __graft_entry__ as g; g.build(); g.smoke(); print('graft clean')" > gpurun_out/smoke.log 2>&1
echo "smoke rc=$?"; tail -2 gpurun_out/smoke.log
cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d /tmp/prof -o r23 -- bash -c "cd /root/repo && python bench.py --steps 80 --warmup 20" > /root/repo/gpurun_out/rocprof.log 2>&1
echo "rocprof rc=$?"
cd /root/repo
python scripts/rocpd_stats.py /tmp/prof/*r23*.db > gpurun_out/kernel_stats_r23.md 2>&1 || python scripts/rocpd_stats.py $(ls /tmp/prof/* | head -1) > gpurun_out/kernel_stats_r23.md 2>&1
echo "stats rc=$?"; head -30 gpurun_out/kernel_stats_r23.md
