#!/bin/bash
mkdir -p /root/repo/gpurun_out
export TMPDIR=/tmp
cd /tmp
timeout 450 rocprofv3 --kernel-trace --stats -d /tmp/prof -o "$1" -- \
  bash -c "cd /root/repo && python -u scripts/bench_algos.py" \
  > /root/repo/gpurun_out/prof_algo_$1.log 2>&1
cd /root/repo
python scripts/rocpd_stats.py /tmp/prof/*$1*.db \
  > gpurun_out/kernel_stats_$1.md 2>&1
head -3 gpurun_out/kernel_stats_$1.md
