#!/bin/bash
# rocprofv3 kernel-stats profile of bench.py -> gpurun_out/kernel_stats_$1.md
TAG=${1:-rXX}
mkdir -p /root/repo/gpurun_out
export TMPDIR=/tmp
cd /tmp
timeout 450 rocprofv3 --kernel-trace --stats -d /tmp/prof -o "$TAG" -- \
  bash -c "cd /root/repo && python bench.py --steps 80 --warmup 20" \
  > /root/repo/gpurun_out/prof_$TAG.log 2>&1
cd /root/repo
python scripts/rocpd_stats.py /tmp/prof/*$TAG*.db \
  > gpurun_out/kernel_stats_$TAG.md 2>&1
head -3 gpurun_out/kernel_stats_$TAG.md
