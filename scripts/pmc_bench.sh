#!/bin/bash
# PMC counter run (counters only, no trace flags - gpurun policy)
mkdir -p /root/repo/gpurun_out
export TMPDIR=/tmp
cd /tmp
timeout 450 rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_BUSY_CYCLES SQ_INSTS_MFMA SQ_INSTS_VALU SQ_LDS_BANK_CONFLICT -d /tmp/pmc -o "$1" -- \
  bash -c "cd /root/repo && python bench.py --steps 30 --warmup 10" \
  > /root/repo/gpurun_out/pmc_$1.log 2>&1
cd /root/repo
python scripts/rocpd_pmc.py /tmp/pmc/*$1*.db gpurun_out/pmc_$1.md >> gpurun_out/pmc_$1.log 2>&1 || true
tail -3 gpurun_out/pmc_$1.log
