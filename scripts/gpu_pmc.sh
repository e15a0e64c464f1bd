#!/bin/bash
# PMC counter capture for the fused kernels (north-star evidence:
# MFMA utilization, LDS bank conflicts, HBM bytes).
# NOTE: --pmc must NOT be combined with trace domains (gpurun refuses).
set -x
cd /root/repo
mkdir -p gpurun_out/pmc
export TMPDIR=/tmp
cd /tmp
rocprofv3 -L 2>/dev/null | grep -iE "MfmaUtil|VALUBusy|LDS_BANK|FETCH_SIZE|WRITE_SIZE|OccupancyPercent|SQ_WAVE_CYCLES" | head -20 \
    > /root/repo/gpurun_out/pmc/available.txt
# pass 1: MFMA / VALU / LDS conflicts (SQ block <=8 slots)
timeout 500 rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_BUSY_CYCLES SQ_WAVE_CYCLES SQ_LDS_BANK_CONFLICT \
    -d /root/repo/gpurun_out/pmc -o pass1 -- \
    python /root/repo/bench.py --steps 2 --warmup 1 --batch-size 32 \
    > /root/repo/gpurun_out/pmc/pass1.log 2>&1
echo "PASS1 EXIT: $?"
# pass 2: HBM fetch (TCC, FETCH_SIZE costs 3 of 4 slots)
timeout 500 rocprofv3 --pmc FETCH_SIZE \
    -d /root/repo/gpurun_out/pmc -o pass2 -- \
    python /root/repo/bench.py --steps 2 --warmup 1 --batch-size 32 \
    > /root/repo/gpurun_out/pmc/pass2.log 2>&1
echo "PASS2 EXIT: $?"
ls -la /root/repo/gpurun_out/pmc/
