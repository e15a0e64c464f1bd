#!/bin/bash
# Round-2 kernel-stats profile of the flagship bench at B=128.
set -x
cd /root/repo
mkdir -p gpurun_out/prof2
export TMPDIR=/tmp
cd /tmp
timeout 700 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof2 -o r2 -- \
    python /root/repo/bench.py --steps 5 --warmup 2 --batch-size 128 --no-rtf \
    > /root/repo/gpurun_out/prof2_bench.log 2>&1
echo "PROF EXIT: $?"
find /root/repo/gpurun_out/prof2 -name "*kernel_stats.csv" -exec head -40 {} \;
tail -2 /root/repo/gpurun_out/prof2_bench.log
