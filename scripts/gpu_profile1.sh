#!/bin/bash
# rocprofv3 kernel-stats profile of the flagship bench + full-batch bench.
set -x
cd /root/repo
mkdir -p gpurun_out/prof
export TMPDIR=/tmp
cd /tmp
timeout 700 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof -o r1 -- \
    python /root/repo/bench.py --steps 5 --warmup 2 --batch-size 32 \
    > /root/repo/gpurun_out/prof_bench.log 2>&1
echo "PROF EXIT: $?"
cd /root/repo
timeout 600 python bench.py --steps 15 --warmup 5 --inference-rtf \
    > gpurun_out/bench_b64.log 2>&1
echo "BENCH EXIT: $?"
ls -la gpurun_out/prof/
head -25 gpurun_out/prof/*kernel_stats.csv 2>/dev/null || find gpurun_out/prof -name "*.csv" | head
tail -3 gpurun_out/prof_bench.log
tail -3 gpurun_out/bench_b64.log
