#!/bin/bash
# MFMA tier validation + bench + profile.
set -x
cd /root/repo
mkdir -p gpurun_out/prof17
timeout 900 python -m pytest tests/test_kernels_gpu.py -q -m gpu --timeout 300 \
    > gpurun_out/gpu_tests17.log 2>&1
echo "PYTEST EXIT: $?"
tail -15 gpurun_out/gpu_tests17.log
timeout 600 python bench.py --steps 15 --warmup 5 --inference-rtf \
    > gpurun_out/bench_tch128.log 2>&1
echo "BENCH EXIT: $?"
tail -2 gpurun_out/bench_tch128.log
export TMPDIR=/tmp
cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof17 -o r17 -- \
    python /root/repo/bench.py --steps 5 --warmup 2 --batch-size 32 \
    > /root/repo/gpurun_out/prof17_bench.log 2>&1
echo "PROF EXIT: $?"
