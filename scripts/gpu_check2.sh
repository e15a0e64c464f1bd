#!/bin/bash
# MFMA tier validation + bench + profile.
set -x
cd /root/repo
mkdir -p gpurun_out/prof18
timeout 900 python -m pytest tests/test_kernels_gpu.py -q -m gpu --timeout 300 \
    > gpurun_out/gpu_tests18.log 2>&1
echo "PYTEST EXIT: $?"
tail -15 gpurun_out/gpu_tests18.log
timeout 600 python bench.py --steps 15 --warmup 5 --inference-rtf \
    > gpurun_out/bench_phasepack.log 2>&1
echo "BENCH EXIT: $?"
tail -2 gpurun_out/bench_phasepack.log
export TMPDIR=/tmp
cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof18 -o r18 -- \
    python /root/repo/bench.py --steps 5 --warmup 2 --batch-size 32 \
    > /root/repo/gpurun_out/prof18_bench.log 2>&1
echo "PROF EXIT: $?"
