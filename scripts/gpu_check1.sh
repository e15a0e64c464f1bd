#!/bin/bash
# First GPU validation: kernel numerics + quick bench.
set -x
cd /root/repo
mkdir -p gpurun_out
python -c "import torch; print(torch.__version__, torch.cuda.is_available(), torch.cuda.get_device_name(0))" 2>&1 | tail -2
timeout 900 python -m pytest tests/test_kernels_gpu.py -q -m gpu -x --timeout 300 \
    > gpurun_out/gpu_tests.log 2>&1
echo "PYTEST EXIT: $?"
tail -30 gpurun_out/gpu_tests.log
timeout 600 python bench.py --steps 10 --warmup 3 --batch-size 16 \
    > gpurun_out/bench_quick.log 2>&1
echo "BENCH EXIT: $?"
tail -5 gpurun_out/bench_quick.log
