#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
for B in 256; do
  timeout 500 python bench.py --steps 8 --warmup 3 --batch-size $B > gpurun_out/bench2_B$B.log 2>&1
  echo "c2 B$B: $(tail -1 gpurun_out/bench2_B$B.log | head -c 300)"
done
for B in 384 512; do
  timeout 600 python bench.py --config configs/mb_24k_vctk.yaml --steps 5 --warmup 2 --batch-size $B > gpurun_out/bench_c4_B$B.log 2>&1
  echo "c4 B$B: $(tail -1 gpurun_out/bench_c4_B$B.log | head -c 300)"
done
