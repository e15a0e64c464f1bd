#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out/proffinal
timeout 400 python bench.py --steps 30 --warmup 8 --inference-rtf > gpurun_out/bench_headline.log 2>&1
echo "HEADLINE: $?"; tail -1 gpurun_out/bench_headline.log
timeout 600 python bench.py --config configs/mb_24k_vctk.yaml --steps 8 --warmup 3 > gpurun_out/bench_c4_final.log 2>&1
echo "C4: $?"; tail -1 gpurun_out/bench_c4_final.log | head -c 250; echo
timeout 600 python bench.py --config configs/fb_44k_8band.yaml --steps 8 --warmup 3 --inference-rtf > gpurun_out/bench_c5_final.log 2>&1
echo "C5: $?"; tail -1 gpurun_out/bench_c5_final.log | head -c 250; echo
export TMPDIR=/tmp
cd /tmp
timeout 500 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/proffinal -o rf -- \
    python /root/repo/bench.py --steps 5 --warmup 2 --batch-size 32 > /root/repo/gpurun_out/proffinal.log 2>&1
echo "PROF: $?"
