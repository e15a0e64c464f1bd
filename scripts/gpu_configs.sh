#!/bin/bash
# Config-matrix validation: configs 4 (24k VCTK large-batch) and 5
# (44.1k 8-band deep-residual), plus bf16-vs-fp32 loss-curve check.
set -x
cd /root/repo
mkdir -p gpurun_out

timeout 700 python bench.py --config configs/mb_24k_vctk.yaml --steps 8 --warmup 3 \
    > gpurun_out/bench_c4_24k.log 2>&1
echo "C4 EXIT: $?"; tail -1 gpurun_out/bench_c4_24k.log

timeout 700 python bench.py --config configs/fb_44k_8band.yaml --steps 8 --warmup 3 --inference-rtf \
    > gpurun_out/bench_c5_44k.log 2>&1
echo "C5 EXIT: $?"; tail -1 gpurun_out/bench_c5_44k.log

timeout 600 python - > gpurun_out/bf16_vs_fp32.log 2>&1 <<'EOF'
# bf16 GAN numerics check (SURVEY.md §7 hard part 3): identical seeds,
# 20 steps, compare loss trajectories bf16 vs fp32.
import torch, json
from melgan_amd.config import HParam, merge_default
from melgan_amd.engine.trainer import Trainer

raw = """
data: {train_dir: null, val_dir: null}
train: {batch_size: 8, num_workers: 0, discriminator_train_start_steps: 0}
audio: {segment_length: 8192, hop_length: 256}
model: {out_channels: 4, base_channels: 256, generator_ratio: [8, 4, 2],
        use_subband_stft_loss: true}
log: {chkpt_dir: /tmp/ck, log_dir: /tmp/lg}
backend: {amp: %s, fused_adam: true}
"""
curves = {}
for amp in ("none", "bf16"):
    hp = merge_default(HParam(raw=raw % amp))
    torch.manual_seed(0)
    tr = Trainer(hp, f"numerics_{amp}", device=torch.device("cuda"),
                 enable_tb=False)
    torch.manual_seed(1)
    mel = torch.randn(8, 80, 32, device="cuda")
    audio = torch.rand(8, 1, 8192, device="cuda") * 1.6 - 0.8
    losses = []
    for i in range(20):
        g_loss, parts, d_loss = tr.train_step(mel, audio)
        losses.append(round(float(g_loss), 4))
    curves[amp] = losses
print(json.dumps(curves, indent=1))
import numpy as np
a = np.array(curves["none"]); b = np.array(curves["bf16"])
rel = np.abs(a - b) / np.abs(a)
print("max rel dev:", rel.max(), "final rel dev:", rel[-1])
print("fp32 decreasing:", a[-1] < a[0], "bf16 decreasing:", b[-1] < b[0])
EOF
echo "NUMERICS EXIT: $?"
tail -6 gpurun_out/bf16_vs_fp32.log
