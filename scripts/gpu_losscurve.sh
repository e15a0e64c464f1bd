#!/bin/bash
# 300-step training run (loss decrease evidence) + 2-rank-on-1-GPU RCCL probe.
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 900 python - > gpurun_out/losscurve.log 2>&1 <<'PYEOF'
import torch, json
from melgan_amd.config import HParam, merge_default
from melgan_amd.engine.trainer import Trainer
raw = """
data: {train_dir: null, val_dir: null}
train: {batch_size: 32, num_workers: 0, discriminator_train_start_steps: 150,
        lambda_adv: 2.5}
audio: {segment_length: 16384, hop_length: 256}
model: {out_channels: 4, base_channels: 512, generator_ratio: [8, 4, 2],
        use_subband_stft_loss: true}
log: {chkpt_dir: /tmp/lc_ck, log_dir: /tmp/lc_lg, summary_interval: 1000,
      validation_interval: 100000, save_interval: 100000}
backend: {amp: bf16, fused_adam: true}
"""
hp = merge_default(HParam(raw=raw))
torch.manual_seed(0)
tr = Trainer(hp, "losscurve", device=torch.device("cuda"), enable_tb=False)
gen = torch.Generator().manual_seed(7)
batches = []
for i in range(8):
    mel_seed = torch.randn(32, 80, 64, generator=gen)
    audio = (torch.rand(32, 1, 16384, generator=gen) * 1.6 - 0.8)
    batches.append((mel_seed.cuda(), audio.cuda()))
hist = []
for i in range(300):
    g_loss, parts, d_loss = tr.train_step(*batches[i % 8])
    if i % 10 == 0:
        hist.append((i, round(float(parts["stft"]), 4),
                     round(float(g_loss), 4),
                     round(float(d_loss), 4) if d_loss is not None else None))
print(json.dumps(hist))
first = [h[1] for h in hist[:3]]
last = [h[1] for h in hist[-3:]]
print("stft first3:", first, "last3:", last,
      "decreased:", sum(last)/3 < sum(first)/3)
PYEOF
echo "LOSSCURVE EXIT: $?"
tail -2 gpurun_out/losscurve.log

# 2 ranks on 1 GPU over RCCL (may refuse duplicate-GPU; probe only)
timeout 300 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 --master-port 29711 bench.py --steps 3 --warmup 1 --batch-size 8 \
    > gpurun_out/rccl2_probe.log 2>&1
echo "RCCL2 EXIT: $?"
tail -4 gpurun_out/rccl2_probe.log
