#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 1000 python - > gpurun_out/soak2000.log 2>&1 <<'PYEOF'
import torch, json
from melgan_amd.config import HParam, merge_default
from melgan_amd.engine.trainer import Trainer
raw = """
data: {train_dir: null, val_dir: null}
train: {batch_size: 64, num_workers: 0, discriminator_train_start_steps: 400}
audio: {segment_length: 16384, hop_length: 256}
model: {out_channels: 4, base_channels: 512, generator_ratio: [8, 4, 2],
        use_subband_stft_loss: true}
log: {chkpt_dir: /tmp/soak_ck, log_dir: /tmp/soak_lg, summary_interval: 10**9,
      validation_interval: 1000000, save_interval: 1000000}
backend: {amp: bf16, fused_adam: true}
"""
hp = merge_default(HParam(raw=raw))
torch.manual_seed(0)
tr = Trainer(hp, "soak", device=torch.device("cuda"), enable_tb=False)
gen = torch.Generator().manual_seed(7)
batches = [((torch.randn(64,80,64,generator=gen)).cuda(),
            (torch.rand(64,1,16384,generator=gen)*1.6-0.8).cuda())
           for _ in range(16)]
mem0 = None
hist = []
for i in range(2000):
    g_loss, parts, d_loss = tr.train_step(*batches[i % 16])
    if i == 100:
        torch.cuda.reset_peak_memory_stats()
        mem0 = torch.cuda.memory_allocated()
    if i % 200 == 0:
        hist.append((i, round(float(parts["stft"]),4),
                     round(float(d_loss),4) if d_loss is not None else None,
                     round(torch.cuda.memory_allocated()/2**30, 3)))
print(json.dumps(hist))
growth = torch.cuda.memory_allocated() - mem0
print("mem growth after step 100:", growth, "bytes; finite:",
      all(torch.isfinite(g_loss)), "stft:", hist[0][1], "->", hist[-1][1])
PYEOF
echo "SOAK: $?"; tail -2 gpurun_out/soak2000.log
