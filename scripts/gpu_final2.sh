#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 600 python -m pytest tests/test_kernels_gpu.py -q -m gpu --timeout 300 > gpurun_out/gpu_tests21.log 2>&1
echo "PYTEST: $?"; tail -2 gpurun_out/gpu_tests21.log
timeout 400 python bench.py --steps 15 --warmup 5 --inference-rtf > gpurun_out/bench_phasepack2.log 2>&1
echo "BENCH: $?"; tail -1 gpurun_out/bench_phasepack2.log | head -c 250; echo
timeout 900 python - > gpurun_out/train1000_pp.log 2>&1 <<'PYEOF'
import torch, json
from melgan_amd.config import HParam, merge_default
from melgan_amd.engine.trainer import Trainer
raw = """
data: {train_dir: null, val_dir: null}
train: {batch_size: 32, num_workers: 0, discriminator_train_start_steps: 300,
        lambda_adv: 2.5}
audio: {segment_length: 16384, hop_length: 256}
model: {out_channels: 4, base_channels: 512, generator_ratio: [8, 4, 2],
        use_subband_stft_loss: true}
log: {chkpt_dir: /tmp/t1k_ck, log_dir: /tmp/t1k_lg, summary_interval: 100000,
      validation_interval: 1000000, save_interval: 1000000}
backend: {amp: bf16, fused_adam: true, phase_timing: true}
"""
hp = merge_default(HParam(raw=raw))
torch.manual_seed(0)
tr = Trainer(hp, "t1k", device=torch.device("cuda"), enable_tb=False)
gen = torch.Generator().manual_seed(7)
batches = [( (torch.randn(32,80,64,generator=gen)).cuda(),
             (torch.rand(32,1,16384,generator=gen)*1.6-0.8).cuda())
           for _ in range(16)]
hist = []
for i in range(500):
    g_loss, parts, d_loss = tr.train_step(*batches[i % 16])
    if i % 50 == 0:
        hist.append((i, round(float(parts["stft"]),4),
                     round(float(d_loss),4) if d_loss is not None else None))
ck = tr.save()
tr2 = Trainer(hp, "t1k_resumed", device=torch.device("cuda"), enable_tb=False)
tr2.load(ck)
assert tr2.step == 500
for i in range(500, 1000):
    g_loss, parts, d_loss = tr2.train_step(*batches[i % 16])
    if i % 50 == 0:
        hist.append((i, round(float(parts["stft"]),4),
                     round(float(d_loss),4) if d_loss is not None else None))
print(json.dumps(hist))
print("phases(ms):", {k: round(v,2) for k,v in tr2.last_phases.items()})
print("stft 0:", hist[0][1], "-> 450:", hist[9][1], "-> 950:", hist[-1][1])
PYEOF
echo "T1K: $?"; tail -3 gpurun_out/train1000_pp.log
timeout 500 python - > gpurun_out/bf16_vs_fp32_pp.log 2>&1 <<'PYEOF'
import torch, json, numpy as np
from melgan_amd.config import HParam, merge_default
from melgan_amd.engine.trainer import Trainer
raw = """
data: {train_dir: null, val_dir: null}
train: {batch_size: 8, num_workers: 0, discriminator_train_start_steps: 0}
audio: {segment_length: 8192, hop_length: 256}
model: {out_channels: 4, base_channels: 256, generator_ratio: [8, 4, 2],
        use_subband_stft_loss: true}
log: {chkpt_dir: /tmp/ck2, log_dir: /tmp/lg2}
backend: {amp: %s, fused_adam: true}
"""
curves = {}
for amp in ("none", "bf16"):
    hp = merge_default(HParam(raw=raw % amp))
    torch.manual_seed(0)
    tr = Trainer(hp, f"nx_{amp}", device=torch.device("cuda"), enable_tb=False)
    torch.manual_seed(1)
    mel = torch.randn(8, 80, 32, device="cuda")
    audio = torch.rand(8, 1, 8192, device="cuda") * 1.6 - 0.8
    losses = []
    for i in range(20):
        g_loss, parts, d_loss = tr.train_step(mel, audio)
        losses.append(round(float(g_loss), 4))
    curves[amp] = losses
a = np.array(curves["none"]); b = np.array(curves["bf16"])
print(json.dumps(curves))
print("max rel dev:", float(np.abs(a-b).max()/np.abs(a).max()),
      "decreasing:", a[-1] < a[0], b[-1] < b[0])
PYEOF
echo "NUMERICS: $?"; tail -2 gpurun_out/bf16_vs_fp32_pp.log
