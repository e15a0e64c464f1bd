#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out/nan
run_soak () {  # $1=tag $2=loss_stream $3=d_stream $4=env
  timeout 400 env $4 python - > gpurun_out/nan/$1.log 2>&1 <<PYEOF
import torch, time
from melgan_amd.config import HParam, merge_default
from melgan_amd.engine.trainer import Trainer
raw = """
data: {train_dir: null, val_dir: null}
train: {batch_size: 64, num_workers: 0, discriminator_train_start_steps: 300}
audio: {segment_length: 16384, hop_length: 256}
model: {out_channels: 4, base_channels: 512, generator_ratio: [8, 4, 2],
        use_subband_stft_loss: true}
log: {chkpt_dir: /tmp/nb_$1, log_dir: /tmp/nbl_$1}
backend: {amp: bf16, fused_adam: true, loss_stream: $2, d_stream: $3}
"""
hp = merge_default(HParam(raw=raw))
torch.manual_seed(0)
tr = Trainer(hp, "nb$1", device=torch.device("cuda"), enable_tb=False)
g = torch.Generator().manual_seed(7)
bs = [(torch.randn(64, 80, 64, generator=g).cuda()*2-6,
       torch.rand(64, 1, 16384, generator=g).cuda()*1.6-.8) for _ in range(16)]
bad = -1
for i in range(2000):
    gl, parts, dl = tr.train_step(*bs[i % 16])
    if i % 100 == 0:
        if not torch.isfinite(gl):
            bad = i
            break
print("$1 first_nan_window<=", bad if bad >= 0 else "none",
      f"final g={float(gl):.4f}")
PYEOF
  echo "$1: $(tail -1 gpurun_out/nan/$1.log)"
}
run_soak lsonly true false ""
run_soak dsonly false true ""
