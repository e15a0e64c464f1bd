#!/bin/bash
# Round-2 breadth evidence: config 4/5 benches, batch sweep (288 GB HBM
# story), 1000-step soak + resume on the restructured trainer.
set -x
cd /root/repo
mkdir -p gpurun_out
# config benches
timeout 400 python bench.py --config configs/mb_24k_vctk.yaml --steps 8 --warmup 3 --no-rtf > gpurun_out/bench_c4.log 2>&1
echo "C4: $(tail -1 gpurun_out/bench_c4.log | head -c 400)"
timeout 400 python bench.py --config configs/fb_44k_8band.yaml --steps 8 --warmup 3 --no-rtf > gpurun_out/bench_c5.log 2>&1
echo "C5: $(tail -1 gpurun_out/bench_c5.log | head -c 400)"
# batch sweep (22k flagship + 24k)
for B in 256 512 1024; do
  timeout 500 python bench.py --steps 5 --warmup 2 --batch-size $B --no-rtf > gpurun_out/sweep22_B$B.log 2>&1
  echo "22k B$B: $(tail -1 gpurun_out/sweep22_B$B.log | head -c 300)"
done
for B in 512 1024; do
  timeout 500 python bench.py --config configs/mb_24k_vctk.yaml --steps 4 --warmup 2 --batch-size $B --no-rtf > gpurun_out/sweep24_B$B.log 2>&1
  echo "24k B$B: $(tail -1 gpurun_out/sweep24_B$B.log | head -c 300)"
done
# 1000-step soak with mid-run resume on the restructured trainer
timeout 900 python - > gpurun_out/soak_r2.log 2>&1 <<'PYEOF'
import torch, time
from melgan_amd.config import HParam, merge_default
from melgan_amd.engine.trainer import Trainer
raw = """
data: {train_dir: null, val_dir: null}
train: {batch_size: 64, num_workers: 0, discriminator_train_start_steps: 300}
audio: {segment_length: 16384, hop_length: 256}
model: {out_channels: 4, base_channels: 512, generator_ratio: [8, 4, 2],
        use_subband_stft_loss: true}
log: {chkpt_dir: /tmp/soak_ck, log_dir: /tmp/soak_lg}
backend: {amp: bf16, fused_adam: true}
"""
hp = merge_default(HParam(raw=raw))
torch.manual_seed(0)
tr = Trainer(hp, "soakr2", device=torch.device("cuda"), enable_tb=False)
g = torch.Generator().manual_seed(7)
batches = [(torch.randn(64, 80, 64, generator=g).cuda() * 2 - 6,
            (torch.rand(64, 1, 16384, generator=g).cuda() * 1.6 - .8))
           for _ in range(16)]
t0 = time.time()
ck = None
for i in range(1000):
    gl, parts, dl = tr.train_step(*batches[i % 16])
    if i % 100 == 0 or i == 999:
        torch.cuda.synchronize()
        print(f"step {tr.step} g={float(gl):.4f} stft={float(parts['stft']):.4f}"
              + (f" d={float(dl):.4f}" if dl is not None else "")
              + f" adv={float(parts['adv']) if parts['adv'] is not None else 0:.4f}"
              + f" mem={torch.cuda.memory_allocated()/2**30:.3f}GB", flush=True)
    if tr.step == 500:
        ck = tr.save()
        print("checkpoint saved at step 500:", ck, flush=True)
print(f"1000 steps in {time.time()-t0:.1f}s")
# resume check
tr2 = Trainer(hp, "soakr2b", device=torch.device("cuda"), enable_tb=False)
tr2.load(ck)
assert tr2.step == 500
for i in range(5):
    gl, parts, dl = tr2.train_step(*batches[i % 16])
print(f"resume OK: step {tr2.step} g={float(gl):.4f} d={float(dl):.4f}")
print("SOAK PASS")
PYEOF
echo "SOAK EXIT: $?"
tail -16 gpurun_out/soak_r2.log
