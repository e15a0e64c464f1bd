#!/bin/bash
# Round-3 diagnostic for the loss_stream NaN race (NOTES.md "STREAM
# STATUS").  Round-2 bisect (gpu_nan_bisect.sh): loss_stream alone NaNs
# a 2000-step soak at step ~600-1600; d_stream alone and both-off are
# clean; record_stream + wait_stream(loss_stream) before optim_g.step
# did NOT cure it.  This harness splits the two remaining hypotheses:
#   PYTORCH_NO_CUDA_MEMORY_CACHING=1  -> every alloc is a fresh
#     hipMalloc; if the NaN disappears, it is a caching-allocator block
#     reuse hazard (a raw-pointer consumer the allocator cannot see).
#   AMD_SERIALIZE_KERNEL=3            -> every kernel launch synchronizes;
#     if the NaN disappears here but NOT above, it is a missing
#     inter-stream ordering edge (event/wait), not allocator reuse.
# Interpretation matrix:
#   nocache clean, serialize clean  -> allocator reuse (fix: record_stream
#     on the exact tensor, or allocate it on loss_stream)
#   nocache NaN,  serialize clean   -> ordering edge (find the missing wait)
#   both NaN                        -> likely NOT a race: with loss_stream
#     on, the grad accumulation ORDER at the fake/fake_sub input buffers
#     (stft-path grad from the side stream + adv-path grad from main)
#     differs from single-stream; non-associative fp addition seeds a
#     different trajectory and a GAN can diverge to NaN from that alone.
#     Confirm with the seed run: if the NaN step moves wildly (or
#     disappears) under a different seed, it is trajectory sensitivity,
#     not a correctness bug — compare g/d loss curves A/B over the first
#     200 steps (they should agree to ~1e-3 before chaos sets in).
# NOTE: nocache runs ~3-5x slower; 2000 steps may need the step count cut
# or the timeout raised.  Budget ~15 GPU-min for all three runs.
set -x
cd /root/repo
mkdir -p gpurun_out/lsdiag
run_soak () {  # $1=tag $2=extra-env $3=steps $4=timeout $5=seed
  local SEED=${5:-0}
  timeout $4 env $2 python - > gpurun_out/lsdiag/$1.log 2>&1 <<PYEOF
import torch
from melgan_amd.config import HParam, merge_default
from melgan_amd.engine.trainer import Trainer
raw = """
data: {train_dir: null, val_dir: null}
train: {batch_size: 64, num_workers: 0, discriminator_train_start_steps: 300}
audio: {segment_length: 16384, hop_length: 256}
model: {out_channels: 4, base_channels: 512, generator_ratio: [8, 4, 2],
        use_subband_stft_loss: true}
log: {chkpt_dir: /tmp/ld_$1, log_dir: /tmp/ldl_$1}
backend: {amp: bf16, fused_adam: true, loss_stream: true, d_stream: true}
"""
hp = merge_default(HParam(raw=raw))
torch.manual_seed($SEED)
tr = Trainer(hp, "ld$1", device=torch.device("cuda"), enable_tb=False)
g = torch.Generator().manual_seed(7 + $SEED)
bs = [(torch.randn(64, 80, 64, generator=g).cuda()*2-6,
       torch.rand(64, 1, 16384, generator=g).cuda()*1.6-.8) for _ in range(16)]
bad = -1
for i in range($3):
    gl, parts, dl = tr.train_step(*bs[i % 16])
    if i % 100 == 0:
        if not torch.isfinite(gl):
            bad = i
            break
print("$1 first_nan_window<=", bad if bad >= 0 else "none",
      f"final g={float(gl):.4f}")
PYEOF
  echo "$1: $(tail -1 gpurun_out/lsdiag/$1.log)"
}
run_soak base      ""                                  2000 400
run_soak nocache   "PYTORCH_NO_CUDA_MEMORY_CACHING=1"  800  900
run_soak serialize "AMD_SERIALIZE_KERNEL=3"            2000 900
run_soak seed11    ""                                  2000 400 11
