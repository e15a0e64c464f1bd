#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out/final
export TMPDIR=/tmp
cd /tmp
timeout 500 rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_LDS_BANK_CONFLICT \
    -d /tmp/fp -o fp -- python /root/repo/bench.py --steps 2 --warmup 1 --batch-size 128 --no-rtf \
    > /root/repo/gpurun_out/final/pmc.log 2>&1
echo "PMC EXIT: $?"
python - <<'PYEOF'
import sqlite3, glob, re, json, collections
out = collections.defaultdict(dict)
for f in glob.glob('/tmp/fp/*_results.db'):
    db = sqlite3.connect(f); cur = db.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    kd = [t for t in tables if t.startswith('rocpd_kernel_dispatch')][0]
    sfx = kd.replace('rocpd_kernel_dispatch_','')
    pe = [t for t in tables if t.startswith('rocpd_pmc_event')][0]
    pi = [t for t in tables if t.startswith('rocpd_info_pmc')][0]
    for name, dur, n in cur.execute(f"SELECT ks.display_name, SUM(k.end-k.start), COUNT(*) FROM {kd} k JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id=ks.id GROUP BY 1"):
        key = re.sub(r'<[^>]*>','',str(name)).split('(')[0].replace('void ','')[:52]
        out[key]['dur_ms'] = round(dur/1e6,2); out[key]['n'] = n
    for name, cname, val in cur.execute(f"""SELECT ks.display_name, pi.name, SUM(p.value) FROM {pe} p JOIN {pi} pi ON p.pmc_id=pi.id JOIN {kd} k ON p.event_id=k.event_id JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id=ks.id GROUP BY 1,2"""):
        key = re.sub(r'<[^>]*>','',str(name)).split('(')[0].replace('void ','')[:52]
        out[key][str(cname)] = int(val)
json.dump(out, open('/root/repo/gpurun_out/final/pmc.json','w'), indent=1)
rows = []
for k, v in out.items():
    wc = v.get('SQ_WAVE_CYCLES', 0)
    if not wc or 'melgan' not in k: continue
    rows.append((v.get('dur_ms',0), k, 100*v.get('SQ_VALU_MFMA_BUSY_CYCLES',0)/wc,
                 100*v.get('SQ_WAIT_ANY',0)/wc, v.get('SQ_LDS_BANK_CONFLICT',0)/1e6))
rows.sort(reverse=True)
print(f"{'ms':>7} {'mfma%':>6} {'wait%':>6} {'conflM':>7}  kernel")
for dur, k, mf, wa, cf in rows[:16]:
    print(f"{dur:7.2f} {mf:6.1f} {wa:6.1f} {cf:7.1f}  {k}")
PYEOF
cd /root/repo
# 2000-step soak on the final trainer
timeout 500 python - > gpurun_out/final/soak2000.log 2>&1 <<'PYEOF'
import torch, time
from melgan_amd.config import HParam, merge_default
from melgan_amd.engine.trainer import Trainer
raw = """
data: {train_dir: null, val_dir: null}
train: {batch_size: 64, num_workers: 0, discriminator_train_start_steps: 300}
audio: {segment_length: 16384, hop_length: 256}
model: {out_channels: 4, base_channels: 512, generator_ratio: [8, 4, 2],
        use_subband_stft_loss: true}
log: {chkpt_dir: /tmp/f2ck, log_dir: /tmp/f2lg}
backend: {amp: bf16, fused_adam: true}
"""
hp = merge_default(HParam(raw=raw))
torch.manual_seed(0)
tr = Trainer(hp, "fsoak", device=torch.device("cuda"), enable_tb=False)
g = torch.Generator().manual_seed(7)
bs = [(torch.randn(64, 80, 64, generator=g).cuda()*2-6,
       torch.rand(64, 1, 16384, generator=g).cuda()*1.6-.8) for _ in range(16)]
t0 = time.time()
for i in range(2000):
    gl, parts, dl = tr.train_step(*bs[i % 16])
    if i % 400 == 0 or i == 1999:
        torch.cuda.synchronize()
        print(f"step {tr.step} g={float(gl):.4f}"
              + (f" d={float(dl):.4f}" if dl is not None else "")
              + f" mem={torch.cuda.memory_allocated()/2**30:.3f}GB", flush=True)
print(f"2000 steps in {time.time()-t0:.1f}s  SOAK PASS")
PYEOF
echo "SOAK EXIT: $?"; tail -8 gpurun_out/final/soak2000.log
# 288 GB sizing: biggest batches
timeout 500 python bench.py --config configs/mb_24k_vctk.yaml --steps 3 --warmup 1 --batch-size 2048 --no-rtf > gpurun_out/final/b24k_2048.log 2>&1
echo "B2048: $(tail -1 gpurun_out/final/b24k_2048.log | head -c 300)"
