#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out/prof3
export TMPDIR=/tmp
cd /tmp
timeout 500 rocprofv3 --kernel-trace --stats -d /tmp/p3 -o tr -- \
    python /root/repo/bench.py --steps 4 --warmup 2 --batch-size 128 --no-rtf \
    > /root/repo/gpurun_out/prof3/trace.log 2>&1
echo "TRACE EXIT: $?"
timeout 500 rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_LDS_BANK_CONFLICT \
    -d /tmp/p3 -o pm1 -- python /root/repo/bench.py --steps 2 --warmup 1 --batch-size 128 --no-rtf \
    > /root/repo/gpurun_out/prof3/pmc1.log 2>&1
echo "PMC1 EXIT: $?"
timeout 500 rocprofv3 --pmc FETCH_SIZE WRITE_SIZE \
    -d /tmp/p3 -o pm2 -- python /root/repo/bench.py --steps 2 --warmup 1 --batch-size 128 --no-rtf \
    > /root/repo/gpurun_out/prof3/pmc2.log 2>&1
echo "PMC2 EXIT: $?"
python - <<'PYEOF'
import sqlite3, glob, re, json, collections
def agg(db_path, like=None, pmc=False):
    db = sqlite3.connect(db_path); cur = db.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    kd = [t for t in tables if t.startswith('rocpd_kernel_dispatch')][0]
    sfx = kd.replace('rocpd_kernel_dispatch_','')
    out = collections.defaultdict(dict)
    for name, dur, n in cur.execute(f"SELECT ks.display_name, SUM(k.end-k.start), COUNT(*) FROM {kd} k JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id=ks.id GROUP BY 1"):
        key = re.sub(r'<[^>]*>','',str(name)).split('(')[0].replace('void ','')[:60]
        out[key]['dur_ms'] = round(dur/1e6,2); out[key]['n'] = n
    if pmc:
        pe = [t for t in tables if t.startswith('rocpd_pmc_event')][0]
        pi = [t for t in tables if t.startswith('rocpd_info_pmc')][0]
        for name, cname, val in cur.execute(f"""SELECT ks.display_name, pi.name, SUM(p.value) FROM {pe} p JOIN {pi} pi ON p.pmc_id=pi.id JOIN {kd} k ON p.event_id=k.event_id JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id=ks.id GROUP BY 1,2"""):
            key = re.sub(r'<[^>]*>','',str(name)).split('(')[0].replace('void ','')[:60]
            out[key][str(cname)] = int(val)
    return out
res = {}
for f in glob.glob('/tmp/p3/*_results.db'):
    tag = 'trace' if '/tr_' in f or 'tr_results' in f else ('pmc1' if 'pm1' in f else 'pmc2')
    res[tag] = agg(f, pmc=tag.startswith('pmc'))
json.dump(res, open('/root/repo/gpurun_out/prof3/summary.json','w'), indent=1)
tr = res.get('trace', {})
for k, v in sorted(tr.items(), key=lambda kv: -kv[1].get('dur_ms',0))[:20]:
    print(f"{v.get('dur_ms',0):8.2f} ms {v.get('n',0):5d}x {k}")
PYEOF
tail -2 /root/repo/gpurun_out/prof3/trace.log
