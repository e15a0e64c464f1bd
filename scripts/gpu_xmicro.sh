#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 300 python benchmarks/xstage_micro.py > gpurun_out/xmicro_new.log 2>&1
echo "NEW EXIT $?"
MELGAN_AMD_NO_XSTAGE=1 timeout 300 python benchmarks/xstage_micro.py > gpurun_out/xmicro_old.log 2>&1
echo "OLD EXIT $?"
# PMC counters on the micro (xstage kernels only appear in the new run)
export TMPDIR=/tmp; cd /tmp
timeout 400 rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_LDS_BANK_CONFLICT \
    -d /tmp/pmcx -o px -- python /root/repo/benchmarks/xstage_micro.py \
    > /root/repo/gpurun_out/xmicro_pmc.log 2>&1
echo "PMC EXIT $?"
python - <<'PYEOF'
import sqlite3, glob, re, json, collections
out = {}
for db_path in glob.glob('/tmp/pmcx/*_results.db'):
    db = sqlite3.connect(db_path); cur = db.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    kd = [t for t in tables if t.startswith('rocpd_kernel_dispatch')][0]
    sfx = kd.replace('rocpd_kernel_dispatch_','')
    pe = [t for t in tables if t.startswith('rocpd_pmc_event')][0]
    pi = [t for t in tables if t.startswith('rocpd_info_pmc')][0]
    try:
        rows = cur.execute(f"""
        SELECT ks.display_name, pi.name, SUM(p.value), COUNT(*)
        FROM {pe} p JOIN {pi} pi ON p.pmc_id = pi.id
        JOIN {kd} k ON p.event_id = k.event_id
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
        WHERE ks.display_name LIKE '%xstage%'
        GROUP BY 1,2""").fetchall()
    except Exception as e:
        print('q1 fail', e); rows = []
    for name, cname, val, n in rows:
        key = re.sub(r'<[^>]*>','',str(name)).split('(')[0][:50]
        out.setdefault(key, {})[str(cname)] = int(val)
        out[key]['n'] = n
    for name, dur in cur.execute(f"SELECT ks.display_name, SUM(k.end-k.start) FROM {kd} k JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id=ks.id WHERE ks.display_name LIKE '%xstage%' GROUP BY 1"):
        key = re.sub(r'<[^>]*>','',str(name)).split('(')[0][:50]
        out.setdefault(key, {})['dur_ms'] = round(dur/1e6,2)
json.dump(out, open('/root/repo/gpurun_out/xmicro_pmc.json','w'), indent=1)
print(json.dumps(out, indent=1))
PYEOF
cat /root/repo/gpurun_out/xmicro_new.log /root/repo/gpurun_out/xmicro_old.log
