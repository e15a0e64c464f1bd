#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out/hbm
export TMPDIR=/tmp
cd /tmp
timeout 600 rocprofv3 --pmc FETCH_SIZE -d /tmp/hb -o f1 -- \
    python /root/repo/bench.py --steps 2 --warmup 1 --batch-size 128 --no-rtf \
    > /root/repo/gpurun_out/hbm/fetch.log 2>&1
echo "FETCH EXIT: $?"
timeout 600 rocprofv3 --pmc WRITE_SIZE -d /tmp/hb -o w1 -- \
    python /root/repo/bench.py --steps 2 --warmup 1 --batch-size 128 --no-rtf \
    > /root/repo/gpurun_out/hbm/write.log 2>&1
echo "WRITE EXIT: $?"
python - <<'PYEOF'
import sqlite3, glob, re, json, collections
out = collections.defaultdict(dict)
for f in glob.glob('/tmp/hb/*_results.db'):
    db = sqlite3.connect(f); cur = db.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    kd = [t for t in tables if t.startswith('rocpd_kernel_dispatch')][0]
    sfx = kd.replace('rocpd_kernel_dispatch_','')
    pe = [t for t in tables if t.startswith('rocpd_pmc_event')][0]
    pi = [t for t in tables if t.startswith('rocpd_info_pmc')][0]
    for name, dur, n in cur.execute(f"SELECT ks.display_name, SUM(k.end-k.start), COUNT(*) FROM {kd} k JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id=ks.id GROUP BY 1"):
        key = re.sub(r'<[^>]*>','',str(name)).split('(')[0].replace('void ','')[:56]
        out[key]['dur_ms'] = round(out[key].get('dur_ms',0) + dur/1e6, 2)
        out[key]['n'] = n
    for name, cname, val in cur.execute(f"""SELECT ks.display_name, pi.name, SUM(p.value) FROM {pe} p JOIN {pi} pi ON p.pmc_id=pi.id JOIN {kd} k ON p.event_id=k.event_id JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id=ks.id GROUP BY 1,2"""):
        key = re.sub(r'<[^>]*>','',str(name)).split('(')[0].replace('void ','')[:56]
        out[key][str(cname)] = int(val)
json.dump(out, open('/root/repo/gpurun_out/hbm/summary.json','w'), indent=1)
rows = []
for k, v in out.items():
    if 'FETCH_SIZE' not in v and 'WRITE_SIZE' not in v: continue
    # FETCH_SIZE/WRITE_SIZE are KB; FETCH under-counts coalesced reads 2x
    fe = v.get('FETCH_SIZE',0)*1024; wr = v.get('WRITE_SIZE',0)*1024
    dur = v.get('dur_ms',0)/2*1e-3/1.0  # both passes summed dur -> halve
    bw = (fe*2 + wr)/dur/1e12 if dur else 0
    rows.append((v.get('dur_ms',0), k, fe/1e9, wr/1e9, bw))
rows.sort(reverse=True)
print(f"{'ms(2 passes)':>12} {'fetchGB':>8} {'writeGB':>8} {'~TB/s':>6}  kernel")
for dur, k, fe, wr, bw in rows[:18]:
    print(f"{dur:12.2f} {fe:8.2f} {wr:8.2f} {bw:6.2f}  {k}")
PYEOF
