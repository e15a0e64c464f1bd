#!/bin/bash
# PMC counter capture + IN-BOX aggregation (results dbs are >64MB, so
# only the per-kernel summary goes back in gpurun_out).
set -x
cd /root/repo
rm -rf gpurun_out
mkdir -p gpurun_out/pmc
export TMPDIR=/tmp
cd /tmp
timeout 500 rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_BUSY_CYCLES SQ_WAVE_CYCLES SQ_LDS_BANK_CONFLICT \
    -d /tmp/pmc -o pass1 -- \
    python /root/repo/bench.py --steps 2 --warmup 1 --batch-size 32 \
    > /root/repo/gpurun_out/pmc/pass1.log 2>&1
echo "PASS1 EXIT: $?"
timeout 500 rocprofv3 --pmc FETCH_SIZE WRITE_SIZE \
    -d /tmp/pmc -o pass2 -- \
    python /root/repo/bench.py --steps 2 --warmup 1 --batch-size 32 \
    > /root/repo/gpurun_out/pmc/pass2.log 2>&1
echo "PASS2 EXIT: $?"
python - <<'EOF'
import sqlite3, glob, re, json
out = {}
for db_path in glob.glob('/tmp/pmc/*_results.db'):
    tag = 'pass1' if 'pass1' in db_path else 'pass2'
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    kd = [t for t in tables if t.startswith('rocpd_kernel_dispatch')][0]
    sfx = kd.replace('rocpd_kernel_dispatch_','')
    # counter values live in rocpd_pmc_event linked to dispatch events
    pe = [t for t in tables if t.startswith('rocpd_pmc_event')][0]
    pi = [t for t in tables if t.startswith('rocpd_info_pmc')][0]
    cols_pe = [c[1] for c in cur.execute(f'PRAGMA table_info({pe})')]
    cols_pi = [c[1] for c in cur.execute(f'PRAGMA table_info({pi})')]
    print(tag, 'pe cols:', cols_pe, 'pi cols:', cols_pi)
    q = f"""
    SELECT ks.display_name, pi.name, SUM(p.value), COUNT(*)
    FROM {pe} p
    JOIN {pi} pi ON p.pmc_id = pi.id
    JOIN {kd} k ON p.event_id = k.event_id
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
    GROUP BY ks.display_name, pi.name
    """
    try:
        rows = cur.execute(q).fetchall()
    except Exception as e:
        print('query failed:', e)
        continue
    for name, cname, val, n in rows:
        key = re.sub(r'<[^>]*>','',str(name)).split('(')[0][:60]
        out.setdefault(key, {})[str(cname)] = val
        out[key]['dispatches'] = n
    # also kernel durations
    q2 = f"SELECT ks.display_name, SUM(k.end-k.start) FROM {kd} k JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id=ks.id GROUP BY 1"
    for name, dur in cur.execute(q2):
        key = re.sub(r'<[^>]*>','',str(name)).split('(')[0][:60]
        out.setdefault(key, {})[f'dur_ns_{tag}'] = dur
json.dump(out, open('/root/repo/gpurun_out/pmc/summary.json','w'), indent=1)
print("summary written,", len(out), "kernels")
EOF
echo "AGG EXIT: $?"
du -sh /root/repo/gpurun_out
