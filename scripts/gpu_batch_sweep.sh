#!/bin/bash
# Validate bwd_weight16 + batch-size sweep + B=64 wall-gap profile.
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 600 python -m pytest tests/test_kernels_gpu.py -q -m gpu -x --timeout 300 \
    > gpurun_out/gpu_tests_bw16.log 2>&1
echo "PYTEST EXIT: $?"; tail -3 gpurun_out/gpu_tests_bw16.log
for B in 64 96 128; do
  timeout 500 python bench.py --steps 12 --warmup 4 --batch-size $B \
      > gpurun_out/bench_B$B.log 2>&1
  echo "B$B: $(tail -1 gpurun_out/bench_B$B.log | head -c 220)"
done
export TMPDIR=/tmp
cd /tmp
timeout 500 rocprofv3 --kernel-trace --stats -d /tmp/prof -o rb64 -- \
    python /root/repo/bench.py --steps 5 --warmup 2 --batch-size 64 \
    > /root/repo/gpurun_out/profB64.log 2>&1
python - <<'EOF'
import sqlite3, glob
db = sqlite3.connect(glob.glob('/tmp/prof/rb64_results.db')[0])
cur = db.cursor()
t = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table' AND name LIKE 'rocpd_kernel_dispatch%'")][0]
rows = cur.execute(f"SELECT start, end FROM {t} ORDER BY start").fetchall()
import numpy as np
starts = np.array([r[0] for r in rows]); ends = np.array([r[1] for r in rows])
tot = (ends - starts).sum()
# steady state: drop first half (init+warmup)
t0 = starts.min() + 0.5*(ends.max()-starts.min())
m = starts > t0
s2, e2 = starts[m], ends[m]
span = e2.max() - s2.min()
busy = 0; cs, ce = s2[0], e2[0]
for a, b in zip(s2[1:], e2[1:]):
    if a > ce: busy += ce - cs; cs, ce = a, b
    else: ce = max(ce, b)
busy += ce - cs
print(f"B=64 steady: span {span/1e6:.1f} ms, busy {busy/1e6:.1f} ms, gap {(span-busy)/span*100:.1f}%")
open('/root/repo/gpurun_out/gap_b64.txt','w').write(f"span {span/1e6:.1f} busy {busy/1e6:.1f} gap {(span-busy)/span*100:.1f}%\n")
EOF
echo "GAP EXIT: $?"
cat /root/repo/gpurun_out/gap_b64.txt
