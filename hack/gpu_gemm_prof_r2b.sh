#!/bin/bash
set -x
cd /tmp && export TMPDIR=/tmp
OUT=/root/repo/gpurun_out/r2s11
mkdir -p "$OUT"
cd /root/repo
timeout 420 python -c "import __graft_entry__ as g; g.build()" >/dev/null 2>&1
cd /tmp
timeout 300 rocprofv3 --kernel-trace --stats -d /tmp/prof842 -o s842 -- \
  python -c "import sys; sys.path.insert(0,'/root/repo'); from k8s_dra_driver_gpu_amd.fabric import probe; print('TF', probe.gemm_bf16_tflops_ex(0, 8192, 5, 842))" > "$OUT/run842.log" 2>&1
grep "^TF" "$OUT/run842.log"
find /tmp/prof842 -type f | tee "$OUT/prof_files.txt"
timeout 300 rocprofv3 --pmc SQ_INSTS_MFMA,SQ_INSTS_VALU,SQ_WAVES,SQ_BUSY_CYCLES -d /tmp/pmc842 -o p842 -- \
  python -c "import sys; sys.path.insert(0,'/root/repo'); from k8s_dra_driver_gpu_amd.fabric import probe; print('TF', probe.gemm_bf16_tflops_ex(0, 8192, 3, 842))" > "$OUT/pmc842.log" 2>&1
grep "^TF" "$OUT/pmc842.log"
find /tmp/pmc842 -type f >> "$OUT/prof_files.txt"
# extract kernel stats from whatever rocprofv3 produced
timeout 120 python - > "$OUT/kernel_stats.txt" 2>&1 <<'PY'
import glob, sqlite3, csv, os
for db in glob.glob('/tmp/prof842/**/*.db', recursive=True) + glob.glob('/tmp/pmc842/**/*.db', recursive=True):
    print('==', db)
    c = sqlite3.connect(db)
    tables = [r[0] for r in c.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    for t in tables:
        if any(k in t.lower() for k in ('kernel', 'counter', 'stat')):
            print('--- table', t)
            try:
                rows = list(c.execute(f"SELECT * FROM {t} LIMIT 6"))
                cols = [d[0] for d in c.execute(f"SELECT * FROM {t} LIMIT 1").description]
                print(cols)
                for r in rows: print(r)
            except Exception as e:
                print('ERR', e)
for f in glob.glob('/tmp/prof842/**/*.csv', recursive=True) + glob.glob('/tmp/pmc842/**/*.csv', recursive=True):
    print('== CSV', f)
    print(open(f).read()[:2000])
PY
head -80 "$OUT/kernel_stats.txt"
