#!/bin/bash
set -x
cd /tmp && export TMPDIR=/tmp
OUT=/root/repo/gpurun_out/r2s25
mkdir -p "$OUT"
cd /root/repo
timeout 420 python -c "import __graft_entry__ as g; g.build()" >/dev/null 2>&1

# 1. full GPU tier
timeout 600 python -m pytest tests -m gpu -q 2>&1 | tail -3 | tee "$OUT/gpu_tier.txt"

# 2. torch fp8 scaled-mm library reference (own process)
timeout 300 python - > "$OUT/torch_fp8.txt" 2>&1 <<'PY'
import torch, time
try:
    size = 8192
    x = torch.randn(size, size, device="cuda").to(torch.float8_e4m3fn)
    w = torch.randn(size, size, device="cuda").to(torch.float8_e4m3fn).t()
    sx = torch.tensor(1.0, device="cuda")
    for _ in range(3):
        y = torch._scaled_mm(x, w, scale_a=sx, scale_b=sx, out_dtype=torch.bfloat16)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(8):
        y = torch._scaled_mm(x, w, scale_a=sx, scale_b=sx, out_dtype=torch.bfloat16)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 8
    print(f"torch _scaled_mm fp8 {size}^3: {2*size**3/dt/1e12:.0f} TF")
except Exception as e:
    print("torch fp8 unavailable:", type(e).__name__, e)
PY
cat "$OUT/torch_fp8.txt"

# 3. PMC on the fp4 champion: MFMA busy vs total cycles
cd /tmp
timeout 300 rocprofv3 --pmc SQ_INSTS_MFMA,SQ_BUSY_CYCLES,GRBM_COUNT,SQ_WAVES -d /tmp/pmc446 -o p446 -- \
  python -c "import sys; sys.path.insert(0,'/root/repo'); from k8s_dra_driver_gpu_amd.fabric import probe; print('TF', probe.gemm_fp8_tflops_ex(0, 8192, 3, 446))" > "$OUT/pmc446.log" 2>&1
grep "^TF" "$OUT/pmc446.log"
timeout 120 python - > "$OUT/pmc446_counters.txt" 2>&1 <<'PY'
import glob, sqlite3
for db in glob.glob('/tmp/pmc446/**/*.db', recursive=True):
    c = sqlite3.connect(db)
    tables = [r[0] for r in c.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    for t in tables:
        if 'counter' in t.lower() or 'pmc' in t.lower():
            print('--- table', t)
            try:
                cols = [d[0] for d in c.execute(f"SELECT * FROM {t} LIMIT 1").description]
                print(cols)
                for r in c.execute(f"SELECT * FROM {t} LIMIT 12"):
                    print(r)
            except Exception as e:
                print('ERR', e)
PY
head -40 "$OUT/pmc446_counters.txt"
