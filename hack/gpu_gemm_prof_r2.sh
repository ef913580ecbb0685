#!/bin/bash
# rocprofv3 evidence for the round-2 GEMM champion (bk=842) + torch library
# reference in a separate process (torch after ctypes-HIP breaks init).
set -x
cd /tmp && export TMPDIR=/tmp
OUT=/root/repo/gpurun_out/r2s10
mkdir -p "$OUT"
cd /root/repo
timeout 420 python -c "import __graft_entry__ as g; g.build()" >/dev/null 2>&1

# torch bf16 matmul (hipBLASLt) reference, own process
timeout 300 python - > "$OUT/torch_ref.txt" 2>&1 <<'PY'
import torch, time
for size, it in ((4096, 20), (8192, 8)):
    x = torch.randn(size, size, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(size, size, dtype=torch.bfloat16, device="cuda")
    for _ in range(3): y = x @ w
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(it): y = x @ w
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / it
    print(f"torch bf16 {size}^3: {2*size**3/dt/1e12:.0f} TF")
PY
cat "$OUT/torch_ref.txt"

# kernel-trace stats for the champion
cd /tmp
timeout 300 rocprofv3 --kernel-trace --stats -d /tmp/prof842 -o s842 -- \
  python -c "import sys; sys.path.insert(0,'/root/repo'); from k8s_dra_driver_gpu_amd.fabric import probe; print(probe.gemm_bf16_tflops_ex(0, 8192, 5, 842))" > "$OUT/run842.log" 2>&1
tail -3 "$OUT/run842.log"
find /tmp/prof842 -name "*stats*" -exec cp {} "$OUT/" \;

# PMC counters in their own run (no trace domains)
timeout 300 rocprofv3 --pmc SQ_INSTS_MFMA,SQ_INSTS_VALU,SQ_WAVES -d /tmp/pmc842 -o p842 -- \
  python -c "import sys; sys.path.insert(0,'/root/repo'); from k8s_dra_driver_gpu_amd.fabric import probe; print(probe.gemm_bf16_tflops_ex(0, 8192, 3, 842))" > "$OUT/pmc842.log" 2>&1
tail -3 "$OUT/pmc842.log"
find /tmp/pmc842 -name "*.csv" | head -3
for f in $(find /tmp/pmc842 -name "*counter*" -o -name "*.csv" | head -4); do cp "$f" "$OUT/"; done
ls "$OUT"
