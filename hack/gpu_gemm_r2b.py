"""Ladder round 2b: 256x256 depth-2 vs depth-1; torch(hipBLASLt) reference."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from k8s_dra_driver_gpu_amd.fabric import probe

rng = np.random.default_rng(7)
M, N, K = 512, 512, 256
a = rng.standard_normal((M, K), dtype=np.float32)
bt = rng.standard_normal((N, K), dtype=np.float32)
ref = probe.bf16_truncate(a) @ probe.bf16_truncate(bt).T
for bk in (832, 842):
    d = probe.gemm_bf16(a, bt, dev=0, bk=bk)
    err = np.abs(d - ref).max() / max(1e-6, np.abs(ref).max())
    print(f"bk={bk} rel_err={err:.2e}", "OK" if err < 1e-2 else "FAIL")

for size, iters in ((4096, 10), (8192, 5)):
    for bk in (732, 832, 842):
        tf = probe.gemm_bf16_tflops_ex(0, size, iters, bk)
        print(f"size={size} bk={bk}: {tf:.0f} TF"); sys.stdout.flush()

# library reference: torch bf16 matmul (hipBLASLt under the hood)
import torch
for size in (4096, 8192):
    x = torch.randn(size, size, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(size, size, dtype=torch.bfloat16, device="cuda")
    for _ in range(3):
        y = x @ w
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    it = 10 if size == 4096 else 5
    for _ in range(it):
        y = x @ w
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / it
    print(f"torch bf16 {size}^3: {2*size**3/dt/1e12:.0f} TF")
