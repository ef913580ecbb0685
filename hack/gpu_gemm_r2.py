"""Round-2 GEMM ladder: register-hoisted fragments (632) and 256x128 tile
(732) vs the depth-2 champion (432). Numerics vs torch fp32 first."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from k8s_dra_driver_gpu_amd.fabric import probe

# numerics (non-square, multiples of the largest tile)
rng = np.random.default_rng(7)
M, N, K = 512, 384, 256
a = rng.standard_normal((M, K), dtype=np.float32)
bt = rng.standard_normal((N, K), dtype=np.float32)
ref = probe.bf16_truncate(a) @ probe.bf16_truncate(bt).T
for bk in (432, 732, 764, 832):
    d = probe.gemm_bf16(a, bt, dev=0, bk=bk)
    err = np.abs(d - ref).max() / max(1e-6, np.abs(ref).max())
    print(f"bk={bk} rel_err={err:.2e}", "OK" if err < 1e-2 else "FAIL")

# perf ladder
for size, iters in ((4096, 10), (8192, 5)):
    for bk in (432, 732, 764, 832):
        tf = probe.gemm_bf16_tflops_ex(0, size, iters, bk)
        print(f"size={size} bk={bk}: {tf:.0f} TF")
        sys.stdout.flush()
