"""fp8 GEMM numerics + throughput ladder."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from k8s_dra_driver_gpu_amd.fabric import probe

rng = np.random.default_rng(5)
M, N, K = 512, 384, 256
a = rng.standard_normal((M, K)).astype(np.float32)
bt = rng.standard_normal((N, K)).astype(np.float32)
ref = (probe.fp8_e4m3_to_f32(probe.to_fp8_e4m3(a)).astype(np.float64)
       @ probe.fp8_e4m3_to_f32(probe.to_fp8_e4m3(bt)).astype(np.float64).T)
for v in (1, 2):
    d = probe.gemm_fp8(a, bt, variant=v)
    err = np.abs(d - ref).max() / np.abs(ref).max()
    print(f"variant={v} rel_err={err:.3e}", "OK" if err < 1e-2 else "FAIL")
    sys.stdout.flush()

for size, iters in ((4096, 10), (8192, 5)):
    for v in (1, 2):
        tf = probe.gemm_fp8_tflops_ex(0, size, iters, v)
        print(f"fp8 size={size} v={v}: {tf:.0f} TF"); sys.stdout.flush()
