import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from k8s_dra_driver_gpu_amd.fabric import probe

rng = np.random.default_rng(11)
M, N, K = 512, 512, 512
a = rng.standard_normal((M, K)).astype(np.float32)
bt = rng.standard_normal((N, K)).astype(np.float32)
ref = (probe.fp4_e2m1_to_f32(probe.to_fp4_e2m1(a)).astype(np.float64)
       @ probe.fp4_e2m1_to_f32(probe.to_fp4_e2m1(bt)).astype(np.float64).T)
for v in (4, 416, 436):
    d = probe.gemm_fp4(a, bt, variant=v)
    err = np.abs(d - ref).max() / np.abs(ref).max()
    print(f"fp4 v={v} rel_err={err:.3e}", "OK" if err < 1e-3 else "FAIL")
sys.stdout.flush()
for size, iters in ((4096, 10), (8192, 5)):
    for v in (4, 416, 436):
        tf = probe.gemm_fp8_tflops_ex(0, size, iters, v)
        print(f"fp4 size={size} v={v}: {tf:.0f} TF"); sys.stdout.flush()
