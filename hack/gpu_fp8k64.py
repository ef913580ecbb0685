import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from k8s_dra_driver_gpu_amd.fabric import probe

rng = np.random.default_rng(11)
M, N, K = 512, 512, 512
a = rng.standard_normal((M, K)).astype(np.float32)
bt = rng.standard_normal((N, K)).astype(np.float32)
ref = (probe.fp8_e4m3_to_f32(probe.to_fp8_e4m3(a)).astype(np.float64)
       @ probe.fp8_e4m3_to_f32(probe.to_fp8_e4m3(bt)).astype(np.float64).T)
d = probe.gemm_fp8(a, bt, variant=336)
err = np.abs(d - ref).max() / np.abs(ref).max()
print(f"fp8 v=336 rel_err={err:.3e}", "OK" if err < 1e-2 else "FAIL")
sys.stdout.flush()
for size, iters in ((4096, 10), (8192, 5)):
    for v in (316, 336):
        tf = probe.gemm_fp8_tflops_ex(0, size, iters, v)
        print(f"fp8 size={size} v={v}: {tf:.0f} TF"); sys.stdout.flush()
    tf = probe.gemm_fp8_tflops_ex(0, size, iters, 436)
    print(f"fp4 size={size} v=436: {tf:.0f} TF"); sys.stdout.flush()
