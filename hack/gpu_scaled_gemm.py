"""MX-scaled fp8 GEMM: numerics vs dequantized reference + throughput."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from k8s_dra_driver_gpu_amd.fabric import probe

rng = np.random.default_rng(33)
M = N = K = 512
a = (rng.standard_normal((M, K)) * np.exp2(rng.integers(-6, 7, (M, K)))).astype(np.float32)
bt = (rng.standard_normal((N, K)) * np.exp2(rng.integers(-6, 7, (N, K)))).astype(np.float32)
for v in (526, 546, 556):
    c, a8, sa, b8t, sbt = probe.gemm_fp8_scaled(a, bt, variant=v)
    ref = (probe.mx_dequantize_fp8(a8, sa).astype(np.float64)
           @ probe.mx_dequantize_fp8(b8t, sbt).astype(np.float64).T)
    err = np.abs(c - ref).max() / np.abs(ref).max()
    print(f"variant {v}: rel_err = {err:.3e}", "OK" if err < 1e-4 else "FAIL")
for v in (526, 529, 556):
    for size in (4096, 8192):
        tf = probe.gemm_fp8_scaled_tflops(size=size, iters=10, variant=v)
        print(f"variant {v} @{size}^3: {tf:.0f} TF")
# unscaled champion for comparison on the same box
for size in (4096, 8192):
    tf = probe.gemm_fp8_tflops_ex(size=size, iters=10, variant=346)
    print(f"unscaled v346 @{size}^3: {tf:.0f} TF")
