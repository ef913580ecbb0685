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

# fp4 scaled path
rng4 = np.random.default_rng(44)
a = (rng4.standard_normal((32, 256)) * np.exp2(rng4.integers(-6, 7, (32, 256)))).astype(np.float32)
b = (rng4.standard_normal((256, 32)) * np.exp2(rng4.integers(-6, 7, (256, 32)))).astype(np.float32)
d = probe.mfma_fp4_scaled_tile(a, b)
a4, sa = probe.mx_quantize_fp4(a)
b4t, sb = probe.mx_quantize_fp4(np.ascontiguousarray(b.T))
ref = (probe.mx_dequantize_fp4(a4, sa).astype(np.float64)
       @ probe.mx_dequantize_fp4(b4t, sb).astype(np.float64).T)
err = np.abs(d - ref).max() / np.abs(ref).max()
print(f"fp4 scaled tile rel_err = {err:.3e}", "OK" if err < 1e-4 else "FAIL")
M = N = K = 512
a = (rng4.standard_normal((M, K)) * np.exp2(rng4.integers(-6, 7, (M, K)))).astype(np.float32)
bt = (rng4.standard_normal((N, K)) * np.exp2(rng4.integers(-6, 7, (N, K)))).astype(np.float32)
c, a4, sa, b4t, sbt = probe.gemm_fp4_scaled(a, bt)
ref = (probe.mx_dequantize_fp4(a4, sa).astype(np.float64)
       @ probe.mx_dequantize_fp4(b4t, sbt).astype(np.float64).T)
err = np.abs(c - ref).max() / np.abs(ref).max()
print(f"fp4 scaled GEMM rel_err = {err:.3e}", "OK" if err < 1e-4 else "FAIL")
for size in (4096, 8192):
    print(f"fp4 scaled @{size}^3: {probe.gemm_fp4_scaled_tflops(size=size, iters=10):.0f} TF")
for size in (4096, 8192):
    print(f"fp4 unscaled v446 @{size}^3: {probe.gemm_fp8_tflops_ex(size=size, iters=10, variant=446):.0f} TF")
