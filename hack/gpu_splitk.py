"""Split-K bf16 GEMM: numerics + small-shape throughput ladder."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from k8s_dra_driver_gpu_amd.fabric import probe

rng = np.random.default_rng(7)
M = N = K = 1024
a = rng.standard_normal((M, K)).astype(np.float32)
bt = rng.standard_normal((N, K)).astype(np.float32)
ref = (probe.bf16_truncate(a).astype(np.float64)
       @ probe.bf16_truncate(bt).astype(np.float64).T)
for ks in (1, 4, 8):
    c = probe.gemm_bf16_splitk(a, bt, ksplit=ks)
    err = np.abs(c - ref).max() / np.abs(ref).max()
    print(f"splitk={ks} rel_err = {err:.3e}", "OK" if err < 1e-4 else "FAIL")
for size in (1024, 2048, 4096):
    base = probe.gemm_bf16_tflops_ex(0, size, 10, 432)
    line = f"@{size}^3: plain432={base:.0f}"
    for ks in (2, 4, 8):
        if size % 128 == 0 and (size // ks) % 32 == 0:
            line += f" ks{ks}={probe.gemm_bf16_splitk_tflops(0, size, 10, ks):.0f}"
    big = probe.gemm_bf16_tflops_ex(0, size, 10, 852)
    print(line + f" big852={big:.0f} TF")

# tall-skinny: the actual split-K regime (16 output tiles, K=32768)
for (M2, N2, K2) in ((512, 512, 32768), (1024, 1024, 16384)):
    line = f"M={M2} N={N2} K={K2}:"
    for ks in (1, 4, 8, 16):
        tf = probe.gemm_bf16_splitk_tflops_mnk(M2, N2, K2, iters=10, ksplit=ks)
        line += f" ks{ks}={tf:.0f}"
    print(line + " TF")
