import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from k8s_dra_driver_gpu_amd.fabric import probe

rng = np.random.default_rng(9)
K = 128
a = rng.standard_normal((32, K)).astype(np.float32)
b = rng.standard_normal((K, 32)).astype(np.float32)
ref = (probe.fp4_e2m1_to_f32(probe.to_fp4_e2m1(a)).astype(np.float64)
       @ probe.fp4_e2m1_to_f32(probe.to_fp4_e2m1(b)).astype(np.float64))
d = probe.mfma_fp4_tile_gemm(a, b)
err = np.abs(d - ref).max() / max(1e-6, np.abs(ref).max())
print(f"fp4 tile rel_err={err:.3e}", "OK" if err < 1e-3 else "FAIL")
# asymmetric identity
a2 = np.zeros((32, 64), dtype=np.float32); np.fill_diagonal(a2[:, :32], 1.0)
b2 = rng.standard_normal((64, 32)).astype(np.float32)
d2 = probe.mfma_fp4_tile_gemm(a2, b2)
ref2 = probe.fp4_e2m1_to_f32(probe.to_fp4_e2m1(b2[:32]))
print("fp4 identity max_err:", np.abs(d2 - ref2).max())
print(f"mfma_fp4 ceiling: {probe.mfma_fp4_tflops(0, 2048, 10):.0f} TF")
