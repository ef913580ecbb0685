"""Pin the mfma_scale_f32_16x16x128_f8f6f4 A/B fragment layout empirically:
try each hypothesis, compare against dequantized-fp8 matmul reference."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from k8s_dra_driver_gpu_amd.fabric import probe

rng = np.random.default_rng(3)
K = 256
a = rng.standard_normal((16, K)).astype(np.float32)
b = rng.standard_normal((K, 16)).astype(np.float32)
ref = (probe.fp8_e4m3_to_f32(probe.to_fp8_e4m3(a)).astype(np.float64)
       @ probe.fp8_e4m3_to_f32(probe.to_fp8_e4m3(b)).astype(np.float64))
for layout in (0, 1, 2):
    try:
        d = probe.mfma_fp8_tile_gemm(a, b, layout=layout)
        err = np.abs(d - ref).max() / np.abs(ref).max()
        print(f"layout={layout}: rel_err={err:.3e}", "<<< MATCH" if err < 1e-2 else "")
    except Exception as e:
        print(f"layout={layout}: EXC {e}")
# identity check with asymmetric B for the matching layout (guide's advice)
a2 = np.zeros((16, 128), dtype=np.float32); np.fill_diagonal(a2[:, :16], 1.0)
b2 = rng.standard_normal((128, 16)).astype(np.float32)
for layout in (0, 1, 2):
    d = probe.mfma_fp8_tile_gemm(a2, b2, layout=layout)
    ref2 = probe.fp8_e4m3_to_f32(probe.to_fp8_e4m3(b2[:16]))
    err = np.abs(d - ref2).max()
    print(f"identity layout={layout}: max_err={err:.3e}")
