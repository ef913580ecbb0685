"""Verify REAL MX block scales through mfma_scale (per-lane scale byte at
OPSEL 0): compare vs the dequantized MX reference."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from k8s_dra_driver_gpu_amd.fabric import probe

rng = np.random.default_rng(21)
K = 256
# wide dynamic range so wrong scale mapping produces huge errors
a = (rng.standard_normal((16, K)) * np.exp2(rng.integers(-8, 9, (16, K)))).astype(np.float32)
b = (rng.standard_normal((K, 16)) * np.exp2(rng.integers(-8, 9, (K, 16)))).astype(np.float32)
a8, sa = probe.mx_quantize_fp8(a)
b8t, sb = probe.mx_quantize_fp8(np.ascontiguousarray(b.T))
ref = (probe.mx_dequantize_fp8(a8, sa).astype(np.float64)
       @ probe.mx_dequantize_fp8(b8t, sb).astype(np.float64).T)
d = probe.mfma_fp8_scaled_tile(a, b)
err = np.abs(d - ref).max() / np.abs(ref).max()
print(f"MX-scaled fp8 tile rel_err = {err:.3e}", "OK" if err < 1e-4 else "FAIL")
# degenerate: all scales 1.0 must equal the unscaled path
a2 = rng.standard_normal((16, 128)).astype(np.float32)
b2 = rng.standard_normal((128, 16)).astype(np.float32)
d1 = probe.mfma_fp8_scaled_tile(a2 * 448, b2)  # big absmax -> nontrivial scales
a28, sa2 = probe.mx_quantize_fp8(a2 * 448)
b28t, sb2 = probe.mx_quantize_fp8(np.ascontiguousarray(b2.T))
ref2 = (probe.mx_dequantize_fp8(a28, sa2).astype(np.float64)
        @ probe.mx_dequantize_fp8(b28t, sb2).astype(np.float64).T)
err2 = np.abs(d1 - ref2).max() / np.abs(ref2).max()
print(f"MX-scaled (nontrivial scales) rel_err = {err2:.3e}", "OK" if err2 < 1e-4 else "FAIL")
