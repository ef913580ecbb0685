"""fp8 ceiling + 256x256 variants + G32."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from k8s_dra_driver_gpu_amd.fabric import probe

print("mfma_fp8 ceiling:", f"{probe.mfma_fp8_tflops(0, 2048, 10):.0f} TF"); sys.stdout.flush()
print("mfma_bf16 ceiling:", f"{probe.mfma_bf16_tflops(0, 2048, 10):.0f} TF"); sys.stdout.flush()

rng = np.random.default_rng(5)
M, N, K = 512, 512, 256
a = rng.standard_normal((M, K)).astype(np.float32)
bt = rng.standard_normal((N, K)).astype(np.float32)
ref8 = (probe.fp8_e4m3_to_f32(probe.to_fp8_e4m3(a)).astype(np.float64)
        @ probe.fp8_e4m3_to_f32(probe.to_fp8_e4m3(bt)).astype(np.float64).T)
for v in (3, 316, 232):
    d = probe.gemm_fp8(a, bt, variant=v)
    err = np.abs(d - ref8).max() / np.abs(ref8).max()
    print(f"fp8 v={v} rel_err={err:.2e}", "OK" if err < 1e-2 else "FAIL")
sys.stdout.flush()
for size, iters in ((4096, 10), (8192, 5)):
    for v in (216, 232, 3, 316):
        tf = probe.gemm_fp8_tflops_ex(0, size, iters, v)
        print(f"fp8 size={size} v={v}: {tf:.0f} TF"); sys.stdout.flush()
