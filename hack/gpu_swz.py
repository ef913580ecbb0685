import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from k8s_dra_driver_gpu_amd.fabric import probe

rng = np.random.default_rng(13)
M, N, K = 512, 512, 512
a = rng.standard_normal((M, K)).astype(np.float32)
bt = rng.standard_normal((N, K)).astype(np.float32)
refb = probe.bf16_truncate(a) @ probe.bf16_truncate(bt).T
d = probe.gemm_bf16(a, bt, bk=852)
print("bf16 swz rel_err:", np.abs(d - refb).max() / np.abs(refb).max())
ref8 = (probe.fp8_e4m3_to_f32(probe.to_fp8_e4m3(a)).astype(np.float64)
        @ probe.fp8_e4m3_to_f32(probe.to_fp8_e4m3(bt)).astype(np.float64).T)
d = probe.gemm_fp8(a, bt, variant=326)
print("fp8 swz rel_err:", np.abs(d - ref8).max() / np.abs(ref8).max())
ref4 = (probe.fp4_e2m1_to_f32(probe.to_fp4_e2m1(a)).astype(np.float64)
        @ probe.fp4_e2m1_to_f32(probe.to_fp4_e2m1(bt)).astype(np.float64).T)
d = probe.gemm_fp4(a, bt, variant=446)
print("fp4 swz rel_err:", np.abs(d - ref4).max() / np.abs(ref4).max())
sys.stdout.flush()
for size, iters in ((8192, 5), (4096, 10)):
    for name, fn, pairs in (("bf16", probe.gemm_bf16_tflops_ex, (842, 852)),
                            ("fp8", probe.gemm_fp8_tflops_ex, (316, 326)),
                            ("fp4", probe.gemm_fp8_tflops_ex, (436, 446))):
        for v in pairs:
            print(f"{name} size={size} v={v}: {fn(0, size, iters, v):.0f} TF"); sys.stdout.flush()
