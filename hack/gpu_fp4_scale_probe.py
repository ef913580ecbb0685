"""Decode the mfma_scale_f32_32x32x64_f8f6f4 (fp4) scale lane layout."""
import ctypes, numpy as np
lib = ctypes.CDLL("k8s_dra_driver_gpu_amd/_libfabricprobe.so")
lib.fp_mfma_fp4_scale_probe_host.restype = ctypes.c_int
lib.fp_mfma_fp4_scale_probe_host.argtypes = [ctypes.c_int, ctypes.POINTER(ctypes.c_float)]
D = np.zeros((129, 32, 32), dtype=np.float32)
rc = lib.fp_mfma_fp4_scale_probe_host(0, D.ctypes.data_as(ctypes.POINTER(ctypes.c_float)))
print("rc=", rc)
base = D[128]
print("baseline uniques:", np.unique(np.round(base, 3)))
WS = [64.0, 32.0, 16.0, 8.0]   # delta of doubled 16-chunk 0..3
for L in range(128):
    delta = D[L] - base
    hot = np.argwhere(np.abs(delta) > 2.0)
    who = "SA" if L < 64 else "SB"
    lane = L % 64
    if not len(hot):
        print(f"{who} lane={lane:2d}: no effect")
        continue
    cells = {}
    for (i, j) in hot:
        d = float(delta[i, j])
        chs = []
        for ch in range(4):
            if d >= WS[ch] - 2.0:
                d -= WS[ch]
                chs.append(ch)
        cells[(int(i), int(j))] = tuple(chs)
    rows = sorted(set(i for i, _ in cells))
    cols = sorted(set(j for _, j in cells))
    chsets = sorted({v for v in cells.values()})
    print(f"{who} lane={lane:2d}: rows={rows if len(rows)<32 else 'ALL'} "
          f"cols={cols if len(cols)<32 else 'ALL'} chunks={chsets}")
