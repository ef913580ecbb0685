"""Decode the mfma_scale_f32_16x16x128_f8f6f4 scale-operand lane layout.

Probe data: A = all 1.0 (e4m3), B[k][col] = 2^-(k/16) (chunk ch weight
16*2^-ch per cell). Probe L<64 doubles SA lane L byte0; L in [64,128)
doubles SB lane L-64 byte0; probe 128 is the unperturbed baseline.
A doubled scale covering (row i, col j, chunk ch) adds 16*2^-ch to D[i][j].
"""
import ctypes, numpy as np
lib = ctypes.CDLL("k8s_dra_driver_gpu_amd/_libfabricprobe.so")
lib.fp_mfma_scale_probe_host.restype = ctypes.c_int
lib.fp_mfma_scale_probe_host.argtypes = [ctypes.c_int, ctypes.POINTER(ctypes.c_float)]
D = np.zeros((129, 16, 16), dtype=np.float32)
rc = lib.fp_mfma_scale_probe_host(0, D.ctypes.data_as(ctypes.POINTER(ctypes.c_float)))
print("rc=", rc)
base = D[128]
print("baseline uniques:", np.unique(np.round(base, 4)))
for L in range(128):
    delta = D[L] - base
    hot = np.argwhere(np.abs(delta) > 0.06)
    who = "SA" if L < 64 else "SB"
    lane = L % 64
    if not len(hot):
        print(f"{who} lane={lane:2d}: no effect")
        continue
    cells = {}
    for (i, j) in hot:
        d = float(delta[i, j])
        chs = []
        for ch in range(8):
            w = 16.0 * 2.0 ** -ch
            if d >= w - 0.06:
                d -= w
                chs.append(ch)
        cells[(int(i), int(j))] = tuple(chs)
    rows = sorted(set(i for i, _ in cells))
    cols = sorted(set(j for _, j in cells))
    chsets = sorted({v for v in cells.values()})
    print(f"{who} lane={lane:2d}: rows={rows if len(rows)<16 else 'ALL'} "
          f"cols={cols if len(cols)<16 else 'ALL'} chunks={chsets}")
