"""Python bindings for the CDNA4 fabric/device probe library.

Loads the in-tree ``_libfabricprobe.so`` (HIP, gfx950) via ctypes.  On a GPU
box the library MUST be present — probes raise loudly rather than falling
back to anything non-native (round-end GPU checks verify the native path is
the one that runs).
"""

from __future__ import annotations

import ctypes
import os
from dataclasses import dataclass
from typing import List, Optional

import numpy as np

_SO = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "_libfabricprobe.so")
_lib: Optional[ctypes.CDLL] = None


class ProbeError(RuntimeError):
    pass


def _load() -> ctypes.CDLL:
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(_SO):
        raise ProbeError(
            f"fabric probe library not built: {_SO} missing — run "
            "`python -m k8s_dra_driver_gpu_amd.ops.build`"
        )
    lib = ctypes.CDLL(_SO)
    lib.fp_device_count.restype = ctypes.c_int
    lib.fp_hbm_read_gbps.restype = ctypes.c_double
    lib.fp_hbm_read_gbps.argtypes = [ctypes.c_int, ctypes.c_size_t, ctypes.c_int]
    lib.fp_hbm_write_gbps.restype = ctypes.c_double
    lib.fp_hbm_write_gbps.argtypes = [ctypes.c_int, ctypes.c_size_t, ctypes.c_int]
    lib.fp_hbm_copy_gbps.restype = ctypes.c_double
    lib.fp_hbm_copy_gbps.argtypes = [ctypes.c_int, ctypes.c_size_t, ctypes.c_int]
    lib.fp_mfma_bf16_tflops.restype = ctypes.c_double
    lib.fp_mfma_bf16_tflops.argtypes = [ctypes.c_int, ctypes.c_int, ctypes.c_int]
    lib.fp_mfma_fp8_tflops.restype = ctypes.c_double
    lib.fp_mfma_fp8_tflops.argtypes = [ctypes.c_int, ctypes.c_int, ctypes.c_int]
    lib.fp_mfma_fp4_tflops.restype = ctypes.c_double
    lib.fp_mfma_fp4_tflops.argtypes = [ctypes.c_int, ctypes.c_int, ctypes.c_int]
    lib.fp_mfma_fp8_scaled_tile_host.restype = ctypes.c_int
    lib.fp_mfma_fp8_scaled_tile_host.argtypes = [
        ctypes.c_int,
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_float),
        ctypes.c_int,
    ]
    lib.fp_mfma_fp4_scaled_tile_host.restype = ctypes.c_int
    lib.fp_mfma_fp4_scaled_tile_host.argtypes = [
        ctypes.c_int,
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_float),
        ctypes.c_int,
    ]
    lib.fp_gemm_fp4_scaled_host.restype = ctypes.c_int
    lib.fp_gemm_fp4_scaled_host.argtypes = [
        ctypes.c_int,
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_float),
        ctypes.c_int, ctypes.c_int, ctypes.c_int,
    ]
    lib.fp_gemm_fp4_scaled_tflops.restype = ctypes.c_double
    lib.fp_gemm_fp4_scaled_tflops.argtypes = [ctypes.c_int, ctypes.c_int, ctypes.c_int]
    lib.fp_mfma_fp4_tile_gemm_host.restype = ctypes.c_int
    lib.fp_mfma_fp4_tile_gemm_host.argtypes = [
        ctypes.c_int,
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_float),
        ctypes.c_int,
    ]
    lib.fp_gemm_fp8_tflops_ex.restype = ctypes.c_double
    lib.fp_gemm_fp8_tflops_ex.argtypes = [
        ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_int]
    lib.fp_gemm_fp8_host_ex.restype = ctypes.c_int
    lib.fp_gemm_fp8_host_ex.argtypes = [
        ctypes.c_int,
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_float),
        ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_int,
    ]
    lib.fp_gemm_bf16_splitk_tflops.restype = ctypes.c_double
    lib.fp_gemm_bf16_splitk_tflops.argtypes = [
        ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_int,
    ]
    lib.fp_gemm_bf16_splitk_tflops_mnk.restype = ctypes.c_double
    lib.fp_gemm_bf16_splitk_tflops_mnk.argtypes = [
        ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_int,
        ctypes.c_int, ctypes.c_int,
    ]
    lib.fp_gemm_bf16_splitk_host.restype = ctypes.c_int
    lib.fp_gemm_bf16_splitk_host.argtypes = [
        ctypes.c_int,
        ctypes.POINTER(ctypes.c_int16),
        ctypes.POINTER(ctypes.c_int16),
        ctypes.POINTER(ctypes.c_float),
        ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_int,
    ]
    lib.fp_gemm_fp8_scaled_host.restype = ctypes.c_int
    lib.fp_gemm_fp8_scaled_host.argtypes = [
        ctypes.c_int,
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_float),
        ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_int,
    ]
    lib.fp_gemm_fp8_scaled_tflops.restype = ctypes.c_double
    lib.fp_gemm_fp8_scaled_tflops.argtypes = [
        ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_int,
    ]
    lib.fp_mfma_fp8_tile_gemm_host.restype = ctypes.c_int
    lib.fp_mfma_fp8_tile_gemm_host.argtypes = [
        ctypes.c_int,
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_uint8),
        ctypes.POINTER(ctypes.c_float),
        ctypes.c_int,
        ctypes.c_int,
    ]
    lib.fp_mfma_tile_gemm_host.restype = ctypes.c_int
    lib.fp_mfma_tile_gemm_host.argtypes = [
        ctypes.c_int,
        ctypes.POINTER(ctypes.c_uint16),
        ctypes.POINTER(ctypes.c_uint16),
        ctypes.POINTER(ctypes.c_float),
        ctypes.c_int,
    ]
    lib.fp_hbm_block_sum_host.restype = ctypes.c_int
    lib.fp_hbm_block_sum_host.argtypes = [
        ctypes.c_int,
        ctypes.POINTER(ctypes.c_float),
        ctypes.c_long,
        ctypes.POINTER(ctypes.c_float),
        ctypes.c_int,
    ]
    lib.fp_p2p_read_gbps.restype = ctypes.c_double
    lib.fp_p2p_read_gbps.argtypes = [ctypes.c_int, ctypes.c_int, ctypes.c_size_t, ctypes.c_int]
    lib.fp_allreduce_pull_gbps.restype = ctypes.c_double
    lib.fp_allreduce_pull_gbps.argtypes = [ctypes.c_size_t, ctypes.c_int]
    lib.fp_gemm_bf16_tflops.restype = ctypes.c_double
    lib.fp_gemm_bf16_tflops.argtypes = [ctypes.c_int, ctypes.c_int, ctypes.c_int]
    lib.fp_gemm_bf16_tflops_ex.restype = ctypes.c_double
    lib.fp_gemm_bf16_tflops_ex.argtypes = [
        ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_int]
    lib.fp_gemm_bf16_host_ex.restype = ctypes.c_int
    lib.fp_gemm_bf16_host_ex.argtypes = [
        ctypes.c_int,
        ctypes.POINTER(ctypes.c_uint16), ctypes.POINTER(ctypes.c_uint16),
        ctypes.POINTER(ctypes.c_float),
        ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_int,
    ]
    lib.fp_gemm_bf16_host.restype = ctypes.c_int
    lib.fp_gemm_bf16_host.argtypes = [
        ctypes.c_int,
        ctypes.POINTER(ctypes.c_uint16), ctypes.POINTER(ctypes.c_uint16),
        ctypes.POINTER(ctypes.c_float),
        ctypes.c_int, ctypes.c_int, ctypes.c_int,
    ]
    lib.fp_burn.restype = ctypes.c_int
    lib.fp_burn.argtypes = [ctypes.c_int, ctypes.c_int,
                            ctypes.POINTER(ctypes.c_double),
                            ctypes.POINTER(ctypes.c_double)]
    _lib = lib
    return lib


def available() -> bool:
    return os.path.exists(_SO)


def device_count() -> int:
    return _load().fp_device_count()


def _check(v: float, what: str) -> float:
    if v < 0:
        raise ProbeError(f"{what} failed with hip error {-int(v)}")
    return v


def hbm_read_gbps(dev: int = 0, bytes_: int = 2 << 30, iters: int = 10) -> float:
    return _check(_load().fp_hbm_read_gbps(dev, bytes_, iters), "hbm_read")


def hbm_write_gbps(dev: int = 0, bytes_: int = 2 << 30, iters: int = 10) -> float:
    return _check(_load().fp_hbm_write_gbps(dev, bytes_, iters), "hbm_write")


def hbm_copy_gbps(dev: int = 0, bytes_: int = 1 << 30, iters: int = 10) -> float:
    return _check(_load().fp_hbm_copy_gbps(dev, bytes_, iters), "hbm_copy")


def mfma_bf16_tflops(dev: int = 0, inner_iters: int = 2048, launches: int = 20) -> float:
    return _check(_load().fp_mfma_bf16_tflops(dev, inner_iters, launches), "mfma_bf16")


def mfma_tile_gemm(a: np.ndarray, b: np.ndarray, dev: int = 0) -> np.ndarray:
    """D[16,16] = a[16,K] @ b[K,16] on the matrix cores (bf16 in, fp32 out).

    a/b are float32 arrays; they are truncated to bf16 on the host exactly as
    the kernel consumes them, so a torch fp32 reference over the truncated
    inputs is bitwise-comparable modulo accumulation order.
    """
    K = a.shape[1]
    assert a.shape == (16, K) and b.shape == (K, 16) and K % 32 == 0
    a_bf = _to_bf16_bits(np.ascontiguousarray(a, dtype=np.float32))
    b_bf = _to_bf16_bits(np.ascontiguousarray(b, dtype=np.float32))
    out = np.zeros((16, 16), dtype=np.float32)
    rc = _load().fp_mfma_tile_gemm_host(
        dev,
        a_bf.ctypes.data_as(ctypes.POINTER(ctypes.c_uint16)),
        b_bf.ctypes.data_as(ctypes.POINTER(ctypes.c_uint16)),
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
        K,
    )
    if rc < 0:
        raise ProbeError(f"mfma_tile_gemm failed with hip error {-rc}")
    return out


def hbm_block_sum(src: np.ndarray, blocks: int = 64, dev: int = 0) -> np.ndarray:
    src = np.ascontiguousarray(src, dtype=np.float32)
    out = np.zeros(blocks, dtype=np.float32)
    rc = _load().fp_hbm_block_sum_host(
        dev,
        src.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
        src.size,
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
        blocks,
    )
    if rc < 0:
        raise ProbeError(f"hbm_block_sum failed with hip error {-rc}")
    return out


def p2p_read_gbps(dst_dev: int, src_dev: int, bytes_: int = 1 << 30, iters: int = 10) -> float:
    v = _load().fp_p2p_read_gbps(dst_dev, src_dev, bytes_, iters)
    if v == -1.0:
        raise ProbeError(f"no p2p access between device {dst_dev} and {src_dev}")
    return _check(v, "p2p_read")


def allreduce_pull_gbps(bytes_: int = 1 << 30, iters: int = 5) -> float:
    v = _load().fp_allreduce_pull_gbps(bytes_, iters)
    if v == -1.0:
        raise ProbeError("allreduce probe needs >= 2 GPUs")
    return _check(v, "allreduce_pull")


def gemm_bf16_tflops(dev: int = 0, size: int = 4096, iters: int = 10) -> float:
    """LDS-staged big-tile bf16 GEMM throughput (size^3 problem; dispatches
    the measured champion: 256x256 tile, depth-2, register-hoisted,
    XOR-swizzled LDS)."""
    return _check(_load().fp_gemm_bf16_tflops(dev, size, iters), "gemm_bf16")


def gemm_bf16_tflops_ex(dev: int = 0, size: int = 4096, iters: int = 10,
                        bk: int = 32) -> float:
    """Throughput of a specific kernel variant. bk selector: 32/64 = the
    16x16x32 tiling at that K-step depth; 232/264 = the 32x32x16 tiling
    (half the LDS read bytes per FLOP) at BK=32/64."""
    return _check(_load().fp_gemm_bf16_tflops_ex(dev, size, iters, bk),
                  "gemm_bf16_ex")


def gemm_bf16(a: np.ndarray, bt: np.ndarray, dev: int = 0, bk: int = 32) -> np.ndarray:
    """C[M,N] = a[M,K] @ bt[N,K]^T on the LDS-staged GEMM kernel (bf16 in,
    fp32 out); a/bt are float32, truncated to bf16 exactly as consumed."""
    M, K = a.shape
    N, K2 = bt.shape
    # selector -> K-step depth: 232/264 = 32x32x16 tiling, 332/364 = the
    # 3-buffer pipelined kernel (counted vmcnt + raw barrier)
    kstep = {32: 32, 64: 64, 232: 32, 264: 64, 332: 32, 364: 64, 432: 32,
             532: 32, 632: 32, 732: 32, 764: 64, 832: 32, 842: 32, 844: 32, 848: 32, 852: 32}[bk]
    tile_m = {732: 256, 764: 256, 832: 256, 842: 256, 844: 256, 848: 256, 852: 256}.get(bk, 128)
    assert K == K2 and M % tile_m == 0 and N % 128 == 0 and K % kstep == 0
    a_bf = _to_bf16_bits(np.ascontiguousarray(a, dtype=np.float32))
    b_bf = _to_bf16_bits(np.ascontiguousarray(bt, dtype=np.float32))
    out = np.zeros((M, N), dtype=np.float32)
    rc = _load().fp_gemm_bf16_host_ex(
        dev,
        a_bf.ctypes.data_as(ctypes.POINTER(ctypes.c_uint16)),
        b_bf.ctypes.data_as(ctypes.POINTER(ctypes.c_uint16)),
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
        M, N, K, bk,
    )
    if rc < 0:
        raise ProbeError(f"gemm_bf16 failed with hip error {-rc}")
    return out


def burn(dev: int = 0, duration_ms: int = 2000) -> tuple:
    """Concurrent MFMA + HBM stress (the dcgmi-diag analog): returns
    (tflops, gbps) achieved while both run together."""
    tf = ctypes.c_double()
    gb = ctypes.c_double()
    rc = _load().fp_burn(dev, duration_ms, ctypes.byref(tf), ctypes.byref(gb))
    if rc < 0:
        raise ProbeError(f"burn failed with hip error {-rc}")
    return tf.value, gb.value


def to_fp8_e4m3(x: np.ndarray) -> np.ndarray:
    """Encode float32 -> OCP fp8 e4m3fn bytes (round-to-nearest-even,
    saturate to +-448, no inf; gfx950 uses OCP, NOT the MI300X fnuz
    encoding — cdna_hip_programming.md "FP8 — OCP, not FNUZ")."""
    x = np.asarray(x, dtype=np.float32)
    out = np.zeros(x.shape, dtype=np.uint8)
    sign = (x < 0) | ((x == 0) & (np.signbit(x)))
    ax = np.abs(x)
    ax = np.minimum(ax, 448.0)  # saturate (e4m3fn has no inf)
    # normals: exponent range [-6, 8]; subnormals below 2^-6
    e = np.floor(np.log2(np.maximum(ax, 1e-45))).astype(np.int32)
    e = np.clip(e, -6, 8)
    scale = np.exp2(e.astype(np.float64) - 3)  # mantissa step 2^(e-3)
    q = np.rint(ax / scale)  # RNE via rint
    # mantissa overflow (q == 16) bumps the exponent
    bump = q >= 16
    e = e + bump.astype(np.int32)
    q = np.where(bump, 8, q)  # 16/2 -> 1.0 -> stored mantissa 0 + implied 1 -> q=8
    over = e > 8
    e = np.where(over, 8, e)
    q = np.where(over, 15, q)  # clamp to max normal 448 = 1.75 * 2^8
    is_sub = ax < 2 ** -6
    # normal: bits = ((e+7)<<3) | (q-8); subnormal: e field 0, mantissa = round(ax/2^-9)
    sub_m = np.clip(np.rint(ax / 2.0 ** -9), 0, 7).astype(np.uint8)
    norm_bits = (((e + 7) << 3) | (q.astype(np.int32) - 8)).astype(np.uint8)
    out = np.where(is_sub, sub_m, norm_bits)
    out = np.where(ax == 0, 0, out)
    out |= (sign.astype(np.uint8) << 7)
    return out.astype(np.uint8)


def fp8_e4m3_to_f32(b: np.ndarray) -> np.ndarray:
    """Decode OCP e4m3fn bytes -> float32 (the dequant reference)."""
    b = np.asarray(b, dtype=np.uint8)
    sign = np.where(b & 0x80, -1.0, 1.0).astype(np.float32)
    e = ((b >> 3) & 0xF).astype(np.int32)
    m = (b & 0x7).astype(np.float32)
    normal = e > 0
    val = np.where(normal,
                   (1.0 + m / 8.0) * np.exp2((e - 7).astype(np.float32)),
                   (m / 8.0) * np.exp2(np.float32(-6)))
    return (sign * val).astype(np.float32)


def mfma_fp8_tile_gemm(a: np.ndarray, b: np.ndarray, dev: int = 0,
                       layout: int = 0) -> np.ndarray:
    """D[16,16] = a[16,K] @ b[K,16] on the MX-fp8 (e4m3, scale=1) matrix
    cores via mfma_scale_f32_16x16x128_f8f6f4. `layout` selects the A/B
    fragment-map hypothesis (see fabric_probe.hip) — pinned empirically."""
    K = a.shape[1]
    assert a.shape == (16, K) and b.shape == (K, 16) and K % 128 == 0
    a8 = to_fp8_e4m3(np.ascontiguousarray(a, dtype=np.float32))
    b8 = to_fp8_e4m3(np.ascontiguousarray(b, dtype=np.float32))
    out = np.zeros((16, 16), dtype=np.float32)
    rc = _load().fp_mfma_fp8_tile_gemm_host(
        dev,
        a8.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        b8.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
        K, layout,
    )
    if rc < 0:
        raise ProbeError(f"mfma_fp8_tile_gemm failed with hip error {-rc}")
    return out


# OCP e2m1 (fp4) value table: s * {0,0.5,1,1.5,2,3,4,6}
_FP4_VALUES = np.array([0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0], dtype=np.float32)


def to_fp4_e2m1(x: np.ndarray) -> np.ndarray:
    """Encode float32 -> OCP fp4 e2m1 nibbles (0..15; round-to-nearest,
    ties toward the larger magnitude per the table midpoints)."""
    x = np.asarray(x, dtype=np.float32)
    ax = np.abs(x).reshape(-1)
    idx = np.argmin(np.abs(ax[:, None] - _FP4_VALUES[None, :]), axis=1).astype(np.uint8)
    nib = idx | (np.signbit(x.reshape(-1)).astype(np.uint8) << 3)
    return nib.reshape(x.shape)


def fp4_e2m1_to_f32(nib: np.ndarray) -> np.ndarray:
    nib = np.asarray(nib, dtype=np.uint8)
    mag = _FP4_VALUES[(nib & 0x7).reshape(-1)].reshape(nib.shape)
    return np.where(nib & 0x8, -mag, mag).astype(np.float32)


def _pack_nibbles(nib: np.ndarray, axis_len: int) -> np.ndarray:
    """Pack pairs along the last axis (even = low nibble)."""
    flat = nib.reshape(-1, axis_len)
    return (flat[:, 0::2] | (flat[:, 1::2] << 4)).astype(np.uint8)


def mx_quantize_fp8(x: np.ndarray, block: int = 32):
    """MX quantization along the last axis: per-block E8M0 scale (power of
    two) + e4m3 elements of x/scale. Returns (elems_u8, scales_u8)."""
    x = np.asarray(x, dtype=np.float32)
    assert x.shape[-1] % block == 0
    xb = x.reshape(*x.shape[:-1], -1, block)
    absmax = np.abs(xb).max(axis=-1, keepdims=True)
    # scale = 2^e with absmax/2^e <= 448 (e4m3 max); e8m0 byte = e + 127
    e = np.ceil(np.log2(np.maximum(absmax, 1e-30) / 448.0))
    e = np.clip(e, -127, 127)
    scale = np.exp2(e).astype(np.float32)
    elems = to_fp8_e4m3((xb / scale).reshape(x.shape))
    scales = (e.astype(np.int32) + 127).astype(np.uint8).reshape(*x.shape[:-1], -1)
    return elems, scales


def mx_dequantize_fp8(elems: np.ndarray, scales: np.ndarray, block: int = 32):
    v = fp8_e4m3_to_f32(elems)
    sc = np.exp2(scales.astype(np.float32) - 127.0)
    vb = v.reshape(*v.shape[:-1], -1, block)
    return (vb * sc[..., None]).reshape(v.shape)


def mx_quantize_fp4(x: np.ndarray, block: int = 32):
    """MX quantization along the last axis for e2m1: per-block E8M0 scale
    with absmax/scale <= 6.0 (e2m1 max). Returns (nibbles_u8, scales_u8)."""
    x = np.asarray(x, dtype=np.float32)
    assert x.shape[-1] % block == 0
    xb = x.reshape(*x.shape[:-1], -1, block)
    absmax = np.abs(xb).max(axis=-1, keepdims=True)
    e = np.ceil(np.log2(np.maximum(absmax, 1e-30) / 6.0))
    e = np.clip(e, -127, 127)
    scale = np.exp2(e).astype(np.float32)
    nib = to_fp4_e2m1((xb / scale).reshape(x.shape))
    scales = (e.astype(np.int32) + 127).astype(np.uint8).reshape(*x.shape[:-1], -1)
    return nib, scales


def mx_dequantize_fp4(nib: np.ndarray, scales: np.ndarray, block: int = 32):
    v = fp4_e2m1_to_f32(nib)
    sc = np.exp2(scales.astype(np.float32) - 127.0)
    vb = v.reshape(*v.shape[:-1], -1, block)
    return (vb * sc[..., None]).reshape(v.shape)


def mfma_fp4_scaled_tile(a: np.ndarray, b: np.ndarray, dev: int = 0) -> np.ndarray:
    """D[32,32] = a[32,K] @ b[K,32] with REAL per-32-block MX scales on the
    fp4 matrix cores (naive scale layout, pinned by the fp4 scale probe)."""
    K = a.shape[1]
    assert a.shape == (32, K) and b.shape == (K, 32) and K % 64 == 0
    a4, sa = mx_quantize_fp4(a)
    b4t, sb = mx_quantize_fp4(np.ascontiguousarray(b.T))
    b4 = np.ascontiguousarray(b4t.T)
    a_packed = _pack_nibbles(a4, K)
    b_pairs = b4.reshape(K // 2, 2, 32)
    b_packed = (b_pairs[:, 0, :] | (b_pairs[:, 1, :] << 4)).astype(np.uint8)
    out = np.zeros((32, 32), dtype=np.float32)
    rc = _load().fp_mfma_fp4_scaled_tile_host(
        dev,
        np.ascontiguousarray(a_packed).ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        np.ascontiguousarray(b_packed).ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        np.ascontiguousarray(sa).ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        np.ascontiguousarray(sb).ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
        K,
    )
    if rc < 0:
        raise ProbeError(f"mfma_fp4_scaled_tile failed with hip error {-rc}")
    return out


def gemm_fp4_scaled(a: np.ndarray, bt: np.ndarray, dev: int = 0):
    """C[M,N] = MX-dequant(a) @ MX-dequant(bt)^T, per-block E8M0 scales
    applied by the fp4 matrix cores (256x256 G16 XOR-swizzled tiles with
    LDS-staged scale panels). Returns (C, a4, sa, b4t, sbt)."""
    M, K = a.shape
    N, K2 = bt.shape
    assert K == K2 and M % 256 == 0 and N % 256 == 0 and K % 128 == 0
    a4, sa = mx_quantize_fp4(np.ascontiguousarray(a, dtype=np.float32))
    b4t, sbt = mx_quantize_fp4(np.ascontiguousarray(bt, dtype=np.float32))
    a_packed = _pack_nibbles(a4, K)
    b_packed = _pack_nibbles(b4t, K)
    out = np.zeros((M, N), dtype=np.float32)
    rc = _load().fp_gemm_fp4_scaled_host(
        dev,
        np.ascontiguousarray(a_packed).ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        np.ascontiguousarray(b_packed).ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        np.ascontiguousarray(sa).ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        np.ascontiguousarray(sbt).ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
        M, N, K,
    )
    if rc < 0:
        raise ProbeError(f"gemm_fp4_scaled failed with hip error {-rc}")
    return out, a4, sa, b4t, sbt


def gemm_fp4_scaled_tflops(dev: int = 0, size: int = 4096, iters: int = 10) -> float:
    """MX-scaled fp4 GEMM throughput (real per-block scales in the loop)."""
    return _check(_load().fp_gemm_fp4_scaled_tflops(dev, size, iters),
                  "gemm_fp4_scaled_tflops")


def mfma_fp8_scaled_tile(a: np.ndarray, b: np.ndarray, dev: int = 0) -> np.ndarray:
    """D[16,16] = a[16,K] @ b[K,16] through the matrix cores with REAL
    per-32-block MX (E8M0) scales — the HW-fused dequant+matmul path.
    a is MX-quantized along K; b along its leading (K) axis per column."""
    K = a.shape[1]
    assert a.shape == (16, K) and b.shape == (K, 16) and K % 128 == 0
    a8, sa = mx_quantize_fp8(a)
    # quantize B per column along K: transpose to [16][K], quantize, back
    b8_t, sb = mx_quantize_fp8(np.ascontiguousarray(b.T))
    b8 = np.ascontiguousarray(b8_t.T)
    out = np.zeros((16, 16), dtype=np.float32)
    rc = _load().fp_mfma_fp8_scaled_tile_host(
        dev,
        np.ascontiguousarray(a8).ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        np.ascontiguousarray(b8).ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        np.ascontiguousarray(sa).ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        np.ascontiguousarray(sb).ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
        K,
    )
    if rc < 0:
        raise ProbeError(f"mfma_fp8_scaled_tile failed with hip error {-rc}")
    return out


def mfma_fp4_tile_gemm(a: np.ndarray, b: np.ndarray, dev: int = 0) -> np.ndarray:
    """D[32,32] = a[32,K] @ b[K,32] on the MX-fp4 (e2m1, scale=1) matrix
    cores via mfma_scale_f32_32x32x64_f8f6f4 (the ~10 PF headline shape)."""
    K = a.shape[1]
    assert a.shape == (32, K) and b.shape == (K, 32) and K % 64 == 0
    a4 = to_fp4_e2m1(np.ascontiguousarray(a, dtype=np.float32))
    b4 = to_fp4_e2m1(np.ascontiguousarray(b, dtype=np.float32))
    a_packed = _pack_nibbles(a4, K)                       # [32][K/2]
    # B packed along k: byte (k/2)*32 + col
    b_pairs = b4.reshape(K // 2, 2, 32)
    b_packed = (b_pairs[:, 0, :] | (b_pairs[:, 1, :] << 4)).astype(np.uint8)
    out = np.zeros((32, 32), dtype=np.float32)
    rc = _load().fp_mfma_fp4_tile_gemm_host(
        dev,
        np.ascontiguousarray(a_packed).ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        np.ascontiguousarray(b_packed).ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
        K,
    )
    if rc < 0:
        raise ProbeError(f"mfma_fp4_tile_gemm failed with hip error {-rc}")
    return out


def gemm_fp4(a: np.ndarray, bt: np.ndarray, dev: int = 0,
             variant: int = 446) -> np.ndarray:
    """C[M,N] = a[M,K] @ bt[N,K]^T on the MX-fp4 GEMM kernel (e2m1 packed
    two-per-byte, fp32 out). variant 4 = 256x256 2-buf, 416 = +G16 swizzle,
    436 = 3-buf counted pipeline G16."""
    M, K = a.shape
    N, K2 = bt.shape
    tile_m = 512 if variant == 456 else 256
    assert K == K2 and M % tile_m == 0 and N % 256 == 0 and K % 256 == 0
    a4 = _pack_nibbles(to_fp4_e2m1(np.ascontiguousarray(a, dtype=np.float32)), K)
    b4 = _pack_nibbles(to_fp4_e2m1(np.ascontiguousarray(bt, dtype=np.float32)), K)
    out = np.zeros((M, N), dtype=np.float32)
    rc = _load().fp_gemm_fp8_host_ex(
        dev,
        np.ascontiguousarray(a4).ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        np.ascontiguousarray(b4).ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
        M, N, K, variant,
    )
    if rc < 0:
        raise ProbeError(f"gemm_fp4 failed with hip error {-rc}")
    return out


def gemm_fp4_tflops_ex(dev: int = 0, size: int = 4096, iters: int = 10,
                       variant: int = 446) -> float:
    """MX-fp4 GEMM throughput. Champion 446 (3-buf counted G16 + XOR LDS
    swizzle): measured 3060/3526 TF @4096^3/8192^3 vs the 9074 TF mfma
    ceiling (gpurun_out/r2s17-20, r2s24)."""
    return _check(_load().fp_gemm_fp8_tflops_ex(dev, size, iters, variant),
                  "gemm_fp4_ex")


def mfma_fp4_tflops(dev: int = 0, inner_iters: int = 2048, launches: int = 20) -> float:
    """Register-resident mfma_scale fp4 32x32x64 issue-rate ceiling."""
    return _check(_load().fp_mfma_fp4_tflops(dev, inner_iters, launches), "mfma_fp4")


def mfma_fp8_tflops(dev: int = 0, inner_iters: int = 2048, launches: int = 20) -> float:
    """Register-resident mfma_scale fp8 issue-rate ceiling (no memory)."""
    return _check(_load().fp_mfma_fp8_tflops(dev, inner_iters, launches), "mfma_fp8")


def gemm_fp8_tflops_ex(dev: int = 0, size: int = 4096, iters: int = 10,
                       variant: int = 346) -> float:
    """MX-fp8 (e4m3, scale=1) GEMM throughput. Variants: 1 = 128x128;
    2/24/28/216/232 = 256x128 3-buf (suffix = tile-group swizzle);
    3/316 = 256x256 2-buf dual-barrier + G16; 326 adds the XOR LDS bank
    swizzle; champion 346 further drops the provably-redundant trailing
    barrier: 1824/1992-2003 TF @4096^3/8192^3 vs the 4780 TF mfma_scale
    ceiling (gpurun_out/r2s15, r2s24, r2s26)."""
    return _check(_load().fp_gemm_fp8_tflops_ex(dev, size, iters, variant),
                  "gemm_fp8_ex")


def gemm_fp8(a: np.ndarray, bt: np.ndarray, dev: int = 0,
             variant: int = 346) -> np.ndarray:
    """C[M,N] = a[M,K] @ bt[N,K]^T on the MX-fp8 GEMM kernel (e4m3 in,
    fp32 out); inputs are float32, quantized to e4m3 exactly as consumed."""
    M, K = a.shape
    N, K2 = bt.shape
    tile_m = {1: 128, 456: 512}.get(variant, 256)
    assert K == K2 and M % tile_m == 0 and N % 128 == 0 and K % 128 == 0
    a8 = to_fp8_e4m3(np.ascontiguousarray(a, dtype=np.float32))
    b8 = to_fp8_e4m3(np.ascontiguousarray(bt, dtype=np.float32))
    out = np.zeros((M, N), dtype=np.float32)
    rc = _load().fp_gemm_fp8_host_ex(
        dev,
        a8.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        b8.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
        M, N, K, variant,
    )
    if rc < 0:
        raise ProbeError(f"gemm_fp8 failed with hip error {-rc}")
    return out


def gemm_fp8_scaled(a: np.ndarray, bt: np.ndarray, dev: int = 0,
                    variant: int = 556):
    """C[M,N] = MX-dequant(a) @ MX-dequant(bt)^T with REAL per-32-block
    E8M0 scales applied by the matrix cores (HW-fused dequant+matmul).
    Inputs are float32; both operands are MX-quantized along K here and
    the (codes, scales) consumed verbatim by the kernel. Returns
    (C, a8, sa, b8t, sbt) so callers can build the exact dequant
    reference.

    Variant ladder (measured 8192^3): 52 per-step global scale byte loads
    748 TF; 526 +XOR swizzle 761; 546 4-step vector loads 414 (VGPR
    spill); 556 scales staged into LDS by stage() 1813 TF = 90% of the
    unscaled champion (v346, 2025) — the default."""
    M, K = a.shape
    N, K2 = bt.shape
    tile_m = 128 if variant == 5 else 256
    assert K == K2 and M % tile_m == 0 and N % 128 == 0 and K % 128 == 0
    # the 4-step-grouped variant iterates K in 512-element groups
    assert variant != 546 or K % 512 == 0
    a8, sa = mx_quantize_fp8(np.ascontiguousarray(a, dtype=np.float32))
    b8t, sbt = mx_quantize_fp8(np.ascontiguousarray(bt, dtype=np.float32))
    out = np.zeros((M, N), dtype=np.float32)
    rc = _load().fp_gemm_fp8_scaled_host(
        dev,
        np.ascontiguousarray(a8).ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        np.ascontiguousarray(b8t).ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        np.ascontiguousarray(sa).ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        np.ascontiguousarray(sbt).ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
        M, N, K, variant,
    )
    if rc < 0:
        raise ProbeError(f"gemm_fp8_scaled failed with hip error {-rc}")
    return out, a8, sa, b8t, sbt


def gemm_fp8_scaled_tflops(dev: int = 0, size: int = 4096, iters: int = 10,
                           variant: int = 556) -> float:
    """MX-scaled fp8 GEMM throughput (real per-block scales in the loop)."""
    return _check(_load().fp_gemm_fp8_scaled_tflops(dev, size, iters, variant),
                  "gemm_fp8_scaled_tflops")


def gemm_bf16_splitk(a: np.ndarray, bt: np.ndarray, ksplit: int = 4,
                     dev: int = 0) -> np.ndarray:
    """C[M,N] = a @ bt^T via the split-K 128x128 kernel: `ksplit` partial
    products per tile accumulated with f32 hardware atomics — fills the
    chip on shapes where plain tiling launches < 256 workgroups."""
    M, K = a.shape
    N, K2 = bt.shape
    assert K == K2 and K % (32 * ksplit) == 0 and (K // ksplit) % 32 == 0
    a16 = _to_bf16_bits(np.ascontiguousarray(a, dtype=np.float32))
    b16 = _to_bf16_bits(np.ascontiguousarray(bt, dtype=np.float32))
    out = np.zeros((M, N), dtype=np.float32)
    rc = _load().fp_gemm_bf16_splitk_host(
        dev,
        a16.ctypes.data_as(ctypes.POINTER(ctypes.c_int16)),
        b16.ctypes.data_as(ctypes.POINTER(ctypes.c_int16)),
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
        M, N, K, ksplit,
    )
    if rc < 0:
        raise ProbeError(f"gemm_bf16_splitk failed with hip error {-rc}")
    return out


def gemm_bf16_splitk_tflops(dev: int = 0, size: int = 2048, iters: int = 10,
                            ksplit: int = 4) -> float:
    """Split-K bf16 GEMM throughput (C zeroing included in each iteration)."""
    return _check(_load().fp_gemm_bf16_splitk_tflops(dev, size, iters, ksplit),
                  "gemm_bf16_splitk_tflops")


def gemm_bf16_splitk_tflops_mnk(M: int, N: int, K: int, dev: int = 0,
                                iters: int = 10, ksplit: int = 8) -> float:
    """Split-K throughput on an explicit (M, N, K) — the kernel's target is
    tall-skinny K (few output tiles, huge reduction dim)."""
    return _check(_load().fp_gemm_bf16_splitk_tflops_mnk(
        dev, M, N, K, iters, ksplit), "gemm_bf16_splitk_tflops_mnk")


def _to_bf16_bits(x: np.ndarray) -> np.ndarray:
    """Round-to-nearest-even f32 -> bf16 bit pattern (uint16)."""
    bits = x.view(np.uint32)
    rounding = ((bits >> 16) & 1) + 0x7FFF
    return ((bits + rounding) >> 16).astype(np.uint16)


def bf16_truncate(x: np.ndarray) -> np.ndarray:
    """f32 -> bf16 -> f32 (for building CPU references)."""
    return (_to_bf16_bits(x).astype(np.uint32) << 16).view(np.float32).reshape(x.shape)


@dataclass
class FabricProbeReport:
    """The fabric daemon's readiness evidence (BASELINE.json north star)."""

    device_count: int
    hbm_read_gbps: float
    hbm_write_gbps: float
    mfma_bf16_tflops: float
    gemm_bf16_tflops: float
    p2p_gbps: List[List[float]]  # [dst][src], -1 on self
    allreduce_gbps: float
    # low-precision matrix-core floors (MX-scaled fp8/fp4; 0.0 = not run)
    mfma_fp8_tflops: float = 0.0
    mfma_fp4_tflops: float = 0.0
    # quantized-GEMM floors with REAL per-block E8M0 scales flowing
    # through the mfma scale operands (the production MX inference path)
    gemm_fp8_mx_tflops: float = 0.0
    gemm_fp4_mx_tflops: float = 0.0


def run_fabric_report(quick: bool = True) -> FabricProbeReport:
    n = device_count()
    if n == 0:
        raise ProbeError("no GPUs visible")
    # stay above the 256 MiB Infinity Cache even in quick mode, else
    # bandwidth numbers read the LLC, not HBM (MI355X_MICROARCH L3 note)
    size = (512 << 20) if quick else (2 << 30)
    iters = 5 if quick else 20
    p2p = [[-1.0] * n for _ in range(n)]
    for d in range(n):
        for s in range(n):
            if d != s:
                try:
                    p2p[d][s] = p2p_read_gbps(d, s, size, iters)
                except ProbeError:
                    p2p[d][s] = 0.0
    ar = 0.0
    if n >= 2:
        try:
            ar = allreduce_pull_gbps(size, iters)
        except ProbeError:
            ar = 0.0
    return FabricProbeReport(
        device_count=n,
        hbm_read_gbps=hbm_read_gbps(0, size, iters),
        hbm_write_gbps=hbm_write_gbps(0, size, iters),
        mfma_bf16_tflops=mfma_bf16_tflops(0, 1024, 10),
        # quick mode uses the 128x128-tile kernel: a 2048^3 problem is only
        # 64 workgroups for the 256x256 champion (256 CUs idle)
        gemm_bf16_tflops=(gemm_bf16_tflops_ex(0, 2048, 5, 432) if quick
                          else gemm_bf16_tflops(0, 8192, 5)),
        p2p_gbps=p2p,
        allreduce_gbps=ar,
        mfma_fp8_tflops=mfma_fp8_tflops(0, 1024, 5),
        mfma_fp4_tflops=mfma_fp4_tflops(0, 1024, 5),
        gemm_fp8_mx_tflops=gemm_fp8_scaled_tflops(0, 2048 if quick else 8192, 5),
        gemm_fp4_mx_tflops=gemm_fp4_scaled_tflops(0, 2048 if quick else 8192, 5),
    )
