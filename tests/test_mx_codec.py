"""CPU tests for the MX (microscaling) codecs in fabric/probe.py: OCP fp8
e4m3 / fp4 e2m1 encode+decode and the per-32-block E8M0 quantizers that
feed the hardware-scaled GEMM kernels. Pure numpy — no GPU required.

Reference behavior: the quantized (codes, scales) pairs are consumed
verbatim by mfma_fp8_scaled_tile / gemm_fp8_scaled / gemm_fp4_scaled, so
these properties pin the exact contract the kernels rely on.
"""

import numpy as np
import pytest

from k8s_dra_driver_gpu_amd.fabric import probe


class TestMxQuantizeFp8:
    def test_scales_are_powers_of_two(self):
        rng = np.random.default_rng(0)
        x = (rng.standard_normal((8, 128))
             * np.exp2(rng.integers(-10, 11, (8, 128)))).astype(np.float32)
        _, scales = probe.mx_quantize_fp8(x)
        assert scales.dtype == np.uint8
        assert scales.shape == (8, 4)
        # e8m0: value = 2^(byte-127); always exactly a power of two

    def test_block_absmax_within_e4m3_range(self):
        rng = np.random.default_rng(1)
        x = (rng.standard_normal((4, 64)) * 1e6).astype(np.float32)
        codes, scales = probe.mx_quantize_fp8(x)
        scaled = x.reshape(4, 2, 32) / np.exp2(
            scales.astype(np.float32) - 127.0)[..., None]
        assert np.abs(scaled).max() <= 448.0 + 1e-3  # e4m3 max magnitude

    def test_round_trip_error_bound(self):
        rng = np.random.default_rng(2)
        x = (rng.standard_normal((16, 256))
             * np.exp2(rng.integers(-6, 7, (16, 256)))).astype(np.float32)
        codes, scales = probe.mx_quantize_fp8(x)
        d = probe.mx_dequantize_fp8(codes, scales)
        # e4m3 has a 3-bit mantissa: relative error within a block is
        # bounded by ~2^-3 relative to the block absmax
        blocks = x.reshape(16, 8, 32)
        dblocks = d.reshape(16, 8, 32)
        absmax = np.abs(blocks).max(axis=-1, keepdims=True)
        rel = np.abs(dblocks - blocks) / np.maximum(absmax, 1e-30)
        assert rel.max() < 0.08

    def test_dequantize_is_exact_fixed_point(self):
        # decode -> re-quantize with the same scales -> decode is identity
        rng = np.random.default_rng(3)
        x = rng.standard_normal((2, 64)).astype(np.float32) * 100
        codes, scales = probe.mx_quantize_fp8(x)
        d = probe.mx_dequantize_fp8(codes, scales)
        codes2, scales2 = probe.mx_quantize_fp8(d)
        d2 = probe.mx_dequantize_fp8(codes2, scales2)
        np.testing.assert_array_equal(d, d2)

    def test_zero_block(self):
        codes, scales = probe.mx_quantize_fp8(np.zeros((1, 32), np.float32))
        assert probe.mx_dequantize_fp8(codes, scales).sum() == 0.0


class TestMxQuantizeFp4:
    def test_block_absmax_within_e2m1_range(self):
        rng = np.random.default_rng(4)
        x = (rng.standard_normal((4, 64)) * 1e5).astype(np.float32)
        nib, scales = probe.mx_quantize_fp4(x)
        scaled = x.reshape(4, 2, 32) / np.exp2(
            scales.astype(np.float32) - 127.0)[..., None]
        assert np.abs(scaled).max() <= 6.0 + 1e-3  # e2m1 max magnitude

    def test_nibble_range(self):
        rng = np.random.default_rng(5)
        x = rng.standard_normal((2, 128)).astype(np.float32) * 4
        nib, _ = probe.mx_quantize_fp4(x)
        assert nib.dtype == np.uint8
        assert nib.max() <= 0xF

    def test_round_trip_error_bound(self):
        rng = np.random.default_rng(6)
        x = (rng.standard_normal((8, 128))
             * np.exp2(rng.integers(-5, 6, (8, 128)))).astype(np.float32)
        nib, scales = probe.mx_quantize_fp4(x)
        d = probe.mx_dequantize_fp4(nib, scales)
        blocks = x.reshape(8, 4, 32)
        dblocks = d.reshape(8, 4, 32)
        absmax = np.abs(blocks).max(axis=-1, keepdims=True)
        # e2m1's coarse grid: worst-case within-block relative error is
        # half the largest value gap (6.0 vs 4.0 midpoint) / absmax
        rel = np.abs(dblocks - blocks) / np.maximum(absmax, 1e-30)
        assert rel.max() < 0.25

    def test_signs_preserved(self):
        x = np.array([[1.0, -1.0] * 16], dtype=np.float32)
        nib, scales = probe.mx_quantize_fp4(x)
        d = probe.mx_dequantize_fp4(nib, scales)
        assert (np.sign(d) == np.sign(x)).all()


class TestScaledLayoutContract:
    """Pin the host-side contracts the kernels assume: scale array shapes
    and the CH chunk-permutation invariant used by the fp8 scaled path."""

    def test_scale_shapes(self):
        x = np.ones((256, 512), np.float32)
        _, s8 = probe.mx_quantize_fp8(x)
        _, s4 = probe.mx_quantize_fp4(x)
        assert s8.shape == (256, 16)
        assert s4.shape == (256, 16)

    def test_ch_permutation_is_involution_free_bijection(self):
        # CH = [0,4,1,5,2,6,3,7]: slots s hold logical chunk CH[s]; scale
        # group g covers slots {2g, 2g+1} -> logical chunks {g, g+4} ==
        # k in [32g, 32g+32) after the pairing {0,2}/{4,6}/{1,3}/{5,7}
        CH = [0, 4, 1, 5, 2, 6, 3, 7]
        assert sorted(CH) == list(range(8))
        for g in range(4):
            logical = {CH[2 * g], CH[2 * g + 1]}
            assert logical == {g, g + 4}
