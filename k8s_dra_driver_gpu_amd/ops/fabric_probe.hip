// CDNA4 (gfx950 / MI355X) fabric & device probe kernels.
//
// These are the hand-written HIP probes behind the ComputeDomain fabric
// validation path (the analog of the reference's NCCL "nickelpie" +
// nvbandwidth test workloads, tests/bats/test_cd_mnnvl_workload.bats:18-55):
//
//  * hbm_read / hbm_write / hbm_copy  — HBM3E streaming bandwidth
//    (vectorized 16 B/lane, grid-stride, sized >> 256 workgroups to fill all
//    8 XCDs; ~8 TB/s peak, ~6.3 TB/s achievable per MI355X_MICROARCH.md),
//  * mfma_bf16_loop                   — matrix-core saturation probe
//    (v_mfma_f32_32x32x16_bf16 on register operands, 4 independent
//    accumulators to cover the issue latency; ~2.4 PF uench ceiling),
//  * mfma_bf16_tile_gemm              — numerics check: one-tile GEMM with
//    the documented fragment layout, verified against a PyTorch fp32
//    reference in tests,
//  * p2p_read                        — xGMI peer-to-peer pull bandwidth
//    (per-link ~153 GB/s x 7 links); used by the fabric daemon to attribute
//    per-peer link health.
//
// Wave size is 64 (CDNA), blocks are multiples of 64; no CUDA compatibility
// paths.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define PROBE_BLOCK 256  // 4 waves

typedef float float4v __attribute__((ext_vector_type(4)));
typedef short bf16x8 __attribute__((ext_vector_type(8)));
typedef short bf16x4 __attribute__((ext_vector_type(4)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

// ---------------------------------------------------------------------------
// HBM streaming probes
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(PROBE_BLOCK)
hbm_read_kernel(const float4v* __restrict__ src, float* __restrict__ sink,
                long n_vec) {
    // Grid-stride read of float4 (16 B/lane); accumulate to defeat DCE.
    long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    float4v acc = {0.f, 0.f, 0.f, 0.f};
    // 4-deep unroll keeps >= 4 loads in flight per lane.
    long i = idx;
    for (; i + 3 * stride < n_vec; i += 4 * stride) {
        float4v a = src[i];
        float4v b = src[i + stride];
        float4v c = src[i + 2 * stride];
        float4v d = src[i + 3 * stride];
        acc += a + b + c + d;
    }
    for (; i < n_vec; i += stride) acc += src[i];
    float r = acc.x + acc.y + acc.z + acc.w;
    if (r == -1.0f) sink[0] = r;  // never true for the test pattern; defeats DCE
}

// Variant: non-temporal loads (bypass L1/L2 allocation hints) — streaming
// reads never re-use lines, so this can reduce cache-path pressure.
extern "C" __global__ void __launch_bounds__(PROBE_BLOCK)
hbm_read_nt_kernel(const float4v* __restrict__ src, float* __restrict__ sink,
                   long n_vec) {
    long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    float4v acc = {0.f, 0.f, 0.f, 0.f};
    long i = idx;
    for (; i + 3 * stride < n_vec; i += 4 * stride) {
        float4v a = __builtin_nontemporal_load(&src[i]);
        float4v b = __builtin_nontemporal_load(&src[i + stride]);
        float4v c = __builtin_nontemporal_load(&src[i + 2 * stride]);
        float4v d = __builtin_nontemporal_load(&src[i + 3 * stride]);
        acc += a + b + c + d;
    }
    for (; i < n_vec; i += stride) acc += __builtin_nontemporal_load(&src[i]);
    float r = acc.x + acc.y + acc.z + acc.w;
    if (r == -1.0f) sink[0] = r;
}

// Variant: 8-deep unroll (more loads in flight per lane).
extern "C" __global__ void __launch_bounds__(PROBE_BLOCK)
hbm_read_u8_kernel(const float4v* __restrict__ src, float* __restrict__ sink,
                   long n_vec) {
    long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    float4v acc = {0.f, 0.f, 0.f, 0.f};
    long i = idx;
    for (; i + 7 * stride < n_vec; i += 8 * stride) {
        float4v v0 = src[i];
        float4v v1 = src[i + stride];
        float4v v2 = src[i + 2 * stride];
        float4v v3 = src[i + 3 * stride];
        float4v v4 = src[i + 4 * stride];
        float4v v5 = src[i + 5 * stride];
        float4v v6 = src[i + 6 * stride];
        float4v v7 = src[i + 7 * stride];
        acc += (v0 + v1) + (v2 + v3) + ((v4 + v5) + (v6 + v7));
    }
    for (; i < n_vec; i += stride) acc += src[i];
    float r = acc.x + acc.y + acc.z + acc.w;
    if (r == -1.0f) sink[0] = r;
}

// Variant: contiguous-chunk reads — each lane reads 4 consecutive float4s
// (64 B per lane per step), wave covers a 4 KiB line-aligned span.
extern "C" __global__ void __launch_bounds__(PROBE_BLOCK)
hbm_read_chunk_kernel(const float4v* __restrict__ src, float* __restrict__ sink,
                      long n_vec) {
    long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    long chunks = n_vec / 4;
    float4v acc = {0.f, 0.f, 0.f, 0.f};
    for (long c = tid; c < chunks; c += stride) {
        const float4v* p = src + c * 4;
        acc += p[0] + p[1] + p[2] + p[3];
    }
    float r = acc.x + acc.y + acc.z + acc.w;
    if (r == -1.0f) sink[0] = r;
}

// Deliberately dereference an unmapped device address: generates a GPU VM
// page fault, the KFD event the health monitor consumes as its XID-13
// analog (fault-injection only — used by the end-to-end health test).
// Reads AND writes a wild address so neither XNACK retry nor load
// elimination can swallow the fault.
extern "C" __global__ void vmfault_kernel(float* __restrict__ sink) {
    volatile float* bad = (volatile float*)(0xdeadbeef000ull);
    float v = bad[threadIdx.x];
    bad[threadIdx.x] = v + 1.0f;
    sink[0] = v;
}

extern "C" __global__ void __launch_bounds__(PROBE_BLOCK)
hbm_write_kernel(float4v* __restrict__ dst, long n_vec, float val) {
    long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    float4v v = {val, val, val, val};
    for (long i = idx; i < n_vec; i += stride) dst[i] = v;
}

extern "C" __global__ void __launch_bounds__(PROBE_BLOCK)
hbm_copy_kernel(float4v* __restrict__ dst, const float4v* __restrict__ src,
                long n_vec) {
    long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (long i = idx; i < n_vec; i += stride) dst[i] = src[i];
}

// Non-temporal store variants (streaming writes never re-read their lines).
extern "C" __global__ void __launch_bounds__(PROBE_BLOCK)
hbm_write_nt_kernel(float4v* __restrict__ dst, long n_vec, float val) {
    long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    float4v v = {val, val, val, val};
    for (long i = idx; i < n_vec; i += stride)
        __builtin_nontemporal_store(v, &dst[i]);
}

extern "C" __global__ void __launch_bounds__(PROBE_BLOCK)
hbm_copy_nt_kernel(float4v* __restrict__ dst, const float4v* __restrict__ src,
                   long n_vec) {
    long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (long i = idx; i < n_vec; i += stride)
        __builtin_nontemporal_store(__builtin_nontemporal_load(&src[i]), &dst[i]);
}

// Correctness companion for the read probe: block-level sums written out so
// the Python test can compare against a torch fp32 reference.
extern "C" __global__ void __launch_bounds__(PROBE_BLOCK)
hbm_block_sum_kernel(const float* __restrict__ src, float* __restrict__ out,
                     long n) {
    __shared__ float red[PROBE_BLOCK / WAVE];
    long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    float acc = 0.f;
    for (long i = idx; i < n; i += stride) acc += src[i];
    // wave reduce (64-wide)
    for (int off = WAVE / 2; off > 0; off >>= 1)
        acc += __shfl_down(acc, off, WAVE);
    int lane = threadIdx.x & (WAVE - 1);
    int wid = threadIdx.x / WAVE;
    if (lane == 0) red[wid] = acc;
    __syncthreads();
    if (threadIdx.x == 0) {
        float s = 0.f;
        for (int w = 0; w < PROBE_BLOCK / WAVE; ++w) s += red[w];
        out[blockIdx.x] = s;
    }
}

// ---------------------------------------------------------------------------
// MFMA saturation probe (bf16 32x32x16)
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(PROBE_BLOCK)
mfma_bf16_loop_kernel(const short* __restrict__ seed, float* __restrict__ sink,
                      int iters) {
    // Register-resident MFMA chain: 4 independent accumulators per wave cover
    // the 32x32 issue interval; operands come from memory once (prevents
    // constant folding).
    bf16x8 a, b;
    #pragma unroll
    for (int e = 0; e < 8; ++e) {
        a[e] = seed[(threadIdx.x & (WAVE - 1)) * 8 + e];
        b[e] = seed[512 + (threadIdx.x & (WAVE - 1)) * 8 + e];
    }
    f32x16 acc0 = {}, acc1 = {}, acc2 = {}, acc3 = {};
    for (int i = 0; i < iters; ++i) {
        acc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc0, 0, 0, 0);
        acc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc1, 0, 0, 0);
        acc2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc2, 0, 0, 0);
        acc3 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc3, 0, 0, 0);
    }
    float r = 0.f;
    #pragma unroll
    for (int e = 0; e < 16; ++e) r += acc0[e] + acc1[e] + acc2[e] + acc3[e];
    if (r == -1.0f) sink[0] = r;
}

// ---------------------------------------------------------------------------
// MFMA one-tile GEMM (numerics check): D[16x16] = A[16xK] * B[Kx16], bf16 in,
// f32 out, K a multiple of 32, one wave.
//
// Fragment layout for v_mfma_f32_16x16x32_bf16 (cdna_hip_programming.md 3):
//   A: lane l holds A[row = l&15][k = (l>>4)*8 + e], e in 0..7
//   B: lane l holds B[k = (l>>4)*8 + e][col = l&15]
//   C/D: lane l reg r -> row = (l>>4)*4 + r, col = l&15
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(WAVE)
mfma_bf16_tile_gemm_kernel(const short* __restrict__ A,
                           const short* __restrict__ B,
                           float* __restrict__ D, int K) {
    int lane = threadIdx.x & (WAVE - 1);
    int row = lane & 15;
    int kgrp = lane >> 4;  // 0..3
    f32x4 acc = {};
    for (int k0 = 0; k0 < K; k0 += 32) {
        bf16x8 a, b;
        #pragma unroll
        for (int e = 0; e < 8; ++e) {
            int k = k0 + kgrp * 8 + e;
            a[e] = A[row * K + k];      // A is [16][K] row-major
            b[e] = B[k * 16 + row];     // B is [K][16] row-major; col = row idx
        }
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
        int out_row = kgrp * 4 + r;
        D[out_row * 16 + row] = acc[r];
    }
}

// ---------------------------------------------------------------------------
// MX-fp8 (OCP e4m3fn) 16x16x128 block-scaled MFMA tile verify. gfx950's
// mfma_scale_f32_16x16x128_f8f6f4 is the only large-K fp8 MFMA (2x the
// bf16 rate; cdna_hip_programming.md gfx950 intrinsic list). Scales are
// E8M0, one per 32-element MX block = one byte per lane per operand; 0x7F
// (2^0) makes the scaled form numerically plain fp8.
//
// `layout` selects the A/B lane->element hypothesis so the real mapping is
// pinned EMPIRICALLY in one GPU call (no ISA doc for the fragment map in
// this image):
//   0: k = (lane>>4)*32 + e            (natural extension of bf16 16x16x32)
//   1: k = e*4 + (lane>>4)             (K-interleaved across lane groups)
//   2: k = (lane>>4)*8 + (e&7) + (e>>3)*32   (8-elem dwords from 4 blocks)
// C/D layout is shape-determined and dtype-independent on gfx950 (guide
// "Fragment layout"): row=(lane>>4)*4+r, col=lane&15, same as bf16.
// ---------------------------------------------------------------------------

typedef int i32x8 __attribute__((ext_vector_type(8)));

extern "C" __global__ void __launch_bounds__(WAVE)
mfma_fp8_tile_gemm_kernel(const unsigned char* __restrict__ A,
                          const unsigned char* __restrict__ B,
                          float* __restrict__ D, int K, int layout) {
    int lane = threadIdx.x & (WAVE - 1);
    int row = lane & 15;
    int kgrp = lane >> 4;  // 0..3
    f32x4 acc = {};
    for (int k0 = 0; k0 < K; k0 += 128) {
        union {
            i32x8 v;
            unsigned char b[32];
        } a, bb;
        for (int e = 0; e < 32; ++e) {
            int k;
            if (layout == 1)
                k = e * 4 + kgrp;
            else if (layout == 2)
                k = kgrp * 8 + (e & 7) + (e >> 3) * 32;
            else
                k = kgrp * 32 + e;
            a.b[e] = A[row * K + k0 + k];   // A is [16][K] row-major
            bb.b[e] = B[(k0 + k) * 16 + row];  // B is [K][16]; col = row idx
        }
        acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            a.v, bb.v, acc, 0 /*cbsz=fp8*/, 0 /*blgp=fp8*/,
            0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        int out_row = kgrp * 4 + r;
        D[out_row * 16 + row] = acc[r];
    }
}

// Register-resident mfma_scale_f32_16x16x128_f8f6f4 loop: the fp8 MFMA
// issue-rate ceiling (no memory traffic). 4 independent accumulators per
// lane hide the MFMA latency, same shape as mfma_bf16_loop_kernel.
extern "C" __global__ void __launch_bounds__(PROBE_BLOCK)
mfma_fp8_loop_kernel(const int* __restrict__ seed, float* __restrict__ sink,
                     int iters) {
    int lane = threadIdx.x & (WAVE - 1);
    union { i32x8 v; int i[8]; } a, b;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
        a.i[e] = seed[(lane * 8 + e) & 1023];
        b.i[e] = seed[(lane * 8 + e + 512) & 1023];
    }
    f32x4 acc0 = {}, acc1 = {}, acc2 = {}, acc3 = {};
    for (int it = 0; it < iters; ++it) {
        acc0 = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            a.v, b.v, acc0, 0, 0, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
        acc1 = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            a.v, b.v, acc1, 0, 0, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
        acc2 = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            a.v, b.v, acc2, 0, 0, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
        acc3 = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            a.v, b.v, acc3, 0, 0, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
    }
    float r = acc0[0] + acc1[1] + acc2[2] + acc3[3];
    if (r == -1.0f) sink[0] = r;
}

// MX-fp4 (OCP e2m1) 32x32x64 block-scaled MFMA: the ~10 PF dense headline
// path (fp4 runs only through mfma_scale_*_f8f6f4 with FMT=4). Tile-verify
// + register ubench. fp4 elements pack two per byte (low nibble = even
// element); the k-map is applied consistently to A and B so GEMM numerics
// pin correctness regardless of the device's internal k-order (see the fp8
// layout note, gpurun_out/r2s12). C/D layout is shape-determined: for
// 32x32 shapes row = (reg&3) + 8*(reg>>2) + 4*(lane>>5), col = lane&31.

extern "C" __global__ void __launch_bounds__(WAVE)
mfma_fp4_tile_gemm_kernel(const unsigned char* __restrict__ A,  // [32][K] e2m1 packed 2/byte, row stride K/2
                          const unsigned char* __restrict__ B,  // [K][32] packed pairs along k: byte (k/2)*32+col
                          float* __restrict__ D, int K) {
    int lane = threadIdx.x & (WAVE - 1);
    int col = lane & 31;
    int kgrp = lane >> 5;  // 0..1, each covering 32 k's
    f32x16 acc = {};
    for (int k0 = 0; k0 < K; k0 += 64) {
        union {
            i32x8 v;
            unsigned char b[32];
        } a, bb;
        for (int e = 0; e < 32; ++e) a.b[e] = 0, bb.b[e] = 0;
        // lane covers k = k0 + kgrp*32 + j, j = 0..31; packed two per byte
        for (int j = 0; j < 32; ++j) {
            int k = k0 + kgrp * 32 + j;
            // A[row=col][k]: row-major, 2 elems/byte
            unsigned char av = A[(size_t)col * (K / 2) + k / 2];
            unsigned char an = (k & 1) ? (av >> 4) : (av & 0xF);
            // B[k][col]: packed along k: byte index (k/2)*32 + col
            unsigned char bv = B[(size_t)(k / 2) * 32 + col];
            unsigned char bn = (k & 1) ? (bv >> 4) : (bv & 0xF);
            if (j & 1) {
                a.b[j / 2] |= an << 4;
                bb.b[j / 2] |= bn << 4;
            } else {
                a.b[j / 2] |= an;
                bb.b[j / 2] |= bn;
            }
        }
        acc = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
            a.v, bb.v, acc, 4 /*cbsz=fp4*/, 4 /*blgp=fp4*/,
            0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
    }
#pragma unroll
    for (int r = 0; r < 16; ++r) {
        int row = (r & 3) + 8 * (r >> 2) + 4 * kgrp;
        D[row * 32 + col] = acc[r];
    }
}

// MX-SCALED fp8 tile verify: REAL per-block E8M0 scales (block = 32
// elements). Scale-operand lane layout was pinned empirically with
// mfma_scale_probe_kernel (profiles/r2: scale_probe_w2.txt): the scale
// byte of lane idx+16*g (byte 0, OPSEL=0) covers the two 16-element
// k-chunks {0,2}/{4,6}/{1,3}/{5,7} (g=0..3) OF THE REGISTER SLOTS, i.e.
// hardware scale blocks are NOT the contiguous 32 elements a lane holds
// under the naive slot->k identity. Placing logical chunk CH[s] at slot
// s with CH = [0,4,1,5,2,6,3,7] (same permutation for A and B, so the
// dot product is unchanged) makes scale group g cover exactly the
// contiguous logical block k in [32g, 32g+32): standard MX-32 blocks.
extern "C" __global__ void __launch_bounds__(WAVE)
mfma_fp8_scaled_tile_kernel(const unsigned char* __restrict__ A,   // [16][K] e4m3
                            const unsigned char* __restrict__ B,   // [K][16] e4m3
                            const unsigned char* __restrict__ SA,  // [16][K/32] e8m0
                            const unsigned char* __restrict__ SB,  // [16][K/32] e8m0 (per col)
                            float* __restrict__ D, int K) {
    int lane = threadIdx.x & (WAVE - 1);
    int row = lane & 15;       // A row for a-slices; B column for b-slices
    int kgrp = lane >> 4;      // scale group g: logical MX block [32g, 32g+32)
    const int CH[8] = {0, 4, 1, 5, 2, 6, 3, 7};
    f32x4 acc = {};
    for (int k0 = 0; k0 < K; k0 += 128) {
        union {
            i32x8 v;
            unsigned char b[32];
        } a, bb;
        for (int e = 0; e < 32; ++e) {
            int slot = kgrp * 2 + (e >> 4);
            int k = k0 + CH[slot] * 16 + (e & 15);
            a.b[e] = A[row * K + k];
            bb.b[e] = B[k * 16 + row];
        }
        int blk = k0 / 32 + kgrp;
        int sa = SA[row * (K / 32) + blk];
        int sb = SB[row * (K / 32) + blk];  // col == row lane mapping
        acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            a.v, bb.v, acc, 0, 0, 0, sa, 0, sb);
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        int out_row = kgrp * 4 + r;
        D[out_row * 16 + row] = acc[r];
    }
}

// Scale-operand layout probe: A = all-ones e4m3, B[k][j] = 1 iff
// k/32 == j%4 (so D[i][j] = 32 * effective_scale(row i, block j%4)).
// For probe index L (0..63): lane L supplies scale byte 0x80 (2.0), all
// other lanes 0x7F (1.0) — first 64 probes perturb scale_a, next 64
// perturb scale_b. Dout[L][16][16] reveals exactly which (row, block)
// each lane's scale byte controls.
extern "C" __global__ void __launch_bounds__(WAVE)
mfma_scale_probe_kernel(float* __restrict__ Dout) {
    int lane = threadIdx.x & (WAVE - 1);
    int row = lane & 15;
    int kgrp = lane >> 4;
    union { i32x8 v; unsigned char b[32]; } a, bb;
    for (int e = 0; e < 32; ++e) a.b[e] = 0x38;  // e4m3 1.0
    for (int e = 0; e < 32; ++e) {
        int k = kgrp * 32 + e;   // this lane's B rows: col = row
        // B[k][col] = 2^-(k/16): every chunk contributes a distinguishable
        // weight to every column, so multi-chunk scale coverage decodes
        // from the delta (16-chunk ch adds 16 * 2^-ch when doubled)
        int ch = k / 16;   // e4m3 2^-ch; ch==7 needs the denormal 0x04
        bb.b[e] = (unsigned char)(ch < 7 ? 0x38 - 8 * ch : 0x04);
    }
    // probes 0..127 perturb one scale byte; probe 128 is the baseline
    for (int L = 0; L < 129; ++L) {
        int sa = (L < 64 && lane == L) ? 0x80 : 0x7F;
        int sb = (L >= 64 && L < 128 && lane == (L - 64)) ? 0x80 : 0x7F;
        f32x4 acc = {};
        acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            a.v, bb.v, acc, 0, 0, 0, sa, 0, sb);
#pragma unroll
        for (int r = 0; r < 4; ++r)
            Dout[L * 256 + (kgrp * 4 + r) * 16 + row] = acc[r];
    }
}

// fp4 tile GEMM with REAL per-block MX scales: the fp4 scale layout is
// naive (lane idx+32*kgrp scales its own k in [32*kgrp, 32*kgrp+32) —
// probe profiles/r2/fp4_scale_probe.txt), so sa = SA[row][k0/32+kgrp].
extern "C" __global__ void __launch_bounds__(WAVE)
mfma_fp4_scaled_tile_kernel(const unsigned char* __restrict__ A,   // [32][K/2] packed
                            const unsigned char* __restrict__ B,   // [K/2][32] packed along k
                            const unsigned char* __restrict__ SA,  // [32][K/32] e8m0
                            const unsigned char* __restrict__ SB,  // [32][K/32] e8m0 per col
                            float* __restrict__ D, int K) {
    int lane = threadIdx.x & (WAVE - 1);
    int col = lane & 31;
    int kgrp = lane >> 5;
    f32x16 acc = {};
    for (int k0 = 0; k0 < K; k0 += 64) {
        union { i32x8 v; unsigned char b[32]; } a, bb;
        for (int e = 0; e < 32; ++e) a.b[e] = 0, bb.b[e] = 0;
        for (int j = 0; j < 32; ++j) {
            int k = k0 + kgrp * 32 + j;
            unsigned char av = A[(size_t)col * (K / 2) + k / 2];
            unsigned char an = (k & 1) ? (av >> 4) : (av & 0xF);
            unsigned char bv = B[(size_t)(k / 2) * 32 + col];
            unsigned char bn = (k & 1) ? (bv >> 4) : (bv & 0xF);
            if (j & 1) {
                a.b[j / 2] |= an << 4;
                bb.b[j / 2] |= bn << 4;
            } else {
                a.b[j / 2] |= an;
                bb.b[j / 2] |= bn;
            }
        }
        int blk = k0 / 32 + kgrp;
        int sa = SA[(size_t)col * (K / 32) + blk];
        int sb = SB[(size_t)col * (K / 32) + blk];
        acc = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
            a.v, bb.v, acc, 4, 4, 0, sa, 0, sb);
    }
#pragma unroll
    for (int r = 0; r < 16; ++r) {
        int row = (r & 3) + 8 * (r >> 2) + 4 * kgrp;
        D[row * 32 + col] = acc[r];
    }
}

// Scale-operand layout probe for the fp4 32x32x64 shape: A = all 1.0
// e2m1; B[k][col] = w(k/16) with w = {4,2,1,0.5} so every 16-chunk
// contributes a distinguishable weight (delta of a doubled chunk ch is
// 16*w(ch) in {64,32,16,8}). Probes 0..63 double SA lane L's byte 0,
// 64..127 double SB lane L-64's, probe 128 is the baseline.
extern "C" __global__ void __launch_bounds__(WAVE)
mfma_fp4_scale_probe_kernel(float* __restrict__ Dout) {
    int lane = threadIdx.x & (WAVE - 1);
    int kgrp = lane >> 5;
    union { i32x8 v; unsigned char b[32]; } a, bb;
    const unsigned char W[4] = {0x6, 0x4, 0x2, 0x1};  // e2m1 4,2,1,0.5
    for (int e = 0; e < 32; ++e) a.b[e] = 0, bb.b[e] = 0;
    for (int j = 0; j < 32; j += 2) {
        int k = kgrp * 32 + j;
        a.b[j / 2] = 0x22;  // two 1.0 nibbles
        bb.b[j / 2] = (unsigned char)(W[k / 16] | (W[(k + 1) / 16] << 4));
    }
    int col = lane & 31;
    for (int L = 0; L < 129; ++L) {
        int sa = (L < 64 && lane == L) ? 0x80 : 0x7F;
        int sb = (L >= 64 && L < 128 && lane == (L - 64)) ? 0x80 : 0x7F;
        f32x16 acc = {};
        acc = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
            a.v, bb.v, acc, 4, 4, 0, sa, 0, sb);
#pragma unroll
        for (int r = 0; r < 16; ++r) {
            int row = (r & 3) + 8 * (r >> 2) + 4 * kgrp;
            Dout[L * 1024 + row * 32 + col] = acc[r];
        }
    }
}

// Register-resident fp4 issue-rate ubench (4 independent f32x16 accumulators)
extern "C" __global__ void __launch_bounds__(PROBE_BLOCK)
mfma_fp4_loop_kernel(const int* __restrict__ seed, float* __restrict__ sink,
                     int iters) {
    int lane = threadIdx.x & (WAVE - 1);
    union { i32x8 v; int i[8]; } a, b;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
        a.i[e] = seed[(lane * 8 + e) & 1023];
        b.i[e] = seed[(lane * 8 + e + 512) & 1023];
    }
    f32x16 acc0 = {}, acc1 = {}, acc2 = {}, acc3 = {};
    for (int it = 0; it < iters; ++it) {
        acc0 = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
            a.v, b.v, acc0, 4, 4, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
        acc1 = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
            a.v, b.v, acc1, 4, 4, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
        acc2 = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
            a.v, b.v, acc2, 4, 4, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
        acc3 = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
            a.v, b.v, acc3, 4, 4, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
    }
    float r = acc0[0] + acc1[1] + acc2[2] + acc3[3];
    if (r == -1.0f) sink[0] = r;
}

// ---------------------------------------------------------------------------
// xGMI p2p pull probe: read from a peer GPU's buffer (mapped via
// hipDeviceEnablePeerAccess) into local HBM. Bandwidth is bound by the xGMI
// links to that peer (~153 GB/s per link).
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(PROBE_BLOCK)
p2p_read_kernel(float4v* __restrict__ local_dst,
                const float4v* __restrict__ peer_src, long n_vec) {
    long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (long i = idx; i < n_vec; i += stride) local_dst[i] = peer_src[i];
}

// Pull-reduce: local += peer (the building block of the fabric all-reduce
// probe: rank r pulls each peer's shard and reduces, then peers pull the
// result — bandwidth-optimal on a full xGMI mesh where every pair has a
// direct link, unlike a ring tuned for NVSwitch).
extern "C" __global__ void __launch_bounds__(PROBE_BLOCK)
p2p_reduce_kernel(float4v* __restrict__ local_acc,
                  const float4v* __restrict__ peer_src, long n_vec) {
    long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
    long stride = (long)gridDim.x * blockDim.x;
    for (long i = idx; i < n_vec; i += stride) {
        float4v v = peer_src[i];
        local_acc[i] += v;
    }
}
