// LDS-staged bf16 GEMM probe for gfx950 (CDNA4) — the "MFMA-saturating,
// LDS-staged" fabric-probe kernel from the BASELINE north star, and a
// realistic matrix-core workload (vs the register-resident mfma loop).
//
// Structure (cdna_hip_programming.md §5, measured ladder step 3):
//  * 128x128 output tile per block, 256 threads = 4 waves, each wave a
//    64x64 sub-tile = 4x4 fragments of v_mfma_f32_16x16x32_bf16;
//  * K-loop in BK=32 steps, A/B tiles staged through LDS with the gfx950
//    16-byte async copy `global_load_lds` (lane-linear LDS images), double
//    buffered so the next tile's DMA overlaps the current tile's MFMAs;
//  * B is consumed K-major (caller passes B^T, the usual inference weight
//    layout) so both operands stage with coalesced 16 B chunks.
//
// C/D fragment mapping (guide §3): col = lane&15, row = (lane>>4)*4 + r.
// A/B input mapping: lane l holds elem k = (l>>4)*8 + e of row/col (l&15).

#include <hip/hip_runtime.h>

#define WAVE 64
#define BM 128
#define BN 128
#define THREADS 256

typedef short bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

// LDS tile images are row-major [BM][BK] (A) and [BN][BK] (Bt), 2 B/elem:
// row r occupies 64 B = 4 chunks of 16 B -> chunk c of row r sits at
// (r*4 + c)*16 bytes. glds writes wave-uniform base + lane*16, so chunk
// index == wave*64 + lane gives a lane-linear image. The matching GLOBAL
// address for chunk (r, c) is &src[(row0 + r) * ld + k0 + c*8].

template <int BK>
__device__ __forceinline__ void gemm_bf16_128_body(
    const short* __restrict__ A,   // [M][K] row-major bf16
    const short* __restrict__ Bt,  // [N][K] row-major bf16
    float* __restrict__ C,         // [M][N] row-major f32
    int M, int N, int K) {
    __shared__ short lds[2 * (BM * BK + BN * BK)];  // 2 x (8 KiB + 8 KiB)
    // one __shared__ object only (guide §5 trap 4a); buffer offsets:
    const int HALF = BM * BK + BN * BK;
    auto ldsA = [&](int buf) -> short* { return lds + buf * HALF; };
    auto ldsB = [&](int buf) -> short* { return lds + buf * HALF + BM * BK; };

    // NOTE on XCD-aware blockIdx remapping: the bijective per-XCD-span
    // remap was tried here and MEASURED SLOWER at 8192^3 (843 -> 646 TF):
    // a contiguous 512-tile span per XCD walks 16 MB of A rows, thrashing
    // the 4 MiB per-XCD L2, while the natural row-major dispatch already
    // gives all 8 XCDs the same A tile-row (L2-resident). Kept linear.
    const int tiles_n = (N + BN - 1) / BN;
    const int tile_m = blockIdx.x / tiles_n;
    const int tile_n = blockIdx.x % tiles_n;
    const int m0 = tile_m * BM;
    const int n0 = tile_n * BN;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;          // 4 waves: 2x2 sub-tiles of 64x64
    const int wr = (wid >> 1) * 64;      // wave row offset in tile
    const int wc = (wid & 1) * 64;       // wave col offset in tile

    // Stage one BK-deep pair of tiles into buffer `buf` via glds.
    // Each thread issues 2 chunks per operand: chunk = phase*256 + tid.
    // chunks per operand tile: BM rows x (BK/8) 16B chunks
    constexpr int CHUNKS = BM * (BK / 8);
    constexpr int PHASES = CHUNKS / THREADS;
    auto stage = [&](int buf, int k0) {
        short* la = ldsA(buf);
        short* lb = ldsB(buf);
#pragma unroll
        for (int phase = 0; phase < PHASES; ++phase) {
            int chunk = phase * THREADS + tid;
            int r = chunk / (BK / 8);               // row in tile
            int c = chunk % (BK / 8);               // 16B chunk in row
            const short* ga = &A[(size_t)(m0 + r) * K + k0 + c * 8];
            // glds: LDS dest = wave-uniform base + lane*16
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)ga,
                (__attribute__((address_space(3))) void*)(la + (size_t)(phase * THREADS + wid * WAVE) * 8),
                16, 0, 0);
            const short* gb = &Bt[(size_t)(n0 + r) * K + k0 + c * 8];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gb,
                (__attribute__((address_space(3))) void*)(lb + (size_t)(phase * THREADS + wid * WAVE) * 8),
                16, 0, 0);
        }
    };

    f32x4 acc[4][4] = {};
    const int kg = (lane >> 4) * 8;  // this lane's K sub-offset

    stage(0, 0);
    __syncthreads();  // drain buffer 0's DMA (vmcnt(0) inside the barrier)
    for (int k0 = 0; k0 < K; k0 += BK) {
        const int buf = (k0 / BK) & 1;
        // issue next tile's DMA BEFORE computing: it runs under the MFMAs
        if (k0 + BK < K) stage(buf ^ 1, k0 + BK);

        const short* la = ldsA(buf);
        const short* lb = ldsB(buf);
#pragma unroll
        for (int ks = 0; ks < BK / 32; ++ks) {
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                const int ar = wr + i * 16 + (lane & 15);
                bf16x8 a = *(const bf16x8*)&la[ar * BK + ks * 32 + kg];
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    const int bc = wc + j * 16 + (lane & 15);
                    bf16x8 b = *(const bf16x8*)&lb[bc * BK + ks * 32 + kg];
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[i][j], 0, 0, 0);
                }
            }
        }
        // one barrier per K-step: drains the prefetch DMA (vmcnt(0)) and
        // guarantees every wave finished reading `buf` before the next
        // iteration's stage overwrites it
        __syncthreads();
    }

    // epilogue: col = lane&15, row = (lane>>4)*4 + r
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = m0 + wr + i * 16 + (lane >> 4) * 4 + r;
                int col = n0 + wc + j * 16 + (lane & 15);
                if (row < M && col < N) C[(size_t)row * N + col] = acc[i][j][r];
            }
        }
    }
}

extern "C" __global__ void __launch_bounds__(THREADS)
gemm_bf16_128_kernel(const short* A, const short* Bt, float* C, int M, int N, int K) {
    gemm_bf16_128_body<32>(A, Bt, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(THREADS)
gemm_bf16_128_bk64_kernel(const short* A, const short* Bt, float* C, int M, int N, int K) {
    gemm_bf16_128_body<64>(A, Bt, C, M, N, K);
}

// ---------------------------------------------------------------------------
// 32x32x16 variant: identical 128x128 tile and glds staging, but each wave's
// 64x64 sub-tile is 2x2 fragments of v_mfma_f32_32x32x16_bf16 instead of 4x4
// of 16x16x32. One a/b fragment pair feeds a 32x32x16 MFMA (32768 flops)
// instead of 16x16x32 (16384), halving the LDS read bytes per FLOP.
//
// MEASURED (MI355X): numerics exact (layout below verified vs torch fp32),
// but NOT faster — 849 TF @4096^3 / 828 @8192^3 (BK=32) vs 856/911 for the
// 16x16x32 tiling, and the BK=64 instantiation collapses to ~570-600 TF
// (16 f32 accumulators x 4 fragments + the deeper unroll overflows the VGPR
// budget the 16x16x32 shape fits in). Conclusion: the 16x16x32 kernel is
// not LDS-read-bound at this tile size; kept selectable (bk=232/264) as a
// measured data point, default dispatch unchanged.
//
// Layouts (cdna_hip_programming.md §3, mfma_f32_32x32x16_bf16):
//   A/B: lane l holds elem k = (l>>5)*8 + e of row/col (l&31), e in 0..7
//   C/D: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5), reg in [0,16)
// ---------------------------------------------------------------------------

typedef float f32x16 __attribute__((ext_vector_type(16)));

template <int BK>
__device__ __forceinline__ void gemm_bf16_128_mfma32_body(
    const short* __restrict__ A,   // [M][K] row-major bf16
    const short* __restrict__ Bt,  // [N][K] row-major bf16
    float* __restrict__ C,         // [M][N] row-major f32
    int M, int N, int K) {
    __shared__ short lds[2 * (BM * BK + BN * BK)];
    const int HALF = BM * BK + BN * BK;
    auto ldsA = [&](int buf) -> short* { return lds + buf * HALF; };
    auto ldsB = [&](int buf) -> short* { return lds + buf * HALF + BM * BK; };

    const int tiles_n = (N + BN - 1) / BN;
    const int tile_m = blockIdx.x / tiles_n;
    const int tile_n = blockIdx.x % tiles_n;
    const int m0 = tile_m * BM;
    const int n0 = tile_n * BN;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int wr = (wid >> 1) * 64;
    const int wc = (wid & 1) * 64;

    constexpr int CHUNKS = BM * (BK / 8);
    constexpr int PHASES = CHUNKS / THREADS;
    auto stage = [&](int buf, int k0) {
        short* la = ldsA(buf);
        short* lb = ldsB(buf);
#pragma unroll
        for (int phase = 0; phase < PHASES; ++phase) {
            int chunk = phase * THREADS + tid;
            int r = chunk / (BK / 8);
            int c = chunk % (BK / 8);
            const short* ga = &A[(size_t)(m0 + r) * K + k0 + c * 8];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)ga,
                (__attribute__((address_space(3))) void*)(la + (size_t)(phase * THREADS + wid * WAVE) * 8),
                16, 0, 0);
            const short* gb = &Bt[(size_t)(n0 + r) * K + k0 + c * 8];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gb,
                (__attribute__((address_space(3))) void*)(lb + (size_t)(phase * THREADS + wid * WAVE) * 8),
                16, 0, 0);
        }
    };

    f32x16 acc[2][2] = {};
    const int kg = (lane >> 5) * 8;  // this lane's K sub-offset (2 groups of 8)
    const int rl = lane & 31;        // this lane's row/col within the fragment

    stage(0, 0);
    __syncthreads();
    for (int k0 = 0; k0 < K; k0 += BK) {
        const int buf = (k0 / BK) & 1;
        if (k0 + BK < K) stage(buf ^ 1, k0 + BK);

        const short* la = ldsA(buf);
        const short* lb = ldsB(buf);
#pragma unroll
        for (int ks = 0; ks < BK / 16; ++ks) {
            bf16x8 a[2], b[2];
#pragma unroll
            for (int i = 0; i < 2; ++i)
                a[i] = *(const bf16x8*)&la[(wr + i * 32 + rl) * BK + ks * 16 + kg];
#pragma unroll
            for (int j = 0; j < 2; ++j)
                b[j] = *(const bf16x8*)&lb[(wc + j * 32 + rl) * BK + ks * 16 + kg];
#pragma unroll
            for (int i = 0; i < 2; ++i)
#pragma unroll
                for (int j = 0; j < 2; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        a[i], b[j], acc[i][j], 0, 0, 0);
        }
        __syncthreads();
    }

    // epilogue: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
#pragma unroll
    for (int i = 0; i < 2; ++i) {
#pragma unroll
        for (int j = 0; j < 2; ++j) {
#pragma unroll
            for (int reg = 0; reg < 16; ++reg) {
                int row = m0 + wr + i * 32 + (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
                int col = n0 + wc + j * 32 + rl;
                if (row < M && col < N) C[(size_t)row * N + col] = acc[i][j][reg];
            }
        }
    }
}

extern "C" __global__ void __launch_bounds__(THREADS)
gemm_bf16_128_mfma32_kernel(const short* A, const short* Bt, float* C, int M, int N, int K) {
    gemm_bf16_128_mfma32_body<32>(A, Bt, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(THREADS)
gemm_bf16_128_mfma32_bk64_kernel(const short* A, const short* Bt, float* C, int M, int N, int K) {
    gemm_bf16_128_mfma32_body<64>(A, Bt, C, M, N, K);
}

// ---------------------------------------------------------------------------
// Pipelined variant: 3 LDS buffers, prefetch depth 2, counted s_waitcnt and a
// RAW s_barrier. __syncthreads()'s fence emits vmcnt(0) while a glds is in
// flight, so the 2-buffer kernel above drains its own prefetch DMA at every
// barrier; keeping ONE tile in flight across the barrier instead
// (guide §6: 3-buf span, counted vmcnt, raw barrier) removes that stall.
//
// Synchronization argument (per wave, per K-step k; S = glds issued per
// stage() per thread):
//   1. s_waitcnt vmcnt(S): my stage(k) writes are done; stage(k+1)'s S may
//      still be in flight (vmcnt retires in order).
//   2. s_waitcnt lgkmcnt(0); s_barrier: every wave has individually waited
//      for ITS share of stage(k), so after the barrier the whole tile k is
//      readable; and every wave has finished computing step k-1, so buffer
//      (k+2)%3 == (k-1)%3 is free to overwrite.
//   3. issue stage(k+2) — lands on the VM counter BEHIND stage(k+1).
//   4. compute step k from buffer k%3.
// ---------------------------------------------------------------------------

template <int BK>
__device__ __forceinline__ void gemm_bf16_128_pipe_body(
    const short* __restrict__ A,   // [M][K] row-major bf16
    const short* __restrict__ Bt,  // [N][K] row-major bf16
    float* __restrict__ C,         // [M][N] row-major f32
    int M, int N, int K) {
    __shared__ short lds[3 * (BM * BK + BN * BK)];
    const int HALF = BM * BK + BN * BK;
    auto ldsA = [&](int buf) -> short* { return lds + buf * HALF; };
    auto ldsB = [&](int buf) -> short* { return lds + buf * HALF + BM * BK; };

    const int tiles_n = (N + BN - 1) / BN;
    const int tile_m = blockIdx.x / tiles_n;
    const int tile_n = blockIdx.x % tiles_n;
    const int m0 = tile_m * BM;
    const int n0 = tile_n * BN;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int wr = (wid >> 1) * 64;
    const int wc = (wid & 1) * 64;

    constexpr int CHUNKS = BM * (BK / 8);
    constexpr int PHASES = CHUNKS / THREADS;
    constexpr int S = PHASES * 2;  // glds per thread per stage (A+B)
    auto stage = [&](int buf, int k0) {
        short* la = ldsA(buf);
        short* lb = ldsB(buf);
#pragma unroll
        for (int phase = 0; phase < PHASES; ++phase) {
            int chunk = phase * THREADS + tid;
            int r = chunk / (BK / 8);
            int c = chunk % (BK / 8);
            const short* ga = &A[(size_t)(m0 + r) * K + k0 + c * 8];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)ga,
                (__attribute__((address_space(3))) void*)(la + (size_t)(phase * THREADS + wid * WAVE) * 8),
                16, 0, 0);
            const short* gb = &Bt[(size_t)(n0 + r) * K + k0 + c * 8];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gb,
                (__attribute__((address_space(3))) void*)(lb + (size_t)(phase * THREADS + wid * WAVE) * 8),
                16, 0, 0);
        }
    };

    f32x4 acc[4][4] = {};
    const int kg = (lane >> 4) * 8;
    const int steps = K / BK;

    stage(0, 0);
    if (steps > 1) stage(1, BK);
    for (int s = 0; s < steps; ++s) {
        const int buf = s % 3;
        // 1. my stage(s) done; stage(s+1) (if issued) may stay in flight
        if (s + 1 < steps)
            asm volatile("s_waitcnt vmcnt(%0)" ::"n"(S) : "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        // 2. align waves; buffer (s+2)%3 is now reusable
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        // 3. prefetch two tiles ahead
        if (s + 2 < steps) stage((s + 2) % 3, (s + 2) * BK);

        const short* la = ldsA(buf);
        const short* lb = ldsB(buf);
#pragma unroll
        for (int ks = 0; ks < BK / 32; ++ks) {
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                const int ar = wr + i * 16 + (lane & 15);
                bf16x8 a = *(const bf16x8*)&la[ar * BK + ks * 32 + kg];
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    const int bc = wc + j * 16 + (lane & 15);
                    bf16x8 b = *(const bf16x8*)&lb[bc * BK + ks * 32 + kg];
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[i][j], 0, 0, 0);
                }
            }
        }
    }
    // final barrier not needed: each wave only writes its own C fragments

#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = m0 + wr + i * 16 + (lane >> 4) * 4 + r;
                int col = n0 + wc + j * 16 + (lane & 15);
                if (row < M && col < N) C[(size_t)row * N + col] = acc[i][j][r];
            }
        }
    }
}

extern "C" __global__ void __launch_bounds__(THREADS)
gemm_bf16_128_pipe_kernel(const short* A, const short* Bt, float* C, int M, int N, int K) {
    gemm_bf16_128_pipe_body<32>(A, Bt, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(THREADS)
gemm_bf16_128_pipe_bk64_kernel(const short* A, const short* Bt, float* C, int M, int N, int K) {
    gemm_bf16_128_pipe_body<64>(A, Bt, C, M, N, K);
}

// ---------------------------------------------------------------------------
// Depth-2 pipeline: 4 LDS buffers, TWO tiles in flight across each barrier
// (wait vmcnt(2S) in steady state). 64 KiB LDS -> 2 workgroups/CU (vs 3 for
// the 3-buffer kernel): measures whether deeper DMA overlap buys more than
// the occupancy it costs.
// ---------------------------------------------------------------------------

template <int BK>
__device__ __forceinline__ void gemm_bf16_128_pipe2_body(
    const short* __restrict__ A, const short* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K) {
    __shared__ short lds[4 * (BM * BK + BN * BK)];
    const int HALF = BM * BK + BN * BK;
    auto ldsA = [&](int buf) -> short* { return lds + buf * HALF; };
    auto ldsB = [&](int buf) -> short* { return lds + buf * HALF + BM * BK; };

    const int tiles_n = (N + BN - 1) / BN;
    const int tile_m = blockIdx.x / tiles_n;
    const int tile_n = blockIdx.x % tiles_n;
    const int m0 = tile_m * BM;
    const int n0 = tile_n * BN;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int wr = (wid >> 1) * 64;
    const int wc = (wid & 1) * 64;

    constexpr int CHUNKS = BM * (BK / 8);
    constexpr int PHASES = CHUNKS / THREADS;
    constexpr int S = PHASES * 2;
    auto stage = [&](int buf, int k0) {
        short* la = ldsA(buf);
        short* lb = ldsB(buf);
#pragma unroll
        for (int phase = 0; phase < PHASES; ++phase) {
            int chunk = phase * THREADS + tid;
            int r = chunk / (BK / 8);
            int c = chunk % (BK / 8);
            const short* ga = &A[(size_t)(m0 + r) * K + k0 + c * 8];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)ga,
                (__attribute__((address_space(3))) void*)(la + (size_t)(phase * THREADS + wid * WAVE) * 8),
                16, 0, 0);
            const short* gb = &Bt[(size_t)(n0 + r) * K + k0 + c * 8];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gb,
                (__attribute__((address_space(3))) void*)(lb + (size_t)(phase * THREADS + wid * WAVE) * 8),
                16, 0, 0);
        }
    };

    f32x4 acc[4][4] = {};
    const int kg = (lane >> 4) * 8;
    const int steps = K / BK;

    stage(0, 0);
    if (steps > 1) stage(1, BK);
    if (steps > 2) stage(2, 2 * BK);
    for (int s = 0; s < steps; ++s) {
        const int buf = s & 3;
        if (s + 2 < steps)
            asm volatile("s_waitcnt vmcnt(%0)" ::"n"(2 * S) : "memory");
        else if (s + 1 < steps)
            asm volatile("s_waitcnt vmcnt(%0)" ::"n"(S) : "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        if (s + 3 < steps) stage((s + 3) & 3, (s + 3) * BK);

        const short* la = ldsA(buf);
        const short* lb = ldsB(buf);
#pragma unroll
        for (int ks = 0; ks < BK / 32; ++ks) {
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                const int ar = wr + i * 16 + (lane & 15);
                bf16x8 a = *(const bf16x8*)&la[ar * BK + ks * 32 + kg];
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    const int bc = wc + j * 16 + (lane & 15);
                    bf16x8 b = *(const bf16x8*)&lb[bc * BK + ks * 32 + kg];
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[i][j], 0, 0, 0);
                }
            }
        }
    }

#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = m0 + wr + i * 16 + (lane >> 4) * 4 + r;
                int col = n0 + wc + j * 16 + (lane & 15);
                if (row < M && col < N) C[(size_t)row * N + col] = acc[i][j][r];
            }
        }
    }
}

// Split-K variant of the depth-2 128x128 kernel for SMALL shapes: at
// 2048^3 the plain kernel launches only 256 WGs (chip barely covered) and
// 1024^3 only 64. Each workgroup computes a partial product over one of
// `ksplit` K-slices and accumulates into C with hardware f32 atomic adds
// (unsafeAtomicAdd -> global_atomic_add_f32); the host zeroes C first.
// Requires (K/ksplit) % BK == 0.
template <int BK>
__device__ __forceinline__ void gemm_bf16_splitk_body(
    const short* __restrict__ A, const short* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K, int ksplit) {
    __shared__ short lds[4 * (BM * BK + BN * BK)];
    const int HALF = BM * BK + BN * BK;
    auto ldsA = [&](int buf) -> short* { return lds + buf * HALF; };
    auto ldsB = [&](int buf) -> short* { return lds + buf * HALF + BM * BK; };

    const int tiles_n = (N + BN - 1) / BN;
    const int tiles_mn = ((M + BM - 1) / BM) * tiles_n;
    const int tile = blockIdx.x % tiles_mn;   // tile-major: slices of one
    const int slice = blockIdx.x / tiles_mn;  // tile land on distinct CUs
    const int tile_m = tile / tiles_n;
    const int tile_n = tile % tiles_n;
    const int m0 = tile_m * BM;
    const int n0 = tile_n * BN;
    const int kper = K / ksplit;
    const int kbase = slice * kper;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int wr = (wid >> 1) * 64;
    const int wc = (wid & 1) * 64;

    constexpr int CHUNKS = BM * (BK / 8);
    constexpr int PHASES = CHUNKS / THREADS;
    constexpr int S = PHASES * 2;
    auto stage = [&](int buf, int k0) {
        short* la = ldsA(buf);
        short* lb = ldsB(buf);
#pragma unroll
        for (int phase = 0; phase < PHASES; ++phase) {
            int chunk = phase * THREADS + tid;
            int r = chunk / (BK / 8);
            int c = chunk % (BK / 8);
            const short* ga = &A[(size_t)(m0 + r) * K + kbase + k0 + c * 8];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)ga,
                (__attribute__((address_space(3))) void*)(la + (size_t)(phase * THREADS + wid * WAVE) * 8),
                16, 0, 0);
            const short* gb = &Bt[(size_t)(n0 + r) * K + kbase + k0 + c * 8];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gb,
                (__attribute__((address_space(3))) void*)(lb + (size_t)(phase * THREADS + wid * WAVE) * 8),
                16, 0, 0);
        }
    };

    f32x4 acc[4][4] = {};
    const int kg = (lane >> 4) * 8;
    const int steps = kper / BK;

    stage(0, 0);
    if (steps > 1) stage(1, BK);
    if (steps > 2) stage(2, 2 * BK);
    for (int s = 0; s < steps; ++s) {
        const int buf = s & 3;
        if (s + 2 < steps)
            asm volatile("s_waitcnt vmcnt(%0)" ::"n"(2 * S) : "memory");
        else if (s + 1 < steps)
            asm volatile("s_waitcnt vmcnt(%0)" ::"n"(S) : "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        if (s + 3 < steps) stage((s + 3) & 3, (s + 3) * BK);

        const short* la = ldsA(buf);
        const short* lb = ldsB(buf);
#pragma unroll
        for (int ks = 0; ks < BK / 32; ++ks) {
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                const int ar = wr + i * 16 + (lane & 15);
                bf16x8 a = *(const bf16x8*)&la[ar * BK + ks * 32 + kg];
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    const int bc = wc + j * 16 + (lane & 15);
                    bf16x8 b = *(const bf16x8*)&lb[bc * BK + ks * 32 + kg];
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[i][j], 0, 0, 0);
                }
            }
        }
    }

#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = m0 + wr + i * 16 + (lane >> 4) * 4 + r;
                int col = n0 + wc + j * 16 + (lane & 15);
                if (row < M && col < N)
                    unsafeAtomicAdd(&C[(size_t)row * N + col], acc[i][j][r]);
            }
        }
    }
}

extern "C" __global__ void __launch_bounds__(THREADS)
gemm_bf16_128_splitk_kernel(const short* A, const short* Bt, float* C,
                            int M, int N, int K, int ksplit) {
    gemm_bf16_splitk_body<32>(A, Bt, C, M, N, K, ksplit);
}

extern "C" __global__ void __launch_bounds__(THREADS)
gemm_bf16_128_pipe2_kernel(const short* A, const short* Bt, float* C, int M, int N, int K) {
    gemm_bf16_128_pipe2_body<32>(A, Bt, C, M, N, K);
}

// Depth-3 pipeline: 5 LDS buffers, THREE tiles in flight across each barrier
// (wait vmcnt(3S) in steady state). 80 KiB LDS -> still 2 workgroups/CU,
// so this isolates pipeline depth from occupancy.
// ---------------------------------------------------------------------------

template <int BK>
__device__ __forceinline__ void gemm_bf16_128_pipe3_body(
    const short* __restrict__ A, const short* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K) {
    __shared__ short lds[5 * (BM * BK + BN * BK)];
    const int HALF = BM * BK + BN * BK;
    auto ldsA = [&](int buf) -> short* { return lds + buf * HALF; };
    auto ldsB = [&](int buf) -> short* { return lds + buf * HALF + BM * BK; };

    const int tiles_n = (N + BN - 1) / BN;
    const int tile_m = blockIdx.x / tiles_n;
    const int tile_n = blockIdx.x % tiles_n;
    const int m0 = tile_m * BM;
    const int n0 = tile_n * BN;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int wr = (wid >> 1) * 64;
    const int wc = (wid & 1) * 64;

    constexpr int CHUNKS = BM * (BK / 8);
    constexpr int PHASES = CHUNKS / THREADS;
    constexpr int S = PHASES * 2;
    auto stage = [&](int buf, int k0) {
        short* la = ldsA(buf);
        short* lb = ldsB(buf);
#pragma unroll
        for (int phase = 0; phase < PHASES; ++phase) {
            int chunk = phase * THREADS + tid;
            int r = chunk / (BK / 8);
            int c = chunk % (BK / 8);
            const short* ga = &A[(size_t)(m0 + r) * K + k0 + c * 8];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)ga,
                (__attribute__((address_space(3))) void*)(la + (size_t)(phase * THREADS + wid * WAVE) * 8),
                16, 0, 0);
            const short* gb = &Bt[(size_t)(n0 + r) * K + k0 + c * 8];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gb,
                (__attribute__((address_space(3))) void*)(lb + (size_t)(phase * THREADS + wid * WAVE) * 8),
                16, 0, 0);
        }
    };

    f32x4 acc[4][4] = {};
    const int kg = (lane >> 4) * 8;
    const int steps = K / BK;

    stage(0, 0);
    if (steps > 1) stage(1, BK);
    if (steps > 2) stage(2, 2 * BK);
    if (steps > 3) stage(3, 3 * BK);
    for (int s = 0; s < steps; ++s) {
        const int buf = s % 5;
        if (s + 3 < steps)
            asm volatile("s_waitcnt vmcnt(%0)" ::"n"(3 * S) : "memory");
        else if (s + 2 < steps)
            asm volatile("s_waitcnt vmcnt(%0)" ::"n"(2 * S) : "memory");
        else if (s + 1 < steps)
            asm volatile("s_waitcnt vmcnt(%0)" ::"n"(S) : "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        if (s + 4 < steps) stage((s + 4) % 5, (s + 4) * BK);

        const short* la = ldsA(buf);
        const short* lb = ldsB(buf);
#pragma unroll
        for (int ks = 0; ks < BK / 32; ++ks) {
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                const int ar = wr + i * 16 + (lane & 15);
                bf16x8 a = *(const bf16x8*)&la[ar * BK + ks * 32 + kg];
#pragma unroll
                for (int j = 0; j < 4; ++j) {
                    const int bc = wc + j * 16 + (lane & 15);
                    bf16x8 b = *(const bf16x8*)&lb[bc * BK + ks * 32 + kg];
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[i][j], 0, 0, 0);
                }
            }
        }
    }

#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = m0 + wr + i * 16 + (lane >> 4) * 4 + r;
                int col = n0 + wc + j * 16 + (lane & 15);
                if (row < M && col < N) C[(size_t)row * N + col] = acc[i][j][r];
            }
        }
    }
}

extern "C" __global__ void __launch_bounds__(THREADS)
gemm_bf16_128_pipe3_kernel(const short* A, const short* Bt, float* C, int M, int N, int K) {
    gemm_bf16_128_pipe3_body<32>(A, Bt, C, M, N, K);
}

// ---------------------------------------------------------------------------
// Ladder step (round 2): register-hoisted fragments. The pipe2 inner loop
// re-reads each B fragment from LDS for every i (16 b-reads + 4 a-reads per
// 16 MFMAs = 1.25 ds_read_b128 per MFMA). Loading a[4] and b[4] into
// registers once per K-substep cuts LDS reads to 8 per 16 MFMAs
// (0.5/MFMA) at +32 VGPRs — measures whether the 16x16x32 tile is
// ds_read-bound at depth-2 (the round-1 32x32x16 experiment said "not
// LDS-bound" at 1 read/MFMA granularity; this probes the other direction).
// ---------------------------------------------------------------------------

template <int BK>
__device__ __forceinline__ void gemm_bf16_128_pipe2r_body(
    const short* __restrict__ A, const short* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K) {
    __shared__ short lds[4 * (BM * BK + BN * BK)];
    const int HALF = BM * BK + BN * BK;
    auto ldsA = [&](int buf) -> short* { return lds + buf * HALF; };
    auto ldsB = [&](int buf) -> short* { return lds + buf * HALF + BM * BK; };

    const int tiles_n = (N + BN - 1) / BN;
    const int tile_m = blockIdx.x / tiles_n;
    const int tile_n = blockIdx.x % tiles_n;
    const int m0 = tile_m * BM;
    const int n0 = tile_n * BN;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int wr = (wid >> 1) * 64;
    const int wc = (wid & 1) * 64;

    constexpr int CHUNKS = BM * (BK / 8);
    constexpr int PHASES = CHUNKS / THREADS;
    constexpr int S = PHASES * 2;
    auto stage = [&](int buf, int k0) {
        short* la = ldsA(buf);
        short* lb = ldsB(buf);
#pragma unroll
        for (int phase = 0; phase < PHASES; ++phase) {
            int chunk = phase * THREADS + tid;
            int r = chunk / (BK / 8);
            int c = chunk % (BK / 8);
            const short* ga = &A[(size_t)(m0 + r) * K + k0 + c * 8];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)ga,
                (__attribute__((address_space(3))) void*)(la + (size_t)(phase * THREADS + wid * WAVE) * 8),
                16, 0, 0);
            const short* gb = &Bt[(size_t)(n0 + r) * K + k0 + c * 8];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gb,
                (__attribute__((address_space(3))) void*)(lb + (size_t)(phase * THREADS + wid * WAVE) * 8),
                16, 0, 0);
        }
    };

    f32x4 acc[4][4] = {};
    const int kg = (lane >> 4) * 8;
    const int steps = K / BK;

    stage(0, 0);
    if (steps > 1) stage(1, BK);
    if (steps > 2) stage(2, 2 * BK);
    for (int s = 0; s < steps; ++s) {
        const int buf = s & 3;
        if (s + 2 < steps)
            asm volatile("s_waitcnt vmcnt(%0)" ::"n"(2 * S) : "memory");
        else if (s + 1 < steps)
            asm volatile("s_waitcnt vmcnt(%0)" ::"n"(S) : "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        if (s + 3 < steps) stage((s + 3) & 3, (s + 3) * BK);

        const short* la = ldsA(buf);
        const short* lb = ldsB(buf);
#pragma unroll
        for (int ks = 0; ks < BK / 32; ++ks) {
            bf16x8 af[4], bf[4];
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                const int ar = wr + i * 16 + (lane & 15);
                af[i] = *(const bf16x8*)&la[ar * BK + ks * 32 + kg];
            }
#pragma unroll
            for (int j = 0; j < 4; ++j) {
                const int bc = wc + j * 16 + (lane & 15);
                bf[j] = *(const bf16x8*)&lb[bc * BK + ks * 32 + kg];
            }
#pragma unroll
            for (int i = 0; i < 4; ++i)
#pragma unroll
                for (int j = 0; j < 4; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        af[i], bf[j], acc[i][j], 0, 0, 0);
        }
    }

#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = m0 + wr + i * 16 + (lane >> 4) * 4 + r;
                int col = n0 + wc + j * 16 + (lane & 15);
                if (row < M && col < N) C[(size_t)row * N + col] = acc[i][j][r];
            }
}

extern "C" __global__ void __launch_bounds__(THREADS)
gemm_bf16_128_pipe2r_kernel(const short* A, const short* Bt, float* C, int M, int N, int K) {
    gemm_bf16_128_pipe2r_body<32>(A, Bt, C, M, N, K);
}

// ---------------------------------------------------------------------------
// Ladder step (round 2): 256x128 output tile, 512 threads = 8 waves (4x2
// grid of 64x64 wave tiles), register-hoisted fragments, 3-buffer counted
// pipeline. Doubles the A-reuse per staged B tile (each B tile feeds 2x the
// MFMAs) and halves the number of workgroups (L2 tile traffic), at
// 3 x 24 KiB = 72 KiB LDS -> 2 workgroups/CU (16 waves).
// ---------------------------------------------------------------------------

#define BM2 256
#define THREADS2 512

template <int BK>
__device__ __forceinline__ void gemm_bf16_256_body(
    const short* __restrict__ A, const short* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K) {
    __shared__ short lds[3 * (BM2 * BK + BN * BK)];
    const int HALF = BM2 * BK + BN * BK;
    auto ldsA = [&](int buf) -> short* { return lds + buf * HALF; };
    auto ldsB = [&](int buf) -> short* { return lds + buf * HALF + BM2 * BK; };

    const int tiles_n = (N + BN - 1) / BN;
    const int tile_m = blockIdx.x / tiles_n;
    const int tile_n = blockIdx.x % tiles_n;
    const int m0 = tile_m * BM2;
    const int n0 = tile_n * BN;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;          // 8 waves: 4x2 grid of 64x64
    const int wr = (wid >> 1) * 64;
    const int wc = (wid & 1) * 64;

    // A: BM2*(BK/8) = 1024 chunks (2 phases of 512); B: BN*(BK/8) = 512 (1)
    constexpr int ACH = BM2 * (BK / 8);
    constexpr int APH = ACH / THREADS2;
    constexpr int BCH = BN * (BK / 8);
    constexpr int BPH = BCH / THREADS2;
    constexpr int S = APH + BPH;  // glds per thread per stage
    auto stage = [&](int buf, int k0) {
        short* la = ldsA(buf);
        short* lb = ldsB(buf);
#pragma unroll
        for (int phase = 0; phase < APH; ++phase) {
            int chunk = phase * THREADS2 + tid;
            int r = chunk / (BK / 8);
            int c = chunk % (BK / 8);
            const short* ga = &A[(size_t)(m0 + r) * K + k0 + c * 8];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)ga,
                (__attribute__((address_space(3))) void*)(la + (size_t)(phase * THREADS2 + wid * WAVE) * 8),
                16, 0, 0);
        }
#pragma unroll
        for (int phase = 0; phase < BPH; ++phase) {
            int chunk = phase * THREADS2 + tid;
            int r = chunk / (BK / 8);
            int c = chunk % (BK / 8);
            const short* gb = &Bt[(size_t)(n0 + r) * K + k0 + c * 8];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gb,
                (__attribute__((address_space(3))) void*)(lb + (size_t)(phase * THREADS2 + wid * WAVE) * 8),
                16, 0, 0);
        }
    };

    f32x4 acc[4][4] = {};
    const int kg = (lane >> 4) * 8;
    const int steps = K / BK;

    stage(0, 0);
    if (steps > 1) stage(1, BK);
    for (int s = 0; s < steps; ++s) {
        const int buf = s % 3;
        if (s + 1 < steps)
            asm volatile("s_waitcnt vmcnt(%0)" ::"n"(S) : "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        if (s + 2 < steps) stage((s + 2) % 3, (s + 2) * BK);

        const short* la = ldsA(buf);
        const short* lb = ldsB(buf);
#pragma unroll
        for (int ks = 0; ks < BK / 32; ++ks) {
            bf16x8 af[4], bf[4];
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                const int ar = wr + i * 16 + (lane & 15);
                af[i] = *(const bf16x8*)&la[ar * BK + ks * 32 + kg];
            }
#pragma unroll
            for (int j = 0; j < 4; ++j) {
                const int bc = wc + j * 16 + (lane & 15);
                bf[j] = *(const bf16x8*)&lb[bc * BK + ks * 32 + kg];
            }
#pragma unroll
            for (int i = 0; i < 4; ++i)
#pragma unroll
                for (int j = 0; j < 4; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        af[i], bf[j], acc[i][j], 0, 0, 0);
        }
    }

#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = m0 + wr + i * 16 + (lane >> 4) * 4 + r;
                int col = n0 + wc + j * 16 + (lane & 15);
                if (row < M && col < N) C[(size_t)row * N + col] = acc[i][j][r];
            }
}

extern "C" __global__ void __launch_bounds__(THREADS2)
gemm_bf16_256_kernel(const short* A, const short* Bt, float* C, int M, int N, int K) {
    gemm_bf16_256_body<32>(A, Bt, C, M, N, K);
}

// ---------------------------------------------------------------------------
// Generic big-tile body (round-2 ladder): TWRxTWC waves of 64-lane wavefronts
// tile a TBMxTBN output block; each wave computes AI x BJ fragments of
// v_mfma_f32_16x16x32_bf16 with register-hoisted fragments; 3-buffer counted
// pipeline (one tile in flight across each barrier).
//   gemm_bf16_256 (732)   = Big<256,128, 4,2> measured 1055/1187 TF
//   gemm_bf16_256x256     = Big<256,256, 4,2> (wave = 64x128: AI=4, BJ=8)
// (a 512x128 variant was rejected: its B tile has fewer 16 B chunks than
// threads, breaking the per-thread counted-vmcnt discipline; and 512-wide
// tiles underfill the 256-CU chip at 4096^2 output)
// ---------------------------------------------------------------------------

template <int TBM, int TBN, int TWR, int TWC, int BK>
__device__ __forceinline__ void gemm_big_body(
    const short* __restrict__ A, const short* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K) {
    constexpr int NT = TWR * TWC * WAVE;        // threads
    constexpr int WM = TBM / TWR;               // wave tile rows
    constexpr int WN = TBN / TWC;               // wave tile cols
    constexpr int AI = WM / 16;                 // a-fragments per wave
    constexpr int BJ = WN / 16;                 // b-fragments per wave
    __shared__ short lds[3 * (TBM * BK + TBN * BK)];
    const int HALF = TBM * BK + TBN * BK;
    auto ldsA = [&](int buf) -> short* { return lds + buf * HALF; };
    auto ldsB = [&](int buf) -> short* { return lds + buf * HALF + TBM * BK; };

    const int tiles_n = (N + TBN - 1) / TBN;
    const int m0 = (blockIdx.x / tiles_n) * TBM;
    const int n0 = (blockIdx.x % tiles_n) * TBN;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int wr = (wid / TWC) * WM;
    const int wc = (wid % TWC) * WN;

    constexpr int ACH = TBM * (BK / 8);
    constexpr int APH = ACH / NT;
    constexpr int BCH = TBN * (BK / 8);
    constexpr int BPH = BCH / NT;
    static_assert(APH * NT == ACH && BPH * NT == BCH, "phase split");
    constexpr int S = APH + BPH;
    auto stage = [&](int buf, int k0) {
        short* la = ldsA(buf);
        short* lb = ldsB(buf);
#pragma unroll
        for (int phase = 0; phase < APH; ++phase) {
            int chunk = phase * NT + tid;
            int r = chunk / (BK / 8);
            int c = chunk % (BK / 8);
            const short* ga = &A[(size_t)(m0 + r) * K + k0 + c * 8];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)ga,
                (__attribute__((address_space(3))) void*)(la + (size_t)(phase * NT + wid * WAVE) * 8),
                16, 0, 0);
        }
#pragma unroll
        for (int phase = 0; phase < BPH; ++phase) {
            int chunk = phase * NT + tid;
            int r = chunk / (BK / 8);
            int c = chunk % (BK / 8);
            const short* gb = &Bt[(size_t)(n0 + r) * K + k0 + c * 8];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gb,
                (__attribute__((address_space(3))) void*)(lb + (size_t)(phase * NT + wid * WAVE) * 8),
                16, 0, 0);
        }
    };

    f32x4 acc[AI][BJ] = {};
    const int kg = (lane >> 4) * 8;
    const int steps = K / BK;

    stage(0, 0);
    if (steps > 1) stage(1, BK);
    for (int s = 0; s < steps; ++s) {
        const int buf = s % 3;
        if (s + 1 < steps)
            asm volatile("s_waitcnt vmcnt(%0)" ::"n"(S) : "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        if (s + 2 < steps) stage((s + 2) % 3, (s + 2) * BK);

        const short* la = ldsA(buf);
        const short* lb = ldsB(buf);
#pragma unroll
        for (int ks = 0; ks < BK / 32; ++ks) {
            bf16x8 af[AI], bf[BJ];
#pragma unroll
            for (int i = 0; i < AI; ++i) {
                const int ar = wr + i * 16 + (lane & 15);
                af[i] = *(const bf16x8*)&la[ar * BK + ks * 32 + kg];
            }
#pragma unroll
            for (int j = 0; j < BJ; ++j) {
                const int bc = wc + j * 16 + (lane & 15);
                bf[j] = *(const bf16x8*)&lb[bc * BK + ks * 32 + kg];
            }
#pragma unroll
            for (int i = 0; i < AI; ++i)
#pragma unroll
                for (int j = 0; j < BJ; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        af[i], bf[j], acc[i][j], 0, 0, 0);
        }
    }

#pragma unroll
    for (int i = 0; i < AI; ++i)
#pragma unroll
        for (int j = 0; j < BJ; ++j)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = m0 + wr + i * 16 + (lane >> 4) * 4 + r;
                int col = n0 + wc + j * 16 + (lane & 15);
                if (row < M && col < N) C[(size_t)row * N + col] = acc[i][j][r];
            }
}

extern "C" __global__ void __launch_bounds__(512)
gemm_bf16_256x256_kernel(const short* A, const short* Bt, float* C, int M, int N, int K) {
    gemm_big_body<256, 256, 4, 2, 32>(A, Bt, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512)
gemm_bf16_256_bk64_kernel(const short* A, const short* Bt, float* C, int M, int N, int K) {
    gemm_big_body<256, 128, 4, 2, 64>(A, Bt, C, M, N, K);
}

// Depth-2 variant of the generic big-tile body: 4 LDS buffers, TWO tiles in
// flight across each barrier (pipe2-style counted vmcnt). For 256x256 this
// is 4 x 32 KiB = 128 KiB LDS (fits the 160 KiB CU budget at 1 WG/CU).
template <int TBM, int TBN, int TWR, int TWC, int BK, int GROUP = 1, int SWZ = 0>
__device__ __forceinline__ void gemm_big2_body(
    const short* __restrict__ A, const short* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K) {
    constexpr int NT = TWR * TWC * WAVE;
    constexpr int WM = TBM / TWR;
    constexpr int WN = TBN / TWC;
    constexpr int AI = WM / 16;
    constexpr int BJ = WN / 16;
    __shared__ short lds[4 * (TBM * BK + TBN * BK)];
    const int HALF = TBM * BK + TBN * BK;
    auto ldsA = [&](int buf) -> short* { return lds + buf * HALF; };
    auto ldsB = [&](int buf) -> short* { return lds + buf * HALF + TBM * BK; };

    const int tiles_n = (N + TBN - 1) / TBN;
    int tile_m, tile_n;
    if (GROUP > 1) {
        // grouped swizzle for L2 reuse (see gemm_fp8_body)
        const int tiles_m = (M + TBM - 1) / TBM;
        const int per_group = GROUP * tiles_n;
        const int gid = blockIdx.x / per_group;
        const int first_m = gid * GROUP;
        const int gsz = min(GROUP, tiles_m - first_m);
        tile_m = first_m + (blockIdx.x % per_group) % gsz;
        tile_n = (blockIdx.x % per_group) / gsz;
    } else {
        tile_m = blockIdx.x / tiles_n;
        tile_n = blockIdx.x % tiles_n;
    }
    const int m0 = tile_m * TBM;
    const int n0 = tile_n * TBN;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int wr = (wid / TWC) * WM;
    const int wc = (wid % TWC) * WN;

    constexpr int ACH = TBM * (BK / 8);
    constexpr int APH = ACH / NT;
    constexpr int BCH = TBN * (BK / 8);
    constexpr int BPH = BCH / NT;
    static_assert(APH * NT == ACH && BPH * NT == BCH, "phase split");
    constexpr int S = APH + BPH;
    auto stage = [&](int buf, int k0) {
        short* la = ldsA(buf);
        short* lb = ldsB(buf);
#pragma unroll
        for (int phase = 0; phase < APH; ++phase) {
            int chunk = phase * NT + tid;
            int r = chunk / (BK / 8);
            int c = chunk % (BK / 8);
            // SWZ: LDS slot s holds global chunk (r, c ^ (r & CPR-1)) —
            // rows land on different banks so the 16-lane fragment reads
            // (row stride = one bank group) stop conflicting
            if (SWZ) c ^= r & (BK / 8 - 1);
            const short* ga = &A[(size_t)(m0 + r) * K + k0 + c * 8];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)ga,
                (__attribute__((address_space(3))) void*)(la + (size_t)(phase * NT + wid * WAVE) * 8),
                16, 0, 0);
        }
#pragma unroll
        for (int phase = 0; phase < BPH; ++phase) {
            int chunk = phase * NT + tid;
            int r = chunk / (BK / 8);
            int c = chunk % (BK / 8);
            if (SWZ) c ^= r & (BK / 8 - 1);
            const short* gb = &Bt[(size_t)(n0 + r) * K + k0 + c * 8];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gb,
                (__attribute__((address_space(3))) void*)(lb + (size_t)(phase * NT + wid * WAVE) * 8),
                16, 0, 0);
        }
    };

    f32x4 acc[AI][BJ] = {};
    const int kg = (lane >> 4) * 8;
    const int steps = K / BK;

    stage(0, 0);
    if (steps > 1) stage(1, BK);
    if (steps > 2) stage(2, 2 * BK);
    for (int s = 0; s < steps; ++s) {
        const int buf = s & 3;
        if (s + 2 < steps)
            asm volatile("s_waitcnt vmcnt(%0)" ::"n"(2 * S) : "memory");
        else if (s + 1 < steps)
            asm volatile("s_waitcnt vmcnt(%0)" ::"n"(S) : "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        if (s + 3 < steps) stage((s + 3) & 3, (s + 3) * BK);

        const short* la = ldsA(buf);
        const short* lb = ldsB(buf);
#pragma unroll
        for (int ks = 0; ks < BK / 32; ++ks) {
            bf16x8 af[AI], bf[BJ];
#pragma unroll
            for (int i = 0; i < AI; ++i) {
                const int ar = wr + i * 16 + (lane & 15);
                int ch = ks * 4 + (lane >> 4);  // 16B chunk within the row
                if (SWZ) ch ^= ar & (BK / 8 - 1);
                af[i] = *(const bf16x8*)&la[ar * BK + ch * 8];
            }
#pragma unroll
            for (int j = 0; j < BJ; ++j) {
                const int bc = wc + j * 16 + (lane & 15);
                int ch = ks * 4 + (lane >> 4);
                if (SWZ) ch ^= bc & (BK / 8 - 1);
                bf[j] = *(const bf16x8*)&lb[bc * BK + ch * 8];
            }
#pragma unroll
            for (int i = 0; i < AI; ++i)
#pragma unroll
                for (int j = 0; j < BJ; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        af[i], bf[j], acc[i][j], 0, 0, 0);
        }
    }

#pragma unroll
    for (int i = 0; i < AI; ++i)
#pragma unroll
        for (int j = 0; j < BJ; ++j)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = m0 + wr + i * 16 + (lane >> 4) * 4 + r;
                int col = n0 + wc + j * 16 + (lane & 15);
                if (row < M && col < N) C[(size_t)row * N + col] = acc[i][j][r];
            }
}

extern "C" __global__ void __launch_bounds__(512)
gemm_bf16_256x256_d2_kernel(const short* A, const short* Bt, float* C, int M, int N, int K) {
    gemm_big2_body<256, 256, 4, 2, 32>(A, Bt, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512)
gemm_bf16_256x256_d2_swz_kernel(const short* A, const short* Bt, float* C, int M, int N, int K) {
    gemm_big2_body<256, 256, 4, 2, 32, 1, 1>(A, Bt, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512)
gemm_bf16_256x256_d2_g4_kernel(const short* A, const short* Bt, float* C, int M, int N, int K) {
    gemm_big2_body<256, 256, 4, 2, 32, 4>(A, Bt, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512)
gemm_bf16_256x256_d2_g8_kernel(const short* A, const short* Bt, float* C, int M, int N, int K) {
    gemm_big2_body<256, 256, 4, 2, 32, 8>(A, Bt, C, M, N, K);
}

// ---------------------------------------------------------------------------
// MX-fp8 (OCP e4m3, scale=1) GEMM — mfma_scale_f32_16x16x128_f8f6f4 runs at
// 2x the bf16 MFMA rate (the MI355X ~5 PF headline path). Same structure as
// the bf16 big-tile bodies: glds staging, counted-vmcnt 3-buffer pipeline,
// register-hoisted fragments. BK = 128 (one MFMA K-depth per step); LDS
// tile rows are 128 B = 8 x 16 B chunks, elements are 1 B so staged bytes
// per FLOP are HALF the bf16 kernel's.
//
// Fragment k-mapping note (measured, gpurun_out/r2s12/fp8_layout.txt): the
// matmul is invariant to any k-permutation applied consistently to both
// A and B fragments, so the natural extension map (lane kgrp*32 + e,
// 32 contiguous bytes per lane) is used for both operands; with
// per-block scales all = 0x7F (2^0) the MX blocking cannot introduce a
// scale/k mismatch.
// ---------------------------------------------------------------------------

typedef int i32x8 __attribute__((ext_vector_type(8)));
typedef int i32x4g __attribute__((ext_vector_type(4)));

template <int TBM, int TBN, int TWR, int TWC, int GROUP = 1>
__device__ __forceinline__ void gemm_fp8_body(
    const unsigned char* __restrict__ A,   // [M][K] row-major e4m3
    const unsigned char* __restrict__ Bt,  // [N][K] row-major e4m3
    float* __restrict__ C,                 // [M][N] row-major f32
    int M, int N, int K) {
    constexpr int BK = 128;
    constexpr int NT = TWR * TWC * WAVE;
    constexpr int WM = TBM / TWR;
    constexpr int WN = TBN / TWC;
    constexpr int AI = WM / 16;
    constexpr int BJ = WN / 16;
    __shared__ unsigned char lds[3 * (TBM * BK + TBN * BK)];
    const int HALF = TBM * BK + TBN * BK;
    auto ldsA = [&](int buf) -> unsigned char* { return lds + buf * HALF; };
    auto ldsB = [&](int buf) -> unsigned char* { return lds + buf * HALF + TBM * BK; };

    const int tiles_n = (N + TBN - 1) / TBN;
    int tile_m, tile_n;
    if (GROUP > 1) {
        // grouped swizzle (L2 reuse): concurrent blocks cover only GROUP
        // m-tiles, so each B column-panel is reused GROUP times and the A
        // working set stays inside the per-XCD L2 while both kernels are
        // HBM-bound at large sizes
        const int tiles_m = (M + TBM - 1) / TBM;
        const int per_group = GROUP * tiles_n;
        const int gid = blockIdx.x / per_group;
        const int first_m = gid * GROUP;
        const int gsz = min(GROUP, tiles_m - first_m);
        tile_m = first_m + (blockIdx.x % per_group) % gsz;
        tile_n = (blockIdx.x % per_group) / gsz;
    } else {
        tile_m = blockIdx.x / tiles_n;
        tile_n = blockIdx.x % tiles_n;
    }
    const int m0 = tile_m * TBM;
    const int n0 = tile_n * TBN;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int wr = (wid / TWC) * WM;
    const int wc = (wid % TWC) * WN;

    constexpr int ACH = TBM * (BK / 16);  // 16 B chunks per A tile
    constexpr int APH = ACH / NT;
    constexpr int BCH = TBN * (BK / 16);
    constexpr int BPH = BCH / NT;
    static_assert(APH * NT == ACH && BPH * NT == BCH, "phase split");
    constexpr int S = APH + BPH;
    auto stage = [&](int buf, int k0) {
        unsigned char* la = ldsA(buf);
        unsigned char* lb = ldsB(buf);
#pragma unroll
        for (int phase = 0; phase < APH; ++phase) {
            int chunk = phase * NT + tid;
            int r = chunk / (BK / 16);
            int c = chunk % (BK / 16);
            const unsigned char* ga = &A[(size_t)(m0 + r) * K + k0 + c * 16];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)ga,
                (__attribute__((address_space(3))) void*)(la + (size_t)(phase * NT + wid * WAVE) * 16),
                16, 0, 0);
        }
#pragma unroll
        for (int phase = 0; phase < BPH; ++phase) {
            int chunk = phase * NT + tid;
            int r = chunk / (BK / 16);
            int c = chunk % (BK / 16);
            const unsigned char* gb = &Bt[(size_t)(n0 + r) * K + k0 + c * 16];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gb,
                (__attribute__((address_space(3))) void*)(lb + (size_t)(phase * NT + wid * WAVE) * 16),
                16, 0, 0);
        }
    };

    f32x4 acc[AI][BJ] = {};
    const int kg = (lane >> 4) * 32;  // this lane's 32-byte k-slice
    const int steps = K / BK;

    stage(0, 0);
    if (steps > 1) stage(1, BK);
    for (int s = 0; s < steps; ++s) {
        const int buf = s % 3;
        if (s + 1 < steps)
            asm volatile("s_waitcnt vmcnt(%0)" ::"n"(S) : "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        if (s + 2 < steps) stage((s + 2) % 3, (s + 2) * BK);

        const unsigned char* la = ldsA(buf);
        const unsigned char* lb = ldsB(buf);
        i32x8 af[AI], bf[BJ];
#pragma unroll
        for (int i = 0; i < AI; ++i) {
            const int ar = wr + i * 16 + (lane & 15);
            af[i] = *(const i32x8*)&la[ar * BK + kg];
        }
#pragma unroll
        for (int j = 0; j < BJ; ++j) {
            const int bc = wc + j * 16 + (lane & 15);
            bf[j] = *(const i32x8*)&lb[bc * BK + kg];
        }
#pragma unroll
        for (int i = 0; i < AI; ++i)
#pragma unroll
            for (int j = 0; j < BJ; ++j)
                acc[i][j] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
                    af[i], bf[j], acc[i][j], 0, 0, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
    }

#pragma unroll
    for (int i = 0; i < AI; ++i)
#pragma unroll
        for (int j = 0; j < BJ; ++j)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = m0 + wr + i * 16 + (lane >> 4) * 4 + r;
                int col = n0 + wc + j * 16 + (lane & 15);
                if (row < M && col < N) C[(size_t)row * N + col] = acc[i][j][r];
            }
}

// MX-scaled fp8 GEMM: real per-block E8M0 scales through the mfma scale
// operands. Same 3-buffer global_load_lds pipeline as gemm_fp8_body; two
// differences, both forced by the hardware scale-lane layout pinned in
// profiles/r2/scale_probe_w2.txt:
//  * a lane in scale group g (= lane>>4) must hold logical k-chunks
//    {g, g+4} in its register slots (CH = [0,4,1,5,2,6,3,7]), so the
//    fragment is two 16-byte LDS reads instead of one 32-byte read;
//  * each mfma takes sa = SA[row][blk], sb = SB[col][blk] at byte 0
//    (OPSEL 0), blk = k0/32 + g. Scale bytes are read straight from
//    global: per step a lane loads AI+BJ bytes against 16 mfmas, and the
//    per-tile scale slice (TBM+TBN bytes per k-step) lives in L2.
template <int TBM, int TBN, int TWR, int TWC, int GROUP = 1, int SWZ = 0, int LOADS = 1>
__device__ __forceinline__ void gemm_fp8_scaled_body(
    const unsigned char* __restrict__ A,   // [M][K] row-major e4m3
    const unsigned char* __restrict__ Bt,  // [N][K] row-major e4m3
    const unsigned char* __restrict__ SA,  // [M][K/32] e8m0
    const unsigned char* __restrict__ SBt, // [N][K/32] e8m0
    float* __restrict__ C,                 // [M][N] row-major f32
    int M, int N, int K) {
    constexpr int BK = 128;
    constexpr int NT = TWR * TWC * WAVE;
    constexpr int WM = TBM / TWR;
    constexpr int WN = TBN / TWC;
    constexpr int AI = WM / 16;
    constexpr int BJ = WN / 16;
    // LOADS==2: per-buffer scale panel ((TBM+TBN)*4 B) staged next to data
    constexpr int SCB = (LOADS == 2) ? (TBM + TBN) * 4 : 0;
    __shared__ unsigned char lds[3 * (TBM * BK + TBN * BK + SCB)];
    const int HALF = TBM * BK + TBN * BK + SCB;
    auto ldsA = [&](int buf) -> unsigned char* { return lds + buf * HALF; };
    auto ldsB = [&](int buf) -> unsigned char* { return lds + buf * HALF + TBM * BK; };
    auto ldsS = [&](int buf) -> unsigned char* { return lds + buf * HALF + TBM * BK + TBN * BK; };

    const int tiles_n = (N + TBN - 1) / TBN;
    int tile_m, tile_n;
    if (GROUP > 1) {
        const int tiles_m = (M + TBM - 1) / TBM;
        const int per_group = GROUP * tiles_n;
        const int gid = blockIdx.x / per_group;
        const int first_m = gid * GROUP;
        const int gsz = min(GROUP, tiles_m - first_m);
        tile_m = first_m + (blockIdx.x % per_group) % gsz;
        tile_n = (blockIdx.x % per_group) / gsz;
    } else {
        tile_m = blockIdx.x / tiles_n;
        tile_n = blockIdx.x % tiles_n;
    }
    const int m0 = tile_m * TBM;
    const int n0 = tile_n * TBN;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int wr = (wid / TWC) * WM;
    const int wc = (wid % TWC) * WN;
    const int kblocks_ = K / 32;

    constexpr int ACH = TBM * (BK / 16);
    constexpr int APH = ACH / NT;
    constexpr int BCH = TBN * (BK / 16);
    constexpr int BPH = BCH / NT;
    static_assert(APH * NT == ACH && BPH * NT == BCH, "phase split");
    constexpr int S = APH + BPH;
    auto stage = [&](int buf, int k0) {
        unsigned char* la = ldsA(buf);
        unsigned char* lb = ldsB(buf);
#pragma unroll
        for (int phase = 0; phase < APH; ++phase) {
            int chunk = phase * NT + tid;
            int r = chunk / (BK / 16);
            int c = chunk % (BK / 16);
            if (SWZ) c ^= r & 7;  // chunk-granular XOR: spreads LDS banks
            const unsigned char* ga = &A[(size_t)(m0 + r) * K + k0 + c * 16];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)ga,
                (__attribute__((address_space(3))) void*)(la + (size_t)(phase * NT + wid * WAVE) * 16),
                16, 0, 0);
        }
#pragma unroll
        for (int phase = 0; phase < BPH; ++phase) {
            int chunk = phase * NT + tid;
            int r = chunk / (BK / 16);
            int c = chunk % (BK / 16);
            if (SWZ) c ^= r & 7;
            const unsigned char* gb = &Bt[(size_t)(n0 + r) * K + k0 + c * 16];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gb,
                (__attribute__((address_space(3))) void*)(lb + (size_t)(phase * NT + wid * WAVE) * 16),
                16, 0, 0);
        }
        if (LOADS == 2 && tid < TBM + TBN) {
            // this k-step's 4 scale bytes for one A row / B column. A
            // masked instruction still bumps vmcnt for waves with any
            // active lane, which only makes the vmcnt(S) waits stricter
            // (measured FASTER than the unmasked/padded alternative for
            // this shape: 1837 vs 1736 TF — the fp4 body keeps the
            // unmasked form, where it won +5%).
            const unsigned char* gs = tid < TBM
                ? &SA[(size_t)(m0 + tid) * kblocks_ + k0 / 32]
                : &SBt[(size_t)(n0 + (tid - TBM)) * kblocks_ + k0 / 32];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gs,
                (__attribute__((address_space(3))) void*)(ldsS(buf) + tid * 4),
                4, 0, 0);
        }
    };

    f32x4 acc[AI][BJ] = {};
    const int g = lane >> 4;  // scale group: logical MX block k0/32 + g
    const int kblocks = K / 32;
    const int steps = K / BK;

    stage(0, 0);
    if (steps > 1) stage(1, BK);
    for (int s = 0; s < steps; ++s) {
        const int buf = s % 3;
        if (s + 1 < steps)
            asm volatile("s_waitcnt vmcnt(%0)" ::"n"(S) : "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        if (s + 2 < steps) stage((s + 2) % 3, (s + 2) * BK);

        const unsigned char* la = ldsA(buf);
        const unsigned char* lb = ldsB(buf);
        const int blk = s * 4 + g;
        union frag { i32x8 v; i32x4g h[2]; };
        frag af[AI], bf[BJ];
        int sa[AI], sb[BJ];
#pragma unroll
        for (int i = 0; i < AI; ++i) {
            const int ar = wr + i * 16 + (lane & 15);
            const int x = SWZ ? (ar & 7) : 0;
            af[i].h[0] = *(const i32x4g*)&la[ar * BK + (g ^ x) * 16];
            af[i].h[1] = *(const i32x4g*)&la[ar * BK + ((g + 4) ^ x) * 16];
            sa[i] = LOADS == 2 ? ldsS(buf)[ar * 4 + g]
                  : LOADS       ? SA[(size_t)(m0 + ar) * kblocks + blk] : 0x7F;
        }
#pragma unroll
        for (int j = 0; j < BJ; ++j) {
            const int bc = wc + j * 16 + (lane & 15);
            const int x = SWZ ? (bc & 7) : 0;
            bf[j].h[0] = *(const i32x4g*)&lb[bc * BK + (g ^ x) * 16];
            bf[j].h[1] = *(const i32x4g*)&lb[bc * BK + ((g + 4) ^ x) * 16];
            sb[j] = LOADS == 2 ? ldsS(buf)[TBM * 4 + bc * 4 + g]
                  : LOADS       ? SBt[(size_t)(n0 + bc) * kblocks + blk] : 0x7F;
        }
#pragma unroll
        for (int i = 0; i < AI; ++i)
#pragma unroll
            for (int j = 0; j < BJ; ++j)
                acc[i][j] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
                    af[i].v, bf[j].v, acc[i][j], 0, 0, 0, sa[i], 0, sb[j]);
    }

#pragma unroll
    for (int i = 0; i < AI; ++i)
#pragma unroll
        for (int j = 0; j < BJ; ++j)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = m0 + wr + i * 16 + (lane >> 4) * 4 + r;
                int col = n0 + wc + j * 16 + (lane & 15);
                if (row < M && col < N) C[(size_t)row * N + col] = acc[i][j][r];
            }
}

extern "C" __global__ void __launch_bounds__(256)
gemm_fp8_scaled_128_kernel(const unsigned char* A, const unsigned char* Bt,
                           const unsigned char* SA, const unsigned char* SBt,
                           float* C, int M, int N, int K) {
    gemm_fp8_scaled_body<128, 128, 2, 2>(A, Bt, SA, SBt, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512)
gemm_fp8_scaled_256_g16_kernel(const unsigned char* A, const unsigned char* Bt,
                               const unsigned char* SA, const unsigned char* SBt,
                               float* C, int M, int N, int K) {
    gemm_fp8_scaled_body<256, 128, 4, 2, 16>(A, Bt, SA, SBt, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512)
gemm_fp8_scaled_256_g16_swz_kernel(const unsigned char* A, const unsigned char* Bt,
                                   const unsigned char* SA, const unsigned char* SBt,
                                   float* C, int M, int N, int K) {
    gemm_fp8_scaled_body<256, 128, 4, 2, 16, 1>(A, Bt, SA, SBt, C, M, N, K);
}

// diagnostic: identical split-chunk data path, scales hardcoded to 1.0 —
// isolates the cost of the per-step scale loads from the fragment split
extern "C" __global__ void __launch_bounds__(512)
gemm_fp8_scaled_noload_kernel(const unsigned char* A, const unsigned char* Bt,
                              const unsigned char* SA, const unsigned char* SBt,
                              float* C, int M, int N, int K) {
    gemm_fp8_scaled_body<256, 128, 4, 2, 16, 1, 0>(A, Bt, SA, SBt, C, M, N, K);
}

// Production MX-scaled body: K-loop grouped by 4 steps so each lane loads
// its scale bytes as ONE 16-byte vector per row per 4 steps (16 blocks =
// 512 k) instead of a byte per step — measured 3x: the per-step scalar
// byte loads were the whole gap to the unscaled champion (529 diagnostic:
// 2205 TF no-loads vs 747 byte-loads). Requires K % 512 == 0.
template <int TBM, int TBN, int TWR, int TWC, int GROUP, int SWZ>
__device__ __forceinline__ void gemm_fp8_scaled4_body(
    const unsigned char* __restrict__ A,
    const unsigned char* __restrict__ Bt,
    const unsigned char* __restrict__ SA,
    const unsigned char* __restrict__ SBt,
    float* __restrict__ C, int M, int N, int K) {
    constexpr int BK = 128;
    constexpr int NT = TWR * TWC * WAVE;
    constexpr int WM = TBM / TWR;
    constexpr int WN = TBN / TWC;
    constexpr int AI = WM / 16;
    constexpr int BJ = WN / 16;
    __shared__ unsigned char lds[3 * (TBM * BK + TBN * BK)];
    const int HALF = TBM * BK + TBN * BK;
    auto ldsA = [&](int buf) -> unsigned char* { return lds + buf * HALF; };
    auto ldsB = [&](int buf) -> unsigned char* { return lds + buf * HALF + TBM * BK; };

    const int tiles_n = (N + TBN - 1) / TBN;
    int tile_m, tile_n;
    if (GROUP > 1) {
        const int tiles_m = (M + TBM - 1) / TBM;
        const int per_group = GROUP * tiles_n;
        const int gid = blockIdx.x / per_group;
        const int first_m = gid * GROUP;
        const int gsz = min(GROUP, tiles_m - first_m);
        tile_m = first_m + (blockIdx.x % per_group) % gsz;
        tile_n = (blockIdx.x % per_group) / gsz;
    } else {
        tile_m = blockIdx.x / tiles_n;
        tile_n = blockIdx.x % tiles_n;
    }
    const int m0 = tile_m * TBM;
    const int n0 = tile_n * TBN;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int wr = (wid / TWC) * WM;
    const int wc = (wid % TWC) * WN;

    constexpr int ACH = TBM * (BK / 16);
    constexpr int APH = ACH / NT;
    constexpr int BCH = TBN * (BK / 16);
    constexpr int BPH = BCH / NT;
    static_assert(APH * NT == ACH && BPH * NT == BCH, "phase split");
    constexpr int S = APH + BPH;
    auto stage = [&](int buf, int k0) {
        unsigned char* la = ldsA(buf);
        unsigned char* lb = ldsB(buf);
#pragma unroll
        for (int phase = 0; phase < APH; ++phase) {
            int chunk = phase * NT + tid;
            int r = chunk / (BK / 16);
            int c = chunk % (BK / 16);
            if (SWZ) c ^= r & 7;
            const unsigned char* ga = &A[(size_t)(m0 + r) * K + k0 + c * 16];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)ga,
                (__attribute__((address_space(3))) void*)(la + (size_t)(phase * NT + wid * WAVE) * 16),
                16, 0, 0);
        }
#pragma unroll
        for (int phase = 0; phase < BPH; ++phase) {
            int chunk = phase * NT + tid;
            int r = chunk / (BK / 16);
            int c = chunk % (BK / 16);
            if (SWZ) c ^= r & 7;
            const unsigned char* gb = &Bt[(size_t)(n0 + r) * K + k0 + c * 16];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gb,
                (__attribute__((address_space(3))) void*)(lb + (size_t)(phase * NT + wid * WAVE) * 16),
                16, 0, 0);
        }
    };

    f32x4 acc[AI][BJ] = {};
    const int g = lane >> 4;
    const int kblocks = K / 32;
    const int steps = K / BK;   // must be a multiple of 4

    stage(0, 0);
    if (steps > 1) stage(1, BK);
    for (int s4 = 0; s4 < steps; s4 += 4) {
        // 16 scale blocks (= these 4 k-steps) per affected row, one 16B load
        union sv { i32x4g v; unsigned char b[16]; };
        sv sad[AI], sbd[BJ];
#pragma unroll
        for (int i = 0; i < AI; ++i) {
            const int ar = wr + i * 16 + (lane & 15);
            sad[i].v = *(const i32x4g*)&SA[(size_t)(m0 + ar) * kblocks + s4 * 4];
        }
#pragma unroll
        for (int j = 0; j < BJ; ++j) {
            const int bc = wc + j * 16 + (lane & 15);
            sbd[j].v = *(const i32x4g*)&SBt[(size_t)(n0 + bc) * kblocks + s4 * 4];
        }
#pragma unroll
        for (int ss = 0; ss < 4; ++ss) {
            const int s = s4 + ss;
            const int buf = s % 3;
            // (4-step variant unchanged) strict count: scale loads sit between
            // stage(s4+1) and stage(s4+2) in issue order, so vmcnt(S)
            // remains sufficient at every inner step (at ss=0/1 it also
            // drains the scale loads — one L2 latency per 4 steps)
            if (s + 1 < steps)
                asm volatile("s_waitcnt vmcnt(%0)" ::"n"(S) : "memory");
            else
                asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
            __builtin_amdgcn_s_barrier();
            if (s + 2 < steps) stage((s + 2) % 3, (s + 2) * BK);

            const unsigned char* la = ldsA(buf);
            const unsigned char* lb = ldsB(buf);
            union frag { i32x8 v; i32x4g h[2]; };
            frag af[AI], bf[BJ];
            int sa[AI], sb[BJ];
            (void)0;
#pragma unroll
            for (int i = 0; i < AI; ++i) {
                const int ar = wr + i * 16 + (lane & 15);
                const int x = SWZ ? (ar & 7) : 0;
                af[i].h[0] = *(const i32x4g*)&la[ar * BK + (g ^ x) * 16];
                af[i].h[1] = *(const i32x4g*)&la[ar * BK + ((g + 4) ^ x) * 16];
            }
#pragma unroll
            for (int j = 0; j < BJ; ++j) {
                const int bc = wc + j * 16 + (lane & 15);
                const int x = SWZ ? (bc & 7) : 0;
                bf[j].h[0] = *(const i32x4g*)&lb[bc * BK + (g ^ x) * 16];
                bf[j].h[1] = *(const i32x4g*)&lb[bc * BK + ((g + 4) ^ x) * 16];
            }
            // byte ss*4 + g of the 16B scale vector = block s*4+g: dword ss
            // shifted by 8*g (g is per-lane, so a variable shift not a
            // dynamic byte index — no scratch)
#pragma unroll
            for (int i = 0; i < AI; ++i)
                sa[i] = (((const unsigned*)&sad[i].v)[ss] >> (8 * g)) & 0xFF;
#pragma unroll
            for (int j = 0; j < BJ; ++j)
                sb[j] = (((const unsigned*)&sbd[j].v)[ss] >> (8 * g)) & 0xFF;
#pragma unroll
            for (int i = 0; i < AI; ++i)
#pragma unroll
                for (int j = 0; j < BJ; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
                        af[i].v, bf[j].v, acc[i][j], 0, 0, 0, sa[i], 0, sb[j]);
        }
    }

#pragma unroll
    for (int i = 0; i < AI; ++i)
#pragma unroll
        for (int j = 0; j < BJ; ++j)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = m0 + wr + i * 16 + (lane >> 4) * 4 + r;
                int col = n0 + wc + j * 16 + (lane & 15);
                if (row < M && col < N) C[(size_t)row * N + col] = acc[i][j][r];
            }
}

extern "C" __global__ void __launch_bounds__(512)
gemm_fp8_scaled_slds_g16_swz_kernel(const unsigned char* A, const unsigned char* Bt,
                                    const unsigned char* SA, const unsigned char* SBt,
                                    float* C, int M, int N, int K) {
    gemm_fp8_scaled_body<256, 128, 4, 2, 16, 1, 2>(A, Bt, SA, SBt, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512)
gemm_fp8_scaled4_g16_swz_kernel(const unsigned char* A, const unsigned char* Bt,
                                const unsigned char* SA, const unsigned char* SBt,
                                float* C, int M, int N, int K) {
    gemm_fp8_scaled4_body<256, 128, 4, 2, 16, 1>(A, Bt, SA, SBt, C, M, N, K);
}

// 128x128: 3 x 32 KiB LDS (2 WG/CU), 4 waves of 64x64
extern "C" __global__ void __launch_bounds__(256)
gemm_fp8_128_kernel(const unsigned char* A, const unsigned char* Bt, float* C,
                    int M, int N, int K) {
    gemm_fp8_body<128, 128, 2, 2>(A, Bt, C, M, N, K);
}

// 256x128: 3 x 48 KiB LDS (1 WG/CU), 8 waves of 64x64
extern "C" __global__ void __launch_bounds__(512)
gemm_fp8_256_kernel(const unsigned char* A, const unsigned char* Bt, float* C,
                    int M, int N, int K) {
    gemm_fp8_body<256, 128, 4, 2>(A, Bt, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512)
gemm_fp8_256_g4_kernel(const unsigned char* A, const unsigned char* Bt, float* C,
                       int M, int N, int K) {
    gemm_fp8_body<256, 128, 4, 2, 4>(A, Bt, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512)
gemm_fp8_256_g8_kernel(const unsigned char* A, const unsigned char* Bt, float* C,
                       int M, int N, int K) {
    gemm_fp8_body<256, 128, 4, 2, 8>(A, Bt, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512)
gemm_fp8_256_g16_kernel(const unsigned char* A, const unsigned char* Bt, float* C,
                        int M, int N, int K) {
    gemm_fp8_body<256, 128, 4, 2, 16>(A, Bt, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512)
gemm_fp8_256_g32_kernel(const unsigned char* A, const unsigned char* Bt, float* C,
                        int M, int N, int K) {
    gemm_fp8_body<256, 128, 4, 2, 32>(A, Bt, C, M, N, K);
}

// 256x256 fp8, 2 x 64 KiB LDS double buffer (dual barrier per step since a
// staged buffer is immediately reused), 8 waves of 64x128 (AI=4, BJ=8)
template <int GROUP, int SWZ = 0, int TAILBAR = 1>
__device__ __forceinline__ void gemm_fp8_256x256_body(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K) {
    constexpr int TBM = 256, TBN = 256, BK = 128, NT = 512;
    constexpr int AI = 4, BJ = 8;
    __shared__ unsigned char lds[2 * (TBM * BK + TBN * BK)];
    const int HALF = TBM * BK + TBN * BK;
    auto ldsA = [&](int buf) -> unsigned char* { return lds + buf * HALF; };
    auto ldsB = [&](int buf) -> unsigned char* { return lds + buf * HALF + TBM * BK; };

    const int tiles_n = (N + TBN - 1) / TBN;
    int tile_m, tile_n;
    if (GROUP > 1) {
        const int tiles_m = (M + TBM - 1) / TBM;
        const int per_group = GROUP * tiles_n;
        const int gid = blockIdx.x / per_group;
        const int first_m = gid * GROUP;
        const int gsz = min(GROUP, tiles_m - first_m);
        tile_m = first_m + (blockIdx.x % per_group) % gsz;
        tile_n = (blockIdx.x % per_group) / gsz;
    } else {
        tile_m = blockIdx.x / tiles_n;
        tile_n = blockIdx.x % tiles_n;
    }
    const int m0 = tile_m * TBM;
    const int n0 = tile_n * TBN;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int wr = (wid >> 1) * 64;
    const int wc = (wid & 1) * 128;

    constexpr int ACH = TBM * (BK / 16);
    constexpr int APH = ACH / NT;   // 4
    auto stage = [&](int buf, int k0) {
        unsigned char* la = ldsA(buf);
        unsigned char* lb = ldsB(buf);
#pragma unroll
        for (int phase = 0; phase < APH; ++phase) {
            int chunk = phase * NT + tid;
            int r = chunk / (BK / 16);
            int c = chunk % (BK / 16);
            if (SWZ) c = ((c >> 1) ^ (r & 3)) * 2 + (c & 1);  // pair swizzle
            const unsigned char* ga = &A[(size_t)(m0 + r) * K + k0 + c * 16];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)ga,
                (__attribute__((address_space(3))) void*)(la + (size_t)(phase * NT + wid * WAVE) * 16),
                16, 0, 0);
            const unsigned char* gb = &Bt[(size_t)(n0 + r) * K + k0 + c * 16];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gb,
                (__attribute__((address_space(3))) void*)(lb + (size_t)(phase * NT + wid * WAVE) * 16),
                16, 0, 0);
        }
    };

    f32x4 acc[AI][BJ] = {};
    const int steps = K / BK;

    stage(0, 0);
    for (int s = 0; s < steps; ++s) {
        const int buf = s & 1;
        // stage(s) fully landed
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        // prefetch s+1 into the other buffer UNDER this step's MFMAs —
        // safe: compute(s-1) finished before the barrier above
        if (s + 1 < steps) stage(buf ^ 1, (s + 1) * BK);

        const unsigned char* la = ldsA(buf);
        const unsigned char* lb = ldsB(buf);
        i32x8 af[AI], bf[BJ];
#pragma unroll
        for (int i = 0; i < AI; ++i) {
            const int ar = wr + i * 16 + (lane & 15);
            int p = lane >> 4;  // 32B pair index of this lane's k-slice
            if (SWZ) p ^= ar & 3;
            af[i] = *(const i32x8*)&la[ar * BK + p * 32];
        }
#pragma unroll
        for (int j = 0; j < BJ; ++j) {
            const int bc = wc + j * 16 + (lane & 15);
            int p = lane >> 4;
            if (SWZ) p ^= bc & 3;
            bf[j] = *(const i32x8*)&lb[bc * BK + p * 32];
        }
#pragma unroll
        for (int i = 0; i < AI; ++i)
#pragma unroll
            for (int j = 0; j < BJ; ++j)
                acc[i][j] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
                    af[i], bf[j], acc[i][j], 0, 0, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
        // TAILBAR=0: the next iteration's top barrier already proves every
        // wave finished this step's reads before stage(s+2) can overwrite
        // the buffer (stages are only issued after a barrier) — measured
        // variant; TAILBAR=1 keeps the conservative trailing barrier.
        if (TAILBAR) __syncthreads();
    }

#pragma unroll
    for (int i = 0; i < AI; ++i)
#pragma unroll
        for (int j = 0; j < BJ; ++j)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = m0 + wr + i * 16 + (lane >> 4) * 4 + r;
                int col = n0 + wc + j * 16 + (lane & 15);
                if (row < M && col < N) C[(size_t)row * N + col] = acc[i][j][r];
            }
}

extern "C" __global__ void __launch_bounds__(512)
gemm_fp8_256x256_kernel(const unsigned char* A, const unsigned char* Bt, float* C,
                        int M, int N, int K) {
    gemm_fp8_256x256_body<1>(A, Bt, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512)
gemm_fp8_256x256_g16_kernel(const unsigned char* A, const unsigned char* Bt, float* C,
                            int M, int N, int K) {
    gemm_fp8_256x256_body<16>(A, Bt, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512)
gemm_fp8_256x256_g16_swz_kernel(const unsigned char* A, const unsigned char* Bt, float* C,
                                int M, int N, int K) {
    gemm_fp8_256x256_body<16, 1>(A, Bt, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512)
gemm_fp8_256x256_g16_swz_nb_kernel(const unsigned char* A, const unsigned char* Bt, float* C,
                                   int M, int N, int K) {
    gemm_fp8_256x256_body<16, 1, 0>(A, Bt, C, M, N, K);
}

// ---------------------------------------------------------------------------
// MX-fp4 (OCP e2m1, scale=1) GEMM — mfma_scale_f32_32x32x64_f8f6f4, the
// ~10 PF dense headline shape (ubench ceiling measured 9074 TF,
// gpurun_out/r2s17). Elements pack two per byte, so HBM+LDS bytes per FLOP
// are HALF the fp8 kernel's. 256x256 tile, 8 waves of 64x128 as 2x4
// fragments of 32x32 (C/D: 16 f32 per lane, row = (reg&3)+8*(reg>>2)+
// 4*(lane>>5), col = lane&31). BK=256 (4 MFMA K-depths per staged step),
// 2 x 64 KiB LDS double buffer with dual barrier, grouped tile swizzle.
// ---------------------------------------------------------------------------

typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef int i32x4 __attribute__((ext_vector_type(4)));

template <int GROUP>
__device__ __forceinline__ void gemm_fp4_256x256_body(
    const unsigned char* __restrict__ A,   // [M][K/2] packed e2m1 (even k = low nibble)
    const unsigned char* __restrict__ Bt,  // [N][K/2] packed e2m1
    float* __restrict__ C,                 // [M][N] f32
    int M, int N, int K) {
    constexpr int TBM = 256, TBN = 256, BK = 256, NT = 512;
    constexpr int AI = 2, BJ = 4;  // wave tile 64x128 in 32x32 fragments
    constexpr int RB = BK / 2;     // tile row bytes (128)
    __shared__ unsigned char lds[2 * (TBM + TBN) * RB];
    const int HALF = (TBM + TBN) * RB;
    auto ldsA = [&](int buf) -> unsigned char* { return lds + buf * HALF; };
    auto ldsB = [&](int buf) -> unsigned char* { return lds + buf * HALF + TBM * RB; };

    const int tiles_n = (N + TBN - 1) / TBN;
    int tile_m, tile_n;
    if (GROUP > 1) {
        const int tiles_m = (M + TBM - 1) / TBM;
        const int per_group = GROUP * tiles_n;
        const int gid = blockIdx.x / per_group;
        const int first_m = gid * GROUP;
        const int gsz = min(GROUP, tiles_m - first_m);
        tile_m = first_m + (blockIdx.x % per_group) % gsz;
        tile_n = (blockIdx.x % per_group) / gsz;
    } else {
        tile_m = blockIdx.x / tiles_n;
        tile_n = blockIdx.x % tiles_n;
    }
    const int m0 = tile_m * TBM;
    const int n0 = tile_n * TBN;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int wr = (wid >> 1) * 64;   // 4 wave rows
    const int wc = (wid & 1) * 128;   // 2 wave cols

    constexpr int ACH = TBM * (RB / 16);  // 16B chunks per A tile (2048)
    constexpr int APH = ACH / NT;         // 4
    auto stage = [&](int buf, int k0) {
        unsigned char* la = ldsA(buf);
        unsigned char* lb = ldsB(buf);
        const size_t kb0 = (size_t)k0 / 2;
#pragma unroll
        for (int phase = 0; phase < APH; ++phase) {
            int chunk = phase * NT + tid;
            int r = chunk / (RB / 16);
            int c = chunk % (RB / 16);
            const unsigned char* ga = &A[(size_t)(m0 + r) * (K / 2) + kb0 + c * 16];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)ga,
                (__attribute__((address_space(3))) void*)(la + (size_t)(phase * NT + wid * WAVE) * 16),
                16, 0, 0);
            const unsigned char* gb = &Bt[(size_t)(n0 + r) * (K / 2) + kb0 + c * 16];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gb,
                (__attribute__((address_space(3))) void*)(lb + (size_t)(phase * NT + wid * WAVE) * 16),
                16, 0, 0);
        }
    };

    f32x16 acc[AI][BJ] = {};
    const int kgrp = lane >> 5;       // 0..1: this lane's 32-k half
    const int ln31 = lane & 31;
    const int steps = K / BK;

    stage(0, 0);
    for (int s = 0; s < steps; ++s) {
        const int buf = s & 1;
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        if (s + 1 < steps) stage(buf ^ 1, (s + 1) * BK);

        const unsigned char* la = ldsA(buf);
        const unsigned char* lb = ldsB(buf);
#pragma unroll
        for (int d = 0; d < BK / 64; ++d) {  // 4 MFMA K-depths per step
            const int kb = d * 32 + kgrp * 16;  // byte offset of this lane's 16B
            union { i32x8 v; unsigned char b[32]; } af[AI], bf[BJ];
#pragma unroll
            for (int i = 0; i < AI; ++i) {
                const int ar = wr + i * 32 + ln31;
                *(i32x4*)&af[i].b[0] = *(const i32x4*)&la[ar * RB + kb];
            }
#pragma unroll
            for (int j = 0; j < BJ; ++j) {
                const int bc = wc + j * 32 + ln31;
                *(i32x4*)&bf[j].b[0] = *(const i32x4*)&lb[bc * RB + kb];
            }
#pragma unroll
            for (int i = 0; i < AI; ++i)
#pragma unroll
                for (int j = 0; j < BJ; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
                        af[i].v, bf[j].v, acc[i][j], 4, 4, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
        }
        __syncthreads();
    }

#pragma unroll
    for (int i = 0; i < AI; ++i)
#pragma unroll
        for (int j = 0; j < BJ; ++j)
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                int row = m0 + wr + i * 32 + (r & 3) + 8 * (r >> 2) + 4 * kgrp;
                int col = n0 + wc + j * 32 + ln31;
                if (row < M && col < N) C[(size_t)row * N + col] = acc[i][j][r];
            }
}

extern "C" __global__ void __launch_bounds__(512)
gemm_fp4_256x256_kernel(const unsigned char* A, const unsigned char* Bt, float* C,
                        int M, int N, int K) {
    gemm_fp4_256x256_body<1>(A, Bt, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512)
gemm_fp4_256x256_g16_kernel(const unsigned char* A, const unsigned char* Bt, float* C,
                            int M, int N, int K) {
    gemm_fp4_256x256_body<16>(A, Bt, C, M, N, K);
}

// fp4 3-buffer counted-vmcnt variant: BK=128 (2 MFMA K-depths per step),
// 3 x 32 KiB LDS, single barrier per step (one tile in flight) — measures
// whether the 2-buffer dual-barrier serialization is the remaining cost.
template <int GROUP, int SWZ = 0, int TBM = 256, int TBN = 256,
          int TWR = 4, int TWC = 2>
__device__ __forceinline__ void gemm_fp4_3buf_body(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K) {
    constexpr int BK = 128, NT = TWR * TWC * WAVE;
    constexpr int AI = (TBM / TWR) / 32, BJ = (TBN / TWC) / 32;
    constexpr int RB = BK / 2;  // 64 B rows
    __shared__ unsigned char lds[3 * (TBM + TBN) * RB];
    const int HALF = (TBM + TBN) * RB;
    auto ldsA = [&](int buf) -> unsigned char* { return lds + buf * HALF; };
    auto ldsB = [&](int buf) -> unsigned char* { return lds + buf * HALF + TBM * RB; };

    const int tiles_n = (N + TBN - 1) / TBN;
    int tile_m, tile_n;
    if (GROUP > 1) {
        const int tiles_m = (M + TBM - 1) / TBM;
        const int per_group = GROUP * tiles_n;
        const int gid = blockIdx.x / per_group;
        const int first_m = gid * GROUP;
        const int gsz = min(GROUP, tiles_m - first_m);
        tile_m = first_m + (blockIdx.x % per_group) % gsz;
        tile_n = (blockIdx.x % per_group) / gsz;
    } else {
        tile_m = blockIdx.x / tiles_n;
        tile_n = blockIdx.x % tiles_n;
    }
    const int m0 = tile_m * TBM;
    const int n0 = tile_n * TBN;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int wr = (wid / TWC) * (TBM / TWR);
    const int wc = (wid % TWC) * (TBN / TWC);

    constexpr int ACH = TBM * (RB / 16);
    constexpr int APH = ACH / NT;
    constexpr int BCH = TBN * (RB / 16);
    constexpr int BPH = BCH / NT;
    static_assert(APH * NT == ACH && BPH * NT == BCH, "phase split");
    constexpr int S = APH + BPH;  // A + B glds per thread per stage
    auto stage = [&](int buf, int k0) {
        unsigned char* la = ldsA(buf);
        unsigned char* lb = ldsB(buf);
        const size_t kb0 = (size_t)k0 / 2;
#pragma unroll
        for (int phase = 0; phase < APH; ++phase) {
            int chunk = phase * NT + tid;
            int r = chunk / (RB / 16);
            int c = chunk % (RB / 16);
            if (SWZ) c ^= r & (RB / 16 - 1);
            const unsigned char* ga = &A[(size_t)(m0 + r) * (K / 2) + kb0 + c * 16];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)ga,
                (__attribute__((address_space(3))) void*)(la + (size_t)(phase * NT + wid * WAVE) * 16),
                16, 0, 0);
        }
#pragma unroll
        for (int phase = 0; phase < BPH; ++phase) {
            int chunk = phase * NT + tid;
            int r = chunk / (RB / 16);
            int c = chunk % (RB / 16);
            if (SWZ) c ^= r & (RB / 16 - 1);
            const unsigned char* gb = &Bt[(size_t)(n0 + r) * (K / 2) + kb0 + c * 16];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gb,
                (__attribute__((address_space(3))) void*)(lb + (size_t)(phase * NT + wid * WAVE) * 16),
                16, 0, 0);
        }
    };

    f32x16 acc[AI][BJ] = {};
    const int kgrp = lane >> 5;
    const int ln31 = lane & 31;
    const int steps = K / BK;

    stage(0, 0);
    if (steps > 1) stage(1, BK);
    for (int s = 0; s < steps; ++s) {
        const int buf = s % 3;
        if (s + 1 < steps)
            asm volatile("s_waitcnt vmcnt(%0)" ::"n"(S) : "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        if (s + 2 < steps) stage((s + 2) % 3, (s + 2) * BK);

        const unsigned char* la = ldsA(buf);
        const unsigned char* lb = ldsB(buf);
#pragma unroll
        for (int d = 0; d < BK / 64; ++d) {
            union { i32x8 v; unsigned char b[32]; } af[AI], bf[BJ];
#pragma unroll
            for (int i = 0; i < AI; ++i) {
                const int ar = wr + i * 32 + ln31;
                int ch = d * 2 + kgrp;
                if (SWZ) ch ^= ar & (RB / 16 - 1);
                *(i32x4*)&af[i].b[0] = *(const i32x4*)&la[ar * RB + ch * 16];
            }
#pragma unroll
            for (int j = 0; j < BJ; ++j) {
                const int bc = wc + j * 32 + ln31;
                int ch = d * 2 + kgrp;
                if (SWZ) ch ^= bc & (RB / 16 - 1);
                *(i32x4*)&bf[j].b[0] = *(const i32x4*)&lb[bc * RB + ch * 16];
            }
#pragma unroll
            for (int i = 0; i < AI; ++i)
#pragma unroll
                for (int j = 0; j < BJ; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
                        af[i].v, bf[j].v, acc[i][j], 4, 4, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
        }
    }

#pragma unroll
    for (int i = 0; i < AI; ++i)
#pragma unroll
        for (int j = 0; j < BJ; ++j)
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                int row = m0 + wr + i * 32 + (r & 3) + 8 * (r >> 2) + 4 * kgrp;
                int col = n0 + wc + j * 32 + ln31;
                if (row < M && col < N) C[(size_t)row * N + col] = acc[i][j][r];
            }
}

extern "C" __global__ void __launch_bounds__(512)
gemm_fp4_3buf_g16_kernel(const unsigned char* A, const unsigned char* Bt, float* C,
                         int M, int N, int K) {
    gemm_fp4_3buf_body<16>(A, Bt, C, M, N, K);
}

extern "C" __global__ void __launch_bounds__(512)
gemm_fp4_3buf_g16_swz_kernel(const unsigned char* A, const unsigned char* Bt, float* C,
                             int M, int N, int K) {
    gemm_fp4_3buf_body<16, 1>(A, Bt, C, M, N, K);
}

// MX-scaled fp4 GEMM: real per-block E8M0 scales. The fp4 scale-lane
// layout is the NAIVE one (probe profiles/r2/fp4_scale_probe.txt: scale
// lane idx+32*g covers exactly the 32 k-elements that lane supplies), so
// no chunk permutation is needed; scale panels are staged into LDS per
// buffer exactly like the fp8 556 variant. blk = s*4 + d*2 + kgrp.
template <int GROUP, int SWZ, int TBM = 256, int TBN = 256,
          int TWR = 4, int TWC = 2>
__device__ __forceinline__ void gemm_fp4_scaled_body(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ Bt,
    const unsigned char* __restrict__ SA, const unsigned char* __restrict__ SBt,
    float* __restrict__ C, int M, int N, int K) {
    constexpr int BK = 128, NT = TWR * TWC * WAVE;
    constexpr int AI = (TBM / TWR) / 32, BJ = (TBN / TWC) / 32;
    constexpr int RB = BK / 2;  // 64 B rows
    constexpr int SCB = NT * 4;  // per-buffer scale panel (thread-padded)
    __shared__ unsigned char lds[3 * ((TBM + TBN) * RB + SCB)];
    const int HALF = (TBM + TBN) * RB + SCB;
    auto ldsA = [&](int buf) -> unsigned char* { return lds + buf * HALF; };
    auto ldsB = [&](int buf) -> unsigned char* { return lds + buf * HALF + TBM * RB; };
    auto ldsS = [&](int buf) -> unsigned char* { return lds + buf * HALF + (TBM + TBN) * RB; };

    const int tiles_n = (N + TBN - 1) / TBN;
    int tile_m, tile_n;
    if (GROUP > 1) {
        const int tiles_m = (M + TBM - 1) / TBM;
        const int per_group = GROUP * tiles_n;
        const int gid = blockIdx.x / per_group;
        const int first_m = gid * GROUP;
        const int gsz = min(GROUP, tiles_m - first_m);
        tile_m = first_m + (blockIdx.x % per_group) % gsz;
        tile_n = (blockIdx.x % per_group) / gsz;
    } else {
        tile_m = blockIdx.x / tiles_n;
        tile_n = blockIdx.x % tiles_n;
    }
    const int m0 = tile_m * TBM;
    const int n0 = tile_n * TBN;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int wr = (wid / TWC) * (TBM / TWR);
    const int wc = (wid % TWC) * (TBN / TWC);
    const int kblocks_ = K / 32;

    constexpr int ACH = TBM * (RB / 16);
    constexpr int APH = ACH / NT;
    constexpr int BCH = TBN * (RB / 16);
    constexpr int BPH = BCH / NT;
    static_assert(APH * NT == ACH && BPH * NT == BCH, "phase split");
    constexpr int S = APH + BPH;
    auto stage = [&](int buf, int k0) {
        unsigned char* la = ldsA(buf);
        unsigned char* lb = ldsB(buf);
        const size_t kb0 = (size_t)k0 / 2;
#pragma unroll
        for (int phase = 0; phase < APH; ++phase) {
            int chunk = phase * NT + tid;
            int r = chunk / (RB / 16);
            int c = chunk % (RB / 16);
            if (SWZ) c ^= r & (RB / 16 - 1);
            const unsigned char* ga = &A[(size_t)(m0 + r) * (K / 2) + kb0 + c * 16];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)ga,
                (__attribute__((address_space(3))) void*)(la + (size_t)(phase * NT + wid * WAVE) * 16),
                16, 0, 0);
        }
#pragma unroll
        for (int phase = 0; phase < BPH; ++phase) {
            int chunk = phase * NT + tid;
            int r = chunk / (RB / 16);
            int c = chunk % (RB / 16);
            if (SWZ) c ^= r & (RB / 16 - 1);
            const unsigned char* gb = &Bt[(size_t)(n0 + r) * (K / 2) + kb0 + c * 16];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gb,
                (__attribute__((address_space(3))) void*)(lb + (size_t)(phase * NT + wid * WAVE) * 16),
                16, 0, 0);
        }
        {
            // unmasked (clamped) so every wave issues exactly S+1 loads
            const int t = tid < TBM + TBN - 1 ? tid : TBM + TBN - 1;
            const unsigned char* gs = t < TBM
                ? &SA[(size_t)(m0 + t) * kblocks_ + k0 / 32]
                : &SBt[(size_t)(n0 + (t - TBM)) * kblocks_ + k0 / 32];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gs,
                (__attribute__((address_space(3))) void*)(ldsS(buf) + tid * 4),
                4, 0, 0);
        }
    };

    f32x16 acc[AI][BJ] = {};
    const int kgrp = lane >> 5;
    const int ln31 = lane & 31;
    const int steps = K / BK;

    stage(0, 0);
    if (steps > 1) stage(1, BK);
    for (int s = 0; s < steps; ++s) {
        const int buf = s % 3;
        if (s + 1 < steps)
            asm volatile("s_waitcnt vmcnt(%0)" ::"n"(S + 1) : "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        if (s + 2 < steps) stage((s + 2) % 3, (s + 2) * BK);

        const unsigned char* la = ldsA(buf);
        const unsigned char* lb = ldsB(buf);
        const unsigned char* lscale = ldsS(buf);
        // one b32 read per row fetches all 4 of this step's scale bytes;
        // both d-iterations extract theirs with a shift (halves the LDS
        // scale traffic vs per-(d,row) byte reads)
        unsigned sa32[AI], sb32[BJ];
#pragma unroll
        for (int i = 0; i < AI; ++i)
            sa32[i] = *(const unsigned*)&lscale[(wr + i * 32 + ln31) * 4];
#pragma unroll
        for (int j = 0; j < BJ; ++j)
            sb32[j] = *(const unsigned*)&lscale[TBM * 4 + (wc + j * 32 + ln31) * 4];
#pragma unroll
        for (int d = 0; d < BK / 64; ++d) {
            union { i32x8 v; unsigned char b[32]; } af[AI], bf[BJ];
            int sa[AI], sb[BJ];
#pragma unroll
            for (int i = 0; i < AI; ++i) {
                const int ar = wr + i * 32 + ln31;
                int ch = d * 2 + kgrp;
                if (SWZ) ch ^= ar & (RB / 16 - 1);
                *(i32x4*)&af[i].b[0] = *(const i32x4*)&la[ar * RB + ch * 16];
                sa[i] = (sa32[i] >> (8 * (d * 2 + kgrp))) & 0xFF;
            }
#pragma unroll
            for (int j = 0; j < BJ; ++j) {
                const int bc = wc + j * 32 + ln31;
                int ch = d * 2 + kgrp;
                if (SWZ) ch ^= bc & (RB / 16 - 1);
                *(i32x4*)&bf[j].b[0] = *(const i32x4*)&lb[bc * RB + ch * 16];
                sb[j] = (sb32[j] >> (8 * (d * 2 + kgrp))) & 0xFF;
            }
#pragma unroll
            for (int i = 0; i < AI; ++i)
#pragma unroll
                for (int j = 0; j < BJ; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
                        af[i].v, bf[j].v, acc[i][j], 4, 4, 0, sa[i], 0, sb[j]);
        }
    }

#pragma unroll
    for (int i = 0; i < AI; ++i)
#pragma unroll
        for (int j = 0; j < BJ; ++j)
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                int row = m0 + wr + i * 32 + (r & 3) + 8 * (r >> 2) + 4 * kgrp;
                int col = n0 + wc + j * 32 + ln31;
                if (row < M && col < N) C[(size_t)row * N + col] = acc[i][j][r];
            }
}

extern "C" __global__ void __launch_bounds__(512)
gemm_fp4_scaled_g16_swz_kernel(const unsigned char* A, const unsigned char* Bt,
                               const unsigned char* SA, const unsigned char* SBt,
                               float* C, int M, int N, int K) {
    gemm_fp4_scaled_body<16, 1>(A, Bt, SA, SBt, C, M, N, K);
}

// 512x256 tile: 16 waves (1024 threads), 3 x 48 KiB LDS. MEASURED
// CATASTROPHIC (273/169 TF vs 3524/2944 for the 256x256 3-buf champion,
// numerics exact — gpurun_out/r2s29): the 1024-thread launch bounds force
// the compiler to fit 128 acc VGPRs + fragments into a half-size budget,
// spilling to scratch. Kept as a measured-negative data point; the
// 256x256 tile remains the fp4 shape.
extern "C" __global__ void __launch_bounds__(1024)
gemm_fp4_512_g16_swz_kernel(const unsigned char* A, const unsigned char* Bt, float* C,
                            int M, int N, int K) {
    gemm_fp4_3buf_body<16, 1, 512, 256, 8, 2>(A, Bt, C, M, N, K);
}

// fp8 on the 32x32x64 shape (FMT=0), same 3-buffer counted structure as
// the fp4 winner: 1 B/elem so rows are BK bytes; per lane 32 k = 32 B
// fragments. 3 x 32 KiB LDS at BK=64.
template <int GROUP>
__device__ __forceinline__ void gemm_fp8k64_3buf_body(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K) {
    constexpr int TBM = 256, TBN = 256, BK = 64, NT = 512;
    constexpr int AI = 2, BJ = 4;
    constexpr int RB = BK;  // 64 B rows (1 B/elem)
    __shared__ unsigned char lds[3 * (TBM + TBN) * RB];
    const int HALF = (TBM + TBN) * RB;
    auto ldsA = [&](int buf) -> unsigned char* { return lds + buf * HALF; };
    auto ldsB = [&](int buf) -> unsigned char* { return lds + buf * HALF + TBM * RB; };

    const int tiles_n = (N + TBN - 1) / TBN;
    int tile_m, tile_n;
    if (GROUP > 1) {
        const int tiles_m = (M + TBM - 1) / TBM;
        const int per_group = GROUP * tiles_n;
        const int gid = blockIdx.x / per_group;
        const int first_m = gid * GROUP;
        const int gsz = min(GROUP, tiles_m - first_m);
        tile_m = first_m + (blockIdx.x % per_group) % gsz;
        tile_n = (blockIdx.x % per_group) / gsz;
    } else {
        tile_m = blockIdx.x / tiles_n;
        tile_n = blockIdx.x % tiles_n;
    }
    const int m0 = tile_m * TBM;
    const int n0 = tile_n * TBN;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    const int wr = (wid >> 1) * 64;
    const int wc = (wid & 1) * 128;

    constexpr int ACH = TBM * (RB / 16);  // 1024 chunks
    constexpr int APH = ACH / NT;         // 2
    constexpr int S = 2 * APH;            // A + B glds per thread per stage
    auto stage = [&](int buf, int k0) {
        unsigned char* la = ldsA(buf);
        unsigned char* lb = ldsB(buf);
        const size_t kb0 = (size_t)k0;
#pragma unroll
        for (int phase = 0; phase < APH; ++phase) {
            int chunk = phase * NT + tid;
            int r = chunk / (RB / 16);
            int c = chunk % (RB / 16);
            const unsigned char* ga = &A[(size_t)(m0 + r) * K + kb0 + c * 16];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)ga,
                (__attribute__((address_space(3))) void*)(la + (size_t)(phase * NT + wid * WAVE) * 16),
                16, 0, 0);
            const unsigned char* gb = &Bt[(size_t)(n0 + r) * K + kb0 + c * 16];
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)gb,
                (__attribute__((address_space(3))) void*)(lb + (size_t)(phase * NT + wid * WAVE) * 16),
                16, 0, 0);
        }
    };

    f32x16 acc[AI][BJ] = {};
    const int kgrp = lane >> 5;
    const int ln31 = lane & 31;
    const int steps = K / BK;

    stage(0, 0);
    if (steps > 1) stage(1, BK);
    for (int s = 0; s < steps; ++s) {
        const int buf = s % 3;
        if (s + 1 < steps)
            asm volatile("s_waitcnt vmcnt(%0)" ::"n"(S) : "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        if (s + 2 < steps) stage((s + 2) % 3, (s + 2) * BK);

        const unsigned char* la = ldsA(buf);
        const unsigned char* lb = ldsB(buf);
        {
            const int kb = kgrp * 32;  // one 64-deep MFMA per step; 32 B/lane
            i32x8 af[AI], bf[BJ];
#pragma unroll
            for (int i = 0; i < AI; ++i) {
                const int ar = wr + i * 32 + ln31;
                af[i] = *(const i32x8*)&la[ar * RB + kb];
            }
#pragma unroll
            for (int j = 0; j < BJ; ++j) {
                const int bc = wc + j * 32 + ln31;
                bf[j] = *(const i32x8*)&lb[bc * RB + kb];
            }
#pragma unroll
            for (int i = 0; i < AI; ++i)
#pragma unroll
                for (int j = 0; j < BJ; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
                        af[i], bf[j], acc[i][j], 0, 0, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
        }
    }

#pragma unroll
    for (int i = 0; i < AI; ++i)
#pragma unroll
        for (int j = 0; j < BJ; ++j)
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                int row = m0 + wr + i * 32 + (r & 3) + 8 * (r >> 2) + 4 * kgrp;
                int col = n0 + wc + j * 32 + ln31;
                if (row < M && col < N) C[(size_t)row * N + col] = acc[i][j][r];
            }
}

extern "C" __global__ void __launch_bounds__(512)
gemm_fp8k64_3buf_g16_kernel(const unsigned char* A, const unsigned char* Bt, float* C,
                            int M, int N, int K) {
    gemm_fp8k64_3buf_body<16>(A, Bt, C, M, N, K);
}
