// C ABI around the CDNA4 probe kernels (fabric_probe.hip).
//
// Consumed two ways:
//  * Python (ctypes) — k8s_dra_driver_gpu_amd/fabric/probe.py, for pytest
//    numerics checks against torch fp32 references and for bench.py fabric
//    validation;
//  * C++ — the xGMI fabric daemon (native/fabricd) links this directly for
//    its readiness probe (the nvidia-imex-ctl -q analog actually exercises
//    the fabric).
//
// All functions return >= 0 on success; negative values are -hipError_t.

#include <hip/hip_runtime.h>
#include <cstdio>
#include <ctime>
#include <cstring>
#include <vector>

#define PROBE_BLOCK 256
// >> 256 workgroups so all 8 XCDs fill regardless of the round-robin
// dispatcher (cdna_hip_programming.md 1: 256 CUs, blockIdx -> XCD b%8).
#define PROBE_GRID 4096
// HBM streaming sweet spot measured on MI355X (see profiles/): non-temporal
// float4 reads at 8192x256 -> 6.16 TB/s (98% of the ~6.3 TB/s achievable).
#define READ_GRID 8192
// writes/copies peak with non-temporal stores at a much larger grid
// (measured: write 5.67 TB/s @65536x256 > 5.5 @32768 > 5.2 @16384;
// copy 5.3 TB/s @16384; plain stores 4.9/4.6)
#define WRITE_GRID 65536

typedef float float4v __attribute__((ext_vector_type(4)));

extern "C" __global__ void hbm_read_kernel(const float4v*, float*, long);
extern "C" __global__ void hbm_read_nt_kernel(const float4v*, float*, long);
extern "C" __global__ void hbm_read_u8_kernel(const float4v*, float*, long);
extern "C" __global__ void hbm_read_chunk_kernel(const float4v*, float*, long);
extern "C" __global__ void hbm_write_kernel(float4v*, long, float);
extern "C" __global__ void hbm_write_nt_kernel(float4v*, long, float);
extern "C" __global__ void hbm_copy_nt_kernel(float4v*, const float4v*, long);
extern "C" __global__ void hbm_copy_kernel(float4v*, const float4v*, long);
extern "C" __global__ void hbm_block_sum_kernel(const float*, float*, long);
extern "C" __global__ void mfma_bf16_loop_kernel(const short*, float*, int);
extern "C" __global__ void mfma_fp8_loop_kernel(const int*, float*, int);
extern "C" __global__ void mfma_fp4_loop_kernel(const int*, float*, int);
extern "C" __global__ void mfma_fp4_tile_gemm_kernel(const unsigned char*, const unsigned char*, float*, int);
extern "C" __global__ void mfma_fp8_scaled_tile_kernel(const unsigned char*, const unsigned char*, const unsigned char*, const unsigned char*, float*, int);
extern "C" __global__ void mfma_scale_probe_kernel(float*);
extern "C" __global__ void mfma_fp4_scale_probe_kernel(float*);
extern "C" __global__ void mfma_fp4_scaled_tile_kernel(const unsigned char*, const unsigned char*, const unsigned char*, const unsigned char*, float*, int);
extern "C" __global__ void gemm_fp4_scaled_g16_swz_kernel(const unsigned char*, const unsigned char*, const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void mfma_bf16_tile_gemm_kernel(const short*, const short*, float*, int);
extern "C" __global__ void mfma_fp8_tile_gemm_kernel(const unsigned char*, const unsigned char*, float*, int, int);
extern "C" __global__ void p2p_read_kernel(float4v*, const float4v*, long);
extern "C" __global__ void vmfault_kernel(float*);
extern "C" __global__ void gemm_bf16_128_kernel(const short*, const short*, float*, int, int, int);
extern "C" __global__ void gemm_bf16_128_bk64_kernel(const short*, const short*, float*, int, int, int);
extern "C" __global__ void gemm_bf16_128_mfma32_kernel(const short*, const short*, float*, int, int, int);
extern "C" __global__ void gemm_bf16_128_mfma32_bk64_kernel(const short*, const short*, float*, int, int, int);
extern "C" __global__ void gemm_bf16_128_pipe_kernel(const short*, const short*, float*, int, int, int);
extern "C" __global__ void gemm_bf16_128_pipe_bk64_kernel(const short*, const short*, float*, int, int, int);
extern "C" __global__ void gemm_bf16_128_pipe2_kernel(const short*, const short*, float*, int, int, int);
extern "C" __global__ void gemm_bf16_128_splitk_kernel(const short*, const short*, float*, int, int, int, int);
extern "C" __global__ void gemm_bf16_128_pipe3_kernel(const short*, const short*, float*, int, int, int);
extern "C" __global__ void gemm_bf16_128_pipe2r_kernel(const short*, const short*, float*, int, int, int);
extern "C" __global__ void gemm_bf16_256_kernel(const short*, const short*, float*, int, int, int);
extern "C" __global__ void gemm_bf16_256x256_kernel(const short*, const short*, float*, int, int, int);
extern "C" __global__ void gemm_bf16_256x256_d2_kernel(const short*, const short*, float*, int, int, int);
extern "C" __global__ void gemm_bf16_256_bk64_kernel(const short*, const short*, float*, int, int, int);
extern "C" __global__ void gemm_fp8_128_kernel(const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp8_256_kernel(const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp8_256_g4_kernel(const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp8_256_g8_kernel(const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp8_256_g16_kernel(const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp8_256_g32_kernel(const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp8_256x256_kernel(const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp8_256x256_g16_kernel(const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp4_256x256_kernel(const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp4_256x256_g16_kernel(const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp4_3buf_g16_kernel(const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp8k64_3buf_g16_kernel(const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp8_256x256_g16_swz_kernel(const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp8_256x256_g16_swz_nb_kernel(const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp8_scaled_128_kernel(const unsigned char*, const unsigned char*, const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp8_scaled_256_g16_kernel(const unsigned char*, const unsigned char*, const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp8_scaled_256_g16_swz_kernel(const unsigned char*, const unsigned char*, const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp8_scaled_noload_kernel(const unsigned char*, const unsigned char*, const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp8_scaled4_g16_swz_kernel(const unsigned char*, const unsigned char*, const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp8_scaled_slds_g16_swz_kernel(const unsigned char*, const unsigned char*, const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp4_3buf_g16_swz_kernel(const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_fp4_512_g16_swz_kernel(const unsigned char*, const unsigned char*, float*, int, int, int);
extern "C" __global__ void gemm_bf16_256x256_d2_swz_kernel(const short*, const short*, float*, int, int, int);
extern "C" __global__ void gemm_bf16_256x256_d2_g4_kernel(const short*, const short*, float*, int, int, int);
extern "C" __global__ void gemm_bf16_256x256_d2_g8_kernel(const short*, const short*, float*, int, int, int);

// bk selector shared by the gemm entry points: 32/64 pick the 16x16x32
// tiling at that K-depth; 232/264 the 32x32x16 tiling (measured slower,
// kept as a data point); 332/364 the 3-buffer pipelined 16x16x32 tiling
// (counted vmcnt + raw barrier) at BK=32/64.
static inline void (*gemm_kern_for(int bk))(const short*, const short*, float*, int, int, int) {
    switch (bk) {
        case 64:  return gemm_bf16_128_bk64_kernel;
        case 232: return gemm_bf16_128_mfma32_kernel;
        case 264: return gemm_bf16_128_mfma32_bk64_kernel;
        case 332: return gemm_bf16_128_pipe_kernel;
        case 364: return gemm_bf16_128_pipe_bk64_kernel;
        case 432: return gemm_bf16_128_pipe2_kernel;
        case 532: return gemm_bf16_128_pipe3_kernel;
        case 632: return gemm_bf16_128_pipe2r_kernel;
        case 732: return gemm_bf16_256_kernel;
        case 764: return gemm_bf16_256_bk64_kernel;
        case 832: return gemm_bf16_256x256_kernel;
        case 842: return gemm_bf16_256x256_d2_kernel;
        case 844: return gemm_bf16_256x256_d2_g4_kernel;
        case 848: return gemm_bf16_256x256_d2_g8_kernel;
        case 852: return gemm_bf16_256x256_d2_swz_kernel;
        default:  return gemm_bf16_128_kernel;
    }
}

// launch geometry per kernel variant: (tile_m, tile_n, threads)
static inline void gemm_geom_for(int bk, int* tm, int* tn, int* threads) {
    if (bk == 732 || bk == 764) { *tm = 256; *tn = 128; *threads = 512; return; }
    if (bk == 832 || bk == 842 || bk == 844 || bk == 848 || bk == 852) { *tm = 256; *tn = 256; *threads = 512; return; }
    *tm = 128; *tn = 128; *threads = 256;
}
extern "C" __global__ void p2p_reduce_kernel(float4v*, const float4v*, long);

#define CHK(x)                                                                 \
    do {                                                                       \
        hipError_t _e = (x);                                                   \
        if (_e != hipSuccess) {                                                \
            fprintf(stderr, "fabricprobe: %s failed: %s\n", #x,                \
                    hipGetErrorString(_e));                                    \
            return -(double)_e;                                                \
        }                                                                      \
    } while (0)

#define CHKI(x)                                                                \
    do {                                                                       \
        hipError_t _e = (x);                                                   \
        if (_e != hipSuccess) {                                                \
            fprintf(stderr, "fabricprobe: %s failed: %s\n", #x,                \
                    hipGetErrorString(_e));                                    \
            return -(int)_e;                                                   \
        }                                                                      \
    } while (0)

extern "C" {

int fp_device_count(void) {
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) return 0;
    return n;
}

// Trigger a GPU VM page fault on `dev` (fault injection for the health
// monitor's end-to-end test; the process absorbs the resulting HIP error).
// Returns the hipDeviceSynchronize() error code so callers can tell whether
// the device actually reported the fault (0 = no error surfaced, which
// means the fault did NOT happen), or -1 if the launch machinery failed.
int fp_trigger_vmfault(int dev) {
    if (hipSetDevice(dev) != hipSuccess) return -1;
    float* sink = nullptr;
    if (hipMalloc(&sink, sizeof(float)) != hipSuccess) return -1;
    hipLaunchKernelGGL(vmfault_kernel, dim3(1), dim3(64), 0, 0, sink);
    hipError_t launch = hipGetLastError();
    if (launch != hipSuccess) {
        hipFree(sink);
        return -1;
    }
    hipError_t sync = hipDeviceSynchronize();
    hipFree(sink);
    return (int)sync;
}

// ---------------------------------------------------------------------------
// HBM bandwidth probes (self-contained: allocate, warm, time, free)
// ---------------------------------------------------------------------------

static double time_kernel_ms(hipEvent_t start, hipEvent_t stop) {
    float ms = 0.f;
    hipEventElapsedTime(&ms, start, stop);
    return (double)ms;
}

double fp_hbm_read_gbps(int dev, size_t bytes, int iters) {
    CHK(hipSetDevice(dev));
    long n_vec = (long)(bytes / sizeof(float4v));
    float4v* buf;
    float* sink;
    CHK(hipMalloc(&buf, n_vec * sizeof(float4v)));
    CHK(hipMalloc(&sink, sizeof(float)));
    CHK(hipMemset(buf, 0x3c, n_vec * sizeof(float4v)));
    hipEvent_t t0, t1;
    CHK(hipEventCreate(&t0));
    CHK(hipEventCreate(&t1));
    hipLaunchKernelGGL(hbm_read_nt_kernel, dim3(READ_GRID), dim3(PROBE_BLOCK), 0, 0, buf, sink, n_vec);
    CHK(hipGetLastError());
    CHK(hipDeviceSynchronize());
    CHK(hipEventRecord(t0));
    for (int i = 0; i < iters; ++i)
        hipLaunchKernelGGL(hbm_read_nt_kernel, dim3(READ_GRID), dim3(PROBE_BLOCK), 0, 0, buf, sink, n_vec);
    CHK(hipEventRecord(t1));
    CHK(hipEventSynchronize(t1));
    double ms = time_kernel_ms(t0, t1);
    hipFree(buf);
    hipFree(sink);
    hipEventDestroy(t0);
    hipEventDestroy(t1);
    return (double)bytes * iters / (ms * 1e6);  // GB/s
}

// Variant sweep: variant 0=plain u4, 1=nontemporal, 2=u8, 3=chunked;
// grid/block configurable for occupancy sweeps.
double fp_hbm_read_gbps_ex(int dev, size_t bytes, int iters, int grid, int block,
                           int variant) {
    CHK(hipSetDevice(dev));
    long n_vec = (long)(bytes / sizeof(float4v));
    float4v* buf;
    float* sink;
    CHK(hipMalloc(&buf, n_vec * sizeof(float4v)));
    CHK(hipMalloc(&sink, sizeof(float)));
    CHK(hipMemset(buf, 0x3c, n_vec * sizeof(float4v)));
    void (*kern)(const float4v*, float*, long) = hbm_read_kernel;
    if (variant == 1) kern = hbm_read_nt_kernel;
    if (variant == 2) kern = hbm_read_u8_kernel;
    if (variant == 3) kern = hbm_read_chunk_kernel;
    hipEvent_t t0, t1;
    CHK(hipEventCreate(&t0));
    CHK(hipEventCreate(&t1));
    hipLaunchKernelGGL(kern, dim3(grid), dim3(block), 0, 0, buf, sink, n_vec);
    CHK(hipGetLastError());
    CHK(hipDeviceSynchronize());
    CHK(hipEventRecord(t0));
    for (int i = 0; i < iters; ++i)
        hipLaunchKernelGGL(kern, dim3(grid), dim3(block), 0, 0, buf, sink, n_vec);
    CHK(hipEventRecord(t1));
    CHK(hipEventSynchronize(t1));
    double ms = time_kernel_ms(t0, t1);
    hipFree(buf);
    hipFree(sink);
    hipEventDestroy(t0);
    hipEventDestroy(t1);
    return (double)bytes * iters / (ms * 1e6);
}

double fp_hbm_write_gbps_ex(int dev, size_t bytes, int iters, int grid, int block,
                            int variant) {
    CHK(hipSetDevice(dev));
    long n_vec = (long)(bytes / sizeof(float4v));
    float4v* buf;
    CHK(hipMalloc(&buf, n_vec * sizeof(float4v)));
    void (*kern)(float4v*, long, float) = variant == 1 ? hbm_write_nt_kernel : hbm_write_kernel;
    hipEvent_t t0, t1;
    CHK(hipEventCreate(&t0));
    CHK(hipEventCreate(&t1));
    hipLaunchKernelGGL(kern, dim3(grid), dim3(block), 0, 0, buf, n_vec, 1.5f);
    CHK(hipGetLastError());
    CHK(hipDeviceSynchronize());
    CHK(hipEventRecord(t0));
    for (int i = 0; i < iters; ++i)
        hipLaunchKernelGGL(kern, dim3(grid), dim3(block), 0, 0, buf, n_vec, 2.5f);
    CHK(hipEventRecord(t1));
    CHK(hipEventSynchronize(t1));
    double ms = time_kernel_ms(t0, t1);
    hipFree(buf);
    hipEventDestroy(t0);
    hipEventDestroy(t1);
    return (double)bytes * iters / (ms * 1e6);
}

double fp_hbm_copy_gbps_ex(int dev, size_t bytes, int iters, int grid, int block,
                           int variant) {
    CHK(hipSetDevice(dev));
    long n_vec = (long)(bytes / sizeof(float4v));
    float4v *src, *dst;
    CHK(hipMalloc(&src, n_vec * sizeof(float4v)));
    CHK(hipMalloc(&dst, n_vec * sizeof(float4v)));
    CHK(hipMemset(src, 0x3c, n_vec * sizeof(float4v)));
    void (*kern)(float4v*, const float4v*, long) =
        variant == 1 ? hbm_copy_nt_kernel : hbm_copy_kernel;
    hipEvent_t t0, t1;
    CHK(hipEventCreate(&t0));
    CHK(hipEventCreate(&t1));
    hipLaunchKernelGGL(kern, dim3(grid), dim3(block), 0, 0, dst, src, n_vec);
    CHK(hipGetLastError());
    CHK(hipDeviceSynchronize());
    CHK(hipEventRecord(t0));
    for (int i = 0; i < iters; ++i)
        hipLaunchKernelGGL(kern, dim3(grid), dim3(block), 0, 0, dst, src, n_vec);
    CHK(hipEventRecord(t1));
    CHK(hipEventSynchronize(t1));
    double ms = time_kernel_ms(t0, t1);
    hipFree(src);
    hipFree(dst);
    hipEventDestroy(t0);
    hipEventDestroy(t1);
    return 2.0 * (double)bytes * iters / (ms * 1e6);
}

double fp_hbm_write_gbps(int dev, size_t bytes, int iters) {
    CHK(hipSetDevice(dev));
    long n_vec = (long)(bytes / sizeof(float4v));
    float4v* buf;
    CHK(hipMalloc(&buf, n_vec * sizeof(float4v)));
    hipEvent_t t0, t1;
    CHK(hipEventCreate(&t0));
    CHK(hipEventCreate(&t1));
    hipLaunchKernelGGL(hbm_write_nt_kernel, dim3(WRITE_GRID), dim3(PROBE_BLOCK), 0, 0, buf, n_vec, 1.5f);
    CHK(hipGetLastError());
    CHK(hipDeviceSynchronize());
    CHK(hipEventRecord(t0));
    for (int i = 0; i < iters; ++i)
        hipLaunchKernelGGL(hbm_write_nt_kernel, dim3(WRITE_GRID), dim3(PROBE_BLOCK), 0, 0, buf, n_vec, 2.5f);
    CHK(hipEventRecord(t1));
    CHK(hipEventSynchronize(t1));
    double ms = time_kernel_ms(t0, t1);
    hipFree(buf);
    hipEventDestroy(t0);
    hipEventDestroy(t1);
    return (double)bytes * iters / (ms * 1e6);
}

double fp_hbm_copy_gbps(int dev, size_t bytes, int iters) {
    CHK(hipSetDevice(dev));
    long n_vec = (long)(bytes / sizeof(float4v));
    float4v *src, *dst;
    CHK(hipMalloc(&src, n_vec * sizeof(float4v)));
    CHK(hipMalloc(&dst, n_vec * sizeof(float4v)));
    CHK(hipMemset(src, 0x3c, n_vec * sizeof(float4v)));
    hipEvent_t t0, t1;
    CHK(hipEventCreate(&t0));
    CHK(hipEventCreate(&t1));
    hipLaunchKernelGGL(hbm_copy_nt_kernel, dim3(WRITE_GRID), dim3(PROBE_BLOCK), 0, 0, dst, src, n_vec);
    CHK(hipGetLastError());
    CHK(hipDeviceSynchronize());
    CHK(hipEventRecord(t0));
    for (int i = 0; i < iters; ++i)
        hipLaunchKernelGGL(hbm_copy_nt_kernel, dim3(WRITE_GRID), dim3(PROBE_BLOCK), 0, 0, dst, src, n_vec);
    CHK(hipEventRecord(t1));
    CHK(hipEventSynchronize(t1));
    double ms = time_kernel_ms(t0, t1);
    hipFree(src);
    hipFree(dst);
    hipEventDestroy(t0);
    hipEventDestroy(t1);
    // copy moves 2x bytes (read + write)
    return 2.0 * (double)bytes * iters / (ms * 1e6);
}

// ---------------------------------------------------------------------------
// MFMA saturation probe
// ---------------------------------------------------------------------------

double fp_mfma_bf16_tflops(int dev, int inner_iters, int launches) {
    CHK(hipSetDevice(dev));
    short* seed;
    float* sink;
    CHK(hipMalloc(&seed, 1024 * sizeof(short)));
    CHK(hipMalloc(&sink, sizeof(float)));
    CHK(hipMemset(seed, 0x3d, 1024 * sizeof(short)));
    hipEvent_t t0, t1;
    CHK(hipEventCreate(&t0));
    CHK(hipEventCreate(&t1));
    hipLaunchKernelGGL(mfma_bf16_loop_kernel, dim3(PROBE_GRID), dim3(PROBE_BLOCK), 0, 0, seed, sink, 16);
    CHK(hipGetLastError());
    CHK(hipDeviceSynchronize());
    CHK(hipEventRecord(t0));
    for (int l = 0; l < launches; ++l)
        hipLaunchKernelGGL(mfma_bf16_loop_kernel, dim3(PROBE_GRID), dim3(PROBE_BLOCK), 0, 0, seed, sink, inner_iters);
    CHK(hipEventRecord(t1));
    CHK(hipEventSynchronize(t1));
    double ms = time_kernel_ms(t0, t1);
    // per wave-instruction: 2*32*32*16 FLOP; 4 MFMA per inner iter;
    // waves = grid*block/64
    double waves = (double)PROBE_GRID * PROBE_BLOCK / 64.0;
    double flops = waves * 4.0 * inner_iters * launches * 2.0 * 32 * 32 * 16;
    hipFree(seed);
    hipFree(sink);
    hipEventDestroy(t0);
    hipEventDestroy(t1);
    return flops / (ms * 1e9);  // TFLOP/s
}

// ---------------------------------------------------------------------------
// LDS-staged bf16 GEMM probe (gemm_probe.hip): throughput + host verify.
// ---------------------------------------------------------------------------

// Split-K bf16 GEMM for small shapes: ksplit partial products per output
// tile, accumulated with f32 hardware atomics into a zeroed C. The memset
// is part of each timed iteration (it is part of the op).
double fp_gemm_bf16_splitk_tflops_mnk(int dev, int M, int N, int K,
                                      int iters, int ksplit);

double fp_gemm_bf16_splitk_tflops(int dev, int size, int iters, int ksplit) {
    return fp_gemm_bf16_splitk_tflops_mnk(dev, size, size, size, iters, ksplit);
}

double fp_gemm_bf16_splitk_tflops_mnk(int dev, int M, int N, int K,
                                      int iters, int ksplit) {
    CHK(hipSetDevice(dev));
    short *A, *Bt;
    float* C;
    CHK(hipMalloc(&A, (size_t)M * K * 2));
    CHK(hipMalloc(&Bt, (size_t)N * K * 2));
    CHK(hipMalloc(&C, (size_t)M * N * sizeof(float)));
    CHK(hipMemset(A, 0x3f, (size_t)M * K * 2));
    CHK(hipMemset(Bt, 0x3f, (size_t)N * K * 2));
    dim3 grid((M / 128) * (N / 128) * ksplit);
    hipEvent_t t0, t1;
    CHK(hipEventCreate(&t0));
    CHK(hipEventCreate(&t1));
    CHK(hipMemsetAsync(C, 0, (size_t)M * N * sizeof(float)));
    hipLaunchKernelGGL(gemm_bf16_128_splitk_kernel, grid, dim3(256), 0, 0, A, Bt, C, M, N, K, ksplit);
    CHK(hipGetLastError());
    CHK(hipDeviceSynchronize());
    CHK(hipEventRecord(t0));
    for (int i = 0; i < iters; ++i) {
        CHK(hipMemsetAsync(C, 0, (size_t)M * N * sizeof(float)));
        hipLaunchKernelGGL(gemm_bf16_128_splitk_kernel, grid, dim3(256), 0, 0, A, Bt, C, M, N, K, ksplit);
    }
    CHK(hipEventRecord(t1));
    CHK(hipEventSynchronize(t1));
    double ms = time_kernel_ms(t0, t1);
    hipFree(A);
    hipFree(Bt);
    hipFree(C);
    hipEventDestroy(t0);
    hipEventDestroy(t1);
    return 2.0 * M * (double)N * K * iters / (ms * 1e9);
}

int fp_gemm_bf16_splitk_host(int dev, const short* A, const short* Bt,
                             float* C, int M, int N, int K, int ksplit) {
    CHKI(hipSetDevice(dev));
    short *dA, *dB;
    float* dC;
    CHKI(hipMalloc(&dA, (size_t)M * K * 2));
    CHKI(hipMalloc(&dB, (size_t)N * K * 2));
    CHKI(hipMalloc(&dC, (size_t)M * N * sizeof(float)));
    CHKI(hipMemcpy(dA, A, (size_t)M * K * 2, hipMemcpyHostToDevice));
    CHKI(hipMemcpy(dB, Bt, (size_t)N * K * 2, hipMemcpyHostToDevice));
    CHKI(hipMemset(dC, 0, (size_t)M * N * sizeof(float)));
    dim3 grid(((M + 127) / 128) * ((N + 127) / 128) * ksplit);
    hipLaunchKernelGGL(gemm_bf16_128_splitk_kernel, grid, dim3(256), 0, 0, dA, dB, dC, M, N, K, ksplit);
    CHKI(hipGetLastError());
    CHKI(hipDeviceSynchronize());
    CHKI(hipMemcpy(C, dC, (size_t)M * N * sizeof(float), hipMemcpyDeviceToHost));
    hipFree(dA);
    hipFree(dB);
    hipFree(dC);
    return 0;
}

double fp_gemm_bf16_tflops_ex(int dev, int size, int iters, int bk) {
    CHK(hipSetDevice(dev));
    int M = size, N = size, K = size;
    short *A, *Bt;
    float* C;
    CHK(hipMalloc(&A, (size_t)M * K * sizeof(short)));
    CHK(hipMalloc(&Bt, (size_t)N * K * sizeof(short)));
    CHK(hipMalloc(&C, (size_t)M * N * sizeof(float)));
    CHK(hipMemset(A, 0x3c, (size_t)M * K * sizeof(short)));
    CHK(hipMemset(Bt, 0x3b, (size_t)N * K * sizeof(short)));
    int tm, tn, threads;
    gemm_geom_for(bk, &tm, &tn, &threads);
    dim3 grid((M / tm) * (N / tn));
    auto kern = gemm_kern_for(bk);
    hipEvent_t t0, t1;
    CHK(hipEventCreate(&t0));
    CHK(hipEventCreate(&t1));
    hipLaunchKernelGGL(kern, grid, dim3(threads), 0, 0, A, Bt, C, M, N, K);
    CHK(hipGetLastError());
    CHK(hipDeviceSynchronize());
    CHK(hipEventRecord(t0));
    for (int i = 0; i < iters; ++i)
        hipLaunchKernelGGL(kern, grid, dim3(threads), 0, 0, A, Bt, C, M, N, K);
    CHK(hipEventRecord(t1));
    CHK(hipEventSynchronize(t1));
    double ms = time_kernel_ms(t0, t1);
    hipFree(A);
    hipFree(Bt);
    hipFree(C);
    hipEventDestroy(t0);
    hipEventDestroy(t1);
    return 2.0 * M * (double)N * K * iters / (ms * 1e9);
}

double fp_gemm_bf16_tflops(int dev, int size, int iters) {
    // measured ladder (profiles/, same-box pairs), TF @4096^3 / @8192^3:
    //   round 1 (128x128 tile, 4 waves):
    //     2-buf vmcnt(0)     851 / 897   3-buf depth-1  859-862 / 901-916
    //     4-buf depth-2      896-903 / 899-918    5-buf depth-3  895 / 846
    //   round 2 (big tiles, 8 waves, register-hoisted fragments):
    //     256x128 depth-1 (732)  1028-1055 / 1100-1187
    //     256x128 BK=64   (764)   691 / 742  (VGPR/unroll collapse, as r1)
    //     256x256 depth-1 (832)  1045-1080 / 1137-1187
    //     256x256 depth-2 (842)  1058-1084 / 1112-1180
    //     842 + XOR LDS swizzle (852)  1101 / 1171  <- default
    // The 256x256 tile halves LDS reads per MFMA (each wave 64x128 = 4x8
    // fragments, af[4]/bf[8] hoisted), runs 1 WG/CU (4 x 32 KiB LDS) with
    // two tiles in flight across each counted-vmcnt raw barrier, and the
    // XOR chunk swizzle de-conflicts the row-strided fragment reads
    // (fp8/fp4 gained +24/+15% from the same swizzle).
    return fp_gemm_bf16_tflops_ex(dev, size, iters, 852);
}

int fp_gemm_bf16_host_ex(int dev, const unsigned short* A, const unsigned short* Bt,
                         float* C, int M, int N, int K, int bk) {
    CHKI(hipSetDevice(dev));
    short *dA, *dB;
    float* dC;
    CHKI(hipMalloc(&dA, (size_t)M * K * sizeof(short)));
    CHKI(hipMalloc(&dB, (size_t)N * K * sizeof(short)));
    CHKI(hipMalloc(&dC, (size_t)M * N * sizeof(float)));
    CHKI(hipMemcpy(dA, A, (size_t)M * K * sizeof(short), hipMemcpyHostToDevice));
    CHKI(hipMemcpy(dB, Bt, (size_t)N * K * sizeof(short), hipMemcpyHostToDevice));
    int tm, tn, threads;
    gemm_geom_for(bk, &tm, &tn, &threads);
    dim3 grid(((M + tm - 1) / tm) * ((N + tn - 1) / tn));
    auto kern = gemm_kern_for(bk);
    hipLaunchKernelGGL(kern, grid, dim3(threads), 0, 0, dA, dB, dC, M, N, K);
    CHKI(hipGetLastError());
    CHKI(hipDeviceSynchronize());
    CHKI(hipMemcpy(C, dC, (size_t)M * N * sizeof(float), hipMemcpyDeviceToHost));
    hipFree(dA);
    hipFree(dB);
    hipFree(dC);
    return 0;
}

int fp_gemm_bf16_host(int dev, const unsigned short* A, const unsigned short* Bt,
                      float* C, int M, int N, int K) {
    CHKI(hipSetDevice(dev));
    short *dA, *dB;
    float* dC;
    CHKI(hipMalloc(&dA, (size_t)M * K * sizeof(short)));
    CHKI(hipMalloc(&dB, (size_t)N * K * sizeof(short)));
    CHKI(hipMalloc(&dC, (size_t)M * N * sizeof(float)));
    CHKI(hipMemcpy(dA, A, (size_t)M * K * sizeof(short), hipMemcpyHostToDevice));
    CHKI(hipMemcpy(dB, Bt, (size_t)N * K * sizeof(short), hipMemcpyHostToDevice));
    dim3 grid(((M + 127) / 128) * ((N + 127) / 128));
    hipLaunchKernelGGL(gemm_bf16_128_kernel, grid, dim3(256), 0, 0, dA, dB, dC, M, N, K);
    CHKI(hipGetLastError());
    CHKI(hipDeviceSynchronize());
    CHKI(hipMemcpy(C, dC, (size_t)M * N * sizeof(float), hipMemcpyDeviceToHost));
    hipFree(dA);
    hipFree(dB);
    hipFree(dC);
    return 0;
}

// ---------------------------------------------------------------------------
// Diagnostic burn: MFMA and HBM streaming CONCURRENTLY on two streams for
// ~duration_ms — the dcgmi-diag / GPU burn-test analog used by fabricd's
// deep health check. Writes achieved rates into out[2] = {TFLOPs, GB/s}.
// Returns 0 on success. Concurrency is real: matrix cores and the memory
// subsystem are exercised together, the worst-case power/thermal shape.
// ---------------------------------------------------------------------------

int fp_burn(int dev, int duration_ms, double* out_tflops, double* out_gbps) {
    CHKI(hipSetDevice(dev));
    size_t bytes = (size_t)2 << 30;
    long n_vec = (long)(bytes / sizeof(float4v));
    float4v* buf;
    short* seed;
    float* sink;
    CHKI(hipMalloc(&buf, bytes));
    CHKI(hipMalloc(&seed, 1024 * sizeof(short)));
    CHKI(hipMalloc(&sink, 2 * sizeof(float)));
    CHKI(hipMemset(buf, 0x3c, bytes));
    CHKI(hipMemset(seed, 0x3d, 1024 * sizeof(short)));
    hipStream_t s_mfma, s_hbm;
    CHKI(hipStreamCreate(&s_mfma));
    CHKI(hipStreamCreate(&s_hbm));
    // Use HALF the grid per workload so both co-reside on the chip.
    const int half_grid = 2048;
    const int mfma_iters = 512;  // ~0.5 ms per launch at half grid
    int launches = 0;
    hipEvent_t t0, t1;
    CHKI(hipEventCreate(&t0));
    CHKI(hipEventCreate(&t1));
    CHKI(hipEventRecord(t0));
    double target_s = duration_ms / 1000.0;
    // submit in small batches until the wall clock says stop
    struct timespec ts0, ts;
    clock_gettime(CLOCK_MONOTONIC, &ts0);
    while (true) {
        for (int i = 0; i < 4; ++i) {
            hipLaunchKernelGGL(mfma_bf16_loop_kernel, dim3(half_grid), dim3(PROBE_BLOCK), 0,
                               s_mfma, seed, sink, mfma_iters);
            hipLaunchKernelGGL(hbm_read_nt_kernel, dim3(half_grid), dim3(PROBE_BLOCK), 0,
                               s_hbm, buf, sink + 1, n_vec);
            launches++;
        }
        CHKI(hipGetLastError());
        CHKI(hipStreamSynchronize(s_mfma));
        CHKI(hipStreamSynchronize(s_hbm));
        clock_gettime(CLOCK_MONOTONIC, &ts);
        double el = (ts.tv_sec - ts0.tv_sec) + (ts.tv_nsec - ts0.tv_nsec) * 1e-9;
        if (el >= target_s) break;
    }
    CHKI(hipEventRecord(t1));
    CHKI(hipEventSynchronize(t1));
    float ms = 0.f;
    hipEventElapsedTime(&ms, t0, t1);
    double waves = (double)half_grid * PROBE_BLOCK / 64.0;
    // per kernel: waves x mfma_iters x 4 MFMA x 2*32*32*16 FLOP
    double flops = waves * (double)mfma_iters * 4.0 * launches * 2.0 * 32 * 32 * 16;
    if (out_tflops) *out_tflops = flops / (ms * 1e9);
    if (out_gbps) *out_gbps = (double)bytes * launches / (ms * 1e6);
    hipFree(buf);
    hipFree(seed);
    hipFree(sink);
    hipStreamDestroy(s_mfma);
    hipStreamDestroy(s_hbm);
    hipEventDestroy(t0);
    hipEventDestroy(t1);
    return 0;
}

// ---------------------------------------------------------------------------
// Numerics check entry points (host buffers in/out)
// ---------------------------------------------------------------------------

int fp_mfma_tile_gemm_host(int dev, const unsigned short* A, const unsigned short* B,
                           float* D, int K) {
    CHKI(hipSetDevice(dev));
    short *dA, *dB;
    float* dD;
    CHKI(hipMalloc(&dA, 16 * K * sizeof(short)));
    CHKI(hipMalloc(&dB, K * 16 * sizeof(short)));
    CHKI(hipMalloc(&dD, 16 * 16 * sizeof(float)));
    CHKI(hipMemcpy(dA, A, 16 * K * sizeof(short), hipMemcpyHostToDevice));
    CHKI(hipMemcpy(dB, B, K * 16 * sizeof(short), hipMemcpyHostToDevice));
    hipLaunchKernelGGL(mfma_bf16_tile_gemm_kernel, dim3(1), dim3(64), 0, 0, dA, dB, dD, K);
    CHKI(hipDeviceSynchronize());
    CHKI(hipMemcpy(D, dD, 16 * 16 * sizeof(float), hipMemcpyDeviceToHost));
    hipFree(dA);
    hipFree(dB);
    hipFree(dD);
    return 0;
}

int fp_mfma_fp8_tile_gemm_host(int dev, const unsigned char* A, const unsigned char* B,
                               float* D, int K, int layout) {
    CHKI(hipSetDevice(dev));
    unsigned char *dA, *dB;
    float* dD;
    CHKI(hipMalloc(&dA, (size_t)16 * K));
    CHKI(hipMalloc(&dB, (size_t)K * 16));
    CHKI(hipMalloc(&dD, 16 * 16 * sizeof(float)));
    CHKI(hipMemcpy(dA, A, (size_t)16 * K, hipMemcpyHostToDevice));
    CHKI(hipMemcpy(dB, B, (size_t)K * 16, hipMemcpyHostToDevice));
    hipLaunchKernelGGL(mfma_fp8_tile_gemm_kernel, dim3(1), dim3(64), 0, 0, dA, dB, dD, K, layout);
    CHKI(hipGetLastError());
    CHKI(hipDeviceSynchronize());
    CHKI(hipMemcpy(D, dD, 16 * 16 * sizeof(float), hipMemcpyDeviceToHost));
    hipFree(dA);
    hipFree(dB);
    hipFree(dD);
    return 0;
}

// MX-fp8 GEMM (scale=1): variant 1 = 128x128/256t, 2 = 256x128/512t
static inline void (*fp8_kern_for(int v))(const unsigned char*, const unsigned char*, float*, int, int, int) {
    switch (v) {
        case 2:  return gemm_fp8_256_kernel;
        case 24: return gemm_fp8_256_g4_kernel;
        case 28: return gemm_fp8_256_g8_kernel;
        case 216: return gemm_fp8_256_g16_kernel;
        case 232: return gemm_fp8_256_g32_kernel;
        case 3:   return gemm_fp8_256x256_kernel;
        case 316: return gemm_fp8_256x256_g16_kernel;
        case 4:   return gemm_fp4_256x256_kernel;
        case 416: return gemm_fp4_256x256_g16_kernel;
        case 436: return gemm_fp4_3buf_g16_kernel;
        case 446: return gemm_fp4_3buf_g16_swz_kernel;
        case 456: return gemm_fp4_512_g16_swz_kernel;
        case 336: return gemm_fp8k64_3buf_g16_kernel;
        case 326: return gemm_fp8_256x256_g16_swz_kernel;
        case 346: return gemm_fp8_256x256_g16_swz_nb_kernel;
        default: return gemm_fp8_128_kernel;
    }
}
static inline void fp8_geom_for(int v, int* tm, int* tn, int* threads) {
    if (v == 2 || v == 24 || v == 28 || v == 216 || v == 232) {
        *tm = 256; *tn = 128; *threads = 512; return;
    }
    if (v == 3 || v == 316 || v == 326 || v == 336 || v == 346 || v == 4 || v == 416 || v == 436 || v == 446) { *tm = 256; *tn = 256; *threads = 512; return; }
    if (v == 456) { *tm = 512; *tn = 256; *threads = 1024; return; }
    *tm = 128; *tn = 128; *threads = 256;
}

double fp_gemm_fp8_tflops_ex(int dev, int size, int iters, int variant) {
    CHK(hipSetDevice(dev));
    int M = size, N = size, K = size;
    const int fp4 = (variant == 4 || variant == 416 || variant == 436 || variant == 446 || variant == 456);
    unsigned char *A, *Bt;
    float* C;
    CHK(hipMalloc(&A, (size_t)M * K / (fp4 ? 2 : 1)));
    CHK(hipMalloc(&Bt, (size_t)N * K / (fp4 ? 2 : 1)));
    CHK(hipMalloc(&C, (size_t)M * N * sizeof(float)));
    CHK(hipMemset(A, fp4 ? 0x22 : 0x38, (size_t)M * K / (fp4 ? 2 : 1)));   // 1.0s
    CHK(hipMemset(Bt, fp4 ? 0x11 : 0x30, (size_t)N * K / (fp4 ? 2 : 1)));  // 0.5s
    int tm, tn, threads;
    fp8_geom_for(variant, &tm, &tn, &threads);
    dim3 grid((M / tm) * (N / tn));
    auto kern = fp8_kern_for(variant);
    hipEvent_t t0, t1;
    CHK(hipEventCreate(&t0));
    CHK(hipEventCreate(&t1));
    hipLaunchKernelGGL(kern, grid, dim3(threads), 0, 0, A, Bt, C, M, N, K);
    CHK(hipGetLastError());
    CHK(hipDeviceSynchronize());
    CHK(hipEventRecord(t0));
    for (int i = 0; i < iters; ++i)
        hipLaunchKernelGGL(kern, grid, dim3(threads), 0, 0, A, Bt, C, M, N, K);
    CHK(hipEventRecord(t1));
    CHK(hipEventSynchronize(t1));
    double ms = time_kernel_ms(t0, t1);
    hipFree(A);
    hipFree(Bt);
    hipFree(C);
    hipEventDestroy(t0);
    hipEventDestroy(t1);
    return 2.0 * M * (double)N * K * iters / (ms * 1e9);
}

double fp_mfma_fp4_tflops(int dev, int inner_iters, int launches) {
    CHK(hipSetDevice(dev));
    int* seed;
    float* sink;
    CHK(hipMalloc(&seed, 1024 * sizeof(int)));
    CHK(hipMalloc(&sink, sizeof(float)));
    CHK(hipMemset(seed, 0x11, 1024 * sizeof(int)));
    hipEvent_t t0, t1;
    CHK(hipEventCreate(&t0));
    CHK(hipEventCreate(&t1));
    hipLaunchKernelGGL(mfma_fp4_loop_kernel, dim3(PROBE_GRID), dim3(PROBE_BLOCK), 0, 0, seed, sink, inner_iters);
    CHK(hipGetLastError());
    CHK(hipDeviceSynchronize());
    CHK(hipEventRecord(t0));
    for (int l = 0; l < launches; ++l)
        hipLaunchKernelGGL(mfma_fp4_loop_kernel, dim3(PROBE_GRID), dim3(PROBE_BLOCK), 0, 0, seed, sink, inner_iters);
    CHK(hipEventRecord(t1));
    CHK(hipEventSynchronize(t1));
    double ms = time_kernel_ms(t0, t1);
    double waves = (double)PROBE_GRID * PROBE_BLOCK / 64.0;
    // 4 MFMA per inner iter, each 2*32*32*64 FLOP
    double flops = waves * 4.0 * inner_iters * launches * 2.0 * 32 * 32 * 64;
    hipFree(seed);
    hipFree(sink);
    hipEventDestroy(t0);
    hipEventDestroy(t1);
    return flops / (ms * 1e9);
}

int fp_mfma_scale_probe_host(int dev, float* D) {
    CHKI(hipSetDevice(dev));
    float* dD;
    CHKI(hipMalloc(&dD, 129 * 256 * sizeof(float)));
    hipLaunchKernelGGL(mfma_scale_probe_kernel, dim3(1), dim3(64), 0, 0, dD);
    CHKI(hipGetLastError());
    CHKI(hipDeviceSynchronize());
    CHKI(hipMemcpy(D, dD, 129 * 256 * sizeof(float), hipMemcpyDeviceToHost));
    hipFree(dD);
    return 0;
}

int fp_mfma_fp4_scale_probe_host(int dev, float* D) {
    CHKI(hipSetDevice(dev));
    float* dD;
    CHKI(hipMalloc(&dD, 129 * 1024 * sizeof(float)));
    hipLaunchKernelGGL(mfma_fp4_scale_probe_kernel, dim3(1), dim3(64), 0, 0, dD);
    CHKI(hipGetLastError());
    CHKI(hipDeviceSynchronize());
    CHKI(hipMemcpy(D, dD, 129 * 1024 * sizeof(float), hipMemcpyDeviceToHost));
    hipFree(dD);
    return 0;
}

int fp_mfma_fp8_scaled_tile_host(int dev, const unsigned char* A, const unsigned char* B,
                                 const unsigned char* SA, const unsigned char* SB,
                                 float* D, int K) {
    CHKI(hipSetDevice(dev));
    unsigned char *dA, *dB, *dSA, *dSB;
    float* dD;
    CHKI(hipMalloc(&dA, (size_t)16 * K));
    CHKI(hipMalloc(&dB, (size_t)K * 16));
    CHKI(hipMalloc(&dSA, (size_t)16 * K / 32));
    CHKI(hipMalloc(&dSB, (size_t)16 * K / 32));
    CHKI(hipMalloc(&dD, 16 * 16 * sizeof(float)));
    CHKI(hipMemcpy(dA, A, (size_t)16 * K, hipMemcpyHostToDevice));
    CHKI(hipMemcpy(dB, B, (size_t)K * 16, hipMemcpyHostToDevice));
    CHKI(hipMemcpy(dSA, SA, (size_t)16 * K / 32, hipMemcpyHostToDevice));
    CHKI(hipMemcpy(dSB, SB, (size_t)16 * K / 32, hipMemcpyHostToDevice));
    hipLaunchKernelGGL(mfma_fp8_scaled_tile_kernel, dim3(1), dim3(64), 0, 0, dA, dB, dSA, dSB, dD, K);
    CHKI(hipGetLastError());
    CHKI(hipDeviceSynchronize());
    CHKI(hipMemcpy(D, dD, 16 * 16 * sizeof(float), hipMemcpyDeviceToHost));
    hipFree(dA); hipFree(dB); hipFree(dSA); hipFree(dSB); hipFree(dD);
    return 0;
}

// MX-scaled fp8 GEMM: variant 5 = 128x128/256t, 52 = 256x128 G16/512t
static inline void (*fp8s_kern_for(int v))(const unsigned char*, const unsigned char*,
                                           const unsigned char*, const unsigned char*,
                                           float*, int, int, int) {
    if (v == 526) return gemm_fp8_scaled_256_g16_swz_kernel;
    if (v == 529) return gemm_fp8_scaled_noload_kernel;
    if (v == 546) return gemm_fp8_scaled4_g16_swz_kernel;
    if (v == 556) return gemm_fp8_scaled_slds_g16_swz_kernel;
    return v == 52 ? gemm_fp8_scaled_256_g16_kernel : gemm_fp8_scaled_128_kernel;
}
static inline void fp8s_geom_for(int v, int* tm, int* tn, int* threads) {
    if (v == 52 || v == 526 || v == 529 || v == 546 || v == 556) { *tm = 256; *tn = 128; *threads = 512; return; }
    *tm = 128; *tn = 128; *threads = 256;
}

int fp_gemm_fp8_scaled_host(int dev, const unsigned char* A, const unsigned char* Bt,
                            const unsigned char* SA, const unsigned char* SBt,
                            float* C, int M, int N, int K, int variant) {
    CHKI(hipSetDevice(dev));
    unsigned char *dA, *dB, *dSA, *dSB;
    float* dC;
    CHKI(hipMalloc(&dA, (size_t)M * K));
    CHKI(hipMalloc(&dB, (size_t)N * K));
    CHKI(hipMalloc(&dSA, (size_t)M * K / 32));
    CHKI(hipMalloc(&dSB, (size_t)N * K / 32));
    CHKI(hipMalloc(&dC, (size_t)M * N * sizeof(float)));
    CHKI(hipMemcpy(dA, A, (size_t)M * K, hipMemcpyHostToDevice));
    CHKI(hipMemcpy(dB, Bt, (size_t)N * K, hipMemcpyHostToDevice));
    CHKI(hipMemcpy(dSA, SA, (size_t)M * K / 32, hipMemcpyHostToDevice));
    CHKI(hipMemcpy(dSB, SBt, (size_t)N * K / 32, hipMemcpyHostToDevice));
    int tm, tn, threads;
    fp8s_geom_for(variant, &tm, &tn, &threads);
    dim3 grid(((M + tm - 1) / tm) * ((N + tn - 1) / tn));
    hipLaunchKernelGGL(fp8s_kern_for(variant), grid, dim3(threads), 0, 0,
                       dA, dB, dSA, dSB, dC, M, N, K);
    CHKI(hipGetLastError());
    CHKI(hipDeviceSynchronize());
    CHKI(hipMemcpy(C, dC, (size_t)M * N * sizeof(float), hipMemcpyDeviceToHost));
    hipFree(dA); hipFree(dB); hipFree(dSA); hipFree(dSB); hipFree(dC);
    return 0;
}

double fp_gemm_fp8_scaled_tflops(int dev, int size, int iters, int variant) {
    CHK(hipSetDevice(dev));
    int M = size, N = size, K = size;
    unsigned char *A, *Bt, *SA, *SBt;
    float* C;
    CHK(hipMalloc(&A, (size_t)M * K));
    CHK(hipMalloc(&Bt, (size_t)N * K));
    CHK(hipMalloc(&SA, (size_t)M * K / 32));
    CHK(hipMalloc(&SBt, (size_t)N * K / 32));
    CHK(hipMalloc(&C, (size_t)M * N * sizeof(float)));
    CHK(hipMemset(A, 0x38, (size_t)M * K));
    CHK(hipMemset(Bt, 0x30, (size_t)N * K));
    CHK(hipMemset(SA, 0x7F, (size_t)M * K / 32));   // e8m0 1.0
    CHK(hipMemset(SBt, 0x7F, (size_t)N * K / 32));
    int tm, tn, threads;
    fp8s_geom_for(variant, &tm, &tn, &threads);
    dim3 grid((M / tm) * (N / tn));
    auto kern = fp8s_kern_for(variant);
    hipEvent_t t0, t1;
    CHK(hipEventCreate(&t0));
    CHK(hipEventCreate(&t1));
    hipLaunchKernelGGL(kern, grid, dim3(threads), 0, 0, A, Bt, SA, SBt, C, M, N, K);
    CHK(hipGetLastError());
    CHK(hipDeviceSynchronize());
    CHK(hipEventRecord(t0));
    for (int i = 0; i < iters; ++i)
        hipLaunchKernelGGL(kern, grid, dim3(threads), 0, 0, A, Bt, SA, SBt, C, M, N, K);
    CHK(hipEventRecord(t1));
    CHK(hipEventSynchronize(t1));
    double ms = time_kernel_ms(t0, t1);
    hipFree(A); hipFree(Bt); hipFree(SA); hipFree(SBt); hipFree(C);
    hipEventDestroy(t0);
    hipEventDestroy(t1);
    return 2.0 * M * (double)N * K * iters / (ms * 1e9);
}

int fp_mfma_fp4_scaled_tile_host(int dev, const unsigned char* A, const unsigned char* B,
                                 const unsigned char* SA, const unsigned char* SB,
                                 float* D, int K) {
    CHKI(hipSetDevice(dev));
    unsigned char *dA, *dB, *dSA, *dSB;
    float* dD;
    CHKI(hipMalloc(&dA, (size_t)32 * K / 2));
    CHKI(hipMalloc(&dB, (size_t)K / 2 * 32));
    CHKI(hipMalloc(&dSA, (size_t)32 * K / 32));
    CHKI(hipMalloc(&dSB, (size_t)32 * K / 32));
    CHKI(hipMalloc(&dD, 32 * 32 * sizeof(float)));
    CHKI(hipMemcpy(dA, A, (size_t)32 * K / 2, hipMemcpyHostToDevice));
    CHKI(hipMemcpy(dB, B, (size_t)K / 2 * 32, hipMemcpyHostToDevice));
    CHKI(hipMemcpy(dSA, SA, (size_t)32 * K / 32, hipMemcpyHostToDevice));
    CHKI(hipMemcpy(dSB, SB, (size_t)32 * K / 32, hipMemcpyHostToDevice));
    hipLaunchKernelGGL(mfma_fp4_scaled_tile_kernel, dim3(1), dim3(64), 0, 0,
                       dA, dB, dSA, dSB, dD, K);
    CHKI(hipGetLastError());
    CHKI(hipDeviceSynchronize());
    CHKI(hipMemcpy(D, dD, 32 * 32 * sizeof(float), hipMemcpyDeviceToHost));
    hipFree(dA); hipFree(dB); hipFree(dSA); hipFree(dSB); hipFree(dD);
    return 0;
}

int fp_gemm_fp4_scaled_host(int dev, const unsigned char* A, const unsigned char* Bt,
                            const unsigned char* SA, const unsigned char* SBt,
                            float* C, int M, int N, int K) {
    CHKI(hipSetDevice(dev));
    unsigned char *dA, *dB, *dSA, *dSB;
    float* dC;
    CHKI(hipMalloc(&dA, (size_t)M * K / 2));
    CHKI(hipMalloc(&dB, (size_t)N * K / 2));
    CHKI(hipMalloc(&dSA, (size_t)M * K / 32));
    CHKI(hipMalloc(&dSB, (size_t)N * K / 32));
    CHKI(hipMalloc(&dC, (size_t)M * N * sizeof(float)));
    CHKI(hipMemcpy(dA, A, (size_t)M * K / 2, hipMemcpyHostToDevice));
    CHKI(hipMemcpy(dB, Bt, (size_t)N * K / 2, hipMemcpyHostToDevice));
    CHKI(hipMemcpy(dSA, SA, (size_t)M * K / 32, hipMemcpyHostToDevice));
    CHKI(hipMemcpy(dSB, SBt, (size_t)N * K / 32, hipMemcpyHostToDevice));
    dim3 grid(((M + 255) / 256) * ((N + 255) / 256));
    hipLaunchKernelGGL(gemm_fp4_scaled_g16_swz_kernel, grid, dim3(512), 0, 0,
                       dA, dB, dSA, dSB, dC, M, N, K);
    CHKI(hipGetLastError());
    CHKI(hipDeviceSynchronize());
    CHKI(hipMemcpy(C, dC, (size_t)M * N * sizeof(float), hipMemcpyDeviceToHost));
    hipFree(dA); hipFree(dB); hipFree(dSA); hipFree(dSB); hipFree(dC);
    return 0;
}

double fp_gemm_fp4_scaled_tflops(int dev, int size, int iters) {
    CHK(hipSetDevice(dev));
    int M = size, N = size, K = size;
    unsigned char *A, *Bt, *SA, *SBt;
    float* C;
    CHK(hipMalloc(&A, (size_t)M * K / 2));
    CHK(hipMalloc(&Bt, (size_t)N * K / 2));
    CHK(hipMalloc(&SA, (size_t)M * K / 32));
    CHK(hipMalloc(&SBt, (size_t)N * K / 32));
    CHK(hipMalloc(&C, (size_t)M * N * sizeof(float)));
    CHK(hipMemset(A, 0x22, (size_t)M * K / 2));
    CHK(hipMemset(Bt, 0x11, (size_t)N * K / 2));
    CHK(hipMemset(SA, 0x7F, (size_t)M * K / 32));
    CHK(hipMemset(SBt, 0x7F, (size_t)N * K / 32));
    dim3 grid((M / 256) * (N / 256));
    hipEvent_t t0, t1;
    CHK(hipEventCreate(&t0));
    CHK(hipEventCreate(&t1));
    hipLaunchKernelGGL(gemm_fp4_scaled_g16_swz_kernel, grid, dim3(512), 0, 0,
                       A, Bt, SA, SBt, C, M, N, K);
    CHK(hipGetLastError());
    CHK(hipDeviceSynchronize());
    CHK(hipEventRecord(t0));
    for (int i = 0; i < iters; ++i)
        hipLaunchKernelGGL(gemm_fp4_scaled_g16_swz_kernel, grid, dim3(512), 0, 0,
                           A, Bt, SA, SBt, C, M, N, K);
    CHK(hipEventRecord(t1));
    CHK(hipEventSynchronize(t1));
    double ms = time_kernel_ms(t0, t1);
    hipFree(A); hipFree(Bt); hipFree(SA); hipFree(SBt); hipFree(C);
    hipEventDestroy(t0);
    hipEventDestroy(t1);
    return 2.0 * M * (double)N * K * iters / (ms * 1e9);
}

int fp_mfma_fp4_tile_gemm_host(int dev, const unsigned char* A, const unsigned char* B,
                               float* D, int K) {
    CHKI(hipSetDevice(dev));
    unsigned char *dA, *dB;
    float* dD;
    CHKI(hipMalloc(&dA, (size_t)32 * K / 2));
    CHKI(hipMalloc(&dB, (size_t)K / 2 * 32));
    CHKI(hipMalloc(&dD, 32 * 32 * sizeof(float)));
    CHKI(hipMemcpy(dA, A, (size_t)32 * K / 2, hipMemcpyHostToDevice));
    CHKI(hipMemcpy(dB, B, (size_t)K / 2 * 32, hipMemcpyHostToDevice));
    hipLaunchKernelGGL(mfma_fp4_tile_gemm_kernel, dim3(1), dim3(64), 0, 0, dA, dB, dD, K);
    CHKI(hipGetLastError());
    CHKI(hipDeviceSynchronize());
    CHKI(hipMemcpy(D, dD, 32 * 32 * sizeof(float), hipMemcpyDeviceToHost));
    hipFree(dA);
    hipFree(dB);
    hipFree(dD);
    return 0;
}

double fp_mfma_fp8_tflops(int dev, int inner_iters, int launches) {
    CHK(hipSetDevice(dev));
    int* seed;
    float* sink;
    CHK(hipMalloc(&seed, 1024 * sizeof(int)));
    CHK(hipMalloc(&sink, sizeof(float)));
    CHK(hipMemset(seed, 0x11, 1024 * sizeof(int)));
    hipEvent_t t0, t1;
    CHK(hipEventCreate(&t0));
    CHK(hipEventCreate(&t1));
    hipLaunchKernelGGL(mfma_fp8_loop_kernel, dim3(PROBE_GRID), dim3(PROBE_BLOCK), 0, 0, seed, sink, inner_iters);
    CHK(hipGetLastError());
    CHK(hipDeviceSynchronize());
    CHK(hipEventRecord(t0));
    for (int l = 0; l < launches; ++l)
        hipLaunchKernelGGL(mfma_fp8_loop_kernel, dim3(PROBE_GRID), dim3(PROBE_BLOCK), 0, 0, seed, sink, inner_iters);
    CHK(hipEventRecord(t1));
    CHK(hipEventSynchronize(t1));
    double ms = time_kernel_ms(t0, t1);
    double waves = (double)PROBE_GRID * PROBE_BLOCK / 64.0;
    // 4 MFMA per inner iter, each 2*16*16*128 FLOP
    double flops = waves * 4.0 * inner_iters * launches * 2.0 * 16 * 16 * 128;
    hipFree(seed);
    hipFree(sink);
    hipEventDestroy(t0);
    hipEventDestroy(t1);
    return flops / (ms * 1e9);
}

int fp_gemm_fp8_host_ex(int dev, const unsigned char* A, const unsigned char* Bt,
                        float* C, int M, int N, int K, int variant) {
    CHKI(hipSetDevice(dev));
    const int den = (variant == 4 || variant == 416 || variant == 436 || variant == 446 || variant == 456) ? 2 : 1;  // fp4: packed bytes
    unsigned char *dA, *dB;
    float* dC;
    CHKI(hipMalloc(&dA, (size_t)M * K / den));
    CHKI(hipMalloc(&dB, (size_t)N * K / den));
    CHKI(hipMalloc(&dC, (size_t)M * N * sizeof(float)));
    CHKI(hipMemcpy(dA, A, (size_t)M * K / den, hipMemcpyHostToDevice));
    CHKI(hipMemcpy(dB, Bt, (size_t)N * K / den, hipMemcpyHostToDevice));
    int tm, tn, threads;
    fp8_geom_for(variant, &tm, &tn, &threads);
    dim3 grid(((M + tm - 1) / tm) * ((N + tn - 1) / tn));
    hipLaunchKernelGGL(fp8_kern_for(variant), grid, dim3(threads), 0, 0, dA, dB, dC, M, N, K);
    CHKI(hipGetLastError());
    CHKI(hipDeviceSynchronize());
    CHKI(hipMemcpy(C, dC, (size_t)M * N * sizeof(float), hipMemcpyDeviceToHost));
    hipFree(dA);
    hipFree(dB);
    hipFree(dC);
    return 0;
}

int fp_hbm_block_sum_host(int dev, const float* src, long n, float* out, int blocks) {
    CHKI(hipSetDevice(dev));
    float *dsrc, *dout;
    CHKI(hipMalloc(&dsrc, n * sizeof(float)));
    CHKI(hipMalloc(&dout, blocks * sizeof(float)));
    CHKI(hipMemcpy(dsrc, src, n * sizeof(float), hipMemcpyHostToDevice));
    hipLaunchKernelGGL(hbm_block_sum_kernel, dim3(blocks), dim3(PROBE_BLOCK), 0, 0, dsrc, dout, n);
    CHKI(hipDeviceSynchronize());
    CHKI(hipMemcpy(out, dout, blocks * sizeof(float), hipMemcpyDeviceToHost));
    hipFree(dsrc);
    hipFree(dout);
    return 0;
}

// ---------------------------------------------------------------------------
// xGMI p2p probes
// ---------------------------------------------------------------------------

double fp_p2p_read_gbps(int dst_dev, int src_dev, size_t bytes, int iters) {
    long n_vec = (long)(bytes / sizeof(float4v));
    CHK(hipSetDevice(src_dev));
    float4v* src;
    CHK(hipMalloc(&src, n_vec * sizeof(float4v)));
    CHK(hipMemset(src, 0x3c, n_vec * sizeof(float4v)));
    CHK(hipSetDevice(dst_dev));
    int can = 0;
    CHK(hipDeviceCanAccessPeer(&can, dst_dev, src_dev));
    if (!can) {
        hipSetDevice(src_dev);
        hipFree(src);
        return -1.0;
    }
    hipError_t pe = hipDeviceEnablePeerAccess(src_dev, 0);
    if (pe != hipSuccess && pe != hipErrorPeerAccessAlreadyEnabled) {
        fprintf(stderr, "fabricprobe: enable peer access %d->%d: %s\n", dst_dev,
                src_dev, hipGetErrorString(pe));
        hipSetDevice(src_dev);
        hipFree(src);
        return -(double)pe;
    }
    float4v* dst;
    CHK(hipMalloc(&dst, n_vec * sizeof(float4v)));
    hipEvent_t t0, t1;
    CHK(hipEventCreate(&t0));
    CHK(hipEventCreate(&t1));
    hipLaunchKernelGGL(p2p_read_kernel, dim3(PROBE_GRID), dim3(PROBE_BLOCK), 0, 0, dst, src, n_vec);
    CHK(hipDeviceSynchronize());
    CHK(hipEventRecord(t0));
    for (int i = 0; i < iters; ++i)
        hipLaunchKernelGGL(p2p_read_kernel, dim3(PROBE_GRID), dim3(PROBE_BLOCK), 0, 0, dst, src, n_vec);
    CHK(hipEventRecord(t1));
    CHK(hipEventSynchronize(t1));
    double ms = time_kernel_ms(t0, t1);
    hipFree(dst);
    CHK(hipSetDevice(src_dev));
    hipFree(src);
    hipEventDestroy(t0);
    hipEventDestroy(t1);
    return (double)bytes * iters / (ms * 1e6);
}

// All-GPU pull-reduce all-reduce probe over the xGMI full mesh.
// Each rank owns shard r of the vector; every rank pulls every peer's shard
// slice and reduces locally (reduce-scatter by direct pull), then pulls the
// reduced shards back (all-gather by direct pull). Full-mesh-optimal: every
// transfer goes over a direct xGMI link. Returns effective all-reduce
// bandwidth (algbw = bytes / time) in GB/s, or < 0 on error.
double fp_allreduce_pull_gbps(size_t bytes, int iters) {
    int n = fp_device_count();
    if (n < 2) return -1.0;
    long n_vec = (long)(bytes / sizeof(float4v));
    long shard = n_vec / n;
    if (shard == 0) return -2.0;
    std::vector<float4v*> bufs(n), acc(n);
    for (int d = 0; d < n; ++d) {
        CHK(hipSetDevice(d));
        for (int p = 0; p < n; ++p) {
            if (p == d) continue;
            hipError_t pe = hipDeviceEnablePeerAccess(p, 0);
            if (pe != hipSuccess && pe != hipErrorPeerAccessAlreadyEnabled) return -(double)pe;
        }
        CHK(hipMalloc(&bufs[d], n_vec * sizeof(float4v)));
        CHK(hipMalloc(&acc[d], shard * sizeof(float4v)));
        CHK(hipMemset(bufs[d], 0x3c, n_vec * sizeof(float4v)));
    }
    hipEvent_t t0, t1;
    CHK(hipSetDevice(0));
    CHK(hipEventCreate(&t0));
    CHK(hipEventCreate(&t1));
    CHK(hipEventRecord(t0));
    for (int it = 0; it < iters; ++it) {
        // reduce-scatter: device d reduces shard d from all peers
        for (int d = 0; d < n; ++d) {
            CHK(hipSetDevice(d));
            hipLaunchKernelGGL(hbm_copy_kernel, dim3(PROBE_GRID), dim3(PROBE_BLOCK), 0, 0,
                               acc[d], bufs[d] + (long)d * shard, shard);
            for (int p = 0; p < n; ++p) {
                if (p == d) continue;
                hipLaunchKernelGGL(p2p_reduce_kernel, dim3(PROBE_GRID), dim3(PROBE_BLOCK), 0, 0,
                                   acc[d], bufs[p] + (long)d * shard, shard);
            }
        }
        for (int d = 0; d < n; ++d) {
            CHK(hipSetDevice(d));
            CHK(hipDeviceSynchronize());
        }
        // all-gather: device d pulls reduced shard p from device p
        for (int d = 0; d < n; ++d) {
            CHK(hipSetDevice(d));
            for (int p = 0; p < n; ++p) {
                const float4v* srcp = acc[p];
                hipLaunchKernelGGL(p2p_read_kernel, dim3(PROBE_GRID), dim3(PROBE_BLOCK), 0, 0,
                                   bufs[d] + (long)p * shard, srcp, shard);
            }
        }
        for (int d = 0; d < n; ++d) {
            CHK(hipSetDevice(d));
            CHK(hipDeviceSynchronize());
        }
    }
    CHK(hipSetDevice(0));
    CHK(hipEventRecord(t1));
    CHK(hipEventSynchronize(t1));
    float ms = 0.f;
    hipEventElapsedTime(&ms, t0, t1);
    for (int d = 0; d < n; ++d) {
        hipSetDevice(d);
        hipFree(bufs[d]);
        hipFree(acc[d]);
    }
    hipEventDestroy(t0);
    hipEventDestroy(t1);
    return (double)bytes * iters / ((double)ms * 1e6);
}

}  // extern "C"
