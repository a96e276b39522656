// Empirical gfx950 probes: (1) effective shader clock via issue-rate,
// (2) f32 MFMA peak throughput (v_mfma_f32_32x32x2_f32, accumulate-only,
// no memory traffic).  Grounds the wide-config GEMM bound analysis in
// measured silicon numbers instead of datasheet peaks.
//   hipcc --offload-arch=gfx950 -O3 scripts/mfma_peak.hip -o scripts/mfma_peak
#include <hip/hip_runtime.h>
#include <cstdio>

#define CHECK(x) do { hipError_t e = (x); if (e) { \
    printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); \
    return 1; } } while (0)

typedef float f32x16 __attribute__((ext_vector_type(16)));

// issue-rate clock probe: 8 independent FMA chains, issue-bound at
// 1 instr/cycle/SIMD (64-lane wave = 4 cycles/instr)
__global__ void k_clock(float* out, unsigned long long* t, int iters) {
    float a0=1.f,a1=1.f,a2=1.f,a3=1.f,a4=1.f,a5=1.f,a6=1.f,a7=1.f;
    float b = out[0] + 1e-9f;
    unsigned long long t0 = __builtin_amdgcn_s_memrealtime();
    for (int i = 0; i < iters; ++i) {
        a0 = fmaf(a0, b, 1e-9f); a1 = fmaf(a1, b, 1e-9f);
        a2 = fmaf(a2, b, 1e-9f); a3 = fmaf(a3, b, 1e-9f);
        a4 = fmaf(a4, b, 1e-9f); a5 = fmaf(a5, b, 1e-9f);
        a6 = fmaf(a6, b, 1e-9f); a7 = fmaf(a7, b, 1e-9f);
    }
    unsigned long long t1 = __builtin_amdgcn_s_memrealtime();
    if (threadIdx.x == 0)
        t[blockIdx.x] = t1 - t0;
    out[blockIdx.x] = a0+a1+a2+a3+a4+a5+a6+a7;
}

// MFMA peak: per wave, `accs` independent 32x32x2 f32 accumulation chains
template <int ACCS>
__global__ void k_mfma_peak(float* out, int iters) {
    f32x16 acc[ACCS];
    for (int j = 0; j < ACCS; ++j)
        for (int u = 0; u < 16; ++u) acc[j][u] = 0.f;
    float a = out[0] + 1e-9f, b = 1.00000001f;
    for (int i = 0; i < iters; ++i) {
#pragma unroll
        for (int j = 0; j < ACCS; ++j)
            acc[j] = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b, acc[j],
                                                          0, 0, 0);
    }
    float s = 0.f;
    for (int j = 0; j < ACCS; ++j) s += acc[j][threadIdx.x & 15];
    out[blockIdx.x * blockDim.x + threadIdx.x] = s;
}

// MFMA with operands streamed from LDS exactly like mfma_pipeline's ks
// loop (3 ds_read_b32 per 2 MFMAs, padded rows).  PIPE=0: naive reads
// right before use (what the compiler makes of it); PIPE=1: 2-stage
// manual operand prefetch (read ks+2's operands before ks's MFMAs issue).
template <int PIPE>
__global__ void k_mfma_lds(float* out, int iters) {
    __shared__ float As[32 * 68];
    __shared__ float Bs[32 * 132];
    int tid = threadIdx.x, wid = tid >> 6, lane = tid & 63;
    for (int e = tid; e < 32 * 68; e += blockDim.x) As[e] = 1e-9f;
    for (int e = tid; e < 32 * 132; e += blockDim.x) Bs[e] = 1e-9f;
    __syncthreads();
    int wm0 = (wid >> 1) * 32, wn0 = (wid & 1) * 64;
    int r = lane & 31, kk2 = lane >> 5;
    f32x16 a00{}, a01{};
    for (int it = 0; it < iters; ++it) {
        if (PIPE == 0) {
#pragma unroll
            for (int ks = 0; ks < 32; ks += 2) {
                float a0 = As[(ks + kk2) * 68 + wm0 + r];
                float b0 = Bs[(ks + kk2) * 132 + wn0 + r];
                float b1 = Bs[(ks + kk2) * 132 + wn0 + 32 + r];
                a00 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, a00,
                                                           0, 0, 0);
                a01 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, a01,
                                                           0, 0, 0);
            }
        } else {
            float a0 = As[kk2 * 68 + wm0 + r];
            float b0 = Bs[kk2 * 132 + wn0 + r];
            float b1 = Bs[kk2 * 132 + wn0 + 32 + r];
#pragma unroll
            for (int ks = 0; ks < 32; ks += 2) {
                float na = 0.f, nb0 = 0.f, nb1 = 0.f;
                if (ks + 2 < 32) {
                    na = As[(ks + 2 + kk2) * 68 + wm0 + r];
                    nb0 = Bs[(ks + 2 + kk2) * 132 + wn0 + r];
                    nb1 = Bs[(ks + 2 + kk2) * 132 + wn0 + 32 + r];
                }
                a00 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, a00,
                                                           0, 0, 0);
                a01 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, a01,
                                                           0, 0, 0);
                a0 = na; b0 = nb0; b1 = nb1;
            }
        }
    }
    float s = a00[lane & 15] + a01[lane & 15];
    out[blockIdx.x * blockDim.x + tid] = s;
}

template <int PIPE>
static int run_lds(int wgs, int iters, const char* tag) {
    float* out;
    CHECK(hipMalloc(&out, (size_t)wgs * 256 * 4));
    dim3 grid(wgs), block(256);
    hipLaunchKernelGGL((k_mfma_lds<PIPE>), grid, block, 0, 0, out, iters);
    CHECK(hipDeviceSynchronize());
    hipEvent_t e0, e1;
    hipEventCreate(&e0); hipEventCreate(&e1);
    hipEventRecord(e0);
    hipLaunchKernelGGL((k_mfma_lds<PIPE>), grid, block, 0, 0, out, iters);
    hipEventRecord(e1);
    CHECK(hipDeviceSynchronize());
    float ms = 0; hipEventElapsedTime(&ms, e0, e1);
    double flop = (double)wgs * 4 * 32.0 * (double)iters * 4096.0;
    printf("%-34s %4d wgs: %7.1f TF/s (%.2f ms)\n", tag, wgs,
           flop / ms / 1e9, ms);
    hipFree(out);
    return 0;
}

template <int ACCS>
static int run_mfma(int wgs, int waves_per_wg, int iters, const char* tag) {
    float* out;
    CHECK(hipMalloc(&out, (size_t)wgs * waves_per_wg * 64 * 4 + 4096));
    CHECK(hipMemset(out, 0, 4096));
    dim3 grid(wgs), block(waves_per_wg * 64);
    hipLaunchKernelGGL((k_mfma_peak<ACCS>), grid, block, 0, 0, out, iters);
    CHECK(hipDeviceSynchronize());
    hipEvent_t e0, e1;
    hipEventCreate(&e0); hipEventCreate(&e1);
    hipEventRecord(e0);
    hipLaunchKernelGGL((k_mfma_peak<ACCS>), grid, block, 0, 0, out, iters);
    hipEventRecord(e1);
    CHECK(hipDeviceSynchronize());
    float ms = 0; hipEventElapsedTime(&ms, e0, e1);
    double flop = (double)wgs * waves_per_wg * ACCS * (double)iters * 4096.0;
    printf("%-34s %4d wgs x %d waves, accs=%d: %7.1f TF/s (%.2f ms)\n",
           tag, wgs, waves_per_wg, ACCS, flop / ms / 1e9, ms);
    hipFree(out);
    return 0;
}

int main() {
    // clock probe: 1 wave per CU
    {
        float* out; unsigned long long* t;
        CHECK(hipMalloc(&out, 256 * 4));
        CHECK(hipMalloc(&t, 256 * 8));
        CHECK(hipMemset(out, 0, 256 * 4));
        int iters = 200000;
        hipLaunchKernelGGL(k_clock, dim3(256), dim3(64), 0, 0, out, t, iters);
        CHECK(hipDeviceSynchronize());
        hipLaunchKernelGGL(k_clock, dim3(256), dim3(64), 0, 0, out, t, iters);
        CHECK(hipDeviceSynchronize());
        unsigned long long h[256];
        CHECK(hipMemcpy(h, t, 256 * 8, hipMemcpyDeviceToHost));
        unsigned long long mx = 0;
        for (int i = 0; i < 256; ++i) if (h[i] > mx) mx = h[i];
        // 8 instr per iter, 4 cycles each (64-wide wave), memrealtime = 100 MHz
        double cycles = 8.0 * iters * 4.0;
        double secs = (double)mx / 100e6;
        printf("issue-rate effective clock: %.2f GHz\n", cycles / secs / 1e9);
    }
    run_mfma<2>(256, 4, 60000, "1 wg/CU (4 waves), 2 chains");
    run_mfma<2>(512, 4, 30000, "2 wgs/CU (8 waves), 2 chains");
    run_mfma<4>(256, 4, 30000, "1 wg/CU, 4 chains");
    run_mfma<4>(512, 4, 15000, "2 wgs/CU, 4 chains");
    run_mfma<8>(512, 4, 8000, "2 wgs/CU, 8 chains");
    run_lds<0>(512, 2000, "LDS operands, naive reads");
    run_lds<1>(512, 2000, "LDS operands, 2-stage prefetch");
    return 0;
}
