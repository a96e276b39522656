// Microbench: one 16x64 fwd tile (64x256x256-GEMM slice) per workgroup,
// repeated; isolates the per-chunk pipeline cost of the persistent-step
// GEMM phases, with and without idle-workgroup poll pressure on a
// barrier go-flag (replicating the in-step environment).
//
// Build: hipcc -O3 --offload-arch=gfx950 scripts/tile_bench.hip -o scripts/tile_bench
// Run:   ./scripts/tile_bench [iters]

#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>

#define CHK(c) do { hipError_t e = (c); if (e) { \
    printf("HIP err %s @%d\n", hipGetErrorString(e), __LINE__); exit(1);} } while (0)

#define IN 256
#define OUT 256

// variant 0: depth-1 prefetch with copy (16-row tile, 4 accs)
// variant 1: depth-2 prefetch, two reg banks
// variant 2: no LDS staging — per-lane strided global reads
// variant 3: 4-row tile, single acc (the shipped p_fwd shape)
template <int V>
__global__ void __launch_bounds__(256, 1)
k_tile(const float* __restrict__ wt, float* __restrict__ y,
       int iters, int nactive, unsigned long long* flag) {
    __shared__ float xs[16 * IN];
    __shared__ float ws0[64 * 65];
    __shared__ float ws1[64 * 65];
    int tid = threadIdx.x;
    if ((int)blockIdx.x >= nactive) {
        if (tid == 0) {
            volatile unsigned long long* f = flag;
            long spins = 0;
            while (*f < 1ull) {
                if (++spins > (1L << 26)) break;
                if (spins < 8) __builtin_amdgcn_s_sleep(2);
                else __builtin_amdgcn_s_sleep(32);
            }
        }
        __syncthreads();
        return;
    }
    for (int e = tid; e < 16 * IN; e += 256) xs[e] = 0.001f * e;
    __syncthreads();
    int rq = tid >> 6, c = tid & 63;
    int kk16 = tid >> 6, cc16 = tid & 63;
    float sink = 0.f;
    for (int it = 0; it < iters; ++it) {
        float acc0 = 0.f, acc1 = 0.f, acc2 = 0.f, acc3 = 0.f;
        const float* xr0 = xs + (rq + 0) * IN;
        const float* xr1 = xs + (rq + 4) * IN;
        const float* xr2 = xs + (rq + 8) * IN;
        const float* xr3 = xs + (rq + 12) * IN;
        if (V == 2) {
            const float* wcol = wt + c;
#pragma unroll 8
            for (int k = 0; k < IN; ++k) {
                float wv = wcol[(long)k * OUT];
                acc0 += xr0[k] * wv; acc1 += xr1[k] * wv;
                acc2 += xr2[k] * wv; acc3 += xr3[k] * wv;
            }
        } else if (V == 0) {
            float wreg[16];
#pragma unroll
            for (int u = 0; u < 16; ++u)
                wreg[u] = wt[(long)(kk16 + 4 * u) * OUT + cc16];
            for (int ch = 0; ch < 4; ++ch) {
                int kc = ch << 6;
                float wb[16];
#pragma unroll
                for (int u = 0; u < 16; ++u) wb[u] = wreg[u];
                if (ch + 1 < 4) {
#pragma unroll
                    for (int u = 0; u < 16; ++u)
                        wreg[u] = wt[(long)(kc + 64 + kk16 + 4 * u) * OUT
                                     + cc16];
                }
                float* ws = (ch & 1) ? ws1 : ws0;
#pragma unroll
                for (int u = 0; u < 16; ++u)
                    ws[(kk16 + 4 * u) * 65 + cc16] = wb[u];
                __syncthreads();
#pragma unroll 8
                for (int k = 0; k < 64; ++k) {
                    float wv = ws[k * 65 + c];
                    acc0 += xr0[kc + k] * wv; acc1 += xr1[kc + k] * wv;
                    acc2 += xr2[kc + k] * wv; acc3 += xr3[kc + k] * wv;
                }
                __syncthreads();
            }
        } else if (V == 3) {
            // 4-row tile: wave rq owns row rq, single acc
            float acc = 0.f;
            const float* xr = xs + rq * IN;
            float wreg[16];
#pragma unroll
            for (int u = 0; u < 16; ++u)
                wreg[u] = wt[(long)(kk16 + 4 * u) * OUT + cc16];
            for (int ch = 0; ch < 4; ++ch) {
                int kc = ch << 6;
                float wb[16];
#pragma unroll
                for (int u = 0; u < 16; ++u) wb[u] = wreg[u];
                if (ch + 1 < 4) {
#pragma unroll
                    for (int u = 0; u < 16; ++u)
                        wreg[u] = wt[(long)(kc + 64 + kk16 + 4 * u) * OUT
                                     + cc16];
                }
                float* ws = (ch & 1) ? ws1 : ws0;
#pragma unroll
                for (int u = 0; u < 16; ++u)
                    ws[(kk16 + 4 * u) * 65 + cc16] = wb[u];
                __syncthreads();
#pragma unroll 8
                for (int k = 0; k < 64; ++k)
                    acc += xr[kc + k] * ws[k * 65 + c];
                __syncthreads();
            }
            acc0 = acc;
        } else {   // V == 1
            float wa[16], wb2[16];
#pragma unroll
            for (int u = 0; u < 16; ++u)
                wa[u] = wt[(long)(kk16 + 4 * u) * OUT + cc16];
#pragma unroll
            for (int u = 0; u < 16; ++u)
                wb2[u] = wt[(long)(64 + kk16 + 4 * u) * OUT + cc16];
            for (int ch = 0; ch < 4; ++ch) {
                int kc = ch << 6;
                float* ws = (ch & 1) ? ws1 : ws0;
                if ((ch & 1) == 0) {
#pragma unroll
                    for (int u = 0; u < 16; ++u)
                        ws[(kk16 + 4 * u) * 65 + cc16] = wa[u];
                    if (ch + 2 < 4) {
#pragma unroll
                        for (int u = 0; u < 16; ++u)
                            wa[u] = wt[(long)(kc + 128 + kk16 + 4 * u) * OUT
                                       + cc16];
                    }
                } else {
#pragma unroll
                    for (int u = 0; u < 16; ++u)
                        ws[(kk16 + 4 * u) * 65 + cc16] = wb2[u];
                    if (ch + 2 < 4) {
#pragma unroll
                        for (int u = 0; u < 16; ++u)
                            wb2[u] = wt[(long)(kc + 128 + kk16 + 4 * u) * OUT
                                        + cc16];
                    }
                }
                __syncthreads();
#pragma unroll 8
                for (int k = 0; k < 64; ++k) {
                    float wv = ws[k * 65 + c];
                    acc0 += xr0[kc + k] * wv; acc1 += xr1[kc + k] * wv;
                    acc2 += xr2[kc + k] * wv; acc3 += xr3[kc + k] * wv;
                }
                __syncthreads();
            }
        }
        sink += acc0 + acc1 + acc2 + acc3;
    }
    if (tid == 0 && blockIdx.x == 0) {
        y[0] = sink;
        __threadfence();
        atomicAdd(flag, 1ull);                 // release the pollers
    }
    __syncthreads();
}

template <int V>
static void run(const char* name, const float* wt, float* y, int iters,
                int nactive, unsigned long long* flag) {
    hipEvent_t a, b;
    CHK(hipEventCreate(&a)); CHK(hipEventCreate(&b));
    CHK(hipMemset(flag, 0, 8));
    hipLaunchKernelGGL(k_tile<V>, dim3(64), dim3(256), 0, 0, wt, y, 100,
                       nactive, flag);
    CHK(hipDeviceSynchronize());
    CHK(hipMemset(flag, 0, 8));
    CHK(hipEventRecord(a));
    hipLaunchKernelGGL(k_tile<V>, dim3(64), dim3(256), 0, 0, wt, y, iters,
                       nactive, flag);
    CHK(hipEventRecord(b));
    CHK(hipDeviceSynchronize());
    float ms;
    CHK(hipEventElapsedTime(&ms, a, b));
    printf("%-28s nactive=%2d  %8.3f us/tile\n", name, nactive,
           ms * 1e3 / iters);
    CHK(hipEventDestroy(a)); CHK(hipEventDestroy(b));
}

int main(int argc, char** argv) {
    int iters = argc > 1 ? atoi(argv[1]) : 5000;
    float* wt;
    float* y;
    unsigned long long* flag;
    CHK(hipMalloc(&wt, IN * OUT * 4));
    CHK(hipMalloc(&y, 256));
    CHK(hipMalloc(&flag, 64));
    CHK(hipMemset(wt, 0x3c, IN * OUT * 4));
    run<0>("v0 depth1+copy", wt, y, iters, 16, flag);
    run<0>("v0 depth1+copy all-busy", wt, y, iters, 64, flag);
    run<1>("v1 depth2", wt, y, iters, 16, flag);
    run<2>("v2 direct strided", wt, y, iters, 16, flag);
    run<2>("v2 direct strided all-busy", wt, y, iters, 64, flag);
    run<3>("v3 4-row single-acc", wt, y, iters, 64, flag);
    return 0;
}
