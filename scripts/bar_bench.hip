// Standalone microbenchmark for grid-barrier variants on gfx950.
// Build (locally, travels in the snapshot):
//   hipcc -O3 --offload-arch=gfx950 scripts/bar_bench.hip -o scripts/bar_bench
// Run on the GPU box: ./scripts/bar_bench [iters]
//
// Measures the per-barrier cost at PNWG=64 workgroups for:
//   v1 central : one atomic counter, every wg arrives AND polls it
//   v2 tree    : 8 group counters (padded lines) -> root -> one go-flag;
//                only the go-flag is polled (written once per round)
//   v3 flag32  : v2 at 32 wgs
// plus an empty-kernel launch reference.

#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>

#define CHK(c) do { hipError_t e = (c); if (e) { \
    printf("HIP err %s @%d\n", hipGetErrorString(e), __LINE__); exit(1);} } while (0)

__device__ inline void bar_central(unsigned long long* ctr,
                                   unsigned long long& tgt, int nwg) {
    __syncthreads();
    tgt += nwg;
    if (threadIdx.x == 0) {
        __threadfence();
        atomicAdd(ctr, 1ull);
        volatile unsigned long long* v = ctr;
        long spins = 0;
        while (*v < tgt) { __builtin_amdgcn_s_sleep(2);
            if (++spins > (1L << 28)) break; }
        __threadfence();
    }
    __syncthreads();
}

// gbar layout: [0..7]*16 group counters (128 B apart), [128] root, [144] flag
__device__ inline void bar_tree(unsigned long long* g,
                                unsigned long long& round, int nwg) {
    __syncthreads();
    round += 1;
    if (threadIdx.x == 0) {
        __threadfence();
        int grp = blockIdx.x & 7;
        int gsz = nwg >> 3;                   // wgs per group
        unsigned long long old = atomicAdd(&g[grp * 16], 1ull);
        if (old + 1 == round * gsz) {         // last of group
            unsigned long long r = atomicAdd(&g[128], 1ull);
            if (r + 1 == round * 8)           // last group overall
                __hip_atomic_store(&g[144], round, __ATOMIC_RELEASE,
                                   __HIP_MEMORY_SCOPE_AGENT);
        }
        volatile unsigned long long* f = &g[144];
        long spins = 0;
        while (*f < round) { __builtin_amdgcn_s_sleep(2);
            if (++spins > (1L << 28)) break; }
        __threadfence();
    }
    __syncthreads();
}

__global__ void __launch_bounds__(256, 1)
k_central(unsigned long long* ctr, int iters, int nwg) {
    unsigned long long tgt = 0;
    for (int i = 0; i < iters; ++i) bar_central(ctr, tgt, nwg);
}

__global__ void __launch_bounds__(256, 1)
k_tree(unsigned long long* g, int iters, int nwg) {
    unsigned long long round = 0;
    for (int i = 0; i < iters; ++i) bar_tree(g, round, nwg);
}

__global__ void k_empty() {}

static float timeit(hipEvent_t a, hipEvent_t b) {
    float ms; CHK(hipEventElapsedTime(&ms, a, b)); return ms;
}

int main(int argc, char** argv) {
    int iters = argc > 1 ? atoi(argv[1]) : 10000;
    unsigned long long* g;
    CHK(hipMalloc(&g, 4096));
    hipEvent_t a, b;
    CHK(hipEventCreate(&a)); CHK(hipEventCreate(&b));

    CHK(hipMemset(g, 0, 4096));
    hipLaunchKernelGGL(k_central, dim3(64), dim3(256), 0, 0, g, 100, 64);
    CHK(hipDeviceSynchronize());
    CHK(hipMemset(g, 0, 4096));
    CHK(hipEventRecord(a));
    hipLaunchKernelGGL(k_central, dim3(64), dim3(256), 0, 0, g, iters, 64);
    CHK(hipEventRecord(b));
    CHK(hipDeviceSynchronize());
    printf("central_64wg_us_per_bar %.3f\n", timeit(a, b) * 1e3 / iters);

    CHK(hipMemset(g, 0, 4096));
    hipLaunchKernelGGL(k_tree, dim3(64), dim3(256), 0, 0, g, 100, 64);
    CHK(hipDeviceSynchronize());
    CHK(hipMemset(g, 0, 4096));
    CHK(hipEventRecord(a));
    hipLaunchKernelGGL(k_tree, dim3(64), dim3(256), 0, 0, g, iters, 64);
    CHK(hipEventRecord(b));
    CHK(hipDeviceSynchronize());
    printf("tree_64wg_us_per_bar %.3f\n", timeit(a, b) * 1e3 / iters);

    CHK(hipMemset(g, 0, 4096));
    hipLaunchKernelGGL(k_tree, dim3(32), dim3(256), 0, 0, g, 100, 32);
    CHK(hipDeviceSynchronize());
    CHK(hipMemset(g, 0, 4096));
    CHK(hipEventRecord(a));
    hipLaunchKernelGGL(k_tree, dim3(32), dim3(256), 0, 0, g, iters, 32);
    CHK(hipEventRecord(b));
    CHK(hipDeviceSynchronize());
    printf("tree_32wg_us_per_bar %.3f\n", timeit(a, b) * 1e3 / iters);

    CHK(hipEventRecord(a));
    for (int i = 0; i < 1000; ++i)
        hipLaunchKernelGGL(k_empty, dim3(64), dim3(256), 0, 0);
    CHK(hipEventRecord(b));
    CHK(hipDeviceSynchronize());
    printf("empty_launch_us %.3f\n", timeit(a, b));
    return 0;
}
