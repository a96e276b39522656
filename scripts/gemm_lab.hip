// GEMM lab: times the wide-config MFMA kernels from engine.hip in
// isolation at the exact BASELINE-config-5 shapes, against the empirical
// 156 TF/s f32 MFMA peak (scripts/mfma_peak).  Used to attribute the
// 124 us (44%-of-peak) in-step H-GEMM time: kernel-inherent vs
// environment (L2 state, adjacent kernels).
//   hipcc --offload-arch=gfx950 -O3 scripts/gemm_lab.hip -o scripts/gemm_lab
#include "../d4pg_amd/ops/hip/engine.hip"

#include <cstdio>

using namespace d4pg;

#define CHECK(x) do { hipError_t e = (x); if (e) { \
    printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); \
    return 1; } } while (0)

static double time_ms(void (*fn)(void*), void* arg, int reps) {
    hipEvent_t e0, e1;
    hipEventCreate(&e0); hipEventCreate(&e1);
    hipEventRecord(e0);
    for (int i = 0; i < reps; ++i) fn(arg);
    hipEventRecord(e1);
    hipDeviceSynchronize();
    float ms = 0; hipEventElapsedTime(&ms, e0, e1);
    return ms / reps;
}

struct Shape { int B, in1, in2, out; const char* tag; };

int main() {
    const int reps = 50;
    Shape shapes[] = {
        {4096, 1024, 0, 1024, "fwd H-GEMM (L3)"},
        {4096, 1024, 6, 1024, "fwd H-GEMM concat (L2)"},
        {4096, 17, 0, 1024, "fwd L1 (k=17)"},
        {4096, 1024, 0, 51, "fwd head (n=51)"},
    };
    for (auto& s : shapes) {
        int in_total = s.in1 + s.in2;
        float *x1, *x2 = nullptr, *wt, *bias, *y;
        CHECK(hipMalloc(&x1, (size_t)s.B * s.in1 * 4));
        if (s.in2) CHECK(hipMalloc(&x2, (size_t)s.B * s.in2 * 4));
        CHECK(hipMalloc(&wt, (size_t)in_total * s.out * 4));
        CHECK(hipMalloc(&bias, s.out * 4));
        CHECK(hipMalloc(&y, (size_t)s.B * s.out * 4));
        CHECK(hipMemset(x1, 0, (size_t)s.B * s.in1 * 4));
        CHECK(hipMemset(wt, 0, (size_t)in_total * s.out * 4));
        int ntm = (s.B + MT_M - 1) / MT_M, ntn = (s.out + MT_N - 1) / MT_N;
        auto launch = [&]() {
            hipLaunchKernelGGL(k_mfma_fwd, dim3(ntm * ntn), dim3(256), 0, 0,
                               x1, x2, wt, bias, y, s.B, s.in1, s.in2,
                               s.out, ACT_RELU, 1, (float*)nullptr);
        };
        launch();
        CHECK(hipDeviceSynchronize());
        hipEvent_t e0, e1;
        hipEventCreate(&e0); hipEventCreate(&e1);
        hipEventRecord(e0);
        for (int i = 0; i < reps; ++i) launch();
        hipEventRecord(e1);
        CHECK(hipDeviceSynchronize());
        float ms = 0; hipEventElapsedTime(&ms, e0, e1);
        ms /= reps;
        double tf = 2.0 * s.B * in_total * s.out / (ms * 1e9);
        printf("%-26s grid %4d wgs: %8.2f us  %6.1f TF/s\n",
               s.tag, ntm * ntn, ms * 1000.0, tf);
        hipFree(x1); if (x2) hipFree(x2);
        hipFree(wt); hipFree(bias); hipFree(y);
    }
    // dW and dX at the hot shape
    {
        int B = 4096, in_total = 1024, out = 1024;
        float *dz, *x1, *wt, *gw, *gb, *dx, *h;
        CHECK(hipMalloc(&dz, (size_t)B * out * 4));
        CHECK(hipMalloc(&x1, (size_t)B * in_total * 4));
        CHECK(hipMalloc(&wt, (size_t)in_total * out * 4));
        CHECK(hipMalloc(&gw, (size_t)in_total * out * 4));
        CHECK(hipMalloc(&gb, out * 4));
        CHECK(hipMalloc(&dx, (size_t)B * in_total * 4));
        CHECK(hipMalloc(&h, (size_t)B * in_total * 4));
        CHECK(hipMemset(dz, 0, (size_t)B * out * 4));
        CHECK(hipMemset(x1, 0, (size_t)B * in_total * 4));
        int ntm = (in_total + MT_M - 1) / MT_M, ntn = (out + MT_N - 1) / MT_N;
        int ksplit = 1;
        while (ntm * ntn * ksplit < 256 && ksplit * 2 * MT_K <= B)
            ksplit *= 2;
        float* parts;
        CHECK(hipMalloc(&parts,
                        (size_t)ksplit * (in_total * out + out) * 4));
        hipEvent_t e0, e1; hipEventCreate(&e0); hipEventCreate(&e1);
        auto dw = [&]() {
            hipLaunchKernelGGL(k_mfma_dw, dim3(ntm * ntn * ksplit),
                               dim3(256), 0, 0, dz, x1, (const float*)nullptr,
                               gw, gb, B, in_total, 0, out, ksplit, parts);
            if (ksplit > 1)
                hipLaunchKernelGGL(k_dw_reduce, dim3(1024), dim3(256), 0, 0,
                                   parts, gw, gb, (long)in_total * out, out,
                                   ksplit);
        };
        dw(); CHECK(hipDeviceSynchronize());
        hipEventRecord(e0);
        for (int i = 0; i < reps; ++i) dw();
        hipEventRecord(e1); CHECK(hipDeviceSynchronize());
        float ms = 0; hipEventElapsedTime(&ms, e0, e1); ms /= reps;
        printf("%-26s grid %4d wgs: %8.2f us  %6.1f TF/s (ksplit=%d)\n",
               "dW H-GEMM", ntm * ntn * ksplit, ms * 1000.0,
               2.0 * B * in_total * out / (ms * 1e9), ksplit);
        int ntmx = (B + MT_M - 1) / MT_M, ntnx = (in_total + MT_N - 1) / MT_N;
        auto dxl = [&]() {
            hipLaunchKernelGGL(k_mfma_dx, dim3(ntmx * ntnx), dim3(256), 0, 0,
                               dz, wt, h, dx, B, 0, in_total, out, ACT_RELU);
        };
        dxl(); CHECK(hipDeviceSynchronize());
        hipEventRecord(e0);
        for (int i = 0; i < reps; ++i) dxl();
        hipEventRecord(e1); CHECK(hipDeviceSynchronize());
        hipEventElapsedTime(&ms, e0, e1); ms /= reps;
        printf("%-26s grid %4d wgs: %8.2f us  %6.1f TF/s\n", "dX H-GEMM",
               ntmx * ntnx, ms * 1000.0, 2.0 * B * in_total * out / (ms * 1e9));
    }
    return 0;
}
