// bf16 MFMA peak-throughput variant sweep (standalone binary, gfx950).
//
// Finds the fastest formulation for the health probe's matrix-pipe check.
// v_mfma_f32_32x32x16_bf16 issues back-to-back at 32 cyc/SIMD (guide:
// MI355X_MICROARCH.md §Per-instruction constants); the sweep varies
// accumulator count (dependency cover) and wave geometry.
//
// Build: hipcc --offload-arch=gfx950 -O3 mfma_bench.hip -o mfma_bench

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <cstring>

#define HIP_CHECK(x)                                                      \
    do {                                                                  \
        hipError_t e = (x);                                               \
        if (e != hipSuccess) {                                            \
            fprintf(stderr, "HIP error %s at %d\n", hipGetErrorString(e), \
                    __LINE__);                                            \
            exit(1);                                                      \
        }                                                                 \
    } while (0)

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x16 = __attribute__((ext_vector_type(16))) float;

template <int NACC, bool ZERO = false>
__global__ void mfma_peak(float *out, int iters) {
#if defined(__gfx950__)
    int lane = threadIdx.x & 63;
    bf16x8 a, b;
    for (int r = 0; r < 8; ++r) {
        a[r] = ZERO ? (__bf16)0.0f : (__bf16)(float)(((lane + r) % 5) - 2);
        b[r] = ZERO ? (__bf16)0.0f : (__bf16)(float)(((lane * 3 + r) % 7) - 3);
    }
    f32x16 acc[NACC];
    for (int k = 0; k < NACC; ++k) acc[k] = f32x16{};
    for (int i = 0; i < iters; ++i) {
#pragma unroll
        for (int k = 0; k < NACC; ++k)
            acc[k] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc[k], 0, 0, 0);
    }
    float s = 0;
    for (int k = 0; k < NACC; ++k)
        for (int r = 0; r < 16; ++r) s += acc[k][r];
    if (s == -1.0f) out[blockIdx.x] = s;
#else
    (void)out; (void)iters;
#endif
}

template <int NACC, bool ZERO = false>
double bench(int blocks, int threads, int iters, float *d) {
    hipLaunchKernelGGL((mfma_peak<NACC, ZERO>), dim3(blocks), dim3(threads), 0, 0, d, 64);
    HIP_CHECK(hipDeviceSynchronize());
    hipEvent_t t0, t1;
    HIP_CHECK(hipEventCreate(&t0));
    HIP_CHECK(hipEventCreate(&t1));
    HIP_CHECK(hipEventRecord(t0));
    hipLaunchKernelGGL((mfma_peak<NACC, ZERO>), dim3(blocks), dim3(threads), 0, 0, d, iters);
    HIP_CHECK(hipEventRecord(t1));
    HIP_CHECK(hipEventSynchronize(t1));
    float ms = 0;
    HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
    HIP_CHECK(hipEventDestroy(t0));
    HIP_CHECK(hipEventDestroy(t1));
    double waves = (double)blocks * threads / 64;
    double flops = waves * iters * NACC * 2.0 * 32 * 32 * 16;
    return flops / (ms * 1e9);  // TF/s
}

int main() {
    hipDeviceProp_t props;
    HIP_CHECK(hipGetDeviceProperties(&props, 0));
    int cu = props.multiProcessorCount;
    float *d;
    HIP_CHECK(hipMalloc(&d, 65536 * sizeof(float)));
    const int iters = 8192;
    struct { char name[64]; double tf; } best{{0}, 0};
    struct Shape { int blocks, threads; const char *desc; };
    Shape shapes[] = {
        {cu, 512, "512x1/CU"},
        {cu * 2, 512, "512x2/CU"},
        {cu * 2, 256, "256x2/CU"},
        {cu * 4, 256, "256x4/CU"},
    };
    for (auto &sh : shapes) {
        double t2 = bench<2>(sh.blocks, sh.threads, iters, d);
        double t4 = bench<4>(sh.blocks, sh.threads, iters, d);
        double t8 = bench<8>(sh.blocks, sh.threads, iters, d);
        printf("%-10s acc2=%6.0f acc4=%6.0f acc8=%6.0f TF/s\n", sh.desc, t2, t4, t8);
        if (t2 > best.tf) { best.tf = t2; snprintf(best.name, 64, "%s acc2", sh.desc); }
        if (t4 > best.tf) { best.tf = t4; snprintf(best.name, 64, "%s acc4", sh.desc); }
        if (t8 > best.tf) { best.tf = t8; snprintf(best.name, 64, "%s acc8", sh.desc); }
    }
    printf("BEST %s %.0f TF/s\n", best.name, best.tf);
    // DVFS demonstration: zero operands draw less power -> higher clock
    double zr = bench<2, true>(cu * 2, 512, iters, d);
    printf("DVFS check: 512x2/CU acc2 zero-operands = %.0f TF/s\n", zr);
    return 0;
}
