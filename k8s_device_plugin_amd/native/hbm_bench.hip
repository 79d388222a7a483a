// HBM streaming-copy variant bench (standalone binary, gfx950).
//
// Finds the fastest copy formulation for the health probe's bandwidth
// check: plain float4 grid-stride vs unrolled vs nontemporal hints.
// Guide: MI355X_MICROARCH.md §HBM (6.29 TB/s measured float4 copy ceiling).
//
// Build: hipcc --offload-arch=gfx950 -O3 hbm_bench.hip -o hbm_bench
// Run:   ./hbm_bench [bytes]

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>

#define HIP_CHECK(x)                                                      \
    do {                                                                  \
        hipError_t e = (x);                                               \
        if (e != hipSuccess) {                                            \
            fprintf(stderr, "HIP error %s at %s:%d\n",                    \
                    hipGetErrorString(e), __FILE__, __LINE__);            \
            exit(1);                                                      \
        }                                                                 \
    } while (0)

__global__ void copy_plain(const float4 *__restrict__ src,
                           float4 *__restrict__ dst, size_t n) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) dst[i] = src[i];
}

__global__ void copy_unroll4(const float4 *__restrict__ src,
                             float4 *__restrict__ dst, size_t n) {
    size_t stride = (size_t)gridDim.x * blockDim.x;
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    // 4 independent loads in flight before the stores
    for (; i + 3 * stride < n; i += 4 * stride) {
        float4 a = src[i];
        float4 b = src[i + stride];
        float4 c = src[i + 2 * stride];
        float4 d = src[i + 3 * stride];
        dst[i] = a;
        dst[i + stride] = b;
        dst[i + 2 * stride] = c;
        dst[i + 3 * stride] = d;
    }
    for (; i < n; i += stride) dst[i] = src[i];
}

using f32x4 = __attribute__((ext_vector_type(4))) float;

__global__ void copy_nt(const float4 *__restrict__ src4,
                        float4 *__restrict__ dst4, size_t n) {
    const f32x4 *__restrict__ src = reinterpret_cast<const f32x4 *>(src4);
    f32x4 *__restrict__ dst = reinterpret_cast<f32x4 *>(dst4);
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride)
        __builtin_nontemporal_store(__builtin_nontemporal_load(&src[i]), &dst[i]);
}

__global__ void copy_nt_unroll4(const float4 *__restrict__ src4,
                                float4 *__restrict__ dst4, size_t n) {
    const f32x4 *__restrict__ src = reinterpret_cast<const f32x4 *>(src4);
    f32x4 *__restrict__ dst = reinterpret_cast<f32x4 *>(dst4);
    size_t stride = (size_t)gridDim.x * blockDim.x;
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (; i + 3 * stride < n; i += 4 * stride) {
        f32x4 a = __builtin_nontemporal_load(&src[i]);
        f32x4 b = __builtin_nontemporal_load(&src[i + stride]);
        f32x4 c = __builtin_nontemporal_load(&src[i + 2 * stride]);
        f32x4 d = __builtin_nontemporal_load(&src[i + 3 * stride]);
        __builtin_nontemporal_store(a, &dst[i]);
        __builtin_nontemporal_store(b, &dst[i + stride]);
        __builtin_nontemporal_store(c, &dst[i + 2 * stride]);
        __builtin_nontemporal_store(d, &dst[i + 3 * stride]);
    }
    for (; i < n; i += stride)
        __builtin_nontemporal_store(__builtin_nontemporal_load(&src[i]), &dst[i]);
}

// no-loop one-shot: thread t copies exactly 4 contiguous-stride elements
__global__ void copy_nt_oneshot(const float4 *__restrict__ src4,
                                float4 *__restrict__ dst4, size_t n) {
    const f32x4 *__restrict__ src = reinterpret_cast<const f32x4 *>(src4);
    f32x4 *__restrict__ dst = reinterpret_cast<f32x4 *>(dst4);
    size_t stride = (size_t)gridDim.x * blockDim.x;
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i + 3 * stride < n) {
        f32x4 a = __builtin_nontemporal_load(&src[i]);
        f32x4 b = __builtin_nontemporal_load(&src[i + stride]);
        f32x4 c = __builtin_nontemporal_load(&src[i + 2 * stride]);
        f32x4 d = __builtin_nontemporal_load(&src[i + 3 * stride]);
        __builtin_nontemporal_store(a, &dst[i]);
        __builtin_nontemporal_store(b, &dst[i + stride]);
        __builtin_nontemporal_store(c, &dst[i + 2 * stride]);
        __builtin_nontemporal_store(d, &dst[i + 3 * stride]);
    } else {
        for (; i < n; i += stride)
            __builtin_nontemporal_store(__builtin_nontemporal_load(&src[i]), &dst[i]);
    }
}

__global__ void fill(float4 *buf, size_t n) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride)
        buf[i] = make_float4((float)(i & 0xFFFF), 1.f, 2.f, 3.f);
}

template <typename K>
double bench(K kernel, const float4 *src, float4 *dst, size_t n, int blocks,
             int threads, int iters) {
    hipLaunchKernelGGL(kernel, dim3(blocks), dim3(threads), 0, 0, src, dst, n);
    HIP_CHECK(hipDeviceSynchronize());
    hipEvent_t t0, t1;
    HIP_CHECK(hipEventCreate(&t0));
    HIP_CHECK(hipEventCreate(&t1));
    HIP_CHECK(hipEventRecord(t0));
    for (int i = 0; i < iters; ++i)
        hipLaunchKernelGGL(kernel, dim3(blocks), dim3(threads), 0, 0, src, dst, n);
    HIP_CHECK(hipEventRecord(t1));
    HIP_CHECK(hipEventSynchronize(t1));
    float ms = 0;
    HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
    HIP_CHECK(hipEventDestroy(t0));
    HIP_CHECK(hipEventDestroy(t1));
    return (2.0 * n * sizeof(float4) * iters) / (ms * 1e6);  // GB/s
}

int main(int argc, char **argv) {
    size_t bytes = argc > 1 ? strtoull(argv[1], nullptr, 0) : (size_t)4 << 30;
    size_t n = bytes / sizeof(float4);
    float4 *src, *dst;
    HIP_CHECK(hipMalloc(&src, n * sizeof(float4)));
    HIP_CHECK(hipMalloc(&dst, n * sizeof(float4)));
    hipLaunchKernelGGL(fill, dim3(8192), dim3(256), 0, 0, src, n);
    HIP_CHECK(hipDeviceSynchronize());

    const int iters = 10;
    struct { const char *name; double gbps; } best{"", 0};
    for (int blocks : {8192, 16384, 32768, 65536, 131072}) {
        double p = bench(copy_plain, src, dst, n, blocks, 256, iters);
        double u = bench(copy_unroll4, src, dst, n, blocks, 256, iters);
        double t = bench(copy_nt, src, dst, n, blocks, 256, iters);
        double tu = bench(copy_nt_unroll4, src, dst, n, blocks, 256, iters);
        printf("blocks=%5d plain=%7.0f unroll4=%7.0f nt=%7.0f nt_unroll4=%7.0f GB/s\n",
               blocks, p, u, t, tu);
        if (tu > best.gbps) best = {"nt_unroll4", tu};
        if (p > best.gbps) best = {"plain", p};
        if (u > best.gbps) best = {"unroll4", u};
        if (t > best.gbps) best = {"nt", t};
        if (tu > best.gbps) best = {"nt_unroll4", tu};
    }
    {
        // exact-fit grid: every thread does its 4 elements, no loop
        int blocks = (int)(n / (256 * 4));
        double o = bench(copy_nt_oneshot, src, dst, n, blocks, 256, iters);
        printf("oneshot blocks=%d nt_oneshot=%7.0f GB/s\n", blocks, o);
        if (o > best.gbps) best = {"nt_oneshot", o};
    }
    printf("BEST %s %.0f GB/s\n", best.name, best.gbps);
    return 0;
}
