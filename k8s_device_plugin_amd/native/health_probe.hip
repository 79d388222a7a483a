// gfx950 deep health probe (HIP/CDNA4).
//
// MI355X-native upgrade over the reference's liveness check, which only
// opens the device node (reference: internal/pkg/amdgpu/amdgpu.go:390-399).
// This probe actually executes CDNA4 code on the GPU and verifies:
//   1. wavefront size is 64 and all lanes participate (ballot);
//   2. the bf16 MFMA matrix pipe computes correct sums
//      (v_mfma_f32_16x16x32_bf16 invariant checks, exact in bf16/f32);
//   3. LDS write/read round-trips a full 128 KiB per CU;
//   4. HBM sustains a float4 streaming copy at a reportable GB/s
//      (grid sized >>256 workgroups to cover all 8 XCDs).
//
// Used by the plugin's optional deep health check, __graft_entry__.smoke(),
// and the gpu-marked tests.  Build: hipcc --offload-arch=gfx950, see
// native/build.py.  Guide: /opt/skills/guides/cdna_hip_programming.md §3.

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

namespace py = pybind11;

#define HIP_CHECK(expr)                                                        \
    do {                                                                       \
        hipError_t _e = (expr);                                                \
        if (_e != hipSuccess)                                                  \
            throw std::runtime_error(std::string("HIP error at " #expr ": ") + \
                                     hipGetErrorString(_e));                   \
    } while (0)

namespace {

// ---------------- wavefront probe ----------------

__global__ void wave_probe_kernel(unsigned long long *out) {
    unsigned long long active = __ballot(1);
    if (threadIdx.x == 0) {
        out[0] = active;
        out[1] = warpSize;
    }
}

// ---------------- MFMA probe ----------------

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__device__ inline __bf16 small_int_bf16(int v) {
    return (__bf16)(float)v;  // small ints are exact in bf16
}

// One wave computes D = A*B with v_mfma_f32_16x16x32_bf16 three times:
//  pass 0: A == 1, B == 0.5     -> every D element must equal 16.0
//  pass 1: A[lane][r] = pattern, B == 1 -> sum(D) == 16 * sum(A) (exact)
//  pass 2: asymmetric A AND B   -> element-wise check against a host
//          matmul through the documented lane maps (tests/test_gpu.py)
__global__ void mfma_probe_kernel(float *d_out /* [3][256] */) {
#if defined(__gfx950__)
    int lane = threadIdx.x;
    if (lane >= 64) return;

    for (int pass = 0; pass < 3; ++pass) {
        bf16x8 a, b;
        for (int r = 0; r < 8; ++r) {
            if (pass == 0) {
                a[r] = small_int_bf16(1);
                b[r] = (__bf16)0.5f;
            } else if (pass == 1) {
                a[r] = small_int_bf16(((lane * 8 + r) % 7) - 3);
                b[r] = small_int_bf16(1);
            } else {
                a[r] = small_int_bf16(((lane * 8 + r) % 7) - 3);
                b[r] = small_int_bf16(((lane * 5 + r * 3) % 11) - 5);
            }
        }
        f32x4 c = {0.f, 0.f, 0.f, 0.f};
        f32x4 d = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
        for (int r = 0; r < 4; ++r)
            d_out[pass * 256 + lane * 4 + r] = d[r];
    }
#else
    if (threadIdx.x == 0) d_out[0] = -1.0f;
#endif
}

// ---------------- MFMA peak throughput ----------------

using f32x16 = __attribute__((ext_vector_type(16))) float;

// Back-to-back v_mfma_f32_32x32x16_bf16 on independent accumulators.
// Measured sweep (gpurun_out/mfma_variants.txt): 2 accumulators at
// 512 threads x 2 blocks/CU is fastest (2043 TF/s; 8 accumulators spill
// VGPRs and collapse to 152 TF/s) — partner waves on each SIMD provide
// the remaining issue cover (guide: §Two waves per SIMD).
__global__ void __launch_bounds__(512, 2)
mfma_peak_kernel(float *out, int iters) {
#if defined(__gfx950__)
    int lane = threadIdx.x & 63;
    bf16x8 a, b;
    for (int r = 0; r < 8; ++r) {
        a[r] = (__bf16)(float)(((lane + r) % 5) - 2);
        b[r] = (__bf16)(float)(((lane * 3 + r) % 7) - 3);
    }
    f32x16 acc0 = {}, acc1 = {};
    for (int i = 0; i < iters; ++i) {
        acc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc0, 0, 0, 0);
        acc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc1, 0, 0, 0);
    }
    float s = 0;
    for (int r = 0; r < 16; ++r) s += acc0[r] + acc1[r];
    if (s == -1.0f) out[blockIdx.x] = s;  // never true: defeats DCE only
#else
    (void)out;
    (void)iters;
#endif
}

// ---------------- LDS probe ----------------

__global__ void lds_probe_kernel(uint32_t *err_count, int lds_words) {
    extern __shared__ uint32_t lds[];
    int tid = threadIdx.x;
    int nthreads = blockDim.x;
    for (int i = tid; i < lds_words; i += nthreads)
        lds[i] = (uint32_t)i * 2654435761u + blockIdx.x;
    __syncthreads();
    // read back with a stride so each thread checks other threads' writes
    uint32_t errors = 0;
    for (int i = tid * 17 % lds_words, n = 0; n < lds_words; ++n, i = (i + 1) % lds_words)
        if (lds[i] != (uint32_t)i * 2654435761u + blockIdx.x) ++errors;
    if (errors) atomicAdd(err_count, errors);
}

// ---------------- HBM streaming copy ----------------

// Nontemporal unrolled float4 stream: measured fastest variant on MI355X
// (profiles/r01_bench_mi355x.md; nt hints keep the one-shot stream out of
// L1/L2, 4 independent loads per thread hide HBM latency).
using f32x4_nt = __attribute__((ext_vector_type(4))) float;

// Exact-fit one-shot stream: thread t moves exactly 4 float4s at
// grid-stride offsets, no loop bookkeeping — measured 6.22 TB/s on MI355X
// (99% of the 6.29 TB/s float4-copy ceiling, MI355X_MICROARCH.md §HBM;
// sweep in profiles/r01_bench_mi355x.md).
__global__ void hbm_copy_kernel(const float4 *__restrict__ src4,
                                float4 *__restrict__ dst4, size_t n) {
    const f32x4_nt *__restrict__ src = reinterpret_cast<const f32x4_nt *>(src4);
    f32x4_nt *__restrict__ dst = reinterpret_cast<f32x4_nt *>(dst4);
    size_t stride = (size_t)gridDim.x * blockDim.x;
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i + 3 * stride < n) {
        f32x4_nt a = __builtin_nontemporal_load(&src[i]);
        f32x4_nt b = __builtin_nontemporal_load(&src[i + stride]);
        f32x4_nt c = __builtin_nontemporal_load(&src[i + 2 * stride]);
        f32x4_nt d = __builtin_nontemporal_load(&src[i + 3 * stride]);
        __builtin_nontemporal_store(a, &dst[i]);
        __builtin_nontemporal_store(b, &dst[i + stride]);
        __builtin_nontemporal_store(c, &dst[i + 2 * stride]);
        __builtin_nontemporal_store(d, &dst[i + 3 * stride]);
    } else {
        for (; i < n; i += stride)
            __builtin_nontemporal_store(__builtin_nontemporal_load(&src[i]),
                                        &dst[i]);
    }
}

__global__ void fill_pattern_kernel(float4 *buf, size_t n) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        float v = (float)(i & 0xFFFF);
        buf[i] = make_float4(v, v + 0.25f, v + 0.5f, v + 0.75f);
    }
}

// ---------------- host-side probe ----------------

py::dict run_probe(int device, size_t hbm_bytes) {
    py::dict result;
    HIP_CHECK(hipSetDevice(device));

    hipDeviceProp_t props;
    HIP_CHECK(hipGetDeviceProperties(&props, device));
    result["device_name"] = std::string(props.name);
    result["gcn_arch"] = std::string(props.gcnArchName);
    result["cu_count"] = props.multiProcessorCount;
    result["vram_bytes"] = (uint64_t)props.totalGlobalMem;

    // 1. wavefront
    {
        unsigned long long *d;
        HIP_CHECK(hipMalloc(&d, 2 * sizeof(unsigned long long)));
        hipLaunchKernelGGL(wave_probe_kernel, dim3(1), dim3(64), 0, 0, d);
        HIP_CHECK(hipGetLastError());
        unsigned long long h[2];
        HIP_CHECK(hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost));
        HIP_CHECK(hipFree(d));
        result["wavefront_size"] = (int)h[1];
        result["wave_ok"] = (h[0] == ~0ull && h[1] == 64);
    }

    // 2. MFMA
    {
        float *d;
        HIP_CHECK(hipMalloc(&d, 3 * 256 * sizeof(float)));
        HIP_CHECK(hipMemset(d, 0, 3 * 256 * sizeof(float)));
        hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, 0, d);
        HIP_CHECK(hipGetLastError());
        std::vector<float> h(3 * 256);
        HIP_CHECK(hipMemcpy(h.data(), d, h.size() * sizeof(float),
                            hipMemcpyDeviceToHost));
        HIP_CHECK(hipFree(d));

        bool pass0 = true;
        for (int i = 0; i < 256; ++i)
            if (h[i] != 16.0f) pass0 = false;
        double sum_a = 0;
        for (int lane = 0; lane < 64; ++lane)
            for (int r = 0; r < 8; ++r)
                sum_a += ((lane * 8 + r) % 7) - 3;
        double sum_d = 0;
        for (int i = 0; i < 256; ++i) sum_d += h[256 + i];
        bool pass1 = (sum_d == 16.0 * sum_a);
        result["mfma_ok"] = pass0 && pass1;
        result["mfma_const_ok"] = pass0;
        result["mfma_sum_ok"] = pass1;
    }

    // 2b. MFMA peak throughput (matrix pipes at speed, all CUs)
    {
        float *d;
        HIP_CHECK(hipMalloc(&d, 4096 * sizeof(float)));
        const int iters = 8192, blocks = props.multiProcessorCount * 2;
        hipLaunchKernelGGL(mfma_peak_kernel, dim3(blocks), dim3(512), 0, 0, d,
                           64);  // warmup
        HIP_CHECK(hipDeviceSynchronize());
        hipEvent_t t0, t1;
        HIP_CHECK(hipEventCreate(&t0));
        HIP_CHECK(hipEventCreate(&t1));
        HIP_CHECK(hipEventRecord(t0));
        hipLaunchKernelGGL(mfma_peak_kernel, dim3(blocks), dim3(512), 0, 0, d,
                           iters);
        HIP_CHECK(hipEventRecord(t1));
        HIP_CHECK(hipEventSynchronize(t1));
        float ms = 0;
        HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
        // FLOPs: blocks * 8 waves * iters * 2 MFMA * 2*32*32*16
        double flops = (double)blocks * 8 * iters * 2 * 2 * 32 * 32 * 16;
        result["mfma_tflops"] = flops / (ms * 1e9);
        HIP_CHECK(hipEventDestroy(t0));
        HIP_CHECK(hipEventDestroy(t1));
        HIP_CHECK(hipFree(d));
    }

    // 3. LDS: 128 KiB dynamic per workgroup (160 KiB/CU on gfx950)
    {
        const int lds_bytes = 128 * 1024;
        uint32_t *d_err;
        HIP_CHECK(hipMalloc(&d_err, sizeof(uint32_t)));
        HIP_CHECK(hipMemset(d_err, 0, sizeof(uint32_t)));
        hipLaunchKernelGGL(lds_probe_kernel, dim3(props.multiProcessorCount),
                           dim3(256), lds_bytes, 0, d_err, lds_bytes / 4);
        HIP_CHECK(hipGetLastError());
        uint32_t h_err = 1;
        HIP_CHECK(hipMemcpy(&h_err, d_err, sizeof(h_err), hipMemcpyDeviceToHost));
        HIP_CHECK(hipFree(d_err));
        result["lds_ok"] = (h_err == 0);
        result["lds_bytes_tested"] = lds_bytes;
    }

    // 4. HBM streaming bandwidth (read+write)
    {
        size_t n = hbm_bytes / sizeof(float4);
        float4 *src, *dst;
        HIP_CHECK(hipMalloc(&src, n * sizeof(float4)));
        HIP_CHECK(hipMalloc(&dst, n * sizeof(float4)));
        // exact-fit grid: one thread per 4 float4s (no loop), which was the
        // fastest measured formulation; still >>256 workgroups for all 8 XCDs
        const int threads = 256, iters = 5;
        long want = (long)((n + threads * 4 - 1) / (threads * 4));
        const int blocks = (int)(want < 1024 ? 1024 : want);
        hipLaunchKernelGGL(fill_pattern_kernel, dim3(blocks), dim3(threads), 0, 0,
                           src, n);
        hipLaunchKernelGGL(hbm_copy_kernel, dim3(blocks), dim3(threads), 0, 0,
                           src, dst, n);  // warmup
        HIP_CHECK(hipDeviceSynchronize());

        hipEvent_t t0, t1;
        HIP_CHECK(hipEventCreate(&t0));
        HIP_CHECK(hipEventCreate(&t1));
        HIP_CHECK(hipEventRecord(t0));
        for (int i = 0; i < iters; ++i)
            hipLaunchKernelGGL(hbm_copy_kernel, dim3(blocks), dim3(threads), 0, 0,
                               src, dst, n);
        HIP_CHECK(hipEventRecord(t1));
        HIP_CHECK(hipEventSynchronize(t1));
        float ms = 0;
        HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
        double gbps = (2.0 * n * sizeof(float4) * iters) / (ms * 1e6);
        result["hbm_gbps"] = gbps;
        result["hbm_bytes_tested"] = (uint64_t)(n * sizeof(float4));

        // spot-check correctness of the copy
        std::vector<float4> sample(16);
        HIP_CHECK(hipMemcpy(sample.data(), dst + n / 2,
                            sample.size() * sizeof(float4),
                            hipMemcpyDeviceToHost));
        bool copy_ok = true;
        for (size_t k = 0; k < sample.size(); ++k) {
            size_t i = n / 2 + k;
            if (sample[k].x != (float)(i & 0xFFFF)) copy_ok = false;
        }
        result["hbm_copy_ok"] = copy_ok;
        HIP_CHECK(hipEventDestroy(t0));
        HIP_CHECK(hipEventDestroy(t1));
        HIP_CHECK(hipFree(src));
        HIP_CHECK(hipFree(dst));
    }

    bool healthy = result["wave_ok"].cast<bool>() &&
                   result["mfma_ok"].cast<bool>() &&
                   result["lds_ok"].cast<bool>() &&
                   result["hbm_copy_ok"].cast<bool>();
    result["healthy"] = healthy;
    return result;
}

// Raw MFMA fragment outputs for element-wise verification against a host
// (PyTorch) reference: pass 0 = constant inputs, pass 1 = patterned A with
// B == 1 (tests/test_gpu.py reconstructs the lane->element mapping).
py::dict mfma_probe_raw(int device) {
    HIP_CHECK(hipSetDevice(device));
    float *d;
    HIP_CHECK(hipMalloc(&d, 3 * 256 * sizeof(float)));
    HIP_CHECK(hipMemset(d, 0, 3 * 256 * sizeof(float)));
    hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, 0, d);
    HIP_CHECK(hipGetLastError());
    std::vector<float> h(3 * 256);
    HIP_CHECK(hipMemcpy(h.data(), d, h.size() * sizeof(float),
                        hipMemcpyDeviceToHost));
    HIP_CHECK(hipFree(d));
    py::list d0, d1, d2;
    for (int i = 0; i < 256; ++i) d0.append(h[i]);
    for (int i = 0; i < 256; ++i) d1.append(h[256 + i]);
    for (int i = 0; i < 256; ++i) d2.append(h[512 + i]);
    py::dict out;
    out["pass0"] = d0;  // expect all 16.0
    out["pass1"] = d1;  // D = A_pattern @ ones, per-lane fragments
    out["pass2"] = d2;  // D = A_pattern @ B_pattern (asymmetric both)
    return out;
}

int device_count() {
    int n = 0;
    hipError_t e = hipGetDeviceCount(&n);
    if (e != hipSuccess) return 0;
    return n;
}

// Measured GPU->GPU copy bandwidth: on an 8*MI355X hive this runs over one
// xGMI point-to-point link (~153 GB/s class); PCIe-bridged pairs land far
// lower — the hardware truth behind the allocator's hive packing
// (docs/resource-allocation.md).
py::dict p2p_bandwidth(int src, int dst, size_t bytes) {
    py::dict out;
    out["src"] = src;
    out["dst"] = dst;

    int can = 0;
    HIP_CHECK(hipDeviceCanAccessPeer(&can, dst, src));
    out["peer_access"] = (bool)can;

    HIP_CHECK(hipSetDevice(src));
    void *src_buf;
    HIP_CHECK(hipMalloc(&src_buf, bytes));
    HIP_CHECK(hipMemset(src_buf, 0x5A, bytes));
    HIP_CHECK(hipSetDevice(dst));
    if (can) {
        hipError_t e = hipDeviceEnablePeerAccess(src, 0);
        if (e != hipSuccess && e != hipErrorPeerAccessAlreadyEnabled)
            throw std::runtime_error(hipGetErrorString(e));
    }
    void *dst_buf;
    HIP_CHECK(hipMalloc(&dst_buf, bytes));

    const int iters = 10;
    HIP_CHECK(hipMemcpyPeerAsync(dst_buf, dst, src_buf, src, bytes, 0));
    HIP_CHECK(hipDeviceSynchronize());
    hipEvent_t t0, t1;
    HIP_CHECK(hipEventCreate(&t0));
    HIP_CHECK(hipEventCreate(&t1));
    HIP_CHECK(hipEventRecord(t0));
    for (int i = 0; i < iters; ++i)
        HIP_CHECK(hipMemcpyPeerAsync(dst_buf, dst, src_buf, src, bytes, 0));
    HIP_CHECK(hipEventRecord(t1));
    HIP_CHECK(hipEventSynchronize(t1));
    float ms = 0;
    HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
    out["gbps"] = ((double)bytes * iters) / (ms * 1e6);

    HIP_CHECK(hipEventDestroy(t0));
    HIP_CHECK(hipEventDestroy(t1));
    HIP_CHECK(hipFree(dst_buf));
    HIP_CHECK(hipSetDevice(src));
    HIP_CHECK(hipFree(src_buf));
    return out;
}

} // namespace

// PCI bus id per HIP ordinal ("domain:bus:device.function", lowercase) —
// lets the plugin map its kfd-derived device ids onto ordinals exactly
// instead of assuming enumeration order (ROCR_VISIBLE_DEVICES can
// reorder it).
std::vector<std::string> pci_bus_ids() {
    int n = 0;
    HIP_CHECK(hipGetDeviceCount(&n));
    std::vector<std::string> out;
    out.reserve(n);
    for (int i = 0; i < n; ++i) {
        char buf[32] = {0};
        if (hipDeviceGetPCIBusId(buf, sizeof(buf), i) == hipSuccess)
            out.emplace_back(buf);
        else
            out.emplace_back("");
    }
    return out;
}

PYBIND11_MODULE(_healthprobe, m) {
    m.doc() = "gfx950 deep GPU health probe (MFMA/LDS/HBM)";
    m.def("run_probe", &run_probe, py::arg("device") = 0,
          py::arg("hbm_bytes") = (size_t)1 << 30);
    m.def("device_count", &device_count);
    m.def("pci_bus_ids", &pci_bus_ids);
    m.def("p2p_bandwidth", &p2p_bandwidth, py::arg("src") = 0,
          py::arg("dst") = 1, py::arg("bytes") = (size_t)1 << 30);
    m.def("mfma_probe_raw", &mfma_probe_raw, py::arg("device") = 0);
}
