// kata_xpu_device_plugin_amd._gpuprobe — MI355X (gfx950) health/burn-in probes.
//
// A device plugin hands whole GPUs to Kata VMs; before a GPU is advertised
// (or after a RAS event) the operator wants evidence the silicon is sane.
// The reference has no such capability (NVML-less); these
// CDNA4-native probes are the MI355X answer:
//
//   * hbm_bandwidth_probe : float4-coalesced streaming copy (16 B/lane per
//     instruction — cdna_hip_programming.md §2 coalescing rule), grid ≫256
//     workgroups to fill all 8 XCDs; a healthy MI355X sustains ≈6.3 TB/s
//     (MI355X_MICROARCH.md: 8 TB/s peak, 79% achievable).
//   * mfma_probe_f32  : v_mfma_f32_16x16x4_f32 tile with the documented
//     lane mapping (A[l&15][l>>4], B[l>>4][l&15]; C/D col=l&15,
//     row=(l>>4)*4+reg — cdna_hip_programming.md §3), verified elementwise
//     against a host fp32 reference. Exact-f32 MFMA ⇒ bitwise match.
//   * mfma_probe_bf16 : v_mfma_f32_32x32x16_bf16 throughput burn
//     (the CDNA4 matrix-core rate, ~2.5 PF dense chip-wide) with a
//     host-checked numeric result on one tile.
//   * memtest         : address-pattern write/readback over a buffer,
//     returns mismatch count (catches dead HBM channels).
//
// Wavefront size is 64 everywhere (gfx950; cdna_hip_programming.md §1).
// Build: hipcc --offload-arch=gfx950 (driven by setup.py / __graft_entry__).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <algorithm>
#include <cmath>
#include <cstdint>
#include <cstring>
#include <string>
#include <vector>

namespace py = pybind11;

#define HIP_CHECK(expr)                                                       \
    do {                                                                      \
        hipError_t _e = (expr);                                               \
        if (_e != hipSuccess)                                                 \
            throw std::runtime_error(std::string(#expr) + " failed: " +       \
                                     hipGetErrorString(_e));                  \
    } while (0)

namespace {

constexpr int WAVE = 64;

// ---------------------------------------------------------------------------
// HBM bandwidth: grid-stride float4 copy. 256 threads/WG, ≥2048 WGs so all
// 8 XCDs × 32 CUs are saturated (a launch needs ≫256 workgroups to fill
// the chip — MI355X_MICROARCH.md §Workgroup dispatch).
// ---------------------------------------------------------------------------
__global__ void copy_f4(const float4* __restrict__ src,
                        float4* __restrict__ dst, size_t n4) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (; i < n4; i += stride) dst[i] = src[i];
}

// ---------------------------------------------------------------------------
// Address-pattern memtest: write f(i), read back, count mismatches on GPU.
// ---------------------------------------------------------------------------
__global__ void pattern_write(uint64_t* __restrict__ buf, size_t n, uint64_t salt) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) buf[i] = (uint64_t)i * 0x9E3779B97F4A7C15ull ^ salt;
}

__global__ void pattern_check(const uint64_t* __restrict__ buf, size_t n,
                              uint64_t salt, unsigned long long* mismatches) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    unsigned long long local = 0;
    for (; i < n; i += stride)
        if (buf[i] != ((uint64_t)i * 0x9E3779B97F4A7C15ull ^ salt)) local++;
    if (local) atomicAdd(mismatches, local);
}

// ---------------------------------------------------------------------------
// LDS bandwidth burn: ds_read_b128 streaming from a 32 KiB per-WG tile.
// Per the §LDS table (MI355X_MICROARCH.md) b128 reads move 256 B/clk/CU
// (≈150 TB/s chip-wide from ≥4 waves/CU); one scalar accumulate per read
// keeps the loop LDS-bound (4 LDS cycles vs 2 VALU cycles per b128).
// A sick CU/LDS array shows up as a chip-wide rate far below spec.
// ---------------------------------------------------------------------------
typedef float lds_f32x4 __attribute__((ext_vector_type(4)));

__global__ void __launch_bounds__(256, 4)
lds_read_burn(float* __restrict__ out, int iters) {
    __shared__ lds_f32x4 buf[2048];  // 32 KiB → 4 WGs/CU fit in 160 KiB
    for (int i = threadIdx.x; i < 2048; i += 256) {
        lds_f32x4 v = {(float)i, 1.f, 2.f, 3.f};
        buf[i] = v;
    }
    __syncthreads();
    // Inline-asm ds_read_b128 (volatile): plain C++ float4 reads get their
    // unused components dead-code-eliminated into ds_read_b32, which
    // measures the 128 B/clk narrow-read rate instead of the 256 B/clk
    // b128 rate (§LDS table). One scalar accumulate per read keeps the
    // loop LDS-bound (8×4 LDS cycles vs 8×2 VALU per batch); the explicit
    // s_waitcnt is mandatory — hipcc pads nothing inside asm (§5.7).
    unsigned base = (unsigned)(threadIdx.x & 255) * 16u;  // lane-contiguous 16-B slots
    float acc = 0.f;
    for (int it = 0; it < iters; it++) {
        lds_f32x4 v0, v1, v2, v3, v4, v5, v6, v7;
        asm volatile(
            "ds_read_b128 %0, %8 offset:0\n\t"
            "ds_read_b128 %1, %8 offset:4096\n\t"
            "ds_read_b128 %2, %8 offset:8192\n\t"
            "ds_read_b128 %3, %8 offset:12288\n\t"
            "ds_read_b128 %4, %8 offset:16384\n\t"
            "ds_read_b128 %5, %8 offset:20480\n\t"
            "ds_read_b128 %6, %8 offset:24576\n\t"
            "ds_read_b128 %7, %8 offset:28672\n\t"
            "s_waitcnt lgkmcnt(0)"
            : "=v"(v0), "=v"(v1), "=v"(v2), "=v"(v3),
              "=v"(v4), "=v"(v5), "=v"(v6), "=v"(v7)
            : "v"(base));
        acc += v0.x + v1.x + v2.x + v3.x + v4.x + v5.x + v6.x + v7.x;
    }
    out[(size_t)blockIdx.x * blockDim.x + threadIdx.x] = acc;
}

// ---------------------------------------------------------------------------
// HBM load-to-use latency: single-wave pointer chase over a permutation
// too large for L2/LLC (guide: ~900 cycles per HBM-miss dependent load).
// ---------------------------------------------------------------------------
__global__ void pointer_chase(const uint32_t* __restrict__ next,
                              uint32_t* __restrict__ out, int steps) {
    if (blockIdx.x != 0 || threadIdx.x != 0) return;
    uint32_t idx = 0;
    for (int i = 0; i < steps; i++) idx = next[idx];
    *out = idx;  // keep the chain live
}

// ---------------------------------------------------------------------------
// f32-input MFMA correctness tile: D = A·B for one 16×16×4 step per wave.
// Lane mapping from cdna_hip_programming.md §3 (documented, exact f32):
//   a = A[l&15][l>>4]   (K=4: k = l>>4)
//   b = B[l>>4][l&15]
//   D reg j ↔ D[(l>>4)*4 + j][l&15]
// ---------------------------------------------------------------------------
typedef float f32x4 __attribute__((ext_vector_type(4)));

__global__ void mfma_f32_tile(const float* __restrict__ A,  // 16×4 row-major
                              const float* __restrict__ B,  // 4×16 row-major
                              float* __restrict__ D) {      // 16×16 row-major
    int l = threadIdx.x & (WAVE - 1);
    int row = l & 15, k = l >> 4;
    float a = A[row * 4 + k];
    float b = B[k * 16 + (l & 15)];
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
    if (blockIdx.x == 0 && threadIdx.x < WAVE) {
        for (int j = 0; j < 4; j++) D[((l >> 4) * 4 + j) * 16 + (l & 15)] = acc[j];
    }
}

// ---------------------------------------------------------------------------
// bf16 MFMA: one verified 32×32×16 tile + a throughput burn loop.
// v_mfma_f32_32x32x16_bf16: each lane holds 8 bf16 of A and B (4 VGPRs),
// 16 f32 accumulators. C/D mapping (cdna_hip_programming.md §3):
//   col = l&31, row = (reg&3) + 8*(reg>>2) + 4*(l>>5)
// A/B input mapping (K=16, 2 lane-halves of 32): lane l holds
//   A[l&31][(l>>5)*8 + j]  and  B[(l>>5)*8 + j][l&31],  j = 0..7.
// The numeric check against a host reference catches any mapping error.
// ---------------------------------------------------------------------------
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

__global__ void mfma_bf16_tile(const __bf16* __restrict__ A,  // 32×16 row-major
                               const __bf16* __restrict__ B,  // 16×32 row-major
                               float* __restrict__ D) {       // 32×32 row-major
    int l = threadIdx.x & (WAVE - 1);
    int half = l >> 5;          // which 8-wide K slice
    int lane32 = l & 31;
    bf16x8 a, b;
    for (int j = 0; j < 8; j++) {
        a[j] = A[lane32 * 16 + half * 8 + j];
        b[j] = B[(half * 8 + j) * 32 + lane32];
    }
    f32x16 acc = {};
    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
    if (blockIdx.x == 0 && threadIdx.x < WAVE) {
        for (int reg = 0; reg < 16; reg++) {
            int row = (reg & 3) + 8 * (reg >> 2) + 4 * half;
            D[row * 32 + lane32] = acc[reg];
        }
    }
}

// Throughput burn: every wave hammers independent accumulators with
// back-to-back 32×32×16 bf16 MFMAs (32 cyc/SIMD issue; 4 accumulators give
// plenty of independence). Result is summed into out so the work is live.
__global__ void __launch_bounds__(256, 2)
mfma_bf16_burn(float* __restrict__ out, int iters) {
    int l = threadIdx.x & (WAVE - 1);
    bf16x8 a, b;
    for (int j = 0; j < 8; j++) {
        a[j] = (__bf16)(0.5f + 0.001f * (float)((l + j) & 7));
        b[j] = (__bf16)(0.25f + 0.002f * (float)((l ^ j) & 7));
    }
    f32x16 acc0 = {}, acc1 = {}, acc2 = {}, acc3 = {};
    for (int it = 0; it < iters; it++) {
        acc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc0, 0, 0, 0);
        acc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc1, 0, 0, 0);
        acc2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc2, 0, 0, 0);
        acc3 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc3, 0, 0, 0);
    }
    float s = 0;
    for (int r = 0; r < 16; r++) s += acc0[r] + acc1[r] + acc2[r] + acc3[r];
    // one store per lane; never zero, so the compiler cannot DCE the loop
    out[(size_t)blockIdx.x * blockDim.x + threadIdx.x] = s;
}

// ---------------------------------------------------------------------------
// Host-side wrappers
// ---------------------------------------------------------------------------

int device_count() {
    int n = 0;
    hipError_t e = hipGetDeviceCount(&n);
    if (e != hipSuccess) return 0;
    return n;
}

py::dict device_info(int dev) {
    hipDeviceProp_t prop;
    HIP_CHECK(hipGetDeviceProperties(&prop, dev));
    size_t free_b = 0, total_b = 0;
    HIP_CHECK(hipSetDevice(dev));
    HIP_CHECK(hipMemGetInfo(&free_b, &total_b));
    py::dict d;
    d["name"] = std::string(prop.name);
    d["gcn_arch"] = std::string(prop.gcnArchName);
    d["compute_units"] = prop.multiProcessorCount;
    d["total_mem_bytes"] = (uint64_t)total_b;
    d["free_mem_bytes"] = (uint64_t)free_b;
    d["pci_bus_id"] = prop.pciBusID;
    d["pci_domain_id"] = prop.pciDomainID;
    d["pci_device_id"] = prop.pciDeviceID;
    d["warp_size"] = prop.warpSize;
    return d;
}

py::dict hbm_bandwidth_probe(int dev, size_t bytes, int iters) {
    HIP_CHECK(hipSetDevice(dev));
    size_t n4 = bytes / sizeof(float4);
    float4 *src = nullptr, *dst = nullptr;
    HIP_CHECK(hipMalloc(&src, n4 * sizeof(float4)));
    HIP_CHECK(hipMalloc(&dst, n4 * sizeof(float4)));
    HIP_CHECK(hipMemset(src, 0x3c, n4 * sizeof(float4)));
    hipDeviceProp_t prop;
    HIP_CHECK(hipGetDeviceProperties(&prop, dev));
    int blocks = prop.multiProcessorCount * 8;  // ≫256 WGs, fills all XCDs
    hipEvent_t t0, t1;
    HIP_CHECK(hipEventCreate(&t0));
    HIP_CHECK(hipEventCreate(&t1));
    // warmup
    hipLaunchKernelGGL(copy_f4, dim3(blocks), dim3(256), 0, 0, src, dst, n4);
    HIP_CHECK(hipDeviceSynchronize());
    float best_ms = 1e30f;
    for (int i = 0; i < iters; i++) {
        HIP_CHECK(hipEventRecord(t0));
        hipLaunchKernelGGL(copy_f4, dim3(blocks), dim3(256), 0, 0, src, dst, n4);
        HIP_CHECK(hipEventRecord(t1));
        HIP_CHECK(hipEventSynchronize(t1));
        float ms = 0;
        HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
        if (ms < best_ms) best_ms = ms;
    }
    HIP_CHECK(hipEventDestroy(t0));
    HIP_CHECK(hipEventDestroy(t1));
    HIP_CHECK(hipFree(src));
    HIP_CHECK(hipFree(dst));
    double gbps = 2.0 * (double)(n4 * sizeof(float4)) / (best_ms * 1e6);
    py::dict d;
    d["gbps"] = gbps;
    d["best_ms"] = best_ms;
    d["bytes_moved"] = (uint64_t)(2 * n4 * sizeof(float4));
    return d;
}

py::dict lds_bandwidth_probe(int dev, int iters) {
    HIP_CHECK(hipSetDevice(dev));
    hipDeviceProp_t prop;
    HIP_CHECK(hipGetDeviceProperties(&prop, dev));
    int blocks = prop.multiProcessorCount * 4;   // 4 × 32 KiB per CU
    float* out = nullptr;
    HIP_CHECK(hipMalloc(&out, (size_t)blocks * 256 * 4));
    hipLaunchKernelGGL(lds_read_burn, dim3(blocks), dim3(256), 0, 0, out, 256);
    HIP_CHECK(hipDeviceSynchronize());
    hipEvent_t t0, t1;
    HIP_CHECK(hipEventCreate(&t0));
    HIP_CHECK(hipEventCreate(&t1));
    HIP_CHECK(hipEventRecord(t0));
    hipLaunchKernelGGL(lds_read_burn, dim3(blocks), dim3(256), 0, 0, out, iters);
    HIP_CHECK(hipEventRecord(t1));
    HIP_CHECK(hipEventSynchronize(t1));
    float ms = 0;
    HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
    HIP_CHECK(hipEventDestroy(t0));
    HIP_CHECK(hipEventDestroy(t1));
    HIP_CHECK(hipFree(out));
    // bytes: blocks × 256 lanes × 8 b128-reads × 16 B per iteration
    double bytes = (double)blocks * 256 * 8 * 16 * (double)iters;
    py::dict d;
    d["tbps"] = bytes / (ms * 1e9);
    d["burn_ms"] = ms;
    return d;
}

py::dict hbm_latency_probe(int dev, size_t bytes, int steps) {
    HIP_CHECK(hipSetDevice(dev));
    size_t n = bytes / sizeof(uint32_t);
    // Sattolo shuffle → one full cycle; stride-randomized to defeat
    // prefetch and the 256 MiB LLC (use ≥1 GiB).
    std::vector<uint32_t> perm(n);
    for (size_t i = 0; i < n; i++) perm[i] = (uint32_t)i;
    uint64_t rng = 0x9E3779B97F4A7C15ull;
    for (size_t i = n - 1; i > 0; i--) {
        rng = rng * 6364136223846793005ull + 1442695040888963407ull;
        size_t j = (size_t)(rng % i);
        std::swap(perm[i], perm[j]);
    }
    std::vector<uint32_t> next(n);
    for (size_t i = 0; i < n; i++) next[perm[i]] = perm[(i + 1) % n];
    uint32_t *d_next = nullptr, *d_out = nullptr;
    HIP_CHECK(hipMalloc(&d_next, n * 4));
    HIP_CHECK(hipMalloc(&d_out, 4));
    HIP_CHECK(hipMemcpy(d_next, next.data(), n * 4, hipMemcpyHostToDevice));
    hipLaunchKernelGGL(pointer_chase, dim3(1), dim3(64), 0, 0, d_next, d_out, 1000);
    HIP_CHECK(hipDeviceSynchronize());
    hipEvent_t t0, t1;
    HIP_CHECK(hipEventCreate(&t0));
    HIP_CHECK(hipEventCreate(&t1));
    HIP_CHECK(hipEventRecord(t0));
    hipLaunchKernelGGL(pointer_chase, dim3(1), dim3(64), 0, 0, d_next, d_out, steps);
    HIP_CHECK(hipEventRecord(t1));
    HIP_CHECK(hipEventSynchronize(t1));
    float ms = 0;
    HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
    HIP_CHECK(hipEventDestroy(t0));
    HIP_CHECK(hipEventDestroy(t1));
    HIP_CHECK(hipFree(d_next));
    HIP_CHECK(hipFree(d_out));
    py::dict d;
    d["latency_ns"] = ms * 1e6 / steps;
    d["steps"] = steps;
    return d;
}

py::dict pcie_bandwidth_probe(int dev, size_t bytes, int iters) {
    // Host-link health: pinned-memory H2D/D2H streaming. MI355X host link
    // is PCIe Gen5 x16 (63 GB/s spec); a degraded link (wrong slot width,
    // retraining) shows up far below that.
    HIP_CHECK(hipSetDevice(dev));
    void* host = nullptr;
    void* d = nullptr;
    HIP_CHECK(hipHostMalloc(&host, bytes));
    HIP_CHECK(hipMalloc(&d, bytes));
    std::memset(host, 0x5a, bytes);
    hipEvent_t t0, t1;
    HIP_CHECK(hipEventCreate(&t0));
    HIP_CHECK(hipEventCreate(&t1));
    auto run = [&](bool h2d) {
        // warmup
        HIP_CHECK(hipMemcpyAsync(h2d ? d : host, h2d ? host : d, bytes,
                                 h2d ? hipMemcpyHostToDevice : hipMemcpyDeviceToHost, 0));
        HIP_CHECK(hipDeviceSynchronize());
        float best = 1e30f;
        for (int i = 0; i < iters; i++) {
            HIP_CHECK(hipEventRecord(t0));
            HIP_CHECK(hipMemcpyAsync(h2d ? d : host, h2d ? host : d, bytes,
                                     h2d ? hipMemcpyHostToDevice : hipMemcpyDeviceToHost, 0));
            HIP_CHECK(hipEventRecord(t1));
            HIP_CHECK(hipEventSynchronize(t1));
            float ms = 0;
            HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
            if (ms < best) best = ms;
        }
        return (double)bytes / (best * 1e6);
    };
    double h2d = run(true);
    double d2h = run(false);
    HIP_CHECK(hipEventDestroy(t0));
    HIP_CHECK(hipEventDestroy(t1));
    HIP_CHECK(hipFree(d));
    HIP_CHECK(hipHostFree(host));
    py::dict r;
    r["h2d_gbps"] = h2d;
    r["d2h_gbps"] = d2h;
    r["bytes"] = (uint64_t)bytes;
    return r;
}

py::dict memtest(int dev, size_t bytes) {
    HIP_CHECK(hipSetDevice(dev));
    size_t n = bytes / sizeof(uint64_t);
    uint64_t* buf = nullptr;
    unsigned long long* mm = nullptr;
    HIP_CHECK(hipMalloc(&buf, n * sizeof(uint64_t)));
    HIP_CHECK(hipMalloc(&mm, sizeof(unsigned long long)));
    HIP_CHECK(hipMemset(mm, 0, sizeof(unsigned long long)));
    uint64_t salt = 0xA5A5A5A55A5A5A5Aull;
    hipDeviceProp_t prop;
    HIP_CHECK(hipGetDeviceProperties(&prop, dev));
    int blocks = prop.multiProcessorCount * 8;
    hipLaunchKernelGGL(pattern_write, dim3(blocks), dim3(256), 0, 0, buf, n, salt);
    hipLaunchKernelGGL(pattern_check, dim3(blocks), dim3(256), 0, 0, buf, n, salt, mm);
    HIP_CHECK(hipDeviceSynchronize());
    unsigned long long mismatches = 0;
    HIP_CHECK(hipMemcpy(&mismatches, mm, sizeof(mismatches), hipMemcpyDeviceToHost));
    HIP_CHECK(hipFree(buf));
    HIP_CHECK(hipFree(mm));
    py::dict d;
    d["bytes"] = (uint64_t)(n * sizeof(uint64_t));
    d["mismatches"] = (uint64_t)mismatches;
    return d;
}

py::dict mfma_probe_f32(int dev) {
    HIP_CHECK(hipSetDevice(dev));
    std::vector<float> hA(16 * 4), hB(4 * 16), hD(16 * 16), ref(16 * 16, 0.f);
    for (int i = 0; i < 16; i++)
        for (int k = 0; k < 4; k++) hA[i * 4 + k] = 0.25f * i - 0.5f * k + 0.125f;
    for (int k = 0; k < 4; k++)
        for (int j = 0; j < 16; j++) hB[k * 16 + j] = 0.0625f * j + 0.75f * k - 1.f;
    // asymmetric B so a row/col swap cannot pass (guide §3 note)
    for (int i = 0; i < 16; i++)
        for (int j = 0; j < 16; j++)
            for (int k = 0; k < 4; k++)
                ref[i * 16 + j] = fmaf(hA[i * 4 + k], hB[k * 16 + j], ref[i * 16 + j]);
    float *dA, *dB, *dD;
    HIP_CHECK(hipMalloc(&dA, hA.size() * 4));
    HIP_CHECK(hipMalloc(&dB, hB.size() * 4));
    HIP_CHECK(hipMalloc(&dD, hD.size() * 4));
    HIP_CHECK(hipMemcpy(dA, hA.data(), hA.size() * 4, hipMemcpyHostToDevice));
    HIP_CHECK(hipMemcpy(dB, hB.data(), hB.size() * 4, hipMemcpyHostToDevice));
    hipLaunchKernelGGL(mfma_f32_tile, dim3(1), dim3(WAVE), 0, 0, dA, dB, dD);
    HIP_CHECK(hipDeviceSynchronize());
    HIP_CHECK(hipMemcpy(hD.data(), dD, hD.size() * 4, hipMemcpyDeviceToHost));
    HIP_CHECK(hipFree(dA)); HIP_CHECK(hipFree(dB)); HIP_CHECK(hipFree(dD));
    double max_abs_err = 0;
    for (int i = 0; i < 256; i++)
        max_abs_err = std::max(max_abs_err, (double)std::fabs(hD[i] - ref[i]));
    py::dict d;
    d["max_abs_err"] = max_abs_err;   // exact f32 MFMA ⇒ expect 0.0
    d["ok"] = max_abs_err == 0.0;
    return d;
}

static float bf16_round(float x) {
    // emulate bf16 storage rounding (round-to-nearest-even on the top 16 bits)
    uint32_t u;
    std::memcpy(&u, &x, 4);
    uint32_t lsb = (u >> 16) & 1u;
    u += 0x7fffu + lsb;
    u &= 0xffff0000u;
    float r;
    std::memcpy(&r, &u, 4);
    return r;
}

py::dict mfma_probe_bf16(int dev, int burn_iters) {
    HIP_CHECK(hipSetDevice(dev));
    // --- correctness tile ---
    std::vector<float> fA(32 * 16), fB(16 * 32);
    for (int i = 0; i < 32; i++)
        for (int k = 0; k < 16; k++) fA[i * 16 + k] = bf16_round(0.03125f * i - 0.0625f * k + 0.5f);
    for (int k = 0; k < 16; k++)
        for (int j = 0; j < 32; j++) fB[k * 32 + j] = bf16_round(0.015625f * j + 0.09375f * k - 1.f);
    std::vector<uint16_t> hA(32 * 16), hB(16 * 32);
    for (size_t i = 0; i < fA.size(); i++) {
        uint32_t u; std::memcpy(&u, &fA[i], 4); hA[i] = (uint16_t)(u >> 16);
    }
    for (size_t i = 0; i < fB.size(); i++) {
        uint32_t u; std::memcpy(&u, &fB[i], 4); hB[i] = (uint16_t)(u >> 16);
    }
    std::vector<float> ref(32 * 32, 0.f), hD(32 * 32);
    for (int i = 0; i < 32; i++)
        for (int j = 0; j < 32; j++)
            for (int k = 0; k < 16; k++)
                ref[i * 32 + j] += fA[i * 16 + k] * fB[k * 32 + j];
    uint16_t *dA, *dB; float* dD;
    HIP_CHECK(hipMalloc(&dA, hA.size() * 2));
    HIP_CHECK(hipMalloc(&dB, hB.size() * 2));
    HIP_CHECK(hipMalloc(&dD, hD.size() * 4));
    HIP_CHECK(hipMemcpy(dA, hA.data(), hA.size() * 2, hipMemcpyHostToDevice));
    HIP_CHECK(hipMemcpy(dB, hB.data(), hB.size() * 2, hipMemcpyHostToDevice));
    hipLaunchKernelGGL(mfma_bf16_tile, dim3(1), dim3(WAVE), 0, 0,
                       (const __bf16*)dA, (const __bf16*)dB, dD);
    HIP_CHECK(hipDeviceSynchronize());
    HIP_CHECK(hipMemcpy(hD.data(), dD, hD.size() * 4, hipMemcpyDeviceToHost));
    HIP_CHECK(hipFree(dA)); HIP_CHECK(hipFree(dB)); HIP_CHECK(hipFree(dD));
    double max_rel_err = 0;
    for (int i = 0; i < 1024; i++) {
        double denom = std::max(1.0, (double)std::fabs(ref[i]));
        max_rel_err = std::max(max_rel_err, (double)std::fabs(hD[i] - ref[i]) / denom);
    }

    // --- throughput burn ---
    hipDeviceProp_t prop;
    HIP_CHECK(hipGetDeviceProperties(&prop, dev));
    int blocks = prop.multiProcessorCount * 2;   // 2 × 256-thread WGs per CU
    float* out;
    HIP_CHECK(hipMalloc(&out, (size_t)blocks * 256 * 4));
    hipLaunchKernelGGL(mfma_bf16_burn, dim3(blocks), dim3(256), 0, 0, out, 64);
    HIP_CHECK(hipDeviceSynchronize());
    hipEvent_t t0, t1;
    HIP_CHECK(hipEventCreate(&t0));
    HIP_CHECK(hipEventCreate(&t1));
    HIP_CHECK(hipEventRecord(t0));
    hipLaunchKernelGGL(mfma_bf16_burn, dim3(blocks), dim3(256), 0, 0, out, burn_iters);
    HIP_CHECK(hipEventRecord(t1));
    HIP_CHECK(hipEventSynchronize(t1));
    float ms = 0;
    HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
    HIP_CHECK(hipEventDestroy(t0));
    HIP_CHECK(hipEventDestroy(t1));
    HIP_CHECK(hipFree(out));
    // FLOPs: blocks × 4 waves/WG × 4 acc × iters × 2·32·32·16
    double flops = (double)blocks * 4.0 * 4.0 * burn_iters * 2.0 * 32 * 32 * 16;
    py::dict d;
    d["max_rel_err"] = max_rel_err;
    d["ok"] = max_rel_err < 1e-6;   // exact: inputs are bf16-representable
    d["tflops"] = flops / (ms * 1e9);
    d["burn_ms"] = ms;
    return d;
}

}  // namespace

PYBIND11_MODULE(_gpuprobe, m) {
    m.doc() = "MI355X (gfx950) health/burn-in probes for kata-xpu-device-plugin-amd";
    m.def("device_count", &device_count);
    m.def("device_info", &device_info, py::arg("dev") = 0);
    m.def("hbm_bandwidth_probe", &hbm_bandwidth_probe, py::arg("dev") = 0,
          py::arg("bytes") = (size_t)1 << 31, py::arg("iters") = 5);
    m.def("memtest", &memtest, py::arg("dev") = 0, py::arg("bytes") = (size_t)1 << 31);
    m.def("pcie_bandwidth_probe", &pcie_bandwidth_probe, py::arg("dev") = 0,
          py::arg("bytes") = (size_t)256 << 20, py::arg("iters") = 5);
    m.def("lds_bandwidth_probe", &lds_bandwidth_probe, py::arg("dev") = 0,
          py::arg("iters") = 100000);
    m.def("hbm_latency_probe", &hbm_latency_probe, py::arg("dev") = 0,
          py::arg("bytes") = (size_t)1 << 30, py::arg("steps") = 2000000);
    m.def("mfma_probe_f32", &mfma_probe_f32, py::arg("dev") = 0);
    m.def("mfma_probe_bf16", &mfma_probe_bf16, py::arg("dev") = 0,
          py::arg("burn_iters") = 20000);
}
