// MI355X node agent — on-node GPU health validation for provisioned nodes.
//
// The MI355X-native analogue of what the reference left to AKS's NVIDIA
// node-image health machinery: before a freshly provisioned 8×MI355X node is
// trusted (and as the node.health controller's on-node evidence), this agent
// validates the GPU stack end to end: device discovery (gfx950 arch, 288 GB
// HBM3E), an HBM bandwidth self-test (float4 streaming copy; healthy nodes
// reach ~6 TB/s of the 8 TB/s peak), a VALU FMA correctness check, an MFMA
// matrix-pipe check (v_mfma_f32_16x16x4_f32 — the CDNA4 matrix core the ML
// workloads will live on), and the xGMI peer-to-peer link matrix.
//
// Built for gfx950 only (hipcc --offload-arch=gfx950); exposed as a C ABI
// consumed by gpu_provisioner_amd/nodeagent.py (ctypes) both as a CLI and in
// the k8s DaemonSet health probe.

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <cstring>

#define NA_OK 0
#define NA_ERR_HIP -1
#define NA_ERR_VERIFY -2

#define HIP_CHECK(expr)                                                                  \
    do {                                                                                 \
        hipError_t _e = (expr);                                                          \
        if (_e != hipSuccess) {                                                          \
            std::snprintf(na_last_error_buf, sizeof(na_last_error_buf), "%s at %s:%d",   \
                          hipGetErrorString(_e), __FILE__, __LINE__);                    \
            return NA_ERR_HIP;                                                           \
        }                                                                                \
    } while (0)

static char na_last_error_buf[512];

extern "C" const char* na_last_error() { return na_last_error_buf; }

// ---------------------------------------------------------------------------
// kernels
// ---------------------------------------------------------------------------

// Streaming float4 copy: the canonical HBM bandwidth probe (reaches ~79% of
// the 8 TB/s peak on healthy silicon). Grid-stride so any grid ≫256 WGs fills
// all 8 XCDs.
__global__ void copy_f4_kernel(const float4* __restrict__ src, float4* __restrict__ dst,
                               size_t n) {
    size_t i = blockIdx.x * (size_t)blockDim.x + threadIdx.x;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) dst[i] = src[i];
}

// VALU self-test: dependent FMA chain with a closed-form result.
__global__ void fma_selftest_kernel(float* __restrict__ out, int iters) {
    int lane = threadIdx.x;
    float x = 1.0f;
    float a = 0.5f, b = 0.25f;
    for (int i = 0; i < iters; ++i) x = fmaf(x, a, b);  // x -> x/2 + 1/4, fixpoint 0.5
    out[blockIdx.x * blockDim.x + lane] = x;
}

// MFMA matrix-pipe self-test: one wave issues v_mfma_f32_16x16x4_f32 with
// uniform operands. With A=alpha and B=beta everywhere and C=0, every output
// element is K*alpha*beta (K=4) regardless of fragment layout, so the check
// is layout-independent and still exercises the matrix pipe + AGPR file.
typedef float f32x4 __attribute__((ext_vector_type(4)));

__global__ void mfma_selftest_kernel(float* __restrict__ out, float alpha, float beta) {
#if defined(__gfx950__)
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(alpha, beta, acc, 0, 0, 0);
    // second accumulation: D = A*B + C must chain through the accumulator
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(alpha, beta, acc, 0, 0, 0);
    int lane = threadIdx.x;
    for (int i = 0; i < 4; ++i) out[lane * 4 + i] = acc[i];
#else
    out[threadIdx.x] = -1.0f;  // wrong arch: fail verification
#endif
}

// ---------------------------------------------------------------------------
// C ABI
// ---------------------------------------------------------------------------

extern "C" int na_device_count(int* count) {
    HIP_CHECK(hipGetDeviceCount(count));
    return NA_OK;
}

extern "C" int na_device_info(int dev, char* name, int name_len, char* arch, int arch_len,
                              long long* hbm_bytes, int* cus, int* max_clock_khz) {
    hipDeviceProp_t prop;
    HIP_CHECK(hipGetDeviceProperties(&prop, dev));
    std::snprintf(name, name_len, "%s", prop.name);
    std::snprintf(arch, arch_len, "%s", prop.gcnArchName);
    *hbm_bytes = (long long)prop.totalGlobalMem;
    *cus = prop.multiProcessorCount;
    *max_clock_khz = prop.clockRate;
    return NA_OK;
}

extern "C" int na_hbm_bandwidth(int dev, long long bytes, int iters, double* gbs) {
    HIP_CHECK(hipSetDevice(dev));
    size_t n = (size_t)bytes / sizeof(float4);
    float4 *src = nullptr, *dst = nullptr;
    HIP_CHECK(hipMalloc(&src, n * sizeof(float4)));
    hipError_t e2 = hipMalloc(&dst, n * sizeof(float4));
    if (e2 != hipSuccess) {
        (void)hipFree(src);
        std::snprintf(na_last_error_buf, sizeof(na_last_error_buf), "%s", hipGetErrorString(e2));
        return NA_ERR_HIP;
    }
    HIP_CHECK(hipMemset(src, 1, n * sizeof(float4)));
    // 2048 workgroups × 256 threads: ≫256 WGs so all 8 XCDs are saturated
    dim3 grid(2048), block(256);
    hipEvent_t t0, t1;
    HIP_CHECK(hipEventCreate(&t0));
    HIP_CHECK(hipEventCreate(&t1));
    copy_f4_kernel<<<grid, block>>>(src, dst, n);  // warmup
    HIP_CHECK(hipDeviceSynchronize());
    HIP_CHECK(hipEventRecord(t0));
    for (int i = 0; i < iters; ++i) copy_f4_kernel<<<grid, block>>>(src, dst, n);
    HIP_CHECK(hipEventRecord(t1));
    HIP_CHECK(hipEventSynchronize(t1));
    float ms = 0.f;
    HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
    // read + write
    *gbs = (2.0 * (double)bytes * iters) / (ms * 1e6);
    (void)hipEventDestroy(t0);
    (void)hipEventDestroy(t1);
    (void)hipFree(src);
    (void)hipFree(dst);
    return NA_OK;
}

extern "C" int na_fma_selftest(int dev) {
    HIP_CHECK(hipSetDevice(dev));
    const int blocks = 1024, threads = 256, iters = 256;
    float* out = nullptr;
    HIP_CHECK(hipMalloc(&out, (size_t)blocks * threads * sizeof(float)));
    fma_selftest_kernel<<<dim3(blocks), dim3(threads)>>>(out, iters);
    HIP_CHECK(hipDeviceSynchronize());
    float* host = new float[(size_t)blocks * threads];
    HIP_CHECK(hipMemcpy(host, out, (size_t)blocks * threads * sizeof(float), hipMemcpyDeviceToHost));
    int rc = NA_OK;
    for (int i = 0; i < blocks * threads; ++i) {
        // fixpoint of x -> x/2 + 1/4 is 0.5; 256 iterations converge exactly
        if (host[i] != 0.5f) {
            std::snprintf(na_last_error_buf, sizeof(na_last_error_buf),
                          "fma selftest: lane %d got %g want 0.5", i, host[i]);
            rc = NA_ERR_VERIFY;
            break;
        }
    }
    delete[] host;
    (void)hipFree(out);
    return rc;
}

extern "C" int na_mfma_selftest(int dev) {
    HIP_CHECK(hipSetDevice(dev));
    const float alpha = 1.5f, beta = 2.0f;
    const int n = 64 * 4;
    float* out = nullptr;
    HIP_CHECK(hipMalloc(&out, n * sizeof(float)));
    // one wave (64 threads): MFMA is a per-wave instruction
    mfma_selftest_kernel<<<dim3(1), dim3(64)>>>(out, alpha, beta);
    HIP_CHECK(hipDeviceSynchronize());
    float host[n];
    HIP_CHECK(hipMemcpy(host, out, n * sizeof(float), hipMemcpyDeviceToHost));
    (void)hipFree(out);
    // two chained MFMAs, K=4, uniform operands: every element = 2*K*alpha*beta
    const float want = 2.0f * 4.0f * alpha * beta;
    for (int i = 0; i < n; ++i) {
        if (host[i] != want) {
            std::snprintf(na_last_error_buf, sizeof(na_last_error_buf),
                          "mfma selftest: elem %d got %g want %g", i, host[i], want);
            return NA_ERR_VERIFY;
        }
    }
    return NA_OK;
}

extern "C" int na_p2p_matrix(int n, int* out /* n*n */) {
    for (int i = 0; i < n; ++i) {
        for (int j = 0; j < n; ++j) {
            if (i == j) {
                out[i * n + j] = 1;
                continue;
            }
            int can = 0;
            HIP_CHECK(hipDeviceCanAccessPeer(&can, i, j));
            out[i * n + j] = can;
        }
    }
    return NA_OK;
}

extern "C" int na_p2p_bandwidth(int src, int dst, long long bytes, int iters, double* gbs) {
    HIP_CHECK(hipSetDevice(src));
    void* sbuf = nullptr;
    HIP_CHECK(hipMalloc(&sbuf, (size_t)bytes));
    HIP_CHECK(hipSetDevice(dst));
    void* dbuf = nullptr;
    hipError_t e = hipMalloc(&dbuf, (size_t)bytes);
    if (e != hipSuccess) {
        (void)hipSetDevice(src);
        (void)hipFree(sbuf);
        std::snprintf(na_last_error_buf, sizeof(na_last_error_buf), "%s", hipGetErrorString(e));
        return NA_ERR_HIP;
    }
    HIP_CHECK(hipSetDevice(src));
    hipEvent_t t0, t1;
    HIP_CHECK(hipEventCreate(&t0));
    HIP_CHECK(hipEventCreate(&t1));
    HIP_CHECK(hipMemcpyPeer(dbuf, dst, sbuf, src, (size_t)bytes));  // warmup
    HIP_CHECK(hipDeviceSynchronize());
    HIP_CHECK(hipEventRecord(t0));
    for (int i = 0; i < iters; ++i)
        HIP_CHECK(hipMemcpyPeer(dbuf, dst, sbuf, src, (size_t)bytes));
    HIP_CHECK(hipEventRecord(t1));
    HIP_CHECK(hipEventSynchronize(t1));
    float ms = 0.f;
    HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
    *gbs = ((double)bytes * iters) / (ms * 1e6);
    (void)hipEventDestroy(t0);
    (void)hipEventDestroy(t1);
    (void)hipFree(sbuf);
    (void)hipSetDevice(dst);
    (void)hipFree(dbuf);
    (void)hipSetDevice(src);
    return NA_OK;
}
