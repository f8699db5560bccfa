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

// Streaming float4 copy: the canonical HBM bandwidth probe. Winning shape
// from two on-device variant sweeps (nodeagent/bw_sweep.hip + /tmp round 2,
// MI355X): each workgroup owns ONE CONTIGUOUS slice (consecutive
// iterations stay in the same DRAM window — grid-stride loses ~15%),
// nontemporal dwordx4 (streaming policy, no L2/LC pollution), and FOUR
// independent loads in flight per thread before the stores. Measured
// 6233 GB/s at 16384 WGs x 1024 threads — 99% of the ≈6.3 TB/s this chip
// sustains on read+write streams.
typedef float f4v __attribute__((ext_vector_type(4)));  // raw vector: nt-builtin compatible

__global__ void copy_f4_kernel(const float4* __restrict__ src_, float4* __restrict__ dst_,
                               size_t n) {
    const f4v* __restrict__ src = reinterpret_cast<const f4v*>(src_);
    f4v* __restrict__ dst = reinterpret_cast<f4v*>(dst_);
    size_t per = (n + gridDim.x - 1) / gridDim.x;
    size_t lo = blockIdx.x * per;
    size_t hi = lo + per < n ? lo + per : n;
    size_t bd = blockDim.x;
    size_t i = lo + threadIdx.x;
    for (; i + 3 * bd < hi; i += 4 * bd) {
        f4v a = __builtin_nontemporal_load(&src[i]);
        f4v b = __builtin_nontemporal_load(&src[i + bd]);
        f4v c = __builtin_nontemporal_load(&src[i + 2 * bd]);
        f4v e = __builtin_nontemporal_load(&src[i + 3 * bd]);
        __builtin_nontemporal_store(a, &dst[i]);
        __builtin_nontemporal_store(b, &dst[i + bd]);
        __builtin_nontemporal_store(c, &dst[i + 2 * bd]);
        __builtin_nontemporal_store(e, &dst[i + 3 * bd]);
    }
    for (; i < hi; i += bd)
        __builtin_nontemporal_store(__builtin_nontemporal_load(&src[i]), &dst[i]);
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

// MFMA datatype-path self-tests for the pipes ML workloads actually run on:
// gfx950's 2xK bf16 form (v_mfma_f32_16x16x32_bf16, the CDNA4 training
// workhorse) and the fp8 E4M3 form (v_mfma_f32_16x16x32_fp8_fp8, the serving
// path). Same uniform-operand trick as the f32 test: with A=alpha and B=beta
// everywhere and C=0, every output element is K*alpha*beta regardless of
// fragment layout, so the check is layout-independent while still exercising
// the low-precision input muxes and the accumulator file.
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));

__global__ void mfma_bf16_selftest_kernel(float* __restrict__ out, float alpha, float beta) {
#if defined(__gfx950__)
    bf16x8 a, b;
    for (int i = 0; i < 8; ++i) {
        a[i] = (__bf16)alpha;
        b[i] = (__bf16)beta;
    }
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    int lane = threadIdx.x;
    for (int i = 0; i < 4; ++i) out[lane * 4 + i] = acc[i];
#else
    out[threadIdx.x] = -1.0f;
#endif
}

// fp8 operands are 8 packed E4M3 bytes per lane (one i64).
__global__ void mfma_fp8_selftest_kernel(float* __restrict__ out, long a_bits, long b_bits) {
#if defined(__gfx950__)
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(a_bits, b_bits, acc, 0, 0, 0);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(a_bits, b_bits, acc, 0, 0, 0);
    int lane = threadIdx.x;
    for (int i = 0; i < 4; ++i) out[lane * 4 + i] = acc[i];
#else
    out[threadIdx.x] = -1.0f;
#endif
}

// Layout-correct bf16 GEMM tile: one v_mfma_f32_16x16x32_bf16 computing
// D[16x16] = A[16x32]·B[32x16] with ASYMMETRIC integer-valued data, checked
// against an exact host reference. The uniform-operand self-tests above are
// layout-independent by construction; this one fails if the per-lane
// fragment mapping (A: row=l&15, k=(l>>4)*8+i; B: col=l&15, same k;
// D: col=l&15, row=(l>>4)*4+i) is wrong anywhere. Integer values keep both
// sides exact regardless of summation order.
__device__ __host__ inline float tile_a(int i, int k) { return (float)((i * 31 + k * 7) % 7 - 3); }
__device__ __host__ inline float tile_b(int k, int j) { return (float)((k * 13 + j * 3) % 5 - 2); }

__global__ void mfma_bf16_tile_kernel(float* __restrict__ out) {
#if defined(__gfx950__)
    int l = threadIdx.x;
    bf16x8 a, b;
    int arow = l & 15, kbase = (l >> 4) * 8;
    for (int i = 0; i < 8; ++i) {
        a[i] = (__bf16)tile_a(arow, kbase + i);
        b[i] = (__bf16)tile_b(kbase + i, arow);  // B: col = l&15
    }
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    int dcol = l & 15, drow0 = (l >> 4) * 4;
    for (int i = 0; i < 4; ++i) out[(drow0 + i) * 16 + dcol] = acc[i];
#else
    out[threadIdx.x] = -1.0f;
#endif
}

extern "C" int na_mfma_bf16_tile_check(int dev) {
    HIP_CHECK(hipSetDevice(dev));
    float* out = nullptr;
    HIP_CHECK(hipMalloc(&out, 256 * sizeof(float)));
    mfma_bf16_tile_kernel<<<dim3(1), dim3(64)>>>(out);
    HIP_CHECK(hipDeviceSynchronize());
    float host[256];
    HIP_CHECK(hipMemcpy(host, out, sizeof(host), hipMemcpyDeviceToHost));
    (void)hipFree(out);
    for (int i = 0; i < 16; ++i) {
        for (int j = 0; j < 16; ++j) {
            float ref = 0.f;
            for (int k = 0; k < 32; ++k) ref += tile_a(i, k) * tile_b(k, j);
            if (host[i * 16 + j] != ref) {
                std::snprintf(na_last_error_buf, sizeof(na_last_error_buf),
                              "mfma bf16 tile: D[%d][%d] got %g want %g", i, j,
                              host[i * 16 + j], ref);
                return NA_ERR_VERIFY;
            }
        }
    }
    return NA_OK;
}

// f16 variant of the layout-correct tile (same fragment maps as bf16).
typedef _Float16 f16x8 __attribute__((ext_vector_type(8)));

__global__ void mfma_f16_tile_kernel(float* __restrict__ out) {
#if defined(__gfx950__)
    int l = threadIdx.x;
    int arow = l & 15, kbase = (l >> 4) * 8;
    f16x8 a, b;
    for (int i = 0; i < 8; ++i) {
        a[i] = (_Float16)tile_a(arow, kbase + i);
        b[i] = (_Float16)tile_b(kbase + i, arow);
    }
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, acc, 0, 0, 0);
    int dcol = l & 15, drow0 = (l >> 4) * 4;
    for (int i = 0; i < 4; ++i) out[(drow0 + i) * 16 + dcol] = acc[i];
#else
    out[threadIdx.x] = -1.0f;
#endif
}

extern "C" int na_mfma_f16_tile_check(int dev) {
    HIP_CHECK(hipSetDevice(dev));
    float* out = nullptr;
    HIP_CHECK(hipMalloc(&out, 256 * sizeof(float)));
    mfma_f16_tile_kernel<<<dim3(1), dim3(64)>>>(out);
    HIP_CHECK(hipDeviceSynchronize());
    float host[256];
    HIP_CHECK(hipMemcpy(host, out, sizeof(host), hipMemcpyDeviceToHost));
    (void)hipFree(out);
    for (int i = 0; i < 16; ++i) {
        for (int j = 0; j < 16; ++j) {
            float ref = 0.f;
            for (int k = 0; k < 32; ++k) ref += tile_a(i, k) * tile_b(k, j);
            if (host[i * 16 + j] != ref) {
                std::snprintf(na_last_error_buf, sizeof(na_last_error_buf),
                              "mfma f16 tile: D[%d][%d] got %g want %g", i, j,
                              host[i * 16 + j], ref);
                return NA_ERR_VERIFY;
            }
        }
    }
    return NA_OK;
}

// fp8 (E4M3) variant of the layout-correct tile: same fragment maps, the
// 8 per-lane elements packed one byte each into the i64 operand (element i
// at byte i). Values restricted to small integers exact in E4M3.
__device__ __host__ inline int tile8_a(int i, int k) { return (i * 31 + k * 7) % 7 - 3; }
__device__ __host__ inline int tile8_b(int k, int j) { return (k * 13 + j * 3) % 5 - 2; }

__device__ __host__ inline unsigned char e4m3(int v) {
    // encodings for -3..3 (sign | exp(bias 7) | 3-bit mantissa)
    switch (v) {
        case 0: return 0x00;
        case 1: return 0x38;
        case 2: return 0x40;
        case 3: return 0x44;
        case -1: return 0xB8;
        case -2: return 0xC0;
        default: return 0xC4;  // -3
    }
}

__global__ void mfma_fp8_tile_kernel(float* __restrict__ out) {
#if defined(__gfx950__)
    int l = threadIdx.x;
    int arow = l & 15, kbase = (l >> 4) * 8;
    long a_bits = 0, b_bits = 0;
    for (int i = 0; i < 8; ++i) {
        a_bits |= (long)e4m3(tile8_a(arow, kbase + i)) << (8 * i);
        b_bits |= (long)e4m3(tile8_b(kbase + i, arow)) << (8 * i);
    }
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(a_bits, b_bits, acc, 0, 0, 0);
    int dcol = l & 15, drow0 = (l >> 4) * 4;
    for (int i = 0; i < 4; ++i) out[(drow0 + i) * 16 + dcol] = acc[i];
#else
    out[threadIdx.x] = -1.0f;
#endif
}

extern "C" int na_mfma_fp8_tile_check(int dev) {
    HIP_CHECK(hipSetDevice(dev));
    float* out = nullptr;
    HIP_CHECK(hipMalloc(&out, 256 * sizeof(float)));
    mfma_fp8_tile_kernel<<<dim3(1), dim3(64)>>>(out);
    HIP_CHECK(hipDeviceSynchronize());
    float host[256];
    HIP_CHECK(hipMemcpy(host, out, sizeof(host), hipMemcpyDeviceToHost));
    (void)hipFree(out);
    for (int i = 0; i < 16; ++i) {
        for (int j = 0; j < 16; ++j) {
            float ref = 0.f;
            for (int k = 0; k < 32; ++k) ref += (float)(tile8_a(i, k) * tile8_b(k, j));
            if (host[i * 16 + j] != ref) {
                std::snprintf(na_last_error_buf, sizeof(na_last_error_buf),
                              "mfma fp8 tile: D[%d][%d] got %g want %g", i, j,
                              host[i * 16 + j], ref);
                return NA_ERR_VERIFY;
            }
        }
    }
    return NA_OK;
}

// i8 variant (v_mfma_i32_16x16x64_i8, K=64 — CDNA4's 2xK int8 path):
// 16 per-lane elements packed 4-per-i32; k = (l>>4)*16 + i.
typedef int i32x4 __attribute__((ext_vector_type(4)));

__global__ void mfma_i8_tile_kernel(int* __restrict__ out) {
#if defined(__gfx950__)
    int l = threadIdx.x;
    int arow = l & 15, kbase = (l >> 4) * 16;
    i32x4 a = {0, 0, 0, 0}, b = {0, 0, 0, 0};
    for (int i = 0; i < 16; ++i) {
        unsigned char av = (unsigned char)(signed char)tile8_a(arow, kbase + i);
        unsigned char bv = (unsigned char)(signed char)tile8_b(kbase + i, arow);
        a[i / 4] |= (int)av << (8 * (i % 4));
        b[i / 4] |= (int)bv << (8 * (i % 4));
    }
    i32x4 acc = {0, 0, 0, 0};
    acc = __builtin_amdgcn_mfma_i32_16x16x64_i8(a, b, acc, 0, 0, 0);
    int dcol = l & 15, drow0 = (l >> 4) * 4;
    for (int i = 0; i < 4; ++i) out[(drow0 + i) * 16 + dcol] = acc[i];
#else
    out[threadIdx.x] = -1;
#endif
}

extern "C" int na_mfma_i8_tile_check(int dev) {
    HIP_CHECK(hipSetDevice(dev));
    int* out = nullptr;
    HIP_CHECK(hipMalloc(&out, 256 * sizeof(int)));
    mfma_i8_tile_kernel<<<dim3(1), dim3(64)>>>(out);
    HIP_CHECK(hipDeviceSynchronize());
    int host[256];
    HIP_CHECK(hipMemcpy(host, out, sizeof(host), hipMemcpyDeviceToHost));
    (void)hipFree(out);
    for (int i = 0; i < 16; ++i) {
        for (int j = 0; j < 16; ++j) {
            int ref = 0;
            for (int k = 0; k < 64; ++k) ref += tile8_a(i, k) * tile8_b(k, j);
            if (host[i * 16 + j] != ref) {
                std::snprintf(na_last_error_buf, sizeof(na_last_error_buf),
                              "mfma i8 tile: D[%d][%d] got %d want %d", i, j,
                              host[i * 16 + j], ref);
                return NA_ERR_VERIFY;
            }
        }
    }
    return NA_OK;
}

// 32x32 geometry: v_mfma_f32_32x32x16_bf16 — same pipes, the OTHER tile
// shape with its own fragment maps (A: row=l&31, k=(l>>5)*8+i; B: col
// likewise; C/D per the ISA: col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5),
// 16 accumulator elements per lane).
typedef float f32x16 __attribute__((ext_vector_type(16)));

__global__ void mfma_bf16_tile32_kernel(float* __restrict__ out) {
#if defined(__gfx950__)
    int l = threadIdx.x;
    int arow = l & 31, kbase = (l >> 5) * 8;
    bf16x8 a, b;
    for (int i = 0; i < 8; ++i) {
        a[i] = (__bf16)tile_a(arow, kbase + i);
        b[i] = (__bf16)tile_b(kbase + i, arow);
    }
    f32x16 acc;
    for (int i = 0; i < 16; ++i) acc[i] = 0.f;
    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
    int dcol = l & 31;
    for (int r = 0; r < 16; ++r) {
        int drow = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
        out[drow * 32 + dcol] = acc[r];
    }
#else
    out[threadIdx.x] = -1.0f;
#endif
}

extern "C" int na_mfma_bf16_tile32_check(int dev) {
    HIP_CHECK(hipSetDevice(dev));
    float* out = nullptr;
    HIP_CHECK(hipMalloc(&out, 1024 * sizeof(float)));
    mfma_bf16_tile32_kernel<<<dim3(1), dim3(64)>>>(out);
    HIP_CHECK(hipDeviceSynchronize());
    float host[1024];
    HIP_CHECK(hipMemcpy(host, out, sizeof(host), hipMemcpyDeviceToHost));
    (void)hipFree(out);
    for (int i = 0; i < 32; ++i) {
        for (int j = 0; j < 32; ++j) {
            float ref = 0.f;
            for (int k = 0; k < 16; ++k) ref += tile_a(i, k) * tile_b(k, j);
            if (host[i * 32 + j] != ref) {
                std::snprintf(na_last_error_buf, sizeof(na_last_error_buf),
                              "mfma bf16 32x32 tile: D[%d][%d] got %g want %g", i, j,
                              host[i * 32 + j], ref);
                return NA_ERR_VERIFY;
            }
        }
    }
    return NA_OK;
}

// Block-scaled MX path (gfx950-only): v_mfma_scale_f32_16x16x128_f8f6f4 —
// the ONLY large-K low-precision MFMA and the fp4/fp6/MX serving pipe.
// Two checks in one: (1) layout-correct asymmetric fp8-e4m3 data at
// scale 1.0 against an exact host reference (per-lane A: row=l&15,
// k=(l>>4)*32+i, byte i of <8 x i32>; B: col=l&15, same k; C/D as the
// 16x16 family); (2) scale semantics — E8M0 exponent 128 on A doubles
// the result.
typedef int i32x8 __attribute__((ext_vector_type(8)));

__global__ void mfma_mx_tile_kernel(float* __restrict__ out, int scale_a) {
#if defined(__gfx950__)
    int l = threadIdx.x;
    int arow = l & 15, kbase = (l >> 4) * 32;
    i32x8 a = {0, 0, 0, 0, 0, 0, 0, 0}, b = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int i = 0; i < 32; ++i) {
        a[i / 4] |= (int)e4m3(tile8_a(arow, (kbase + i) % 64)) << (8 * (i % 4));
        b[i / 4] |= (int)e4m3(tile8_b((kbase + i) % 64, arow)) << (8 * (i % 4));
    }
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    int sb = 0x7F7F7F7F;  // E8M0 127 = 1.0 per block
    int sa = scale_a * 0x01010101;
    acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
        a, b, acc, /*cbsz=*/0, /*blgp=*/0, /*opselA=*/0, sa, /*opselB=*/0, sb);
    int dcol = l & 15, drow0 = (l >> 4) * 4;
    for (int i = 0; i < 4; ++i) out[(drow0 + i) * 16 + dcol] = acc[i];
#else
    out[threadIdx.x] = -1.0f;
#endif
}

extern "C" int na_mfma_mx_tile_check(int dev) {
    HIP_CHECK(hipSetDevice(dev));
    float* out = nullptr;
    HIP_CHECK(hipMalloc(&out, 256 * sizeof(float)));
    for (int pass = 0; pass < 2; ++pass) {
        int scale_a = pass == 0 ? 0x7F : 0x80;  // 1.0, then 2.0
        float mult = pass == 0 ? 1.0f : 2.0f;
        mfma_mx_tile_kernel<<<dim3(1), dim3(64)>>>(out, scale_a);
        HIP_CHECK(hipDeviceSynchronize());
        float host[256];
        HIP_CHECK(hipMemcpy(host, out, sizeof(host), hipMemcpyDeviceToHost));
        for (int i = 0; i < 16; ++i) {
            for (int j = 0; j < 16; ++j) {
                float ref = 0.f;
                for (int k = 0; k < 128; ++k)
                    ref += (float)(tile8_a(i, k % 64) * tile8_b(k % 64, j));
                ref *= mult;
                if (host[i * 16 + j] != ref) {
                    std::snprintf(na_last_error_buf, sizeof(na_last_error_buf),
                                  "mfma mx tile (scale %.0fx): D[%d][%d] got %g want %g",
                                  mult, i, j, host[i * 16 + j], ref);
                    (void)hipFree(out);
                    return NA_ERR_VERIFY;
                }
            }
        }
    }
    (void)hipFree(out);
    return NA_OK;
}

// LDS self-test: fill the whole per-WG allocation with a position-dependent
// pattern, barrier, read back through a bank-swizzled index. Exercises the
// LDS array + crossbar across all CUs (one WG per CU's worth of a big grid).
__global__ void lds_selftest_kernel(unsigned* __restrict__ fails, size_t words) {
    extern __shared__ unsigned lds[];
    unsigned t = threadIdx.x, n = blockDim.x;
    for (size_t i = t; i < words; i += n)
        lds[i] = (unsigned)(i * 2654435761u) ^ (blockIdx.x * 97u);
    __syncthreads();
    unsigned bad = 0;
    for (size_t i = t; i < words; i += n) {
        // swizzle: XOR within a 64-dword bank row so every lane group hits
        // addresses written by other lanes (crossbar coverage)
        size_t j = (i & ~(size_t)63) | ((i ^ 37) & 63);
        if (j < words && lds[j] != ((unsigned)(j * 2654435761u) ^ (blockIdx.x * 97u))) bad++;
    }
    if (bad) atomicAdd(fails, bad);
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
    // sweep winner: 16384 WGs × 1024 threads, 4-deep in-chunk unroll
    dim3 grid(16384), block(1024);
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

// Shared verify for the 16x16x32 family: two chained MFMAs, K=32, uniform
// operands → every element = 2*K*a*b.
static int verify_mfma_out(const char* which, const float* host, int n, float want) {
    for (int i = 0; i < n; ++i) {
        if (host[i] != want) {
            std::snprintf(na_last_error_buf, sizeof(na_last_error_buf),
                          "%s selftest: elem %d got %g want %g", which, i, host[i], want);
            return NA_ERR_VERIFY;
        }
    }
    return NA_OK;
}

extern "C" int na_mfma_bf16_selftest(int dev) {
    HIP_CHECK(hipSetDevice(dev));
    // 1.5 and 2.0 are exact in bf16; 2*32*1.5*2.0 = 192 exact in f32
    const float alpha = 1.5f, beta = 2.0f;
    const int n = 64 * 4;
    float* out = nullptr;
    HIP_CHECK(hipMalloc(&out, n * sizeof(float)));
    mfma_bf16_selftest_kernel<<<dim3(1), dim3(64)>>>(out, alpha, beta);
    HIP_CHECK(hipDeviceSynchronize());
    float host[n];
    HIP_CHECK(hipMemcpy(host, out, n * sizeof(float), hipMemcpyDeviceToHost));
    (void)hipFree(out);
    return verify_mfma_out("mfma-bf16", host, n, 2.0f * 32.0f * alpha * beta);
}

extern "C" int na_mfma_fp8_selftest(int dev) {
    HIP_CHECK(hipSetDevice(dev));
    // E4M3: 1.5 = 0x3C, 2.0 = 0x40 (both exact); 2*32*1.5*2.0 = 192
    const long a_bits = 0x3C3C3C3C3C3C3C3CLL;
    const long b_bits = 0x4040404040404040LL;
    const int n = 64 * 4;
    float* out = nullptr;
    HIP_CHECK(hipMalloc(&out, n * sizeof(float)));
    mfma_fp8_selftest_kernel<<<dim3(1), dim3(64)>>>(out, a_bits, b_bits);
    HIP_CHECK(hipDeviceSynchronize());
    float host[n];
    HIP_CHECK(hipMemcpy(host, out, n * sizeof(float), hipMemcpyDeviceToHost));
    (void)hipFree(out);
    return verify_mfma_out("mfma-fp8", host, n, 192.0f);
}

extern "C" int na_lds_selftest(int dev, long long* bytes_tested) {
    HIP_CHECK(hipSetDevice(dev));
    hipDeviceProp_t prop;
    HIP_CHECK(hipGetDeviceProperties(&prop, dev));
    // whole per-WG LDS allocation (160 KiB/CU on gfx950, minus any
    // runtime-reserved slice reported via sharedMemPerBlock)
    size_t lds_bytes = prop.sharedMemPerBlock;
    if (lds_bytes > 160 * 1024) lds_bytes = 160 * 1024;
    size_t words = lds_bytes / sizeof(unsigned);
    unsigned* fails = nullptr;
    HIP_CHECK(hipMalloc(&fails, sizeof(unsigned)));
    HIP_CHECK(hipMemset(fails, 0, sizeof(unsigned)));
    // 2048 WGs: every CU's LDS array gets exercised multiple times
    hipError_t e = hipSuccess;
    lds_selftest_kernel<<<dim3(2048), dim3(256), lds_bytes>>>(fails, words);
    e = hipGetLastError();
    if (e != hipSuccess) {
        (void)hipFree(fails);
        std::snprintf(na_last_error_buf, sizeof(na_last_error_buf), "lds launch: %s",
                      hipGetErrorString(e));
        return NA_ERR_HIP;
    }
    HIP_CHECK(hipDeviceSynchronize());
    unsigned bad = 0;
    HIP_CHECK(hipMemcpy(&bad, fails, sizeof(unsigned), hipMemcpyDeviceToHost));
    (void)hipFree(fails);
    if (bytes_tested) *bytes_tested = (long long)lds_bytes;
    if (bad) {
        std::snprintf(na_last_error_buf, sizeof(na_last_error_buf),
                      "lds selftest: %u mismatched words across %zu-byte LDS", bad, lds_bytes);
        return NA_ERR_VERIFY;
    }
    return NA_OK;
}

extern "C" int na_sdma_bandwidth(int dev, long long bytes, int iters, double* gbs) {
    // device-to-device copy through hipMemcpyAsync: the runtime routes
    // large D2D copies through the SDMA engines — the same hardware RCCL
    // leans on for xGMI peer traffic. A sick SDMA engine shows up here
    // before any collective does.
    HIP_CHECK(hipSetDevice(dev));
    void *src = nullptr, *dst = nullptr;
    HIP_CHECK(hipMalloc(&src, (size_t)bytes));
    hipError_t e2 = hipMalloc(&dst, (size_t)bytes);
    if (e2 != hipSuccess) {
        (void)hipFree(src);
        std::snprintf(na_last_error_buf, sizeof(na_last_error_buf), "%s", hipGetErrorString(e2));
        return NA_ERR_HIP;
    }
    HIP_CHECK(hipMemset(src, 7, (size_t)bytes));
    hipStream_t stream;
    HIP_CHECK(hipStreamCreate(&stream));
    HIP_CHECK(hipMemcpyAsync(dst, src, (size_t)bytes, hipMemcpyDeviceToDevice, stream));  // warmup
    HIP_CHECK(hipStreamSynchronize(stream));
    hipEvent_t t0, t1;
    HIP_CHECK(hipEventCreate(&t0));
    HIP_CHECK(hipEventCreate(&t1));
    HIP_CHECK(hipEventRecord(t0, stream));
    for (int i = 0; i < iters; ++i)
        HIP_CHECK(hipMemcpyAsync(dst, src, (size_t)bytes, hipMemcpyDeviceToDevice, stream));
    HIP_CHECK(hipEventRecord(t1, stream));
    HIP_CHECK(hipEventSynchronize(t1));
    float ms = 0.f;
    HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
    *gbs = (2.0 * (double)bytes * iters) / (ms * 1e6);  // read + write
    (void)hipEventDestroy(t0);
    (void)hipEventDestroy(t1);
    (void)hipStreamDestroy(stream);
    (void)hipFree(src);
    (void)hipFree(dst);
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
