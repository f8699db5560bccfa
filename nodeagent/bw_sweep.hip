// HBM copy-kernel variant sweep (dev tool, not shipped in the agent):
// picks the fastest streaming-copy shape for the na_hbm_bandwidth probe.
// Build: hipcc --offload-arch=gfx950 -O3 nodeagent/bw_sweep.hip -o bw_sweep
#include <hip/hip_runtime.h>

#include <cstdio>

typedef float f4v __attribute__((ext_vector_type(4)));

#define CK(x)                                                                  \
    do {                                                                       \
        hipError_t e = (x);                                                    \
        if (e != hipSuccess) {                                                 \
            printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__);    \
            return 1;                                                          \
        }                                                                      \
    } while (0)

__global__ void copy_plain(const f4v* __restrict__ s, f4v* __restrict__ d, size_t n) {
    size_t i = blockIdx.x * (size_t)blockDim.x + threadIdx.x;
    size_t st = (size_t)gridDim.x * blockDim.x;
    for (; i < n; i += st) d[i] = s[i];
}

__global__ void copy_nt(const f4v* __restrict__ s, f4v* __restrict__ d, size_t n) {
    size_t i = blockIdx.x * (size_t)blockDim.x + threadIdx.x;
    size_t st = (size_t)gridDim.x * blockDim.x;
    for (; i < n; i += st)
        __builtin_nontemporal_store(__builtin_nontemporal_load(&s[i]), &d[i]);
}

__global__ void copy_nt_u4(const f4v* __restrict__ s, f4v* __restrict__ d, size_t n) {
    size_t i = blockIdx.x * (size_t)blockDim.x + threadIdx.x;
    size_t st = (size_t)gridDim.x * blockDim.x;
    for (; i + 3 * st < n; i += 4 * st) {
        f4v a = __builtin_nontemporal_load(&s[i]);
        f4v b = __builtin_nontemporal_load(&s[i + st]);
        f4v c = __builtin_nontemporal_load(&s[i + 2 * st]);
        f4v e = __builtin_nontemporal_load(&s[i + 3 * st]);
        __builtin_nontemporal_store(a, &d[i]);
        __builtin_nontemporal_store(b, &d[i + st]);
        __builtin_nontemporal_store(c, &d[i + 2 * st]);
        __builtin_nontemporal_store(e, &d[i + 3 * st]);
    }
    for (; i < n; i += st)
        __builtin_nontemporal_store(__builtin_nontemporal_load(&s[i]), &d[i]);
}

// contiguous-chunk-per-block (no grid-stride): each WG owns one contiguous
// slice; consecutive iterations stay in the same DRAM window
__global__ void copy_nt_chunk(const f4v* __restrict__ s, f4v* __restrict__ d, size_t n) {
    size_t per = (n + gridDim.x - 1) / gridDim.x;
    size_t lo = blockIdx.x * per, hi = min(lo + per, n);
    for (size_t i = lo + threadIdx.x; i < hi; i += blockDim.x)
        __builtin_nontemporal_store(__builtin_nontemporal_load(&s[i]), &d[i]);
}

__global__ void copy_plain_u2(const f4v* __restrict__ s, f4v* __restrict__ d, size_t n) {
    size_t i = blockIdx.x * (size_t)blockDim.x + threadIdx.x;
    size_t st = (size_t)gridDim.x * blockDim.x;
    for (; i + st < n; i += 2 * st) {
        f4v a = s[i], b = s[i + st];
        d[i] = a;
        d[i + st] = b;
    }
    for (; i < n; i += st) d[i] = s[i];
}

template <typename K>
double run(K kernel, const f4v* s, f4v* d, size_t n, int wgs, int block, int iters) {
    kernel<<<dim3(wgs), dim3(block)>>>(const_cast<f4v*>(s), d, n);  // warmup
    (void)hipDeviceSynchronize();
    hipEvent_t t0, t1;
    (void)hipEventCreate(&t0);
    (void)hipEventCreate(&t1);
    (void)hipEventRecord(t0);
    for (int i = 0; i < iters; ++i) kernel<<<dim3(wgs), dim3(block)>>>(const_cast<f4v*>(s), d, n);
    (void)hipEventRecord(t1);
    (void)hipEventSynchronize(t1);
    float ms = 0;
    (void)hipEventElapsedTime(&ms, t0, t1);
    (void)hipEventDestroy(t0);
    (void)hipEventDestroy(t1);
    return (2.0 * n * sizeof(f4v) * iters) / (ms * 1e6);  // GB/s
}

int main() {
    size_t bytes = 1ull << 31;  // 2 GiB each way: well past Infinity Cache
    size_t n = bytes / sizeof(f4v);
    f4v *s, *d;
    CK(hipMalloc(&s, bytes));
    CK(hipMalloc(&d, bytes));
    CK(hipMemset(s, 1, bytes));
    const int iters = 10;
    struct {
        const char* name;
        double gbs;
    } best{"", 0};
    for (int block : {256, 512, 1024}) {
        for (int wgs : {1024, 2048, 4096, 8192}) {
            double g;
            g = run(copy_plain, s, d, n, wgs, block, iters);
            printf("plain     block=%4d wgs=%4d  %7.1f GB/s\n", block, wgs, g);
            if (g > best.gbs) best = {"plain", g};
            g = run(copy_nt, s, d, n, wgs, block, iters);
            printf("nt        block=%4d wgs=%4d  %7.1f GB/s\n", block, wgs, g);
            if (g > best.gbs) best = {"nt", g};
            g = run(copy_nt_u4, s, d, n, wgs, block, iters);
            printf("nt_u4     block=%4d wgs=%4d  %7.1f GB/s\n", block, wgs, g);
            if (g > best.gbs) best = {"nt_u4", g};
            g = run(copy_nt_chunk, s, d, n, wgs, block, iters);
            printf("nt_chunk  block=%4d wgs=%4d  %7.1f GB/s\n", block, wgs, g);
            if (g > best.gbs) best = {"nt_chunk", g};
            g = run(copy_plain_u2, s, d, n, wgs, block, iters);
            printf("plain_u2  block=%4d wgs=%4d  %7.1f GB/s\n", block, wgs, g);
            if (g > best.gbs) best = {"plain_u2", g};
        }
    }
    printf("BEST: %s %.1f GB/s\n", best.name, best.gbs);
    (void)hipFree(s);
    (void)hipFree(d);
    return 0;
}
