/*
 * micro_atomic — decides the C4 (Q18 150M-group agg) question: does the
 * ~1.8 TB/s random-atomic line ceiling lift when the target table is
 * Infinity-Cache-resident (<= 256 MB)? If yes, partitioning the input by
 * group-key range and building L3-resident sub-tables sequentially would
 * beat the single fused pass; if no, the fused agg is at its true wall.
 *
 *   hipcc --offload-arch=gfx950 -O3 micro_atomic.hip -o micro_atomic
 *   ./micro_atomic
 */
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>
#include <vector>

#define HIP_CHECK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
    printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); \
    return 1; } } while (0)

__global__ void k_atomic_rmw(unsigned long long *tab, uint64_t mask,
                             int64_t n_ops) {
    uint64_t x = (uint64_t)(blockIdx.x * blockDim.x + threadIdx.x) *
                 0x9E3779B97F4A7C15ull + 12345;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         i < n_ops; i += stride) {
        x ^= x >> 29; x *= 0xBF58476D1CE4E5B9ull; x ^= x >> 32;
        atomicAdd(&tab[x & mask], 1ull);
    }
}

__global__ void k_plain_read(const unsigned long long *tab, uint64_t mask,
                             int64_t n_ops, unsigned long long *sink) {
    uint64_t x = (uint64_t)(blockIdx.x * blockDim.x + threadIdx.x) *
                 0x9E3779B97F4A7C15ull + 777;
    unsigned long long acc = 0;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         i < n_ops; i += stride) {
        x ^= x >> 29; x *= 0xBF58476D1CE4E5B9ull; x ^= x >> 32;
        acc += tab[x & mask];
    }
    if (acc == 0xDEADBEEF) *sink = acc; /* keep the loads alive */
}

int main() {
    const int64_t n_ops = 1LL << 29; /* 512M ops */
    std::vector<int64_t> sizes_mb = {32, 64, 128, 192, 256, 512, 1024,
                                     2048, 4800};
    unsigned long long *sink;
    HIP_CHECK(hipMalloc(&sink, 8));
    printf("%8s %14s %14s %14s %14s\n", "size_MB", "atomic_Gops", "atomic_TBps",
           "read_Gops", "read_TBps");
    for (int64_t mb : sizes_mb) {
        uint64_t words = (uint64_t)mb << 20 >> 3;
        uint64_t mask = 1;
        while ((mask << 1) <= words) mask <<= 1;
        mask -= 1;
        unsigned long long *tab;
        HIP_CHECK(hipMalloc(&tab, (mask + 1) * 8));
        HIP_CHECK(hipMemset(tab, 0, (mask + 1) * 8));
        hipEvent_t e0, e1;
        HIP_CHECK(hipEventCreate(&e0));
        HIP_CHECK(hipEventCreate(&e1));
        /* warmup */
        hipLaunchKernelGGL(k_atomic_rmw, dim3(4096), dim3(256), 0, 0, tab,
                           mask, n_ops / 8);
        HIP_CHECK(hipDeviceSynchronize());
        HIP_CHECK(hipEventRecord(e0));
        hipLaunchKernelGGL(k_atomic_rmw, dim3(4096), dim3(256), 0, 0, tab,
                           mask, n_ops);
        HIP_CHECK(hipEventRecord(e1));
        HIP_CHECK(hipEventSynchronize(e1));
        float ms_a = 0;
        HIP_CHECK(hipEventElapsedTime(&ms_a, e0, e1));
        hipLaunchKernelGGL(k_plain_read, dim3(4096), dim3(256), 0, 0, tab,
                           mask, n_ops / 8, sink);
        HIP_CHECK(hipDeviceSynchronize());
        HIP_CHECK(hipEventRecord(e0));
        hipLaunchKernelGGL(k_plain_read, dim3(4096), dim3(256), 0, 0, tab,
                           mask, n_ops, sink);
        HIP_CHECK(hipEventRecord(e1));
        HIP_CHECK(hipEventSynchronize(e1));
        float ms_r = 0;
        HIP_CHECK(hipEventElapsedTime(&ms_r, e0, e1));
        double gops_a = n_ops / (ms_a * 1e6);
        double gops_r = n_ops / (ms_r * 1e6);
        printf("%8lld %14.2f %14.2f %14.2f %14.2f\n", (long long)mb,
               gops_a, gops_a * 64 / 1000.0, gops_r, gops_r * 64 / 1000.0);
        HIP_CHECK(hipFree(tab));
        HIP_CHECK(hipEventDestroy(e0));
        HIP_CHECK(hipEventDestroy(e1));
    }
    HIP_CHECK(hipFree(sink));
    return 0;
}
