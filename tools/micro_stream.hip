/*
 * micro_stream — backs the k_scan ceiling note in profiles/README.md
 * with measured numbers on THIS part:
 *   1. read-only dense stream (grid-stride i64 reads, reduced to sink)
 *   2. copy stream (read + write, i64)
 *   3. scan-shape: 50%-selective predicate (i32) + 5-column survivor
 *      gather + per-wave staged compaction write — the exact structure
 *      of k_scan on the C3 lineitem pass, minus projection arithmetic.
 * Reported as effective TB/s over ALGORITHMIC bytes (reads count the
 * full touched columns, writes count compacted survivors only).
 *
 *   hipcc --offload-arch=gfx950 -O3 micro_stream.hip -o micro_stream
 *   ./micro_stream [n_rows_millions]
 */
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>

#define HIP_CHECK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
    printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); \
    return 1; } } while (0)

__global__ void k_read(const uint64_t *a, int64_t n, uint64_t *sink) {
    uint64_t s = 0;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += stride)
        s += a[i];
    if (s == 0xdeadbeef) *sink = s; /* never true; defeats DCE */
}

__global__ void k_copy(const uint64_t *a, uint64_t *b, int64_t n) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += stride)
        b[i] = a[i];
}

#define STAGE 512
__global__ void k_scanshape(const int32_t *pred, const uint64_t *c0,
                            const uint64_t *c1, const uint64_t *c2,
                            const uint64_t *c3, const uint64_t *c4,
                            int64_t n, int32_t cut, uint32_t *counter,
                            uint64_t *o0, uint64_t *o1, uint64_t *o2,
                            uint32_t cap) {
    __shared__ uint32_t s_rows[4][STAGE];
    const int wid = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;
    uint32_t *stage = s_rows[wid];
    uint32_t cnt = 0;

    auto flush = [&]() {
        if (cnt == 0) return;
        uint32_t base = 0;
        if (lane == 0) base = atomicAdd(counter, cnt);
        base = (uint32_t)__shfl((int)base, 0, 64);
        for (uint32_t k = (uint32_t)lane; k < cnt; k += 64) {
            uint32_t at = base + k;
            if (at >= cap) continue;
            int64_t src = stage[k];
            o0[at] = c0[src];
            o1[at] = c1[src] * (100u - (uint32_t)c2[src]);
            o2[at] = c3[src] + c4[src];
        }
        cnt = 0;
    };
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;;
         i += stride) {
        bool active = i < n;
        if (!__ballot(active)) break;
        bool want = active && pred[i] > cut;
        unsigned long long m = __ballot(want);
        uint32_t add = (uint32_t)__popcll(m);
        if (cnt + add > STAGE) flush();
        if (want)
            stage[cnt + (uint32_t)__popcll(m & ((1ull << lane) - 1ull))] =
                (uint32_t)i;
        cnt += add;
    }
    flush();
}

__global__ void k_fill(uint64_t *a, int64_t n, uint64_t seed) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += stride) {
        uint64_t x = (uint64_t)i * 0x9E3779B97F4A7C15ull + seed;
        x ^= x >> 29; x *= 0xBF58476D1CE4E5B9ull; x ^= x >> 32;
        a[i] = x;
    }
}
__global__ void k_fill_pred(int32_t *p, int64_t n) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += stride) {
        uint64_t x = (uint64_t)i * 0x9E3779B97F4A7C15ull + 7;
        x ^= x >> 29; x *= 0xBF58476D1CE4E5B9ull; x ^= x >> 32;
        p[i] = (int32_t)(x % 1000);
    }
}

static float time_ms(hipEvent_t a, hipEvent_t b) {
    float ms = 0;
    hipEventElapsedTime(&ms, a, b);
    return ms;
}

int main(int argc, char **argv) {
    int64_t n = (argc > 1 ? atoll(argv[1]) : 400) * 1000000LL;
    const int blocks = 4096, tpb = 256, reps = 5;
    uint64_t *c[5], *o[3], *sink;
    int32_t *pred;
    uint32_t *counter;
    for (int i = 0; i < 5; i++) HIP_CHECK(hipMalloc(&c[i], n * 8));
    for (int i = 0; i < 3; i++) HIP_CHECK(hipMalloc(&o[i], n * 8));
    HIP_CHECK(hipMalloc(&pred, n * 4));
    HIP_CHECK(hipMalloc(&counter, 4));
    HIP_CHECK(hipMalloc(&sink, 8));
    for (int i = 0; i < 5; i++)
        hipLaunchKernelGGL(k_fill, dim3(blocks), dim3(tpb), 0, 0, c[i], n,
                           (uint64_t)i);
    hipLaunchKernelGGL(k_fill_pred, dim3(blocks), dim3(tpb), 0, 0, pred, n);
    HIP_CHECK(hipDeviceSynchronize());
    hipEvent_t e0, e1;
    hipEventCreate(&e0);
    hipEventCreate(&e1);

    /* 1. read-only */
    double best = 1e30;
    for (int r = 0; r < reps; r++) {
        hipEventRecord(e0);
        hipLaunchKernelGGL(k_read, dim3(blocks), dim3(tpb), 0, 0, c[0], n,
                           sink);
        hipEventRecord(e1);
        HIP_CHECK(hipDeviceSynchronize());
        double ms = time_ms(e0, e1);
        if (ms < best) best = ms;
    }
    printf("read_only: %.2f TB/s (%.2f ms, %.1f GB)\n",
           n * 8 / best / 1e9, best, n * 8 / 1e9);

    /* 2. copy */
    best = 1e30;
    for (int r = 0; r < reps; r++) {
        hipEventRecord(e0);
        hipLaunchKernelGGL(k_copy, dim3(blocks), dim3(tpb), 0, 0, c[0], o[0],
                           n);
        hipEventRecord(e1);
        HIP_CHECK(hipDeviceSynchronize());
        double ms = time_ms(e0, e1);
        if (ms < best) best = ms;
    }
    printf("copy: %.2f TB/s (%.2f ms, %.1f GB moved)\n",
           n * 16 / best / 1e9, best, n * 16 / 1e9);

    /* 3. scan shape at ~54%% selectivity (pred > 460 of 0..999) */
    best = 1e30;
    uint32_t kept = 0;
    for (int r = 0; r < reps; r++) {
        HIP_CHECK(hipMemset(counter, 0, 4));
        hipEventRecord(e0);
        hipLaunchKernelGGL(k_scanshape, dim3(blocks), dim3(tpb), 0, 0, pred,
                           c[0], c[1], c[2], c[3], c[4], n, 460, counter,
                           o[0], o[1], o[2], (uint32_t)n);
        hipEventRecord(e1);
        HIP_CHECK(hipDeviceSynchronize());
        HIP_CHECK(hipMemcpy(&kept, counter, 4, hipMemcpyDeviceToHost));
        double ms = time_ms(e0, e1);
        if (ms < best) best = ms;
    }
    double rd = n * 4.0 + n * 5 * 8.0;      /* pred + 5 cols ~fully touched */
    double wr = kept * 3 * 8.0;
    printf("scan_shape: %.2f TB/s algorithmic (%.2f ms, kept %.1f%%, "
           "%.1f GB read + %.1f GB written)\n",
           (rd + wr) / best / 1e9, best, 100.0 * kept / n, rd / 1e9,
           wr / 1e9);
    return 0;
}
