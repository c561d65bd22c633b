/*
 * gx_common.h — shared device/host plumbing for the gfx950 operator pack.
 *
 * Hash chain (device) is Java-exact where routing parity requires it
 * (partition exchange), and reused for join/agg bucket placement (free —
 * placement never changes result rows):
 *   row hash   h = 31*h + blockHash      (chunk/Chunk.java:116-129)
 *   i64 hash   (int)(v ^ v>>>32)         (chunk/LongBlock.java:110-127)
 *   i32 hash   v                         (chunk/IntegerBlock.java:112-117)
 *   f64 hash   Long.hashCode(bits)       (chunk/DoubleBlock.java)
 *   bucket mix fastutil HashCommon.mix   (ConcurrentRawHashTable.java:93)
 *   partition  fastutil murmurHash3      (utils/ExecUtils.java:1023-1033)
 */
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <map>
#include <mutex>
#include <string>
#include <vector>

#include "../../include/gxop.h"

#define WAVE 64

/* ---- error plumbing ---------------------------------------------------- */

extern thread_local std::string gx_err;

static inline void gx_set_err(const std::string &m) { gx_err = m; }

#define HIP_OK(call)                                                         \
    do {                                                                     \
        hipError_t _e = (call);                                              \
        if (_e != hipSuccess) {                                              \
            gx_set_err(std::string(#call) + ": " + hipGetErrorString(_e));   \
            return -1;                                                       \
        }                                                                    \
    } while (0)

#define HIP_OK_NULL(call)                                                    \
    do {                                                                     \
        hipError_t _e = (call);                                              \
        if (_e != hipSuccess) {                                              \
            gx_set_err(std::string(#call) + ": " + hipGetErrorString(_e));   \
            return nullptr;                                                  \
        }                                                                    \
    } while (0)

/* ---- Java-exact hashes (device + host versions) ------------------------ */

__host__ __device__ static inline int32_t gx_mix(int32_t x) {
    int32_t h = (int32_t)((uint32_t)x * 0x9E3779B9u);
    return h ^ (int32_t)((uint32_t)h >> 16);
}
__host__ __device__ static inline int32_t gx_murmur3(int32_t x) {
    uint32_t h = (uint32_t)x;
    h ^= h >> 16; h *= 0x85ebca6bu; h ^= h >> 13; h *= 0xc2b2ae35u; h ^= h >> 16;
    return (int32_t)h;
}
__host__ __device__ static inline int32_t gx_hash_i64(int64_t v) {
    return (int32_t)((uint64_t)v ^ ((uint64_t)v >> 32));
}
__host__ __device__ static inline int32_t gx_hash_f64(double v) {
    uint64_t bits = __builtin_bit_cast(uint64_t, v);
    /* Java doubleToLongBits canonicalizes NaN */
    if (v != v) bits = 0x7ff8000000000000ull;
    return (int32_t)(bits ^ (bits >> 32));
}

/* ---- device column store ------------------------------------------------
 * SoA accumulation of consumed chunks in HBM (the device ChunksIndex).
 * Fixed-width types only on the device path for now (i64/i32/f64);
 * GX_SLICE columns are appended as offsets+bytes for payload gather.      */

/* Caching device allocator: hipMalloc is slow for GB-size buffers and
 * hipFree synchronizes the device — per-step churn dominated early bench
 * steps (rocprof: 10 ms kernels vs 170 ms wall). Exact-size-class reuse;
 * sizes are rounded up to 1 MiB classes (small: pow2) so repeated operator
 * passes hit the cache. Keyed by device. */
class DevPool {
    std::mutex mu_;
    std::multimap<std::pair<int, size_t>, void *> free_;
    size_t cached_ = 0;
    static constexpr size_t CAP = 48ull << 30;

public:
    static DevPool &inst() {
        static DevPool p;
        return p;
    }
    static size_t size_class(size_t sz) {
        if (sz < (1 << 20)) {
            size_t c = 4096;
            while (c < sz) c <<= 1;
            return c;
        }
        return (sz + (1 << 20) - 1) & ~(size_t)((1 << 20) - 1);
    }
    hipError_t alloc(size_t sz, void **out) {
        int dev = 0;
        (void)hipGetDevice(&dev);
        sz = size_class(sz);
        {
            std::lock_guard<std::mutex> g(mu_);
            auto it = free_.find({dev, sz});
            if (it != free_.end()) {
                *out = it->second;
                cached_ -= sz;
                free_.erase(it);
                return hipSuccess;
            }
        }
        hipError_t e = hipMalloc(out, sz);
        if (e == hipErrorOutOfMemory) {
            trim();
            e = hipMalloc(out, sz);
        }
        return e;
    }
    void dealloc(size_t sz, void *p) {
        if (!p) return;
        int dev = 0;
        (void)hipGetDevice(&dev);
        sz = size_class(sz);
        std::lock_guard<std::mutex> g(mu_);
        if (cached_ + sz > CAP) {
            (void)hipFree(p);
            return;
        }
        free_.insert({{dev, sz}, p});
        cached_ += sz;
    }
    void trim() {
        std::lock_guard<std::mutex> g(mu_);
        for (auto &kv : free_) (void)hipFree(kv.second);
        free_.clear();
        cached_ = 0;
    }
};

struct DevBuf {
    void *p = nullptr;
    size_t cap = 0;   /* bytes allocated (size-class rounded) */
    int grow(size_t need, hipStream_t s) {
        if (need <= cap) return 0;
        size_t ncap = DevPool::size_class(
            cap && need < cap + cap / 2 ? cap + cap / 2 : need);
        void *np = nullptr;
        HIP_OK(DevPool::inst().alloc(ncap, &np));
        size_t actual = DevPool::size_class(ncap);
        if (p) {
            HIP_OK(hipMemcpyAsync(np, p, cap, hipMemcpyDeviceToDevice, s));
            HIP_OK(hipStreamSynchronize(s));
            DevPool::inst().dealloc(cap, p);
        }
        p = np; cap = actual;
        return 0;
    }
    void release() {
        if (p) DevPool::inst().dealloc(cap, p);
        p = nullptr;
        cap = 0;
    }
};

struct DevColumn {
    int32_t type = GX_I64;
    DevBuf values;        /* elem_size * n */
    DevBuf nulls;         /* u8 * n; allocated lazily on first null-bearing chunk */
    bool has_nulls = false;
    DevBuf offsets;       /* SLICE: i32 end-offsets (global) */
    DevBuf bytes;         /* SLICE payload */
    int64_t byte_len = 0;

    size_t elem_size() const {
        switch (type) {
        case GX_I64: case GX_F64: return 8;
        case GX_I32: return 4;
        case GX_DECIMAL: return 40;
        default: return 0;
        }
    }
};

/* Device-side view of one column (POD, passed to kernels). */
struct DevColView {
    int32_t type;
    int32_t has_nulls;
    const void *values;
    const uint8_t *nulls;
    const int32_t *offsets;
    const uint8_t *bytes;
};

/* rebase chunk-local slice end-offsets to store-global offsets */
__global__ void k_rebase_i32(int32_t *off, int64_t n, int32_t base) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x)
        off[i] += base;
}

struct DevStore {
    std::vector<DevColumn> cols;
    int64_t n_rows = 0;
    hipStream_t stream = nullptr;

    void init(int32_t n, const int32_t *types, hipStream_t s) {
        cols.resize(n);
        for (int32_t i = 0; i < n; i++) cols[i].type = types[i];
        stream = s;
    }

    int reserve(int64_t rows) {
        for (auto &c : cols) {
            if (c.elem_size())
                if (c.values.grow((size_t)rows * c.elem_size(), stream)) return -1;
        }
        return 0;
    }

    /* append one gx_chunk (host or device pointers). */
    int append(const gx_chunk *ch) {
        if ((size_t)ch->n_blocks != cols.size()) { gx_set_err("column count mismatch"); return -1; }
        int64_t old = n_rows, n = ch->n_rows;
        for (int32_t ci = 0; ci < ch->n_blocks; ci++) {
            const gx_block *b = &ch->blocks[ci];
            DevColumn &c = cols[ci];
            if (b->type != c.type) { gx_set_err("column type mismatch"); return -1; }
            hipMemcpyKind kind = b->mem == GX_MEM_DEVICE ? hipMemcpyDeviceToDevice
                                                         : hipMemcpyHostToDevice;
            if (c.elem_size()) {
                size_t es = c.elem_size();
                if (c.values.grow((size_t)(old + n) * es, stream)) return -1;
                HIP_OK(hipMemcpyAsync((char *)c.values.p + (size_t)old * es,
                                      b->values, (size_t)n * es, kind, stream));
            } else { /* SLICE */
                if (c.offsets.grow((size_t)(old + n) * 4, stream)) return -1;
                /* copy raw offsets then rebase by byte_len on device (or host)
                 * — for simplicity copy via host when small; offsets arrive
                 * chunk-local. Rebase kernel is in gxhip.hip. */
                HIP_OK(hipMemcpyAsync((char *)c.offsets.p + (size_t)old * 4,
                                      b->offsets, (size_t)n * 4, kind, stream));
                int64_t blen = 0;
                if (n > 0) {
                    if (b->mem == GX_MEM_DEVICE) {
                        int32_t last;
                        HIP_OK(hipMemcpyAsync(&last, (const char *)b->offsets + (n - 1) * 4,
                                              4, hipMemcpyDeviceToHost, stream));
                        HIP_OK(hipStreamSynchronize(stream));
                        blen = last;
                    } else {
                        blen = b->offsets[n - 1];
                    }
                }
                if (blen > 0) {
                    if (c.bytes.grow((size_t)(c.byte_len + blen), stream)) return -1;
                    HIP_OK(hipMemcpyAsync((char *)c.bytes.p + c.byte_len, b->data,
                                          (size_t)blen, kind, stream));
                }
                if (c.byte_len > 0 && n > 0) {
                    int64_t g = (n + 255) / 256;
                    if (g > 4096) g = 4096;
                    hipLaunchKernelGGL(k_rebase_i32, dim3((uint32_t)g), dim3(256),
                                       0, stream,
                                       (int32_t *)c.offsets.p + old, n,
                                       (int32_t)c.byte_len);
                }
                c.byte_len += blen;
            }
            /* nulls: u8 per row; lazily materialize zeros when first needed */
            if (b->nulls) {
                if (!c.has_nulls) {
                    if (c.nulls.grow((size_t)(old + n), stream)) return -1;
                    if (old > 0)
                        HIP_OK(hipMemsetAsync(c.nulls.p, 0, (size_t)old, stream));
                    c.has_nulls = true;
                } else if (c.nulls.grow((size_t)(old + n), stream)) return -1;
                HIP_OK(hipMemcpyAsync((char *)c.nulls.p + old, b->nulls, (size_t)n,
                                      kind, stream));
            } else if (c.has_nulls) {
                if (c.nulls.grow((size_t)(old + n), stream)) return -1;
                HIP_OK(hipMemsetAsync((char *)c.nulls.p + old, 0, (size_t)n, stream));
            }
        }
        n_rows += n;
        return 0;
    }

    DevColView view(int32_t ci) const {
        const DevColumn &c = cols[ci];
        DevColView v;
        v.type = c.type;
        v.has_nulls = c.has_nulls ? 1 : 0;
        v.values = c.values.p;
        v.nulls = (const uint8_t *)c.nulls.p;
        v.offsets = (const int32_t *)c.offsets.p;
        v.bytes = (const uint8_t *)c.bytes.p;
        return v;
    }

    void release() {
        for (auto &c : cols) {
            c.values.release(); c.nulls.release();
            c.offsets.release(); c.bytes.release();
        }
        cols.clear(); n_rows = 0;
    }
};

/* round up to next power of two (>=2) */
static inline int64_t gx_pow2(int64_t x) {
    if (x < 2) return 2;
    x--;
    x |= x >> 1; x |= x >> 2; x |= x >> 4; x |= x >> 8; x |= x >> 16; x |= x >> 32;
    return x + 1;
}
