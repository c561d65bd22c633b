/*
 * gxhip.hip — MI355X (gfx950) native implementation of the gxop C-ABI:
 * the PolarDB-X CN MPP hot path (hash join, hash aggregation, partition
 * exchange) as hand-written HIP kernels. This is the PRODUCT path — no CPU
 * fallback exists; loading this library off-GPU fails at first use.
 *
 * Design (MI355X-first; everything here is HBM-latency/bandwidth bound,
 * MFMA unused — SURVEY.md §8d):
 *  - Join build: CSR bucket table (histogram → exclusive scan → scatter into
 *    contiguous 16-B {key,pos} entries). A bucket's entries are ADJACENT, so
 *    one probe usually costs one 16-B random read (vs the reference's
 *    pointer-chase chains, AbstractHashJoinExec.java:80-106).
 *  - Probe: grid-stride waves, ballot-compacted match emission (one
 *    atomicAdd per wave per batch), then per-column coalesced-write gathers.
 *  - Agg: streaming two-kernel insert/accumulate per consumed chunk over an
 *    open-addressing claim table — no intra-wave spins (SIMT-safe), group
 *    ids stable across rehash.
 *  - Partition: Java-exact routing (ExecUtils.partition, murmurHash3) so
 *    rows land on the same partition the reference would send them to.
 *
 * Semantics restated from (paths under /root/reference/polardbx-executor/
 * src/main/java/com/alibaba/polardbx/executor/):
 *   ParallelHashJoinExec.java (build/probe/outer-null bitmap)
 *   AbstractBufferedJoinExec.java:185-266 (probe row loop, semi/anti rules)
 *   AbstractJoinExec.java:102-227 (output schemas per join type)
 *   ExecUtils.buildOneChunk:914-944 (NULL-key build rows skipped)
 *   AggOpenHashMap.java:100-139 / GroupOpenHashMap.java:142-193
 *   calc/aggfunctions/* (accumulator null/init semantics)
 *   PartitioningExchanger.java:71-134
 */
#include "gx_common.h"

#include <algorithm>
#include <cstdlib>
#include <deque>
#include <mutex>
#include <hipcub/hipcub.hpp>

thread_local std::string gx_err;

/* ================= kernel-side column descriptors ====================== */

static inline size_t gx_fixed_size(int32_t t) {
    switch (t) {
    case GX_I64: case GX_F64: return 8;
    case GX_I32: return 4;
    case GX_DECIMAL: return 40;
    default: return 0;
    }
}

#define GX_MAX_KEYS 4
#define GX_MAX_COLS 16

struct KeyViews {
    int32_t n;
    DevColView col[GX_MAX_KEYS];
};


/* airlift Slice.hashCode = (int) XxHash64(seed 0) over raw bytes
 * (chunk/SliceBlock.java:183-195, non-"compatible" path) — device
 * restatement of the published algorithm; placement-only. */
#define XXP1 0x9E3779B185EBCA87ull
#define XXP2 0xC2B2AE3D27D4EB4Full
#define XXP3 0x165667B19E3779F9ull
#define XXP4 0x85EBCA77C2B2AE63ull
#define XXP5 0x27D4EB2F165667C5ull

__device__ static inline uint64_t xx_rotl(uint64_t x, int r) {
    return (x << r) | (x >> (64 - r));
}
__device__ static uint64_t xxhash64_dev(const uint8_t *data, int64_t len) {
    const uint8_t *p = data, *end = data + len;
    uint64_t h;
    if (len >= 32) {
        uint64_t v1 = XXP1 + XXP2, v2 = XXP2, v3 = 0, v4 = (uint64_t)0 - XXP1;
        const uint8_t *limit = end - 32;
        do {
            uint64_t a, b, c, d;
            __builtin_memcpy(&a, p, 8); __builtin_memcpy(&b, p + 8, 8);
            __builtin_memcpy(&c, p + 16, 8); __builtin_memcpy(&d, p + 24, 8);
            v1 = xx_rotl(v1 + a * XXP2, 31) * XXP1;
            v2 = xx_rotl(v2 + b * XXP2, 31) * XXP1;
            v3 = xx_rotl(v3 + c * XXP2, 31) * XXP1;
            v4 = xx_rotl(v4 + d * XXP2, 31) * XXP1;
            p += 32;
        } while (p <= limit);
        h = xx_rotl(v1, 1) + xx_rotl(v2, 7) + xx_rotl(v3, 12) + xx_rotl(v4, 18);
        h ^= xx_rotl(v1 * XXP2, 31) * XXP1; h = h * XXP1 + XXP4;
        h ^= xx_rotl(v2 * XXP2, 31) * XXP1; h = h * XXP1 + XXP4;
        h ^= xx_rotl(v3 * XXP2, 31) * XXP1; h = h * XXP1 + XXP4;
        h ^= xx_rotl(v4 * XXP2, 31) * XXP1; h = h * XXP1 + XXP4;
    } else {
        h = XXP5;
    }
    h += (uint64_t)len;
    while (p + 8 <= end) {
        uint64_t k; __builtin_memcpy(&k, p, 8);
        h ^= xx_rotl(k * XXP2, 31) * XXP1;
        h = xx_rotl(h, 27) * XXP1 + XXP4; p += 8;
    }
    if (p + 4 <= end) {
        uint32_t k; __builtin_memcpy(&k, p, 4);
        h ^= (uint64_t)k * XXP1;
        h = xx_rotl(h, 23) * XXP2 + XXP3; p += 4;
    }
    while (p < end) { h ^= (*p) * XXP5; h = xx_rotl(h, 11) * XXP1; p++; }
    h ^= h >> 33; h *= XXP2; h ^= h >> 29; h *= XXP3; h ^= h >> 32;
    return h;
}

__device__ static inline int32_t slice_begin(const DevColView &c, int64_t i) {
    return i > 0 ? c.offsets[i - 1] : 0;
}

__device__ static inline bool col_is_null(const DevColView &c, int64_t i) {
    return c.has_nulls && c.nulls[i];
}

__device__ static inline int32_t col_hash(const DevColView &c, int64_t i) {
    if (col_is_null(c, i)) return 0;
    switch (c.type) {
    case GX_I64: return gx_hash_i64(((const int64_t *)c.values)[i]);
    case GX_I32: return ((const int32_t *)c.values)[i];
    case GX_F64: return gx_hash_f64(((const double *)c.values)[i]);
    case GX_SLICE: {
        int32_t b = slice_begin(c, i), e = c.offsets[i];
        return (int32_t)xxhash64_dev(c.bytes + b, e - b);
    }
    }
    return 0;
}

__device__ static inline bool col_eq(const DevColView &a, int64_t i,
                                     const DevColView &b, int64_t j) {
    bool n1 = col_is_null(a, i), n2 = col_is_null(b, j);
    if (n1 && n2) return true;
    if (n1 != n2) return false;
    switch (a.type) {
    case GX_I64: return ((const int64_t *)a.values)[i] == ((const int64_t *)b.values)[j];
    case GX_I32: return ((const int32_t *)a.values)[i] == ((const int32_t *)b.values)[j];
    case GX_F64: return ((const double *)a.values)[i] == ((const double *)b.values)[j];
    case GX_SLICE: {
        int32_t ab = slice_begin(a, i), ae = a.offsets[i];
        int32_t bb = slice_begin(b, j), be = b.offsets[j];
        if (ae - ab != be - bb) return false;
        for (int32_t k = 0; k < ae - ab; k++)
            if (a.bytes[ab + k] != b.bytes[bb + k]) return false;
        return true;
    }
    }
    return false;
}

/* row hash over key columns: h = 31*h + colHash (chunk/Chunk.java:116-129) */
__device__ static inline int32_t row_hash(const KeyViews &k, int64_t i) {
    int32_t h = 0;
    for (int c = 0; c < k.n; c++)
        h = (int32_t)((uint32_t)h * 31u + (uint32_t)col_hash(k.col[c], i));
    return h;
}

__device__ static inline bool row_has_null_key(const KeyViews &k, int64_t i) {
    for (int c = 0; c < k.n; c++)
        if (col_is_null(k.col[c], i)) return true;
    return false;
}

__device__ static inline bool rows_key_equal(const KeyViews &a, int64_t i,
                                             const KeyViews &b, int64_t j) {
    for (int c = 0; c < a.n; c++)
        if (!col_eq(a.col[c], i, b.col[c], j)) return false;
    return true;
}

/* ---- residual (non-equi) join condition -------------------------------
 * AbstractJoinExec.checkJoinCondition:227-250: evaluated per matched
 * candidate; a failing candidate is skipped and does not set `matched`.
 * Host code maps the condition-row column indices (leftSide cols then
 * rightSide cols) onto probe/build sides before launch. */
#define GX_MAX_CONDS 4

struct JoinCondDev {
    int32_t cmp;
    int32_t a_is_build;
    int32_t b_is_const;
    int32_t b_is_build;
    int32_t const_is_null;
    DevColView a, b;
    int64_t v_i64;
    double v_f64;
    const uint8_t *v_bytes;
    int32_t v_len;
};

__device__ static inline bool cond_cmp_i64d(int64_t a, int32_t cmp, int64_t b) {
    switch (cmp) {
    case GX_CMP_LT: return a < b;
    case GX_CMP_LE: return a <= b;
    case GX_CMP_GT: return a > b;
    case GX_CMP_GE: return a >= b;
    case GX_CMP_NE: case GX_CMP_NE_NULLSAFE: return a != b;
    default: return a == b; /* EQ / EQ_NULLSAFE */
    }
}

__device__ static inline bool cond_cmp_f64d(double a, int32_t cmp, double b) {
    switch (cmp) {
    case GX_CMP_LT: return a < b;
    case GX_CMP_LE: return a <= b;
    case GX_CMP_GT: return a > b;
    case GX_CMP_GE: return a >= b;
    case GX_CMP_NE: case GX_CMP_NE_NULLSAFE: return a != b;
    default: return a == b;
    }
}

__device__ static bool join_cond_term(const JoinCondDev &c, int64_t pi,
                                      uint32_t bpos) {
    int64_t ia = c.a_is_build ? (int64_t)bpos : pi;
    int64_t ib = c.b_is_const ? 0 : (c.b_is_build ? (int64_t)bpos : pi);
    bool na = col_is_null(c.a, ia);
    bool nb = c.b_is_const ? (bool)c.const_is_null : col_is_null(c.b, ib);
    bool nullsafe = c.cmp == GX_CMP_EQ_NULLSAFE || c.cmp == GX_CMP_NE_NULLSAFE;
    if (na || nb) {
        if (!nullsafe) return false; /* SQL: NULL fails every comparison */
        bool eq = na && nb;          /* Objects.equals: NULL == NULL only */
        return c.cmp == GX_CMP_EQ_NULLSAFE ? eq : !eq;
    }
    switch (c.a.type) {
    case GX_SLICE: {
        int32_t ab = slice_begin(c.a, ia), ae = c.a.offsets[ia];
        const uint8_t *bp;
        int32_t blen;
        if (c.b_is_const) {
            bp = c.v_bytes;
            blen = c.v_len;
        } else {
            int32_t bb = slice_begin(c.b, ib);
            bp = c.b.bytes + bb;
            blen = c.b.offsets[ib] - bb;
        }
        bool eq = (ae - ab) == blen;
        for (int32_t k = 0; eq && k < blen; k++) eq = c.a.bytes[ab + k] == bp[k];
        return (c.cmp == GX_CMP_NE || c.cmp == GX_CMP_NE_NULLSAFE) ? !eq : eq;
    }
    case GX_F64: {
        double av = ((const double *)c.a.values)[ia];
        double bv = c.b_is_const ? c.v_f64 : ((const double *)c.b.values)[ib];
        return cond_cmp_f64d(av, c.cmp, bv);
    }
    default: {
        int64_t av = c.a.type == GX_I32
            ? (int64_t)((const int32_t *)c.a.values)[ia]
            : ((const int64_t *)c.a.values)[ia];
        int64_t bv;
        if (c.b_is_const)
            bv = c.v_i64;
        else
            bv = c.b.type == GX_I32
                ? (int64_t)((const int32_t *)c.b.values)[ib]
                : ((const int64_t *)c.b.values)[ib];
        return cond_cmp_i64d(av, c.cmp, bv);
    }
    }
}

/* ======================= generic small kernels ========================= */

__global__ void k_hash_rows(KeyViews keys, int64_t n, int32_t *hashes,
                            uint8_t *keynull, int null_safe) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
        if (null_safe) {
            /* group-join: Chunk.equals matching -- null keys participate */
            keynull[i] = 0;
            hashes[i] = row_hash(keys, i);
            continue;
        }
        bool isnull = row_has_null_key(keys, i);
        keynull[i] = isnull;
        hashes[i] = isnull ? 0 : row_hash(keys, i);
    }
}

__global__ void k_any_null(DevColView c, int64_t n, uint32_t *flag) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x)
        if (col_is_null(c, i)) { *flag = 1; return; }
}

/* ========================== join: build =============================== */

/* Join table entry, one 16-B aligned load per candidate:
 * fast path (single I64-typed key): key inline; generic: key compared via
 * build key columns at pos.
 *
 * Two layouts share this struct:
 *
 * INLINE-BUCKET (plain joins, the k_probe path): the table is an array of
 * 64-B buckets = 4 JoinEntry slots; bucket metadata lives in the slots'
 * otherwise-unused pad words (e[0].pad = entry count, e[1].pad = start of
 * this bucket's overflow run in the separate overflow array). With
 * n_buckets = pow2(n) (load <= 1/bucket-line) a probe resolves count,
 * keys and positions from ONE random 64-B line — r01 PMC showed the CSR
 * layout (starts[] + entries[]) spending 2.7-3.2x the algorithmic bytes
 * on 64-B lines fetched per 16-B entry; this layout removes the whole
 * starts[] indirection. Overflow (count > 4, ~0.4% of buckets at load 1)
 * continues into entries[ovf_start ..]. e[0] is loaded unconditionally
 * (count rides in it); e[1..3] load lazily from the same line.
 *
 * CSR (null_safe_keys group-join builds only): starts[b]..starts[b+1]
 * index a contiguous entries[] run — kept because the group-join kernels
 * (gxhip_groupjoin.inc) enumerate whole-table entry lists linearly. */
struct __align__(16) JoinEntry {
    int64_t key;
    uint32_t pos;
    uint32_t pad;
};

__global__ void k_join_hist(const int32_t *hashes, const uint8_t *keynull,
                            int64_t n, uint32_t *counts, uint32_t mask) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
        if (keynull[i]) continue; /* NULL-key build rows never inserted
                                     (ExecUtils.buildOneChunk:933-941) */
        atomicAdd(&counts[(uint32_t)gx_mix(hashes[i]) & mask], 1u);
    }
}

__global__ void k_join_scatter(const int32_t *hashes, const uint8_t *keynull,
                               int64_t n, uint32_t *cursors, uint32_t mask,
                               JoinEntry *entries, DevColView key0,
                               int fast_i64) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
        if (keynull[i]) continue;
        uint32_t b = (uint32_t)gx_mix(hashes[i]) & mask;
        uint32_t at = atomicAdd(&cursors[b], 1u);
        JoinEntry e;
        e.key = fast_i64 ? ((const int64_t *)key0.values)[i] : (int64_t)hashes[i];
        e.pos = (uint32_t)i;
        e.pad = 0;
        entries[at] = e;
    }
}

/* ---- bloom pre-filter (fast path) --------------------------------------
 * The reference tests a FastIntBloomFilter before the bucket walk
 * (AbstractHashJoinExec.java:80-106, FastIntBloomFilter.java:30-61).
 * Here: a cache-line-BLOCKED bloom — one 64-B line per key, two bits
 * inside it — sized ~8 bits/build key so it stays Infinity-Cache-resident
 * while the inline-bucket table does not: a negative probe then costs an
 * L3 hit instead of a random HBM line. Bit placement is internal (the
 * filter only gates the bucket read; false positives fall through), so
 * it uses a 64-bit mix of the key rather than the Java rotate chain. */
__device__ static inline uint64_t bloom_mix(int64_t key) {
    uint64_t x = (uint64_t)key * 0x9E3779B97F4A7C15ull;
    x ^= x >> 29; x *= 0xBF58476D1CE4E5B9ull; x ^= x >> 32;
    return x;
}

/* line_mask = (n_lines - 1); lines of 16 u32 words (64 B) */
__global__ void k_bloom_set(DevColView key0, int64_t n, uint32_t *bloom,
                            uint32_t line_mask) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
        if (col_is_null(key0, i)) continue;
        uint64_t x = bloom_mix(((const int64_t *)key0.values)[i]);
        uint32_t *line = bloom + ((uint32_t)x & line_mask) * 16;
        uint32_t b1 = (uint32_t)(x >> 32) & 511u;
        uint32_t b2 = (uint32_t)(x >> 41) & 511u;
        atomicOr(&line[b1 >> 5], 1u << (b1 & 31));
        atomicOr(&line[b2 >> 5], 1u << (b2 & 31));
    }
}

__device__ static inline bool bloom_test(const uint32_t *bloom,
                                         uint32_t line_mask, int64_t key) {
    uint64_t x = bloom_mix(key);
    const uint32_t *line = bloom + ((uint32_t)x & line_mask) * 16;
    uint32_t b1 = (uint32_t)(x >> 32) & 511u;
    uint32_t b2 = (uint32_t)(x >> 41) & 511u;
    if (!(line[b1 >> 5] & (1u << (b1 & 31)))) return false;
    return (line[b2 >> 5] & (1u << (b2 & 31))) != 0;
}

/* ---- inline-bucket build (plain joins) -------------------------------- */

/* overflow run length per bucket (entries beyond the 4 inline slots) */
__global__ void k_ovf_counts(const uint32_t *counts, int64_t n_buckets,
                             uint32_t *ovf) {
    for (int64_t b = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         b < n_buckets; b += (int64_t)gridDim.x * blockDim.x)
        ovf[b] = counts[b] > 4 ? counts[b] - 4 : 0;
}

/* stamp bucket metadata into the slot pads: e[0].pad = count,
 * e[1].pad = overflow start (both land in the line's first 32-B sector).
 * Runs BEFORE the scatter, which writes only key+pos of each slot. */
__global__ void k_bucket_meta(const uint32_t *counts, const uint32_t *ovf_starts,
                              int64_t n_buckets, JoinEntry *table) {
    for (int64_t b = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         b < n_buckets; b += (int64_t)gridDim.x * blockDim.x) {
        table[b * 4 + 0].pad = counts[b];
        table[b * 4 + 1].pad = ovf_starts[b];
    }
}

__global__ void k_join_scatter_ib(const int32_t *hashes, const uint8_t *keynull,
                                  int64_t n, uint32_t *cursors,
                                  const uint32_t *ovf_starts, uint32_t mask,
                                  JoinEntry *table, JoinEntry *ovf,
                                  DevColView key0, int fast_i64) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
        if (keynull[i]) continue;
        uint32_t b = (uint32_t)gx_mix(hashes[i]) & mask;
        uint32_t rank = atomicAdd(&cursors[b], 1u);
        int64_t key = fast_i64 ? ((const int64_t *)key0.values)[i]
                               : (int64_t)hashes[i];
        if (rank < 4) {
            JoinEntry *t = &table[(uint64_t)b * 4 + rank];
            t->key = key;       /* pad untouched: carries bucket metadata */
            t->pos = (uint32_t)i;
        } else {
            JoinEntry e;
            e.key = key;
            e.pos = (uint32_t)i;
            e.pad = 0;
            ovf[ovf_starts[b] + (rank - 4)] = e;
        }
    }
}

/* ---- bucket-clustered staged probe -------------------------------------
 * For LARGE builds the inline-bucket table far exceeds the caches, so
 * every probe costs one random HBM line and the kernel runs at ~half the
 * random-line ceiling (r2: 11.0 ms / 324M probes). Staging the probe rows
 * by BUCKET RANGE first makes each range's table slice L2-RESIDENT while
 * its rows probe: partition p owns buckets [p<<shift, (p+1)<<shift) — a
 * ~2 MB slice of table lines — and the staged probe kernel processes one
 * partition per workgroup, so a slice's ~2.5 probes/bucket hit the same
 * XCD's L2 after the first fetch. The staged 16-B row carries everything
 * the fast-path probe needs (rowid, hash, inline key). NULL-key rows
 * (bit31 of rowid) spread round-robin across partitions — they match
 * nothing and only emit LEFT/ANTI rows. Fast-path only: the hash is
 * computed inline from the key column (no k_hash_rows pass). */
struct __align__(16) RadixRow {
    uint32_t rowid;   /* bit31 = keynull */
    int32_t hash;
    int64_t key;
};

#define GX_RADIX_MAXP 4096
#define GX_RADIX_TILE 16384

__global__ void k_radix_count(DevColView key0, int64_t n, uint32_t mask,
                              int shift, int n_parts, uint32_t *counts) {
    __shared__ uint32_t s_cnt[GX_RADIX_MAXP];
    for (int p = threadIdx.x; p < n_parts; p += blockDim.x) s_cnt[p] = 0;
    __syncthreads();
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
        uint32_t p;
        if (col_is_null(key0, i)) {
            p = (uint32_t)i & (uint32_t)(n_parts - 1);
        } else {
            int32_t h = gx_hash_i64(((const int64_t *)key0.values)[i]);
            p = ((uint32_t)gx_mix(h) & mask) >> shift;
        }
        atomicAdd(&s_cnt[p], 1u);
    }
    __syncthreads();
    for (int p = threadIdx.x; p < n_parts; p += blockDim.x)
        if (s_cnt[p]) atomicAdd(&counts[p], s_cnt[p]);
}

/* Block-hierarchical scatter: per-tile LDS histogram, ONE global atomicAdd
 * per (block, partition) to reserve a range, then LDS-cursor placement —
 * a flat per-row atomicAdd on the cursor words serialized the whole grid
 * (324M returning atomics over 4 words cost ~3.5 s). */
__global__ void k_radix_scatter(DevColView key0, int64_t n,
                                uint32_t mask, int shift, int n_parts,
                                uint32_t *cursors, RadixRow *out) {
    __shared__ uint32_t s_cnt[GX_RADIX_MAXP];
    __shared__ uint32_t s_base[GX_RADIX_MAXP];
    const int64_t n_tiles = (n + GX_RADIX_TILE - 1) / GX_RADIX_TILE;
    for (int64_t tile = blockIdx.x; tile < n_tiles; tile += gridDim.x) {
        const int64_t t0 = tile * GX_RADIX_TILE;
        const int64_t t1 = t0 + GX_RADIX_TILE < n ? t0 + GX_RADIX_TILE : n;
        for (int p = threadIdx.x; p < n_parts; p += blockDim.x) s_cnt[p] = 0;
        __syncthreads();
        for (int64_t i = t0 + threadIdx.x; i < t1; i += blockDim.x) {
            uint32_t p;
            if (col_is_null(key0, i)) {
                p = (uint32_t)i & (uint32_t)(n_parts - 1);
            } else {
                int32_t h = gx_hash_i64(((const int64_t *)key0.values)[i]);
                p = ((uint32_t)gx_mix(h) & mask) >> shift;
            }
            atomicAdd(&s_cnt[p], 1u);
        }
        __syncthreads();
        for (int p = threadIdx.x; p < n_parts; p += blockDim.x) {
            s_base[p] = s_cnt[p] ? atomicAdd(&cursors[p], s_cnt[p]) : 0u;
            s_cnt[p] = 0;
        }
        __syncthreads();
        for (int64_t i = t0 + threadIdx.x; i < t1; i += blockDim.x) {
            bool kn = col_is_null(key0, i);
            int32_t h = 0;
            uint32_t p;
            if (kn) {
                p = (uint32_t)i & (uint32_t)(n_parts - 1);
            } else {
                h = gx_hash_i64(((const int64_t *)key0.values)[i]);
                p = ((uint32_t)gx_mix(h) & mask) >> shift;
            }
            uint32_t at = s_base[p] + atomicAdd(&s_cnt[p], 1u);
            RadixRow r;
            r.rowid = (uint32_t)i | (kn ? 0x80000000u : 0u);
            r.hash = h;
            r.key = kn ? 0 : ((const int64_t *)key0.values)[i];
            out[at] = r;
        }
        __syncthreads();
    }
}

/* ========================== join: probe =============================== */

/* Output pair: probe row index + matched build position (0xFFFFFFFF for the
 * null-extended side of LEFT/RIGHT unmatched rows; SEMI/ANTI emit only
 * probe_idx). */
struct ProbeParams {
    const JoinEntry *table;     /* inline buckets: 4 slots x n_buckets */
    const JoinEntry *entries;   /* overflow runs (count > 4) */
    uint32_t mask;
    int64_t n_probe;
    const int32_t *hashes;      /* probe row hashes (generic path only) */
    const uint8_t *keynull;     /* probe row has a NULL key (generic path) */
    const RadixRow *staged;     /* radix-staged rows (fast path); else NULL */
    const uint32_t *part_starts;/* staged: n_parts+1 partition row offsets */
    int32_t n_parts;
    int fast_i64;               /* entries carry the key inline */
    KeyViews build_keys;        /* for generic compare */
    KeyViews probe_keys;
    int join_type;              /* gx_join_type */
    int semi_join;              /* SEMI/ANTI (not single) */
    int outer_join;             /* LEFT/RIGHT */
    int single_join;
    int build_outer;
    int anti_null_col;
    DevColView anti_col;        /* probe col for NOT-IN null suppression */
    uint32_t *out_probe;        /* pair list */
    uint32_t *out_build;
    uint32_t cap;
    uint32_t *counter;          /* total pairs (may exceed cap) */
    uint32_t *err;              /* 1 = single-join >1 match */
    uint32_t *build_matched;    /* bitmap (u32 words), build_outer only */
    int32_t n_conds;            /* residual condition terms (AND) */
    JoinCondDev conds[GX_MAX_CONDS];
    const uint32_t *bloom;      /* blocked bloom pre-filter; NULL = off */
    uint32_t bloom_mask;
};

__device__ static inline bool join_conds_pass(const ProbeParams &P,
                                              int64_t pi, uint32_t bpos) {
    for (int t = 0; t < P.n_conds; t++)
        if (!join_cond_term(P.conds[t], pi, bpos)) return false;
    return true;
}

/* Per-wave LDS pair staging: batch STAGE pairs per global atomicAdd.
 * A bare per-iteration atomic on one counter serializes the whole grid
 * (~88 adds/us on one word — MI355X_MICROARCH.md row "dequeue"); staging
 * cuts the atomic count by ~STAGE/2x. Wave-synchronous: no __syncthreads,
 * counts are wave-uniform via ballot. */
#define GX_EMIT_STAGE 512

struct EmitStage {
    uint32_t *s_p;   /* this wave's LDS slice */
    uint32_t *s_b;
    uint32_t cnt;    /* wave-uniform */
    int lane;
};

__device__ static inline void emit_flush(const ProbeParams &P, EmitStage &E) {
    if (E.cnt == 0) return;
    uint32_t base = 0;
    if (E.lane == 0) base = atomicAdd(P.counter, E.cnt);
    base = (uint32_t)__shfl((int)base, 0, 64);
    for (uint32_t i = (uint32_t)E.lane; i < E.cnt; i += 64) {
        uint32_t at = base + i;
        if (at < P.cap) {
            P.out_probe[at] = E.s_p[i];
            P.out_build[at] = E.s_b[i];
        }
    }
    E.cnt = 0;
}

__device__ static inline void emit_pair(const ProbeParams &P, EmitStage &E,
                                        bool want, uint32_t pidx, uint32_t bpos) {
    unsigned long long m = __ballot(want);
    if (!m) return;
    uint32_t n = (uint32_t)__popcll(m);
    if (E.cnt + n > GX_EMIT_STAGE) emit_flush(P, E);
    if (want) {
        uint32_t at = E.cnt + (uint32_t)__popcll(m & ((1ull << E.lane) - 1ull));
        E.s_p[at] = pidx;
        E.s_b[at] = bpos;
    }
    E.cnt += n;
}

/* Process one probe row (or staged row at index bi), wave-synchronously:
 * the whole wave calls this together so ballot-compaction in emit_pair
 * sees every lane. One random 64-B line resolves the whole bucket: e0
 * carries the entry count in its pad; e1 (lazily loaded, same line)
 * carries the overflow start. In the fast path the bucket hash is
 * computed inline from the key column — no k_hash_rows pass, no
 * hashes/keynull loads.
 *
 * (Negative result, r2 A/B: per-thread 4-row batching of the e0 load
 * chains — extra MLP — ran join2 at 13.8 ms vs 11.0 ms unbatched even
 * with the slot arrays in registers; the ~8 resident waves/SIMD already
 * overlap the walk. The lever that pays is bucket-clustered staging.) */
__device__ static inline void probe_one(const ProbeParams &P, EmitStage &E,
                                        bool active, int64_t bi) {
    uint32_t i = 0;
    bool matched = false;
    uint32_t it = 0, end = 0;
    int64_t want_key = 0;
    JoinEntry e0 = {0, 0, 0};
    uint64_t ebase = 0;
    uint32_t ovf0 = 0;
    if (P.staged) {
        RadixRow r = active ? P.staged[bi] : RadixRow{0x80000000u, 0, 0};
        i = r.rowid & 0x7FFFFFFFu;
        if (active && !(r.rowid & 0x80000000u)) {
            want_key = r.key;
            if (!P.bloom || bloom_test(P.bloom, P.bloom_mask, want_key)) {
                uint32_t b = (uint32_t)gx_mix(r.hash) & P.mask;
                ebase = (uint64_t)b * 4;
                e0 = P.table[ebase];
                end = e0.pad;
            }
        }
    } else if (P.fast_i64) {
        i = active ? (uint32_t)bi : 0;
        const DevColView &kc = P.probe_keys.col[0];
        if (active && !col_is_null(kc, i)) {
            want_key = ((const int64_t *)kc.values)[i];
            if (!P.bloom || bloom_test(P.bloom, P.bloom_mask, want_key)) {
                uint32_t b = (uint32_t)gx_mix(gx_hash_i64(want_key)) & P.mask;
                ebase = (uint64_t)b * 4;
                e0 = P.table[ebase];
                end = e0.pad;
            }
        }
    } else {
        i = active ? (uint32_t)bi : 0;
        if (active && !P.keynull[i]) {
            uint32_t b = (uint32_t)gx_mix(P.hashes[i]) & P.mask;
            ebase = (uint64_t)b * 4;
            e0 = P.table[ebase];
            end = e0.pad;
        }
    }

    /* walk candidates; lanes iterate together so emissions batch */
    while (__ballot(it < end)) {
        bool have = active && it < end;
        bool is_match = false;
        uint32_t bpos = 0;
        if (have) {
            JoinEntry e;
            if (it == 0) {
                e = e0;
            } else if (it < 4) {
                e = P.table[ebase + it];
                if (it == 1) ovf0 = e.pad;
            } else {
                e = P.entries[ovf0 + (it - 4)];
            }
            it++;
            if (P.fast_i64)
                is_match = (e.key == want_key);
            else
                is_match = ((int32_t)e.key == P.hashes[i]) &&
                           rows_key_equal(P.build_keys, e.pos,
                                          P.probe_keys, i);
            /* residual condition: a failing candidate is skipped and
             * never counts as a match (checkJoinCondition:206-208) */
            if (is_match && P.n_conds)
                is_match = join_conds_pass(P, i, e.pos);
            bpos = e.pos;
        }
        if (is_match) {
            if (P.single_join && matched) atomicExch(P.err, 1u);
            if (P.build_matched)
                atomicOr(&P.build_matched[bpos >> 5], 1u << (bpos & 31));
            if (P.semi_join) { it = end; } /* first match is enough */
            matched = true;
        }
        bool emit_now = is_match && !P.semi_join;
        emit_pair(P, E, emit_now, i, bpos);
    }

    /* post-row emissions (LEFT/RIGHT null rows, SEMI/ANTI rows) */
    bool want_null_row = active && P.outer_join && !P.build_outer && !matched;
    emit_pair(P, E, want_null_row, i, 0xFFFFFFFFu);
    if (P.semi_join) {
        bool want;
        if (P.join_type == GX_JOIN_SEMI) want = active && matched;
        else { /* ANTI */
            want = active && !matched;
            if (want && P.anti_null_col >= 0)
                want = !col_is_null(P.anti_col, i);
        }
        emit_pair(P, E, want, i, 0xFFFFFFFFu);
    }
}

__global__ void k_probe(ProbeParams P) {
    __shared__ uint32_t s_pairs[2][4][GX_EMIT_STAGE]; /* [p/b][wave][slot] */
    EmitStage E;
    {
        int wid = threadIdx.x >> 6;
        E.s_p = s_pairs[0][wid];
        E.s_b = s_pairs[1][wid];
        E.cnt = 0;
        E.lane = threadIdx.x & 63;
    }
    if (P.staged && P.part_starts) {
        /* bucket-clustered mode: one partition (a ~2 MB table slice) per
         * workgroup iteration — the slice's repeat probes hit this XCD's
         * L2 instead of refetching HBM lines */
        for (int64_t p = blockIdx.x; p < P.n_parts; p += gridDim.x) {
            const int64_t r0 = P.part_starts[p];
            const int64_t r1 = P.part_starts[p + 1];
            for (int64_t base = r0 + threadIdx.x;; base += blockDim.x) {
                if (!__ballot(base < r1)) break;
                probe_one(P, E, base < r1, base);
            }
        }
    } else {
        const int64_t stride = (int64_t)gridDim.x * blockDim.x;
        for (int64_t base = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;;
             base += stride) {
            if (!__ballot(base < P.n_probe)) break;
            probe_one(P, E, base < P.n_probe, base);
        }
    }
    emit_flush(P, E);
}

/* pass-through (ANTI over empty build): emit every probe row */
__global__ void k_iota(uint32_t *out, int64_t n) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x)
        out[i] = (uint32_t)i;
}

/* buildOuter tail: collect unmatched build positions */
__global__ void k_unmatched(const uint32_t *bitmap, int64_t n_build,
                            uint32_t *out, uint32_t cap, uint32_t *counter) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t base_i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;;
         base_i += stride) {
        bool active = base_i < n_build;
        if (!__ballot(active)) break;
        bool want = active && !((bitmap[base_i >> 5] >> (base_i & 31)) & 1u);
        unsigned long long m = __ballot(want);
        if (m) {
            int lane = threadIdx.x & 63;
            int leader = __ffsll(m) - 1;
            uint32_t base = 0;
            if (lane == leader) base = atomicAdd(counter, (uint32_t)__popcll(m));
            base = (uint32_t)__shfl((int)base, leader, 64);
            if (want) {
                uint32_t at = base + (uint32_t)__popcll(m & ((1ull << lane) - 1ull));
                if (at < cap) out[at] = (uint32_t)base_i;
            }
        }
    }
}

/* ===================== gather (output materialization) ================= */

/* fused multi-column gather: read the pair indices ONCE per output row and
 * materialize every output column (coalesced writes; the random reads are
 * the algorithmic payload-gather cost). side: 0 = probe idx, 1 = build idx,
 * 2 = null fill. */
struct GatherParams {
    const uint32_t *pidx;
    const uint32_t *bidx;
    int64_t n;
    int32_t n_cols;
    int32_t side[GX_MAX_COLS];
    DevColView src[GX_MAX_COLS];
    void *out_vals[GX_MAX_COLS];
    uint8_t *out_nulls[GX_MAX_COLS];
};

__global__ void k_gather_multi(GatherParams G) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < G.n;
         i += (int64_t)gridDim.x * blockDim.x) {
        uint32_t pi = G.pidx ? G.pidx[i] : 0xFFFFFFFFu;
        uint32_t bi = G.bidx ? G.bidx[i] : 0xFFFFFFFFu;
        for (int c = 0; c < G.n_cols; c++) {
            uint32_t s = G.side[c] == 0 ? pi : (G.side[c] == 1 ? bi : 0xFFFFFFFFu);
            const DevColView &src = G.src[c];
            if (s == 0xFFFFFFFFu) {
                G.out_nulls[c][i] = 1;
                switch (src.type) {
                case GX_I64: ((int64_t *)G.out_vals[c])[i] = 0; break;
                case GX_I32: ((int32_t *)G.out_vals[c])[i] = 0; break;
                case GX_F64: ((double *)G.out_vals[c])[i] = 0; break;
                case GX_DECIMAL:
                    for (int k = 0; k < 5; k++)
                        ((uint64_t *)G.out_vals[c])[i * 5 + k] = 0;
                    break;
                }
                continue;
            }
            bool nn = col_is_null(src, s);
            G.out_nulls[c][i] = nn;
            switch (src.type) {
            case GX_I64: ((int64_t *)G.out_vals[c])[i] = nn ? 0 : ((const int64_t *)src.values)[s]; break;
            case GX_I32: ((int32_t *)G.out_vals[c])[i] = nn ? 0 : ((const int32_t *)src.values)[s]; break;
            case GX_F64: ((double *)G.out_vals[c])[i] = nn ? 0 : ((const double *)src.values)[s]; break;
            case GX_DECIMAL:
                for (int k = 0; k < 5; k++)
                    ((uint64_t *)G.out_vals[c])[i * 5 + k] = nn ? 0
                        : ((const uint64_t *)src.values)[(int64_t)s * 5 + k];
                break;
            }
        }
    }
}

/* out[i] = src[idx[i]]; idx 0xFFFFFFFF -> NULL. Coalesced writes, the random
 * reads are the algorithmic cost of the join payload gather. */
__global__ void k_gather(DevColView src, const uint32_t *idx, int64_t n,
                         void *out_vals, uint8_t *out_nulls, int fill_null_only) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
        uint32_t s = fill_null_only ? 0xFFFFFFFFu : idx[i];
        if (s == 0xFFFFFFFFu) {
            out_nulls[i] = 1;
            switch (src.type) {
            case GX_I64: ((int64_t *)out_vals)[i] = 0; break;
            case GX_I32: ((int32_t *)out_vals)[i] = 0; break;
            case GX_F64: ((double *)out_vals)[i] = 0; break;
            case GX_DECIMAL:
                for (int k = 0; k < 5; k++)
                    ((uint64_t *)out_vals)[i * 5 + k] = 0;
                break;
            }
            continue;
        }
        bool nn = col_is_null(src, s);
        out_nulls[i] = nn;
        switch (src.type) {
        case GX_I64: ((int64_t *)out_vals)[i] = nn ? 0 : ((const int64_t *)src.values)[s]; break;
        case GX_I32: ((int32_t *)out_vals)[i] = nn ? 0 : ((const int32_t *)src.values)[s]; break;
        case GX_F64: ((double *)out_vals)[i] = nn ? 0 : ((const double *)src.values)[s]; break;
        case GX_DECIMAL:
            for (int k = 0; k < 5; k++)
                ((uint64_t *)out_vals)[i * 5 + k] = nn ? 0
                    : ((const uint64_t *)src.values)[(int64_t)s * 5 + k];
            break;
        }
    }
}

/* ---- SLICE (varlen) gather: lens -> inclusive scan -> byte copy ------- */

__global__ void k_slice_lens(DevColView src, const uint32_t *idx,
                             int fill_null_only, int64_t n, int32_t *lens,
                             uint8_t *out_nulls) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
        uint32_t s = fill_null_only ? 0xFFFFFFFFu : (idx ? idx[i] : (uint32_t)i);
        if (s == 0xFFFFFFFFu || col_is_null(src, s)) {
            out_nulls[i] = 1;
            lens[i] = 0;
        } else {
            out_nulls[i] = 0;
            lens[i] = src.offsets[s] - slice_begin(src, s);
        }
    }
}

__global__ void k_slice_copy(DevColView src, const uint32_t *idx,
                             int fill_null_only, int64_t n,
                             const int32_t *out_off, uint8_t *out_bytes) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
        uint32_t s = fill_null_only ? 0xFFFFFFFFu : (idx ? idx[i] : (uint32_t)i);
        if (s == 0xFFFFFFFFu || col_is_null(src, s)) continue;
        int32_t sb = slice_begin(src, s), se = src.offsets[s];
        int32_t db = i > 0 ? out_off[i - 1] : 0;
        for (int32_t k = 0; k < se - sb; k++)
            out_bytes[db + k] = src.bytes[sb + k];
    }
}

/* ======================= result holder ================================= */

struct HipResult {
    std::vector<DevBuf> bufs;               /* device-resident payloads */
    std::vector<std::vector<uint8_t>> host; /* after to_host */
    std::vector<gx_block> blocks;
    gx_result res;
    int device = 0;
};

static void free_result(HipResult *h) {
    for (auto &b : h->bufs) b.release();
    delete h;
}

/* grid size for memory-bound grid-stride kernels (Guideline 11) */
static inline int gx_grid(int64_t n, int block = 256) {
    int64_t g = (n + block - 1) / block;
    if (g > 4096) g = 4096;
    if (g < 1) g = 1;
    return (int)g;
}

/* ========================= operator base =============================== */

enum OpKind { OP_JOIN = 1, OP_AGG = 2, OP_PART = 3, OP_SCAN = 4,
              OP_GROUPJOIN = 5, OP_WINDOW = 6, OP_FWINDOW = 7 };

struct gx_op {
    int kind;
    int variant = 0;  /* OP_JOIN: 0 = in-HBM JoinOp, 1 = HybridJoinOp */
    int device;
    hipStream_t stream;
    /* Host-side call serialization: the reference
     * calls consumeChunk from multiple driver threads against the shared
     * Synchronizer, serialized by `synchronized(shared)`
     * (ParallelHashJoinExec.java:158). Here the gx_op IS the shared state
     * (table + per-op scratch buffers), so the C-ABI entry points lock
     * this mutex — concurrent consume/probe calls from N driver threads
     * are safe; device work stays async on the op's stream. */
    std::mutex mu;
    virtual ~gx_op() = default;
protected:
    gx_op(int k, int dev, uint64_t s)
        : kind(k), device(dev), stream((hipStream_t)s) {}
};

namespace {

int ensure_device(int device) {
    if (device < 0) { gx_set_err("gxhip requires device >= 0 (no CPU path)"); return -1; }
    HIP_OK(hipSetDevice(device));
    return 0;
}

/* Build a gx_result from exact-size device buffers. Each column owns a
 * values buf + nulls buf; SLICE columns use slot 0 for END-OFFSETS and a
 * data buf appended later by attach_slice_data (lengths known post-scan). */
HipResult *alloc_result_cols(const std::vector<int32_t> &types, int64_t n,
                             int device, hipStream_t stream) {
    auto *h = new HipResult();
    h->device = device;
    h->blocks.resize(types.size());
    h->bufs.resize(types.size() * 2);
    for (size_t c = 0; c < types.size(); c++) {
        size_t es = types[c] == GX_SLICE ? 4 : gx_fixed_size(types[c]);
        if (n > 0) {
            if (h->bufs[c * 2].grow((size_t)n * es, stream) ||
                h->bufs[c * 2 + 1].grow((size_t)n, stream)) {
                free_result(h);
                return nullptr;
            }
        }
        gx_block &b = h->blocks[c];
        std::memset(&b, 0, sizeof(b));
        b.type = types[c];
        b.mem = GX_MEM_DEVICE;
        if (types[c] == GX_SLICE)
            b.offsets = (const int32_t *)h->bufs[c * 2].p;
        else
            b.values = h->bufs[c * 2].p;
        b.nulls = (const uint8_t *)h->bufs[c * 2 + 1].p;
    }
    h->res.chunk.n_rows = (int32_t)n;
    h->res.chunk.n_blocks = (int32_t)types.size();
    h->res.chunk.blocks = h->blocks.data();
    h->res.opaque = h;
    return h;
}

/* allocate and register the byte payload for a SLICE result column */
int attach_slice_data(HipResult *h, size_t col, int64_t data_len,
                      hipStream_t stream) {
    h->bufs.emplace_back();
    if (data_len > 0 && h->bufs.back().grow((size_t)data_len, stream)) return -1;
    h->blocks[col].data = (const uint8_t *)h->bufs.back().p;
    h->blocks[col].data_len = data_len;
    return 0;
}

/* gather one SLICE column into result column `col`: lens + nulls, scan to
 * end-offsets (in place), attach data, copy bytes. idx==nullptr -> identity;
 * fill_null_only -> all-null column. */
int gather_slice_col(const DevColView &src, const uint32_t *idx,
                     int fill_null_only, int64_t n, HipResult *h, size_t col,
                     DevBuf &scan_tmp, hipStream_t stream) {
    if (n == 0) return attach_slice_data(h, col, 0, stream);
    int32_t *off = (int32_t *)h->bufs[col * 2].p;
    uint8_t *nulls = (uint8_t *)h->bufs[col * 2 + 1].p;
    hipLaunchKernelGGL(k_slice_lens, dim3(gx_grid(n)), dim3(256), 0, stream,
                       src, idx, fill_null_only, n, off, nulls);
    size_t tmp_bytes = 0;
    HIP_OK(hipcub::DeviceScan::InclusiveSum(nullptr, tmp_bytes, off, off, n,
                                            stream));
    if (scan_tmp.grow(tmp_bytes, stream)) return -1;
    HIP_OK(hipcub::DeviceScan::InclusiveSum(scan_tmp.p, tmp_bytes, off, off,
                                            n, stream));
    int32_t total = 0;
    HIP_OK(hipMemcpyAsync(&total, off + (n - 1), 4, hipMemcpyDeviceToHost,
                          stream));
    HIP_OK(hipStreamSynchronize(stream));
    if (attach_slice_data(h, col, total, stream)) return -1;
    if (total > 0)
        hipLaunchKernelGGL(k_slice_copy, dim3(gx_grid(n)), dim3(256), 0,
                           stream, src, idx, fill_null_only, n, off,
                           (uint8_t *)h->blocks[col].data);
    return 0;
}

/* stage an input chunk: if every block is device-resident we use the
 * caller's pointers directly (zero copy); otherwise copy host->device. */
struct StagedChunk {
    std::vector<DevBuf> owned;
    std::vector<DevColView> views;
    int64_t n_rows = 0;

    int stage(const gx_chunk *ch, hipStream_t stream) {
        n_rows = ch->n_rows;
        views.resize(ch->n_blocks);
        for (int32_t c = 0; c < ch->n_blocks; c++) {
            const gx_block *b = &ch->blocks[c];
            DevColView &v = views[c];
            v.type = b->type;
            v.offsets = nullptr; v.bytes = nullptr;
            if (b->type == GX_SLICE) {
                v.has_nulls = b->nulls != nullptr;
                if (b->mem == GX_MEM_DEVICE) {
                    v.offsets = b->offsets;
                    v.bytes = b->data;
                    v.nulls = (const uint8_t *)b->nulls;
                } else {
                    int64_t blen = n_rows > 0 ? b->offsets[n_rows - 1] : 0;
                    owned.emplace_back();
                    DevBuf &ob = owned.back();
                    if (ob.grow((size_t)std::max<int64_t>(n_rows, 1) * 4, stream)) return -1;
                    HIP_OK(hipMemcpyAsync(ob.p, b->offsets, (size_t)n_rows * 4,
                                          hipMemcpyHostToDevice, stream));
                    v.offsets = (const int32_t *)ob.p;
                    owned.emplace_back();
                    DevBuf &db = owned.back();
                    if (blen > 0) {
                        if (db.grow((size_t)blen, stream)) return -1;
                        HIP_OK(hipMemcpyAsync(db.p, b->data, (size_t)blen,
                                              hipMemcpyHostToDevice, stream));
                    }
                    v.bytes = (const uint8_t *)db.p;
                    if (b->nulls) {
                        owned.emplace_back();
                        DevBuf &nb = owned.back();
                        if (nb.grow((size_t)n_rows, stream)) return -1;
                        HIP_OK(hipMemcpyAsync(nb.p, b->nulls, (size_t)n_rows,
                                              hipMemcpyHostToDevice, stream));
                        v.nulls = (const uint8_t *)nb.p;
                    }
                }
                continue;
            }
            size_t es = gx_fixed_size(b->type);
            if (b->mem == GX_MEM_DEVICE) {
                v.values = b->values;
                v.nulls = (const uint8_t *)b->nulls;
                v.has_nulls = b->nulls != nullptr;
            } else {
                owned.emplace_back();
                DevBuf &vb = owned.back();
                if (vb.grow((size_t)n_rows * es, stream)) return -1;
                HIP_OK(hipMemcpyAsync(vb.p, b->values, (size_t)n_rows * es,
                                      hipMemcpyHostToDevice, stream));
                v.values = vb.p;
                v.has_nulls = b->nulls != nullptr;
                v.nulls = nullptr;
                if (b->nulls) {
                    owned.emplace_back();
                    DevBuf &nb = owned.back();
                    if (nb.grow((size_t)n_rows, stream)) return -1;
                    HIP_OK(hipMemcpyAsync(nb.p, b->nulls, (size_t)n_rows,
                                          hipMemcpyHostToDevice, stream));
                    v.nulls = (const uint8_t *)nb.p;
                }
            }
        }
        return 0;
    }
    void release() { for (auto &b : owned) b.release(); owned.clear(); }
};

/* ========================= JoinOp ====================================== */

struct JoinOp : gx_op {
    gx_join_cfg cfg;
    std::vector<gx_equi_key> keys;
    std::vector<gx_join_cond> conds;             /* residual condition */
    std::vector<std::vector<uint8_t>> cond_bytes; /* deep-copied SLICE consts */
    std::vector<DevBuf> d_cond_pats;
    std::vector<int32_t> out_proj;  /* projection pushdown; empty = full */
    std::vector<int32_t> outer_types, inner_types;
    std::vector<int> build_key_cols, probe_key_cols;

    DevStore build;          /* build-side columns in HBM */
    DevStore probe_buf;      /* buffered-probe staging (probe_push/flush) */
    bool probe_buf_init = false;
    DevBuf d_hashes, d_keynull;
    DevBuf d_table;          /* inline-bucket table (plain joins) */
    DevBuf d_counts, d_starts, d_entries; /* CSR (null_safe) / IB overflow */
    DevBuf d_bitmap;         /* build_outer matched bitmap */
    DevBuf d_scan_tmp;
    DevBuf d_pidx, d_bpos, d_meta, d_ph, d_pn; /* reused across probe calls */
    DevBuf d_staged, d_radix_cnt;               /* radix-staged probe rows */
    DevBuf d_bloom;            /* blocked bloom pre-filter (fast path) */
    uint32_t bloom_mask = 0;
    uint32_t mask = 0;
    int64_t n_buckets = 0;
    bool fast_i64 = false;
    bool built = false;
    bool pass_nothing = false, pass_through = false;
    bool tail_done = false;
    /* group-join hook: HashGroupJoinExec matching is null-safe
     * (Chunk.equals) — build inserts every row and equal NULL keys match */
    bool null_safe_keys = false;

    /* live probe-kernel stats for bench.py's roofline leg */
    hipEvent_t ev0 = nullptr, ev1 = nullptr;
    double probe_kernel_ms = 0.0;
    int64_t probe_launches = 0, probe_rows_total = 0, matches_total = 0;

    JoinOp(const gx_join_cfg *c) : gx_op(OP_JOIN, c->device, c->stream), cfg(*c) {
        keys.assign(c->keys, c->keys + c->n_keys);
        if (c->n_conds > 0) {
            conds.assign(c->conds, c->conds + c->n_conds);
            cond_bytes.resize(conds.size());
            for (size_t t = 0; t < conds.size(); t++) {
                if (conds[t].col_b < 0 && conds[t].v_bytes && conds[t].v_len > 0) {
                    cond_bytes[t].assign(conds[t].v_bytes,
                                         conds[t].v_bytes + conds[t].v_len);
                    conds[t].v_bytes = cond_bytes[t].data();
                }
            }
        }
        if (c->n_out_proj > 0)
            out_proj.assign(c->out_proj, c->out_proj + c->n_out_proj);
        outer_types.assign(c->outer_types, c->outer_types + c->n_outer_cols);
        inner_types.assign(c->inner_types, c->inner_types + c->n_inner_cols);
        for (auto &k : keys) {
            build_key_cols.push_back(cfg.build_outer ? k.outer_index : k.inner_index);
            probe_key_cols.push_back(cfg.build_outer ? k.inner_index : k.outer_index);
        }
        const auto &bt = cfg.build_outer ? outer_types : inner_types;
        build.init((int32_t)bt.size(), bt.data(), stream);
        if (cfg.expected_build_rows > 0) build.reserve(cfg.expected_build_rows);
    }
    ~JoinOp() override {
        build.release();
        probe_buf.release();
        d_table.release();
        d_hashes.release(); d_keynull.release(); d_counts.release();
        d_starts.release(); d_entries.release(); d_bitmap.release();
        d_scan_tmp.release();
        d_pidx.release(); d_bpos.release(); d_meta.release();
        d_ph.release(); d_pn.release();
        d_staged.release(); d_radix_cnt.release();
        d_bloom.release();
        for (auto &b : d_cond_pats) b.release();
        if (ev0) (void)hipEventDestroy(ev0);
        if (ev1) (void)hipEventDestroy(ev1);
    }

    /* condition-row column (JoinRelType.leftSide cols then rightSide cols)
     * -> (build-side?, source column) honoring build_outer */
    int cond_map(int32_t col, bool &is_build) const {
        const int n_outer = (int)outer_types.size();
        const int n_inner = (int)inner_types.size();
        const int n_left = cfg.join_type == GX_JOIN_RIGHT ? n_inner : n_outer;
        bool is_outer_col;
        int src;
        if (cfg.join_type == GX_JOIN_RIGHT) {
            is_outer_col = col >= n_left;
            src = is_outer_col ? col - n_left : col;
        } else {
            is_outer_col = col < n_left;
            src = is_outer_col ? col : col - n_left;
        }
        is_build = is_outer_col == (cfg.build_outer != 0);
        return src;
    }

    int upload_cond_pats() {
        if (conds.empty() || !d_cond_pats.empty()) return 0;
        d_cond_pats.resize(conds.size());
        for (size_t t = 0; t < conds.size(); t++) {
            const gx_join_cond &c = conds[t];
            if (c.col_b < 0 && c.v_bytes && c.v_len > 0) {
                if (d_cond_pats[t].grow((size_t)c.v_len, stream)) return -1;
                HIP_OK(hipMemcpyAsync(d_cond_pats[t].p, c.v_bytes,
                                      (size_t)c.v_len, hipMemcpyHostToDevice,
                                      stream));
            }
        }
        return 0;
    }

    KeyViews key_views(const DevStore &s, const std::vector<int> &colidx) const {
        KeyViews kv;
        kv.n = (int32_t)colidx.size();
        for (int i = 0; i < kv.n; i++) kv.col[i] = s.view(colidx[i]);
        return kv;
    }
    KeyViews key_views_staged(const StagedChunk &s, const std::vector<int> &colidx) const {
        KeyViews kv;
        kv.n = (int32_t)colidx.size();
        for (int i = 0; i < kv.n; i++) kv.col[i] = s.views[colidx[i]];
        return kv;
    }

    int consume(const gx_chunk *ch) {
        if (ensure_device(device)) return -1;
        return build.append(ch);
    }

    int do_build() {
        /* first-come barrier contract (Synchronizer.buildCount): the first
         * caller builds, later callers no-op */
        if (built) return 0;
        if (ensure_device(device)) return -1;
        const int64_t n = build.n_rows;
        fast_i64 = (keys.size() == 1 && keys[0].unified_type == GX_I64)
                   && !null_safe_keys;

        bool semi_join = (cfg.join_type == GX_JOIN_SEMI || cfg.join_type == GX_JOIN_ANTI)
                         && !cfg.single_join;
        if (n == 0 && cfg.join_type == GX_JOIN_INNER) pass_nothing = true;
        if (semi_join && n == 0) {
            if (cfg.join_type == GX_JOIN_SEMI) pass_nothing = true;
            else pass_through = true;
        }

        if (n == 0) {
            /* LEFT/RIGHT over an empty build still probes (emitting null
             * rows) — give the probe kernel a valid all-empty table. */
            n_buckets = 2;
            mask = 1;
            if (null_safe_keys) {
                if (d_starts.grow((size_t)(n_buckets + 1) * 4, stream)) return -1;
                HIP_OK(hipMemsetAsync(d_starts.p, 0, (size_t)(n_buckets + 1) * 4, stream));
            } else {
                if (d_table.grow((size_t)n_buckets * 4 * sizeof(JoinEntry),
                                 stream)) return -1;
                HIP_OK(hipMemsetAsync(d_table.p, 0,
                                      (size_t)n_buckets * 4 * sizeof(JoinEntry),
                                      stream));
            }
        }
        if (n > 0) {
            if (d_hashes.grow((size_t)n * 4, stream) ||
                d_keynull.grow((size_t)n, stream)) return -1;
            KeyViews bk = key_views(build, build_key_cols);
            hipLaunchKernelGGL(k_hash_rows, dim3(gx_grid(n)), dim3(256), 0, stream,
                               bk, n, (int32_t *)d_hashes.p, (uint8_t *)d_keynull.p,
                               (int)null_safe_keys);
            if (null_safe_keys) {
                /* CSR layout for the group-join kernels (linear entry
                 * enumeration); plain joins use the inline-bucket table. */
                n_buckets = gx_pow2(n * 2); /* avg load 0.5 */
                mask = (uint32_t)(n_buckets - 1);
                if (d_counts.grow((size_t)(n_buckets + 1) * 4, stream) ||
                    d_starts.grow((size_t)(n_buckets + 1) * 4, stream) ||
                    d_entries.grow((size_t)n * sizeof(JoinEntry), stream)) return -1;
                HIP_OK(hipMemsetAsync(d_counts.p, 0, (size_t)(n_buckets + 1) * 4, stream));
                hipLaunchKernelGGL(k_join_hist, dim3(gx_grid(n)), dim3(256), 0, stream,
                                   (const int32_t *)d_hashes.p, (const uint8_t *)d_keynull.p,
                                   n, (uint32_t *)d_counts.p, mask);
                /* exclusive scan counts[0..n_buckets] -> starts (incl. total) */
                size_t tmp_bytes = 0;
                HIP_OK(hipcub::DeviceScan::ExclusiveSum(nullptr, tmp_bytes,
                                                        (uint32_t *)d_counts.p,
                                                        (uint32_t *)d_starts.p,
                                                        n_buckets + 1, stream));
                if (d_scan_tmp.grow(tmp_bytes, stream)) return -1;
                HIP_OK(hipcub::DeviceScan::ExclusiveSum(d_scan_tmp.p, tmp_bytes,
                                                        (uint32_t *)d_counts.p,
                                                        (uint32_t *)d_starts.p,
                                                        n_buckets + 1, stream));
                /* reuse counts as cursors (= starts) */
                HIP_OK(hipMemcpyAsync(d_counts.p, d_starts.p, (size_t)n_buckets * 4,
                                      hipMemcpyDeviceToDevice, stream));
                hipLaunchKernelGGL(k_join_scatter, dim3(gx_grid(n)), dim3(256), 0, stream,
                                   (const int32_t *)d_hashes.p, (const uint8_t *)d_keynull.p,
                                   n, (uint32_t *)d_counts.p, mask,
                                   (JoinEntry *)d_entries.p,
                                   build.view(build_key_cols[0]), (int)fast_i64);
            } else {
                /* inline-bucket build: hist -> overflow-run scan -> bucket
                 * metadata stamp -> scatter (key+pos only; pads carry the
                 * metadata). Load cap 0.75: at load ~0.9 the lazy e1..e3
                 * slot loads frequently miss L2 (the wave's other random
                 * fetches churn ~a full XCD L2 per iteration) and C2's
                 * probe paid ~1.7 HBM lines/row (r2 PMC: 7.3 GB for a
                 * 60M-row probe) — one extra doubling buys ~1 line/row. */
                n_buckets = gx_pow2(n); /* 4 slots/bucket */
                if (n > (n_buckets * 3) / 4) n_buckets <<= 1;
                mask = (uint32_t)(n_buckets - 1);
                if (d_counts.grow((size_t)(n_buckets + 1) * 4, stream) ||
                    d_starts.grow((size_t)(n_buckets + 1) * 4, stream) ||
                    d_table.grow((size_t)n_buckets * 4 * sizeof(JoinEntry),
                                 stream)) return -1;
                HIP_OK(hipMemsetAsync(d_counts.p, 0, (size_t)(n_buckets + 1) * 4, stream));
                hipLaunchKernelGGL(k_join_hist, dim3(gx_grid(n)), dim3(256), 0, stream,
                                   (const int32_t *)d_hashes.p, (const uint8_t *)d_keynull.p,
                                   n, (uint32_t *)d_counts.p, mask);
                hipLaunchKernelGGL(k_ovf_counts, dim3(gx_grid(n_buckets)), dim3(256),
                                   0, stream, (const uint32_t *)d_counts.p,
                                   n_buckets, (uint32_t *)d_starts.p);
                size_t tmp_bytes = 0;
                HIP_OK(hipcub::DeviceScan::ExclusiveSum(nullptr, tmp_bytes,
                                                        (uint32_t *)d_starts.p,
                                                        (uint32_t *)d_starts.p,
                                                        n_buckets + 1, stream));
                if (d_scan_tmp.grow(tmp_bytes, stream)) return -1;
                HIP_OK(hipcub::DeviceScan::ExclusiveSum(d_scan_tmp.p, tmp_bytes,
                                                        (uint32_t *)d_starts.p,
                                                        (uint32_t *)d_starts.p,
                                                        n_buckets + 1, stream));
                uint32_t total_ovf = 0;
                HIP_OK(hipMemcpyAsync(&total_ovf,
                                      (uint32_t *)d_starts.p + n_buckets, 4,
                                      hipMemcpyDeviceToHost, stream));
                HIP_OK(hipStreamSynchronize(stream));
                if (d_entries.grow((size_t)std::max<uint32_t>(total_ovf, 1) *
                                       sizeof(JoinEntry), stream)) return -1;
                hipLaunchKernelGGL(k_bucket_meta, dim3(gx_grid(n_buckets)),
                                   dim3(256), 0, stream,
                                   (const uint32_t *)d_counts.p,
                                   (const uint32_t *)d_starts.p, n_buckets,
                                   (JoinEntry *)d_table.p);
                HIP_OK(hipMemsetAsync(d_counts.p, 0, (size_t)n_buckets * 4, stream));
                hipLaunchKernelGGL(k_join_scatter_ib, dim3(gx_grid(n)), dim3(256),
                                   0, stream,
                                   (const int32_t *)d_hashes.p,
                                   (const uint8_t *)d_keynull.p,
                                   n, (uint32_t *)d_counts.p,
                                   (const uint32_t *)d_starts.p, mask,
                                   (JoinEntry *)d_table.p,
                                   (JoinEntry *)d_entries.p,
                                   build.view(build_key_cols[0]), (int)fast_i64);
                /* bloom pre-filter: GX_BLOOM=1 forces on, =0 off; default
                 * on for builds whose table exceeds the Infinity Cache
                 * while the bloom (8 bits/key) still fits (the regime
                 * where a negative saves a random HBM line) */
                const char *be = getenv("GX_BLOOM");
                bool use_bloom = fast_i64 &&
                    (be ? be[0] == '1'
                        : (cfg.enable_bloom == 1 && n >= (1 << 16) &&
                           n <= INT64_C(200) << 20));
                if (use_bloom) {
                    int64_t n_lines = gx_pow2(std::max<int64_t>(n / 64, 1));
                    bloom_mask = (uint32_t)(n_lines - 1);
                    if (d_bloom.grow((size_t)n_lines * 64, stream)) return -1;
                    HIP_OK(hipMemsetAsync(d_bloom.p, 0, (size_t)n_lines * 64,
                                          stream));
                    hipLaunchKernelGGL(k_bloom_set, dim3(gx_grid(n)),
                                       dim3(256), 0, stream,
                                       build.view(build_key_cols[0]), n,
                                       (uint32_t *)d_bloom.p, bloom_mask);
                }
            }
        }

        /* ANTI NOT-IN: build contains NULL -> pass nothing
         * (doSpecialCheckForSemiJoin:305-312, single build column) */
        if (cfg.join_type == GX_JOIN_ANTI && cfg.anti_null_col >= 0 &&
            build.cols.size() == 1 && n > 0 && build.cols[0].has_nulls) {
            DevBuf flag;
            if (flag.grow(4, stream)) return -1;
            HIP_OK(hipMemsetAsync(flag.p, 0, 4, stream));
            hipLaunchKernelGGL(k_any_null, dim3(gx_grid(n)), dim3(256), 0, stream,
                               build.view(0), n, (uint32_t *)flag.p);
            uint32_t f = 0;
            HIP_OK(hipMemcpyAsync(&f, flag.p, 4, hipMemcpyDeviceToHost, stream));
            HIP_OK(hipStreamSynchronize(stream));
            flag.release();
            if (f) pass_nothing = true;
        }

        if (cfg.build_outer) {
            size_t words = (size_t)((n + 31) / 32);
            if (words == 0) words = 1;
            if (d_bitmap.grow(words * 4, stream)) return -1;
            HIP_OK(hipMemsetAsync(d_bitmap.p, 0, words * 4, stream));
        }
        HIP_OK(hipStreamSynchronize(stream));
        built = true;
        return 0;
    }

    struct OutSpec { bool outer; int src; int32_t type; };

    /* full output schema in reference order (AbstractJoinExec.java:102-227) */
    std::vector<OutSpec> full_specs() const {
        std::vector<OutSpec> s;
        auto add = [&](bool is_outer, int i, int32_t t) { s.push_back({is_outer, i, t}); };
        if (cfg.join_type == GX_JOIN_SEMI || cfg.join_type == GX_JOIN_ANTI) {
            for (size_t i = 0; i < outer_types.size(); i++)
                add(true, (int)i, outer_types[i]);
            return s;
        }
        size_t n_inner_out = cfg.single_join ? 1 : inner_types.size();
        if (cfg.join_type != GX_JOIN_RIGHT) {
            for (size_t i = 0; i < outer_types.size(); i++)
                add(true, (int)i, outer_types[i]);
            for (size_t i = 0; i < n_inner_out; i++)
                add(false, (int)i, inner_types[i]);
        } else {
            for (size_t i = 0; i < inner_types.size(); i++)
                add(false, (int)i, inner_types[i]);
            for (size_t i = 0; i < outer_types.size(); i++)
                add(true, (int)i, outer_types[i]);
        }
        return s;
    }

    std::vector<OutSpec> projected_specs() const {
        std::vector<OutSpec> full = full_specs();
        if (out_proj.empty()) return full;
        std::vector<OutSpec> sel;
        for (int32_t i : out_proj) sel.push_back(full[(size_t)i]);
        return sel;
    }

    std::vector<int32_t> output_types() const {
        std::vector<int32_t> t;
        for (auto &s : projected_specs()) t.push_back(s.type);
        return t;
    }

    /* gather output columns for a pair list into a new result */
    HipResult *materialize(const StagedChunk &probe, const uint32_t *d_pidx,
                           const uint32_t *d_bpos, int64_t n_out) {
        std::vector<int32_t> otypes = output_types();
        if (otypes.size() > GX_MAX_COLS) { gx_set_err("too many output cols"); return nullptr; }
        HipResult *h = alloc_result_cols(otypes, n_out, device, stream);
        if (!h) return nullptr;
        if (n_out == 0) return h;

        GatherParams G;
        std::memset(&G, 0, sizeof(G));
        /* "probe"/"build" sides: d_pidx always indexes the probe chunk and
         * d_bpos the build store; which output column comes from which side
         * honors build_outer. side 2 (null fill) when that index ptr is null
         * (pass-through). */
        G.pidx = d_pidx;
        G.bidx = d_bpos;
        G.n = n_out;
        int col = 0;
        int n_fused = 0;
        struct SliceJob { DevColView src; const uint32_t *idx; int null_fill; size_t col; };
        std::vector<SliceJob> slice_jobs;
        /* spec.outer refers to the logical OUTER input; map to probe/build
         * honoring build_outer (ParallelHashJoinExec buildJoinRow overrides) */
        for (auto &spec : projected_specs()) {
            const bool from_build = (spec.outer && cfg.build_outer) ||
                                    (!spec.outer && !cfg.build_outer);
            DevColView src = from_build ? build.view(spec.src)
                                        : probe.views[spec.src];
            const uint32_t *idx = from_build ? d_bpos : d_pidx;
            if (src.type == GX_SLICE) {
                slice_jobs.push_back({src, idx, idx == nullptr ? 1 : 0,
                                      (size_t)col});
                col++;
                continue;
            }
            G.src[n_fused] = src;
            G.side[n_fused] = from_build ? (d_bpos ? 1 : 2) : (d_pidx ? 0 : 2);
            G.out_vals[n_fused] = h->bufs[col * 2].p;
            G.out_nulls[n_fused] = (uint8_t *)h->bufs[col * 2 + 1].p;
            n_fused++;
            col++;
        }
        G.n_cols = n_fused;
        if (n_fused > 0)
            hipLaunchKernelGGL(k_gather_multi, dim3(gx_grid(n_out)), dim3(256),
                               0, stream, G);
        for (auto &sj : slice_jobs) {
            if (gather_slice_col(sj.src, sj.null_fill ? nullptr : sj.idx,
                                 sj.null_fill, n_out, h, sj.col, d_scan_tmp,
                                 stream)) {
                free_result(h);
                return nullptr;
            }
        }
        return h;
    }

    int probe(const gx_chunk *ch, gx_result **out) {
        *out = nullptr;
        if (ensure_device(device)) return -1;
        if (!built) { gx_set_err("probe before build"); return -1; }

        StagedChunk probe_st;
        if (probe_st.stage(ch, stream)) return -1;
        const int64_t n = probe_st.n_rows;

        if (pass_nothing || n == 0) {
            HipResult *h = alloc_result_cols(output_types(), 0, device, stream);
            if (!h) { probe_st.release(); return -1; }
            *out = &h->res;
            probe_st.release();
            return 0;
        }

        /* probe hashes — generic path only; the fast path computes the
         * bucket hash inline from the key column (one pass saved) */
        KeyViews pk = key_views_staged(probe_st, probe_key_cols);
        if (!fast_i64) {
            if (d_ph.grow((size_t)n * 4, stream) ||
                d_pn.grow((size_t)n, stream))
                return -1;
            hipLaunchKernelGGL(k_hash_rows, dim3(gx_grid(n)), dim3(256), 0,
                               stream, pk, n, (int32_t *)d_ph.p,
                               (uint8_t *)d_pn.p, (int)null_safe_keys);
        }

        int rc = -1;
        do {
            if (pass_through) {
                /* ANTI over empty build: all probe rows pass */
                if (d_pidx.grow((size_t)n * 4, stream)) break;
                hipLaunchKernelGGL(k_iota, dim3(gx_grid(n)), dim3(256), 0, stream,
                                   (uint32_t *)d_pidx.p, n);
                HipResult *h = materialize(probe_st, (uint32_t *)d_pidx.p,
                                           nullptr, n);
                if (!h) break;
                HIP_OK(hipStreamSynchronize(stream));
                *out = &h->res;
                rc = 0;
                break;
            }

            uint32_t cap = (uint32_t)std::min<int64_t>(
                std::max<int64_t>((int64_t)n * 2, 1 << 16), INT64_C(1) << 31);
            if (d_meta.grow(8, stream)) break;

            /* Bucket-clustered staging (fast path): partition the probe
             * rows so each workgroup probes one ~2 MB bucket-range slice.
             * MEASURED OFF by default (r2 A/B, GX_RADIX=1 forces): the
             * ~2.5 probes/bucket do stop refetching HBM lines, but the
             * 30+ concurrent workgroups per XCD each hold a DIFFERENT
             * 2 MB slice — 60+ MB of hot slices against 4 MB of XCD L2 —
             * so the slices thrash each other and the staged probe ran
             * 15.5 ms vs 11.0 ms unstaged (plus a 6.1 ms scatter).
             * Making concurrent slices fit L2 needs <=128 KB slices
             * (64K partitions, two-level scatter) — not attempted. */
            const int64_t table_bytes = n_buckets * 4 * (int64_t)sizeof(JoinEntry);
            int64_t n_part_radix = 1;
            int radix_shift = 0;
            const char *radix_env = getenv("GX_RADIX");
            bool use_radix = fast_i64 && radix_env && radix_env[0] == '1';
            (void)table_bytes;
            if (use_radix) {
                const int64_t slice = 2 << 20;
                n_part_radix = gx_pow2((table_bytes + slice - 1) / slice);
                if (n_part_radix > GX_RADIX_MAXP) n_part_radix = GX_RADIX_MAXP;
                if (n_part_radix > n_buckets) n_part_radix = 1;
                while ((1 << radix_shift) < n_buckets / n_part_radix)
                    radix_shift++;
            }
            std::vector<uint32_t> radix_starts;
            if (n_part_radix > 1) {
                if (d_staged.grow((size_t)n * sizeof(RadixRow), stream) ||
                    d_radix_cnt.grow((size_t)(n_part_radix + 1) * 4, stream))
                    break;
                HIP_OK(hipMemsetAsync(d_radix_cnt.p, 0, (size_t)n_part_radix * 4,
                                      stream));
                hipLaunchKernelGGL(k_radix_count, dim3(gx_grid(n)), dim3(256), 0,
                                   stream, pk.col[0], n, mask, radix_shift,
                                   (int)n_part_radix, (uint32_t *)d_radix_cnt.p);
                std::vector<uint32_t> counts((size_t)n_part_radix);
                HIP_OK(hipMemcpyAsync(counts.data(), d_radix_cnt.p,
                                      (size_t)n_part_radix * 4,
                                      hipMemcpyDeviceToHost, stream));
                HIP_OK(hipStreamSynchronize(stream));
                radix_starts.assign((size_t)n_part_radix + 1, 0);
                for (int64_t p = 0; p < n_part_radix; p++)
                    radix_starts[p + 1] = radix_starts[p] + counts[p];
                HIP_OK(hipMemcpyAsync(d_radix_cnt.p, radix_starts.data(),
                                      (size_t)n_part_radix * 4,
                                      hipMemcpyHostToDevice, stream));
                hipLaunchKernelGGL(k_radix_scatter,
                                   dim3(gx_grid((n + GX_RADIX_TILE - 1) /
                                                GX_RADIX_TILE * 256) ),
                                   dim3(256), 0, stream,
                                   pk.col[0], n, mask, radix_shift,
                                   (int)n_part_radix,
                                   (uint32_t *)d_radix_cnt.p,
                                   (RadixRow *)d_staged.p);
                /* cursors now hold end offsets == starts[1..n]; rebuild the
                 * starts array (incl. leading 0) for the probe kernel */
                HIP_OK(hipMemcpyAsync(d_radix_cnt.p, radix_starts.data(),
                                      (size_t)(n_part_radix + 1) * 4,
                                      hipMemcpyHostToDevice, stream));
            }

            for (int attempt = 0; attempt < 4; attempt++) {
                if (d_pidx.grow((size_t)cap * 4, stream) ||
                    d_bpos.grow((size_t)cap * 4, stream)) { attempt = 99; break; }
                HIP_OK(hipMemsetAsync(d_meta.p, 0, 8, stream));
                ProbeParams P;
                P.table = (const JoinEntry *)d_table.p;
                P.entries = (const JoinEntry *)d_entries.p;
                P.mask = mask;
                P.n_probe = n;
                P.hashes = (const int32_t *)d_ph.p;
                P.keynull = (const uint8_t *)d_pn.p;
                P.staged = nullptr;
                P.part_starts = nullptr;
                P.n_parts = 0;
                P.bloom = (const uint32_t *)d_bloom.p;
                P.bloom_mask = bloom_mask;
                P.fast_i64 = (int)fast_i64;
                P.build_keys = key_views(build, build_key_cols);
                P.probe_keys = pk;
                P.join_type = cfg.join_type;
                P.semi_join = (cfg.join_type == GX_JOIN_SEMI ||
                               cfg.join_type == GX_JOIN_ANTI) && !cfg.single_join;
                P.outer_join = cfg.join_type == GX_JOIN_LEFT ||
                               cfg.join_type == GX_JOIN_RIGHT;
                P.single_join = cfg.single_join;
                P.build_outer = cfg.build_outer;
                P.anti_null_col = cfg.anti_null_col;
                if (cfg.anti_null_col >= 0)
                    P.anti_col = probe_st.views[cfg.anti_null_col];
                else
                    std::memset(&P.anti_col, 0, sizeof(P.anti_col));
                P.out_probe = (uint32_t *)d_pidx.p;
                P.out_build = (uint32_t *)d_bpos.p;
                P.cap = cap;
                P.counter = (uint32_t *)d_meta.p;
                P.err = (uint32_t *)d_meta.p + 1;
                P.build_matched = cfg.build_outer ? (uint32_t *)d_bitmap.p : nullptr;
                P.n_conds = (int32_t)conds.size();
                if (!conds.empty() && upload_cond_pats()) { attempt = 99; break; }
                for (size_t t = 0; t < conds.size(); t++) {
                    const gx_join_cond &src = conds[t];
                    JoinCondDev &d = P.conds[t];
                    std::memset(&d, 0, sizeof(d));
                    d.cmp = src.cmp;
                    d.const_is_null = src.const_is_null;
                    bool a_build = false, b_build = false;
                    int a_src = cond_map(src.col_a, a_build);
                    d.a_is_build = a_build;
                    d.a = a_build ? build.view(a_src) : probe_st.views[a_src];
                    d.b_is_const = src.col_b < 0;
                    if (d.b_is_const) {
                        d.v_i64 = src.v_i64;
                        d.v_f64 = src.v_f64;
                        d.v_bytes = (const uint8_t *)d_cond_pats[t].p;
                        d.v_len = src.v_len;
                    } else {
                        int b_src = cond_map(src.col_b, b_build);
                        d.b_is_build = b_build;
                        d.b = b_build ? build.view(b_src) : probe_st.views[b_src];
                    }
                }
                if (!ev0) {
                    HIP_OK(hipEventCreate(&ev0));
                    HIP_OK(hipEventCreate(&ev1));
                }
                HIP_OK(hipEventRecord(ev0, stream));
                if (n_part_radix > 1) {
                    P.staged = (const RadixRow *)d_staged.p;
                    P.part_starts = (const uint32_t *)d_radix_cnt.p;
                    P.n_parts = (int32_t)n_part_radix;
                    hipLaunchKernelGGL(k_probe, dim3((uint32_t)n_part_radix),
                                       dim3(256), 0, stream, P);
                } else {
                    hipLaunchKernelGGL(k_probe, dim3(gx_grid(n)), dim3(256), 0,
                                       stream, P);
                }
                HIP_OK(hipEventRecord(ev1, stream));
                uint32_t meta[2];
                HIP_OK(hipMemcpyAsync(meta, d_meta.p, 8, hipMemcpyDeviceToHost, stream));
                HIP_OK(hipStreamSynchronize(stream));
                if (meta[1]) {
                    gx_set_err("ERR_SCALAR_SUBQUERY_RETURN_MORE_THAN_ONE_ROW");
                    probe_st.release();
                    return -2;
                }
                {
                    float ms = 0;
                    HIP_OK(hipEventElapsedTime(&ms, ev0, ev1));
                    probe_kernel_ms += ms;
                    probe_launches++;
                }
                if (meta[0] <= cap) {
                    probe_rows_total += n;
                    matches_total += meta[0];
                    HipResult *h = materialize(probe_st, (uint32_t *)d_pidx.p,
                                               (uint32_t *)d_bpos.p, meta[0]);
                    if (!h) { attempt = 99; break; }
                    HIP_OK(hipStreamSynchronize(stream));
                    *out = &h->res;
                    rc = 0;
                    break;
                }
                cap = meta[0]; /* retry with exact size */
            }
        } while (0);
        probe_st.release();
        return rc;
    }

    /* Buffered probe (gxop.h: the LocalBufferExec pattern): push appends
     * the chunk's columns into a device store (async copies only); flush
     * runs ONE probe over the accumulated batch. Amortizes the ~60 us
     * launch+sync cost of a per-chunk probe call across many CHUNK_SIZE
     * chunks. */
    int probe_push(const gx_chunk *ch) {
        if (ensure_device(device)) return -1;
        if (!built) { gx_set_err("probe before build"); return -1; }
        if (!probe_buf_init) {
            const auto &pt = cfg.build_outer ? inner_types : outer_types;
            probe_buf.init((int32_t)pt.size(), pt.data(), stream);
            probe_buf_init = true;
        }
        return probe_buf.append(ch);
    }

    int probe_flush(gx_result **out) {
        *out = nullptr;
        if (!probe_buf_init || probe_buf.n_rows == 0) return 0;
        if (probe_buf.n_rows > INT32_MAX) {
            gx_set_err("buffered probe exceeds 2^31 rows; flush earlier");
            return -1;
        }
        std::vector<gx_block> blocks(probe_buf.cols.size());
        for (size_t c = 0; c < probe_buf.cols.size(); c++) {
            DevColView v = probe_buf.view((int32_t)c);
            gx_block &b = blocks[c];
            std::memset(&b, 0, sizeof(b));
            b.type = v.type;
            b.mem = GX_MEM_DEVICE;
            b.values = v.values;
            b.nulls = v.has_nulls ? v.nulls : nullptr;
            b.offsets = v.offsets;
            b.data = v.bytes;
        }
        gx_chunk ch{(int32_t)probe_buf.n_rows, (int32_t)blocks.size(),
                    blocks.data()};
        int rc = probe(&ch, out);
        probe_buf.release();
        probe_buf_init = false;
        return rc;
    }

    int tail(gx_result **out) {
        *out = nullptr;
        if (!cfg.build_outer || tail_done) { tail_done = true; return 0; }
        if (ensure_device(device)) return -1;
        tail_done = true;
        const int64_t n = build.n_rows;
        if (n == 0) return 0;
        DevBuf d_idx, d_cnt;
        if (d_idx.grow((size_t)n * 4, stream) || d_cnt.grow(4, stream)) return -1;
        HIP_OK(hipMemsetAsync(d_cnt.p, 0, 4, stream));
        hipLaunchKernelGGL(k_unmatched, dim3(gx_grid(n)), dim3(256), 0, stream,
                           (const uint32_t *)d_bitmap.p, n, (uint32_t *)d_idx.p,
                           (uint32_t)n, (uint32_t *)d_cnt.p);
        uint32_t cnt = 0;
        HIP_OK(hipMemcpyAsync(&cnt, d_cnt.p, 4, hipMemcpyDeviceToHost, stream));
        HIP_OK(hipStreamSynchronize(stream));
        if (cnt == 0) { d_idx.release(); d_cnt.release(); return 0; }

        std::vector<int32_t> otypes = output_types();
        HipResult *h = alloc_result_cols(otypes, cnt, device, stream);
        if (!h) { d_idx.release(); d_cnt.release(); return -1; }
        size_t col = 0;
        int rc_col = 0;
        for (auto &spec : projected_specs()) {
            /* buildOuter tail: outer columns come from the BUILD store,
             * inner columns are null-filled (nextJoinNullRows:372-401) */
            const bool from_build = spec.outer;
            const bool null_fill = !spec.outer;
            DevColView src = from_build ? build.view(spec.src) : build.view(0);
            if (null_fill) src.type = spec.type;
            if (spec.type == GX_SLICE) {
                if (gather_slice_col(src, (const uint32_t *)d_idx.p,
                                     (int)null_fill, cnt, h, col, d_scan_tmp,
                                     stream))
                    rc_col = -1;
                col++;
                continue;
            }
            hipLaunchKernelGGL(k_gather, dim3(gx_grid(cnt)), dim3(256), 0, stream,
                               src, (const uint32_t *)d_idx.p, (int64_t)cnt,
                               h->bufs[col * 2].p, (uint8_t *)h->bufs[col * 2 + 1].p,
                               (int)null_fill);
            col++;
        }
        HIP_OK(hipStreamSynchronize(stream));
        d_idx.release(); d_cnt.release();
        if (rc_col != 0) { free_result(h); return -1; }
        *out = &h->res;
        return 0;
    }
};

} // namespace

/* ======================= agg + partition in gxhip_agg.inc ============== */
#include "gxhip_agg.inc"
#include "gxhip_part.inc"
#include "gxhip_hybrid.inc"
#include "gxhip_groupjoin.inc"
#include "gxhip_window.inc"
#include "gxhip_fwindow.inc"
#include "gxhip_scan.inc"
#include "gx_serde.inc"

/* ========================= C ABI ======================================= */

extern "C" {

gx_op *gxop_join_create(const gx_join_cfg *cfg) {
    if (!cfg || cfg->n_keys <= 0 || cfg->n_keys > GX_MAX_KEYS) {
        gx_set_err("bad join cfg");
        return nullptr;
    }
    if (cfg->join_type < GX_JOIN_INNER || cfg->join_type > GX_JOIN_ANTI) {
        gx_set_err("unknown join type");
        return nullptr;
    }
    if (cfg->n_conds < 0 || cfg->n_conds > GX_MAX_CONDS ||
        (cfg->n_conds > 0 && !cfg->conds)) {
        gx_set_err("bad join condition list (max 4 AND terms)");
        return nullptr;
    }
    if (cfg->device < 0) { gx_set_err("gxhip requires a GPU device"); return nullptr; }
    HIP_OK_NULL(hipSetDevice(cfg->device));
    /* out-of-core decision (HybridHashJoinExec vs ParallelHashJoinExec):
     * needs both a budget and a size hint; otherwise stay in-HBM */
    if (cfg->memory_budget_bytes > 0 && cfg->expected_build_rows > 0) {
        std::vector<int32_t> bt(cfg->build_outer ? cfg->outer_types
                                                 : cfg->inner_types,
                                (cfg->build_outer ? cfg->outer_types
                                                  : cfg->inner_types) +
                                (cfg->build_outer ? cfg->n_outer_cols
                                                  : cfg->n_inner_cols));
        int64_t est = cfg->expected_build_rows * join_row_bytes(bt);
        if (est > cfg->memory_budget_bytes)
            return new HybridJoinOp(cfg, hybrid_partition_count(
                                             est, cfg->memory_budget_bytes));
    }
    return new JoinOp(cfg);
}
int gxop_join_consume(gx_op *op, const gx_chunk *c) {
    if (!op || op->kind != OP_JOIN) { gx_set_err("not a join op"); return -1; }
    std::lock_guard<std::mutex> lk(op->mu);
    if (op->variant == 1) return static_cast<HybridJoinOp *>(op)->consume(c);
    return static_cast<JoinOp *>(op)->consume(c);
}
int gxop_join_build(gx_op *op) {
    if (!op || op->kind != OP_JOIN) { gx_set_err("not a join op"); return -1; }
    std::lock_guard<std::mutex> lk(op->mu);
    if (op->variant == 1) return static_cast<HybridJoinOp *>(op)->do_build();
    return static_cast<JoinOp *>(op)->do_build();
}
int gxop_join_probe(gx_op *op, const gx_chunk *c, gx_result **out) {
    if (!op || op->kind != OP_JOIN) { gx_set_err("not a join op"); return -1; }
    std::lock_guard<std::mutex> lk(op->mu);
    if (op->variant == 1) return static_cast<HybridJoinOp *>(op)->probe(c, out);
    return static_cast<JoinOp *>(op)->probe(c, out);
}
int gxop_join_tail(gx_op *op, gx_result **out) {
    if (!op || op->kind != OP_JOIN) { gx_set_err("not a join op"); return -1; }
    std::lock_guard<std::mutex> lk(op->mu);
    if (op->variant == 1) return static_cast<HybridJoinOp *>(op)->tail(out);
    return static_cast<JoinOp *>(op)->tail(out);
}
int gxop_join_probe_push(gx_op *op, const gx_chunk *c) {
    if (!op || op->kind != OP_JOIN) { gx_set_err("not a join op"); return -1; }
    std::lock_guard<std::mutex> lk(op->mu);
    if (op->variant == 1) {
        /* hybrid defers all probe output anyway: push = spill the chunk */
        gx_result *ignored = nullptr;
        return static_cast<HybridJoinOp *>(op)->probe(c, &ignored);
    }
    return static_cast<JoinOp *>(op)->probe_push(c);
}
int gxop_join_probe_flush(gx_op *op, gx_result **out) {
    if (!op || op->kind != OP_JOIN) { gx_set_err("not a join op"); return -1; }
    std::lock_guard<std::mutex> lk(op->mu);
    if (op->variant == 1) { *out = nullptr; return 0; } /* drained via tail */
    return static_cast<JoinOp *>(op)->probe_flush(out);
}
int gxop_join_close(gx_op *op) { delete op; return 0; }

int gxop_result_to_host(gx_result *res) {
    if (!res) return 0;
    HipResult *h = static_cast<HipResult *>(res->opaque);
    if (!h) return 0;
    bool any_dev = false;
    for (auto &b : h->blocks) any_dev |= (b.mem == GX_MEM_DEVICE);
    if (!any_dev) return 0;
    HIP_OK(hipSetDevice(h->device));
    int64_t n = res->chunk.n_rows;
    for (size_t c = 0; c < h->blocks.size(); c++) {
        gx_block &b = h->blocks[c];
        if (b.mem != GX_MEM_DEVICE) continue;
        if (b.type == GX_SLICE) {
            h->host.emplace_back((size_t)n * 4);
            if (n > 0)
                HIP_OK(hipMemcpy(h->host.back().data(), (const void *)b.offsets,
                                 (size_t)n * 4, hipMemcpyDeviceToHost));
            b.offsets = (const int32_t *)h->host.back().data();
            h->host.emplace_back((size_t)(b.data_len > 0 ? b.data_len : 1));
            if (b.data_len > 0)
                HIP_OK(hipMemcpy(h->host.back().data(), (const void *)b.data,
                                 (size_t)b.data_len, hipMemcpyDeviceToHost));
            b.data = (const uint8_t *)h->host.back().data();
        } else {
            size_t es = gx_fixed_size(b.type);
            h->host.emplace_back((size_t)n * es);
            if (n > 0)
                HIP_OK(hipMemcpy(h->host.back().data(), b.values, (size_t)n * es,
                                 hipMemcpyDeviceToHost));
            b.values = h->host.back().data();
        }
        h->host.emplace_back((size_t)n);
        if (n > 0)
            HIP_OK(hipMemcpy(h->host.back().data(), (void *)b.nulls, (size_t)n,
                             hipMemcpyDeviceToHost));
        b.nulls = h->host.back().data();
        b.mem = GX_MEM_HOST;
    }
    for (auto &db : h->bufs) db.release();
    h->bufs.clear();
    return 0;
}

void gxop_result_release(gx_result *res) {
    if (res && res->opaque) free_result(static_cast<HipResult *>(res->opaque));
}

int gxop_result_copy_col(const gx_result *res, int32_t col, void *dst_values,
                         void *dst_nulls) {
    if (!res || col < 0 || col >= res->chunk.n_blocks) {
        gx_set_err("bad result column");
        return -1;
    }
    const gx_block *b = &res->chunk.blocks[col];
    int64_t n = res->chunk.n_rows;
    if (n == 0) return 0;
    size_t es = gx_fixed_size(b->type);
    HIP_OK(hipMemcpy(dst_values, b->values, (size_t)n * es, hipMemcpyDefault));
    if (dst_nulls && b->nulls)
        HIP_OK(hipMemcpy(dst_nulls, (const void *)b->nulls, (size_t)n,
                         hipMemcpyDefault));
    return 0;
}

int gxop_join_get_stats(gx_op *op, gx_join_stats *out) {
    if (!op || op->kind != OP_JOIN || !out) { gx_set_err("not a join op"); return -1; }
    if (op->variant == 1) {
        HybridJoinOp *h = static_cast<HybridJoinOp *>(op);
        out->probe_kernel_ms = h->probe_kernel_ms;
        out->probe_launches = h->probe_launches;
        out->probe_rows = h->probe_rows_total;
        out->matches = h->matches_total;
        return 0;
    }
    JoinOp *j = static_cast<JoinOp *>(op);
    out->probe_kernel_ms = j->probe_kernel_ms;
    out->probe_launches = j->probe_launches;
    out->probe_rows = j->probe_rows_total;
    out->matches = j->matches_total;
    return 0;
}
const char *gx_last_error(void) { return gx_err.c_str(); }
int gxop_abi_version(void) { return 950; }

} /* extern "C" */
