/*
 * gx_oracle.cpp — CPU oracle: a C++ restatement of the PolarDB-X CN hot-path
 * operator semantics (hash join / hash agg / partition exchange), used ONLY
 * as the parity checker and the reported CPU baseline.
 *
 * TEST INFRASTRUCTURE — NOT THE PRODUCT PATH. Only tests/, __graft_entry__'s
 * smoke() and bench.py's cpu_baseline leg may load this library. The product
 * path is the HIP library (galaxysql_amd/csrc); it must fail loudly when its
 * extension is missing, never fall back here.
 *
 * Every function cites the reference file:line it restates (paths under
 * /root/reference/polardbx-executor/src/main/java/com/alibaba/polardbx/
 * executor/ unless noted). Parity is pinned by golden vectors transcribed
 * from the reference's own tests (tests/golden/*, from HashJoinTest.java and
 * HashAggExecTest.java) — see tests/test_oracle_golden.py.
 *
 * Third-party arithmetic: fastutil HashCommon.mix / murmurHash3 / arraySize
 * and airlift Slice.hashCode (XxHash64) are NOT vendored in the reference
 * (transitive deps, version unpinned). They are restated here from their
 * published algorithms; they affect bucket/partition PLACEMENT only, never
 * result rows (SURVEY.md §8c "hash-placement parity: unpinned").
 */
#include "../include/gxop.h"

#include <cstring>
#include <cstdlib>
#include <cstdio>
#include <cmath>
#include <string>
#include <vector>
#include <memory>
#include <mutex>
#include <algorithm>

namespace {

thread_local std::string g_err;
void set_err(const char *msg) { g_err = msg ? msg : ""; }

/* ---- Java 32-bit arithmetic + hash functions --------------------------- */

static inline int32_t jmul(int32_t a, int32_t b) {
    return (int32_t)((uint32_t)a * (uint32_t)b);
}
static inline int32_t jadd(int32_t a, int32_t b) {
    return (int32_t)((uint32_t)a + (uint32_t)b);
}


/* Java Math.min/max semantics for doubles (NaN propagates, -0.0 < +0.0) —
 * what Double2DoubleMin/Max accumulate with (Math.min,
 * Double2DoubleMin.java:40-44); std::min/max differ on NaN and -0.0. */
static inline double jmin_f64o(double a, double b) {
    if (a != a) return a;
    if (b != b) return b;
    if (a == 0.0 && b == 0.0)
        return std::signbit(a) ? a : b;
    return a < b ? a : b;
}
static inline double jmax_f64o(double a, double b) {
    if (a != a) return a;
    if (b != b) return b;
    if (a == 0.0 && b == 0.0)
        return std::signbit(a) ? b : a;
    return a > b ? a : b;
}

/* fastutil HashCommon.mix (int phi 0x9E3779B9, h ^= h >>> 16) —
 * ConcurrentRawHashTable.java:93,114 uses it for bucket placement. */
static inline int32_t hc_mix(int32_t x) {
    int32_t h = jmul(x, (int32_t)0x9E3779B9);
    return h ^ (int32_t)((uint32_t)h >> 16);
}

/* fastutil HashCommon.murmurHash3(int) — ExecUtils.partition
 * (utils/ExecUtils.java:1023-1033). */
static inline int32_t hc_murmur3(int32_t x) {
    uint32_t h = (uint32_t)x;
    h ^= h >> 16; h *= 0x85ebca6bu; h ^= h >> 13; h *= 0xc2b2ae35u; h ^= h >> 16;
    return (int32_t)h;
}

/* fastutil HashCommon.arraySize(expected, f) = max(2, nextPow2(ceil(n/f))),
 * used by ConcurrentRawHashTable.java:67-75 and GroupOpenHashMap.java. */
static inline int64_t next_pow2(int64_t x) {
    if (x <= 1) return 1;
    x--;
    x |= x >> 1; x |= x >> 2; x |= x >> 4; x |= x >> 8; x |= x >> 16; x |= x >> 32;
    return x + 1;
}
static inline int64_t hc_array_size(int64_t expected, double f) {
    int64_t s = next_pow2((int64_t)std::ceil((double)expected / f));
    if (s < 2) s = 2;
    return s;
}
/* fastutil HashCommon.maxFill(n, f) = min(ceil(n*f), n-1). */
static inline int64_t hc_max_fill(int64_t n, double f) {
    int64_t m = (int64_t)std::ceil((double)n * f);
    return m < n - 1 ? m : n - 1;
}

/* ConcurrentRawHashTable.selectLoadFactor (ConcurrentRawHashTable.java:67-75):
 * >=100M -> 0.75 (DEFAULT), >=10M -> 0.5 (FAST), else 0.25 (VERY_FAST). */
static inline double join_load_factor(int64_t size) {
    if (size >= 100000000) return 0.75;
    if (size >= 10000000) return 0.5;
    return 0.25;
}

/* Long.hashCode(v) = (int)(v ^ (v >>> 32)) — LongBlock.java:110-127. */
static inline int32_t hash_i64(int64_t v) {
    return (int32_t)((uint64_t)v ^ ((uint64_t)v >> 32));
}
/* IntegerBlock.hashCode = the value itself (IntegerBlock.java:112-117). */
static inline int32_t hash_i32(int32_t v) { return v; }
/* Double.hashCode = Long.hashCode(doubleToLongBits(v)) — DoubleBlock.java;
 * doubleToLongBits canonicalizes NaN to 0x7ff8000000000000. */
static inline int32_t hash_f64(double v) {
    uint64_t bits;
    if (std::isnan(v)) bits = 0x7ff8000000000000ull;
    else std::memcpy(&bits, &v, 8);
    return (int32_t)(bits ^ (bits >> 32));
}

/* airlift Slice.hashCode(offset,len) = (int) XxHash64.hash(seed=0) over the
 * raw bytes (SliceBlock.java:183-195, non-"compatible" path). XxHash64
 * restated from its published spec; placement-only, unpinned (see header). */
static const uint64_t XXP1 = 0x9E3779B185EBCA87ull, XXP2 = 0xC2B2AE3D27D4EB4Full,
                      XXP3 = 0x165667B19E3779F9ull, XXP4 = 0x85EBCA77C2B2AE63ull,
                      XXP5 = 0x27D4EB2F165667C5ull;
static inline uint64_t rotl64(uint64_t x, int r) { return (x << r) | (x >> (64 - r)); }
static inline uint64_t xx_read64(const uint8_t *p) { uint64_t v; std::memcpy(&v, p, 8); return v; }
static inline uint32_t xx_read32(const uint8_t *p) { uint32_t v; std::memcpy(&v, p, 4); return v; }
static uint64_t xxhash64(const uint8_t *data, size_t len) {
    const uint8_t *p = data, *end = data + len;
    uint64_t h;
    if (len >= 32) {
        uint64_t v1 = XXP1 + XXP2, v2 = XXP2, v3 = 0, v4 = (uint64_t)0 - XXP1;
        const uint8_t *limit = end - 32;
        do {
            v1 = rotl64(v1 + xx_read64(p) * XXP2, 31) * XXP1; p += 8;
            v2 = rotl64(v2 + xx_read64(p) * XXP2, 31) * XXP1; p += 8;
            v3 = rotl64(v3 + xx_read64(p) * XXP2, 31) * XXP1; p += 8;
            v4 = rotl64(v4 + xx_read64(p) * XXP2, 31) * XXP1; p += 8;
        } while (p <= limit);
        h = rotl64(v1, 1) + rotl64(v2, 7) + rotl64(v3, 12) + rotl64(v4, 18);
        auto merge = [&](uint64_t v) {
            h ^= rotl64(v * XXP2, 31) * XXP1; h = h * XXP1 + XXP4;
        };
        merge(v1); merge(v2); merge(v3); merge(v4);
    } else {
        h = XXP5;
    }
    h += (uint64_t)len;
    while (p + 8 <= end) {
        h ^= rotl64(xx_read64(p) * XXP2, 31) * XXP1;
        h = rotl64(h, 27) * XXP1 + XXP4; p += 8;
    }
    if (p + 4 <= end) {
        h ^= (uint64_t)xx_read32(p) * XXP1;
        h = rotl64(h, 23) * XXP2 + XXP3; p += 4;
    }
    while (p < end) {
        h ^= (*p) * XXP5;
        h = rotl64(h, 11) * XXP1; p++;
    }
    h ^= h >> 33; h *= XXP2; h ^= h >> 29; h *= XXP3; h ^= h >> 32;
    return h;
}

/* ---- column store ------------------------------------------------------ */

/* Host-side accumulation of consumed chunks, SoA per column — the oracle's
 * ChunksIndex (operator/util/ChunksIndex.java:38-60) flattened to global
 * positions (position = running row index across consumed chunks). */
struct Column {
    int32_t type = GX_I64;
    std::vector<int64_t> i64v;
    std::vector<int32_t> i32v;
    std::vector<double>  f64v;
    std::vector<int32_t> off;   /* slice end-offsets (global) */
    std::vector<uint8_t> bytes; /* slice data / decimal 40B slots */
    std::vector<uint8_t> null_; /* 1 = NULL */

    size_t size() const { return null_.size(); }

    void append(const gx_block *b, int32_t n) {
        size_t old = null_.size();
        null_.resize(old + n, 0);
        if (b->nulls) std::memcpy(null_.data() + old, b->nulls, n);
        switch (b->type) {
        case GX_I64: {
            i64v.resize(old + n);
            std::memcpy(i64v.data() + old, b->values, (size_t)n * 8);
            break; }
        case GX_I32: {
            i32v.resize(old + n);
            std::memcpy(i32v.data() + old, b->values, (size_t)n * 4);
            break; }
        case GX_F64: {
            f64v.resize(old + n);
            std::memcpy(f64v.data() + old, b->values, (size_t)n * 8);
            break; }
        case GX_SLICE: {
            int32_t base = (int32_t)bytes.size();
            int64_t blen = (n > 0) ? b->offsets[n - 1] : 0;
            bytes.insert(bytes.end(), b->data, b->data + blen);
            off.reserve(off.size() + n);
            for (int32_t i = 0; i < n; i++) off.push_back(base + b->offsets[i]);
            break; }
        case GX_DECIMAL: {
            const uint8_t *src = (const uint8_t *)b->values;
            bytes.insert(bytes.end(), src, src + (size_t)n * 40);
            break; }
        }
    }

    bool is_null(size_t i) const { return null_[i] != 0; }
    int32_t begin_off(size_t i) const {
        /* global begin offset of slice value i */
        if (i == 0) return 0;
        /* find the previous row's end; rows are global so previous end works
         * only within the flattened store where offsets are cumulative */
        return off[i - 1];
    }

    /* Block.hashCode(position): null -> 0 (LongBlock.java:110-127 et al). */
    int32_t hash_at(size_t i) const {
        if (is_null(i)) return 0;
        switch (type) {
        case GX_I64: return hash_i64(i64v[i]);
        case GX_I32: return hash_i32(i32v[i]);
        case GX_F64: return hash_f64(f64v[i]);
        case GX_SLICE: {
            int32_t b = begin_off(i), e = off[i];
            return (int32_t)xxhash64(bytes.data() + b, (size_t)(e - b));
        }
        }
        return 0;
    }

    /* Block.equals(position, other, otherPosition): null==null true, else
     * value equality (LongBlock.java:69-127, IntegerBlock.java:102-166,
     * DoubleBlock, SliceBlock byte compare). */
    bool equal_at(size_t i, const Column &o, size_t j) const {
        bool n1 = is_null(i), n2 = o.is_null(j);
        if (n1 && n2) return true;
        if (n1 != n2) return false;
        switch (type) {
        case GX_I64: return i64v[i] == o.i64v[j];
        case GX_I32: return i32v[i] == o.i32v[j];
        case GX_F64: return f64v[i] == o.f64v[j]; /* Java '==' on double */
        case GX_SLICE: {
            int32_t b1 = begin_off(i), e1 = off[i];
            int32_t b2 = o.begin_off(j), e2 = o.off[j];
            if (e1 - b1 != e2 - b2) return false;
            return std::memcmp(bytes.data() + b1, o.bytes.data() + b2,
                               (size_t)(e1 - b1)) == 0;
        }
        }
        return false;
    }
};

struct Store {
    std::vector<Column> cols;
    size_t n_rows = 0;
    void init(int32_t n_cols, const int32_t *types) {
        cols.resize(n_cols);
        for (int32_t c = 0; c < n_cols; c++) cols[c].type = types[c];
    }
    int append(const gx_chunk *ch) {
        if ((size_t)ch->n_blocks != cols.size()) { set_err("column count mismatch"); return -1; }
        for (int32_t c = 0; c < ch->n_blocks; c++) {
            if (ch->blocks[c].mem != GX_MEM_HOST) { set_err("oracle accepts host memory only"); return -1; }
            if (ch->blocks[c].type != cols[c].type) { set_err("column type mismatch"); return -1; }
            cols[c].append(&ch->blocks[c], ch->n_rows);
        }
        n_rows += ch->n_rows;
        return 0;
    }
    /* Chunk.hashCode(position): h = 31*h + block.hashCode(position)
     * (chunk/Chunk.java:116-129), over the given columns in order. */
    int32_t row_hash(size_t row, const std::vector<int> &key_cols) const {
        int32_t h = 0;
        for (int c : key_cols) h = jadd(jmul(h, 31), cols[c].hash_at(row));
        return h;
    }
    bool row_has_null(size_t row, const std::vector<int> &key_cols) const {
        for (int c : key_cols) if (cols[c].is_null(row)) return true;
        return false;
    }
    bool keys_equal(size_t row, const Store &o, size_t orow,
                    const std::vector<int> &my_cols, const std::vector<int> &o_cols) const {
        for (size_t k = 0; k < my_cols.size(); k++)
            if (!cols[my_cols[k]].equal_at(row, o.cols[o_cols[k]], orow)) return false;
        return true;
    }
};

/* ---- output builder ---------------------------------------------------- */

struct OutCol {
    int32_t type;
    std::vector<int64_t> i64v;
    std::vector<int32_t> i32v;
    std::vector<double>  f64v;
    std::vector<int32_t> off;
    std::vector<uint8_t> bytes;
    std::vector<uint8_t> null_;

    void append_null() {
        null_.push_back(1);
        switch (type) {
        case GX_I64: i64v.push_back(0); break;
        case GX_I32: i32v.push_back(0); break;
        case GX_F64: f64v.push_back(0); break;
        case GX_SLICE: off.push_back((int32_t)bytes.size()); break;
        case GX_DECIMAL: bytes.resize(bytes.size() + 40, 0); break;
        }
    }
    void append_from(const Column &c, size_t i) {
        if (c.is_null(i)) { append_null(); return; }
        null_.push_back(0);
        switch (type) {
        case GX_I64: i64v.push_back(c.i64v[i]); break;
        case GX_I32: i32v.push_back(c.i32v[i]); break;
        case GX_F64: f64v.push_back(c.f64v[i]); break;
        case GX_SLICE: {
            int32_t b = c.begin_off(i), e = c.off[i];
            bytes.insert(bytes.end(), c.bytes.data() + b, c.bytes.data() + e);
            off.push_back((int32_t)bytes.size());
            break; }
        case GX_DECIMAL:
            bytes.insert(bytes.end(), c.bytes.data() + (size_t)i * 40,
                         c.bytes.data() + (size_t)i * 40 + 40);
            break;
        }
    }
    void append_i64(int64_t v) { null_.push_back(0); i64v.push_back(v); }
    void append_dec40(const uint8_t *p) {
        null_.push_back(0);
        bytes.insert(bytes.end(), p, p + 40);
    }
    void append_f64(double v)  { null_.push_back(0); f64v.push_back(v); }
};

struct ResultHolder {
    std::vector<OutCol> cols;
    std::vector<gx_block> blocks;
    gx_result res;
};

static gx_result *make_result(std::vector<OutCol> &&cols, int32_t n_rows) {
    auto *h = new ResultHolder();
    h->cols = std::move(cols);
    h->blocks.resize(h->cols.size());
    for (size_t c = 0; c < h->cols.size(); c++) {
        OutCol &oc = h->cols[c];
        gx_block &b = h->blocks[c];
        std::memset(&b, 0, sizeof(b));
        b.type = oc.type;
        b.mem = GX_MEM_HOST;
        b.nulls = oc.null_.data();
        switch (oc.type) {
        case GX_I64: b.values = oc.i64v.data(); break;
        case GX_I32: b.values = oc.i32v.data(); break;
        case GX_F64: b.values = oc.f64v.data(); break;
        case GX_SLICE:
            b.offsets = oc.off.data();
            b.data = oc.bytes.data();
            b.data_len = (int64_t)oc.bytes.size();
            break;
        case GX_DECIMAL:
            b.values = oc.bytes.data();
            break;
        }
    }
    h->res.chunk.n_rows = n_rows;
    h->res.chunk.n_blocks = (int32_t)h->blocks.size();
    h->res.chunk.blocks = h->blocks.data();
    h->res.opaque = h;
    return &h->res;
}

/* ---- operator base ----------------------------------------------------- */

enum OpKind { OP_JOIN = 1, OP_AGG = 2, OP_PART = 3, OP_SCAN = 4,
              OP_GROUPJOIN = 5, OP_WINDOW = 6, OP_FWINDOW = 7 };

} // anonymous namespace

struct gx_op {
    int kind;
    /* consume/build serialization matching the C-ABI thread contract
     * (the reference's synchronized(shared) in
     * ParallelHashJoinExec.consumeChunk:158). probe stays lock-free here:
     * it only reads the built table, and the all-cores CPU baseline
     * (bench.py) probes one shared op from every host thread. */
    std::mutex mu;
    virtual ~gx_op() = default;
protected:
    explicit gx_op(int k) : kind(k) {}
};

namespace {

/* ---- join operator ------------------------------------------------------
 * Restates ParallelHashJoinExec + AbstractBufferedJoinExec.nextRows
 * (AbstractBufferedJoinExec.java:185-266) + AbstractHashJoinExec.matchInit/
 * matchNext (AbstractHashJoinExec.java:80-106) + Synchronizer.buildHashTable
 * (ParallelHashJoinExec.java:406-426) with numPartitions=1 (single-threaded:
 * insertion order = consume order, as in the reference's SingleExecTest). */
/* residual-condition comparisons (see gx_join_cond in gxop.h) */
static inline bool cond_cmp_i64o(int64_t a, int32_t cmp, int64_t b) {
    switch (cmp) {
    case GX_CMP_LT: return a < b;
    case GX_CMP_LE: return a <= b;
    case GX_CMP_GT: return a > b;
    case GX_CMP_GE: return a >= b;
    case GX_CMP_NE: case GX_CMP_NE_NULLSAFE: return a != b;
    default: return a == b;
    }
}
static inline bool cond_cmp_f64o(double a, int32_t cmp, double b) {
    switch (cmp) {
    case GX_CMP_LT: return a < b;
    case GX_CMP_LE: return a <= b;
    case GX_CMP_GT: return a > b;
    case GX_CMP_GE: return a >= b;
    case GX_CMP_NE: case GX_CMP_NE_NULLSAFE: return a != b;
    default: return a == b;
    }
}

struct JoinOp : gx_op {
    gx_join_cfg cfg;
    std::vector<gx_equi_key> keys;
    std::vector<gx_join_cond> conds;              /* residual condition */
    std::vector<std::vector<uint8_t>> cond_bytes; /* deep-copied SLICE consts */
    std::vector<int32_t> out_proj;  /* projection pushdown; empty = full */
    std::vector<int32_t> outer_types, inner_types;

    Store build;                 /* build-side input columns */
    std::vector<int> build_key_cols, probe_key_cols;
    std::vector<int32_t> key_types;

    /* hash table: bucket = mix(hash) & mask holds LATEST inserted position
     * (ConcurrentRawHashTable.put = getAndSet, java:100-121); chains via
     * position_links (head insertion, ExecUtils.buildOneChunk:914-944). */
    std::vector<int32_t> table;
    std::vector<int32_t> links;
    int32_t mask = 0;
    bool built = false;

    bool pass_nothing = false, pass_through = false;
    /* buildOuter unmatched tracking (Synchronizer.joinNullRowBitSet,
     * ParallelHashJoinExec.java:437-461). */
    std::vector<uint8_t> matched_build;
    size_t tail_cursor = 0;
    bool tail_done = false;

    JoinOp(const gx_join_cfg *c) : gx_op(OP_JOIN), cfg(*c) {
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
            /* build side = inner unless build_outer (AbstractBufferedJoinExec
             * getBuildInput/getProbeInput; buildOuter flips them) */
            build_key_cols.push_back(cfg.build_outer ? k.outer_index : k.inner_index);
            probe_key_cols.push_back(cfg.build_outer ? k.inner_index : k.outer_index);
            key_types.push_back(k.unified_type);
        }
        const auto &bt = cfg.build_outer ? outer_types : inner_types;
        build.init((int32_t)bt.size(), bt.data());
    }

    int consume(const gx_chunk *ch) { return build.append(ch); }

    /* buffered probe (gxop.h: the LocalBufferExec pattern) */
    Store probe_bufs;
    bool probe_buf_init = false;

    int probe_push(const gx_chunk *ch) {
        if (!built) { set_err("probe before build"); return -1; }
        if (!probe_buf_init) {
            const auto &pt = cfg.build_outer ? inner_types : outer_types;
            probe_bufs.init((int32_t)pt.size(), pt.data());
            probe_buf_init = true;
        }
        return probe_bufs.append(ch);
    }

    int probe_flush(gx_result **out) {
        *out = nullptr;
        if (!probe_buf_init || probe_bufs.n_rows == 0) return 0;
        std::vector<gx_block> blocks(probe_bufs.cols.size());
        for (size_t c = 0; c < probe_bufs.cols.size(); c++) {
            Column &col = probe_bufs.cols[c];
            gx_block &b = blocks[c];
            std::memset(&b, 0, sizeof(b));
            b.type = col.type;
            b.mem = GX_MEM_HOST;
            b.nulls = col.null_.data();
            switch (col.type) {
            case GX_I64: b.values = col.i64v.data(); break;
            case GX_I32: b.values = col.i32v.data(); break;
            case GX_F64: b.values = col.f64v.data(); break;
            case GX_DECIMAL: b.values = col.bytes.data(); break;
            case GX_SLICE:
                /* store offsets start at 0 = chunk-local for one chunk */
                b.offsets = col.off.data();
                b.data = col.bytes.data();
                break;
            }
        }
        gx_chunk ch{(int32_t)probe_bufs.n_rows, (int32_t)blocks.size(),
                    blocks.data()};
        int rc = probe(&ch, out);
        probe_bufs = Store();
        probe_buf_init = false;
        return rc;
    }


    int do_build() {
        if (built) return 0; /* first-come barrier */
        const int64_t size = (int64_t)build.n_rows;
        int64_t n = hc_array_size(size, join_load_factor(size));
        mask = (int32_t)(n - 1);
        table.assign((size_t)n, -1);
        links.assign((size_t)size, -1);
        /* ExecUtils.buildOneChunk (utils/ExecUtils.java:914-944): rows with a
         * NULL in ANY key column are skipped unconditionally (:933-941). */
        for (size_t pos = 0; pos < build.n_rows; pos++) {
            if (build.row_has_null(pos, build_key_cols)) continue;
            int32_t h = build.row_hash(pos, build_key_cols);
            int32_t slot = hc_mix(h) & mask;
            links[pos] = table[slot];
            table[slot] = (int32_t)pos;
        }
        built = true;

        /* special modes (ParallelHashJoinExec.buildConsume:107-128 +
         * AbstractBufferedJoinExec.doSpecialCheckForSemiJoin:293-316) */
        bool semi_join = (cfg.join_type == GX_JOIN_SEMI || cfg.join_type == GX_JOIN_ANTI)
                         && !cfg.single_join;
        if (build.n_rows == 0 && cfg.join_type == GX_JOIN_INNER) pass_nothing = true;
        if (semi_join && build.n_rows == 0) {
            if (cfg.join_type == GX_JOIN_SEMI) pass_nothing = true;
            else pass_through = true; /* ANTI with empty build passes all */
        } else if (cfg.join_type == GX_JOIN_ANTI && cfg.anti_null_col >= 0
                   && build.cols.size() == 1) {
            /* x NOT IN (...NULL...) -> empty result
             * (doSpecialCheckForSemiJoin:305-312) */
            for (size_t i = 0; i < build.n_rows && !pass_nothing; i++)
                if (build.cols[0].is_null(i)) pass_nothing = true;
        }
        if (cfg.build_outer) matched_build.assign(build.n_rows, 0);
        return 0;
    }

    /* condition-row column (JoinRelType.leftSide cols then rightSide cols,
     * polardbx-calcite JoinRelType.java:145-151) -> (build-side?, src col) */
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

    /* AbstractJoinExec.checkJoinCondition:227-250: evaluated per matched
     * candidate; failing candidates are skipped (don't set `matched`). */
    bool cond_pass(const Store &probe_store, size_t prow, size_t bpos) const {
        for (const auto &c : conds) {
            bool a_build = false, b_build = false;
            int a_src = cond_map(c.col_a, a_build);
            const Column &A = a_build ? build.cols[a_src]
                                      : probe_store.cols[a_src];
            size_t ia = a_build ? bpos : prow;
            const Column *B = nullptr;
            size_t ib = 0;
            bool na = A.is_null(ia), nb;
            if (c.col_b < 0) {
                nb = c.const_is_null != 0;
            } else {
                int b_src = cond_map(c.col_b, b_build);
                B = &(b_build ? build.cols[b_src] : probe_store.cols[b_src]);
                ib = b_build ? bpos : prow;
                nb = B->is_null(ib);
            }
            bool nullsafe = c.cmp == GX_CMP_EQ_NULLSAFE ||
                            c.cmp == GX_CMP_NE_NULLSAFE;
            bool ok;
            if (na || nb) {
                if (!nullsafe) return false; /* SQL NULL fails comparisons */
                bool eq = na && nb;          /* Objects.equals semantics */
                ok = c.cmp == GX_CMP_EQ_NULLSAFE ? eq : !eq;
            } else if (A.type == GX_SLICE) {
                int32_t ab = A.begin_off(ia), ae = A.off[ia];
                const uint8_t *bp;
                int32_t blen;
                if (c.col_b < 0) {
                    bp = c.v_bytes;
                    blen = c.v_len;
                } else {
                    int32_t bb = B->begin_off(ib);
                    bp = B->bytes.data() + bb;
                    blen = B->off[ib] - bb;
                }
                bool eq = (ae - ab) == blen &&
                          std::memcmp(A.bytes.data() + ab, bp, (size_t)blen) == 0;
                ok = (c.cmp == GX_CMP_NE || c.cmp == GX_CMP_NE_NULLSAFE) ? !eq : eq;
            } else if (A.type == GX_F64) {
                double bv = c.col_b < 0 ? c.v_f64 : B->f64v[ib];
                ok = cond_cmp_f64o(A.f64v[ia], c.cmp, bv);
            } else {
                int64_t av = A.type == GX_I32 ? (int64_t)A.i32v[ia] : A.i64v[ia];
                int64_t bv;
                if (c.col_b < 0) bv = c.v_i64;
                else bv = B->type == GX_I32 ? (int64_t)B->i32v[ib] : B->i64v[ib];
                ok = cond_cmp_i64o(av, c.cmp, bv);
            }
            if (!ok) return false;
        }
        return true;
    }

    int32_t match_init(int32_t h) const {
        int32_t m = table[hc_mix(h) & mask];
        return m;
    }

    /* emit one joined row into out columns. Column order per
     * AbstractJoinExec.buildJoinRow/buildRightJoinRow (java:175-227) and the
     * buildOuter overrides (ParallelHashJoinExec.java:234-287). */
    void emit_match(std::vector<OutCol> &out, const Store &probe_store,
                    size_t prow, int32_t bpos) {
        size_t col = 0;
        const size_t n_outer = outer_types.size();
        const size_t n_inner_out = cfg.single_join ? 1 : inner_types.size();
        if (cfg.join_type == GX_JOIN_SEMI || cfg.join_type == GX_JOIN_ANTI) {
            for (size_t i = 0; i < n_outer; i++)
                out[col++].append_from(probe_store.cols[i], prow);
            return;
        }
        if (cfg.join_type != GX_JOIN_RIGHT) {
            /* outer cols then inner cols */
            if (!cfg.build_outer) {
                for (size_t i = 0; i < n_outer; i++)
                    out[col++].append_from(probe_store.cols[i], prow);
                for (size_t i = 0; i < n_inner_out; i++)
                    out[col++].append_from(build.cols[i], (size_t)bpos);
            } else {
                for (size_t i = 0; i < n_outer; i++)
                    out[col++].append_from(build.cols[i], (size_t)bpos);
                for (size_t i = 0; i < n_inner_out; i++)
                    out[col++].append_from(probe_store.cols[i], prow);
            }
        } else {
            /* RIGHT: inner cols then outer cols */
            if (!cfg.build_outer) {
                for (size_t i = 0; i < inner_types.size(); i++)
                    out[col++].append_from(build.cols[i], (size_t)bpos);
                for (size_t i = 0; i < n_outer; i++)
                    out[col++].append_from(probe_store.cols[i], prow);
            } else {
                for (size_t i = 0; i < inner_types.size(); i++)
                    out[col++].append_from(probe_store.cols[i], prow);
                for (size_t i = 0; i < n_outer; i++)
                    out[col++].append_from(build.cols[i], (size_t)bpos);
            }
        }
    }

    /* LEFT/RIGHT unmatched probe row (AbstractJoinExec.buildLeftNullRow /
     * buildRightNullRow, java:203-227). Only reachable when !build_outer
     * (outputNullRowInTime). */
    void emit_null_row(std::vector<OutCol> &out, const Store &probe_store, size_t prow) {
        size_t col = 0;
        const size_t n_inner_out = cfg.single_join ? 1 : inner_types.size();
        if (cfg.join_type != GX_JOIN_RIGHT) {
            for (size_t i = 0; i < outer_types.size(); i++)
                out[col++].append_from(probe_store.cols[i], prow);
            for (size_t i = 0; i < n_inner_out; i++)
                out[col++].append_null();
        } else {
            for (size_t i = 0; i < inner_types.size(); i++)
                out[col++].append_null();
            for (size_t i = 0; i < outer_types.size(); i++)
                out[col++].append_from(probe_store.cols[i], prow);
        }
    }

    std::vector<int32_t> output_types() const {
        std::vector<int32_t> t;
        if (cfg.join_type == GX_JOIN_SEMI || cfg.join_type == GX_JOIN_ANTI)
            return outer_types;
        const std::vector<int32_t> inner_out = cfg.single_join
            ? std::vector<int32_t>(inner_types.begin(), inner_types.begin() + 1)
            : inner_types;
        if (cfg.join_type != GX_JOIN_RIGHT) {
            t = outer_types;
            t.insert(t.end(), inner_out.begin(), inner_out.end());
        } else {
            t = inner_types;
            t.insert(t.end(), outer_types.begin(), outer_types.end());
        }
        return t;
    }

    gx_result *project_result(std::vector<OutCol> &&cols, int32_t n_rows) {
        if (out_proj.empty()) return make_result(std::move(cols), n_rows);
        std::vector<OutCol> sel;
        sel.reserve(out_proj.size());
        for (int32_t idx : out_proj) sel.push_back(std::move(cols[idx]));
        return make_result(std::move(sel), n_rows);
    }

    int probe(const gx_chunk *ch, gx_result **out) {
        *out = nullptr;
        if (!built) { set_err("probe before build"); return -1; }

        std::vector<int32_t> otypes = output_types();
        /* probe side column types: outer unless build_outer */
        const auto &pt = cfg.build_outer ? inner_types : outer_types;
        Store probe_store;
        probe_store.init((int32_t)pt.size(), pt.data());
        if (probe_store.append(ch) != 0) return -1;

        std::vector<OutCol> cols(otypes.size());
        for (size_t c = 0; c < otypes.size(); c++) cols[c].type = otypes[c];

        if (pass_nothing) { *out = project_result(std::move(cols), 0); return 0; }
        if (pass_through) {
            /* ANTI with empty build: every probe row passes, no operand check
             * (AbstractBufferedJoinExec.doSpecialCheckForSemiJoin:296-301) */
            for (size_t r = 0; r < probe_store.n_rows; r++)
                emit_match(cols, probe_store, r, -1);
            *out = project_result(std::move(cols), (int32_t)probe_store.n_rows);
            return 0;
        }

        bool semi_join = (cfg.join_type == GX_JOIN_SEMI || cfg.join_type == GX_JOIN_ANTI)
                         && !cfg.single_join;
        bool outer_join = (cfg.join_type == GX_JOIN_LEFT || cfg.join_type == GX_JOIN_RIGHT);
        int32_t n_rows = 0;

        for (size_t r = 0; r < probe_store.n_rows; r++) {
            int32_t h = probe_store.row_hash(r, probe_key_cols);
            bool matched = false;
            int32_t m = match_init(h);
            for (; m != -1; m = links[m]) {
                if (!build.keys_equal((size_t)m, probe_store, r,
                                      build_key_cols, probe_key_cols))
                    continue;
                if (!conds.empty() && !cond_pass(probe_store, r, (size_t)m))
                    continue;
                /* NOTE: probe rows with a NULL key can never reach here —
                 * build NULL keys were never inserted and Block.equals(null,
                 * value) is false; null==null would match but null build keys
                 * are skipped (ExecUtils.buildOneChunk:933-941). */
                if (cfg.join_type == GX_JOIN_INNER || cfg.join_type == GX_JOIN_LEFT) {
                    if (cfg.single_join && matched) {
                        set_err("ERR_SCALAR_SUBQUERY_RETURN_MORE_THAN_ONE_ROW");
                        return -2;
                    }
                    emit_match(cols, probe_store, r, m);
                    n_rows++;
                } else if (cfg.join_type == GX_JOIN_RIGHT) {
                    emit_match(cols, probe_store, r, m);
                    n_rows++;
                }
                if (cfg.build_outer) matched_build[(size_t)m] = 1;
                matched = true;
                if (semi_join) break;
            }
            if (outer_join && !cfg.build_outer && !matched) {
                emit_null_row(cols, probe_store, r);
                n_rows++;
            }
            if (cfg.single_join && !semi_join && !matched
                && cfg.join_type == GX_JOIN_LEFT) {
                /* covered by outer_join branch above */
            }
            if (semi_join) {
                if (cfg.join_type == GX_JOIN_SEMI && matched) {
                    emit_match(cols, probe_store, r, -1);
                    n_rows++;
                } else if (cfg.join_type == GX_JOIN_ANTI && !matched) {
                    /* checkAntiJoinOperands: operand NULL suppresses the row
                     * (AbstractBufferedJoinExec.java:247-252,268-271) */
                    bool ok = true;
                    if (cfg.anti_null_col >= 0)
                        ok = !probe_store.cols[cfg.anti_null_col].is_null(r);
                    if (ok) {
                        emit_match(cols, probe_store, r, -1);
                        n_rows++;
                    }
                }
            }
        }
        *out = project_result(std::move(cols), n_rows);
        return 0;
    }

    /* drain buildOuter unmatched rows (ParallelHashJoinExec.nextJoinNullRows
     * :372-401): JoinRelType != RIGHT -> build cols + nulls; RIGHT -> nulls +
     * build cols. */
    int tail(gx_result **out) {
        *out = nullptr;
        if (!cfg.build_outer || tail_done) { tail_done = true; return 0; }
        std::vector<int32_t> otypes = output_types();
        std::vector<OutCol> cols(otypes.size());
        for (size_t c = 0; c < otypes.size(); c++) cols[c].type = otypes[c];
        int32_t n_rows = 0;
        const size_t n_inner_out = cfg.single_join ? 1 : inner_types.size();
        for (size_t pos = tail_cursor; pos < build.n_rows; pos++) {
            if (matched_build[pos]) continue;
            size_t col = 0;
            if (cfg.join_type != GX_JOIN_RIGHT) {
                for (size_t i = 0; i < outer_types.size(); i++)
                    cols[col++].append_from(build.cols[i], pos);
                for (size_t i = 0; i < n_inner_out; i++)
                    cols[col++].append_null();
            } else {
                for (size_t i = 0; i < inner_types.size(); i++)
                    cols[col++].append_null();
                for (size_t i = 0; i < outer_types.size(); i++)
                    cols[col++].append_from(build.cols[i], pos);
            }
            n_rows++;
        }
        tail_cursor = build.n_rows;
        tail_done = true;
        if (n_rows > 0) *out = project_result(std::move(cols), n_rows);
        return 0;
    }
};

/* ---- agg operator -------------------------------------------------------
 * Restates HashAggExec.consumeChunk (operator/HashAggExec.java:133-145) →
 * AggOpenHashMap.putChunk (operator/util/AggOpenHashMap.java:100-139) with
 * the GroupOpenHashMap open-addressing array path (GroupOpenHashMap.java:
 * 142-193: linear probe on mix(hash)&mask, x2 rehash at maxFill). */
struct AggOp : gx_op {
    gx_agg_cfg cfg;
    std::vector<int32_t> group_cols_, input_types;
    std::vector<gx_agg_spec> aggs;

    Store input;                     /* group-key columns only, appended per group */
    Store group_keys;                /* TypedBuffer: one row per group */
    std::vector<int> key_col_idx;    /* 0..n_group_cols-1 into group_keys */

    std::vector<int32_t> table;      /* open addressing: groupId or -1 */
    int64_t n_slots = 0, mask = 0, max_fill = 0, fill = 0;
    double load_f = 0.75;            /* GroupOpenHashMap DEFAULT_LOAD_FACTOR */

    /* aggregator states, one entry per group per agg */
    struct AggState { std::vector<int64_t> i64; std::vector<double> f64; std::vector<uint8_t> isnull; };
    std::vector<AggState> states;

    bool built = false;
    size_t emit_cursor = 0;
    static constexpr int32_t CHUNK_SIZE = 1000; /* ConnectionParams.java:1088 */

    AggOp(const gx_agg_cfg *c) : gx_op(OP_AGG), cfg(*c) {
        group_cols_.assign(c->group_cols, c->group_cols + c->n_group_cols);
        input_types.assign(c->input_types, c->input_types + c->n_input_cols);
        aggs.assign(c->aggs, c->aggs + c->n_aggs);

        std::vector<int32_t> key_types;
        for (int32_t gc : group_cols_) key_types.push_back(input_types[gc]);
        group_keys.init((int32_t)key_types.size(), key_types.data());
        for (size_t k = 0; k < key_types.size(); k++) key_col_idx.push_back((int)k);

        int64_t expected = cfg.expected_groups > 0 ? cfg.expected_groups : 1024;
        n_slots = hc_array_size(expected, load_f);
        mask = n_slots - 1;
        max_fill = hc_max_fill(n_slots, load_f);
        table.assign((size_t)n_slots, -1);

        states.resize(aggs.size());
        if (cfg.n_group_cols == 0) append_group_direct(); /* noGroupBy: one group */
    }

    int32_t n_groups() const { return (int32_t)(states.empty()
        ? group_keys.n_rows : states[0].isnull.size()); }

    void append_group_direct() {
        for (size_t a = 0; a < aggs.size(); a++) {
            AggState &s = states[a];
            switch (aggs[a].func) {
            case GX_AGG_COUNT_ROW: case GX_AGG_COUNT_COL:
                s.i64.push_back(0); s.isnull.push_back(0); break;
            case GX_AGG_SUM_I64:
                s.i64.push_back(0); s.isnull.push_back(0); break; /* Long2LongSum0 init 0 */
            case GX_AGG_SUM_I64N:
                s.i64.push_back(0); s.isnull.push_back(1); break; /* Sum: init NULL */
            case GX_AGG_SUM_F64: case GX_AGG_MIN_F64: case GX_AGG_MAX_F64:
                s.f64.push_back(0); s.isnull.push_back(1); break; /* init NULL */
            case GX_AGG_AVG_F64:
                s.f64.push_back(0); s.i64.push_back(0);
                s.isnull.push_back(1); break; /* {sum, count}; NULL at 0 */
            case GX_AGG_BIT_AND:
                s.i64.push_back(-1); s.isnull.push_back(0); break;
            case GX_AGG_BIT_OR: case GX_AGG_BIT_XOR:
                s.i64.push_back(0); s.isnull.push_back(0); break;
            case GX_AGG_MIN_I64: case GX_AGG_MAX_I64:
                s.i64.push_back(0); s.isnull.push_back(1); break; /* init NULL */
            }
        }
    }

    /* GroupOpenHashMap.doInnerPutArray (java:142-169) + rehash (:171-187).
     * 'from' = store+row providing the key (input chunk or group_keys on
     * rehash); group id -1 = allocate. */
    int32_t inner_put(const Store &from, size_t row, const std::vector<int> &from_cols,
                      int32_t group_id) {
        int32_t h = (int32_t)(hc_mix(from.row_hash(row, from_cols)) & (int32_t)mask);
        while (true) {
            int32_t k = table[(size_t)h];
            if (k == -1) break;
            if (group_keys.keys_equal((size_t)k, from, row, key_col_idx, from_cols))
                return k;
            h = (int32_t)((h + 1) & mask);
        }
        if (group_id == -1) {
            /* appendGroup: copy key into TypedBuffer, assign next id */
            for (size_t kc = 0; kc < from_cols.size(); kc++) {
                const Column &src = from.cols[from_cols[kc]];
                Column &dst = group_keys.cols[kc];
                dst.null_.push_back(src.is_null(row) ? 1 : 0);
                switch (src.type) {
                case GX_I64: dst.i64v.push_back(src.is_null(row) ? 0 : src.i64v[row]); break;
                case GX_I32: dst.i32v.push_back(src.is_null(row) ? 0 : src.i32v[row]); break;
                case GX_F64: dst.f64v.push_back(src.is_null(row) ? 0 : src.f64v[row]); break;
                case GX_SLICE: {
                    if (!src.is_null(row)) {
                        int32_t b = src.begin_off(row), e = src.off[row];
                        dst.bytes.insert(dst.bytes.end(), src.bytes.data() + b, src.bytes.data() + e);
                    }
                    dst.off.push_back((int32_t)dst.bytes.size());
                    break; }
                }
            }
            group_keys.n_rows++;
            group_id = (int32_t)group_keys.n_rows - 1;
            append_group_direct();
        }
        table[(size_t)h] = group_id;
        if (fill++ >= max_fill) rehash();
        return group_id;
    }

    void rehash() {
        n_slots *= 2; mask = n_slots - 1;
        max_fill = hc_max_fill(n_slots, load_f);
        fill = 0;
        table.assign((size_t)n_slots, -1);
        std::vector<int> kcols(key_col_idx.begin(), key_col_idx.end());
        for (size_t g = 0; g < group_keys.n_rows; g++)
            inner_put(group_keys, g, kcols, (int32_t)g);
    }

    void accumulate(size_t a, int32_t gid, const Store &in, size_t row) {
        const gx_agg_spec &sp = aggs[a];
        AggState &s = states[a];
        const Column *c = sp.input_col >= 0 ? &in.cols[sp.input_col] : nullptr;
        switch (sp.func) {
        case GX_AGG_COUNT_ROW: s.i64[gid]++; break;
        case GX_AGG_COUNT_COL: if (!c->is_null(row)) s.i64[gid]++; break;
        case GX_AGG_SUM_I64:
            if (!c->is_null(row))
                s.i64[gid] = (int64_t)((uint64_t)s.i64[gid] +
                    (uint64_t)(c->type == GX_I32 ? (int64_t)c->i32v[row] : c->i64v[row]));
            break;
        case GX_AGG_SUM_I64N: /* null-init Sum (AggregateUtils SqlKind.SUM) */
            if (!c->is_null(row)) {
                s.i64[gid] = (int64_t)((uint64_t)s.i64[gid] +
                    (uint64_t)(c->type == GX_I32 ? (int64_t)c->i32v[row] : c->i64v[row]));
                s.isnull[gid] = 0;
            }
            break;
        case GX_AGG_SUM_F64:
            if (!c->is_null(row)) {
                double v = c->type == GX_F64 ? c->f64v[row]
                         : c->type == GX_I32 ? (double)c->i32v[row] : (double)c->i64v[row];
                if (s.isnull[gid]) { s.f64[gid] = v; s.isnull[gid] = 0; }
                else s.f64[gid] += v;
            }
            break;
        case GX_AGG_BIT_AND: case GX_AGG_BIT_OR: case GX_AGG_BIT_XOR:
            if (!c->is_null(row)) {
                int64_t v = c->type == GX_I32 ? (int64_t)c->i32v[row]
                                              : c->i64v[row];
                if (sp.func == GX_AGG_BIT_AND) s.i64[gid] &= v;
                else if (sp.func == GX_AGG_BIT_OR) s.i64[gid] |= v;
                else s.i64[gid] ^= v;
            }
            break;
        case GX_AGG_AVG_F64:
            if (!c->is_null(row)) {
                double v = c->type == GX_F64 ? c->f64v[row]
                         : c->type == GX_I32 ? (double)c->i32v[row]
                                             : (double)c->i64v[row];
                s.f64[gid] += v;
                s.i64[gid]++;
                s.isnull[gid] = 0;
            }
            break;
        case GX_AGG_MIN_I64: case GX_AGG_MAX_I64:
            if (!c->is_null(row)) {
                int64_t v = c->type == GX_I32 ? (int64_t)c->i32v[row] : c->i64v[row];
                if (s.isnull[gid]) { s.i64[gid] = v; s.isnull[gid] = 0; }
                else s.i64[gid] = sp.func == GX_AGG_MIN_I64 ? std::min(s.i64[gid], v)
                                                            : std::max(s.i64[gid], v);
            }
            break;
        case GX_AGG_MIN_F64: case GX_AGG_MAX_F64:
            if (!c->is_null(row)) {
                double v = c->f64v[row];
                if (s.isnull[gid]) { s.f64[gid] = v; s.isnull[gid] = 0; }
                else s.f64[gid] = sp.func == GX_AGG_MIN_F64
                        ? jmin_f64o(s.f64[gid], v) : jmax_f64o(s.f64[gid], v);
            }
            break;
        }
    }

    int consume(const gx_chunk *ch) {
        Store in;
        in.init((int32_t)input_types.size(), input_types.data());
        if (in.append(ch) != 0) return -1;
        std::vector<int> gcols(group_cols_.begin(), group_cols_.end());
        for (size_t r = 0; r < in.n_rows; r++) {
            int32_t gid = cfg.n_group_cols == 0 ? 0 : inner_put(in, r, gcols, -1);
            for (size_t a = 0; a < aggs.size(); a++) accumulate(a, gid, in, r);
        }
        return 0;
    }

    int next(gx_result **out) {
        *out = nullptr;
        int32_t total = n_groups();
        if ((int32_t)emit_cursor >= total) return 0;
        int32_t n = std::min<int32_t>(CHUNK_SIZE, total - (int32_t)emit_cursor);
        std::vector<int32_t> otypes;
        for (int32_t gc : group_cols_) otypes.push_back(input_types[gc]);
        for (auto &sp : aggs) {
            switch (sp.func) {
            case GX_AGG_COUNT_ROW: case GX_AGG_COUNT_COL:
            case GX_AGG_SUM_I64: case GX_AGG_SUM_I64N:
            case GX_AGG_MIN_I64: case GX_AGG_MAX_I64:
            case GX_AGG_BIT_AND: case GX_AGG_BIT_OR: case GX_AGG_BIT_XOR:
            case GX_AGG_RANK: case GX_AGG_DENSE_RANK:
                otypes.push_back(GX_I64); break;
            default: otypes.push_back(GX_F64); break;
            }
        }
        std::vector<OutCol> cols(otypes.size());
        for (size_t c = 0; c < otypes.size(); c++) cols[c].type = otypes[c];
        for (int32_t i = 0; i < n; i++) {
            size_t g = emit_cursor + (size_t)i;
            size_t col = 0;
            for (size_t kc = 0; kc < group_cols_.size(); kc++)
                cols[col++].append_from(group_keys.cols[kc], g);
            for (size_t a = 0; a < aggs.size(); a++) {
                AggState &s = states[a];
                bool is_i64 = otypes[col] == GX_I64;
                if (s.isnull[g] &&
                    !(aggs[a].func == GX_AGG_COUNT_ROW || aggs[a].func == GX_AGG_COUNT_COL
                      || aggs[a].func == GX_AGG_SUM_I64
                      || aggs[a].func == GX_AGG_BIT_AND
                      || aggs[a].func == GX_AGG_BIT_OR
                      || aggs[a].func == GX_AGG_BIT_XOR))
                    cols[col++].append_null();
                else if (aggs[a].func == GX_AGG_AVG_F64)
                    cols[col++].append_f64(s.f64[g] / (double)s.i64[g]);
                else if (is_i64) cols[col++].append_i64(s.i64[g]);
                else cols[col++].append_f64(s.f64[g]);
            }
        }
        emit_cursor += (size_t)n;
        *out = make_result(std::move(cols), n);
        return 0;
    }
};


/* ---- fused group-join ---------------------------------------------------
 * Restates HashGroupJoinExec (operator/HashGroupJoinExec.java): the
 * CONSUMED side's rows are the groups (one per position, buildOneChunk
 * :296-311 puts EVERY row — no null-key skip, unlike ExecUtils
 * .buildOneChunk — and matching goes through Chunk.equals, so NULL keys
 * match NULL keys, null-safe); each matching probe row accumulates into
 * that position's aggregators (buildJoinRow:469-490 -> doAggregate).
 * INNER emits matched positions; LEFT emits all, unmatched ones after one
 * null-row accumulation (buildNullRow via doNextChunk:324-330): COUNT(*)
 * counts the null row, COUNT(col)/SUM/MIN/MAX see NULL. Emission pairing
 * is each group's own values (see include/gxop.h note on the reference's
 * buildValueChunks counter mismatch, HashGroupJoinExec.java:410-451). */
struct GroupJoinOp : gx_op {
    gx_groupjoin_cfg cfg;
    std::vector<gx_equi_key> keys;
    std::vector<int32_t> build_types, probe_types, group_cols_;
    std::vector<gx_agg_spec> aggs;
    Store build;
    std::vector<int> build_key_cols, probe_key_cols;
    std::vector<int32_t> table, links;
    int32_t mask = 0;
    std::vector<uint8_t> used;
    struct AggState { std::vector<int64_t> i64; std::vector<double> f64;
                      std::vector<uint8_t> isnull; };
    std::vector<AggState> states;
    bool built = false, pass_nothing = false;
    size_t emit_cursor = 0;
    static constexpr int32_t CHUNK_SIZE = 1000;

    GroupJoinOp(const gx_groupjoin_cfg *c) : gx_op(OP_GROUPJOIN), cfg(*c) {
        keys.assign(c->keys, c->keys + c->n_keys);
        build_types.assign(c->build_types, c->build_types + c->n_build_cols);
        probe_types.assign(c->probe_types, c->probe_types + c->n_probe_cols);
        group_cols_.assign(c->group_cols, c->group_cols + c->n_group_cols);
        aggs.assign(c->aggs, c->aggs + c->n_aggs);
        for (auto &k : keys) {
            build_key_cols.push_back(k.outer_index);  /* consumed = "outer" */
            probe_key_cols.push_back(k.inner_index);
        }
        build.init((int32_t)build_types.size(), build_types.data());
    }

    int consume(const gx_chunk *ch) { return build.append(ch); }

    int do_build() {
        if (built) return 0; /* first-come barrier */
        const int64_t size = (int64_t)build.n_rows;
        if (size == 0 && cfg.join_type == GX_JOIN_INNER) pass_nothing = true;
        int64_t n = hc_array_size(std::max<int64_t>(size, 1),
                                  join_load_factor(size));
        mask = (int32_t)(n - 1);
        table.assign((size_t)n, -1);
        links.assign((size_t)size, -1);
        for (size_t pos = 0; pos < build.n_rows; pos++) {
            /* every position inserted; nulls hash per Block.hashCode */
            int32_t h = build.row_hash(pos, build_key_cols);
            int32_t slot = hc_mix(h) & mask;
            links[pos] = table[slot];
            table[slot] = (int32_t)pos;
        }
        used.assign(build.n_rows, 0);
        states.resize(aggs.size());
        for (size_t a = 0; a < aggs.size(); a++) {
            AggState &s = states[a];
            s.i64.assign(build.n_rows, 0);
            s.f64.assign(build.n_rows, 0);
            bool init_null = !(aggs[a].func == GX_AGG_COUNT_ROW ||
                               aggs[a].func == GX_AGG_COUNT_COL ||
                               aggs[a].func == GX_AGG_SUM_I64 ||
                               aggs[a].func == GX_AGG_BIT_AND ||
                               aggs[a].func == GX_AGG_BIT_OR ||
                               aggs[a].func == GX_AGG_BIT_XOR);
            s.isnull.assign(build.n_rows, init_null ? 1 : 0);
            if (aggs[a].func == GX_AGG_MIN_I64)
                s.i64.assign(build.n_rows, INT64_MAX);
            if (aggs[a].func == GX_AGG_MAX_I64)
                s.i64.assign(build.n_rows, INT64_MIN);
            if (aggs[a].func == GX_AGG_BIT_AND)
                s.i64.assign(build.n_rows, -1);
        }
        built = true;
        return 0;
    }

    void accumulate(size_t a, int32_t gid, const Store &in, size_t row) {
        const gx_agg_spec &sp = aggs[a];
        AggState &s = states[a];
        const Column *c = sp.input_col >= 0 ? &in.cols[sp.input_col] : nullptr;
        switch (sp.func) {
        case GX_AGG_COUNT_ROW: s.i64[gid]++; break;
        case GX_AGG_COUNT_COL: if (!c->is_null(row)) s.i64[gid]++; break;
        case GX_AGG_SUM_I64:
            if (!c->is_null(row))
                s.i64[gid] = (int64_t)((uint64_t)s.i64[gid] +
                    (uint64_t)(c->type == GX_I32 ? (int64_t)c->i32v[row]
                                                 : c->i64v[row]));
            break;
        case GX_AGG_SUM_F64:
            if (!c->is_null(row)) {
                double v = c->type == GX_F64 ? c->f64v[row]
                         : c->type == GX_I32 ? (double)c->i32v[row]
                                             : (double)c->i64v[row];
                if (s.isnull[gid]) { s.f64[gid] = v; s.isnull[gid] = 0; }
                else s.f64[gid] += v;
            }
            break;
        case GX_AGG_BIT_AND: case GX_AGG_BIT_OR: case GX_AGG_BIT_XOR:
            if (!c->is_null(row)) {
                int64_t v = c->type == GX_I32 ? (int64_t)c->i32v[row]
                                              : c->i64v[row];
                if (sp.func == GX_AGG_BIT_AND) s.i64[gid] &= v;
                else if (sp.func == GX_AGG_BIT_OR) s.i64[gid] |= v;
                else s.i64[gid] ^= v;
            }
            break;
        case GX_AGG_AVG_F64:
            if (!c->is_null(row)) {
                double v = c->type == GX_F64 ? c->f64v[row]
                         : c->type == GX_I32 ? (double)c->i32v[row]
                                             : (double)c->i64v[row];
                s.f64[gid] += v;
                s.i64[gid]++;
                s.isnull[gid] = 0;
            }
            break;
        case GX_AGG_MIN_I64: case GX_AGG_MAX_I64:
            if (!c->is_null(row)) {
                int64_t v = c->type == GX_I32 ? (int64_t)c->i32v[row]
                                              : c->i64v[row];
                if (s.isnull[gid]) { s.i64[gid] = v; s.isnull[gid] = 0; }
                else s.i64[gid] = sp.func == GX_AGG_MIN_I64
                        ? std::min(s.i64[gid], v) : std::max(s.i64[gid], v);
            }
            break;
        case GX_AGG_MIN_F64: case GX_AGG_MAX_F64:
            if (!c->is_null(row)) {
                double v = c->f64v[row];
                if (s.isnull[gid]) { s.f64[gid] = v; s.isnull[gid] = 0; }
                else s.f64[gid] = sp.func == GX_AGG_MIN_F64
                        ? jmin_f64o(s.f64[gid], v) : jmax_f64o(s.f64[gid], v);
            }
            break;
        }
    }

    int probe(const gx_chunk *ch) {
        if (!built) { set_err("probe before build"); return -1; }
        if (pass_nothing) return 0;
        Store in;
        in.init((int32_t)probe_types.size(), probe_types.data());
        if (in.append(ch) != 0) return -1;
        for (size_t r = 0; r < in.n_rows; r++) {
            int32_t h = in.row_hash(r, probe_key_cols);
            for (int32_t m = table[(size_t)(hc_mix(h) & mask)]; m != -1;
                 m = links[(size_t)m]) {
                if (!build.keys_equal((size_t)m, in, r, build_key_cols,
                                      probe_key_cols))
                    continue;
                used[(size_t)m] = 1;
                for (size_t a = 0; a < aggs.size(); a++)
                    accumulate(a, m, in, r);
            }
        }
        return 0;
    }

    int next(gx_result **out) {
        *out = nullptr;
        if (pass_nothing) return 0;
        std::vector<int32_t> otypes;
        for (int32_t gc : group_cols_) otypes.push_back(build_types[gc]);
        for (auto &sp : aggs) {
            switch (sp.func) {
            case GX_AGG_COUNT_ROW: case GX_AGG_COUNT_COL:
            case GX_AGG_SUM_I64: case GX_AGG_SUM_I64N:
            case GX_AGG_MIN_I64: case GX_AGG_MAX_I64:
            case GX_AGG_BIT_AND: case GX_AGG_BIT_OR: case GX_AGG_BIT_XOR:
            case GX_AGG_RANK: case GX_AGG_DENSE_RANK:
                otypes.push_back(GX_I64); break;
            default: otypes.push_back(GX_F64); break;
            }
        }
        std::vector<OutCol> cols(otypes.size());
        for (size_t c = 0; c < otypes.size(); c++) cols[c].type = otypes[c];
        int32_t n = 0;
        const bool emit_all = cfg.join_type == GX_JOIN_LEFT;
        while (emit_cursor < build.n_rows && n < CHUNK_SIZE) {
            size_t g = emit_cursor++;
            bool matched = used[g] != 0;
            if (!matched && !emit_all) continue;
            size_t col = 0;
            for (int32_t gc : group_cols_)
                cols[col++].append_from(build.cols[gc], g);
            for (size_t a = 0; a < aggs.size(); a++) {
                AggState &s = states[a];
                int64_t i64v = s.i64[g];
                bool nullv = s.isnull[g] != 0;
                if (!matched && aggs[a].func == GX_AGG_COUNT_ROW)
                    i64v += 1;  /* buildNullRow: null row counts */
                if (nullv && !(aggs[a].func == GX_AGG_COUNT_ROW ||
                               aggs[a].func == GX_AGG_COUNT_COL ||
                               aggs[a].func == GX_AGG_SUM_I64))
                    cols[col++].append_null();
                else if (aggs[a].func == GX_AGG_AVG_F64)
                    cols[col++].append_f64(s.f64[g] / (double)s.i64[g]);
                else if (otypes[group_cols_.size() + a] == GX_I64)
                    cols[col++].append_i64(i64v);
                else
                    cols[col++].append_f64(s.f64[g]);
            }
            n++;
        }
        if (n == 0) return 0;
        *out = make_result(std::move(cols), n);
        return 0;
    }
};


/* ---- window: running aggregates over partition-sorted input -------------
 * Restates NonFrameOverWindowExec (operator/NonFrameOverWindowExec.java:
 * 34-160): per row, accumulate into ONE running aggregator slot, resetting
 * when the partition-key row differs from the previous row
 * (isDifferentPartition:136-145, null-safe) or when reset[a] is set
 * (resetAccumulators / CURRENT ROW mode); emit the running value after
 * each accumulate (processFirstLine/doNextChunk:80-120). Streaming: carry
 * (last partition key + running state) survives across chunks. */
struct WindowOp : gx_op {
    gx_window_cfg cfg;
    std::vector<int32_t> part_cols_, order_cols_, carry_cols_, input_types;
    std::vector<gx_agg_spec> aggs;
    std::vector<uint8_t> reset_;

    bool have_carry = false;
    Store carry_key;   /* 1 row: LAST ROW's partition + order key cols */
    struct RunState { int64_t i64 = 0; double f64 = 0; uint8_t isnull = 0; };
    std::vector<RunState> run;
    int64_t part_count = 0;   /* rows since partition start (Rank.count) */

    WindowOp(const gx_window_cfg *c) : gx_op(OP_WINDOW), cfg(*c) {
        part_cols_.assign(c->part_cols, c->part_cols + c->n_part_cols);
        if (c->n_order_cols > 0)
            order_cols_.assign(c->order_cols, c->order_cols + c->n_order_cols);
        input_types.assign(c->input_types, c->input_types + c->n_input_cols);
        aggs.assign(c->aggs, c->aggs + c->n_aggs);
        reset_.assign(c->reset, c->reset + c->n_aggs);
        carry_cols_ = part_cols_;
        carry_cols_.insert(carry_cols_.end(), order_cols_.begin(),
                           order_cols_.end());
        std::vector<int32_t> kt;
        for (int32_t pc : carry_cols_) kt.push_back(input_types[pc]);
        carry_key.init((int32_t)kt.size(), kt.data());
        run.resize(aggs.size());
    }

    void reset_state(size_t a) {
        RunState &s = run[a];
        switch (aggs[a].func) {
        case GX_AGG_COUNT_ROW: case GX_AGG_COUNT_COL: case GX_AGG_SUM_I64:
        case GX_AGG_BIT_OR: case GX_AGG_BIT_XOR:
            s.i64 = 0; s.isnull = 0; break;
        case GX_AGG_BIT_AND:
            s.i64 = -1; s.isnull = 0; break;
        default: s.i64 = 0; s.f64 = 0; s.isnull = 1; break; /* incl. AVG:
            i64 = count, f64 = sum */
        }
    }

    void acc(size_t a, const Store &in, size_t r) {
        const gx_agg_spec &sp = aggs[a];
        RunState &s = run[a];
        const Column *c = sp.input_col >= 0 ? &in.cols[sp.input_col] : nullptr;
        switch (sp.func) {
        case GX_AGG_COUNT_ROW: s.i64++; break;
        case GX_AGG_COUNT_COL: if (!c->is_null(r)) s.i64++; break;
        case GX_AGG_SUM_I64:
            if (!c->is_null(r))
                s.i64 = (int64_t)((uint64_t)s.i64 + (uint64_t)(
                    c->type == GX_I32 ? (int64_t)c->i32v[r] : c->i64v[r]));
            break;
        case GX_AGG_SUM_I64N: /* null-init running Sum */
            if (!c->is_null(r)) {
                s.i64 = (int64_t)((uint64_t)s.i64 + (uint64_t)(
                    c->type == GX_I32 ? (int64_t)c->i32v[r] : c->i64v[r]));
                s.isnull = 0;
            }
            break;
        case GX_AGG_SUM_F64:
            if (!c->is_null(r)) {
                double v = c->type == GX_F64 ? c->f64v[r]
                         : c->type == GX_I32 ? (double)c->i32v[r]
                                             : (double)c->i64v[r];
                if (s.isnull) { s.f64 = v; s.isnull = 0; } else s.f64 += v;
            }
            break;
        case GX_AGG_MIN_I64: case GX_AGG_MAX_I64:
            if (!c->is_null(r)) {
                int64_t v = c->type == GX_I32 ? (int64_t)c->i32v[r]
                                              : c->i64v[r];
                if (s.isnull) { s.i64 = v; s.isnull = 0; }
                else s.i64 = sp.func == GX_AGG_MIN_I64 ? std::min(s.i64, v)
                                                       : std::max(s.i64, v);
            }
            break;
        case GX_AGG_MIN_F64: case GX_AGG_MAX_F64:
            if (!c->is_null(r)) {
                double v = c->f64v[r];
                if (s.isnull) { s.f64 = v; s.isnull = 0; }
                else s.f64 = sp.func == GX_AGG_MIN_F64 ? jmin_f64o(s.f64, v)
                                                       : jmax_f64o(s.f64, v);
            }
            break;
        case GX_AGG_AVG_F64:
            if (!c->is_null(r)) {
                double v = c->type == GX_F64 ? c->f64v[r]
                         : c->type == GX_I32 ? (double)c->i32v[r]
                                             : (double)c->i64v[r];
                s.f64 += v; s.i64++; s.isnull = 0;
            }
            break;
        case GX_AGG_BIT_AND: case GX_AGG_BIT_OR: case GX_AGG_BIT_XOR:
            if (!c->is_null(r)) {
                int64_t v = c->type == GX_I32 ? (int64_t)c->i32v[r]
                                              : c->i64v[r];
                if (sp.func == GX_AGG_BIT_AND) s.i64 &= v;
                else if (sp.func == GX_AGG_BIT_OR) s.i64 |= v;
                else s.i64 ^= v;
            }
            break;
        }
    }

    bool same_partition(const Store &in, size_t r) {
        if (!have_carry) return false;
        std::vector<int> kidx;
        for (size_t k = 0; k < part_cols_.size(); k++) kidx.push_back((int)k);
        std::vector<int> pcols(part_cols_.begin(), part_cols_.end());
        return carry_key.keys_equal(0, in, r, kidx, pcols);
    }

    /* previous row (carry) has the same ORDER-BY values (Rank.sameRank) */
    bool same_run(const Store &in, size_t r) {
        if (!have_carry) return false;
        std::vector<int> kidx;
        for (size_t k = 0; k < order_cols_.size(); k++)
            kidx.push_back((int)(part_cols_.size() + k));
        std::vector<int> ocols(order_cols_.begin(), order_cols_.end());
        return carry_key.keys_equal(0, in, r, kidx, ocols);
    }

    void save_carry(const Store &in, size_t r) {
        for (size_t k = 0; k < carry_cols_.size(); k++) {
            Column &dst = carry_key.cols[k];
            const Column &src = in.cols[carry_cols_[k]];
            dst.i64v.clear(); dst.i32v.clear(); dst.f64v.clear();
            dst.off.clear(); dst.bytes.clear(); dst.null_.clear();
            dst.null_.push_back(src.is_null(r) ? 1 : 0);
            switch (src.type) {
            case GX_I64: dst.i64v.push_back(src.is_null(r) ? 0 : src.i64v[r]); break;
            case GX_I32: dst.i32v.push_back(src.is_null(r) ? 0 : src.i32v[r]); break;
            case GX_F64: dst.f64v.push_back(src.is_null(r) ? 0 : src.f64v[r]); break;
            case GX_SLICE: {
                if (!src.is_null(r)) {
                    int32_t b = src.begin_off(r), e = src.off[r];
                    dst.bytes.assign(src.bytes.data() + b, src.bytes.data() + e);
                }
                dst.off.push_back((int32_t)dst.bytes.size());
                break; }
            }
        }
        carry_key.n_rows = 1;
        have_carry = true;
    }

    int consume(const gx_chunk *ch, gx_result **out) {
        *out = nullptr;
        Store in;
        in.init((int32_t)input_types.size(), input_types.data());
        if (in.append(ch) != 0) return -1;

        std::vector<int32_t> otypes = input_types;
        for (auto &sp : aggs) {
            switch (sp.func) {
            case GX_AGG_COUNT_ROW: case GX_AGG_COUNT_COL:
            case GX_AGG_SUM_I64: case GX_AGG_SUM_I64N:
            case GX_AGG_MIN_I64: case GX_AGG_MAX_I64:
            case GX_AGG_BIT_AND: case GX_AGG_BIT_OR: case GX_AGG_BIT_XOR:
            case GX_AGG_RANK: case GX_AGG_DENSE_RANK:
                otypes.push_back(GX_I64); break;
            default: otypes.push_back(GX_F64); break;
            }
        }
        std::vector<OutCol> cols(otypes.size());
        for (size_t c = 0; c < otypes.size(); c++) cols[c].type = otypes[c];

        for (size_t r = 0; r < in.n_rows; r++) {
            bool change = !same_partition(in, r);
            bool run_change = change || !same_run(in, r);
            save_carry(in, r);
            if (change) part_count = 0;
            part_count++;
            for (size_t c = 0; c < input_types.size(); c++)
                cols[c].append_from(in.cols[c], r);
            for (size_t a = 0; a < aggs.size(); a++) {
                size_t col = input_types.size() + a;
                RunState &s = run[a];
                if (aggs[a].func == GX_AGG_RANK ||
                    aggs[a].func == GX_AGG_DENSE_RANK) {
                    /* Rank.accumulate:40-47 / DenseRank; resetAccumulators
                     * makes every row rank 1 (count and lastRow reset) */
                    if (reset_[a]) { cols[col].append_i64(1); continue; }
                    if (change) s.i64 = 0;
                    if (run_change)
                        s.i64 = aggs[a].func == GX_AGG_RANK ? part_count
                                                            : s.i64 + 1;
                    cols[col].append_i64(s.i64);
                    continue;
                }
                if (reset_[a] || change) reset_state(a);
                acc(a, in, r);
                bool i64out = otypes[col] == GX_I64;
                if (s.isnull) cols[col].append_null();
                else if (aggs[a].func == GX_AGG_AVG_F64)
                    cols[col].append_f64(s.f64 / (double)s.i64);
                else if (i64out) cols[col].append_i64(s.i64);
                else cols[col].append_f64(s.f64);
            }
        }
        *out = make_result(std::move(cols), (int32_t)in.n_rows);
        return 0;
    }
};


/* ---- frame windows ------------------------------------------------------
 * Restates OverWindowFramesExec + operator/frame/ UnboundedOverFrame /
 * RowSlidingOverFrame / RowUnboundedFollowingOverFrame: buffer input,
 * then per partition segment compute each frame's value per row.
 * Emission preserves input order (the operator appends window columns). */
struct FWindowOp : gx_op {
    gx_fwindow_cfg cfg;
    std::vector<int32_t> part_cols_, order_cols_, input_types;
    std::vector<gx_frame_spec> frames;
    Store input;
    bool finished = false;
    size_t emit_cursor = 0;
    std::vector<size_t> seg_start_of_row;   /* first row of my segment */
    std::vector<size_t> seg_end_of_row;     /* one past last row */
    std::vector<size_t> run_start_of_row;   /* ORDER-BY run bounds */
    std::vector<size_t> run_end_of_row;     /* INCLUSIVE last row of run */
    static constexpr int32_t CHUNK_SIZE = 1000;

    FWindowOp(const gx_fwindow_cfg *c) : gx_op(OP_FWINDOW), cfg(*c) {
        part_cols_.assign(c->part_cols, c->part_cols + c->n_part_cols);
        if (c->n_order_cols > 0)
            order_cols_.assign(c->order_cols, c->order_cols + c->n_order_cols);
        input_types.assign(c->input_types, c->input_types + c->n_input_cols);
        frames.assign(c->frames, c->frames + c->n_frames);
        input.init((int32_t)input_types.size(), input_types.data());
    }

    int consume(const gx_chunk *ch) { return input.append(ch); }

    bool same_part(size_t a, size_t b) const {
        std::vector<int> pc(part_cols_.begin(), part_cols_.end());
        return input.keys_equal(a, input, b, pc, pc);
    }

    int finish() {
        const size_t n = input.n_rows;
        seg_start_of_row.resize(n);
        seg_end_of_row.resize(n);
        size_t start = 0;
        for (size_t i = 0; i < n; i++) {
            if (i > 0 && !same_part(i - 1, i)) {
                for (size_t j = start; j < i; j++) seg_end_of_row[j] = i;
                start = i;
            }
            seg_start_of_row[i] = start;
        }
        for (size_t j = start; j < n; j++) seg_end_of_row[j] = n;
        if (!order_cols_.empty()) {
            std::vector<int> oc(order_cols_.begin(), order_cols_.end());
            run_start_of_row.resize(n);
            run_end_of_row.resize(n);
            size_t rs = 0;
            for (size_t i = 0; i < n; i++) {
                if (i > 0 && (seg_start_of_row[i] != seg_start_of_row[i - 1] ||
                              !input.keys_equal(i - 1, input, i, oc, oc))) {
                    for (size_t j = rs; j < i; j++) run_end_of_row[j] = i - 1;
                    rs = i;
                }
                run_start_of_row[i] = rs;
            }
            for (size_t j = rs; j < n; j++) run_end_of_row[j] = n - 1;
        }
        finished = true;
        return 0;
    }

    /* RANGE frame index bounds, restated with the reference's own linear
     * scans (RangeSlidingOverFrame.getBound:117-138,
     * RangeUnboundedPrecedingOverFrame.getBound:144-168,
     * RangeUnboundedFollowingOverFrame.getBound:133-155; null-run
     * handling per updateIndex/updateNullRows in each class — nulls sort
     * first when ascending, last when descending). Returns INCLUSIVE
     * [lo, hi]; hi < lo = empty frame. */
    void range_bounds(const gx_frame_spec &f, size_t i, size_t s, size_t e,
                      int64_t &lo, int64_t &hi) const {
        const Column &oc = input.cols[f.order_col];
        const bool asc = f.order_asc != 0;
        const int64_t dir = asc ? 1 : -1;
        auto is_null = [&](int64_t j) { return oc.is_null((size_t)j); };
        /* value(a) <= value(b) + range — the compare() ladder
         * (RangeSlidingOverFrame.java:159-186, numeric types) */
        auto le_plus = [&](int64_t a, int64_t b, int64_t range) {
            if (oc.type == GX_F64)
                return oc.f64v[a] <= oc.f64v[b] + (double)range;
            int64_t va = oc.type == GX_I32 ? (int64_t)oc.i32v[a] : oc.i64v[a];
            int64_t vb = oc.type == GX_I32 ? (int64_t)oc.i32v[b] : oc.i64v[b];
            return va <= vb + range;
        };
        if (is_null((int64_t)i)) {
            int64_t rs = (int64_t)i, re = (int64_t)i;
            while (rs - 1 >= (int64_t)s && is_null(rs - 1)) rs--;
            while (re + 1 < (int64_t)e && is_null(re + 1)) re++;
            switch (f.kind) {
            case GX_FRAME_RANGE_SLIDING: lo = rs; hi = re; break;
            case GX_FRAME_RANGE_UNBOUNDED_PRECEDING:
                lo = (int64_t)s; hi = asc ? re : (int64_t)e - 1; break;
            default: /* RANGE_UNBOUNDED_FOLLOWING */
                lo = asc ? (int64_t)s : rs; hi = (int64_t)e - 1; break;
            }
            return;
        }
        switch (f.kind) {
        case GX_FRAME_RANGE_SLIDING: {
            int64_t hiI = (int64_t)i;
            while (hiI >= (int64_t)s && hiI < (int64_t)e && !is_null(hiI) &&
                   le_plus(hiI, (int64_t)i, f.following))
                hiI += dir;
            hiI -= dir;
            int64_t loI = (int64_t)i;
            while (loI >= (int64_t)s && loI < (int64_t)e && !is_null(loI) &&
                   le_plus((int64_t)i, loI, f.preceding))
                loI -= dir;
            loI += dir;
            lo = asc ? loI : hiI;
            hi = asc ? hiI : loI;
            break; }
        case GX_FRAME_RANGE_UNBOUNDED_PRECEDING: {
            int64_t other = (int64_t)i;
            while (other >= (int64_t)s && other < (int64_t)e &&
                   !is_null(other) &&
                   (asc ? le_plus(other, (int64_t)i, f.following)
                        : le_plus((int64_t)i, other, f.following)))
                other += 1;
            if (other != (int64_t)i) other -= 1;
            lo = (int64_t)s;
            hi = other;
            break; }
        default: { /* RANGE_UNBOUNDED_FOLLOWING */
            int64_t other = (int64_t)i;
            while (other >= (int64_t)s && other < (int64_t)e &&
                   !is_null(other) &&
                   (asc ? le_plus((int64_t)i, other, f.preceding)
                        : le_plus(other, (int64_t)i, f.preceding)))
                other -= 1;
            if (other != (int64_t)i) other += 1;
            lo = other;
            hi = (int64_t)e - 1;
            break; }
        }
    }

    /* accumulate rows [lo, hi) of frame f into (i64 or f64, isnull) */
    void range_agg(const gx_frame_spec &f, size_t lo, size_t hi,
                   int64_t &iv, double &dv, bool &isnull) const {
        bool never_null = f.func == GX_AGG_COUNT_ROW ||
                          f.func == GX_AGG_COUNT_COL ||
                          f.func == GX_AGG_SUM_I64 ||
                          f.func == GX_AGG_BIT_AND ||
                          f.func == GX_AGG_BIT_OR ||
                          f.func == GX_AGG_BIT_XOR;
        iv = 0; dv = 0; isnull = !never_null;
        const Column *c = f.input_col >= 0 ? &input.cols[f.input_col] : nullptr;
        if (f.func == GX_AGG_MIN_I64) iv = INT64_MAX;
        if (f.func == GX_AGG_MAX_I64) iv = INT64_MIN;
        if (f.func == GX_AGG_BIT_AND) iv = -1;
        for (size_t r = lo; r < hi; r++) {
            switch (f.func) {
            case GX_AGG_COUNT_ROW: iv++; break;
            case GX_AGG_COUNT_COL: if (!c->is_null(r)) iv++; break;
            case GX_AGG_SUM_I64:
                if (!c->is_null(r))
                    iv = (int64_t)((uint64_t)iv + (uint64_t)(
                        c->type == GX_I32 ? (int64_t)c->i32v[r] : c->i64v[r]));
                break;
            case GX_AGG_SUM_I64N:
                if (!c->is_null(r)) {
                    iv = (int64_t)((uint64_t)iv + (uint64_t)(
                        c->type == GX_I32 ? (int64_t)c->i32v[r] : c->i64v[r]));
                    isnull = false;
                }
                break;
            case GX_AGG_SUM_F64:
                if (!c->is_null(r)) {
                    double v = c->type == GX_F64 ? c->f64v[r]
                             : c->type == GX_I32 ? (double)c->i32v[r]
                                                 : (double)c->i64v[r];
                    if (isnull) { dv = v; isnull = false; } else dv += v;
                }
                break;
            case GX_AGG_MIN_I64: case GX_AGG_MAX_I64:
                if (!c->is_null(r)) {
                    int64_t v = c->type == GX_I32 ? (int64_t)c->i32v[r]
                                                  : c->i64v[r];
                    iv = f.func == GX_AGG_MIN_I64 ? std::min(iv, v)
                                                  : std::max(iv, v);
                    isnull = false;
                }
                break;
            case GX_AGG_MIN_F64: case GX_AGG_MAX_F64:
                if (!c->is_null(r)) {
                    double v = c->f64v[r];
                    if (isnull) { dv = v; isnull = false; }
                    else dv = f.func == GX_AGG_MIN_F64 ? jmin_f64o(dv, v)
                                                       : jmax_f64o(dv, v);
                }
                break;
            case GX_AGG_AVG_F64:
                if (!c->is_null(r)) {
                    double v = c->type == GX_F64 ? c->f64v[r]
                             : c->type == GX_I32 ? (double)c->i32v[r]
                                                 : (double)c->i64v[r];
                    dv += v; iv++; isnull = false;
                }
                break;
            case GX_AGG_BIT_AND: case GX_AGG_BIT_OR: case GX_AGG_BIT_XOR:
                if (!c->is_null(r)) {
                    int64_t v = c->type == GX_I32 ? (int64_t)c->i32v[r]
                                                  : c->i64v[r];
                    if (f.func == GX_AGG_BIT_AND) iv &= v;
                    else if (f.func == GX_AGG_BIT_OR) iv |= v;
                    else iv ^= v;
                }
                break;
            }
        }
    }

    int next(gx_result **out) {
        *out = nullptr;
        if (!finished) { set_err("fwindow next before finish"); return -1; }
        const size_t n = input.n_rows;
        if (emit_cursor >= n) return 0;
        size_t m = std::min<size_t>(CHUNK_SIZE, n - emit_cursor);
        std::vector<int32_t> otypes = input_types;
        for (auto &f : frames) {
            switch (f.func) {
            case GX_AGG_COUNT_ROW: case GX_AGG_COUNT_COL:
            case GX_AGG_SUM_I64: case GX_AGG_SUM_I64N:
            case GX_AGG_MIN_I64: case GX_AGG_MAX_I64:
            case GX_AGG_BIT_AND: case GX_AGG_BIT_OR: case GX_AGG_BIT_XOR:
            case GX_AGG_RANK: case GX_AGG_DENSE_RANK: case GX_AGG_NTILE:
                otypes.push_back(GX_I64); break;
            case GX_AGG_FIRST_VALUE: case GX_AGG_LAST_VALUE:
            case GX_AGG_NTH_VALUE: case GX_AGG_LAG: case GX_AGG_LEAD:
                otypes.push_back(input_types[f.input_col]); break;
            default: otypes.push_back(GX_F64); break;
            }
        }
        std::vector<OutCol> cols(otypes.size());
        for (size_t c = 0; c < otypes.size(); c++) cols[c].type = otypes[c];
        for (size_t k = 0; k < m; k++) {
            size_t i = emit_cursor + k;
            for (size_t c = 0; c < input_types.size(); c++)
                cols[c].append_from(input.cols[c], i);
            for (size_t a = 0; a < frames.size(); a++) {
                const gx_frame_spec &f = frames[a];
                size_t s = seg_start_of_row[i], e = seg_end_of_row[i];
                size_t col2 = input_types.size() + a;
                if (f.func >= GX_AGG_FIRST_VALUE &&
                    f.func <= GX_AGG_PERCENT_RANK) {
                    /* navigation / distribution over the whole partition */
                    int64_t size = (int64_t)(e - s), pos = (int64_t)(i - s);
                    int64_t src = -1;
                    switch (f.func) {
                    case GX_AGG_FIRST_VALUE: src = (int64_t)s; break;
                    case GX_AGG_LAST_VALUE: src = (int64_t)e - 1; break;
                    case GX_AGG_NTH_VALUE:
                        src = (int64_t)s + f.preceding - 1;
                        if (src >= (int64_t)e) src = -1;
                        break;
                    case GX_AGG_LAG:
                        src = (int64_t)i - f.preceding;
                        if (src < (int64_t)s) src = -1;
                        break;
                    case GX_AGG_LEAD:
                        src = (int64_t)i + f.preceding;
                        if (src >= (int64_t)e) src = -1;
                        break;
                    case GX_AGG_NTILE: {
                        int64_t nt = f.preceding;
                        int64_t base = size / nt, rem = size % nt;
                        int64_t bucket;
                        if (base == 0) bucket = pos + 1;
                        else if (pos < rem * (base + 1))
                            bucket = pos / (base + 1) + 1;
                        else bucket = rem + (pos - rem * (base + 1)) / base + 1;
                        cols[col2].append_i64(bucket);
                        continue; }
                    case GX_AGG_CUME_DIST:
                        cols[col2].append_f64(
                            (double)((int64_t)run_end_of_row[i] -
                                     (int64_t)s + 1) / (double)size);
                        continue;
                    case GX_AGG_PERCENT_RANK:
                        cols[col2].append_f64(size <= 1 ? 0.0 :
                            (double)((int64_t)run_start_of_row[i] -
                                     (int64_t)s) / (double)(size - 1));
                        continue;
                    }
                    if (src < 0) cols[col2].append_null();
                    else cols[col2].append_from(input.cols[f.input_col],
                                                (size_t)src);
                    continue;
                }
                size_t lo = s, hi = e;
                if (f.kind == GX_FRAME_ROWS_SLIDING) {
                    lo = i >= s + (size_t)f.preceding ? i - (size_t)f.preceding : s;
                    hi = std::min<size_t>(e, i + (size_t)f.following + 1);
                } else if (f.kind == GX_FRAME_ROWS_UNBOUNDED_FOLLOWING) {
                    lo = i;
                } else if (f.kind == GX_FRAME_RANGE_SLIDING ||
                           f.kind == GX_FRAME_RANGE_UNBOUNDED_PRECEDING ||
                           f.kind == GX_FRAME_RANGE_UNBOUNDED_FOLLOWING) {
                    int64_t l, h;
                    range_bounds(f, i, s, e, l, h);
                    lo = (size_t)l;
                    hi = h < l ? (size_t)l : (size_t)(h + 1); /* exclusive */
                }
                int64_t iv; double dv; bool isnull;
                range_agg(f, lo, hi, iv, dv, isnull);
                size_t col = input_types.size() + a;
                if (isnull) cols[col].append_null();
                else if (f.func == GX_AGG_AVG_F64)
                    cols[col].append_f64(dv / (double)iv);
                else if (otypes[col] == GX_I64) cols[col].append_i64(iv);
                else cols[col].append_f64(dv);
            }
        }
        emit_cursor += m;
        *out = make_result(std::move(cols), (int32_t)m);
        return 0;
    }
};

/* ---- partition operator -------------------------------------------------
 * Restates PartitioningExchanger.consumeChunk (mpp/operator/
 * PartitioningExchanger.java:71-134): row hash over key cols (HashBucketFunction
 * shape, PartitionedOutputCollector.java:270-302) -> ExecUtils.partition. */
struct PartOp : gx_op {
    gx_part_cfg cfg;
    std::vector<int32_t> key_cols_, input_types;
    bool pow2;
    PartOp(const gx_part_cfg *c) : gx_op(OP_PART), cfg(*c) {
        key_cols_.assign(c->key_cols, c->key_cols + c->n_key_cols);
        input_types.assign(c->input_types, c->input_types + c->n_input_cols);
        pow2 = (cfg.n_parts & (-cfg.n_parts)) == cfg.n_parts;
    }
    int consume_concat(const gx_chunk *ch, gx_result **out, int64_t *counts) {
        *out = nullptr;
        Store in;
        in.init((int32_t)input_types.size(), input_types.data());
        if (in.append(ch) != 0) return -1;
        std::vector<int> kc(key_cols_.begin(), key_cols_.end());
        std::vector<std::vector<size_t>> parts((size_t)cfg.n_parts);
        for (size_t r = 0; r < in.n_rows; r++) {
            int32_t h = in.row_hash(r, kc);
            int32_t p;
            if (pow2) p = hc_murmur3(h) & (cfg.n_parts - 1);
            else p = (int32_t)(((uint32_t)hc_murmur3(h) & 0x7fffffffu) % (uint32_t)cfg.n_parts);
            parts[(size_t)p].push_back(r);
        }
        std::vector<OutCol> cols(input_types.size());
        for (size_t c = 0; c < input_types.size(); c++) cols[c].type = input_types[c];
        for (int32_t p = 0; p < cfg.n_parts; p++) {
            counts[p] = (int64_t)parts[(size_t)p].size();
            for (size_t r : parts[(size_t)p])
                for (size_t c = 0; c < input_types.size(); c++)
                    cols[c].append_from(in.cols[c], r);
        }
        *out = make_result(std::move(cols), (int32_t)in.n_rows);
        return 0;
    }

    int consume(const gx_chunk *ch, gx_result **outs) {
        Store in;
        in.init((int32_t)input_types.size(), input_types.data());
        if (in.append(ch) != 0) return -1;
        std::vector<int> kc(key_cols_.begin(), key_cols_.end());
        std::vector<std::vector<size_t>> parts((size_t)cfg.n_parts);
        for (size_t r = 0; r < in.n_rows; r++) {
            int32_t h = in.row_hash(r, kc);
            int32_t p;
            if (pow2) p = hc_murmur3(h) & (cfg.n_parts - 1);
            else p = (int32_t)(((uint32_t)hc_murmur3(h) & 0x7fffffffu) % (uint32_t)cfg.n_parts);
            parts[(size_t)p].push_back(r);
        }
        for (int32_t p = 0; p < cfg.n_parts; p++) {
            outs[p] = nullptr;
            if (parts[(size_t)p].empty()) continue;
            std::vector<OutCol> cols(input_types.size());
            for (size_t c = 0; c < input_types.size(); c++) cols[c].type = input_types[c];
            for (size_t r : parts[(size_t)p])
                for (size_t c = 0; c < input_types.size(); c++)
                    cols[c].append_from(in.cols[c], r);
            outs[p] = make_result(std::move(cols), (int32_t)parts[(size_t)p].size());
        }
        return 0;
    }
};

/* DecimalBox "simple" layout conversions (DecimalBox.java:43-71,
 * doAddToSum1/2; DecimalTypeBase offsets 36..39). Exact for values with
 * <= 18 integer digits and fraction scale <= 9. */
static const int64_t POW10[19] = {1LL,10LL,100LL,1000LL,10000LL,100000LL,
    1000000LL,10000000LL,100000000LL,1000000000LL,10000000000LL,
    100000000000LL,1000000000000LL,10000000000000LL,100000000000000LL,
    1000000000000000LL,10000000000000000LL,100000000000000000LL,
    1000000000000000000LL};

/* Wide-DECIMAL fence: values needing a third integer word (integers > 18)
 * or more than 18 total significant digits do not fit the scaled-int64
 * fast path; the reference falls back to full 9-limb Decimal arithmetic
 * there (DecimalBox.java:43-71). We reject loudly (set *err) instead of
 * silently truncating. */
static inline int64_t dec40_to_scaled(const uint8_t *p, int scale,
                                      bool *err) {
    int32_t w[3];
    std::memcpy(w, p, 12);
    uint8_t integers = p[36];
    uint8_t isneg = p[39];
    int64_t ip, fr;
    if (integers > 9) {
        ip = (int64_t)w[0] * 1000000000LL + w[1];
        fr = w[2];
    } else {
        ip = w[0];
        fr = w[1];
    }
    if (integers > 18 || ip >= POW10[18 - scale]) {
        *err = true;
        return 0;
    }
    /* frac word holds the fraction digits x 10^(9-scale) */
    int64_t v = ip * POW10[scale] + fr / POW10[9 - scale];
    return isneg ? -v : v;
}

static inline void scaled_to_dec40(int64_t v, int scale, uint8_t *p) {
    std::memset(p, 0, 40);
    uint8_t isneg = v < 0;
    uint64_t a = isneg ? (uint64_t)(-v) : (uint64_t)v;
    int64_t ip = (int64_t)(a / (uint64_t)POW10[scale]);
    int64_t fr = (int64_t)(a % (uint64_t)POW10[scale]) * POW10[9 - scale];
    int32_t w[3];
    uint8_t integers;
    if (ip >= 1000000000LL) {
        w[0] = (int32_t)(ip / 1000000000LL);
        w[1] = (int32_t)(ip % 1000000000LL);
        w[2] = (int32_t)fr;
        integers = 18;
    } else {
        w[0] = (int32_t)ip;
        w[1] = (int32_t)fr;
        w[2] = 0;
        integers = 9;
    }
    std::memcpy(p, w, 12);
    p[36] = integers;
    p[37] = (uint8_t)scale;
    p[38] = (uint8_t)scale;
    p[39] = isneg;
}

/* ---- scan (vectorized filter + project) ---------------------------------
 * Restates the vectorized filter/projection stage (executor/vectorized/,
 * SURVEY.md §8f row 1): AND of predicates, SQL NULL-fails semantics,
 * projected output. */
struct ScanOp : gx_op {
    gx_scan_cfg cfg;
    std::vector<gx_pred> preds;
    std::vector<gx_proj> projs;
    std::vector<int32_t> input_types, out_types;

    std::vector<std::vector<uint8_t>> pat_store;
    ScanOp(const gx_scan_cfg *c) : gx_op(OP_SCAN), cfg(*c) {
        preds.assign(c->preds, c->preds + c->n_preds);
        for (auto &p : preds) {
            if (p.cmp == GX_CMP_CONTAINS && p.v_bytes && p.v_len > 0) {
                pat_store.emplace_back(p.v_bytes, p.v_bytes + p.v_len);
                p.v_bytes = pat_store.back().data();
            }
        }
        projs.assign(c->projs, c->projs + c->n_projs);
        input_types.assign(c->input_types, c->input_types + c->n_input_cols);
        for (auto &p : projs) {
            switch (p.op) {
            case GX_PROJ_COPY: out_types.push_back(input_types[p.a]); break;
            case GX_PROJ_REV_F64: out_types.push_back(GX_F64); break;
            case GX_PROJ_SCALED_TO_DEC: out_types.push_back(GX_DECIMAL); break;
            default: out_types.push_back(GX_I64); break;
            }
        }
    }

    static bool cmp_ok_i(int64_t a, int32_t cmp, int64_t b) {
        switch (cmp) {
        case GX_CMP_LT: return a < b;
        case GX_CMP_LE: return a <= b;
        case GX_CMP_GT: return a > b;
        case GX_CMP_GE: return a >= b;
        case GX_CMP_EQ: return a == b;
        default: return a != b;
        }
    }
    static bool cmp_ok_f(double a, int32_t cmp, double b) {
        switch (cmp) {
        case GX_CMP_LT: return a < b;
        case GX_CMP_LE: return a <= b;
        case GX_CMP_GT: return a > b;
        case GX_CMP_GE: return a >= b;
        case GX_CMP_EQ: return a == b;
        default: return a != b;
        }
    }

    int consume(const gx_chunk *ch, gx_result **out) {
        *out = nullptr;
        Store in;
        in.init((int32_t)input_types.size(), input_types.data());
        if (in.append(ch) != 0) return -1;
        std::vector<OutCol> cols(out_types.size());
        for (size_t c = 0; c < out_types.size(); c++) cols[c].type = out_types[c];
        int32_t kept = 0;
        bool wide_err = false;
        for (size_t r = 0; r < in.n_rows; r++) {
            bool pass = true;
            for (auto &p : preds) {
                const Column &c = in.cols[p.col];
                if (c.is_null(r)) { pass = false; break; }
                bool ok;
                switch (c.type) {
                case GX_I64: ok = cmp_ok_i(c.i64v[r], p.cmp, p.v_i64); break;
                case GX_I32: ok = cmp_ok_i((int64_t)c.i32v[r], p.cmp, p.v_i64); break;
                case GX_SLICE: {
                    /* LIKE '%pat%' naive byte scan over the slice */
                    int32_t b = c.begin_off(r), e = c.off[r];
                    int32_t len = e - b, pl = p.v_len;
                    ok = false;
                    for (int32_t s = 0; s + pl <= len && !ok; s++)
                        ok = std::memcmp(c.bytes.data() + b + s, p.v_bytes,
                                         (size_t)pl) == 0;
                    break; }
                default: ok = cmp_ok_f(c.f64v[r], p.cmp, p.v_f64); break;
                }
                if (!ok) { pass = false; break; }
            }
            if (!pass) continue;
            kept++;
            for (size_t c = 0; c < projs.size(); c++) {
                const gx_proj &pj = projs[c];
                const Column &a = in.cols[pj.a];
                switch (pj.op) {
                case GX_PROJ_COPY:
                    cols[c].append_from(a, r);
                    break;
                case GX_PROJ_REV_F64: {
                    const Column &b = in.cols[pj.b];
                    if (a.is_null(r) || b.is_null(r)) cols[c].append_null();
                    else cols[c].append_f64(a.f64v[r] * (1.0 - b.f64v[r]));
                    break; }
                case GX_PROJ_REV_SCALED4: {
                    const Column &b = in.cols[pj.b];
                    if (a.is_null(r) || b.is_null(r)) cols[c].append_null();
                    else cols[c].append_i64((int64_t)((uint64_t)a.i64v[r] *
                                            (uint64_t)(100 - b.i64v[r])));
                    break; }
                case GX_PROJ_DEC_TO_SCALED: {
                    if (a.is_null(r)) cols[c].append_null();
                    else cols[c].append_i64(dec40_to_scaled(
                        a.bytes.data() + (size_t)r * 40, pj.c, &wide_err));
                    break; }
                case GX_PROJ_SCALED_TO_DEC: {
                    if (a.is_null(r)) cols[c].append_null();
                    else {
                        uint8_t buf[40];
                        scaled_to_dec40(a.i64v[r], pj.c, buf);
                        cols[c].append_dec40(buf);
                    }
                    break; }
                default: { /* Q9_AMOUNT4 */
                    const Column &b = in.cols[pj.b];
                    const Column &cc = in.cols[pj.c];
                    const Column &dd = in.cols[pj.d];
                    if (a.is_null(r) || b.is_null(r) || cc.is_null(r) ||
                        dd.is_null(r))
                        cols[c].append_null();
                    else
                        cols[c].append_i64(
                            (int64_t)((uint64_t)a.i64v[r] * (uint64_t)(100 - b.i64v[r]))
                            - (int64_t)((uint64_t)cc.i64v[r] * (uint64_t)dd.i64v[r] * 100u));
                    break; }
                }
            }
        }
        if (wide_err) {
            set_err("wide DECIMAL: value exceeds 18 significant digits / "
                    "the DecimalBox simple layout; the scaled-int64 fast "
                    "path cannot represent it (reference falls back to full "
                    "Decimal arithmetic, DecimalBox.java:43-71)");
            return -1;
        }
        *out = make_result(std::move(cols), kept);
        return 0;
    }
};

} // anonymous namespace

/* ---- C ABI ------------------------------------------------------------- */

extern "C" {

gx_op *gxop_join_create(const gx_join_cfg *cfg) {
    if (!cfg || cfg->n_keys <= 0) { set_err("bad join cfg"); return nullptr; }
    if (cfg->join_type < GX_JOIN_INNER || cfg->join_type > GX_JOIN_ANTI) {
        set_err("unknown join type");
        return nullptr;
    }
    return new JoinOp(cfg);
}
int gxop_join_consume(gx_op *op, const gx_chunk *c) {
    if (!op || op->kind != OP_JOIN) { set_err("not a join op"); return -1; }
    std::lock_guard<std::mutex> lk(op->mu);
    return static_cast<JoinOp *>(op)->consume(c);
}
int gxop_join_build(gx_op *op) {
    if (!op || op->kind != OP_JOIN) { set_err("not a join op"); return -1; }
    std::lock_guard<std::mutex> lk(op->mu);
    return static_cast<JoinOp *>(op)->do_build();
}
int gxop_join_probe(gx_op *op, const gx_chunk *c, gx_result **out) {
    if (!op || op->kind != OP_JOIN) { set_err("not a join op"); return -1; }
    return static_cast<JoinOp *>(op)->probe(c, out);
}
int gxop_join_tail(gx_op *op, gx_result **out) {
    if (!op || op->kind != OP_JOIN) { set_err("not a join op"); return -1; }
    return static_cast<JoinOp *>(op)->tail(out);
}
int gxop_join_probe_push(gx_op *op, const gx_chunk *c) {
    if (!op || op->kind != OP_JOIN) { set_err("not a join op"); return -1; }
    return static_cast<JoinOp *>(op)->probe_push(c);
}
int gxop_join_probe_flush(gx_op *op, gx_result **out) {
    if (!op || op->kind != OP_JOIN) { set_err("not a join op"); return -1; }
    return static_cast<JoinOp *>(op)->probe_flush(out);
}
int gxop_join_close(gx_op *op) { delete op; return 0; }

static int agg_funcs_ok(const gx_agg_spec *aggs, int32_t n) {
    for (int32_t i = 0; i < n; i++) {
        int32_t f = aggs[i].func;
        /* group-by supports the accumulator set only: COUNT..BIT_XOR and
         * the null-init Sum; rank/navigation families are window-only and
         * anything else is unknown */
        if ((f < GX_AGG_COUNT_ROW || f > GX_AGG_BIT_XOR) &&
            f != GX_AGG_SUM_I64N)
            return 0;
    }
    return 1;
}

gx_op *gxop_agg_create(const gx_agg_cfg *cfg) {
    if (cfg && !agg_funcs_ok(cfg->aggs, cfg->n_aggs)) {
        set_err("unknown or window-only agg func in group-by");
        return nullptr;
    }
    if (!cfg) { set_err("bad agg cfg"); return nullptr; }
    return new AggOp(cfg);
}
int gxop_agg_consume(gx_op *op, const gx_chunk *c) {
    if (!op || op->kind != OP_AGG) { set_err("not an agg op"); return -1; }
    return static_cast<AggOp *>(op)->consume(c);
}
int gxop_agg_build(gx_op *op) {
    if (!op || op->kind != OP_AGG) { set_err("not an agg op"); return -1; }
    static_cast<AggOp *>(op)->built = true;
    return 0;
}
int gxop_agg_next(gx_op *op, gx_result **out) {
    if (!op || op->kind != OP_AGG) { set_err("not an agg op"); return -1; }
    return static_cast<AggOp *>(op)->next(out);
}
int gxop_agg_close(gx_op *op) { delete op; return 0; }


gx_op *gxop_groupjoin_create(const gx_groupjoin_cfg *cfg) {
    if (cfg && !agg_funcs_ok(cfg->aggs, cfg->n_aggs)) {
        set_err("unknown or window-only agg func in group-by");
        return nullptr;
    }
    if (!cfg || cfg->n_keys <= 0 ||
        (cfg->join_type != GX_JOIN_INNER && cfg->join_type != GX_JOIN_LEFT)) {
        set_err("bad groupjoin cfg");
        return nullptr;
    }
    return new GroupJoinOp(cfg);
}
int gxop_groupjoin_consume(gx_op *op, const gx_chunk *c) {
    if (!op || op->kind != OP_GROUPJOIN) { set_err("not a groupjoin op"); return -1; }
    return static_cast<GroupJoinOp *>(op)->consume(c);
}
int gxop_groupjoin_build(gx_op *op) {
    if (!op || op->kind != OP_GROUPJOIN) { set_err("not a groupjoin op"); return -1; }
    return static_cast<GroupJoinOp *>(op)->do_build();
}
int gxop_groupjoin_probe(gx_op *op, const gx_chunk *c) {
    if (!op || op->kind != OP_GROUPJOIN) { set_err("not a groupjoin op"); return -1; }
    return static_cast<GroupJoinOp *>(op)->probe(c);
}
int gxop_groupjoin_next(gx_op *op, gx_result **out) {
    if (!op || op->kind != OP_GROUPJOIN) { set_err("not a groupjoin op"); return -1; }
    return static_cast<GroupJoinOp *>(op)->next(out);
}
int gxop_groupjoin_close(gx_op *op) { delete op; return 0; }


gx_op *gxop_window_create(const gx_window_cfg *cfg) {
    if (!cfg || cfg->n_aggs <= 0) { set_err("bad window cfg"); return nullptr; }
    for (int32_t i = 0; i < cfg->n_aggs; i++) {
        int32_t f = cfg->aggs[i].func;
        /* running window: accumulators + rank family + null-init Sum;
         * navigation/distribution funcs are frame-window only */
        if ((f < GX_AGG_COUNT_ROW || f > GX_AGG_DENSE_RANK) &&
            f != GX_AGG_SUM_I64N) {
            set_err("unknown or frame-only func in running window");
            return nullptr;
        }
    }
    return new WindowOp(cfg);
}
int gxop_window_consume(gx_op *op, const gx_chunk *c, gx_result **out) {
    if (!op || op->kind != OP_WINDOW) { set_err("not a window op"); return -1; }
    return static_cast<WindowOp *>(op)->consume(c, out);
}
int gxop_window_close(gx_op *op) { delete op; return 0; }


static int fwindow_cfg_ok(const gx_fwindow_cfg *c) {
    if (!c || c->n_frames <= 0) return 0;
    for (int32_t i = 0; i < c->n_frames; i++) {
        const gx_frame_spec &f = c->frames[i];
        if (f.func >= GX_AGG_FIRST_VALUE && f.func <= GX_AGG_PERCENT_RANK) {
            /* navigation funcs: whole-partition only; NTH/LAG/LEAD/NTILE
             * need a positive parameter; CUME/PERCENT need ORDER cols */
            if (f.kind != GX_FRAME_WHOLE_PARTITION) return 0;
            if ((f.func == GX_AGG_NTH_VALUE || f.func == GX_AGG_LAG ||
                 f.func == GX_AGG_LEAD || f.func == GX_AGG_NTILE) &&
                f.preceding <= 0)
                return 0;
            if ((f.func == GX_AGG_CUME_DIST ||
                 f.func == GX_AGG_PERCENT_RANK) && c->n_order_cols <= 0)
                return 0;
            continue;
        }
        if (f.kind == GX_FRAME_WHOLE_PARTITION) continue;
        /* sliding/RANGE: additive + SUM/AVG(F64) + MIN/MAX;
         * following: additive + SUM/AVG(F64) (matches the HIP create) */
        bool additive = f.func == GX_AGG_COUNT_ROW ||
                        f.func == GX_AGG_COUNT_COL ||
                        f.func == GX_AGG_SUM_I64 ||
                        f.func == GX_AGG_SUM_I64N;
        bool f64sum = f.func == GX_AGG_SUM_F64 || f.func == GX_AGG_AVG_F64;
        bool minmax = f.func == GX_AGG_MIN_I64 || f.func == GX_AGG_MAX_I64 ||
                      f.func == GX_AGG_MIN_F64 || f.func == GX_AGG_MAX_F64;
        bool is_range = f.kind == GX_FRAME_RANGE_SLIDING ||
                        f.kind == GX_FRAME_RANGE_UNBOUNDED_PRECEDING ||
                        f.kind == GX_FRAME_RANGE_UNBOUNDED_FOLLOWING;
        if (f.kind == GX_FRAME_ROWS_SLIDING) {
            if (!additive && !minmax && !f64sum) return 0;
            if (f.preceding < 0 || f.following < 0) return 0;
        } else if (is_range) {
            if (!additive && !minmax && !f64sum) return 0;
            if (f.order_col < 0 || f.order_col >= c->n_input_cols) return 0;
            int32_t ot = c->input_types[f.order_col];
            if (ot != GX_I64 && ot != GX_I32 && ot != GX_F64) return 0;
        } else {
            if (!additive && !f64sum) return 0;
        }
    }
    return 1;
}

gx_op *gxop_fwindow_create(const gx_fwindow_cfg *cfg) {
    if (cfg)
        for (int32_t i = 0; i < cfg->n_frames; i++)
            if (cfg->frames[i].func == GX_AGG_RANK ||
                cfg->frames[i].func == GX_AGG_DENSE_RANK) {
                set_err("unknown or window-only agg func in group-by");
                return nullptr;
            }
    if (!fwindow_cfg_ok(cfg)) { set_err("bad fwindow cfg (unsupported frame/func)"); return nullptr; }
    return new FWindowOp(cfg);
}
int gxop_fwindow_consume(gx_op *op, const gx_chunk *c) {
    if (!op || op->kind != OP_FWINDOW) { set_err("not an fwindow op"); return -1; }
    return static_cast<FWindowOp *>(op)->consume(c);
}
int gxop_fwindow_finish(gx_op *op) {
    if (!op || op->kind != OP_FWINDOW) { set_err("not an fwindow op"); return -1; }
    return static_cast<FWindowOp *>(op)->finish();
}
int gxop_fwindow_next(gx_op *op, gx_result **out) {
    if (!op || op->kind != OP_FWINDOW) { set_err("not an fwindow op"); return -1; }
    return static_cast<FWindowOp *>(op)->next(out);
}
int gxop_fwindow_close(gx_op *op) { delete op; return 0; }

gx_op *gxop_part_create(const gx_part_cfg *cfg) {
    if (!cfg || cfg->n_parts <= 0) { set_err("bad part cfg"); return nullptr; }
    return new PartOp(cfg);
}
int gxop_part_consume(gx_op *op, const gx_chunk *c, gx_result **outs) {
    if (!op || op->kind != OP_PART) { set_err("not a part op"); return -1; }
    return static_cast<PartOp *>(op)->consume(c, outs);
}
int gxop_part_consume_concat(gx_op *op, const gx_chunk *c, gx_result **out,
                             int64_t *counts) {
    if (!op || op->kind != OP_PART) { set_err("not a part op"); return -1; }
    return static_cast<PartOp *>(op)->consume_concat(c, out, counts);
}
int gxop_part_close(gx_op *op) { delete op; return 0; }

int gxop_result_copy_col(const gx_result *res, int32_t col, void *dst_values,
                         void *dst_nulls) {
    if (!res || col < 0 || col >= res->chunk.n_blocks) { set_err("bad column"); return -1; }
    const gx_block *b = &res->chunk.blocks[col];
    int64_t n = res->chunk.n_rows;
    if (n == 0) return 0;
    size_t es = (b->type == GX_I32) ? 4 : 8;
    std::memcpy(dst_values, b->values, (size_t)n * es);
    if (dst_nulls) {
        if (b->nulls) std::memcpy(dst_nulls, b->nulls, (size_t)n);
        else std::memset(dst_nulls, 0, (size_t)n);
    }
    return 0;
}

int gxop_join_get_stats(gx_op *op, gx_join_stats *out) {
    if (!op || op->kind != OP_JOIN || !out) { set_err("not a join op"); return -1; }
    std::memset(out, 0, sizeof(*out));
    return 0;
}

int gxop_agg_get_stats(gx_op *op, gx_agg_stats *out) {
    if (!op || op->kind != OP_AGG || !out) { set_err("not an agg op"); return -1; }
    std::memset(out, 0, sizeof(*out));
    out->groups = static_cast<AggOp *>(op)->n_groups();
    return 0;
}

gx_op *gxop_scan_create(const gx_scan_cfg *cfg) {
    if (!cfg || cfg->n_projs <= 0) { set_err("bad scan cfg"); return nullptr; }
    for (int32_t i = 0; i < cfg->n_preds; i++) {
        const gx_pred &p = cfg->preds[i];
        if (p.col < 0 || p.col >= cfg->n_input_cols ||
            p.cmp < GX_CMP_LT || p.cmp > GX_CMP_CONTAINS) {
            set_err("bad scan predicate (column or comparison)");
            return nullptr;
        }
    }
    for (int32_t i = 0; i < cfg->n_projs; i++) {
        const gx_proj &p = cfg->projs[i];
        if (p.op < GX_PROJ_COPY || p.op > GX_PROJ_SCALED_TO_DEC ||
            p.a < 0 || p.a >= cfg->n_input_cols ||
            ((p.op == GX_PROJ_REV_F64 || p.op == GX_PROJ_REV_SCALED4 ||
              p.op == GX_PROJ_Q9_AMOUNT4) &&
             (p.b < 0 || p.b >= cfg->n_input_cols))) {
            set_err("bad scan projection (op or column)");
            return nullptr;
        }
    }
    return new ScanOp(cfg);
}
int gxop_scan_consume(gx_op *op, const gx_chunk *c, gx_result **out) {
    if (!op || op->kind != OP_SCAN) { set_err("not a scan op"); return -1; }
    return static_cast<ScanOp *>(op)->consume(c, out);
}
int gxop_scan_close(gx_op *op) { delete op; return 0; }

int gxop_result_to_host(gx_result *) { return 0; /* oracle results are host */ }
void gxop_result_release(gx_result *res) {
    if (res) delete static_cast<ResultHolder *>(res->opaque);
}
const char *gx_last_error(void) { return g_err.c_str(); }
int gxop_abi_version(void) { return 0; /* CPU oracle */ }

} /* extern "C" */

#include "../galaxysql_amd/csrc/gx_serde.inc"
