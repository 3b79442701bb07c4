// oracle/common.hpp — shared datum/row/chunk machinery for the CPU oracle.
//
// ORACLE — TEST INFRASTRUCTURE ONLY. This library is a CPU restatement of the
// reference executors used to pin parity. Only tests/,
// __graft_entry__.smoke() and bench.py's cpu_baseline leg may load it. It is
// never the product path.
//
// Restates (semantics; file:line cite the reference under /root/reference):
//  - Datum equality/order: ScalarImpl PartialEq / DefaultOrd with NULLs
//    largest (common/src/util/sort_util.rs:67-74, memcmp_encoding.rs:31-52).
//  - StreamChunkBuilder::append_iter_inner chunking incl. the U-pair
//    no-split rule (common/src/array/stream_chunk_builder.rs:188-218).
//  - StreamChunk::eliminate_adjacent_noop_update
//    (common/src/array/stream_chunk.rs:331-384).
#pragma once
#include <algorithm>
#include <cassert>
#include <cmath>
#include <cstdint>
#include <cstring>
#include <map>
#include <set>
#include <memory>
#include <string>
#include <vector>

#include "../include/rw_chunk.h"

namespace orc {

struct Datum {
    bool null = true;
    union {
        int64_t i;
        double d;
    };
    Datum() : null(true), i(0) {}
    static Datum of_i(int64_t v) {
        Datum x;
        x.null = false;
        x.i = v;
        return x;
    }
    static Datum of_d(double v) {
        Datum x;
        x.null = false;
        x.d = v;
        return x;
    }
    // decimal datums carry the 16-byte rust_decimal serialize image as two
    // LE halves: i = bytes 0..8 (flags ++ lo), i2 = bytes 8..16 (mid ++ hi)
    int64_t i2 = 0;
    static Datum of_dec(int64_t lohalf, int64_t hihalf) {
        Datum x;
        x.null = false;
        x.i = lohalf;
        x.i2 = hihalf;
        return x;
    }
};

// ---- decimal arithmetic restatement (rust_decimal 1.40.0, a Cargo.lock
// dependency not vendored under /root/reference; the reference wraps it as
// Decimal{NegativeInf, Normalized, PositiveInf, NaN}, types/decimal.rs:36-44,
// with sum via checked_add, expr general.rs:23). The EXACT domain is
// restated: a decimal is sign x mantissa96 x 10^-scale (scale 0..28);
// addition aligns scales and adds exactly. Sums are accumulated as exact
// signed 256-bit integers at scale 28 (each addend = mantissa x
// 10^(28-scale), <= 10^56 << 2^191; 2^64 such addends cannot overflow
// i256), so the result is the exact rational sum, converted back at the
// group's max input scale. If the exact result's mantissa exceeds 96 bits
// the reference's sequential rust_decimal add would have entered its
// order-dependent precision-loss rescale path — that domain raises loudly
// instead of silently diverging. Specials follow the
// reference's Add table (decimal.rs:259-276): any NaN -> NaN, +Inf + -Inf
// -> NaN, else Inf dominates; they are counted exactly so retraction
// works. ----

struct DecVal {
    int special = 0; // 0 normal, 1 NaN, 2 +Inf, 3 -Inf (serialize byte 0)
    bool neg = false;
    uint32_t scale = 0;
    uint64_t lo = 0;  // mantissa low 64
    uint32_t hi = 0;  // mantissa high 32
};

inline DecVal dec_parse(int64_t lohalf, int64_t hihalf) {
    uint8_t b[16];
    std::memcpy(b, &lohalf, 8);
    std::memcpy(b + 8, &hihalf, 8);
    DecVal v;
    if (b[0] == 1 || b[0] == 2 || b[0] == 3) {
        v.special = b[0];
        return v;
    }
    uint32_t flags, lo32, mid, hi;
    std::memcpy(&flags, b, 4);
    std::memcpy(&lo32, b + 4, 4);
    std::memcpy(&mid, b + 8, 4);
    std::memcpy(&hi, b + 12, 4);
    v.neg = (flags >> 31) & 1;
    v.scale = (flags >> 16) & 0xFF;
    v.lo = (uint64_t)mid << 32 | lo32;
    v.hi = hi;
    return v;
}

inline void dec_serialize(const DecVal& v, int64_t* lohalf, int64_t* hihalf) {
    uint8_t b[16] = {};
    if (v.special) {
        b[0] = (uint8_t)v.special;
    } else {
        uint32_t flags = (v.scale << 16) | ((uint32_t)v.neg << 31);
        uint32_t lo32 = (uint32_t)v.lo, mid = (uint32_t)(v.lo >> 32);
        std::memcpy(b, &flags, 4);
        std::memcpy(b + 4, &lo32, 4);
        std::memcpy(b + 8, &mid, 4);
        std::memcpy(b + 12, &v.hi, 4);
    }
    std::memcpy(lohalf, b, 8);
    std::memcpy(hihalf, b + 8, 8);
}

// exact signed-256-bit accumulator (two's complement over 4 u64 words)
struct I256 {
    uint64_t w[4] = {0, 0, 0, 0};
    void add(const I256& o) {
        unsigned __int128 c = 0;
        for (int k = 0; k < 4; k++) {
            unsigned __int128 t = (unsigned __int128)w[k] + o.w[k] + c;
            w[k] = (uint64_t)t;
            c = t >> 64;
        }
    }
    I256 negated() const {
        I256 r;
        unsigned __int128 c = 1;
        for (int k = 0; k < 4; k++) {
            unsigned __int128 t = (unsigned __int128)(~w[k]) + c;
            r.w[k] = (uint64_t)t;
            c = t >> 64;
        }
        return r;
    }
    bool is_neg() const { return w[3] >> 63; }
    bool is_zero() const { return !(w[0] | w[1] | w[2] | w[3]); }
};

inline uint64_t dec_pow10_u64(int k) { // k <= 19
    uint64_t p = 1;
    while (k-- > 0) p *= 10;
    return p;
}

// addend = sign * mantissa96 * 10^(28 - scale) as I256
inline I256 dec_addend(const DecVal& v) {
    // mantissa96 * 10^k with k = 28-scale, split as 10^a * 10^b (a,b <= 14)
    int k = 28 - (int)v.scale;
    uint64_t pa = dec_pow10_u64(k / 2), pb = dec_pow10_u64(k - k / 2);
    // m = hi*2^64 + lo; multiply by pa then pb using 128-bit partials
    uint64_t m[4] = {v.lo, v.hi, 0, 0};
    for (uint64_t p : {pa, pb}) {
        unsigned __int128 c = 0;
        for (int i = 0; i < 4; i++) {
            unsigned __int128 t = (unsigned __int128)m[i] * p + c;
            m[i] = (uint64_t)t;
            c = t >> 64;
        }
    }
    I256 r;
    for (int i = 0; i < 4; i++) r.w[i] = m[i];
    if (v.neg) r = r.negated();
    return r;
}

// exact divide of |S| by 10^(28 - out_scale); returns false if the result
// mantissa exceeds 96 bits (the reference's precision-loss domain)
inline bool dec_from_sum(const I256& S, uint32_t out_scale, DecVal* out) {
    I256 a = S;
    bool neg = a.is_neg();
    if (neg) a = a.negated();
    int k = 28 - (int)out_scale;
    // long division by 10^k in two <=10^14 chunks (the sum is a multiple of
    // 10^k by construction, so remainders are exactly 0)
    for (uint64_t p : {dec_pow10_u64(k / 2), dec_pow10_u64(k - k / 2)}) {
        if (p == 1) continue;
        unsigned __int128 rem = 0;
        for (int i = 3; i >= 0; i--) {
            unsigned __int128 cur = (rem << 64) | a.w[i];
            a.w[i] = (uint64_t)(cur / p);
            rem = cur % p;
        }
        if (rem != 0) return false; // not exact: internal invariant breach
    }
    if (a.w[3] || a.w[2] || (a.w[1] >> 32)) return false; // > 96 bits
    out->special = 0;
    out->neg = neg && !a.is_zero();
    out->scale = out_scale;
    out->lo = a.w[0];
    out->hi = (uint32_t)a.w[1];
    return true;
}

// value equality (rust_decimal PartialEq compares VALUES: 1.2 == 1.20)
inline bool dec_value_eq(const DecVal& a, const DecVal& b) {
    if (a.special || b.special) return a.special == b.special;
    I256 x = dec_addend(a), y = dec_addend(b);
    return x.w[0] == y.w[0] && x.w[1] == y.w[1] && x.w[2] == y.w[2] &&
           x.w[3] == y.w[3];
}

inline bool type_is_float(uint8_t t) { return t == RW_T_F64 || t == RW_T_F32; }

// ScalarImpl PartialEq: floats are OrderedFloat (NaN == NaN).
inline bool datum_eq(const Datum& a, const Datum& b, uint8_t t);

inline bool datum_eq_impl_decimal(const Datum& a, const Datum& b) {
    return dec_value_eq(dec_parse(a.i, a.i2), dec_parse(b.i, b.i2));
}

inline bool datum_eq(const Datum& a, const Datum& b, uint8_t t) {
    if (t == RW_T_DECIMAL) {
        if (a.null != b.null) return false;
        if (a.null) return true;
        return datum_eq_impl_decimal(a, b);
    }
    if (a.null || b.null) return a.null == b.null;
    if (type_is_float(t)) {
        return a.d == b.d || (std::isnan(a.d) && std::isnan(b.d));
    }
    return a.i == b.i;
}

// DefaultOrd: total order, NULLs largest, NaN largest among floats
// (types/ord.rs DefaultOrd; sort_util.rs NullsAre::Largest default).
inline int datum_cmp(const Datum& a, const Datum& b, uint8_t t) {
    if (a.null || b.null) {
        if (a.null && b.null) return 0;
        return a.null ? 1 : -1; // NULL largest
    }
    if (type_is_float(t)) {
        bool an = std::isnan(a.d), bn = std::isnan(b.d);
        if (an || bn) return an == bn ? 0 : (an ? 1 : -1);
        return a.d < b.d ? -1 : (a.d > b.d ? 1 : 0);
    }
    return a.i < b.i ? -1 : (a.i > b.i ? 1 : 0);
}

using Row = std::vector<Datum>;

inline bool row_eq(const Row& a, const Row& b, const std::vector<uint8_t>& types) {
    if (a.size() != b.size()) return false;
    for (size_t i = 0; i < a.size(); i++)
        if (!datum_eq(a[i], b[i], types[i])) return false;
    return true;
}

// Order spec for a sequence of datums (e.g. minput cache key, join pk).
struct OrderCol {
    uint8_t type;
    bool desc = false; // nulls largest in the VALUE order; desc reverses values
};

// memcomparable order: per column, compare by (asc|desc) with NULLs largest
// IN THE STORED ORDER — i.e. for DESC columns NULL sorts FIRST
// (nulls_are_first = desc && nulls_are_largest, sort_util.rs:208-211).
inline int ordered_cmp(const Row& a, const Row& b, const std::vector<OrderCol>& order) {
    assert(a.size() == b.size() && a.size() == order.size());
    for (size_t i = 0; i < a.size(); i++) {
        int c = datum_cmp(a[i], b[i], order[i].type);
        if (order[i].desc) c = -c;
        if (c) return c;
    }
    return 0;
}

struct RowOrderLess {
    std::vector<OrderCol> order;
    bool operator()(const Row& a, const Row& b) const { return ordered_cmp(a, b, order) < 0; }
};

// ----- owned chunk (builder output / C-ABI marshalling) -----

struct OwnedChunk {
    std::vector<uint8_t> ops;
    std::vector<uint8_t> vis; // empty = all visible
    std::vector<uint8_t> types;
    std::vector<std::vector<Datum>> cols; // column-major
    size_t n_rows() const { return ops.size(); }
    Row row(size_t r) const {
        Row out(cols.size());
        for (size_t c = 0; c < cols.size(); c++) out[c] = cols[c][r];
        return out;
    }
    bool visible(size_t r) const { return vis.empty() || vis[r]; }
};

// View over an incoming C chunk.
struct ChunkView {
    const RwChunk* c;
    uint8_t op(size_t r) const { return c->ops[r]; }
    bool visible(size_t r) const { return c->vis == nullptr || c->vis[r]; }
    size_t n_rows() const { return c->n_rows; }
    size_t n_cols() const { return c->n_cols; }
    uint8_t type(size_t col) const { return c->cols[col].type; }
    Datum at(size_t r, size_t col) const {
        const RwColumn& cc = c->cols[col];
        if (!cc.valid[r]) return Datum();
        switch (cc.type) {
            case RW_T_I64:
            case RW_T_TS: return Datum::of_i(((const int64_t*)cc.data)[r]);
            case RW_T_DECIMAL: {
                const int64_t* p = (const int64_t*)cc.data;
                return Datum::of_dec(p[2 * r], p[2 * r + 1]);
            }
            case RW_T_I32: return Datum::of_i(((const int32_t*)cc.data)[r]);
            case RW_T_BOOL: return Datum::of_i(((const uint8_t*)cc.data)[r]);
            case RW_T_F64: return Datum::of_d(((const double*)cc.data)[r]);
            case RW_T_F32: return Datum::of_d(((const float*)cc.data)[r]);
            default: return Datum();
        }
    }
    Row row(size_t r) const {
        Row out(n_cols());
        for (size_t col = 0; col < n_cols(); col++) out[col] = at(r, col);
        return out;
    }
};

// StreamChunkBuilder (stream_chunk_builder.rs:188-218): yields a chunk when
// size reaches max_chunk_size, EXCEPT when the just-appended op is
// UpdateDelete — then it waits for the paired UpdateInsert (max+1 rows).
struct ChunkBuilder {
    size_t max_chunk_size;
    std::vector<uint8_t> types;
    OwnedChunk cur;
    ChunkBuilder(size_t max_size, std::vector<uint8_t> ts)
        : max_chunk_size(std::max<size_t>(max_size, 1)), types(std::move(ts)) {
        reset();
    }
    void reset() {
        cur = OwnedChunk();
        cur.types = types;
        cur.cols.assign(types.size(), {});
    }
    // returns true if a full chunk was produced into `out`
    bool append_row(uint8_t op, const Row& row, std::unique_ptr<OwnedChunk>* out) {
        assert(row.size() == types.size());
        cur.ops.push_back(op);
        for (size_t i = 0; i < row.size(); i++) cur.cols[i].push_back(row[i]);
        size_t size = cur.ops.size();
        if ((size == max_chunk_size && op != RW_OP_UPDATE_DELETE) || size > max_chunk_size) {
            *out = take();
            return true;
        }
        return false;
    }
    std::unique_ptr<OwnedChunk> take() {
        if (cur.ops.empty()) return nullptr;
        auto out = std::make_unique<OwnedChunk>(std::move(cur));
        reset();
        return out;
    }
};

// StreamChunk::eliminate_adjacent_noop_update (stream_chunk.rs:331-384).
// vnode of a key row (Crc32 of the hash_datum byte feed % vnode_count,
// consistent_hash/vnode.rs:146-181; NULL sentinel 0xfffffff0,
// array/mod.rs:99) — used by rescale re-scoping (update_vnode_bitmap)
inline uint32_t vnode_of_key_row(const Row& key,
                                 const std::vector<uint8_t>& key_types,
                                 uint32_t vnode_count) {
    static uint32_t tab[256];
    static bool init = false;
    if (!init) {
        for (uint32_t i = 0; i < 256; i++) {
            uint32_t c = i;
            for (int k = 0; k < 8; k++)
                c = (c & 1) ? 0xEDB88320u ^ (c >> 1) : c >> 1;
            tab[i] = c;
        }
        init = true;
    }
    auto feed = [&](uint32_t crc, const void* p, size_t n) {
        const uint8_t* b = (const uint8_t*)p;
        for (size_t i = 0; i < n; i++)
            crc = tab[(crc ^ b[i]) & 0xFF] ^ (crc >> 8);
        return crc;
    };
    uint32_t crc = 0xFFFFFFFFu;
    for (size_t k = 0; k < key.size(); k++) {
        const Datum& d = key[k];
        if (d.null) {
            uint32_t sentinel = 0xfffffff0u;
            crc = feed(crc, &sentinel, 4);
            continue;
        }
        switch (key_types[k]) {
            case RW_T_I32: {
                int32_t v = (int32_t)d.i;
                crc = feed(crc, &v, 4);
                break;
            }
            case RW_T_BOOL: {
                uint8_t v = (uint8_t)d.i;
                crc = feed(crc, &v, 1);
                break;
            }
            case RW_T_F64: {
                double v = d.d;
                crc = feed(crc, &v, 8);
                break;
            }
            case RW_T_F32: {
                float v = (float)d.d;
                crc = feed(crc, &v, 4);
                break;
            }
            default: { // I64 / TS (i64 micros)
                int64_t v = d.i;
                crc = feed(crc, &v, 8);
                break;
            }
        }
    }
    return (crc ^ 0xFFFFFFFFu) % vnode_count;
}

inline void eliminate_adjacent_noop_update(OwnedChunk& c) {
    size_t len = c.n_rows();
    if (c.vis.empty()) c.vis.assign(len, 1);
    long prev_r = -1;
    for (size_t curr = 0; curr < len; curr++) {
        if (!c.vis[curr]) continue;
        bool matched = false;
        if (prev_r >= 0) {
            uint8_t po = c.ops[prev_r], co = c.ops[curr];
            bool del_then_ins = (po == RW_OP_UPDATE_DELETE || po == RW_OP_DELETE) &&
                                (co == RW_OP_UPDATE_INSERT || co == RW_OP_INSERT);
            bool ins_then_del = (po == RW_OP_UPDATE_INSERT || po == RW_OP_INSERT) &&
                                (co == RW_OP_UPDATE_DELETE || co == RW_OP_DELETE);
            if ((del_then_ins || ins_then_del) &&
                row_eq(c.row(prev_r), c.row(curr), c.types)) {
                c.vis[prev_r] = 0;
                c.vis[curr] = 0;
                prev_r = -1;
                matched = true;
            }
        }
        if (!matched) prev_r = (long)curr;
    }
    // Normalize half-visible update pairs.
    for (size_t idx = 0; idx + 1 < len; idx++) {
        if (c.ops[idx] == RW_OP_UPDATE_DELETE && c.ops[idx + 1] == RW_OP_UPDATE_INSERT) {
            bool dv = c.vis[idx], iv = c.vis[idx + 1];
            if (dv && !iv) c.ops[idx] = RW_OP_DELETE;
            else if (!dv && iv) c.ops[idx + 1] = RW_OP_INSERT;
        }
    }
    // drop the vis vector if everything is visible (cosmetic)
    bool all = true;
    for (auto v : c.vis)
        if (!v) { all = false; break; }
    if (all) c.vis.clear();
}

// ----- C-ABI marshalling of an OwnedChunk -----

inline RwChunk* chunk_to_c(const OwnedChunk& oc) {
    size_t n = oc.n_rows(), m = oc.cols.size();
    // single allocation block layout: RwChunk, RwColumn[m], then arrays
    auto* ch = new RwChunk();
    auto* cols = new RwColumn[m];
    auto* ops = new uint8_t[n];
    std::memcpy(ops, oc.ops.data(), n);
    uint8_t* vis = nullptr;
    if (!oc.vis.empty()) {
        vis = new uint8_t[n];
        std::memcpy(vis, oc.vis.data(), n);
    }
    for (size_t c = 0; c < m; c++) {
        uint8_t t = oc.types[c];
        auto* valid = new uint8_t[n];
        void* data = nullptr;
        switch (t) {
            case RW_T_I64:
            case RW_T_TS: {
                auto* p = new int64_t[n];
                for (size_t r = 0; r < n; r++) {
                    valid[r] = !oc.cols[c][r].null;
                    p[r] = valid[r] ? oc.cols[c][r].i : 0;
                }
                data = p;
                break;
            }
            case RW_T_I32: {
                auto* p = new int32_t[n];
                for (size_t r = 0; r < n; r++) {
                    valid[r] = !oc.cols[c][r].null;
                    p[r] = valid[r] ? (int32_t)oc.cols[c][r].i : 0;
                }
                data = p;
                break;
            }
            case RW_T_BOOL: {
                auto* p = new uint8_t[n];
                for (size_t r = 0; r < n; r++) {
                    valid[r] = !oc.cols[c][r].null;
                    p[r] = valid[r] ? (uint8_t)oc.cols[c][r].i : 0;
                }
                data = p;
                break;
            }
            case RW_T_DECIMAL: {
                auto* p = new int64_t[2 * n];
                for (size_t r = 0; r < n; r++) {
                    valid[r] = !oc.cols[c][r].null;
                    p[2 * r] = valid[r] ? oc.cols[c][r].i : 0;
                    p[2 * r + 1] = valid[r] ? oc.cols[c][r].i2 : 0;
                }
                data = p;
                break;
            }
            case RW_T_F64: {
                auto* p = new double[n];
                for (size_t r = 0; r < n; r++) {
                    valid[r] = !oc.cols[c][r].null;
                    p[r] = valid[r] ? oc.cols[c][r].d : 0;
                }
                data = p;
                break;
            }
            case RW_T_F32: {
                auto* p = new float[n];
                for (size_t r = 0; r < n; r++) {
                    valid[r] = !oc.cols[c][r].null;
                    p[r] = valid[r] ? (float)oc.cols[c][r].d : 0;
                }
                data = p;
                break;
            }
        }
        cols[c].type = t;
        cols[c].valid = valid;
        cols[c].data = data;
    }
    ch->n_rows = (uint32_t)n;
    ch->n_cols = (uint32_t)m;
    ch->ops = ops;
    ch->vis = vis;
    ch->cols = cols;
    return ch;
}

inline void chunk_free_c(RwChunk* ch) {
    if (!ch) return;
    for (uint32_t c = 0; c < ch->n_cols; c++) {
        delete[] ch->cols[c].valid;
        switch (ch->cols[c].type) {
            case RW_T_I64:
            case RW_T_TS: delete[] (int64_t*)ch->cols[c].data; break;
            case RW_T_DECIMAL: delete[] (int64_t*)ch->cols[c].data; break;
            case RW_T_I32: delete[] (int32_t*)ch->cols[c].data; break;
            case RW_T_BOOL: delete[] (uint8_t*)ch->cols[c].data; break;
            case RW_T_F64: delete[] (double*)ch->cols[c].data; break;
            case RW_T_F32: delete[] (float*)ch->cols[c].data; break;
        }
    }
    delete[] ch->cols;
    delete[] ch->ops;
    delete[] ch->vis;
    delete ch;
}

} // namespace orc

namespace orc {

// Re-emit a spill stream ([put u8][klen u32 LE][key][vlen u32 LE][val]
// frames) in memcomparable-key order (stable: same-key PUT/DELETE within an
// epoch keep their relative order) — the canonical drain order both the GPU
// library and this oracle emit, enabling byte-compare parity.
inline void sort_spill_frames(std::vector<uint8_t>& sp) {
    struct Frame { const uint8_t* p; size_t n; };
    std::vector<Frame> frames;
    size_t off = 0;
    auto rd32 = [&](size_t o) {
        return (uint32_t)sp[o] | ((uint32_t)sp[o + 1] << 8) |
               ((uint32_t)sp[o + 2] << 16) | ((uint32_t)sp[o + 3] << 24);
    };
    while (off + 5 <= sp.size()) {
        size_t start = off;
        uint32_t klen = rd32(off + 1);
        off += 5 + klen;
        if (off + 4 > sp.size()) return; // malformed: leave unsorted
        uint32_t vlen = rd32(off);
        off += 4 + vlen;
        if (off > sp.size()) return;
        frames.push_back({sp.data() + start, off - start});
    }
    std::stable_sort(frames.begin(), frames.end(),
                     [](const Frame& a, const Frame& b) {
        uint32_t ka = (uint32_t)a.p[1] | ((uint32_t)a.p[2] << 8) |
                      ((uint32_t)a.p[3] << 16) | ((uint32_t)a.p[4] << 24);
        uint32_t kb = (uint32_t)b.p[1] | ((uint32_t)b.p[2] << 8) |
                      ((uint32_t)b.p[3] << 16) | ((uint32_t)b.p[4] << 24);
        int c = memcmp(a.p + 5, b.p + 5, ka < kb ? ka : kb);
        if (c) return c < 0;
        return ka < kb;
    });
    std::vector<uint8_t> out;
    out.reserve(sp.size());
    for (auto& f : frames) out.insert(out.end(), f.p, f.p + f.n);
    sp.swap(out);
}

} // namespace orc
