// rw_amd.hip — MI355X-native (gfx950) stream HashAgg + HashJoin executors.
//
// PRODUCT PATH. Implements the C-ABI of include/rw_stream.h with all state
// resident in HBM and all per-row work in HIP kernels. The
// reference semantics being implemented (cited per function below) are those
// of /root/reference/src/stream/src/executor/{aggregate/hash_agg.rs,
// hash_join.rs}; the DESIGN is MI355X-first: SoA layouts, coalesced loads,
// open-addressed HBM tables, wave-level pre-aggregation, atomic-cursor
// emission (epoch outputs are order-free row multisets, matching the
// reference's own nondeterminism bar).
//
// There is NO CPU fallback: every entry point requires a visible GPU and
// fails loudly otherwise (RW_E_NOGPU).
//
// Round-1 kernel scope: group keys / join keys of 1..4 i64
// words (i64/timestamp columns), agg calls count(*)/count/sum/sum0 (i64) and
// append-only min/max (i64); join types Inner (incl. append-only) with
// optional i64 comparison condition. Everything else returns RW_E_INVAL with
// a message — the oracle covers it until the kernels land.

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <map>
#include <optional>
#include <string>
#include <unordered_set>
#include <vector>

#include "../../include/rw_codec.hpp"
#include "../../include/rw_stream.h"

#define WAVE 64

static thread_local std::string g_err;

#define FAIL(code, ...)                                  \
    do {                                                 \
        char _b[256];                                    \
        snprintf(_b, sizeof _b, __VA_ARGS__);            \
        g_err = _b;                                      \
        return code;                                     \
    } while (0)

#define HIP_TRY(x)                                                          \
    do {                                                                    \
        hipError_t _e = (x);                                                \
        if (_e != hipSuccess)                                               \
            FAIL(RW_E_INTERNAL, "HIP error %s at %s:%d", hipGetErrorString(_e), \
                 __FILE__, __LINE__);                                       \
    } while (0)

static bool gpu_ok() {
    int n = 0;
    return hipGetDeviceCount(&n) == hipSuccess && n > 0;
}

// ---------------------------------------------------------------------------
// device helpers
// ---------------------------------------------------------------------------

__device__ __forceinline__ uint64_t mix64(uint64_t x) {
    // splitmix64 finalizer — internal hash placement only (free choice per
    // XxHash64 is cache placement, not output-visible)
    x += 0x9e3779b97f4a7c15ULL;
    x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ULL;
    x = (x ^ (x >> 27)) * 0x94d049bb133111ebULL;
    return x ^ (x >> 31);
}

__device__ __forceinline__ uint64_t hash_key(const int64_t* kw, uint32_t nullmask,
                                             int KW) {
    uint64_t h = 0x20210401u ^ (uint64_t)nullmask * 0x9e3779b97f4a7c15ULL;
    for (int i = 0; i < KW; i++) h = mix64(h ^ (uint64_t)kw[i]);
    return h;
}

__device__ __forceinline__ long long atomic_add_i64(long long* p, long long v) {
    return (long long)atomicAdd((unsigned long long*)p, (unsigned long long)v);
}

__device__ __forceinline__ void atomic_min_i64(long long* p, long long v) {
    atomicMin(p, v);
}
__device__ __forceinline__ void atomic_max_i64(long long* p, long long v) {
    atomicMax(p, v);
}

// ---- decimal (device mirror of the exact-domain restatement in
// oracle/common.hpp; rust_decimal 1.40.0 layout — see include/rw_chunk.h
// RW_T_DECIMAL and DESIGN.md §9). Sums accumulate as exact signed 256-bit
// integers at scale 28 via per-word atomics with cascading carries. ----
struct DecValD {
    int special; // 0 normal, 1 NaN, 2 +Inf, 3 -Inf
    bool neg;
    uint32_t scale;
    uint64_t lo;
    uint32_t hi;
};

__device__ __forceinline__ DecValD dec_parse_d(int64_t a, int64_t b) {
    DecValD v{};
    uint32_t flags = (uint32_t)a;
    uint8_t b0 = (uint8_t)flags;
    if (b0 == 1 || b0 == 2 || b0 == 3) {
        v.special = b0;
        return v;
    }
    v.neg = (flags >> 31) & 1;
    v.scale = (flags >> 16) & 0xFF;
    uint32_t lo32 = (uint32_t)((uint64_t)a >> 32);
    uint32_t mid = (uint32_t)b;
    v.hi = (uint32_t)((uint64_t)b >> 32);
    v.lo = ((uint64_t)mid << 32) | lo32;
    return v;
}

__device__ __forceinline__ void dec_serialize_d(const DecValD& v, long long* a,
                                                long long* b) {
    if (v.special) {
        *a = (long long)(uint32_t)v.special;
        *b = 0;
        return;
    }
    uint32_t flags = (v.scale << 16) | ((uint32_t)v.neg << 31);
    *a = (long long)(((uint64_t)(uint32_t)v.lo << 32) | flags);
    *b = (long long)(((uint64_t)v.hi << 32) | (uint32_t)(v.lo >> 32));
}

__device__ __forceinline__ uint64_t dec_pow10_d(int k) {
    uint64_t p = 1;
    while (k-- > 0) p *= 10;
    return p;
}

// addend words = sign * mantissa96 * 10^(28-scale) (two's complement i256)
__device__ __forceinline__ void dec_addend_d(const DecValD& v, uint64_t w[4]) {
    int k = 28 - (int)v.scale;
    uint64_t pa = dec_pow10_d(k / 2), pb = dec_pow10_d(k - k / 2);
    w[0] = v.lo;
    w[1] = v.hi;
    w[2] = w[3] = 0;
    for (int t = 0; t < 2; t++) {
        uint64_t p = t ? pb : pa;
        unsigned __int128 c = 0;
        for (int i = 0; i < 4; i++) {
            unsigned __int128 x = (unsigned __int128)w[i] * p + c;
            w[i] = (uint64_t)x;
            c = x >> 64;
        }
    }
    if (v.neg) {
        unsigned __int128 c = 1;
        for (int i = 0; i < 4; i++) {
            unsigned __int128 x = (unsigned __int128)(~w[i]) + c;
            w[i] = (uint64_t)x;
            c = x >> 64;
        }
    }
}

// per-word atomic accumulation with cascading carries (mod 2^256 two's
// complement — negative addends just work)
__device__ __forceinline__ void dec_i256_atomic_add(
    unsigned long long* base, size_t cap, uint32_t slot, const uint64_t w[4]) {
    unsigned long long carry = 0;
    for (int k = 0; k < 4; k++) {
        unsigned long long add = (unsigned long long)w[k] + carry;
        carry = add < w[k] ? 1 : 0;
        if (add) {
            unsigned long long old = atomicAdd(&base[(size_t)k * cap + slot],
                                               add);
            if (old + add < add) carry += 1;
        }
    }
}

// exact conversion of the slot's i256 sum back to a decimal at out_scale;
// false when the mantissa exceeds 96 bits (the reference's order-dependent
// precision-loss rescale domain)
__device__ __forceinline__ bool dec_from_sum_d(const unsigned long long* base,
                                               size_t cap, uint32_t slot,
                                               uint32_t out_scale,
                                               DecValD* out) {
    uint64_t a[4];
    for (int k = 0; k < 4; k++) a[k] = base[(size_t)k * cap + slot];
    bool neg = a[3] >> 63;
    if (neg) {
        unsigned __int128 c = 1;
        for (int k = 0; k < 4; k++) {
            unsigned __int128 x = (unsigned __int128)(~a[k]) + c;
            a[k] = (uint64_t)x;
            c = x >> 64;
        }
    }
    int k = 28 - (int)out_scale;
    for (int t = 0; t < 2; t++) {
        uint64_t p = t ? dec_pow10_d(k - k / 2) : dec_pow10_d(k / 2);
        if (p == 1) continue;
        unsigned __int128 rem = 0;
        for (int i = 3; i >= 0; i--) {
            unsigned __int128 cur = (rem << 64) | a[i];
            a[i] = (uint64_t)(cur / p);
            rem = cur % p;
        }
        if (rem != 0) return false;
    }
    if (a[3] || a[2] || (a[1] >> 32)) return false;
    out->special = 0;
    out->neg = neg && (a[0] | a[1]);
    out->scale = out_scale;
    out->lo = a[0];
    out->hi = (uint32_t)a[1];
    return true;
}

// rust_decimal PartialEq compares VALUES (1.2 == 1.20)
__device__ __forceinline__ bool dec_value_eq_d(int64_t a0, int64_t a1,
                                               int64_t b0, int64_t b1) {
    DecValD x = dec_parse_d(a0, a1), y = dec_parse_d(b0, b1);
    if (x.special || y.special) return x.special == y.special;
    uint64_t wx[4], wy[4];
    dec_addend_d(x, wx);
    dec_addend_d(y, wy);
    return wx[0] == wy[0] && wx[1] == wy[1] && wx[2] == wy[2] &&
           wx[3] == wy[3];
}

// slot states
#define SLOT_EMPTY 0u
#define SLOT_CLAIMED 1u
#define SLOT_READY 2u

// All cross-lane-shared table words use relaxed AGENT-scope atomics (sc1:
// L1-bypassing, write-through to the coherent point) — the guide's R1/R2
// hand-off forms. The claim path is R1: sc1 key stores -> s_waitcnt vmcnt(0)
// drain -> sc1 READY store; readers use sc1 loads throughout, which per the
// microarch visibility table replace the acquire when the producer stored
// sc1. No acquire/release fences anywhere on the hot path.
#define RLX __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT

__device__ __forceinline__ uint32_t ld_u32(const uint32_t* p) {
    return __hip_atomic_load((uint32_t*)p, RLX);
}
__device__ __forceinline__ void st_u32(uint32_t* p, uint32_t v) {
    __hip_atomic_store(p, v, RLX);
}
__device__ __forceinline__ int64_t ld_i64(const int64_t* p) {
    return (int64_t)__hip_atomic_load((unsigned long long*)p, RLX);
}
__device__ __forceinline__ void st_i64(int64_t* p, int64_t v) {
    __hip_atomic_store((unsigned long long*)p, (unsigned long long)v, RLX);
}

// Find-or-insert into an open-addressed key table (linear probe).
// key_nulls is a u32-per-slot null mask. Returns slot, or -1 on table-full.
__device__ __forceinline__ uint32_t table_find_or_insert(
    uint32_t* state, int64_t* keys, uint32_t* key_nulls, uint32_t cap_mask,
    const int64_t* kw, uint32_t nullmask, int KW) {
    uint32_t slot = (uint32_t)(hash_key(kw, nullmask, KW) & cap_mask);
    for (uint32_t probes = 0; probes <= cap_mask; probes++) {
        uint32_t st = ld_u32(&state[slot]);
        if (st == SLOT_EMPTY) {
            uint32_t prev = atomicCAS(&state[slot], SLOT_EMPTY, SLOT_CLAIMED);
            if (prev == SLOT_EMPTY) {
                for (int i = 0; i < KW; i++) st_i64(&keys[(size_t)slot * KW + i], kw[i]);
                st_u32(&key_nulls[slot], nullmask);
                asm volatile("s_waitcnt vmcnt(0)" ::: "memory"); // R1 drain
                st_u32(&state[slot], SLOT_READY);
                return slot;
            }
            st = prev;
        }
        while (st == SLOT_CLAIMED) {
            __builtin_amdgcn_s_sleep(1);
            st = ld_u32(&state[slot]);
        }
        // st == SLOT_READY
        bool eq = ld_u32(&key_nulls[slot]) == nullmask;
        for (int i = 0; eq && i < KW; i++)
            eq = ld_i64(&keys[(size_t)slot * KW + i]) == kw[i];
        if (eq) return slot;
        slot = (slot + 1) & cap_mask;
    }
    return (uint32_t)-1;
}

// Find-only (no insert). Returns slot or -1.
__device__ __forceinline__ uint32_t table_find(const uint32_t* state,
                                               const int64_t* keys,
                                               const uint32_t* key_nulls,
                                               uint32_t cap_mask, const int64_t* kw,
                                               uint32_t nullmask, int KW) {
    uint32_t slot = (uint32_t)(hash_key(kw, nullmask, KW) & cap_mask);
    for (uint32_t probes = 0; probes <= cap_mask; probes++) {
        uint32_t st = ld_u32(&state[slot]);
        if (st == SLOT_EMPTY) return (uint32_t)-1;
        if (st == SLOT_READY) {
            bool eq = ld_u32(&key_nulls[slot]) == nullmask;
            for (int i = 0; eq && i < KW; i++)
                eq = ld_i64(&keys[(size_t)slot * KW + i]) == kw[i];
            if (eq) return slot;
        }
        slot = (slot + 1) & cap_mask;
    }
    return (uint32_t)-1;
}

// ---------------------------------------------------------------------------
// HashAgg
// ---------------------------------------------------------------------------

#define MAX_CALLS 8
#define MAX_KW 4

struct AggCallDev {
    uint8_t kind;
    int32_t arg;  // column index in the input BATCH (gk cols ∥ arg cols ∥ sk cols)
    uint8_t minput; // materialized-input state (retractable min/max)
    int8_t mord;    // ordinal among minput calls
    uint8_t decimal; // RW_T_DECIMAL sum/count argument (16-B datums)
    int8_t dord;     // ordinal among decimal calls
};

// Device-side batch of input rows (SoA), i64-widened values. `stride` (in
// elements) lets the batch view row-major buffers — e.g. the join's output
// block, where column c of row r sits at vals[r*n_out + c] — without a
// repack; `valid_inverted` views a null-flag array as validity.
struct AggBatch {
    int64_t* col_vals[MAX_KW + MAX_CALLS]; // group key cols then arg cols
    uint8_t* col_valid[MAX_KW + MAX_CALLS];
    uint8_t* ops;
    uint8_t* vis; // may be null
    uint32_t n_rows;
    uint32_t stride = 1;
    uint8_t valid_inverted = 0;
    uint8_t dense = 0; // all rows visible+Insert+non-null (checked at upload)
    // DISTINCT dedup visibility (aggregate/distinct.rs): per call, byte per
    // row (indexed by r*stride) — 1 = hidden duplicate; null = no dedup
    const uint8_t* call_hidden[MAX_CALLS] = {};
};

// AoS layouts: one 64-B line per key slot, one packed record per row —
// a random probe touches 1 table line + 1 record line instead of the 8–10
// lines the SoA arrays cost (DESIGN §8 lever).
struct alignas(64) JoinSlot {
    uint32_t state; // SLOT_EMPTY/CLAIMED/READY
    uint32_t head;  // chain head row index, UINT32_MAX = none
    uint32_t nulls; // key null mask
    uint32_t _pad;
    long long key[4];
};
static_assert(sizeof(JoinSlot) == 64, "one cache line per slot");

// row record header; vals[n_cols] i64 follow at offset 16
struct JoinRowHdr {
    uint32_t alive; // CAS-claimed tombstone
    uint32_t next;
    uint32_t validbits; // bit c = column c non-NULL
    uint32_t degree;    // matches on the other side (outer/semi/anti types)
};

struct JoinSideDev {
    JoinSlot* slots;     // 64-B AoS slots (GroupTopN tables)
    uint64_t* slots8;    // compact 8-B slots (join executor tables; see below)
    uint32_t cap_mask;
    uint8_t* rows;       // packed records, row_stride bytes each (16-B aligned)
    uint32_t row_stride;  // 16 + 8*n_cols
    uint32_t* row_cursor; // single counter
    uint32_t row_cap;
    // checkpoint-delta tracking (§8f-2): rows killed by deletes this epoch
    // (null for executors without spill, e.g. GroupTopN this round)
    uint32_t* killed;
    uint32_t* killed_cursor;
    uint32_t killed_cap;
    // degree-table delta tracking (§8f-2): pre-epoch rows whose degree
    // changed this epoch (null when the join type needs no degree here)
    uint32_t* deg_dirty_flag;
    uint32_t* deg_dirty_list;
    uint32_t* deg_dirty_n;
};

__device__ __forceinline__ JoinRowHdr* jrow(const JoinSideDev& s, uint32_t r) {
    return (JoinRowHdr*)(s.rows + (size_t)r * s.row_stride);
}
__device__ __forceinline__ long long* jvals(JoinRowHdr* h) {
    return (long long*)((uint8_t*)h + 16);
}

// First slot for a join-table probe: TOP bits of the hash, so the slot index
// is monotone in h and a hash-prefix partition maps to a CONTIGUOUS slot
// region — the locality lever of the partitioned probe/insert pipeline
// (jpart_* kernels below). Distribution-equivalent to h & cap_mask.
__device__ __forceinline__ uint32_t jslot_start(uint64_t h, uint32_t cap_mask) {
    return (uint32_t)(h >> (64 - __popc(cap_mask)));
}

// ---- compact 8-B join slots: CHAINED buckets with bloom tags ----------
// The join executor's slot table packs {head u32 (high), bloom u32 (low)}
// into one u64; bloom==0 means empty. The KEY WORDS live only in the row
// records, and a bucket CHAINS every row whose hash lands on the slot —
// chained hashing, not open addressing. Rationale (round-2 roofline work):
// the round-1 64-B open-addressed slots made every probe/insert touch
// 1–3 random HBM lines (PMC: 500–1000 B/row vs the 128-B model), and even
// compact open addressing needs a verify read of the head record per
// insert. With buckets:
//  - INSERT = one 64-bit CAS on the slot (8-B table, 128–256 MB at q8
//    capacity -> Infinity-Cache-resident) — no probing, no verify read.
//  - MISS   = one slot load: the bloom bit (bit (h>>32)&31, set by every
//    insert, never cleared) rejects absent keys with ~k/32 false-positive
//    walks for a k-key bucket.
//  - MATCH  = walk the chain and filter by FULL key from the records the
//    match reads anyway (no tag-identity assumption anywhere).
// A bucket may mix keys (same-slot hash collisions), so every chain walk
// carries the key filter; deletes/cleans kill records individually.

__device__ __forceinline__ uint32_t jbloom_bit(uint64_t h) {
    return 1u << ((uint32_t)(h >> 32) & 31);
}
__device__ __forceinline__ uint32_t jhead_of(uint64_t packed) {
    return (uint32_t)(packed >> 32);
}
// head row of the bucket for hash h, or UINT32_MAX if the bloom rejects.
// sc1: coherent slot load (the side is mutated during this launch).
__device__ __forceinline__ uint32_t jbucket_head(const JoinSideDev& sd,
                                                 uint64_t h, bool sc1) {
    uint32_t slot = jslot_start(h, sd.cap_mask);
    uint64_t packed = sc1 ? __hip_atomic_load(&sd.slots8[slot], RLX)
                          : sd.slots8[slot];
    if (!((uint32_t)packed & jbloom_bit(h))) return UINT32_MAX;
    return jhead_of(packed);
}
// push row onto its bucket: record payload fully written by the caller
// (drained to the coherence point before the CAS lands, so a walker that
// sees the new head fetches real bytes). sc1_next: the launch has
// same-side walkers (mixed-op chunks) — store `next` coherently.
__device__ __forceinline__ void jbucket_insert(const JoinSideDev& sd,
                                               uint64_t h, uint32_t row,
                                               bool sc1_next) {
    uint32_t slot = jslot_start(h, sd.cap_mask);
    JoinRowHdr* hd = jrow(sd, row);
    uint64_t old = __hip_atomic_load(&sd.slots8[slot], RLX);
    for (;;) {
        uint32_t old_head =
            (uint32_t)old ? jhead_of(old) : UINT32_MAX; // bloom 0 = empty
        if (sc1_next) {
            st_u32(&hd->next, old_head);
            // R1 drain: same-launch walkers (mixed-op chunks) must fetch
            // real payload bytes once they see the new head. All-Insert
            // launches have NO same-side walkers — their visibility comes
            // from the inter-dispatch flush, so the pipeline-stalling
            // drain is skipped (sc1_next == false).
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        } else {
            hd->next = old_head;
        }
        uint64_t want =
            ((uint64_t)row << 32) | ((uint32_t)old | jbloom_bit(h));
        uint64_t prev = atomicCAS((unsigned long long*)&sd.slots8[slot], old,
                                  want);
        if (prev == old) return;
        old = prev;
    }
}

// key filter for chain walks given the record header pointer (buckets mix
// keys; every walk filters by full key)
__device__ __forceinline__ bool jhdr_key_eq(JoinRowHdr* hd,
                                            const uint8_t* key_cols, int KW,
                                            const int64_t* kw,
                                            uint32_t nullmask, bool sc1) {
    uint32_t vb = sc1 ? ld_u32(&hd->validbits) : hd->validbits;
    const long long* hv = (const long long*)((const uint8_t*)hd + 16);
    for (int i = 0; i < KW; i++) {
        uint8_t col = key_cols[i];
        bool valid = (vb >> col) & 1;
        if (valid == (bool)((nullmask >> i) & 1)) return false;
        if (valid) {
            long long v = sc1 ? ld_i64((const int64_t*)&hv[col]) : hv[col];
            if (v != kw[i]) return false;
        }
    }
    return true;
}

// own-side find-or-insert. The probe walk uses PLAIN cached loads: a slot's
// state and keys share one 64-B line, and the claim protocol drains the sc1
// key stores to the coherence point BEFORE the sc1 READY store — so any
// line fill that observes READY also contains the keys (same-line, written
// earlier at the coherence point), and a stale cached line can only show
// the older EMPTY/CLAIMED state, which funnels into the coherent CAS/spin
// path below. Keys never change once READY. This removes the per-visit sc1
// word loads that dominated the insert path (~0.45 ms/1M inserts measured).
__device__ __forceinline__ uint32_t jslot_find_or_insert(JoinSlot* slots,
                                                         uint32_t cap_mask,
                                                         const int64_t* kw,
                                                         uint32_t nullmask,
                                                         int KW) {
    uint32_t slot = jslot_start(hash_key(kw, nullmask, KW), cap_mask);
    for (uint32_t probes = 0; probes <= cap_mask; probes++) {
        JoinSlot* sl = &slots[slot];
        uint32_t st = sl->state; // plain (fast path)
        if (st == SLOT_READY) {
            bool eq = sl->nulls == nullmask;
            for (int i = 0; eq && i < KW; i++) eq = sl->key[i] == kw[i];
            if (eq) return slot;
        } else {
            if (st == SLOT_EMPTY) {
                uint32_t prev = atomicCAS(&sl->state, SLOT_EMPTY, SLOT_CLAIMED);
                if (prev == SLOT_EMPTY) {
                    for (int i = 0; i < KW; i++)
                        st_i64((int64_t*)&sl->key[i], kw[i]);
                    st_u32(&sl->nulls, nullmask);
                    asm volatile("s_waitcnt vmcnt(0)" ::: "memory"); // R1 drain
                    st_u32(&sl->state, SLOT_READY);
                    return slot;
                }
                st = prev;
            } else {
                st = ld_u32(&sl->state); // sc1 refresh of a stale CLAIMED
            }
            while (st == SLOT_CLAIMED) {
                __builtin_amdgcn_s_sleep(1);
                st = ld_u32(&sl->state);
            }
            // coherent compare (the plain line may be stale here)
            bool eq = ld_u32(&sl->nulls) == nullmask;
            for (int i = 0; eq && i < KW; i++)
                eq = ld_i64((const int64_t*)&sl->key[i]) == kw[i];
            if (eq) return slot;
        }
        slot = (slot + 1) & cap_mask;
    }
    return (uint32_t)-1;
}

// match-side find with plain cached loads (immutable during the launch)
__device__ __forceinline__ uint32_t jslot_find_cached(const JoinSlot* slots,
                                                      uint32_t cap_mask,
                                                      const int64_t* kw,
                                                      uint32_t nullmask, int KW) {
    uint32_t slot = jslot_start(hash_key(kw, nullmask, KW), cap_mask);
    for (uint32_t probes = 0; probes <= cap_mask; probes++) {
        const JoinSlot* sl = &slots[slot];
        uint32_t st = sl->state;
        if (st == SLOT_EMPTY) return (uint32_t)-1;
        if (st == SLOT_READY) {
            bool eq = sl->nulls == nullmask;
            for (int i = 0; eq && i < KW; i++) eq = sl->key[i] == kw[i];
            if (eq) return slot;
        }
        slot = (slot + 1) & cap_mask;
    }
    return (uint32_t)-1;
}

// sc1 variant of the find (append-only path mutates the match side)
__device__ __forceinline__ uint32_t jslot_find_sc1(const JoinSlot* slots,
                                                   uint32_t cap_mask,
                                                   const int64_t* kw,
                                                   uint32_t nullmask, int KW) {
    uint32_t slot = jslot_start(hash_key(kw, nullmask, KW), cap_mask);
    for (uint32_t probes = 0; probes <= cap_mask; probes++) {
        const JoinSlot* sl = &slots[slot];
        uint32_t st = ld_u32(&sl->state);
        if (st == SLOT_EMPTY) return (uint32_t)-1;
        if (st == SLOT_READY) {
            bool eq = ld_u32(&sl->nulls) == nullmask;
            for (int i = 0; eq && i < KW; i++)
                eq = ld_i64((const int64_t*)&sl->key[i]) == kw[i];
            if (eq) return slot;
        }
        slot = (slot + 1) & cap_mask;
    }
    return (uint32_t)-1;
}

__global__ void jslot_init_kernel(JoinSlot* slots, size_t cap) {
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (size_t i = blockIdx.x * blockDim.x + threadIdx.x; i < cap; i += stride) {
        slots[i].state = SLOT_EMPTY;
        slots[i].head = UINT32_MAX;
    }
}

// DISTINCT dedup pass (aggregate/distinct.rs:131-158): per (group key,
// datum) a counter slot (JoinSlot layout; `head` is the count, initialized
// 0). An insert is visible iff the count rises 0→1, a delete iff it falls
// 1→0 — transition ownership via the atomic's return value, so concurrent
// duplicates resolve to the same net visibility multiset as the
// reference's sequential pass (the accumulators only see the net).
__global__ void dedup_init_kernel(JoinSlot* slots, size_t cap) {
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (size_t i = blockIdx.x * blockDim.x + threadIdx.x; i < cap; i += stride) {
        slots[i].state = SLOT_EMPTY;
        slots[i].head = 0; // count
    }
}

__global__ void agg_dedup_kernel(AggBatch b, JoinSlot* slots,
                                 uint32_t cap_mask, int KW, int dslot,
                                 uint8_t* hidden, uint32_t* err,
                                 uint32_t* dirty_flag, uint32_t* dirty_list,
                                 uint32_t* dirty_n) {
    uint32_t stride = gridDim.x * blockDim.x;
    for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < b.n_rows;
         r += stride) {
        size_t rs = (size_t)r * b.stride;
        if (b.vis && !b.vis[r]) continue;
        int64_t kw[MAX_KW];
        uint32_t nullmask = 0;
        for (int i = 0; i < KW; i++) {
            bool valid = b.col_valid[i][rs] ^ b.valid_inverted;
            kw[i] = valid ? b.col_vals[i][rs] : 0;
            nullmask |= (!valid) << i;
        }
        bool dvalid = b.col_valid[dslot][rs] ^ b.valid_inverted;
        kw[KW] = dvalid ? b.col_vals[dslot][rs] : 0;
        nullmask |= (!dvalid) << KW;
        uint32_t slot =
            jslot_find_or_insert(slots, cap_mask, kw, nullmask, KW + 1);
        if (slot == UINT32_MAX) {
            atomicExch(err, 4u); // dedup table full
            continue;
        }
        uint8_t op = b.ops[r];
        bool retract = (op == RW_OP_DELETE || op == RW_OP_UPDATE_DELETE);
        uint32_t old = retract ? atomicSub(&slots[slot].head, 1u)
                               : atomicAdd(&slots[slot].head, 1u);
        hidden[rs] = retract ? (old != 1) : (old != 0);
        // §8f-2 spill: record each (group, datum) touched this epoch, once
        // (the reference writes an update per touched datum at dedup(),
        // distinct.rs:158-185; the per-epoch drain nets those to one)
        if (ld_u32(&dirty_flag[slot]) == 0 &&
            atomicCAS(&dirty_flag[slot], 0u, 1u) == 0u) {
            uint32_t i = atomicAdd(dirty_n, 1u);
            dirty_list[i] = slot;
        }
    }
}

// Dedup-table checkpoint gather: pack the dirty slots' (key, count) into a
// compact buffer for one D2H copy, clearing the dirty flags on the way.
struct DedupDirtyRec {
    long long key[MAX_KW]; // group key words ++ datum (KW+1 used)
    uint32_t nulls;
    uint32_t count;
};
__global__ void dedup_gather_kernel(const JoinSlot* slots,
                                    uint32_t* dirty_flag,
                                    const uint32_t* dirty_list,
                                    const uint32_t* dirty_n,
                                    DedupDirtyRec* out) {
    uint32_t n = *dirty_n;
    uint32_t stride = gridDim.x * blockDim.x;
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride) {
        uint32_t s = dirty_list[i];
        const JoinSlot* sl = &slots[s];
        for (int w = 0; w < MAX_KW; w++) out[i].key[w] = sl->key[w];
        out[i].nulls = sl->nulls;
        out[i].count = sl->head;
        dirty_flag[s] = 0;
    }
}

// §8f-5 recovery: seed dedup counter slots from a netted spill replay.
// Keys in the netted map are unique, so each lane owns its slot; the slot
// index is reported back so the host can mark it persisted (the next drain
// must emit DELETE, not silence, when a restored count dies).
__global__ void dedup_restore_kernel(const long long* keys,
                                     const uint32_t* nulls,
                                     const uint32_t* counts, uint32_t n,
                                     int KW1, JoinSlot* slots,
                                     uint32_t cap_mask, uint32_t* out_slots,
                                     uint32_t* err) {
    uint32_t stride = gridDim.x * blockDim.x;
    for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < n;
         r += stride) {
        int64_t kw[MAX_KW];
        for (int i = 0; i < KW1; i++) kw[i] = keys[(size_t)r * KW1 + i];
        uint32_t slot =
            jslot_find_or_insert(slots, cap_mask, kw, nulls[r], KW1);
        if (slot == UINT32_MAX) {
            atomicExch(err, 4u);
            continue;
        }
        slots[slot].head = counts[r];
        out_slots[r] = slot;
    }
}

struct AggTableDev {
    uint32_t* state;
    int64_t* keys;      // [cap * KW]
    uint32_t* key_nulls; // [cap]
    long long* acc;     // [n_calls][cap]
    uint8_t* has;       // [n_calls][cap] — any non-null input applied
    long long* prev;    // [n_calls][cap]
    uint8_t* prev_null; // [n_calls][cap]
    uint8_t* has_prev;  // [cap]
    uint32_t* dirty_flag;
    uint32_t* dirty_list;
    uint32_t* counters; // [0]=dirty_count [1]=out_cursor [2]=overflow flag
    // flush output buffers (records of width gk+calls)
    long long* out_vals;
    uint8_t* out_nulls;
    uint8_t* out_ops;
    uint32_t out_capacity;
    uint32_t cap_mask;
    // materialized-input state (retractable min/max, aggregate/minput.rs):
    // shared row store + per-(minput call, slot) chain heads; rows carry
    // (arg value, stream key) and a CAS-claimed alive flag, mirroring the
    // state-table rows keyed [group, value, stream key]
    uint32_t* mheads;   // [n_minput][cap], UINT32_MAX = empty
    long long* mval;    // [mrow_cap]
    uint8_t* mval_null; // [mrow_cap]
    long long* msk;     // [n_sk][mrow_cap]
    uint8_t* msk_null;  // [n_sk][mrow_cap]
    uint32_t* mnext;
    uint32_t* malive;
    uint32_t* mcursor; // single counter
    uint32_t mrow_cap;
    int n_sk;
    // minput-table spill bookkeeping (§8f-2 round 2): owner of each mrow
    // + per-epoch kill list (deletes of pre-drain rows)
    uint32_t* mslot;          // [mrow_cap] owning group slot
    uint8_t* mcall;           // [mrow_cap] minput ordinal
    uint32_t* mkilled;        // [mkilled_cap]
    uint32_t* mkilled_cursor; // single counter
    uint32_t mkilled_cap;
    // decimal-sum state (exact i256 at scale 28):
    unsigned long long* dsum; // [n_dec][4][cap]
    uint32_t* dscl;           // [n_dec][cap] running max input scale
    long long* dspec;         // [n_dec][3][cap] NaN/+Inf/-Inf counts
    long long* out_vals2;     // [out_capacity][width] decimal high halves
    long long* prev2;         // [n_calls][cap] decimal high halves
    int n_dec;
};

// agg_apply: HashAggExecutor::apply_chunk (hash_agg.rs:332-409) as one
// grid-stride pass over a multi-chunk batch. Per row: group lookup
// (get_group_visibilities/touch_agg_groups collapse to find-or-insert),
// then AggregateFunction::update per call (general.rs:18-41,154-162) via
// atomics — order-free for count/sum/min/max over the epoch's row multiset.
// Wave-level segmented pre-aggregation: q7 input is date_time-monotone, so
// consecutive rows (= consecutive lanes) share the window slot. Each
// contiguous equal-slot run inside a wave is reduced with shuffles and its
// TAIL lane issues ONE atomic per call — cutting hot-slot atomics by up to
// 64× (the unsorted case degrades to runs of 1 = plain per-lane atomics,
// still correct). count/sum combine by +, min/max by min/max, null-presence
// by OR; all order-free, so the result is exactly the reference's.
template <int KW, int n_calls, bool DENSE>
__global__ void agg_apply_kernel(AggBatch b, AggTableDev t, AggCallDev c0,
                                 AggCallDev c1, AggCallDev c2, AggCallDev c3,
                                 int mode, uint32_t r0, uint32_t r1) {
    AggCallDev calls[4] = {c0, c1, c2, c3};
    uint32_t stride = gridDim.x * blockDim.x;
    uint32_t iters = (r1 - r0 + stride - 1) / stride;
    int lane = threadIdx.x & 63;
    size_t cap = (size_t)t.cap_mask + 1;
    const uint32_t SLOT_NONE = (uint32_t)-1;

    // per-lane probe memo: with sorted input a head's key rarely changes
    // between grid-stride iterations — skip the (fabric-bound) table probe
    // and dirty re-check when it doesn't
    int64_t memo_kw[KW];
    for (int i = 0; i < KW; i++) memo_kw[i] = 0;
    uint32_t memo_null = 0;
    uint32_t memo_slot = (uint32_t)-1;
    bool memo_dirtied = false;

    for (uint32_t it = 0; it < iters; it++) {
        uint32_t r = r0 + it * stride + blockIdx.x * blockDim.x + threadIdx.x;
        bool active = DENSE ? (r < r1) : ((r < r1) && !(b.vis && !b.vis[r]));
        int64_t kw[MAX_KW];
        uint32_t nullmask = 0;
        for (int i = 0; i < KW; i++) kw[i] = 0;
        size_t rs = (size_t)r * b.stride;
        if (active) {
            for (int i = 0; i < KW; i++) {
                bool valid = DENSE || (b.col_valid[i][rs] ^ b.valid_inverted);
                kw[i] = valid ? b.col_vals[i][rs] : 0;
                nullmask |= (!valid) << i;
            }
        }
        // run-head detection by neighbor key compare: only one lane per
        // contiguous equal-key run probes the table (q7 input is
        // window-sorted, so runs are long; unsorted degrades to per-lane)
        uint64_t act_b = __ballot(active);
        bool same = true;
        for (int i = 0; i < KW; i++) {
            int64_t pk = __shfl_up(kw[i], 1);
            same = same && (pk == kw[i]);
        }
        uint32_t pn = (uint32_t)__shfl_up((int)nullmask, 1);
        same = same && (pn == nullmask);
        bool prev_active = lane > 0 && ((act_b >> (lane - 1)) & 1);
        bool head = active && !(same && prev_active);
        if (mode == 2) head = active; // debug: no dedupe, per-lane probe
        uint32_t slot = SLOT_NONE;
        if (mode == 4) { // debug: loads only (no probe/scan/atomics)
            long long acc4 = nullmask;
            for (int i = 0; i < KW; i++) acc4 += kw[i];
            for (int ci = 0; ci < n_calls; ci++)
                if (active && calls[ci].arg >= 0) acc4 += b.col_vals[KW + ci][rs];
            if (acc4 == 0x7fffffffffffffffLL) t.counters[2] = 9;
            continue;
        }
        if (head) {
            bool memo_hit = memo_slot != SLOT_NONE && nullmask == memo_null;
            for (int i = 0; i < KW; i++) memo_hit = memo_hit && kw[i] == memo_kw[i];
            if (memo_hit) {
                slot = memo_slot;
            } else {
                slot = table_find_or_insert(t.state, t.keys, t.key_nulls,
                                            t.cap_mask, kw, nullmask, KW);
                if (slot == SLOT_NONE) atomicExch(&t.counters[2], 1u); // full
                memo_slot = slot;
                memo_null = nullmask;
                for (int i = 0; i < KW; i++) memo_kw[i] = kw[i];
                memo_dirtied = false;
            }
            // dirty tracking moved to the head (it owns the memo); exactly-
            // once overall is still the CAS's job
            if (slot != SLOT_NONE && !memo_dirtied) {
                if (ld_u32(&t.dirty_flag[slot]) == 0 &&
                    atomicCAS(&t.dirty_flag[slot], 0u, 1u) == 0u) {
                    uint32_t i = atomicAdd(&t.counters[0], 1u);
                    t.dirty_list[i] = slot;
                }
                memo_dirtied = true;
            }
        }
        uint64_t heads_b = __ballot(head);
        uint64_t le_mask = heads_b & (~0ULL >> (63 - lane));
        int run_start = 63 - __clzll(le_mask | 1ULL);
        // the shuffle MUST be executed by all lanes: ds_bpermute returns
        // undefined data when the SOURCE lane (the head) is exec-masked off,
        // so a divergent `if (!head) slot = shfl(...)` reads garbage
        uint32_t bcast_slot = (uint32_t)__shfl((int)slot, run_start);
        if (active && !head) slot = bcast_slot;
        int run_pos = lane - run_start;

        // per-lane contributions (identity when inactive / NULL arg)
        uint8_t op = (!DENSE && active && slot != SLOT_NONE) ? b.ops[r]
                                                             : RW_OP_INSERT;
        long long sign =
            (op == RW_OP_DELETE || op == RW_OP_UPDATE_DELETE) ? -1 : 1;
        bool contributing = active && slot != SLOT_NONE;
        long long v[4];
        uint8_t hasmask = 0;
        for (int ci = 0; ci < n_calls; ci++) {
            const AggCallDev& c = calls[ci];
            int col = KW + ci;
            bool arg_valid =
                contributing &&
                (DENSE || (b.col_valid[col][rs] ^ b.valid_inverted));
            if (c.minput) {
                // materialized-input ops are exact per-row chain mutations
                // (minput.rs apply_batch): no wave aggregation
                v[ci] = 0;
                if (contributing) {
                    uint32_t* headp = t.mheads + (size_t)c.mord * cap + slot;
                    bool valid = b.col_valid[col][rs] ^ b.valid_inverted;
                    long long val = valid ? b.col_vals[col][rs] : 0;
                    if (sign > 0) {
                        uint32_t row = atomicAdd(t.mcursor, 1u);
                        if (row >= t.mrow_cap) {
                            atomicExch(&t.counters[2], 3u); // minput store full
                        } else {
                            st_u32(&t.mslot[row], slot);
                            t.mcall[row] = (uint8_t)c.mord;
                            st_i64((int64_t*)&t.mval[row], val);
                            t.mval_null[row] = !valid;
                            for (int k = 0; k < t.n_sk; k++) {
                                int skc = KW + n_calls + k;
                                bool skv = b.col_valid[skc][rs] ^ b.valid_inverted;
                                st_i64((int64_t*)&t.msk[(size_t)k * t.mrow_cap + row],
                                       skv ? b.col_vals[skc][rs] : 0);
                                t.msk_null[(size_t)k * t.mrow_cap + row] = !skv;
                            }
                            st_u32(&t.malive[row], 1);
                            uint32_t old_head = ld_u32(headp);
                            for (;;) {
                                st_u32(&t.mnext[row], old_head);
                                asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
                                uint32_t prev = atomicCAS(headp, old_head, row);
                                if (prev == old_head) break;
                                old_head = prev;
                            }
                        }
                    } else {
                        // delete: CAS-kill one matching (value, stream key)
                        uint32_t row = ld_u32(headp);
                        while (row != UINT32_MAX) {
                            if (ld_u32(&t.malive[row])) {
                                bool eq = (t.mval_null[row] == (uint8_t)!valid) &&
                                          (!valid || ld_i64((const int64_t*)&t.mval[row]) == val);
                                for (int k = 0; eq && k < t.n_sk; k++) {
                                    int skc = KW + n_calls + k;
                                    uint8_t va = b.col_valid[skc][rs] ^ b.valid_inverted;
                                    uint8_t vb = !t.msk_null[(size_t)k * t.mrow_cap + row];
                                    eq = (va == vb) &&
                                         (!va || b.col_vals[skc][rs] ==
                                                     ld_i64((const int64_t*)&t.msk[(size_t)k * t.mrow_cap + row]));
                                }
                                if (eq && atomicCAS(&t.malive[row], 1u, 0u) == 1u) {
                                    uint32_t ki =
                                        atomicAdd(t.mkilled_cursor, 1u);
                                    if (ki < t.mkilled_cap)
                                        t.mkilled[ki] = row;
                                    else
                                        atomicExch(&t.counters[2], 3u);
                                    break;
                                }
                            }
                            row = ld_u32(&t.mnext[row]);
                        }
                    }
                }
                continue;
            }
            if (c.decimal) {
                // exact decimal sum (device restatement above): per-lane
                // atomics, no wave pre-reduction
                v[ci] = 0;
                if (contributing &&
                    (DENSE || (b.col_valid[col][rs] ^ b.valid_inverted))) {
                    const int64_t* dp = b.col_vals[col]; // 2 words/row
                    DecValD dv = dec_parse_d(dp[2 * rs], dp[2 * rs + 1]);
                    size_t dcap = cap;
                    if (dv.special) {
                        atomic_add_i64(
                            &t.dspec[((size_t)c.dord * 3 + (dv.special - 1)) *
                                         dcap +
                                     slot],
                            sign);
                    } else {
                        uint64_t w[4];
                        dec_addend_d(dv, w);
                        if (sign < 0) {
                            unsigned __int128 cc = 1;
                            for (int k = 0; k < 4; k++) {
                                unsigned __int128 x =
                                    (unsigned __int128)(~w[k]) + cc;
                                w[k] = (uint64_t)x;
                                cc = x >> 64;
                            }
                        }
                        dec_i256_atomic_add(
                            t.dsum + (size_t)c.dord * 4 * dcap, dcap, slot, w);
                        atomicMax(&t.dscl[(size_t)c.dord * dcap + slot],
                                  dv.scale);
                    }
                    t.has[(size_t)ci * cap + slot] = 1;
                }
                continue;
            }
            bool shown = contributing &&
                         !(b.call_hidden[ci] && b.call_hidden[ci][rs]);
            arg_valid = arg_valid && shown;
            switch (c.kind) {
                case RW_AGG_COUNT_STAR: v[ci] = shown ? sign : 0; break;
                case RW_AGG_COUNT: v[ci] = arg_valid ? sign : 0; break;
                case RW_AGG_SUM:
                case RW_AGG_SUM0:
                    v[ci] = arg_valid ? sign * b.col_vals[col][rs] : 0;
                    if (arg_valid) hasmask |= 1 << ci;
                    break;
                case RW_AGG_MIN:
                    v[ci] = arg_valid ? b.col_vals[col][rs] : INT64_MAX;
                    if (arg_valid) hasmask |= 1 << ci;
                    break;
                case RW_AGG_MAX:
                    v[ci] = arg_valid ? b.col_vals[col][rs] : INT64_MIN;
                    if (arg_valid) hasmask |= 1 << ci;
                    break;
            }
        }
        if (mode == 1) {
            // debug fallback: per-lane atomics, no wave aggregation
            if (contributing) {
                for (int ci = 0; ci < n_calls; ci++) {
                    if (calls[ci].minput || calls[ci].decimal) continue;
                    long long* acc = t.acc + (size_t)ci * cap;
                    uint8_t* has = t.has + (size_t)ci * cap;
                    switch (calls[ci].kind) {
                        case RW_AGG_COUNT_STAR:
                        case RW_AGG_COUNT:
                            if (v[ci]) atomic_add_i64(&acc[slot], v[ci]);
                            break;
                        case RW_AGG_SUM:
                        case RW_AGG_SUM0:
                            if (hasmask & (1 << ci)) {
                                atomic_add_i64(&acc[slot], v[ci]);
                                if (!has[slot]) has[slot] = 1;
                            }
                            break;
                        case RW_AGG_MIN:
                            if (hasmask & (1 << ci)) {
                                atomic_min_i64(&acc[slot], v[ci]);
                                if (!has[slot]) has[slot] = 1;
                            }
                            break;
                        case RW_AGG_MAX:
                            if (hasmask & (1 << ci)) {
                                atomic_max_i64(&acc[slot], v[ci]);
                                if (!has[slot]) has[slot] = 1;
                            }
                            break;
                    }
                }
                if (ld_u32(&t.dirty_flag[slot]) == 0 &&
                    atomicCAS(&t.dirty_flag[slot], 0u, 1u) == 0u) {
                    uint32_t i = atomicAdd(&t.counters[0], 1u);
                    t.dirty_list[i] = slot;
                }
            }
            continue;
        }
        // segmented inclusive reduction over the runs; the run TAIL issues
        // one atomic per call (order-free combines: +, min, max, OR)
        for (int d = 1; d < 64; d <<= 1) {
            long long ov[4];
            uint8_t ohas;
            for (int ci = 0; ci < n_calls; ci++) ov[ci] = __shfl_up(v[ci], d);
            // hasmask must ALWAYS be shuffled: even in a DENSE batch the
            // per-lane masks differ when DISTINCT dedup hides rows (a
            // dense own-mask shortcut dropped the run's sum/min/max here
            // whenever the tail lane's own row was a hidden duplicate)
            ohas = (uint8_t)__shfl_up((int)hasmask, d);
            if (run_pos >= d) {
                for (int ci = 0; ci < n_calls; ci++) {
                    switch (calls[ci].kind) {
                        case RW_AGG_MIN: v[ci] = v[ci] < ov[ci] ? v[ci] : ov[ci]; break;
                        case RW_AGG_MAX: v[ci] = v[ci] > ov[ci] ? v[ci] : ov[ci]; break;
                        default: v[ci] += ov[ci];
                    }
                }
                hasmask |= ohas;
            }
        }
        // a run ends before a head, before a non-contributing lane, or at 63
        uint64_t cont_b = __ballot(contributing);
        uint64_t ends_after = (heads_b | ~cont_b) >> 1 | (1ULL << 63);
        bool tail = contributing && ((ends_after >> lane) & 1);
        if (mode == 3) { // debug: skip the tail atomics
            if (tail && v[0] == 0x7fffffffffffffffLL) t.counters[2] = 9;
            continue;
        }
        if (tail) {
            for (int ci = 0; ci < n_calls; ci++) {
                if (calls[ci].minput || calls[ci].decimal) continue;
                long long* acc = t.acc + (size_t)ci * cap;
                uint8_t* has = t.has + (size_t)ci * cap;
                switch (calls[ci].kind) {
                    case RW_AGG_COUNT_STAR:
                    case RW_AGG_COUNT:
                        if (v[ci]) atomic_add_i64(&acc[slot], v[ci]);
                        break;
                    case RW_AGG_SUM:
                    case RW_AGG_SUM0:
                        if (hasmask & (1 << ci)) {
                            atomic_add_i64(&acc[slot], v[ci]);
                            if (!has[slot]) has[slot] = 1; // benign race
                        }
                        break;
                    case RW_AGG_MIN:
                        if (hasmask & (1 << ci)) {
                            atomic_min_i64(&acc[slot], v[ci]);
                            if (!has[slot]) has[slot] = 1;
                        }
                        break;
                    case RW_AGG_MAX:
                        if (hasmask & (1 << ci)) {
                            atomic_max_i64(&acc[slot], v[ci]);
                            if (!has[slot]) has[slot] = 1;
                        }
                        break;
                }
            }
        }
    }
}


// ---- dense 4-rows-per-lane apply (q7 fast path) ----
// Specialization for dense (all-visible/Insert/non-null), KW==1, unit-stride
// batches: each lane owns 4 consecutive rows loaded as 16-B pairs, combines
// lane-local equal-key segments, and the wave closes cross-lane runs with a
// segmented scan over lane suffixes — one table probe + one atomic set per
// committed run. Verified against a host-side lane simulation (30k random
// patterns) before hardware.

__device__ __forceinline__ long long agg4_unit(uint8_t kind, long long v) {
    switch (kind) {
        case RW_AGG_COUNT_STAR:
        case RW_AGG_COUNT: return 1;
        default: return v;
    }
}
__device__ __forceinline__ long long agg4_identity(uint8_t kind) {
    switch (kind) {
        case RW_AGG_MIN: return INT64_MAX;
        case RW_AGG_MAX: return INT64_MIN;
        default: return 0;
    }
}
__device__ __forceinline__ long long agg4_comb(uint8_t kind, long long a,
                                               long long b) {
    switch (kind) {
        case RW_AGG_MIN: return a < b ? a : b;
        case RW_AGG_MAX: return a > b ? a : b;
        default: return a + b;
    }
}

template <bool NT>
__device__ __forceinline__ ulonglong2 ld_b128(const ulonglong2* p) {
    // NT = nontemporal (streaming) load: the input chunk is read exactly
    // once per epoch, so bypassing L2 retention leaves the cache to the
    // state table (A/B via RW_AGG_NT=1).
    if constexpr (NT) {
        const unsigned long long* q = (const unsigned long long*)p;
        ulonglong2 r;
        r.x = __builtin_nontemporal_load(q);
        r.y = __builtin_nontemporal_load(q + 1);
        return r;
    }
    return *p;
}

template <int n_calls, int RPL, bool NT = false, int CS = -1>
__device__ __forceinline__ void dense_load(const AggBatch& b,
                                           const AggCallDev* calls,
                                           uint32_t rb, uint32_t r1, bool* act,
                                           long long* k,
                                           long long (*cv)[n_calls]) {
    bool all_in = rb + RPL - 1 < r1;
    if (all_in) {
        // back-to-back b128 loads: RPL/2 per column, all in flight at once
        const ulonglong2* kp = (const ulonglong2*)(b.col_vals[0] + rb);
#pragma unroll
        for (int h = 0; h < RPL / 2; h++) {
            ulonglong2 kk = ld_b128<NT>(kp + h);
            k[2 * h] = (long long)kk.x;
            k[2 * h + 1] = (long long)kk.y;
        }
#pragma unroll
        for (int r = 0; r < RPL; r++) act[r] = true;
        _Pragma("unroll") for (int ci = 0; ci < n_calls; ci++) {
            if (ci == CS) continue; // count-star carried as run length
            if (calls[ci].arg < 0) {
#pragma unroll
                for (int r = 0; r < RPL; r++) cv[r][ci] = 1;
                continue;
            }
            const ulonglong2* vp = (const ulonglong2*)(b.col_vals[1 + ci] + rb);
#pragma unroll
            for (int h = 0; h < RPL / 2; h++) {
                ulonglong2 vv = ld_b128<NT>(vp + h);
                cv[2 * h][ci] = agg4_unit(calls[ci].kind, (long long)vv.x);
                cv[2 * h + 1][ci] = agg4_unit(calls[ci].kind, (long long)vv.y);
            }
        }
    } else {
        for (int r = 0; r < RPL; r++) {
            act[r] = rb + r < r1;
            k[r] = act[r] ? b.col_vals[0][rb + r] : 0;
            _Pragma("unroll") for (int ci = 0; ci < n_calls; ci++) {
                if (ci == CS) continue;
                cv[r][ci] = act[r]
                                ? agg4_unit(calls[ci].kind,
                                            calls[ci].arg < 0
                                                ? 1
                                                : b.col_vals[1 + ci][rb + r])
                                : 0;
            }
        }
    }
}

// CS >= 0 names a count-star call (arg < 0): its per-row contribution is
// the constant 1, so the run's value is the run LENGTH — carried as one
// u32 beside the scan instead of an i64 inside it (halves the scan's
// cross-lane traffic for q7's [max, count] shape and frees the VGPRs the
// count column held). CS == -1 is the generic path, unchanged.
template <int n_calls, int RPL = 4, bool PF = false, bool NT = false,
          int CS = -1>
__device__ __forceinline__ void agg_apply_dense4_body(
        const AggBatch& b, const AggTableDev& t, AggCallDev c0, AggCallDev c1,
        AggCallDev c2, AggCallDev c3, uint32_t r0, uint32_t r1) {
    AggCallDev calls[4] = {c0, c1, c2, c3};
    int lane = threadIdx.x & 63;
    size_t cap = (size_t)t.cap_mask + 1;
    const uint32_t SLOT_NONE = (uint32_t)-1;
    uint32_t stride_rows = (gridDim.x * blockDim.x) * RPL;
    uint32_t iters = (r1 - r0 + stride_rows - 1) / stride_rows;

    long long memo_key = 0;
    uint32_t memo_slot = SLOT_NONE;
    bool memo_dirtied = false;

    auto commit = [&](long long key, const long long* vals, uint32_t cnt) {
        (void)cnt;
        uint32_t slot;
        if (memo_slot != SLOT_NONE && key == memo_key) {
            slot = memo_slot;
        } else {
            int64_t kw1[1] = {key};
            slot = table_find_or_insert(t.state, t.keys, t.key_nulls, t.cap_mask,
                                        kw1, 0, 1);
            if (slot == SLOT_NONE) {
                atomicExch(&t.counters[2], 1u);
                return;
            }
            memo_slot = slot;
            memo_key = key;
            memo_dirtied = false;
        }
        if (!memo_dirtied) {
            if (ld_u32(&t.dirty_flag[slot]) == 0 &&
                atomicCAS(&t.dirty_flag[slot], 0u, 1u) == 0u) {
                uint32_t i = atomicAdd(&t.counters[0], 1u);
                t.dirty_list[i] = slot;
            }
            memo_dirtied = true;
        }
        _Pragma("unroll") for (int ci = 0; ci < n_calls; ci++) {
            long long* acc = t.acc + (size_t)ci * cap;
            uint8_t* has = t.has + (size_t)ci * cap;
            if (ci == CS) {
                atomic_add_i64(&acc[slot], (long long)cnt);
                continue;
            }
            switch (calls[ci].kind) {
                case RW_AGG_MIN:
                    atomic_min_i64(&acc[slot], vals[ci]);
                    if (!has[slot]) has[slot] = 1;
                    break;
                case RW_AGG_MAX:
                    atomic_max_i64(&acc[slot], vals[ci]);
                    if (!has[slot]) has[slot] = 1;
                    break;
                case RW_AGG_SUM:
                case RW_AGG_SUM0:
                    atomic_add_i64(&acc[slot], vals[ci]);
                    if (!has[slot]) has[slot] = 1;
                    break;
                default:
                    atomic_add_i64(&acc[slot], vals[ci]);
            }
        }
    };

    uint32_t base = r0 + (blockIdx.x * blockDim.x + threadIdx.x) * RPL;
    bool act[RPL];
    long long k[RPL];
    long long cv[RPL][n_calls]; // [row][call]
    if (PF && iters) dense_load<n_calls, RPL, NT, CS>(b, calls, base, r1, act, k, cv);
    for (uint32_t it = 0; it < iters; it++) {
        uint32_t rb = base + it * stride_rows;
        bool act2[RPL];
        long long k2[RPL];
        long long cv2[RPL][n_calls];
        if (PF) {
            // double-buffered prefetch: next tile's loads issue before this
            // tile's scan/commit chain, hiding the HBM round-trip
            if (it + 1 < iters)
                dense_load<n_calls, RPL, NT, CS>(b, calls, rb + stride_rows, r1, act2,
                                         k2, cv2);
        } else {
            dense_load<n_calls, RPL, NT, CS>(b, calls, rb, r1, act, k, cv);
        }
        // lane-local segments over the RPL rows (inactive rows break runs)
        long long last_key = 0;
        bool any = false, has_bnd = false;
        long long cur_key = 0;
        long long cur[n_calls];
        uint32_t cur_n = 0, pre_n = 0;
        bool have_cur = false, have_pre = false;
        long long preq[n_calls];
        long long pre_key = 0;
        for (int r = 0; r < RPL; r++) {
            if (!act[r]) {
                if (have_cur) {
                    if (!have_pre) {
                        _Pragma("unroll") for (int ci = 0; ci < n_calls; ci++) {
                            if (ci == CS) continue;
                            preq[ci] = cur[ci];
                        }
                        pre_key = cur_key;
                        pre_n = cur_n;
                        have_pre = true;
                    } else {
                        commit(cur_key, cur, cur_n); // interior complete run
                    }
                    has_bnd = true;
                    have_cur = false;
                }
                continue;
            }
            any = true;
            if (have_cur && k[r] == cur_key) {
                _Pragma("unroll") for (int ci = 0; ci < n_calls; ci++) {
                    if (ci == CS) continue;
                    cur[ci] = agg4_comb(calls[ci].kind, cur[ci], cv[r][ci]);
                }
                cur_n++;
            } else {
                if (have_cur) {
                    if (!have_pre) {
                        _Pragma("unroll") for (int ci = 0; ci < n_calls; ci++) {
                            if (ci == CS) continue;
                            preq[ci] = cur[ci];
                        }
                        pre_key = cur_key;
                        pre_n = cur_n;
                        have_pre = true;
                    } else {
                        commit(cur_key, cur, cur_n);
                    }
                    has_bnd = true;
                }
                cur_key = k[r];
                _Pragma("unroll") for (int ci = 0; ci < n_calls; ci++) {
                    if (ci == CS) continue;
                    cur[ci] = cv[r][ci];
                }
                cur_n = 1;
                have_cur = true;
            }
        }
        // after the loop: `cur` (if live) is the open suffix run; `preq`
        // holds the first closed run (prefix). Gaps between same-key runs
        // are irrelevant: every partition of a key's rows commits correctly
        // under commutative combines.
        long long sufv[n_calls];
        bool have_suf = have_cur;
        long long suf_key = cur_key;
        _Pragma("unroll") for (int ci = 0; ci < n_calls; ci++) {
            if (ci == CS) continue;
            sufv[ci] = have_cur ? cur[ci] : agg4_identity(calls[ci].kind);
        }
        uint32_t suf_n = have_cur ? cur_n : 0;
        last_key = have_suf ? suf_key : (have_pre ? pre_key : 0);
        bool lane_any = any;
        bool lane_has_bnd = has_bnd || (have_pre && have_suf);

        // inter-lane continuity
        long long prev_last = __shfl_up(last_key, 1);
        uint64_t act_b = __ballot(lane_any);
        bool prev_any = lane > 0 && ((act_b >> (lane - 1)) & 1);
        long long my_first = any ? (have_pre ? pre_key : suf_key) : 0;
        bool first_at_row0 = any && act[0] && k[0] == my_first;
        bool cont = lane_any && prev_any && first_at_row0 &&
                    my_first == prev_last;

        // scan over suffix values
        bool scan_head = !lane_any || lane_has_bnd || !cont;
        uint64_t heads_b = __ballot(scan_head);
        uint64_t le_mask = heads_b & (~0ULL >> (63 - lane));
        int run_start = 63 - __clzll(le_mask | 1ULL);
        int run_pos = lane - run_start;
        long long incl[n_calls];
        _Pragma("unroll") for (int ci = 0; ci < n_calls; ci++) {
            if (ci == CS) continue;
            incl[ci] = sufv[ci];
        }
        uint32_t incl_n = suf_n;
        for (int d = 1; d < 64; d <<= 1) {
            long long ov[n_calls];
            _Pragma("unroll") for (int ci = 0; ci < n_calls; ci++) {
                if (ci == CS) continue;
                ov[ci] = __shfl_up(incl[ci], d);
            }
            uint32_t ov_n = CS >= 0 ? __shfl_up(incl_n, d) : 0u;
            if (run_pos >= d) {
                _Pragma("unroll") for (int ci = 0; ci < n_calls; ci++) {
                    if (ci == CS) continue;
                    incl[ci] = agg4_comb(calls[ci].kind, incl[ci], ov[ci]);
                }
                if (CS >= 0) incl_n += ov_n;
            }
        }
        long long incoming[n_calls];
        _Pragma("unroll") for (int ci = 0; ci < n_calls; ci++) {
            if (ci == CS) continue;
            incoming[ci] = __shfl_up(incl[ci], 1);
        }
        uint32_t incoming_n = CS >= 0 ? __shfl_up(incl_n, 1) : 0u;

        uint64_t cont_b = __ballot(cont);
        uint64_t bnd_b = __ballot(lane_has_bnd);
        bool next_cont = lane < 63 && ((cont_b >> (lane + 1)) & 1);
        bool next_closes = next_cont && ((bnd_b >> (lane + 1)) & 1);

        if (lane_any) {
            if (cont && lane_has_bnd) {
                // closer: commit the incoming run + own prefix
                long long tot[n_calls];
                _Pragma("unroll") for (int ci = 0; ci < n_calls; ci++) {
                    if (ci == CS) continue;
                    tot[ci] = agg4_comb(calls[ci].kind, incoming[ci], preq[ci]);
                }
                commit(pre_key, tot, incoming_n + pre_n);
            } else if (!cont && lane_has_bnd && have_pre) {
                commit(pre_key, preq, pre_n);
            }
            bool run_continues = next_cont;
            if (have_suf && !run_continues) {
                commit(suf_key, incl, incl_n);
            } else if (have_suf && next_closes) {
                // the next lane commits incoming (== this incl) + its prefix
            }
        }
        if (PF && it + 1 < iters) {
#pragma unroll
            for (int r = 0; r < RPL; r++) {
                act[r] = act2[r];
                k[r] = k2[r];
                _Pragma("unroll") for (int ci = 0; ci < n_calls; ci++) {
                    if (ci == CS) continue;
                    cv[r][ci] = cv2[r][ci];
                }
            }
        }
    }
}

template <int n_calls, int RPL = 4, bool PF = false, bool NT = false,
          int CS = -1>
__global__ void agg_apply_dense4_kernel(AggBatch b, AggTableDev t, AggCallDev c0,
                                        AggCallDev c1, AggCallDev c2,
                                        AggCallDev c3, uint32_t r0, uint32_t r1) {
    agg_apply_dense4_body<n_calls, RPL, PF, NT, CS>(b, t, c0, c1, c2, c3, r0, r1);
}

// Occupancy variant (measured +2.5% at 71→64 VGPRs even with 8 spilled
// regs, gpurun_out/q7_w8.json): cap the register budget at 64 VGPRs so 8
// waves/SIMD are resident instead of the default build's 7 — more
// latency-hiding on a latency-structured kernel. A/B via RW_AGG_W8=0/1.
template <int n_calls, int RPL = 4, bool NT = false, int CS = -1>
__global__ __launch_bounds__(256, 8) void agg_apply_dense4_kernel_w8(
        AggBatch b, AggTableDev t, AggCallDev c0, AggCallDev c1, AggCallDev c2,
        AggCallDev c3, uint32_t r0, uint32_t r1) {
    agg_apply_dense4_body<n_calls, RPL, false, NT, CS>(b, t, c0, c1, c2, c3, r0,
                                                       r1);
}

// agg_flush: flush_data's emit-on-update branch (hash_agg.rs:475-501) +
// OnlyOutputIfHasInput::infer_change_type (agg_group.rs:131-165) +
// reset-at-zero (agg_group.rs:431-445). One thread per dirty slot; Update
// writes the U−/U+ pair into consecutive reserved rows (pair adjacency).
__global__ void agg_flush_kernel(AggTableDev t, int KW, int n_calls,
                                 int row_count_index, AggCallDev c0, AggCallDev c1,
                                 AggCallDev c2, AggCallDev c3) {
    AggCallDev calls[4] = {c0, c1, c2, c3};
    uint32_t n_dirty = t.counters[0];
    uint32_t stride = gridDim.x * blockDim.x;
    size_t cap = (size_t)t.cap_mask + 1;
    int width = KW + n_calls;
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n_dirty;
         i += stride) {
        uint32_t slot = t.dirty_list[i];
        t.dirty_flag[slot] = 0;
        // current outputs (get_outputs, agg_group.rs:431-467)
        long long rc = t.acc[(size_t)row_count_index * cap + slot];
        if (rc < 0) rc = 0; // row_count_of clamp (agg_group.rs:63-75)
        long long curr[MAX_CALLS];
        long long curr2[MAX_CALLS];
        uint8_t curr_null[MAX_CALLS];
        for (int ci = 0; ci < n_calls; ci++) {
            const AggCallDev& c = calls[ci];
            long long* acc = t.acc + (size_t)ci * cap;
            uint8_t* has = t.has + (size_t)ci * cap;
            curr2[ci] = 0;
            if (rc == 0) {
                // reset value states (agg_state.rs:149-155)
                switch (c.kind) {
                    case RW_AGG_MIN: acc[slot] = INT64_MAX; break;
                    case RW_AGG_MAX: acc[slot] = INT64_MIN; break;
                    default: acc[slot] = 0;
                }
                has[slot] = 0;
                if (c.decimal) {
                    for (int k = 0; k < 4; k++)
                        t.dsum[((size_t)c.dord * 4 + k) * cap + slot] = 0;
                    t.dscl[(size_t)c.dord * cap + slot] = 0;
                    for (int k = 0; k < 3; k++)
                        t.dspec[((size_t)c.dord * 3 + k) * cap + slot] = 0;
                }
            }
            if (c.decimal) {
                // exact decimal sum output (specials per decimal.rs:259-276)
                if (!has[slot]) {
                    curr[ci] = 0;
                    curr_null[ci] = 1;
                    continue;
                }
                long long nan = t.dspec[((size_t)c.dord * 3 + 0) * cap + slot];
                long long pinf = t.dspec[((size_t)c.dord * 3 + 1) * cap + slot];
                long long ninf = t.dspec[((size_t)c.dord * 3 + 2) * cap + slot];
                DecValD r2{};
                if (nan > 0 || (pinf > 0 && ninf > 0)) {
                    r2.special = 1;
                } else if (pinf > 0) {
                    r2.special = 2;
                } else if (ninf > 0) {
                    r2.special = 3;
                } else if (!dec_from_sum_d(
                               t.dsum + (size_t)c.dord * 4 * cap, cap, slot,
                               t.dscl[(size_t)c.dord * cap + slot], &r2)) {
                    atomicExch(&t.counters[2], 5u); // outside exact domain
                    curr[ci] = 0;
                    curr_null[ci] = 1;
                    continue;
                }
                dec_serialize_d(r2, &curr[ci], &curr2[ci]);
                curr_null[ci] = 0;
                continue;
            }
            if (c.minput) {
                // output_first over the materialized rows (minput.rs:236-241):
                // min = value ASC NULLS LAST → smallest non-null, NULL if all
                // null; max = value DESC NULLS FIRST → NULL if any NULL row
                // (the reference's order types, test_utils/agg_executor.rs:
                // 96-99 + sort_util.rs NullsAre::Largest)
                uint32_t row = t.mheads[(size_t)c.mord * cap + slot];
                bool any = false, any_null = false;
                long long best = 0;
                while (row != UINT32_MAX) {
                    if (t.malive[row]) {
                        if (t.mval_null[row]) {
                            any_null = true;
                        } else {
                            long long v = t.mval[row];
                            if (!any) best = v;
                            else if (c.kind == RW_AGG_MIN) best = v < best ? v : best;
                            else best = v > best ? v : best;
                            any = true;
                        }
                    }
                    row = t.mnext[row];
                }
                if (c.kind == RW_AGG_MAX && any_null) {
                    curr[ci] = 0;
                    curr_null[ci] = 1;
                } else {
                    curr[ci] = best;
                    curr_null[ci] = !any;
                }
                continue;
            }
            switch (c.kind) {
                case RW_AGG_COUNT_STAR:
                case RW_AGG_COUNT:
                case RW_AGG_SUM0:
                    curr[ci] = acc[slot];
                    curr_null[ci] = 0;
                    break;
                default:
                    curr[ci] = acc[slot];
                    curr_null[ci] = !has[slot];
            }
        }
        uint8_t hp = t.has_prev[slot];
        long long prev_rc = 0;
        if (hp) {
            long long p = t.prev[(size_t)row_count_index * cap + slot];
            prev_rc = p < 0 ? 0 : p;
        }
        // infer_change_type (agg_group.rs:138-163)
        int change; // 0 none, 1 insert, 2 delete, 3 update
        if (prev_rc == 0 && rc == 0) change = 0;
        else if (prev_rc == 0) change = 1;
        else if (rc == 0) change = 2;
        else {
            bool eq = true;
            for (int ci = 0; eq && ci < n_calls; ci++) {
                uint8_t pn = t.prev_null[(size_t)ci * cap + slot];
                if (calls[ci].decimal) {
                    // rust_decimal Eq compares VALUES (1.2 == 1.20)
                    eq = (pn == curr_null[ci]) &&
                         (pn ||
                          dec_value_eq_d(t.prev[(size_t)ci * cap + slot],
                                         t.prev2[(size_t)ci * cap + slot],
                                         curr[ci], curr2[ci]));
                } else {
                    eq = (pn == curr_null[ci]) &&
                         (pn || t.prev[(size_t)ci * cap + slot] == curr[ci]);
                }
            }
            change = eq ? 0 : 3;
        }
        if (change == 0) continue;
        int n_out_rows = (change == 3) ? 2 : 1;
        uint32_t base = atomicAdd(&t.counters[1], (uint32_t)n_out_rows);
        if (base + n_out_rows > t.out_capacity) {
            atomicExch(&t.counters[2], 2u); // output overflow
            continue;
        }
        auto write_row = [&](uint32_t orow, uint8_t op, const long long* vals,
                             const uint8_t* nulls, const long long* vals2) {
            t.out_ops[orow] = op;
            for (int k = 0; k < KW; k++) {
                t.out_vals[(size_t)orow * width + k] = t.keys[(size_t)slot * KW + k];
                t.out_nulls[(size_t)orow * width + k] =
                    (t.key_nulls[slot] >> k) & 1;
            }
            for (int ci = 0; ci < n_calls; ci++) {
                t.out_vals[(size_t)orow * width + KW + ci] = vals[ci];
                t.out_nulls[(size_t)orow * width + KW + ci] = nulls[ci];
                if (t.out_vals2)
                    t.out_vals2[(size_t)orow * width + KW + ci] = vals2[ci];
            }
        };
        long long prevv[MAX_CALLS];
        long long prevv2[MAX_CALLS];
        uint8_t prevn[MAX_CALLS];
        for (int ci = 0; ci < n_calls; ci++) {
            prevv[ci] = t.prev[(size_t)ci * cap + slot];
            prevv2[ci] = t.prev2 ? t.prev2[(size_t)ci * cap + slot] : 0;
            prevn[ci] = t.prev_null[(size_t)ci * cap + slot];
        }
        if (change == 1) {
            write_row(base, RW_OP_INSERT, curr, curr_null, curr2);
        } else if (change == 2) {
            write_row(base, RW_OP_DELETE, prevv, prevn, prevv2);
        } else {
            write_row(base, RW_OP_UPDATE_DELETE, prevv, prevn, prevv2);
            write_row(base + 1, RW_OP_UPDATE_INSERT, curr, curr_null, curr2);
        }
        // prev := curr (or cleared on delete) — agg_group.rs:571-607
        if (change == 2) {
            t.has_prev[slot] = 0;
        } else {
            t.has_prev[slot] = 1;
            for (int ci = 0; ci < n_calls; ci++) {
                t.prev[(size_t)ci * cap + slot] = curr[ci];
                if (t.prev2) t.prev2[(size_t)ci * cap + slot] = curr2[ci];
                t.prev_null[(size_t)ci * cap + slot] = curr_null[ci];
            }
        }
    }
}

// init kernel: min/max identities
__global__ void agg_init_kernel(AggTableDev t, int n_calls, AggCallDev c0,
                                AggCallDev c1, AggCallDev c2, AggCallDev c3) {
    AggCallDev calls[4] = {c0, c1, c2, c3};
    size_t cap = (size_t)t.cap_mask + 1;
    size_t total = cap * n_calls;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (size_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
        int ci = (int)(i / cap);
        long long init = 0;
        if (calls[ci].kind == RW_AGG_MIN) init = INT64_MAX;
        if (calls[ci].kind == RW_AGG_MAX) init = INT64_MIN;
        t.acc[i] = init;
    }
}

__global__ void agg_eowc_close_kernel(AggTableDev t, int KW, int n_calls,
                                      int row_count_index, long long wm,
                                      AggCallDev c0, AggCallDev c1,
                                      AggCallDev c2, AggCallDev c3);

__global__ void agg_eowc_dump_kernel(AggTableDev t, int KW, int n_calls,
                                     int row_count_index, AggCallDev c0,
                                     AggCallDev c1, AggCallDev c2,
                                     AggCallDev c3);

extern "C" __global__ void agg_counters_reset_kernel(uint32_t* counters);

// state restore (rw_stream.h contract): seed value states + prev outputs
// from decoded state rows. err code 1 = table full.
__global__ void agg_restore_kernel(AggTableDev t, int KW, int n_calls,
                                   const int64_t* keys,
                                   const uint32_t* knulls,
                                   const long long* vals,
                                   const uint8_t* vnulls,
                                   const long long* vals2, uint32_t n,
                                   int set_prev, AggCallDev c0, AggCallDev c1,
                                   AggCallDev c2, AggCallDev c3) {
    AggCallDev calls[4] = {c0, c1, c2, c3};
    size_t cap = (size_t)t.cap_mask + 1;
    uint32_t stride = gridDim.x * blockDim.x;
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride) {
        int64_t kw[MAX_KW];
        for (int k = 0; k < KW; k++) kw[k] = keys[(size_t)i * KW + k];
        uint32_t slot = table_find_or_insert(t.state, t.keys, t.key_nulls,
                                             t.cap_mask, kw, knulls[i], KW);
        if (slot == UINT32_MAX) {
            atomicExch(&t.counters[2], 1u);
            continue;
        }
        for (int ci = 0; ci < n_calls; ci++) {
            if (calls[ci].decimal) {
                // seed the exact-sum state from the stored decimal output
                // (sum == output in the exact domain; oracle restore does
                // the same)
                long long w0 = vals[(size_t)ci * n + i];
                long long w1 = vals2 ? vals2[(size_t)ci * n + i] : 0;
                uint8_t nu = vnulls[(size_t)ci * n + i];
                int dord = calls[ci].dord;
                t.has[(size_t)ci * cap + slot] = !nu;
                if (!nu) {
                    DecValD dv = dec_parse_d(w0, w1);
                    if (dv.special) {
                        t.dspec[((size_t)dord * 3 + (dv.special - 1)) * cap +
                                slot] = 1;
                    } else {
                        uint64_t w[4];
                        dec_addend_d(dv, w);
                        for (int k = 0; k < 4; k++)
                            t.dsum[((size_t)dord * 4 + k) * cap + slot] =
                                w[k];
                        t.dscl[(size_t)dord * cap + slot] = dv.scale;
                    }
                }
                if (set_prev) {
                    t.prev[(size_t)ci * cap + slot] = w0;
                    if (t.prev2) t.prev2[(size_t)ci * cap + slot] = w1;
                    t.prev_null[(size_t)ci * cap + slot] = nu;
                }
                continue;
            }
            if (calls[ci].minput) {
                // prev output = first entry of the (already restored)
                // minput chain (output_first, minput.rs:236-241; see
                // agg_flush_kernel's identical walk)
                const AggCallDev& c = calls[ci];
                uint32_t row = t.mheads[(size_t)c.mord * cap + slot];
                bool any = false, any_null = false;
                long long best = 0;
                while (row != UINT32_MAX) {
                    if (t.malive[row]) {
                        if (t.mval_null[row]) {
                            any_null = true;
                        } else {
                            long long v2 = t.mval[row];
                            if (!any) best = v2;
                            else if (c.kind == RW_AGG_MIN)
                                best = v2 < best ? v2 : best;
                            else
                                best = v2 > best ? v2 : best;
                            any = true;
                        }
                    }
                    row = t.mnext[row];
                }
                long long pv;
                uint8_t pn;
                if (c.kind == RW_AGG_MAX && any_null) {
                    pv = 0;
                    pn = 1;
                } else {
                    pv = best;
                    pn = !any;
                }
                if (set_prev) {
                    t.prev[(size_t)ci * cap + slot] = pv;
                    t.prev_null[(size_t)ci * cap + slot] = pn;
                }
                continue;
            }
            long long v = vals[(size_t)ci * n + i];
            uint8_t nu = vnulls[(size_t)ci * n + i];
            long long init = 0;
            if (calls[ci].kind == RW_AGG_MIN) init = INT64_MAX;
            if (calls[ci].kind == RW_AGG_MAX) init = INT64_MIN;
            t.acc[(size_t)ci * cap + slot] = nu ? init : v;
            t.has[(size_t)ci * cap + slot] = !nu;
            if (set_prev) {
                t.prev[(size_t)ci * cap + slot] = v;
                t.prev_null[(size_t)ci * cap + slot] = nu;
            }
        }
        if (set_prev) t.has_prev[slot] = 1;
    }
}

// minput-table restore: rebuild the materialized-input chains from decoded
// rows (rw_stream.h restore contract)
__global__ void agg_minput_restore_kernel(AggTableDev t, int KW, int mord,
                                          const int64_t* gkeys,
                                          const uint32_t* gnulls,
                                          const long long* vals,
                                          const uint8_t* vnulls,
                                          const long long* sks,
                                          const uint8_t* sknulls, uint32_t n) {
    size_t cap = (size_t)t.cap_mask + 1;
    uint32_t stride = gridDim.x * blockDim.x;
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride) {
        int64_t kw[MAX_KW];
        for (int k = 0; k < KW; k++) kw[k] = gkeys[(size_t)i * KW + k];
        uint32_t slot = table_find_or_insert(t.state, t.keys, t.key_nulls,
                                             t.cap_mask, kw, gnulls[i], KW);
        if (slot == UINT32_MAX) {
            atomicExch(&t.counters[2], 1u);
            continue;
        }
        uint32_t row = atomicAdd(t.mcursor, 1u);
        if (row >= t.mrow_cap) {
            atomicExch(&t.counters[2], 3u);
            continue;
        }
        st_u32(&t.mslot[row], slot);
        t.mcall[row] = (uint8_t)mord;
        st_i64((int64_t*)&t.mval[row], vals[i]);
        t.mval_null[row] = vnulls[i];
        for (int k = 0; k < t.n_sk; k++) {
            st_i64((int64_t*)&t.msk[(size_t)k * t.mrow_cap + row],
                   sks[(size_t)k * n + i]);
            t.msk_null[(size_t)k * t.mrow_cap + row] =
                sknulls[(size_t)k * n + i];
        }
        st_u32(&t.malive[row], 1);
        uint32_t* headp = t.mheads + (size_t)mord * cap + slot;
        uint32_t old_head = ld_u32(headp);
        for (;;) {
            st_u32(&t.mnext[row], old_head);
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            uint32_t prev = atomicCAS(headp, old_head, row);
            if (prev == old_head) break;
            old_head = prev;
        }
    }
}

struct HashAgg {
    RwHashAggDesc desc;
    std::vector<uint8_t> input_types;
    std::vector<uint32_t> group_key;
    std::vector<RwAggCall> calls;
    int KW, n_calls, out_width;
    std::vector<uint8_t> out_types;
    hipStream_t stream;
    AggTableDev t{};
    uint32_t capacity;
    // staging (host-pinned mirrors reused per push)
    AggBatch stage{};
    uint32_t stage_cap = 0;
    // kernel timing (HIP events on this executor's own stream). Event pairs
    // live in a ring and are read back lazily — a per-step
    // hipEventSynchronize costs a full host round-trip (~45 us, 3x the
    // kernel itself at q7 sizes), so the timed region must stay async.
    static constexpr int EV_RING = 256;
    hipEvent_t ev0[EV_RING] = {}, ev1[EV_RING] = {};
    uint8_t ev_pending[EV_RING] = {};
    uint64_t ev_head = 0;
    double apply_ms_total = 0;
    uint64_t apply_launches = 0, apply_rows = 0;

    int ev_harvest(int slot) {
        if (!ev_pending[slot]) return RW_OK;
        HIP_TRY(hipEventSynchronize(ev1[slot]));
        float ms = 0;
        HIP_TRY(hipEventElapsedTime(&ms, ev0[slot], ev1[slot]));
        apply_ms_total += ms;
        ev_pending[slot] = 0;
        return RW_OK;
    }
    int ev_harvest_all() {
        for (int s = 0; s < EV_RING; s++)
            if (ev_harvest(s) != RW_OK) return RW_E_INTERNAL;
        return RW_OK;
    }
    // pending outputs
    std::vector<RwChunk*> outq;
    // checkpoint spill buffer (§8f-2): state-table KV deltas, accumulated at
    // flush, drained by rw_agg_checkpoint_drain
    std::vector<uint8_t> spill;
    int debug_mode = 0; // RW_AGG_DEBUG_MODE: 1 per-lane atomics, 2 no-dedupe
    uint8_t* d_vnode_bitmap = nullptr; // rescale scope (update_vnode_bitmap)
    uint16_t* d_vnode_hop = nullptr;   // q7-pipeline exchange-hop vnodes
    uint32_t d_vnode_hop_cap = 0;
    bool eowc = false; // emit-on-window-close (hash_agg.rs:421-474)
    bool has_pending_wm = false;
    int64_t pending_wm = 0;
    int n_minput = 0;   // materialized-input (retractable min/max) calls
    std::vector<uint8_t> call_decimal;
    int n_dec = 0;      // decimal-sum calls (exact i256 state)
    std::vector<uint8_t> call_minput;
    std::vector<uint32_t> stream_key;
    // DISTINCT dedup state (aggregate/distinct.rs)
    std::vector<int> distinct_slots;  // batch slot per dedup table
    std::vector<int> call_dedup_idx;  // per call: dedup table index or -1
    std::vector<JoinSlot*> dedup_slots;
    uint32_t dedup_cap_mask = 0;
    std::vector<uint8_t*> hidden_bufs; // device, per dedup table
    uint32_t hidden_cap = 0;
    // §8f-2 dedup-table spill: per table, device dirty tracking + a host
    // persisted bitmap (was the row in the table at the last checkpoint?)
    std::vector<uint8_t> distinct_col_types; // datum col type per table
    std::vector<uint32_t*> ddirty_flag, ddirty_list, ddirty_n;
    std::vector<std::vector<uint8_t>> dedup_persisted;

    int grid_for(uint32_t work) const {
        uint32_t blocks = (work + 255) / 256;
        if (blocks > 2048) blocks = 2048; // G11: cap + grid-stride
        return (int)(blocks ? blocks : 1);
    }

    AggCallDev cd(int i) const {
        if (i < n_calls) {
            int8_t mord = 0, dord = 0;
            for (int j = 0; j < i; j++) {
                mord += call_minput[j];
                dord += call_decimal[j];
            }
            return AggCallDev{calls[i].kind, calls[i].arg, call_minput[i],
                              mord, call_decimal[i], dord};
        }
        return AggCallDev{0, -1, 0, 0, 0, 0};
    }

    int init(const RwHashAggDesc* d) {
        if (!gpu_ok()) FAIL(RW_E_NOGPU, "risingwave_amd: no GPU visible (product path has no CPU fallback)");
        if (const char* m = getenv("RW_AGG_DEBUG_MODE")) debug_mode = atoi(m);
        desc = *d;
        eowc = d->emit_on_window_close != 0;
        input_types.assign(d->input_types, d->input_types + d->n_input_cols);
        group_key.assign(d->group_key_indices, d->group_key_indices + d->n_group_key);
        calls.assign(d->calls, d->calls + d->n_calls);
        KW = (int)group_key.size();
        n_calls = (int)calls.size();
        if (KW < 1 || KW > MAX_KW) FAIL(RW_E_INVAL, "group key width %d unsupported (1..%d)", KW, MAX_KW);
        if (n_calls < 1 || n_calls > 4) FAIL(RW_E_INVAL, "n_calls %d unsupported (1..4 in round-1 kernels)", n_calls);
        for (auto k : group_key) {
            uint8_t ty = input_types[k];
            if (ty != RW_T_I64 && ty != RW_T_TS)
                FAIL(RW_E_INVAL, "group key type %d unsupported on GPU (i64/ts only)", ty);
        }
        stream_key.assign(d->stream_key, d->stream_key + d->n_stream_key);
        for (auto& c : calls) {
            if (c.kind > RW_AGG_MAX) FAIL(RW_E_INVAL, "agg kind %d", c.kind);
            bool minput = (c.kind == RW_AGG_MIN || c.kind == RW_AGG_MAX) &&
                          !d->append_only;
            call_minput.push_back(minput);
            n_minput += minput;
            bool dec = false;
            if (c.arg >= 0) {
                uint8_t ty = input_types[c.arg];
                if (ty == RW_T_DECIMAL) {
                    // decimal scope: non-DISTINCT sum/count
                    // arguments, non-EOWC executors
                    if (c.distinct ||
                        !(c.kind == RW_AGG_SUM || c.kind == RW_AGG_SUM0 ||
                          c.kind == RW_AGG_COUNT))
                        FAIL(RW_E_INVAL,
                             "decimal supported as non-DISTINCT sum/count "
                             "arguments only");
                    if (eowc)
                        FAIL(RW_E_INVAL, "decimal + EOWC unsupported");
                    dec = c.kind != RW_AGG_COUNT; // count needs only validity
                } else if (ty != RW_T_I64 && ty != RW_T_TS) {
                    FAIL(RW_E_INVAL, "agg arg type %d unsupported on GPU (i64/ts only)", ty);
                }
            }
            call_decimal.push_back(dec);
            n_dec += dec;
        }
        for (auto k : stream_key)
            if (input_types[k] == RW_T_DECIMAL)
                FAIL(RW_E_INVAL, "decimal stream keys unsupported");
        // DISTINCT dedup (aggregate/distinct.rs): one counter table per
        // distinct column. min/max DISTINCT ≡ min/max — the frontend strips
        // it — so distinct over a materialized-input call is rejected.
        call_dedup_idx.assign(calls.size(), -1);
        for (size_t ci = 0; ci < calls.size(); ci++) {
            if (!calls[ci].distinct || calls[ci].arg < 0) continue;
            if (call_minput[ci])
                FAIL(RW_E_INVAL,
                     "DISTINCT min/max is min/max — plan should strip it");
            if (KW + 1 > MAX_KW)
                FAIL(RW_E_INVAL, "group key too wide for DISTINCT dedup");
            int slot = KW + (int)ci; // arg column's batch slot
            size_t di = 0;
            for (; di < distinct_slots.size(); di++)
                if (distinct_slots[di] == slot) break;
            // distinct calls on the same arg COLUMN may use different batch
            // slots (args are duplicated per call) — dedupe by source col
            for (size_t cj = 0; cj < ci; cj++)
                if (call_dedup_idx[cj] >= 0 && calls[cj].arg == calls[ci].arg) {
                    di = (size_t)call_dedup_idx[cj];
                    break;
                }
            if (di == distinct_slots.size()) {
                distinct_slots.push_back(slot);
                distinct_col_types.push_back(input_types[calls[ci].arg]);
            }
            call_dedup_idx[ci] = (int)di;
        }
        if (n_minput) {
            if (stream_key.size() > 2)
                FAIL(RW_E_INVAL, "stream key wider than 2 unsupported for GPU minput state");
            for (auto sk : stream_key) {
                uint8_t ty = input_types[sk];
                if (ty != RW_T_I64 && ty != RW_T_TS)
                    FAIL(RW_E_INVAL, "stream key type %d unsupported on GPU", ty);
            }
        }
        out_width = KW + n_calls;
        for (auto k : group_key) out_types.push_back(input_types[k]);
        for (auto& c : calls) out_types.push_back(c.ret_type);
        capacity = 1u << 20;
        if (desc.state_capacity_hint) {
            capacity = 1;
            while (capacity < desc.state_capacity_hint) capacity <<= 1;
        }
        HIP_TRY(hipStreamCreate(&stream));
        size_t cap = capacity;
        t.cap_mask = capacity - 1;
        HIP_TRY(hipMalloc(&t.state, cap * 4));
        HIP_TRY(hipMemset(t.state, 0, cap * 4));
        HIP_TRY(hipMalloc(&t.keys, cap * KW * 8));
        HIP_TRY(hipMalloc(&t.key_nulls, cap * 4));
        HIP_TRY(hipMalloc(&t.acc, cap * n_calls * 8));
        HIP_TRY(hipMalloc(&t.has, cap * n_calls));
        HIP_TRY(hipMemset(t.has, 0, cap * n_calls));
        HIP_TRY(hipMalloc(&t.prev, cap * n_calls * 8));
        HIP_TRY(hipMalloc(&t.prev_null, cap * n_calls));
        HIP_TRY(hipMalloc(&t.has_prev, cap));
        HIP_TRY(hipMemset(t.has_prev, 0, cap));
        HIP_TRY(hipMalloc(&t.dirty_flag, cap * 4));
        HIP_TRY(hipMemset(t.dirty_flag, 0, cap * 4));
        HIP_TRY(hipMalloc(&t.dirty_list, cap * 4));
        HIP_TRY(hipMalloc(&t.counters, 3 * 4));
        HIP_TRY(hipMemset(t.counters, 0, 3 * 4));
        t.out_capacity = 1u << 22;
        HIP_TRY(hipMalloc(&t.out_vals, (size_t)t.out_capacity * out_width * 8));
        HIP_TRY(hipMalloc(&t.out_nulls, (size_t)t.out_capacity * out_width));
        HIP_TRY(hipMalloc(&t.out_ops, t.out_capacity));
        t.n_dec = n_dec;
        if (n_dec) {
            HIP_TRY(hipMalloc(&t.dsum, (size_t)n_dec * 4 * cap * 8));
            HIP_TRY(hipMemset(t.dsum, 0, (size_t)n_dec * 4 * cap * 8));
            HIP_TRY(hipMalloc(&t.dscl, (size_t)n_dec * cap * 4));
            HIP_TRY(hipMemset(t.dscl, 0, (size_t)n_dec * cap * 4));
            HIP_TRY(hipMalloc(&t.dspec, (size_t)n_dec * 3 * cap * 8));
            HIP_TRY(hipMemset(t.dspec, 0, (size_t)n_dec * 3 * cap * 8));
            HIP_TRY(hipMalloc(&t.out_vals2,
                              (size_t)t.out_capacity * out_width * 8));
            HIP_TRY(hipMalloc(&t.prev2, (size_t)n_calls * cap * 8));
            HIP_TRY(hipMemset(t.prev2, 0, (size_t)n_calls * cap * 8));
        }
        t.n_sk = (int)stream_key.size();
        if (n_minput) {
            t.mrow_cap = 1u << 22;
            if (desc.state_capacity_hint) {
                uint64_t want = desc.state_capacity_hint * 8;
                t.mrow_cap = 1;
                while (t.mrow_cap < want && t.mrow_cap < (1u << 28)) t.mrow_cap <<= 1;
            }
            HIP_TRY(hipMalloc(&t.mheads, (size_t)n_minput * cap * 4));
            HIP_TRY(hipMemset(t.mheads, 0xFF, (size_t)n_minput * cap * 4));
            HIP_TRY(hipMalloc(&t.mval, (size_t)t.mrow_cap * 8));
            HIP_TRY(hipMalloc(&t.mval_null, t.mrow_cap));
            size_t nsk = t.n_sk ? t.n_sk : 1;
            HIP_TRY(hipMalloc(&t.msk, nsk * t.mrow_cap * 8));
            HIP_TRY(hipMalloc(&t.msk_null, nsk * t.mrow_cap));
            HIP_TRY(hipMalloc(&t.mnext, (size_t)t.mrow_cap * 4));
            HIP_TRY(hipMalloc(&t.malive, (size_t)t.mrow_cap * 4));
            HIP_TRY(hipMalloc(&t.mcursor, 4));
            HIP_TRY(hipMemset(t.mcursor, 0, 4));
            HIP_TRY(hipMalloc(&t.mslot, (size_t)t.mrow_cap * 4));
            HIP_TRY(hipMalloc(&t.mcall, t.mrow_cap));
            t.mkilled_cap = 1u << 22;
            HIP_TRY(hipMalloc(&t.mkilled, (size_t)t.mkilled_cap * 4));
            HIP_TRY(hipMalloc(&t.mkilled_cursor, 4));
            HIP_TRY(hipMemset(t.mkilled_cursor, 0, 4));
        }
        agg_init_kernel<<<2048, 256, 0, stream>>>(t, n_calls, cd(0), cd(1), cd(2),
                                                  cd(3));
        HIP_TRY(hipStreamSynchronize(stream));
        return RW_OK;
    }

    int n_batch_slots() const { return KW + n_calls + (n_minput ? (int)stream_key.size() : 0); }

    int ensure_stage(uint32_t n_rows) {
        if (stage_cap >= n_rows) return RW_OK;
        free_stage();
        uint32_t cap = 4096;
        while (cap < n_rows) cap <<= 1;
        for (int i = 0; i < n_batch_slots(); i++) {
            // 16 B/row so decimal slots (2 words/row) fit
            HIP_TRY(hipMalloc(&stage.col_vals[i], (size_t)cap * 16));
            HIP_TRY(hipMalloc(&stage.col_valid[i], cap));
        }
        HIP_TRY(hipMalloc(&stage.ops, cap));
        HIP_TRY(hipMalloc(&stage.vis, cap));
        stage_cap = cap;
        return RW_OK;
    }
    void free_stage() {
        for (int i = 0; i < n_batch_slots(); i++) {
            if (stage.col_vals[i]) hipFree(stage.col_vals[i]);
            if (stage.col_valid[i]) hipFree(stage.col_valid[i]);
            stage.col_vals[i] = nullptr;
            stage.col_valid[i] = nullptr;
        }
        if (stage.ops) hipFree(stage.ops);
        if (stage.vis) hipFree(stage.vis);
        stage.ops = stage.vis = nullptr;
        stage_cap = 0;
    }

    // upload one host chunk into a device AggBatch (batch cols = group key
    // cols then per-call arg cols, i64)
    int upload(const RwChunk* c, AggBatch* out, bool use_stage) {
        uint32_t n = c->n_rows;
        AggBatch b{};
        if (use_stage) {
            int rc = ensure_stage(n);
            if (rc != RW_OK) return rc;
            b = stage;
        }
        auto upcol = [&](int bi, uint32_t col_idx) -> int {
            const RwColumn& col = c->cols[col_idx];
            size_t w = 8;
            if (col.type == RW_T_DECIMAL) w = 16; // 2 words/row
            else if (col.type != RW_T_I64 && col.type != RW_T_TS)
                FAIL(RW_E_INVAL, "column type %d unsupported on GPU", col.type);
            HIP_TRY(hipMemcpyAsync(b.col_vals[bi], col.data, (size_t)n * w,
                                   hipMemcpyHostToDevice, stream));
            HIP_TRY(hipMemcpyAsync(b.col_valid[bi], col.valid, n,
                                   hipMemcpyHostToDevice, stream));
            return RW_OK;
        };
        for (int i = 0; i < KW; i++) {
            int rc = upcol(i, group_key[i]);
            if (rc != RW_OK) return rc;
        }
        for (int ci = 0; ci < n_calls; ci++) {
            if (calls[ci].arg >= 0) {
                int rc = upcol(KW + ci, (uint32_t)calls[ci].arg);
                if (rc != RW_OK) return rc;
            } else {
                HIP_TRY(hipMemsetAsync(b.col_valid[KW + ci], 1, n, stream));
            }
        }
        if (n_minput) {
            for (size_t j = 0; j < stream_key.size(); j++) {
                int rc = upcol(KW + n_calls + (int)j, stream_key[j]);
                if (rc != RW_OK) return rc;
            }
        }
        HIP_TRY(hipMemcpyAsync(b.ops, c->ops, n, hipMemcpyHostToDevice, stream));
        if (c->vis) {
            HIP_TRY(hipMemcpyAsync(b.vis, c->vis, n, hipMemcpyHostToDevice, stream));
        } else {
            b.vis = nullptr;
        }
        b.n_rows = n;
        // dense fast path: every row visible, Insert, and non-null in every
        // referenced column — the kernel then skips the three byte streams
        b.dense = (c->vis == nullptr);
        for (uint32_t r = 0; b.dense && r < n; r++) b.dense = c->ops[r] == 0;
        auto col_all_valid = [&](uint32_t col_idx) {
            const uint8_t* v = c->cols[col_idx].valid;
            for (uint32_t r = 0; r < n; r++)
                if (!v[r]) return false;
            return true;
        };
        for (int i = 0; b.dense && i < KW; i++)
            b.dense = col_all_valid(group_key[i]);
        for (int ci = 0; b.dense && ci < n_calls; ci++)
            if (calls[ci].arg >= 0) b.dense = col_all_valid((uint32_t)calls[ci].arg);
        if (n_minput)
            for (size_t jx = 0; b.dense && jx < stream_key.size(); jx++)
                b.dense = col_all_valid(stream_key[jx]);
        *out = b;
        return RW_OK;
    }

    void launch_apply(const AggBatch& b, uint32_t r0, uint32_t r1) {
        auto a0 = cd(0), a1 = cd(1), a2 = cd(2), a3 = cd(3);
        if (b.dense && KW == 1 && b.stride == 1 && n_minput == 0 &&
            n_dec == 0 && distinct_slots.empty() &&
            debug_mode == 0 && ((uintptr_t)(b.col_vals[0] + r0) & 31) == 0 &&
            (r1 - r0) >= 1024) {
            // rows-per-lane (A/B-selectable via RW_AGG_RPL). Measured on
            // MI355X at q7 sizes: 4 → 37.8 G rows/s, 8 → 28.9, 16 → 23.0 —
            // the VGPR cost of wider per-lane tiles outweighs the deeper
            // load pipeline, so 4 is the default.
            static int rpl = [] {
                const char* e = getenv("RW_AGG_RPL");
                int v = e ? atoi(e) : 4;
                return (v == 4 || v == 8 || v == 16) ? v : 4;
            }();
            // PF: double-buffered prefetch variant (A/B via RW_AGG_PF=1)
            static int pf = [] {
                const char* e = getenv("RW_AGG_PF");
                return e && *e == '1';
            }();
            // W8: 8-waves/SIMD register-capped variant. Default ON for the
            // count-specialized (CS) path — measured 160.3 vs 135.0 G rows/s
            // over the unbounded CS build (gpurun_out/q7_csw8.json; the
            // launch-bounds contract also changes scheduling, not just the
            // register cap). RW_AGG_W8=0 disables, =1 forces it for the
            // generic path too (there it spills 8 VGPRs and wins only ~2%).
            static int w8_env = [] {
                const char* e = getenv("RW_AGG_W8");
                return e ? (*e == '1' ? 1 : 0) : -1;
            }();
            int w8 = w8_env == 1;
            // NT: nontemporal input loads (A/B via RW_AGG_NT=1)
            static int nt = [] {
                const char* e = getenv("RW_AGG_NT");
                return e && *e == '1';
            }();
            int grid = grid_for((r1 - r0 + rpl - 1) / rpl);
            // CS: count-star carried as run length beside the scan instead
            // of an i64 inside it (A/B via RW_AGG_CS=0; default on).
            static int cs_en = [] {
                const char* e = getenv("RW_AGG_CS");
                return !(e && *e == '0');
            }();
            int cs = -1;
            for (int ci = 0; ci < n_calls && cs < 0; ci++)
                if (cd(ci).arg < 0 && cd(ci).kind == RW_AGG_COUNT_STAR) cs = ci;
            if (cs_en && cs >= 0 && rpl == 4 && !pf && !nt) {
                bool cs_w8 = w8_env != 0; // default ON here
                #define RW_DCS(nc, csv)                                        \
                    agg_apply_dense4_kernel<nc, 4, false, false, csv>          \
                        <<<grid, 256, 0, stream>>>(b, t, a0, a1, a2, a3, r0, r1)
                #define RW_DCS8(nc, csv)                                       \
                    agg_apply_dense4_kernel_w8<nc, 4, false, csv>              \
                        <<<grid, 256, 0, stream>>>(b, t, a0, a1, a2, a3, r0, r1)
                switch ((cs_w8 ? 128 : 0) + n_calls * 8 + cs) {
                    case 1 * 8 + 0: RW_DCS(1, 0); return;
                    case 2 * 8 + 0: RW_DCS(2, 0); return;
                    case 2 * 8 + 1: RW_DCS(2, 1); return;
                    case 128 + 1 * 8 + 0: RW_DCS8(1, 0); return;
                    case 128 + 2 * 8 + 0: RW_DCS8(2, 0); return;
                    case 128 + 2 * 8 + 1: RW_DCS8(2, 1); return;
                }
                #undef RW_DCS
                #undef RW_DCS8
            }
            if (w8 && rpl == 4) {
                #define RW_DW8(nc, ntv)                                        \
                    agg_apply_dense4_kernel_w8<nc, 4, ntv>                     \
                        <<<grid, 256, 0, stream>>>(b, t, a0, a1, a2, a3, r0, r1)
                switch (n_calls * 2 + (nt ? 1 : 0)) {
                    case 1 * 2 + 0: RW_DW8(1, false); return;
                    case 1 * 2 + 1: RW_DW8(1, true); return;
                    case 2 * 2 + 0: RW_DW8(2, false); return;
                    case 2 * 2 + 1: RW_DW8(2, true); return;
                    case 3 * 2 + 0: RW_DW8(3, false); return;
                    case 3 * 2 + 1: RW_DW8(3, true); return;
                    case 4 * 2 + 0: RW_DW8(4, false); return;
                    case 4 * 2 + 1: RW_DW8(4, true); return;
                }
                #undef RW_DW8
            }
            if (nt && rpl == 4) {
                #define RW_DNT(nc)                                             \
                    agg_apply_dense4_kernel<nc, 4, false, true>                \
                        <<<grid, 256, 0, stream>>>(b, t, a0, a1, a2, a3, r0, r1)
                switch (n_calls) {
                    case 1: RW_DNT(1); return;
                    case 2: RW_DNT(2); return;
                    case 3: RW_DNT(3); return;
                    case 4: RW_DNT(4); return;
                }
                #undef RW_DNT
            }
            if (pf && rpl == 4) {
                switch (n_calls) {
                    case 1: agg_apply_dense4_kernel<1, 4, true><<<grid, 256, 0, stream>>>(b, t, a0, a1, a2, a3, r0, r1); return;
                    case 2: agg_apply_dense4_kernel<2, 4, true><<<grid, 256, 0, stream>>>(b, t, a0, a1, a2, a3, r0, r1); return;
                    case 3: agg_apply_dense4_kernel<3, 4, true><<<grid, 256, 0, stream>>>(b, t, a0, a1, a2, a3, r0, r1); return;
                    case 4: agg_apply_dense4_kernel<4, 4, true><<<grid, 256, 0, stream>>>(b, t, a0, a1, a2, a3, r0, r1); return;
                }
            }
            #define RW_DENSE(nc, rp)                                          \
                agg_apply_dense4_kernel<nc, rp><<<grid, 256, 0, stream>>>(    \
                    b, t, a0, a1, a2, a3, r0, r1)
            switch (n_calls * 32 + rpl) {
                case 1 * 32 + 4: RW_DENSE(1, 4); return;
                case 1 * 32 + 8: RW_DENSE(1, 8); return;
                case 1 * 32 + 16: RW_DENSE(1, 16); return;
                case 2 * 32 + 4: RW_DENSE(2, 4); return;
                case 2 * 32 + 8: RW_DENSE(2, 8); return;
                case 2 * 32 + 16: RW_DENSE(2, 16); return;
                case 3 * 32 + 4: RW_DENSE(3, 4); return;
                case 3 * 32 + 8: RW_DENSE(3, 8); return;
                case 3 * 32 + 16: RW_DENSE(3, 16); return;
                case 4 * 32 + 4: RW_DENSE(4, 4); return;
                case 4 * 32 + 8: RW_DENSE(4, 8); return;
                case 4 * 32 + 16: RW_DENSE(4, 16); return;
            }
            #undef RW_DENSE
        }
        int grid = grid_for(r1 - r0);
        #define RW_LAUNCH(kw, nc)                                             \
            do {                                                              \
                if (b.dense)                                                  \
                    agg_apply_kernel<kw, nc, true><<<grid, 256, 0, stream>>>( \
                        b, t, a0, a1, a2, a3, debug_mode, r0, r1);            \
                else                                                          \
                    agg_apply_kernel<kw, nc, false><<<grid, 256, 0, stream>>>(\
                        b, t, a0, a1, a2, a3, debug_mode, r0, r1);            \
            } while (0)
        switch (KW * 8 + n_calls) {
            case 1 * 8 + 1: RW_LAUNCH(1, 1); break;
            case 1 * 8 + 2: RW_LAUNCH(1, 2); break;
            case 1 * 8 + 3: RW_LAUNCH(1, 3); break;
            case 1 * 8 + 4: RW_LAUNCH(1, 4); break;
            case 2 * 8 + 1: RW_LAUNCH(2, 1); break;
            case 2 * 8 + 2: RW_LAUNCH(2, 2); break;
            case 2 * 8 + 3: RW_LAUNCH(2, 3); break;
            case 2 * 8 + 4: RW_LAUNCH(2, 4); break;
            case 3 * 8 + 1: RW_LAUNCH(3, 1); break;
            case 3 * 8 + 2: RW_LAUNCH(3, 2); break;
            case 3 * 8 + 3: RW_LAUNCH(3, 3); break;
            case 3 * 8 + 4: RW_LAUNCH(3, 4); break;
            case 4 * 8 + 1: RW_LAUNCH(4, 1); break;
            case 4 * 8 + 2: RW_LAUNCH(4, 2); break;
            case 4 * 8 + 3: RW_LAUNCH(4, 3); break;
            case 4 * 8 + 4: RW_LAUNCH(4, 4); break;
        }
        #undef RW_LAUNCH
    }

    int apply(const AggBatch& b, bool timed,
              const std::vector<uint32_t>& seg_bounds = {}) {
        int slot = -1;
        if (timed) {
            slot = (int)(ev_head++ % EV_RING);
            if (!ev0[slot]) {
                HIP_TRY(hipEventCreate(&ev0[slot]));
                HIP_TRY(hipEventCreate(&ev1[slot]));
            }
            // reclaim the slot's previous measurement (long complete by the
            // time the ring wraps) without stalling the current step
            if (ev_harvest(slot) != RW_OK) return RW_E_INTERNAL;
            HIP_TRY(hipEventRecord(ev0[slot], stream));
        }
        uint32_t start = 0;
        for (uint32_t bnd : seg_bounds) {
            if (bnd > start) launch_apply(b, start, bnd);
            start = bnd;
        }
        if (start < b.n_rows) launch_apply(b, start, b.n_rows);
        if (timed) {
            HIP_TRY(hipEventRecord(ev1[slot], stream));
            ev_pending[slot] = 1;
            apply_launches++;
            apply_rows += b.n_rows;
        }
        return RW_OK;
    }

    // apply a sub-range of a device batch as one launch (epoch-granular
    // ingestion: chunks buffered within an epoch are applied together)
    int apply_range(const AggBatch& b, uint32_t r0, uint32_t r1, bool timed) {
        int slot = -1;
        if (timed) {
            slot = (int)(ev_head++ % EV_RING);
            if (!ev0[slot]) {
                HIP_TRY(hipEventCreate(&ev0[slot]));
                HIP_TRY(hipEventCreate(&ev1[slot]));
            }
            if (ev_harvest(slot) != RW_OK) return RW_E_INTERNAL;
            HIP_TRY(hipEventRecord(ev0[slot], stream));
        }
        launch_apply(b, r0, r1);
        if (timed) {
            HIP_TRY(hipEventRecord(ev1[slot], stream));
            ev_pending[slot] = 1;
            apply_launches++;
            apply_rows += r1 - r0;
        }
        return RW_OK;
    }

    // A minput DELETE targeting a row INSERTed earlier in the same chunk
    // must observe that insert (the reference applies rows in order,
    // hash_agg.rs:332-398); the parallel kernel keeps that only across
    // launches, so such deletes run in their own single-row segments.
    std::vector<uint32_t> minput_conflict_segments(const RwChunk* c) {
        std::vector<uint32_t> bounds;
        if (!n_minput) return bounds;
        bool any_delete = false;
        for (uint32_t r = 0; r < c->n_rows && !any_delete; r++)
            any_delete = c->ops[r] == RW_OP_DELETE || c->ops[r] == RW_OP_UPDATE_DELETE;
        if (!any_delete) return bounds;
        auto row_key = [&](uint32_t r) {
            // (group key, per-minput arg, stream key) — conservative match
            std::string k;
            auto add = [&](uint32_t col) {
                uint8_t valid = c->cols[col].valid[r];
                k.push_back((char)valid);
                int64_t v = valid ? ((const int64_t*)c->cols[col].data)[r] : 0;
                k.append((const char*)&v, 8);
            };
            for (auto g : group_key) add(g);
            for (int ci = 0; ci < n_calls; ci++)
                if (call_minput[ci] && calls[ci].arg >= 0) add((uint32_t)calls[ci].arg);
            for (auto sk : stream_key) add(sk);
            return k;
        };
        std::unordered_multiset<std::string> inserts;
        for (uint32_t r = 0; r < c->n_rows; r++) {
            if (c->vis && !c->vis[r]) continue;
            if (c->ops[r] == RW_OP_INSERT || c->ops[r] == RW_OP_UPDATE_INSERT)
                inserts.insert(row_key(r));
        }
        if (inserts.empty()) return bounds;
        for (uint32_t r = 0; r < c->n_rows; r++) {
            if (c->vis && !c->vis[r]) continue;
            if ((c->ops[r] == RW_OP_DELETE || c->ops[r] == RW_OP_UPDATE_DELETE) &&
                inserts.count(row_key(r))) {
                bounds.push_back(r);
                bounds.push_back(r + 1);
            }
        }
        return bounds;
    }

    int ensure_dedup(uint32_t n) {
        if (dedup_slots.empty()) {
            size_t cap = 1;
            size_t hint = (size_t)capacity * 4;
            if (hint < (1u << 16)) hint = 1u << 16;
            while (cap < hint) cap <<= 1;
            dedup_cap_mask = (uint32_t)(cap - 1);
            for (size_t di = 0; di < distinct_slots.size(); di++) {
                JoinSlot* sl = nullptr;
                HIP_TRY(hipMalloc(&sl, cap * sizeof(JoinSlot)));
                dedup_init_kernel<<<2048, 256, 0, stream>>>(sl, cap);
                dedup_slots.push_back(sl);
                uint32_t *df = nullptr, *dl = nullptr, *dn = nullptr;
                HIP_TRY(hipMalloc(&df, cap * 4));
                HIP_TRY(hipMemset(df, 0, cap * 4));
                HIP_TRY(hipMalloc(&dl, cap * 4));
                HIP_TRY(hipMalloc(&dn, 4));
                HIP_TRY(hipMemset(dn, 0, 4));
                ddirty_flag.push_back(df);
                ddirty_list.push_back(dl);
                ddirty_n.push_back(dn);
                dedup_persisted.emplace_back(cap, 0);
            }
        }
        if (hidden_cap < n) {
            for (auto*& hb : hidden_bufs)
                if (hb) hipFree(hb);
            hidden_bufs.assign(distinct_slots.size(), nullptr);
            for (size_t di = 0; di < distinct_slots.size(); di++)
                HIP_TRY(hipMalloc(&hidden_bufs[di], n));
            hidden_cap = n;
        }
        return RW_OK;
    }

    // ----- product-reachable epoch-batched ingestion (VERDICT r01 item
    // 7): with ingest mode ON, push_chunk STAGES the chunk into a growing
    // device epoch buffer (H2D + D2D append; no apply launch), and the
    // barrier flush applies the whole epoch as ONE kernel launch — the
    // same one-launch-per-epoch shape as the bench's preloaded path, but
    // reachable from the INTEGRATION.md binding. Legal because
    // sum/count/min/max value states are order-free over the epoch's row
    // multiset; rejected for materialized-input or DISTINCT
    // aggregates (their per-chunk passes are order-sensitive). -----
    bool epoch_ingest = false;
    AggBatch ebuf{};
    uint64_t ecap = 0, erows = 0;
    bool edense = true;

    int ensure_epoch(uint64_t need) {
        if (ecap >= need) return RW_OK;
        uint64_t cap = ecap ? ecap : (1ull << 21);
        while (cap < need) cap <<= 1;
        AggBatch nb{};
        int slots = n_batch_slots();
        for (int i = 0; i < slots; i++) {
            HIP_TRY(hipMalloc(&nb.col_vals[i], cap * 8));
            HIP_TRY(hipMalloc(&nb.col_valid[i], cap));
            if (ecap) {
                HIP_TRY(hipMemcpyAsync(nb.col_vals[i], ebuf.col_vals[i],
                                       erows * 8, hipMemcpyDeviceToDevice,
                                       stream));
                HIP_TRY(hipMemcpyAsync(nb.col_valid[i], ebuf.col_valid[i],
                                       erows, hipMemcpyDeviceToDevice,
                                       stream));
            }
        }
        HIP_TRY(hipMalloc(&nb.ops, cap));
        if (ecap)
            HIP_TRY(hipMemcpyAsync(nb.ops, ebuf.ops, erows,
                                   hipMemcpyDeviceToDevice, stream));
        HIP_TRY(hipStreamSynchronize(stream));
        for (int i = 0; i < slots && ecap; i++) {
            hipFree(ebuf.col_vals[i]);
            hipFree(ebuf.col_valid[i]);
        }
        if (ecap) hipFree(ebuf.ops);
        ebuf = nb;
        ecap = cap;
        return RW_OK;
    }

    int epoch_append(const RwChunk* c) {
        if (c->vis)
            FAIL(RW_E_INVAL,
                 "epoch-batched ingest requires visibility-compacted chunks");
        AggBatch b;
        int rc = upload(c, &b, true); // H2D into the pinned-staged buffers
        if (rc != RW_OK) return rc;
        uint32_t n = c->n_rows;
        rc = ensure_epoch(erows + n);
        if (rc != RW_OK) return rc;
        int slots = n_batch_slots();
        for (int i = 0; i < slots; i++) {
            HIP_TRY(hipMemcpyAsync(ebuf.col_vals[i] + erows, b.col_vals[i],
                                   (size_t)n * 8, hipMemcpyDeviceToDevice,
                                   stream));
            HIP_TRY(hipMemcpyAsync(ebuf.col_valid[i] + erows, b.col_valid[i],
                                   n, hipMemcpyDeviceToDevice, stream));
        }
        HIP_TRY(hipMemcpyAsync(ebuf.ops + erows, b.ops, n,
                               hipMemcpyDeviceToDevice, stream));
        HIP_TRY(hipStreamSynchronize(stream)); // staging buffer reuse
        edense = edense && b.dense;
        erows += n;
        return RW_OK;
    }

    // apply the staged epoch as one launch (called from flush)
    int epoch_apply() {
        if (!erows) return RW_OK;
        AggBatch b = ebuf;
        b.vis = nullptr;
        b.n_rows = (uint32_t)erows;
        b.dense = edense ? 1 : 0;
        int rc = apply(b, true);
        if (rc != RW_OK) return rc;
        erows = 0;
        edense = true;
        return RW_OK;
    }

    int push_chunk(const RwChunk* c) {
        if (epoch_ingest) return epoch_append(c);
        AggBatch b;
        int rc = upload(c, &b, true);
        if (rc != RW_OK) return rc;
        if (!distinct_slots.empty()) {
            rc = ensure_dedup(c->n_rows);
            if (rc != RW_OK) return rc;
            int grid = grid_for(c->n_rows);
            for (size_t di = 0; di < distinct_slots.size(); di++)
                agg_dedup_kernel<<<grid, 256, 0, stream>>>(
                    b, dedup_slots[di], dedup_cap_mask, KW, distinct_slots[di],
                    hidden_bufs[di], &t.counters[2], ddirty_flag[di],
                    ddirty_list[di], ddirty_n[di]);
            for (size_t ci = 0; ci < calls.size(); ci++)
                if (call_dedup_idx[ci] >= 0)
                    b.call_hidden[ci] = hidden_bufs[call_dedup_idx[ci]];
        }
        rc = apply(b, true, minput_conflict_segments(c));
        if (rc != RW_OK) return rc;
        HIP_TRY(hipStreamSynchronize(stream)); // staging buffer reuse
        return check_overflow();
    }

    int check_overflow() {
        uint32_t ctr[3];
        HIP_TRY(hipMemcpy(ctr, t.counters, 12, hipMemcpyDeviceToHost));
        if (ctr[2] == 1) FAIL(RW_E_INTERNAL, "agg state table full (capacity %u)", capacity);
        if (ctr[2] == 2) FAIL(RW_E_INTERNAL, "agg output buffer overflow");
        if (ctr[2] == 3) FAIL(RW_E_INTERNAL, "agg minput row store full");
        if (ctr[2] == 4) FAIL(RW_E_INTERNAL, "agg distinct dedup table full");
        if (ctr[2] == 5)
            FAIL(RW_E_OVERFLOW,
                 "decimal sum outside the exact 96-bit domain (the "
                 "reference's order-dependent rescale path; unsupported)");
        return RW_OK;
    }

    // Grow the flush output buffers to hold `need` rows. Called on the sync
    // flush paths BEFORE launching the flush kernel (the kernel clears dirty
    // flags and updates prev as it emits, so an overflowed launch is not
    // retryable): large dirty sets — e.g. q3's 512k-order windows × 16-step
    // epochs — emit up to 2 rows per dirty group.
    int ensure_out_capacity(uint64_t need) {
        if (need <= t.out_capacity) return RW_OK;
        uint64_t cap = t.out_capacity;
        while (cap < need) cap <<= 1;
        if (cap > UINT32_MAX) FAIL(RW_E_INVAL, "flush output > 4G rows");
        hipFree(t.out_vals);
        hipFree(t.out_nulls);
        hipFree(t.out_ops);
        t.out_vals = nullptr; t.out_nulls = nullptr; t.out_ops = nullptr;
        HIP_TRY(hipMalloc(&t.out_vals, (size_t)cap * out_width * 8));
        HIP_TRY(hipMalloc(&t.out_nulls, (size_t)cap * out_width));
        HIP_TRY(hipMalloc(&t.out_ops, cap));
        if (n_dec) {
            hipFree(t.out_vals2);
            t.out_vals2 = nullptr;
            HIP_TRY(hipMalloc(&t.out_vals2, (size_t)cap * out_width * 8));
        }
        t.out_capacity = (uint32_t)cap;
        return RW_OK;
    }

    // sync the stream, read the dirty count, and size the output buffers for
    // the worst case (2 rows per dirty group) ahead of the flush launch
    int presize_flush_out() {
        HIP_TRY(hipStreamSynchronize(stream));
        uint32_t nd = 0;
        HIP_TRY(hipMemcpy(&nd, t.counters, 4, hipMemcpyDeviceToHost));
        return ensure_out_capacity((uint64_t)nd * 2);
    }

    int flush(uint64_t) {
        if (epoch_ingest) {
            int rce = epoch_apply();
            if (rce != RW_OK) return rce;
        }
        int rcp = presize_flush_out();
        if (rcp != RW_OK) return rcp;
        if (eowc) {
            // EOWC barrier (hash_agg.rs:429-474): nothing is EMITTED until a
            // watermark closes windows, but the dirty groups' current states
            // still upsert into the intermediate table (mid-window PUTs,
            // :429-460); state accumulates across epochs
            agg_eowc_dump_kernel<<<2048, 256, 0, stream>>>(
                t, KW, n_calls, (int)desc.row_count_index, cd(0), cd(1), cd(2),
                cd(3));
            HIP_TRY(hipStreamSynchronize(stream));
            int rcd = check_overflow();
            if (rcd != RW_OK) return rcd;
            uint32_t ctrd[3];
            HIP_TRY(hipMemcpy(ctrd, t.counters, 12, hipMemcpyDeviceToHost));
            if (uint32_t n_put = ctrd[1]) {
                std::vector<long long> vals((size_t)n_put * out_width);
                std::vector<uint8_t> nulls((size_t)n_put * out_width);
                HIP_TRY(hipMemcpy(vals.data(), t.out_vals, vals.size() * 8,
                                  hipMemcpyDeviceToHost));
                HIP_TRY(hipMemcpy(nulls.data(), t.out_nulls, nulls.size(),
                                  hipMemcpyDeviceToHost));
                std::vector<uint8_t> put_ops(n_put, RW_OP_INSERT);
                spill_records(vals, nulls, put_ops, n_put);
            }
            HIP_TRY(hipMemset(t.counters, 0, 12));
            if (!has_pending_wm) return RW_OK;
            has_pending_wm = false;
            agg_eowc_close_kernel<<<2048, 256, 0, stream>>>(
                t, KW, n_calls, (int)desc.row_count_index, pending_wm, cd(0),
                cd(1), cd(2), cd(3));
            HIP_TRY(hipStreamSynchronize(stream));
            int rc2 = check_overflow();
            if (rc2 != RW_OK) return rc2;
            uint32_t ctr2[3];
            HIP_TRY(hipMemcpy(ctr2, t.counters, 12, hipMemcpyDeviceToHost));
            uint32_t n_out = ctr2[1];
            if (n_out) {
                std::vector<long long> vals((size_t)n_out * out_width);
                std::vector<uint8_t> nulls((size_t)n_out * out_width);
                std::vector<uint8_t> ops(n_out);
                HIP_TRY(hipMemcpy(vals.data(), t.out_vals, vals.size() * 8,
                                  hipMemcpyDeviceToHost));
                HIP_TRY(hipMemcpy(nulls.data(), t.out_nulls, nulls.size(),
                                  hipMemcpyDeviceToHost));
                HIP_TRY(hipMemcpy(ops.data(), t.out_ops, n_out,
                                  hipMemcpyDeviceToHost));
                // group-key-sorted emission (SortBuffer::consume order)
                std::vector<uint32_t> order(n_out);
                for (uint32_t i = 0; i < n_out; i++) order[i] = i;
                std::sort(order.begin(), order.end(),
                          [&](uint32_t a, uint32_t b) {
                              for (int k = 0; k < KW; k++) {
                                  bool na = nulls[(size_t)a * out_width + k];
                                  bool nb = nulls[(size_t)b * out_width + k];
                                  if (na != nb) return nb; // NULLs largest
                                  if (na) continue;
                                  long long va = vals[(size_t)a * out_width + k];
                                  long long vb = vals[(size_t)b * out_width + k];
                                  if (va != vb) return va < vb;
                              }
                              return false;
                          });
                std::vector<long long> sv((size_t)n_out * out_width);
                std::vector<uint8_t> sn((size_t)n_out * out_width);
                std::vector<uint8_t> so(n_out);
                for (uint32_t i = 0; i < n_out; i++) {
                    memcpy(&sv[(size_t)i * out_width],
                           &vals[(size_t)order[i] * out_width],
                           (size_t)out_width * 8);
                    memcpy(&sn[(size_t)i * out_width],
                           &nulls[(size_t)order[i] * out_width], out_width);
                    so[i] = ops[order[i]];
                }
                // spill: every closed window leaves the state table (the
                // reference deletes all intermediate rows < wm) — DELETE
                // records for all rows, incl. the rc==0 marker rows, which
                // are then filtered out of emission
                spill_records_eowc(sv, sn, n_out);
                uint32_t m = 0;
                for (uint32_t i = 0; i < n_out; i++) {
                    if (so[i] != RW_OP_INSERT) continue;
                    if (m != i) {
                        memcpy(&sv[(size_t)m * out_width],
                               &sv[(size_t)i * out_width],
                               (size_t)out_width * 8);
                        memcpy(&sn[(size_t)m * out_width],
                               &sn[(size_t)i * out_width], out_width);
                    }
                    so[m] = RW_OP_INSERT;
                    m++;
                }
                slice_outputs(sv, sn, so, m);
            }
            HIP_TRY(hipMemset(t.counters, 0, 12));
            return RW_OK;
        }
        agg_flush_kernel<<<2048, 256, 0, stream>>>(t, KW, n_calls,
                                                   (int)desc.row_count_index, cd(0),
                                                   cd(1), cd(2), cd(3));
        HIP_TRY(hipStreamSynchronize(stream));
        int rc = check_overflow();
        if (rc != RW_OK) return rc;
        uint32_t ctr[3];
        HIP_TRY(hipMemcpy(ctr, t.counters, 12, hipMemcpyDeviceToHost));
        uint32_t n_out = ctr[1];
        if (n_out) {
            std::vector<long long> vals((size_t)n_out * out_width);
            std::vector<uint8_t> nulls((size_t)n_out * out_width);
            std::vector<uint8_t> ops(n_out);
            std::vector<long long> vals2;
            HIP_TRY(hipMemcpy(vals.data(), t.out_vals, vals.size() * 8,
                              hipMemcpyDeviceToHost));
            HIP_TRY(hipMemcpy(nulls.data(), t.out_nulls, nulls.size(),
                              hipMemcpyDeviceToHost));
            HIP_TRY(hipMemcpy(ops.data(), t.out_ops, n_out, hipMemcpyDeviceToHost));
            if (n_dec) {
                vals2.resize(vals.size());
                HIP_TRY(hipMemcpy(vals2.data(), t.out_vals2,
                                  vals2.size() * 8, hipMemcpyDeviceToHost));
            }
            spill_records(vals, nulls, ops, n_out, vals2);
            slice_outputs(vals, nulls, ops, n_out, vals2);
        }
        HIP_TRY(hipMemset(t.counters, 0, 12));
        return RW_OK;
    }

    // EOWC close spill: one DELETE per closed (emitted) window. Paired
    // with agg_eowc_dump_kernel's per-barrier mid-window PUTs of the
    // dirty groups' current states (hash_agg.rs:429-460), the stream is
    // the full state-table view — an EOWC executor restores from it
    // (tests/test_restore.py::test_agg_eowc_restore_*).
    void spill_records_eowc(const std::vector<long long>& vals,
                            const std::vector<uint8_t>& nulls,
                            uint32_t n_out) {
        auto put32 = [&](uint32_t x) {
            for (int b = 0; b < 4; b++) spill.push_back((uint8_t)(x >> (8 * b)));
        };
        for (uint32_t r = 0; r < n_out; r++) {
            spill.push_back(0);
            std::vector<uint8_t> k;
            for (int i = 0; i < KW; i++) {
                rwcodec::DatumC d{nulls[(size_t)r * out_width + i] != 0,
                                  vals[(size_t)r * out_width + i], 0};
                rwcodec::memcmp_encode_datum(k, out_types[i], d, {});
            }
            put32((uint32_t)k.size());
            spill.insert(spill.end(), k.begin(), k.end());
            put32(0);
        }
    }

    // Append state-table KV deltas for the flushed records (the spill
    // boundary of StateTable::commit, state_table.rs:1718): PUT for
    // Insert/U+, DELETE for Delete; U− rows carry no delta of their own.
    // Keys are memcomparable (ASC NULLS LAST per group-key column); values
    // are value-encoded rows of group key ++ encoded states (materialized-
    // input states encode None, agg_group.rs:417-421).
    void spill_records(const std::vector<long long>& vals,
                       const std::vector<uint8_t>& nulls,
                       const std::vector<uint8_t>& ops, uint32_t n_out,
                       const std::vector<long long>& vals2 = {}) {
        auto put32 = [&](uint32_t x) {
            for (int b = 0; b < 4; b++) spill.push_back((uint8_t)(x >> (8 * b)));
        };
        for (uint32_t r = 0; r < n_out; r++) {
            uint8_t op = ops[r];
            if (op == RW_OP_UPDATE_DELETE) continue;
            uint8_t put = op != RW_OP_DELETE;
            spill.push_back(put);
            std::vector<uint8_t> k, v;
            for (int i = 0; i < KW; i++) {
                rwcodec::DatumC d{nulls[(size_t)r * out_width + i] != 0,
                                  vals[(size_t)r * out_width + i], 0};
                rwcodec::memcmp_encode_datum(k, out_types[i], d, {});
            }
            if (put) {
                for (int i = 0; i < KW; i++) {
                    rwcodec::DatumC d{nulls[(size_t)r * out_width + i] != 0,
                                      vals[(size_t)r * out_width + i], 0};
                    rwcodec::value_encode_datum(v, out_types[i], d);
                }
                for (int ci = 0; ci < n_calls; ci++) {
                    if (call_minput[ci]) {
                        rwcodec::value_encode_datum(v, calls[ci].ret_type,
                                                    {true, 0, 0});
                    } else {
                        size_t ix = (size_t)r * out_width + KW + ci;
                        rwcodec::DatumC d = rwcodec::datum_of_word(
                            calls[ci].ret_type, nulls[ix] != 0, vals[ix],
                            vals2.empty() ? 0 : vals2[ix]);
                        rwcodec::value_encode_datum(v, calls[ci].ret_type, d);
                    }
                }
            }
            put32((uint32_t)k.size());
            spill.insert(spill.end(), k.begin(), k.end());
            put32((uint32_t)v.size());
            spill.insert(spill.end(), v.begin(), v.end());
        }
    }

    // §8f-2 drain for one DISTINCT dedup table (its own state table in the
    // reference, one per distinct column: pk = group key ∥ datum, value =
    // full row ++ one i64 count per call distincting on the column,
    // distinct.rs:89-93,158-185; DELETE when the count drops to 0, nothing
    // when a row created this epoch also dies in it — mem-table netting,
    // as the join/topn drains). Records sorted by memcomparable pk: the
    // reference iterates a HashMap (nondeterministic), so sorted order is
    // the canonical form both this and the oracle emit.
    int dedup_drain(int di, std::vector<uint8_t>& out) {
        if (di < 0 || (size_t)di >= distinct_slots.size())
            FAIL(RW_E_INVAL, "dedup table index %d (have %zu)", di,
                 distinct_slots.size());
        if ((size_t)di >= dedup_slots.size()) return RW_OK; // nothing pushed
        HIP_TRY(hipStreamSynchronize(stream));
        uint32_t n = 0;
        HIP_TRY(hipMemcpy(&n, ddirty_n[di], 4, hipMemcpyDeviceToHost));
        if (!n) return RW_OK;
        std::vector<uint32_t> slots_h(n);
        HIP_TRY(hipMemcpy(slots_h.data(), ddirty_list[di], (size_t)n * 4,
                          hipMemcpyDeviceToHost));
        DedupDirtyRec* d_out = nullptr;
        HIP_TRY(hipMalloc(&d_out, (size_t)n * sizeof(DedupDirtyRec)));
        dedup_gather_kernel<<<grid_for(n), 256, 0, stream>>>(
            dedup_slots[di], ddirty_flag[di], ddirty_list[di], ddirty_n[di],
            d_out);
        std::vector<DedupDirtyRec> recs(n);
        int rc_cp = hipMemcpy(recs.data(), d_out,
                              (size_t)n * sizeof(DedupDirtyRec),
                              hipMemcpyDeviceToHost);
        hipFree(d_out);
        if (rc_cp != hipSuccess) FAIL(RW_E_INTERNAL, "dedup gather copy");
        HIP_TRY(hipMemset(ddirty_n[di], 0, 4));
        std::vector<uint8_t> key_types(out_types.begin(),
                                       out_types.begin() + KW);
        key_types.push_back(distinct_col_types[di]);
        auto& persisted = dedup_persisted[di];
        struct Rec { std::vector<uint8_t> k, v; uint8_t put; };
        std::vector<Rec> rs;
        rs.reserve(n);
        for (uint32_t i = 0; i < n; i++) {
            const DedupDirtyRec& r = recs[i];
            uint32_t slot = slots_h[i];
            Rec e;
            // signed view: a retract of a never-inserted datum wraps the u32
            // below 0 — match the oracle's signed-count semantics (the
            // reference's count is i64; inconsistent input tolerance)
            if ((int32_t)r.count > 0) {
                e.put = 1;
                persisted[slot] = 1;
            } else {
                if (!persisted[slot]) continue; // created+died this epoch
                e.put = 0;
                persisted[slot] = 0;
            }
            for (int c = 0; c <= KW; c++) {
                rwcodec::DatumC d = rwcodec::datum_of_word(
                    key_types[c], ((r.nulls >> c) & 1) != 0, r.key[c]);
                rwcodec::memcmp_encode_datum(e.k, key_types[c], d, {});
            }
            if (e.put) {
                for (int c = 0; c <= KW; c++) {
                    rwcodec::DatumC d = rwcodec::datum_of_word(
                        key_types[c], ((r.nulls >> c) & 1) != 0, r.key[c]);
                    rwcodec::value_encode_datum(e.v, key_types[c], d);
                }
                for (int ci = 0; ci < n_calls; ci++)
                    if (call_dedup_idx[ci] == di)
                        rwcodec::value_encode_datum(
                            e.v, RW_T_I64, {false, (long long)r.count, 0});
            }
            rs.push_back(std::move(e));
        }
        std::sort(rs.begin(), rs.end(),
                  [](const Rec& a, const Rec& b) { return a.k < b.k; });
        auto put32 = [&](uint32_t x) {
            for (int b = 0; b < 4; b++) out.push_back((uint8_t)(x >> (8 * b)));
        };
        for (auto& e : rs) {
            out.push_back(e.put);
            put32((uint32_t)e.k.size());
            out.insert(out.end(), e.k.begin(), e.k.end());
            put32((uint32_t)e.v.size());
            out.insert(out.end(), e.v.begin(), e.v.end());
        }
        return RW_OK;
    }

    // §8f-5 recovery for one DISTINCT dedup table: replay concatenated
    // dedup_drain streams (PUT last-write-wins / DELETE removes), decode
    // the surviving (group ∥ datum, count) records and seed the counter
    // slots; restored slots are marked persisted so a later 1→0 drop
    // drains as DELETE exactly as in an uninterrupted run.
    int dedup_restore(int di, const uint8_t* buf, uint64_t len) {
        if (di < 0 || (size_t)di >= distinct_slots.size())
            FAIL(RW_E_INVAL, "dedup table index %d (have %zu)", di,
                 distinct_slots.size());
        int rc = ensure_dedup(1);
        if (rc != RW_OK) return rc;
        std::map<std::string, std::vector<uint8_t>> merged;
        bool ok = rwcodec::for_each_frame(
            buf, len,
            [&](uint8_t put, const uint8_t* k, uint32_t klen,
                const uint8_t* v, uint32_t vlen) {
                std::string key((const char*)k, klen);
                if (put)
                    merged[key].assign(v, v + vlen);
                else
                    merged.erase(key);
            });
        if (!ok) FAIL(RW_E_INVAL, "malformed spill stream");
        uint32_t n = (uint32_t)merged.size();
        if (!n) return RW_OK;
        int KW1 = KW + 1;
        std::vector<uint8_t> key_types(out_types.begin(),
                                       out_types.begin() + KW);
        key_types.push_back(distinct_col_types[di]);
        std::vector<long long> keys((size_t)n * KW1);
        std::vector<uint32_t> nulls(n, 0), counts(n);
        uint32_t i = 0;
        for (auto& [kbytes, val] : merged) {
            (void)kbytes;
            size_t off = 0;
            for (int c = 0; c < KW1; c++) {
                rwcodec::DatumC d;
                size_t got = rwcodec::value_decode_datum(
                    val.data() + off, val.size() - off, key_types[c], &d);
                if (!got) FAIL(RW_E_INVAL, "dedup restore: bad key datum");
                off += got;
                keys[(size_t)i * KW1 + c] =
                    d.null ? 0 : rwcodec::word_of_datum(key_types[c], d);
                nulls[i] |= (uint32_t)(d.null != 0) << c;
            }
            rwcodec::DatumC d;
            size_t got = rwcodec::value_decode_datum(
                val.data() + off, val.size() - off, RW_T_I64, &d);
            if (!got || d.null)
                FAIL(RW_E_INVAL, "dedup restore: bad count datum");
            counts[i] = (uint32_t)d.i;
            i++;
        }
        long long* dkeys = nullptr;
        uint32_t *dnulls = nullptr, *dcounts = nullptr, *dslots_out = nullptr,
                 *derr = nullptr;
        HIP_TRY(hipMalloc(&dkeys, keys.size() * 8));
        HIP_TRY(hipMalloc(&dnulls, (size_t)n * 4));
        HIP_TRY(hipMalloc(&dcounts, (size_t)n * 4));
        HIP_TRY(hipMalloc(&dslots_out, (size_t)n * 4));
        HIP_TRY(hipMalloc(&derr, 4));
        HIP_TRY(hipMemcpy(dkeys, keys.data(), keys.size() * 8,
                          hipMemcpyHostToDevice));
        HIP_TRY(hipMemcpy(dnulls, nulls.data(), (size_t)n * 4,
                          hipMemcpyHostToDevice));
        HIP_TRY(hipMemcpy(dcounts, counts.data(), (size_t)n * 4,
                          hipMemcpyHostToDevice));
        HIP_TRY(hipMemset(derr, 0, 4));
        dedup_restore_kernel<<<grid_for(n), 256, 0, stream>>>(
            dkeys, dnulls, dcounts, n, KW1, dedup_slots[di], dedup_cap_mask,
            dslots_out, derr);
        std::vector<uint32_t> slots_h(n);
        uint32_t errv = 0;
        int rc_s = hipStreamSynchronize(stream) == hipSuccess ? RW_OK
                                                              : RW_E_INTERNAL;
        if (rc_s == RW_OK) {
            hipMemcpy(slots_h.data(), dslots_out, (size_t)n * 4,
                      hipMemcpyDeviceToHost);
            hipMemcpy(&errv, derr, 4, hipMemcpyDeviceToHost);
        }
        hipFree(dkeys);
        hipFree(dnulls);
        hipFree(dcounts);
        hipFree(dslots_out);
        hipFree(derr);
        if (rc_s != RW_OK) FAIL(RW_E_INTERNAL, "dedup restore sync failed");
        if (errv) FAIL(RW_E_INTERNAL, "dedup restore overflow (code %u)", errv);
        auto& persisted = dedup_persisted[di];
        for (uint32_t r = 0; r < n; r++) persisted[slots_h[r]] = 1;
        return RW_OK;
    }

    // ----- §8f-2 round 2: materialized-input state-TABLE spill. One
    // table per retractable min/max call (AggStateStorage::
    // MaterializedInput, test_utils/agg_executor.rs:63-121): pk =
    // [group ASC ∥ value ASC(min)/DESC(max) ∥ stream key ASC], NULLs
    // largest; value = the full row. Per-epoch deltas with mem-table
    // netting: fresh alive rows → PUT, pre-drain rows killed this epoch →
    // DELETE, created+died nets away. The drain copies the row arrays up
    // to the cursor (host-marshal path; cost noted). -----
    std::vector<uint32_t> mflush_mark, mkill_mark;

    int minput_drain(int mi, std::vector<uint8_t>& out) {
        if (mi < 0 || mi >= n_minput) FAIL(RW_E_INVAL, "minput table %d", mi);
        HIP_TRY(hipStreamSynchronize(stream));
        if ((int)mflush_mark.size() < n_minput) {
            mflush_mark.assign(n_minput, 0);
            mkill_mark.assign(n_minput, 0);
        }
        uint32_t cur = 0, kcur = 0;
        HIP_TRY(hipMemcpy(&cur, t.mcursor, 4, hipMemcpyDeviceToHost));
        HIP_TRY(hipMemcpy(&kcur, t.mkilled_cursor, 4, hipMemcpyDeviceToHost));
        if (kcur > t.mkilled_cap)
            FAIL(RW_E_INTERNAL, "minput kill list overflow (lost deltas)");
        if (cur > t.mrow_cap) cur = t.mrow_cap;
        size_t cap = (size_t)t.cap_mask + 1;
        std::vector<long long> mval(cur);
        std::vector<uint8_t> mnull(cur), mc(cur), alive4(cur * 4);
        std::vector<uint32_t> mslot_h(cur), alive(cur);
        std::vector<long long> sk((size_t)t.n_sk * cur);
        std::vector<uint8_t> skn((size_t)t.n_sk * cur);
        if (cur) {
            HIP_TRY(hipMemcpy(mval.data(), t.mval, (size_t)cur * 8,
                              hipMemcpyDeviceToHost));
            HIP_TRY(hipMemcpy(mnull.data(), t.mval_null, cur,
                              hipMemcpyDeviceToHost));
            HIP_TRY(hipMemcpy(mc.data(), t.mcall, cur, hipMemcpyDeviceToHost));
            HIP_TRY(hipMemcpy(mslot_h.data(), t.mslot, (size_t)cur * 4,
                              hipMemcpyDeviceToHost));
            HIP_TRY(hipMemcpy(alive.data(), t.malive, (size_t)cur * 4,
                              hipMemcpyDeviceToHost));
            for (int k = 0; k < t.n_sk; k++) {
                HIP_TRY(hipMemcpy(sk.data() + (size_t)k * cur,
                                  t.msk + (size_t)k * t.mrow_cap,
                                  (size_t)cur * 8, hipMemcpyDeviceToHost));
                HIP_TRY(hipMemcpy(skn.data() + (size_t)k * cur,
                                  t.msk_null + (size_t)k * t.mrow_cap, cur,
                                  hipMemcpyDeviceToHost));
            }
        }
        std::vector<uint32_t> kills(kcur);
        if (kcur)
            HIP_TRY(hipMemcpy(kills.data(), t.mkilled, (size_t)kcur * 4,
                              hipMemcpyDeviceToHost));
        std::vector<int64_t> gkeys(cap * KW);
        std::vector<uint32_t> gnulls(cap);
        HIP_TRY(hipMemcpy(gkeys.data(), t.keys, cap * KW * 8,
                          hipMemcpyDeviceToHost));
        HIP_TRY(hipMemcpy(gnulls.data(), t.key_nulls, cap * 4,
                          hipMemcpyDeviceToHost));
        // ordinal -> (call index)
        int ci = -1, o = -1;
        for (size_t j = 0; j < calls.size(); j++) {
            if (call_minput[j]) o++;
            if (o == mi) {
                ci = (int)j;
                break;
            }
        }
        auto encode = [&](uint32_t row, std::string* k,
                          std::vector<uint8_t>* v) {
            std::vector<uint8_t> kb;
            uint32_t slot = mslot_h[row];
            for (int i2 = 0; i2 < KW; i2++) {
                rwcodec::DatumC d{((gnulls[slot] >> i2) & 1) != 0,
                                  gkeys[(size_t)slot * KW + i2], 0};
                rwcodec::memcmp_encode_datum(kb, out_types[i2], d, {});
            }
            rwcodec::DatumC dv = rwcodec::datum_of_word(
                input_types[calls[ci].arg], mnull[row] != 0, mval[row]);
            rwcodec::memcmp_encode_datum(
                kb, input_types[calls[ci].arg], dv,
                {calls[ci].kind == RW_AGG_MAX, true});
            for (int k2 = 0; k2 < t.n_sk; k2++) {
                rwcodec::DatumC d{skn[(size_t)k2 * cur + row] != 0,
                                  sk[(size_t)k2 * cur + row], 0};
                rwcodec::memcmp_encode_datum(
                    kb, input_types[stream_key[k2]], d, {});
            }
            k->assign((const char*)kb.data(), kb.size());
            if (v) {
                for (int i2 = 0; i2 < KW; i2++) {
                    rwcodec::DatumC d{((gnulls[slot] >> i2) & 1) != 0,
                                      gkeys[(size_t)slot * KW + i2], 0};
                    rwcodec::value_encode_datum(*v, out_types[i2], d);
                }
                rwcodec::DatumC dv2 = rwcodec::datum_of_word(
                    input_types[calls[ci].arg], mnull[row] != 0, mval[row]);
                rwcodec::value_encode_datum(*v, input_types[calls[ci].arg],
                                            dv2);
                for (int k2 = 0; k2 < t.n_sk; k2++) {
                    rwcodec::DatumC d{skn[(size_t)k2 * cur + row] != 0,
                                      sk[(size_t)k2 * cur + row], 0};
                    rwcodec::value_encode_datum(
                        *v, input_types[stream_key[k2]], d);
                }
            }
        };
        struct Rec {
            std::string k;
            std::vector<uint8_t> v;
            uint8_t put;
        };
        std::vector<Rec> rs;
        for (uint32_t row = mflush_mark[mi]; row < cur; row++) {
            if (mc[row] != (uint8_t)mi || !alive[row]) continue;
            Rec e;
            e.put = 1;
            encode(row, &e.k, &e.v);
            rs.push_back(std::move(e));
        }
        for (uint32_t i2 = mkill_mark[mi]; i2 < kcur; i2++) {
            uint32_t row = kills[i2];
            if (row >= cur || mc[row] != (uint8_t)mi) continue;
            if (row >= mflush_mark[mi]) continue; // created+died this epoch
            Rec e;
            e.put = 0;
            encode(row, &e.k, nullptr);
            rs.push_back(std::move(e));
        }
        std::stable_sort(rs.begin(), rs.end(),
                         [](const Rec& a, const Rec& b) { return a.k < b.k; });
        auto put32 = [&](uint32_t x) {
            for (int b = 0; b < 4; b++) out.push_back((uint8_t)(x >> (8 * b)));
        };
        for (auto& e : rs) {
            out.push_back(e.put);
            put32((uint32_t)e.k.size());
            out.insert(out.end(), e.k.begin(), e.k.end());
            put32((uint32_t)e.v.size());
            out.insert(out.end(), e.v.begin(), e.v.end());
        }
        mflush_mark[mi] = cur;
        mkill_mark[mi] = kcur;
        return RW_OK;
    }

    // §8f-4 memory reclamation for the materialized-input row store:
    // retractions retire rows in place (malive=0) and the store would
    // otherwise grow without bound. Host-marshalled pack (maintenance
    // cadence, same copy costs as a drain): alive rows compact down,
    // chains rebuild host-side, heads re-upload. Requires drained minput
    // tables (flush marks at the cursor, kill list consumed).
    int minput_compact(uint64_t* reclaimed) {
        if (!n_minput) {
            if (reclaimed) *reclaimed = 0;
            return RW_OK;
        }
        HIP_TRY(hipStreamSynchronize(stream));
        uint32_t cur = 0, kcur = 0;
        HIP_TRY(hipMemcpy(&cur, t.mcursor, 4, hipMemcpyDeviceToHost));
        HIP_TRY(hipMemcpy(&kcur, t.mkilled_cursor, 4,
                          hipMemcpyDeviceToHost));
        if ((int)mflush_mark.size() < n_minput) {
            mflush_mark.assign(n_minput, 0);
            mkill_mark.assign(n_minput, 0);
        }
        for (int mi = 0; mi < n_minput; mi++)
            if (mflush_mark[mi] != cur || mkill_mark[mi] != kcur)
                FAIL(RW_E_INVAL, "minput compact requires drained tables "
                                 "(table %d has undrained deltas)", mi);
        if (cur > t.mrow_cap) cur = t.mrow_cap;
        size_t cap = (size_t)t.cap_mask + 1;
        std::vector<long long> mval(cur), sk((size_t)t.n_sk * cur);
        std::vector<uint8_t> mnull(cur), skn((size_t)t.n_sk * cur),
            mc(cur);
        std::vector<uint32_t> mslot_h(cur), alive(cur);
        if (cur) {
            HIP_TRY(hipMemcpy(mval.data(), t.mval, (size_t)cur * 8,
                              hipMemcpyDeviceToHost));
            HIP_TRY(hipMemcpy(mnull.data(), t.mval_null, cur,
                              hipMemcpyDeviceToHost));
            HIP_TRY(hipMemcpy(mc.data(), t.mcall, cur,
                              hipMemcpyDeviceToHost));
            HIP_TRY(hipMemcpy(mslot_h.data(), t.mslot, (size_t)cur * 4,
                              hipMemcpyDeviceToHost));
            HIP_TRY(hipMemcpy(alive.data(), t.malive, (size_t)cur * 4,
                              hipMemcpyDeviceToHost));
            for (int k = 0; k < t.n_sk; k++) {
                HIP_TRY(hipMemcpy(sk.data() + (size_t)k * cur,
                                  t.msk + (size_t)k * t.mrow_cap,
                                  (size_t)cur * 8, hipMemcpyDeviceToHost));
                HIP_TRY(hipMemcpy(skn.data() + (size_t)k * cur,
                                  t.msk_null + (size_t)k * t.mrow_cap,
                                  (size_t)cur, hipMemcpyDeviceToHost));
            }
        }
        uint32_t n_alive = 0;
        std::vector<uint32_t> mnext_h;
        std::vector<uint32_t> heads((size_t)n_minput * cap, UINT32_MAX);
        for (uint32_t r = 0; r < cur; r++) {
            if (!alive[r]) continue;
            uint32_t nr = n_alive++;
            mval[nr] = mval[r];
            mnull[nr] = mnull[r];
            mc[nr] = mc[r];
            mslot_h[nr] = mslot_h[r];
            for (int k = 0; k < t.n_sk; k++) {
                sk[(size_t)k * cur + nr] = sk[(size_t)k * cur + r];
                skn[(size_t)k * cur + nr] = skn[(size_t)k * cur + r];
            }
            size_t hi = (size_t)mc[nr] * cap + mslot_h[nr];
            mnext_h.push_back(heads[hi]);
            heads[hi] = nr;
        }
        if (n_alive) {
            HIP_TRY(hipMemcpy(t.mval, mval.data(), (size_t)n_alive * 8,
                              hipMemcpyHostToDevice));
            HIP_TRY(hipMemcpy(t.mval_null, mnull.data(), n_alive,
                              hipMemcpyHostToDevice));
            HIP_TRY(hipMemcpy(t.mcall, mc.data(), n_alive,
                              hipMemcpyHostToDevice));
            HIP_TRY(hipMemcpy(t.mslot, mslot_h.data(), (size_t)n_alive * 4,
                              hipMemcpyHostToDevice));
            HIP_TRY(hipMemcpy(t.mnext, mnext_h.data(), (size_t)n_alive * 4,
                              hipMemcpyHostToDevice));
            std::vector<uint32_t> ones(n_alive, 1);
            HIP_TRY(hipMemcpy(t.malive, ones.data(), (size_t)n_alive * 4,
                              hipMemcpyHostToDevice));
            for (int k = 0; k < t.n_sk; k++) {
                HIP_TRY(hipMemcpy(t.msk + (size_t)k * t.mrow_cap,
                                  sk.data() + (size_t)k * cur,
                                  (size_t)n_alive * 8,
                                  hipMemcpyHostToDevice));
                HIP_TRY(hipMemcpy(t.msk_null + (size_t)k * t.mrow_cap,
                                  skn.data() + (size_t)k * cur,
                                  (size_t)n_alive, hipMemcpyHostToDevice));
            }
        }
        HIP_TRY(hipMemcpy(t.mheads, heads.data(), heads.size() * 4,
                          hipMemcpyHostToDevice));
        HIP_TRY(hipMemcpy(t.mcursor, &n_alive, 4, hipMemcpyHostToDevice));
        HIP_TRY(hipMemset(t.mkilled_cursor, 0, 4));
        for (int mi = 0; mi < n_minput; mi++) {
            mflush_mark[mi] = n_alive;
            mkill_mark[mi] = 0;
        }
        if (reclaimed)
            *reclaimed = (uint64_t)(cur - n_alive) * (8 + 1 + 9 * t.n_sk);
        return RW_OK;
    }

    int minput_restore(int mi, const uint8_t* buf, uint64_t len) {
        if (mi < 0 || mi >= n_minput) FAIL(RW_E_INVAL, "minput table %d", mi);
        int ci = -1, o = -1;
        for (size_t j = 0; j < calls.size(); j++) {
            if (call_minput[j]) o++;
            if (o == mi) {
                ci = (int)j;
                break;
            }
        }
        std::map<std::string, std::vector<uint8_t>> merged;
        bool ok = rwcodec::for_each_frame(
            buf, len,
            [&](uint8_t put, const uint8_t* k, uint32_t klen,
                const uint8_t* v, uint32_t vlen) {
                std::string key((const char*)k, klen);
                if (put)
                    merged[key].assign(v, v + vlen);
                else
                    merged.erase(key);
            });
        if (!ok) FAIL(RW_E_INVAL, "malformed minput spill stream");
        uint32_t n = (uint32_t)merged.size();
        if (!n) return RW_OK;
        std::vector<int64_t> gk((size_t)n * KW);
        std::vector<uint32_t> gn(n, 0);
        std::vector<long long> mv(n);
        std::vector<uint8_t> mn(n);
        std::vector<long long> sks((size_t)t.n_sk * n, 0);
        std::vector<uint8_t> skns((size_t)t.n_sk * n, 0);
        uint32_t i2 = 0;
        for (auto& [kb, val] : merged) {
            (void)kb;
            size_t off = 0;
            auto rd = [&](uint8_t ty, long long* vv, uint8_t* nn) -> bool {
                rwcodec::DatumC d;
                size_t got = rwcodec::value_decode_datum(
                    val.data() + off, val.size() - off, ty, &d);
                if (!got) return false;
                off += got;
                *vv = d.null ? 0 : rwcodec::word_of_datum(ty, d);
                *nn = d.null;
                return true;
            };
            for (int k2 = 0; k2 < KW; k2++) {
                long long vv;
                uint8_t nn;
                if (!rd(out_types[k2], &vv, &nn))
                    FAIL(RW_E_INVAL, "minput restore: bad group datum");
                gk[(size_t)i2 * KW + k2] = vv;
                gn[i2] |= (uint32_t)(nn != 0) << k2;
            }
            if (!rd(input_types[calls[ci].arg], &mv[i2], &mn[i2]))
                FAIL(RW_E_INVAL, "minput restore: bad value datum");
            for (int k2 = 0; k2 < t.n_sk; k2++) {
                long long vv;
                uint8_t nn;
                if (!rd(input_types[stream_key[k2]], &vv, &nn))
                    FAIL(RW_E_INVAL, "minput restore: bad stream-key datum");
                sks[(size_t)k2 * n + i2] = vv;
                skns[(size_t)k2 * n + i2] = nn;
            }
            i2++;
        }
        int64_t* dgk = nullptr;
        uint32_t* dgn = nullptr;
        long long* dmv = nullptr;
        uint8_t* dmn = nullptr;
        long long* dsk = nullptr;
        uint8_t* dskn = nullptr;
        HIP_TRY(hipMalloc(&dgk, gk.size() * 8));
        HIP_TRY(hipMalloc(&dgn, gn.size() * 4));
        HIP_TRY(hipMalloc(&dmv, mv.size() * 8));
        HIP_TRY(hipMalloc(&dmn, mn.size()));
        HIP_TRY(hipMalloc(&dsk, sks.size() * 8 + 8));
        HIP_TRY(hipMalloc(&dskn, skns.size() + 1));
        HIP_TRY(hipMemcpy(dgk, gk.data(), gk.size() * 8,
                          hipMemcpyHostToDevice));
        HIP_TRY(hipMemcpy(dgn, gn.data(), gn.size() * 4,
                          hipMemcpyHostToDevice));
        HIP_TRY(hipMemcpy(dmv, mv.data(), mv.size() * 8,
                          hipMemcpyHostToDevice));
        HIP_TRY(hipMemcpy(dmn, mn.data(), mn.size(), hipMemcpyHostToDevice));
        if (!sks.empty()) {
            HIP_TRY(hipMemcpy(dsk, sks.data(), sks.size() * 8,
                              hipMemcpyHostToDevice));
            HIP_TRY(hipMemcpy(dskn, skns.data(), skns.size(),
                              hipMemcpyHostToDevice));
        }
        agg_minput_restore_kernel<<<grid_for(n), 256, 0, stream>>>(
            t, KW, mi, dgk, dgn, dmv, dmn, dsk, dskn, n);
        int rc2 = hipStreamSynchronize(stream) == hipSuccess ? RW_OK
                                                             : RW_E_INTERNAL;
        hipFree(dgk);
        hipFree(dgn);
        hipFree(dmv);
        hipFree(dmn);
        hipFree(dsk);
        hipFree(dskn);
        if (rc2 != RW_OK) FAIL(RW_E_INTERNAL, "minput restore sync failed");
        int rc3 = check_overflow();
        if (rc3 != RW_OK) return rc3;
        // restored rows predate the epoch
        if ((int)mflush_mark.size() < n_minput) {
            mflush_mark.assign(n_minput, 0);
            mkill_mark.assign(n_minput, 0);
        }
        uint32_t cur = 0;
        HIP_TRY(hipMemcpy(&cur, t.mcursor, 4, hipMemcpyDeviceToHost));
        for (int j = 0; j < n_minput; j++) mflush_mark[j] = cur;
        return RW_OK;
    }

    // Host-side chunking with the U-pair no-split rule
    // (stream_chunk_builder.rs:188-218)
    void slice_outputs(const std::vector<long long>& vals,
                       const std::vector<uint8_t>& nulls,
                       const std::vector<uint8_t>& ops, uint32_t n_out,
                       const std::vector<long long>& vals2 = {}) {
        uint32_t start = 0;
        uint32_t max_rows = desc.chunk_size ? desc.chunk_size : 1024;
        while (start < n_out) {
            uint32_t take = n_out - start;
            if (take > max_rows) {
                take = max_rows;
                // don't split a U−/U+ pair: if the last row taken is U−,
                // extend by one (the builder's size max+1 case)
                if (ops[start + take - 1] == RW_OP_UPDATE_DELETE) take += 1;
            }
            outq.push_back(make_chunk(vals, nulls, ops, start, take, vals2));
            start += take;
        }
    }

    RwChunk* make_chunk(const std::vector<long long>& vals,
                        const std::vector<uint8_t>& nulls,
                        const std::vector<uint8_t>& ops, uint32_t start,
                        uint32_t n,
                        const std::vector<long long>& vals2 = {}) {
        auto* ch = new RwChunk();
        auto* cols = new RwColumn[out_width];
        auto* o = new uint8_t[n];
        memcpy(o, ops.data() + start, n);
        for (int ci = 0; ci < out_width; ci++) {
            uint8_t ty = out_types[ci];
            if (ty == RW_T_DECIMAL && !vals2.empty()) {
                auto* data = new int64_t[(size_t)2 * n];
                auto* valid = new uint8_t[n];
                for (uint32_t r = 0; r < n; r++) {
                    size_t ix = (size_t)(start + r) * out_width + ci;
                    valid[r] = !nulls[ix];
                    data[2 * r] = vals[ix];
                    data[2 * r + 1] = vals2[ix];
                }
                cols[ci].type = ty;
                cols[ci].valid = valid;
                cols[ci].data = data;
                continue;
            }
            auto* data = new int64_t[n];
            auto* valid = new uint8_t[n];
            for (uint32_t r = 0; r < n; r++) {
                data[r] = vals[(size_t)(start + r) * out_width + ci];
                valid[r] = !nulls[(size_t)(start + r) * out_width + ci];
            }
            cols[ci].type = out_types[ci];
            cols[ci].valid = valid;
            cols[ci].data = data;
        }
        ch->n_rows = n;
        ch->n_cols = out_width;
        ch->ops = o;
        ch->vis = nullptr;
        ch->cols = cols;
        return ch;
    }

    RwChunk* poll() {
        if (outq.empty()) return nullptr;
        RwChunk* c = outq.front();
        outq.erase(outq.begin());
        return c;
    }

    ~HashAgg() {
        free_stage();
        if (ecap) {
            for (int i = 0; i < n_batch_slots(); i++) {
                hipFree(ebuf.col_vals[i]);
                hipFree(ebuf.col_valid[i]);
            }
            hipFree(ebuf.ops);
        }
        for (int s = 0; s < EV_RING; s++)
            if (ev0[s]) {
                hipEventDestroy(ev0[s]);
                hipEventDestroy(ev1[s]);
            }
        if (t.state) {
            hipFree(t.state);
            hipFree(t.keys);
            hipFree(t.key_nulls);
            hipFree(t.acc);
            hipFree(t.has);
            hipFree(t.prev);
            hipFree(t.prev_null);
            hipFree(t.has_prev);
            hipFree(t.dirty_flag);
            hipFree(t.dirty_list);
            hipFree(t.counters);
            hipFree(t.out_vals);
            hipFree(t.out_nulls);
            hipFree(t.out_ops);
            if (t.dsum) {
                hipFree(t.dsum);
                hipFree(t.dscl);
                hipFree(t.dspec);
                hipFree(t.out_vals2);
                hipFree(t.prev2);
            }
            if (t.mheads) {
                hipFree(t.mheads);
                hipFree(t.mval);
                hipFree(t.mval_null);
                hipFree(t.msk);
                hipFree(t.msk_null);
                hipFree(t.mnext);
                hipFree(t.malive);
                hipFree(t.mcursor);
                hipFree(t.mslot);
                hipFree(t.mcall);
                hipFree(t.mkilled);
                hipFree(t.mkilled_cursor);
            }
            hipStreamDestroy(stream);
        }
        for (auto* c : outq) rw_chunk_free(c);
    }
};

// ---------------------------------------------------------------------------
// C ABI
// ---------------------------------------------------------------------------

// Shared checked export for the drain functions: copies a host spill buffer
// into a malloc'd block the caller frees with rw_spill_free. Fails loudly on
// allocation failure instead of crashing in memcpy.
static int spill_export(const std::vector<uint8_t>& sp, uint8_t** buf,
                        uint64_t* len) {
    *len = sp.size();
    *buf = (uint8_t*)malloc(sp.size() ? sp.size() : 1);
    if (!*buf) FAIL(RW_E_INTERNAL, "spill export: host alloc of %zu bytes failed",
                    sp.size());
    memcpy(*buf, sp.data(), sp.size());
    return RW_OK;
}

// Re-emit a spill stream ([put u8][klen u32 LE][key][vlen u32 LE][val] frames)
// in memcomparable-key order. stable: a PUT and DELETE of the same key within
// one epoch (EOWC mid-window PUT then window-close DELETE) keep their order.
static void sort_spill_frames(std::vector<uint8_t>& sp) {
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
    std::stable_sort(frames.begin(), frames.end(), [](const Frame& a,
                                                      const Frame& b) {
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

extern "C" {

const char* rw_last_error(void) { return g_err.c_str(); }

void rw_chunk_free(RwChunk* ch) {
    if (!ch) return;
    for (uint32_t c = 0; c < ch->n_cols; c++) {
        delete[] ch->cols[c].valid;
        delete[] (int64_t*)ch->cols[c].data;
    }
    delete[] ch->cols;
    delete[] ch->ops;
    delete[] ch->vis;
    delete ch;
}

void* rw_hash_agg_create(const RwHashAggDesc* d) {
    auto* h = new HashAgg();
    if (h->init(d) != RW_OK) {
        delete h;
        return nullptr;
    }
    return h;
}
int rw_hash_agg_push_chunk(void* h, const RwChunk* c) {
    return ((HashAgg*)h)->push_chunk(c);
}
int rw_hash_agg_flush(void* h, uint64_t epoch) { return ((HashAgg*)h)->flush(epoch); }

// epoch-batched ingest mode (rw_stream.h): push_chunk stages, flush applies
int rw_hash_agg_ingest_mode(void* h, int epoch_batched) {
    auto* agg = (HashAgg*)h;
    if (epoch_batched && agg->n_dec)
        FAIL(RW_E_INVAL, "epoch-batched ingest + decimal not yet supported");
    if (epoch_batched && (agg->n_minput > 0 || !agg->distinct_slots.empty()))
        FAIL(RW_E_INVAL,
             "epoch-batched ingest needs order-free value states "
             "(no materialized-input or DISTINCT aggregates)");
    if (!epoch_batched && agg->erows)
        FAIL(RW_E_INVAL, "staged epoch pending; flush before disabling");
    agg->epoch_ingest = epoch_batched != 0;
    return RW_OK;
}
RwChunk* rw_hash_agg_poll(void* h) { return ((HashAgg*)h)->poll(); }
void rw_hash_agg_destroy(void* h) { delete (HashAgg*)h; }

// --- bench support: device-resident batches + kernel stats ---

void* rw_agg_bench_preload(void* h, const RwChunk* c) {
    auto* agg = (HashAgg*)h;
    auto* b = new AggBatch{};
    uint32_t n = c->n_rows;
    for (int i = 0; i < agg->n_batch_slots(); i++) {
        // 16 B/row so decimal slots (2 words/row) fit
        if (hipMalloc(&b->col_vals[i], (size_t)n * 16) != hipSuccess) return nullptr;
        if (hipMalloc(&b->col_valid[i], n) != hipSuccess) return nullptr;
    }
    if (hipMalloc(&b->ops, n) != hipSuccess) return nullptr;
    b->vis = nullptr;
    b->n_rows = n;
    AggBatch stage_save = agg->stage;
    uint32_t cap_save = agg->stage_cap;
    agg->stage = *b;
    agg->stage_cap = n;
    AggBatch out;
    int rc = agg->upload(c, &out, true);
    agg->stage = stage_save;
    agg->stage_cap = cap_save;
    if (rc != RW_OK) {
        delete b;
        return nullptr;
    }
    *b = out; // keep the computed fields (n_rows, dense, vis)
    hipStreamSynchronize(agg->stream);
    return b;
}

int rw_agg_bench_apply(void* h, void* batch) {
    auto* agg = (HashAgg*)h;
    return agg->apply(*(AggBatch*)batch, true);
}

int rw_agg_flush_launch(void* h, uint64_t epoch); // defined below

// C-side step loop: the Python interpreter costs ~15-20 us per step at q7
// sizes (ctypes + loop bookkeeping), which exceeds the 16 us apply kernel
// and makes the host the bottleneck. One call runs `steps` applies with the
// periodic stream-ordered checkpoint flush; everything stays async.
int rw_agg_bench_run(void* h, void** batches, int n_batches, int steps,
                     int barrier_every, int step0) {
    auto* agg = (HashAgg*)h;
    for (int i = 0; i < steps; i++) {
        int rc = agg->apply(*(AggBatch*)batches[(step0 + i) % n_batches], true);
        if (rc != RW_OK) return rc;
        if (barrier_every > 0 && (step0 + i + 1) % barrier_every == 0) {
            rc = rw_agg_flush_launch(h, (uint64_t)(step0 + i));
            if (rc != RW_OK) return rc;
        }
    }
    return RW_OK;
}

// Epoch-granular step loop over one contiguous preloaded region: the
// engine buffers an epoch's chunks and applies them as ONE launch before
// the checkpoint flush (order-free value states) — a
// barrier_every×1M-row launch pipelines HBM loads far better than
// barrier_every separate 1M-row launches. `giant` holds n_slots logical
// steps of rows_per_step rows each; step s reads slot s % n_slots.
// Requires barrier_every | n_slots so every epoch is a contiguous range.
int rw_agg_bench_run_epochs(void* h, void* giant, uint64_t rows_per_step,
                            int n_slots, int steps, int barrier_every,
                            int step0) {
    auto* agg = (HashAgg*)h;
    auto* b = (AggBatch*)giant;
    if (barrier_every <= 0 || n_slots % barrier_every ||
        step0 % barrier_every)
        FAIL(RW_E_INVAL, "epoch run needs barrier_every | n_slots, aligned step0");
    for (int s = 0; s < steps; s += barrier_every) {
        int width = steps - s < barrier_every ? steps - s : barrier_every;
        uint32_t r0 = (uint32_t)(((step0 + s) % n_slots) * rows_per_step);
        uint32_t r1 = r0 + (uint32_t)(width * rows_per_step);
        int rc = agg->apply_range(*b, r0, r1, true);
        if (rc != RW_OK) return rc;
        rc = rw_agg_flush_launch(h, (uint64_t)(step0 + s));
        if (rc != RW_OK) return rc;
    }
    return RW_OK;
}

int rw_agg_sync(void* h) {
    auto* agg = (HashAgg*)h;
    if (hipStreamSynchronize(agg->stream) != hipSuccess)
        FAIL(RW_E_INTERNAL, "sync failed");
    return agg->check_overflow();
}

typedef struct {
    uint64_t launches;
    double total_ms;
    uint64_t rows;
} RwKernelStats;

int rw_agg_n_batch_slots(void* h) { return ((HashAgg*)h)->n_batch_slots(); }

// expose a preloaded batch's device pointers (for the exchange path)
int rw_agg_batch_ptrs(void* batch, const int64_t** vals, const uint8_t** valids,
                      const uint8_t** ops, uint32_t* n_rows) {
    auto* b = (AggBatch*)batch;
    for (int i = 0; i < MAX_KW + MAX_CALLS; i++) {
        vals[i] = b->col_vals[i];
        valids[i] = b->col_valid[i];
    }
    *ops = b->ops;
    *n_rows = b->n_rows;
    return RW_OK;
}

// apply an exchange payload (device-resident; per-block layout
// vals[col][npad]*8 ∥ valid[col][n] ∥ ops[n] with npad = n rounded up to 4
// so every column base is 32-B aligned; cols in the executor's batch slot
// order: group key cols then call args) — one apply launch per block.
// `dense` asserts every payload row is a visible non-NULL Insert (the
// SENDER knows: its source batches carried the dense flag) and routes the
// receiver through the vectorized dense kernel.
int rw_agg_apply_payload(void* h, const uint8_t* payload,
                         const uint64_t* block_rows, int n_blocks, int n_cols,
                         int dense) {
    auto* agg = (HashAgg*)h;
    if (agg->n_dec)
        FAIL(RW_E_INVAL, "decimal columns not yet carried by the exchange");
    if (n_cols != agg->KW + agg->n_calls)
        FAIL(RW_E_INVAL, "payload n_cols %d != %d", n_cols, agg->KW + agg->n_calls);
    uint64_t off = 0;
    for (int bi = 0; bi < n_blocks; bi++) {
        uint64_t n = block_rows[bi];
        if (!n) continue;
        uint64_t npad = (n + 3) & ~3ull;
        AggBatch b{};
        for (int c = 0; c < n_cols; c++) {
            b.col_vals[c] = (int64_t*)(payload + off + (uint64_t)c * npad * 8);
            b.col_valid[c] =
                (uint8_t*)(payload + off + (uint64_t)n_cols * npad * 8 +
                           (uint64_t)c * n);
        }
        b.ops = (uint8_t*)(payload + off + (uint64_t)n_cols * npad * 8 +
                           (uint64_t)n_cols * n);
        b.vis = nullptr;
        b.n_rows = (uint32_t)n;
        b.dense = dense ? 1 : 0;
        int rc = agg->apply(b, true);
        if (rc != RW_OK) return rc;
        off += ((uint64_t)n_cols * npad * 8 + (uint64_t)n_cols * n + n + 31) &
               ~31ull;
    }
    return RW_OK;
}

// bench sink: run the flush (change inference + prev update) but leave the
// emitted records in HBM for the device-resident downstream — returns the
// emitted row count, resets the cursor. The host-marshalling path
// (rw_hash_agg_flush + poll) remains the parity-test surface.
long long rw_agg_flush_device(void* h, uint64_t epoch) {
    auto* agg = (HashAgg*)h;
    (void)epoch;
    if (agg->presize_flush_out() != RW_OK) return -1;
    agg_flush_kernel<<<2048, 256, 0, agg->stream>>>(
        agg->t, agg->KW, agg->n_calls, (int)agg->desc.row_count_index,
        agg->cd(0), agg->cd(1), agg->cd(2), agg->cd(3));
    if (hipStreamSynchronize(agg->stream) != hipSuccess) return -1;
    uint32_t ctr[3];
    if (hipMemcpy(ctr, agg->t.counters, 12, hipMemcpyDeviceToHost) != hipSuccess)
        return -1;
    if (ctr[2] != 0) {
        g_err = "agg overflow (code " + std::to_string(ctr[2]) + ")";
        return -(long long)ctr[2] - 1;
    }
    long long n = ctr[1];
    if (hipMemset(agg->t.counters, 0, 12) != hipSuccess) return -1;
    return n;
}

__global__ void agg_counters_reset_kernel(uint32_t* counters) {
    counters[0] = 0; // dirty count
    counters[1] = 0; // out cursor
}

// fully async flush: the change-inference kernel + counter reset run
// stream-ordered with no host round-trip; the overflow flag (counters[2])
// is left set and surfaces at the next rw_agg_sync. The emitted rows are
// device-resident for the downstream, as with rw_agg_flush_device.
int rw_agg_flush_launch(void* h, uint64_t epoch) {
    auto* agg = (HashAgg*)h;
    (void)epoch;
    agg_flush_kernel<<<2048, 256, 0, agg->stream>>>(
        agg->t, agg->KW, agg->n_calls, (int)agg->desc.row_count_index,
        agg->cd(0), agg->cd(1), agg->cd(2), agg->cd(3));
    agg_counters_reset_kernel<<<1, 1, 0, agg->stream>>>(agg->t.counters);
    return RW_OK;
}

int rw_agg_kernel_stats(void* h, RwKernelStats* out) {
    auto* agg = (HashAgg*)h;
    if (agg->ev_harvest_all() != RW_OK)
        FAIL(RW_E_INTERNAL, "event harvest failed");
    out->launches = agg->apply_launches;
    out->total_ms = agg->apply_ms_total;
    out->rows = agg->apply_rows;
    return RW_OK;
}

// debug: scan the device table for duplicate READY keys and report counts
typedef struct {
    uint64_t ready_slots;
    uint64_t dup_keys;
    uint64_t dirty_count;
    uint64_t out_cursor;
} RwAggDebug;

int rw_agg_debug_scan(void* h, RwAggDebug* out) {
    auto* agg = (HashAgg*)h;
    size_t cap = (size_t)agg->t.cap_mask + 1;
    std::vector<uint32_t> state(cap);
    std::vector<int64_t> keys(cap * agg->KW);
    std::vector<uint32_t> knulls(cap);
    uint32_t ctr[3];
    hipDeviceSynchronize();
    HIP_TRY(hipMemcpy(state.data(), agg->t.state, cap * 4, hipMemcpyDeviceToHost));
    HIP_TRY(hipMemcpy(keys.data(), agg->t.keys, cap * agg->KW * 8,
                      hipMemcpyDeviceToHost));
    HIP_TRY(hipMemcpy(knulls.data(), agg->t.key_nulls, cap * 4,
                      hipMemcpyDeviceToHost));
    HIP_TRY(hipMemcpy(ctr, agg->t.counters, 12, hipMemcpyDeviceToHost));
    std::unordered_multiset<std::string> seen;
    uint64_t ready = 0, dups = 0;
    for (size_t sl = 0; sl < cap; sl++) {
        if (state[sl] != SLOT_READY) continue;
        ready++;
        std::string k((const char*)&keys[sl * agg->KW], agg->KW * 8);
        k.append((const char*)&knulls[sl], 4);
        if (seen.count(k)) dups++;
        seen.insert(k);
    }
    out->ready_slots = ready;
    out->dup_keys = dups;
    out->dirty_count = ctr[0];
    out->out_cursor = ctr[1];
    return RW_OK;
}

// debug: dump dirty_list entries -> (slot, state, key0, acc[0], acc[1])
int rw_agg_debug_dirty(void* h, uint32_t max_n, uint32_t* slots, uint32_t* states,
                       int64_t* key0, long long* acc0, long long* acc1,
                       uint32_t* n_out) {
    auto* agg = (HashAgg*)h;
    size_t cap = (size_t)agg->t.cap_mask + 1;
    hipDeviceSynchronize();
    uint32_t ctr[3];
    HIP_TRY(hipMemcpy(ctr, agg->t.counters, 12, hipMemcpyDeviceToHost));
    uint32_t n = ctr[0] < max_n ? ctr[0] : max_n;
    std::vector<uint32_t> dl(n);
    if (n) HIP_TRY(hipMemcpy(dl.data(), agg->t.dirty_list, n * 4, hipMemcpyDeviceToHost));
    std::vector<uint32_t> st(cap);
    std::vector<int64_t> keys(cap * agg->KW);
    std::vector<long long> acc(cap * agg->n_calls);
    HIP_TRY(hipMemcpy(st.data(), agg->t.state, cap * 4, hipMemcpyDeviceToHost));
    HIP_TRY(hipMemcpy(keys.data(), agg->t.keys, cap * agg->KW * 8, hipMemcpyDeviceToHost));
    HIP_TRY(hipMemcpy(acc.data(), agg->t.acc, cap * agg->n_calls * 8, hipMemcpyDeviceToHost));
    for (uint32_t i = 0; i < n; i++) {
        uint32_t sl = dl[i];
        slots[i] = sl;
        states[i] = st[sl];
        key0[i] = keys[(size_t)sl * agg->KW];
        acc0[i] = acc[sl];
        acc1[i] = agg->n_calls > 1 ? acc[cap + sl] : 0;
    }
    *n_out = n;
    return RW_OK;
}

int rw_agg_checkpoint_drain(void* h, uint8_t** buf, uint64_t* len) {
    auto* agg = (HashAgg*)h;
    // memcmp-key order, as the join/topn/dedup drains (rw_stream.h's ordering
    // guarantee): the flush kernel's atomic-cursor emission order is
    // nondeterministic, so the canonical sorted form is what both this and
    // the oracle emit — enabling byte-compare parity.
    sort_spill_frames(agg->spill);
    int rc = spill_export(agg->spill, buf, len);
    if (rc != RW_OK) return rc;
    agg->spill.clear();
    return RW_OK;
}

int rw_hash_agg_restore(void* h, const uint8_t* buf, uint64_t len) {
    auto* agg = (HashAgg*)h;
    // minput-bearing executors: rw_agg_minput_restore must run for every
    // minput table BEFORE this call (prev outputs are recomputed from the
    // hydrated chains, mirroring agg_group.rs:219-221)
    std::map<std::string, std::vector<uint8_t>> merged;
    bool ok = rwcodec::for_each_frame(
        buf, len,
        [&](uint8_t put, const uint8_t* k, uint32_t klen, const uint8_t* v,
            uint32_t vlen) {
            std::string key((const char*)k, klen);
            if (put)
                merged[key].assign(v, v + vlen);
            else
                merged.erase(key);
        });
    if (!ok) FAIL(RW_E_INVAL, "malformed spill stream");
    uint32_t n = (uint32_t)merged.size();
    if (!n) return RW_OK;
    int KW = agg->KW, nc = agg->n_calls;
    std::vector<int64_t> keys((size_t)n * KW);
    std::vector<uint32_t> knulls(n, 0);
    std::vector<long long> vals((size_t)nc * n);
    std::vector<uint8_t> vnulls((size_t)nc * n);
    std::vector<long long> vals2;
    if (agg->n_dec) vals2.assign((size_t)nc * n, 0);
    uint32_t i = 0;
    for (auto& [kbytes, val] : merged) {
        (void)kbytes;
        size_t off = 0;
        for (int c = 0; c < KW + nc; c++) {
            rwcodec::DatumC d;
            uint8_t ty = agg->out_types[c];
            size_t got = rwcodec::value_decode_datum(val.data() + off,
                                                     val.size() - off, ty, &d);
            if (!got) FAIL(RW_E_INVAL, "restore: bad state datum");
            off += got;
            if (c < KW) {
                keys[(size_t)i * KW + c] = d.null ? 0 : d.i;
                knulls[i] |= (uint32_t)(d.null != 0) << c;
            } else {
                vals[(size_t)(c - KW) * n + i] =
                    rwcodec::word_of_datum(ty, d);
                vnulls[(size_t)(c - KW) * n + i] = d.null;
                if (agg->n_dec) vals2[(size_t)(c - KW) * n + i] = d.i2;
            }
        }
        i++;
    }
    int64_t* dkeys = nullptr;
    uint32_t* dknulls = nullptr;
    long long* dvals = nullptr;
    uint8_t* dvnulls = nullptr;
    long long* dvals2 = nullptr;
    HIP_TRY(hipMalloc(&dkeys, keys.size() * 8));
    HIP_TRY(hipMalloc(&dknulls, knulls.size() * 4));
    HIP_TRY(hipMalloc(&dvals, vals.size() * 8));
    HIP_TRY(hipMalloc(&dvnulls, vnulls.size()));
    HIP_TRY(hipMemcpy(dkeys, keys.data(), keys.size() * 8,
                      hipMemcpyHostToDevice));
    HIP_TRY(hipMemcpy(dknulls, knulls.data(), knulls.size() * 4,
                      hipMemcpyHostToDevice));
    HIP_TRY(hipMemcpy(dvals, vals.data(), vals.size() * 8,
                      hipMemcpyHostToDevice));
    HIP_TRY(hipMemcpy(dvnulls, vnulls.data(), vnulls.size(),
                      hipMemcpyHostToDevice));
    if (agg->n_dec) {
        HIP_TRY(hipMalloc(&dvals2, vals2.size() * 8));
        HIP_TRY(hipMemcpy(dvals2, vals2.data(), vals2.size() * 8,
                          hipMemcpyHostToDevice));
    }
    agg_restore_kernel<<<agg->grid_for(n), 256, 0, agg->stream>>>(
        agg->t, KW, nc, dkeys, dknulls, dvals, dvnulls, dvals2, n,
        agg->eowc ? 0 : 1, agg->cd(0), agg->cd(1), agg->cd(2), agg->cd(3));
    int rc = hipStreamSynchronize(agg->stream) == hipSuccess ? RW_OK
                                                             : RW_E_INTERNAL;
    hipFree(dkeys);
    hipFree(dknulls);
    hipFree(dvals);
    hipFree(dvnulls);
    if (dvals2) hipFree(dvals2);
    if (rc != RW_OK) FAIL(RW_E_INTERNAL, "restore sync failed");
    return agg->check_overflow();
}

int rw_agg_n_minput_tables(void* h) { return ((HashAgg*)h)->n_minput; }
int rw_agg_minput_drain(void* h, int mi, uint8_t** buf, uint64_t* len) {
    auto* agg = (HashAgg*)h;
    std::vector<uint8_t> sp;
    int rc = agg->minput_drain(mi, sp);
    if (rc != RW_OK) return rc;
    return spill_export(sp, buf, len);
}
int rw_agg_minput_restore(void* h, int mi, const uint8_t* buf, uint64_t len) {
    return ((HashAgg*)h)->minput_restore(mi, buf, len);
}

int rw_agg_minput_compact(void* h, uint64_t* reclaimed) {
    return ((HashAgg*)h)->minput_compact(reclaimed);
}

int rw_agg_n_dedup_tables(void* h) {
    return (int)((HashAgg*)h)->distinct_slots.size();
}

int rw_agg_dedup_drain(void* h, int di, uint8_t** buf, uint64_t* len) {
    auto* agg = (HashAgg*)h;
    std::vector<uint8_t> sp;
    int rc = agg->dedup_drain(di, sp);
    if (rc != RW_OK) return rc;
    return spill_export(sp, buf, len);
}

int rw_agg_dedup_restore(void* h, int di, const uint8_t* buf, uint64_t len) {
    return ((HashAgg*)h)->dedup_restore(di, buf, len);
}

void rw_spill_free(uint8_t* buf) { free(buf); }

int rw_agg_stats_reset(void* h) {
    auto* agg = (HashAgg*)h;
    agg->ev_harvest_all();
    agg->apply_launches = 0;
    agg->apply_ms_total = 0;
    agg->apply_rows = 0;
    return RW_OK;
}

} // extern "C"

// ---------------------------------------------------------------------------
// Vnode hashing (VirtualNode::compute_chunk, consistent_hash/vnode.rs:146-181)
// ---------------------------------------------------------------------------
// vnode = IEEE CRC-32 (crc32fast) of the row's dist-key datums fed exactly
// as hash_datum does (native-endian primitive bytes; NULL = u32 0xfffffff0,
// array/mod.rs:99) mod vnode_count. Oracle counterpart:
// oracle/oracle_dispatch.cpp; pinned against zlib.crc32 in tests.

__device__ uint32_t g_crc_table[256];
static bool g_crc_table_init = false;

static int ensure_crc_table() {
    if (g_crc_table_init) return RW_OK;
    uint32_t tab[256];
    for (uint32_t i = 0; i < 256; i++) {
        uint32_t c = i;
        for (int k = 0; k < 8; k++) c = (c & 1) ? 0xEDB88320u ^ (c >> 1) : c >> 1;
        tab[i] = c;
    }
    if (hipMemcpyToSymbol(HIP_SYMBOL(g_crc_table), tab, sizeof tab) !=
        hipSuccess)
        return RW_E_INTERNAL;
    g_crc_table_init = true;
    return RW_OK;
}

// LDS-staged CRC: divergent indexing of __constant__ memory serializes
// (each distinct address replays); LDS banks handle it at full rate
__device__ __forceinline__ uint32_t crc32_bytes(const uint32_t* lut, uint32_t crc,
                                               const uint8_t* p, int n) {
    for (int i = 0; i < n; i++)
        crc = lut[(crc ^ p[i]) & 0xFF] ^ (crc >> 8);
    return crc;
}

__device__ __forceinline__ void stage_crc_lut(uint32_t* lut) {
    for (int i = threadIdx.x; i < 256; i += blockDim.x) lut[i] = g_crc_table[i];
    __syncthreads();
}

struct VnodeBatch {
    int64_t* col_vals[MAX_KW];
    uint8_t* col_valid[MAX_KW];
    uint8_t* col_types[1]; // unused placeholder alignment
    uint32_t n_rows;
};

__global__ void vnode_kernel(VnodeBatch b, int n_keys, uint32_t vnode_count,
                             uint16_t* out, uint8_t t0, uint8_t t1, uint8_t t2,
                             uint8_t t3) {
    __shared__ uint32_t lut[256];
    stage_crc_lut(lut);
    uint8_t types[4] = {t0, t1, t2, t3};
    uint32_t stride = gridDim.x * blockDim.x;
    for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < b.n_rows;
         r += stride) {
        uint32_t crc = 0xFFFFFFFFu;
        for (int k = 0; k < n_keys; k++) {
            if (!b.col_valid[k][r]) {
                uint32_t sentinel = 0xfffffff0u;
                crc = crc32_bytes(lut, crc, (const uint8_t*)&sentinel, 4);
            } else {
                int64_t v = b.col_vals[k][r];
                int nbytes = (types[k] == RW_T_I32) ? 4 : 8;
                if (types[k] == RW_T_I32) {
                    int32_t v32 = (int32_t)v;
                    crc = crc32_bytes(lut, crc, (const uint8_t*)&v32, 4);
                } else {
                    crc = crc32_bytes(lut, crc, (const uint8_t*)&v, 8);
                }
                (void)nbytes;
            }
        }
        out[r] = (uint16_t)((uint64_t)(crc ^ 0xFFFFFFFFu) % vnode_count);
    }
}

extern "C" {

typedef struct {
    uint32_t n_keys;
    const uint32_t* key_indices;
    uint32_t vnode_count;
} RwVnodeDesc;

int rw_vnode_compute(const RwVnodeDesc* d, const RwChunk* chunk, uint16_t* out) {
    if (!gpu_ok()) FAIL(RW_E_NOGPU, "no GPU visible");
    if (d->n_keys < 1 || d->n_keys > MAX_KW) FAIL(RW_E_INVAL, "n_keys");
    if (ensure_crc_table() != RW_OK) FAIL(RW_E_INTERNAL, "crc table");
    uint32_t n = chunk->n_rows;
    VnodeBatch b{};
    uint8_t types[4] = {0, 0, 0, 0};
    for (uint32_t k = 0; k < d->n_keys; k++) {
        const RwColumn& c = chunk->cols[d->key_indices[k]];
        types[k] = c.type;
        if (c.type == RW_T_BOOL) FAIL(RW_E_INVAL, "bool vnode key unsupported on GPU");
        HIP_TRY(hipMalloc(&b.col_vals[k], (size_t)n * 8));
        HIP_TRY(hipMalloc(&b.col_valid[k], n));
        if (c.type == RW_T_I32 || c.type == RW_T_F32) {
            // widen to i64 storage but hash native width
            std::vector<int64_t> tmp(n);
            for (uint32_t r = 0; r < n; r++)
                tmp[r] = (int64_t)(uint32_t)((const uint32_t*)c.data)[r];
            HIP_TRY(hipMemcpy(b.col_vals[k], tmp.data(), (size_t)n * 8,
                              hipMemcpyHostToDevice));
        } else {
            HIP_TRY(hipMemcpy(b.col_vals[k], c.data, (size_t)n * 8,
                              hipMemcpyHostToDevice));
        }
        HIP_TRY(hipMemcpy(b.col_valid[k], c.valid, n, hipMemcpyHostToDevice));
    }
    b.n_rows = n;
    uint16_t* dout;
    HIP_TRY(hipMalloc(&dout, (size_t)n * 2));
    uint32_t blocks = (n + 255) / 256;
    if (blocks > 2048) blocks = 2048;
    if (!blocks) blocks = 1;
    vnode_kernel<<<blocks, 256>>>(b, (int)d->n_keys, d->vnode_count, dout,
                                  types[0], types[1], types[2], types[3]);
    HIP_TRY(hipDeviceSynchronize());
    HIP_TRY(hipMemcpy(out, dout, (size_t)n * 2, hipMemcpyDeviceToHost));
    for (uint32_t k = 0; k < d->n_keys; k++) {
        hipFree(b.col_vals[k]);
        hipFree(b.col_valid[k]);
    }
    hipFree(dout);
    return RW_OK;
}

typedef struct {
    RwVnodeDesc v;
    uint32_t n_outputs;
    const uint32_t* vnode_to_output;
} RwDispatchDesc;

// HashDataDispatcher::dispatch_data (dispatch.rs:949-1050): vnodes from the
// GPU kernel; the tiny per-row visibility/op rewrite is host-side (the
// device-side dense compaction lives in rw_exchange.hip's partition path).
int rw_dispatch_compute(const RwDispatchDesc* d, const RwChunk* chunk,
                        RwChunk** outs) {
    uint32_t n = chunk->n_rows;
    std::vector<uint16_t> vnodes(n);
    int rc = rw_vnode_compute(&d->v, chunk, vnodes.data());
    if (rc != RW_OK) return rc;

    std::vector<uint8_t> ops(chunk->ops, chunk->ops + n);
    auto visible = [&](uint32_t r) { return !chunk->vis || chunk->vis[r]; };
    auto datum_eq_i64 = [&](uint32_t col, uint32_t a, uint32_t b) {
        const RwColumn& c = chunk->cols[col];
        if (c.valid[a] != c.valid[b]) return false;
        if (!c.valid[a]) return true;
        switch (c.type) {
            case RW_T_I32: return ((const int32_t*)c.data)[a] == ((const int32_t*)c.data)[b];
            default: return ((const int64_t*)c.data)[a] == ((const int64_t*)c.data)[b];
        }
    };
    long last_ud = -1;
    for (uint32_t r = 0; r < n; r++) {
        if (!visible(r)) continue;
        if (ops[r] == RW_OP_UPDATE_DELETE) {
            last_ud = (long)r;
        } else if (ops[r] == RW_OP_UPDATE_INSERT && last_ud >= 0) {
            bool changed = false;
            for (uint32_t k = 0; k < d->v.n_keys && !changed; k++)
                changed = !datum_eq_i64(d->v.key_indices[k], (uint32_t)last_ud, r);
            if (changed) {
                ops[last_ud] = RW_OP_DELETE;
                ops[r] = RW_OP_INSERT;
            }
            last_ud = -1;
        }
    }

    for (uint32_t o = 0; o < d->n_outputs; o++) {
        auto* ch = new RwChunk();
        auto* cols = new RwColumn[chunk->n_cols];
        auto* op_arr = new uint8_t[n];
        auto* vis = new uint8_t[n];
        memcpy(op_arr, ops.data(), n);
        for (uint32_t r = 0; r < n; r++)
            vis[r] = visible(r) && d->vnode_to_output[vnodes[r]] == o ? 1 : 0;
        for (uint32_t c = 0; c < chunk->n_cols; c++) {
            uint32_t sz = rw_type_size(chunk->cols[c].type);
            auto* data = new int64_t[n]; // freed as int64_t* by rw_chunk_free
            auto* valid = new uint8_t[n];
            memcpy(data, chunk->cols[c].data, (size_t)n * sz);
            memcpy(valid, chunk->cols[c].valid, n);
            cols[c].type = chunk->cols[c].type;
            cols[c].valid = valid;
            cols[c].data = data;
        }
        ch->n_rows = n;
        ch->n_cols = chunk->n_cols;
        ch->ops = op_arr;
        ch->vis = vis;
        ch->cols = cols;
        outs[o] = ch;
    }
    return RW_OK;
}

} // extern "C"

// ---------------------------------------------------------------------------
// HashJoin (round-1 kernels: Inner + append-only Inner)
// ---------------------------------------------------------------------------
//
// Implements eq_join_oneside for T=Inner (hash_join.rs:949-1075): per probe
// row, walk the other side's rows under the join key, evaluate the optional
// comparison condition (:1294-1302), emit one concatenated output row per
// match with the probe row's (downgraded) op, then apply the probe row to
// its own side's state (:1248-1259). The append-only optimize deletes the
// single matched row and skips the own-side insert (:1241-1245,1359-1364).
// Matched-row iteration order is a chain walk, not memcomparable pk order —
// valid because inner-join epoch outputs are compared as row multisets
// (the reference sorts its own snapshots).

#define MAX_COLS 8
#define MAX_OUT 16

// (JoinSlot/jslot helpers and the DISTINCT dedup kernels moved before
// HashAgg — shared by the agg dedup tables)


struct JoinMeta {
    int KW;
    int n_cols[2];
    int n_pk[2];
    uint8_t key_cols[2][MAX_KW];
    uint8_t pk_cols[2][MAX_KW];
    uint8_t null_safe_mask;
    int n_out;
    uint8_t out_src[MAX_OUT]; // 0 = left, 1 = right
    uint8_t out_col[MAX_OUT];
    uint8_t n_cond; // 0..2 conjuncts (AND)
    struct {
        uint8_t op, src_l, col_l, src_r, col_r;
        long long rconst; // added to the right operand
    } cond[2];
    uint8_t append_only;
    uint8_t join_type;    // RwJoinType
    uint8_t need_deg[2];  // need_left/right_degree (join/mod.rs:153-165)
};

struct JoinBatchDev {
    int64_t* col_vals[MAX_COLS];
    uint8_t* col_valid[MAX_COLS];
    uint8_t* ops;
    uint8_t* vis;
    uint32_t n_rows;
    // every visible row is an Insert: no same-launch thread walks the own
    // side, so record payloads may use plain cached stores (no sc1, no
    // vmcnt drain) — cross-launch visibility comes from the kernel boundary
    uint8_t all_insert;
    // every referenced cell non-NULL: the probe skips the ops/validity
    // byte streams entirely (checked at upload)
    uint8_t all_valid;
    // join keys are pairwise distinct within the batch (host-verified; true
    // by construction for agg-output inputs like Nexmark q8's): at most one
    // chain push per slot per launch, so the head publish needs no CAS
    uint8_t unique_keys;
};

struct JoinOutDev {
    // COLUMNAR output block: column c occupies vals[c*cap ..), so emit
    // stores are lane-coalesced and the q3 agg hop reads it as plain
    // columnar input (stride 1)
    int64_t* vals;   // [n_out][cap]
    uint8_t* nulls;  // [n_out][cap]
    uint8_t* ops;    // [cap]
    uint32_t* counters; // [0]=cursor [1]=overflow
    uint32_t cap;
};

// Find-only with PLAIN (cached) loads — valid only for a side that is not
// mutated during this launch: cross-launch visibility comes from the kernel
// boundary, and cached loads fetch whole lines once instead of per-word
// fabric transactions.
__device__ __forceinline__ uint32_t table_find_cached(
    const uint32_t* state, const int64_t* keys, const uint32_t* key_nulls,
    uint32_t cap_mask, const int64_t* kw, uint32_t nullmask, int KW) {
    uint32_t slot = (uint32_t)(hash_key(kw, nullmask, KW) & cap_mask);
    for (uint32_t probes = 0; probes <= cap_mask; probes++) {
        uint32_t st = state[slot];
        if (st == SLOT_EMPTY) return (uint32_t)-1;
        if (st == SLOT_READY) {
            bool eq = key_nulls[slot] == nullmask;
            for (int i = 0; eq && i < KW; i++)
                eq = keys[(size_t)slot * KW + i] == kw[i];
            if (eq) return slot;
        }
        slot = (slot + 1) & cap_mask;
    }
    return (uint32_t)-1;
}

// Select word `idx` of a register-resident record copy. Unrolled selects:
// a dynamic index into a private array would demote it to scratch, which
// defeats the point of the register walk.
__device__ __forceinline__ long long rec_sel(const long long* recv, int idx) {
    long long v = 0;
#pragma unroll
    for (int w = 0; w < 6; w++)
        if (w == idx) v = recv[w];
    return v;
}

// join_cond_ok over a register-resident match record (no memory re-reads)
__device__ __forceinline__ bool join_cond_ok_rec(const JoinMeta& m,
                                                 int probe_side,
                                                 const JoinBatchDev& b,
                                                 uint32_t r,
                                                 uint32_t validbits,
                                                 const long long* recv) {
    for (int ci = 0; ci < m.n_cond; ci++) {
        auto& cd = m.cond[ci];
        auto fetch = [&](uint8_t src, uint8_t col, int64_t* v) -> bool {
            if ((int)src == probe_side) {
                if (!b.col_valid[col][r]) return false;
                *v = b.col_vals[col][r];
            } else {
                if (!((validbits >> col) & 1)) return false;
                *v = rec_sel(recv, col);
            }
            return true;
        };
        int64_t a, c;
        if (!fetch(cd.src_l, cd.col_l, &a)) return false; // NULL ⇒ false
        if (!fetch(cd.src_r, cd.col_r, &c)) return false;
        c += cd.rconst;
        bool ok;
        switch (cd.op) {
            case RW_CMP_LT: ok = a < c; break;
            case RW_CMP_LE: ok = a <= c; break;
            case RW_CMP_GT: ok = a > c; break;
            case RW_CMP_GE: ok = a >= c; break;
            default: ok = false;
        }
        if (!ok) return false;
    }
    return true;
}

__device__ __forceinline__ bool join_cond_ok(const JoinMeta& m, int probe_side,
                                             const JoinBatchDev& b, uint32_t r,
                                             uint32_t validbits,
                                             const long long* mvals) {
    for (int ci = 0; ci < m.n_cond; ci++) {
        auto& cd = m.cond[ci];
        auto fetch = [&](uint8_t src, uint8_t col, int64_t* v) -> bool {
            if ((int)src == probe_side) {
                if (!b.col_valid[col][r]) return false;
                *v = b.col_vals[col][r];
            } else {
                if (!((validbits >> col) & 1)) return false;
                *v = mvals[col];
            }
            return true;
        };
        int64_t a, c;
        if (!fetch(cd.src_l, cd.col_l, &a)) return false; // NULL ⇒ false
        if (!fetch(cd.src_r, cd.col_r, &c)) return false;
        c += cd.rconst;
        bool ok;
        switch (cd.op) {
            case RW_CMP_LT: ok = a < c; break;
            case RW_CMP_LE: ok = a <= c; break;
            case RW_CMP_GT: ok = a > c; break;
            case RW_CMP_GE: ok = a >= c; break;
            default: ok = false;
        }
        if (!ok) return false;
    }
    return true;
}

// probe the match side + update own side, one thread per probe row.
// Emission reserves output rows with ONE wave-level atomicAdd per iteration;
// the match side is immutable during a non-append-only launch, so it is read
// with plain cached loads (one line per slot, one per record). The
// append-only path (in-launch kill-on-match) uses the sc1 protocol.

// ---- join-type predicates (join/mod.rs:120-165; oracle_join.cpp mirrors) ----
__host__ __device__ __forceinline__ bool j_is_semi(uint8_t t) {
    return t == RW_JOIN_LEFT_SEMI || t == RW_JOIN_RIGHT_SEMI;
}
__host__ __device__ __forceinline__ bool j_is_anti(uint8_t t) {
    return t == RW_JOIN_LEFT_ANTI || t == RW_JOIN_RIGHT_ANTI;
}
__host__ __device__ __forceinline__ bool j_is_outer_side(uint8_t t, int side) {
    return t == RW_JOIN_FULL_OUTER ||
           (t == RW_JOIN_LEFT_OUTER && side == RW_SIDE_LEFT) ||
           (t == RW_JOIN_RIGHT_OUTER && side == RW_SIDE_RIGHT);
}
__host__ __device__ __forceinline__ bool j_outer_side_null(uint8_t t, int side) {
    return t == RW_JOIN_FULL_OUTER ||
           (t == RW_JOIN_LEFT_OUTER && side == RW_SIDE_RIGHT) ||
           (t == RW_JOIN_RIGHT_OUTER && side == RW_SIDE_LEFT);
}
__host__ __device__ __forceinline__ bool j_forward_exactly_once(uint8_t t, int side) {
    return ((t == RW_JOIN_LEFT_SEMI || t == RW_JOIN_LEFT_ANTI) && side == RW_SIDE_LEFT) ||
           ((t == RW_JOIN_RIGHT_SEMI || t == RW_JOIN_RIGHT_ANTI) && side == RW_SIDE_RIGHT);
}
__host__ __device__ __forceinline__ bool j_only_forward_matched_side(uint8_t t, int side) {
    return ((t == RW_JOIN_LEFT_SEMI || t == RW_JOIN_LEFT_ANTI) && side == RW_SIDE_RIGHT) ||
           ((t == RW_JOIN_RIGHT_SEMI || t == RW_JOIN_RIGHT_ANTI) && side == RW_SIDE_LEFT);
}
__host__ __device__ __forceinline__ bool j_need_degree(uint8_t t, int side) {
    if (side == RW_SIDE_LEFT)
        return t == RW_JOIN_FULL_OUTER || t == RW_JOIN_LEFT_OUTER ||
               t == RW_JOIN_LEFT_ANTI || t == RW_JOIN_LEFT_SEMI;
    return t == RW_JOIN_FULL_OUTER || t == RW_JOIN_RIGHT_OUTER ||
           t == RW_JOIN_RIGHT_ANTI || t == RW_JOIN_RIGHT_SEMI;
}

// emit one output row with per-emission reservation. form bit0 = include
// the probe side's columns, bit1 = include the matched record's columns;
// excluded sides are NULL (JoinOutBuilder append_row / _update / _matched,
// join/builder.rs:153-318).
__device__ __forceinline__ void jemit_row(JoinOutDev& out, const JoinMeta& m,
                                          int S, uint8_t op,
                                          const JoinBatchDev& b, uint32_t r,
                                          uint32_t match_vb,
                                          const long long* mv, int form) {
    uint32_t orow = atomicAdd(&out.counters[0], 1u);
    if (orow >= out.cap) {
        atomicExch(&out.counters[1], 1u);
        return;
    }
    out.ops[orow] = op;
    for (int c = 0; c < m.n_out; c++) {
        bool from_probe = (int)m.out_src[c] == S;
        uint8_t col = m.out_col[c];
        int64_t v = 0;
        uint8_t valid = 0;
        if (from_probe && (form & 1)) {
            valid = b.col_valid[col][r];
            v = b.col_vals[col][r];
        } else if (!from_probe && (form & 2)) {
            valid = (match_vb >> col) & 1;
            v = mv[col];
        }
        out.vals[(size_t)c * out.cap + orow] = valid ? v : 0;
        out.nulls[(size_t)c * out.cap + orow] = !valid;
    }
}

// jemit_row over a REGISTER-RESIDENT match record (see the register walk
// in join_probe_kernel): match values come from the register copy via
// unrolled selects — a dynamic index into a private array would demote
// it to scratch and defeat the single-fetch walk.
__device__ __forceinline__ void jemit_row_rec(JoinOutDev& out,
                                              const JoinMeta& m, int S,
                                              uint8_t op,
                                              const JoinBatchDev& b,
                                              uint32_t r, uint32_t match_vb,
                                              const long long* recv,
                                              int form) {
    uint32_t orow = atomicAdd(&out.counters[0], 1u);
    if (orow >= out.cap) {
        atomicExch(&out.counters[1], 1u);
        return;
    }
    out.ops[orow] = op;
    for (int c = 0; c < m.n_out; c++) {
        bool from_probe = (int)m.out_src[c] == S;
        uint8_t col = m.out_col[c];
        int64_t v = 0;
        uint8_t valid = 0;
        if (from_probe && (form & 1)) {
            valid = b.col_valid[col][r];
            v = b.col_vals[col][r];
        } else if (!from_probe && (form & 2)) {
            valid = (match_vb >> col) & 1;
            v = rec_sel(recv, col);
        }
        out.vals[(size_t)c * out.cap + orow] = valid ? v : 0;
        out.nulls[(size_t)c * out.cap + orow] = !valid;
    }
}

// own-side insert/delete (join/hash_join.rs:591-681 without the LRU tier)
__device__ __forceinline__ void jown_insert(JoinSideDev own, const JoinMeta& m,
                                            int S, const JoinBatchDev& b,
                                            uint32_t r, const int64_t* kw,
                                            uint32_t nullmask, JoinOutDev out,
                                            uint32_t init_deg) {
    // wave-aggregated row reservation: the single cursor sustains only
    // ~2.1 G same-line atomicAdds/s, so 1M per-lane adds cost ~0.47 ms per
    // launch (measured via RW_JOIN_SKIP) — one leader add per wave instead
    uint64_t wmask = __ballot(true); // lanes currently inserting
    int lane = threadIdx.x & 63;
    int leader = 63 - __clzll(wmask);
    uint32_t base = 0;
    if (lane == leader)
        base = atomicAdd(own.row_cursor, (uint32_t)__popcll(wmask));
    base = (uint32_t)__shfl((int)base, leader);
    uint32_t row = base + (uint32_t)__popcll(wmask & ((1ULL << lane) - 1));
    if (row >= own.row_cap) {
        atomicExch(&out.counters[1], 3u); // row store full
        return;
    }
    uint64_t h64 = hash_key(kw, nullmask, m.KW);
    JoinRowHdr* h = jrow(own, row);
    uint32_t vb = 0;
    long long* hv = jvals(h);
    if (b.all_insert) {
        // plain cached stores: inserts never read records (chained
        // buckets), and all-Insert launches have no same-side walkers;
        // jbucket_insert drains to the coherence point before the CAS
        // publish so later foreign-key walks fetch real bytes.
        for (int c = 0; c < m.n_cols[S]; c++) {
            hv[c] = b.col_vals[c][r];
            vb |= (uint32_t)(b.col_valid[c][r] != 0) << c;
        }
        h->validbits = vb;
        h->degree = init_deg;
        h->alive = 1;
        jbucket_insert(own, h64, row, /*sc1_next=*/false);
    } else {
        for (int c = 0; c < m.n_cols[S]; c++) {
            st_i64((int64_t*)&hv[c], b.col_vals[c][r]);
            vb |= (uint32_t)(b.col_valid[c][r] != 0) << c;
        }
        st_u32(&h->validbits, vb);
        st_u32(&h->degree, init_deg);
        st_u32(&h->alive, 1);
        jbucket_insert(own, h64, row, /*sc1_next=*/true);
    }
}

__device__ __forceinline__ void jown_delete(JoinSideDev own, const JoinMeta& m,
                                            int S, const JoinBatchDev& b,
                                            uint32_t r, const int64_t* kw,
                                            uint32_t nullmask) {
    // delete own row: FULL-row compare + CAS claim; the
    // full-row compare subsumes the key filter (key cols are row cols)
    uint32_t row = jbucket_head(own, hash_key(kw, nullmask, m.KW), true);
    while (row != UINT32_MAX) {
        JoinRowHdr* h = jrow(own, row);
        if (ld_u32(&h->alive)) {
            uint32_t vb = ld_u32(&h->validbits);
            const long long* hv = jvals(h);
            bool eq = true;
            for (int c = 0; eq && c < m.n_cols[S]; c++) {
                uint8_t va = b.col_valid[c][r];
                uint8_t vbc = (vb >> c) & 1;
                eq = (va == vbc) &&
                     (!va || b.col_vals[c][r] == ld_i64((const int64_t*)&hv[c]));
            }
            if (eq && atomicCAS(&h->alive, 1u, 0u) == 1u) {
                if (own.killed) {
                    uint32_t kidx = atomicAdd(own.killed_cursor, 1u);
                    if (kidx < own.killed_cap) own.killed[kidx] = row;
                }
                break;
            }
        }
        row = ld_u32(&h->next);
    }
}

// one probe row of a non-inner join: eq_join_oneside for
// LeftOuter/RightOuter/FullOuter/Semi/Anti (hash_join.rs:949-1374 +
// with_match_on_insert/delete, join/builder.rs:173-318). Degree transitions
// use the matched record's atomic degree; the emitted transition rows are
// independent of which chunk row wins a concurrent transition, so the
// per-chunk output multiset matches the reference's sequential apply
// (mixed insert/delete chunks sharing a join key run in serial segments —
// conflict_segments).
__device__ void join_probe_row_noninner(const JoinBatchDev& b, JoinSideDev own,
                                        JoinSideDev match, const JoinMeta& m,
                                        int S, JoinOutDev out, uint32_t r,
                                        bool is_insert, uint8_t op,
                                        const int64_t* kw, uint32_t nullmask,
                                        bool never_match) {
    uint8_t T = m.join_type;
    bool fwd_once = j_forward_exactly_once(T, S);
    bool anti = j_is_anti(T), semi = j_is_semi(T);
    bool outer_self = j_is_outer_side(T, S);
    bool only_fwd_m = j_only_forward_matched_side(T, S);
    bool outer_null = j_outer_side_null(T, S);
    if (never_match) {
        // null-safe NeverMatch forwards for anti/outer (hash_join.rs:1143-1152)
        if ((anti && fwd_once) || outer_self)
            jemit_row(out, m, S, op, b, r, 0, nullptr, 1);
        return; // no state write
    }
    uint32_t mhead =
        jbucket_head(match, hash_key(kw, nullmask, m.KW), false);
    uint32_t my_deg = 0;
    {
        // REGISTER-RESIDENT walk for narrow aligned records (same refetch
        // pathology and fix as the inner probe's walk);
        // wide records keep the field walk. The degree update stays an
        // atomic on the record itself (other lanes observe it through the
        // atomic's return value, never through their register copies).
        const bool regw =
            match.row_stride <= 64 && !(match.row_stride & 15);
        const int nw16 = (int)(match.row_stride >> 4);
        long long rec[8];
        uint32_t row = mhead;
        while (row != UINT32_MAX) {
            JoinRowHdr* h = jrow(match, row);
            uint32_t alive, nxt, vb;
            if (regw) {
                const long long* rp = (const long long*)h;
#pragma unroll
                for (int w = 0; w < 4; w++)
                    if (w < nw16)
                        *(longlong2*)&rec[2 * w] =
                            *(const longlong2*)(rp + 2 * w);
                alive = (uint32_t)(uint64_t)rec[0];
                nxt = (uint32_t)((uint64_t)rec[0] >> 32);
                vb = (uint32_t)(uint64_t)rec[1];
            } else {
                alive = h->alive;
                nxt = h->next;
                vb = h->validbits;
            }
            bool eq = alive != 0;
            for (int i = 0; eq && i < m.KW; i++) {
                uint8_t col = m.key_cols[1 - S][i];
                bool valid = (vb >> col) & 1;
                if (valid == (bool)((nullmask >> i) & 1)) {
                    eq = false;
                } else if (valid) {
                    long long v =
                        regw ? rec_sel(rec + 2, col) : jvals(h)[col];
                    if (v != kw[i]) eq = false;
                }
            }
            if (eq && (regw ? join_cond_ok_rec(m, S, b, r, vb, rec + 2)
                            : join_cond_ok(m, S, b, r, vb, jvals(h)))) {
                my_deg++;
                bool zero = false;
                if (m.need_deg[1 - S]) {
                    // insert: emit BEFORE increment (zero == pre-value 0);
                    // delete: decrement BEFORE emit (zero == post-value 0)
                    // (hash_join.rs:1311-1342)
                    uint32_t old = atomicAdd(&h->degree,
                                             is_insert ? 1u : (uint32_t)-1);
                    zero = is_insert ? old == 0 : old == 1;
                    // §8f-2 degree-table delta: record the touched row once
                    if (match.deg_dirty_flag &&
                        ld_u32(&match.deg_dirty_flag[row]) == 0 &&
                        atomicCAS(&match.deg_dirty_flag[row], 0u, 1u) == 0u) {
                        uint32_t di = atomicAdd(match.deg_dirty_n, 1u);
                        match.deg_dirty_list[di] = row;
                    }
                }
                auto emit2 = [&](uint8_t op2, int form) {
                    if (regw)
                        jemit_row_rec(out, m, S, op2, b, r, vb, rec + 2,
                                      form);
                    else
                        jemit_row(out, m, S, op2, b, r, vb, jvals(h), form);
                };
                if (is_insert) {
                    if (anti) {
                        if (zero && only_fwd_m) emit2(RW_OP_DELETE, 2);
                    } else if (semi) {
                        if (zero && only_fwd_m) emit2(RW_OP_INSERT, 2);
                    } else if (zero && outer_null) {
                        emit2(RW_OP_DELETE, 2);
                        emit2(RW_OP_INSERT, 3);
                    } else if (!fwd_once) {
                        emit2(RW_OP_INSERT, 3);
                    }
                } else {
                    if (anti) {
                        if (zero && only_fwd_m) emit2(RW_OP_INSERT, 2);
                    } else if (semi) {
                        if (zero && only_fwd_m) emit2(RW_OP_DELETE, 2);
                    } else if (zero && outer_null) {
                        emit2(RW_OP_DELETE, 3);
                        emit2(RW_OP_INSERT, 2);
                    } else if (!fwd_once) {
                        emit2(RW_OP_DELETE, 3);
                    }
                }
            }
            row = nxt;
        }
    }
    if (my_deg == 0) {
        // forward_if_not_matched (join/builder.rs:303-312)
        if ((anti && fwd_once) || outer_self)
            jemit_row(out, m, S, op, b, r, 0, nullptr, 1);
    } else if (semi && fwd_once) {
        // forward_exactly_once_if_matched (join/builder.rs:287-300)
        jemit_row(out, m, S, op, b, r, 0, nullptr, 1);
    }
    if (is_insert)
        jown_insert(own, m, S, b, r, kw, nullmask, out, my_deg);
    else
        jown_delete(own, m, S, b, r, kw, nullmask);
}

// Per-batch setup for the inner probe's PRE-ASSIGNED layout (see the
// kernel comment below): output region [0, n) is one sparse slot per
// probe row (ops sentinel 0xFF = empty; the emission cursor starts at n
// for >1-match overflow), and the batch's record rows are reserved as ONE
// cursor bump so the kernel needs no cross-lane reservation at all.
__global__ void jprobe_setup_kernel(JoinOutDev out, uint32_t* row_cursor,
                                    uint32_t row_cap, uint32_t* row_base,
                                    uint32_t n) {
    out.counters[0] = n; // overflow emissions append past the sparse region
    uint32_t base = atomicAdd(row_cursor, n);
    if ((uint64_t)base + n > row_cap) atomicExch(&out.counters[1], 3u);
    *row_base = base;
}

__global__ void join_count_emitted_kernel(const uint8_t* ops, uint32_t n,
                                          const uint32_t* counters,
                                          unsigned long long* out_count) {
    uint32_t stride = gridDim.x * blockDim.x;
    unsigned long long local = 0;
    for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < n;
         r += stride)
        local += ops[r] != 0xFF;
    if (local) atomicAdd(out_count, local);
    if (blockIdx.x == 0 && threadIdx.x == 0)
        atomicAdd(out_count, (unsigned long long)(counters[0] - n));
}

// non-inner probe (outer/semi/anti): one thread per row over the shared
// per-row prologue, then the matched-side walk + degree/emission tree in
// join_probe_row_noninner. Separate kernel so its wider live range
// (register-resident records + the emission decision tree) gets its own
// launch bounds instead of spilling the inner kernel.
__global__ __launch_bounds__(256, 4) void join_probe_noninner_kernel(
    JoinBatchDev b, JoinSideDev own, JoinSideDev match, JoinMeta m, int S,
    JoinOutDev out, uint32_t r0, uint32_t r1, int dbg_skip) {
    uint32_t stride = gridDim.x * blockDim.x;
    for (uint32_t r = r0 + blockIdx.x * blockDim.x + threadIdx.x; r < r1;
         r += stride) {
        if (b.vis && !b.vis[r]) continue;
        uint8_t op_in = b.ops[r];
        bool is_insert = (op_in == RW_OP_INSERT || op_in == RW_OP_UPDATE_INSERT);
        uint8_t op = is_insert ? RW_OP_INSERT : RW_OP_DELETE;
        int64_t kw[MAX_KW];
        uint32_t nullmask = 0;
        for (int i = 0; i < m.KW; i++) {
            uint8_t col = m.key_cols[S][i];
            bool valid = b.col_valid[col][r];
            kw[i] = valid ? b.col_vals[col][r] : 0;
            nullmask |= (!valid) << i;
        }
        bool never_match = (nullmask & ~(uint32_t)m.null_safe_mask) != 0;
        if (!(dbg_skip & 2))
            join_probe_row_noninner(b, own, match, m, S, out, r, is_insert,
                                    op, kw, nullmask, never_match);
    }
}

__global__ __launch_bounds__(256, 6) void join_probe_kernel(
    JoinBatchDev b, JoinSideDev own, JoinSideDev match, JoinMeta m, int S,
    JoinOutDev out, uint32_t r0, uint32_t r1, const uint32_t* row_base,
    int dbg_skip = 0) {
    uint32_t stride = gridDim.x * blockDim.x;
    uint32_t n = r1 - r0;
    uint32_t iters = (n + stride - 1) / stride;

    const bool dense = b.all_insert && b.all_valid && !b.vis;
    for (uint32_t it = 0; it < iters; it++) {
        uint32_t r = r0 + it * stride + blockIdx.x * blockDim.x + threadIdx.x;
        bool active = dense ? (r < r1)
                            : ((r < r1) && !(b.vis && !b.vis[r]));
        uint8_t op_in =
            (active && !dense) ? b.ops[r] : RW_OP_INSERT;
        bool is_insert = (op_in == RW_OP_INSERT || op_in == RW_OP_UPDATE_INSERT);
        uint8_t op = is_insert ? RW_OP_INSERT : RW_OP_DELETE;
        int64_t kw[MAX_KW];
        uint32_t nullmask = 0;
        bool never_match = false;
        if (active) {
            for (int i = 0; i < m.KW; i++) {
                uint8_t col = m.key_cols[S][i];
                bool valid = dense || b.col_valid[col][r];
                kw[i] = valid ? b.col_vals[col][r] : 0;
                nullmask |= (!valid) << i;
            }
            // null-safe NeverMatch (hash_join.rs:1004-1016)
            if (nullmask & ~(uint32_t)m.null_safe_mask) never_match = true;
        }

        // non-inner types run in join_probe_noninner_kernel (own launch
        // bounds: its register walk + emission tree would spill this
        // kernel's budget)
        if (never_match) active = false;

        uint32_t mhead = UINT32_MAX;
        uint32_t my_n = 0;
        uint32_t matched_row = UINT32_MAX;
        uint64_t h64 = 0;
        // register-resident copy of the last record the walk loaded, and
        // whether it is the (single) match — lets the emit below run with
        // zero record re-reads in the dominant unique-key case
        long long rec[8];
        bool rec_match = false;
        if (active) {
            h64 = hash_key(kw, nullmask, m.KW);
            mhead = jbucket_head(match, h64, m.append_only != 0);
            if (mhead != UINT32_MAX && !m.append_only && !(dbg_skip & 16)) {
                // dbg 16 = hash+slot only (stage isolation: no record walk)
                uint32_t row = mhead;
                // REGISTER-RESIDENT walk (narrow records, 16-B-aligned
                // stride): each record is fetched ONCE as wide b128 loads
                // into registers and evaluated from there. The serialized
                // field-by-field walk refetches its line from HBM 2-3x —
                // with ~2048 lanes in flight per CU the 32-KB vL0 (and the
                // 4-MB XCD L2) evict a line between a record's `alive`
                // check and its key loads; PMC measured ~323 B fetched per
                // 64-B record (gpurun_out/pmcW67.txt) vs the one-line 128-B
                // model. dbg 32 forces the legacy field walk for A/B.
                if (match.row_stride <= 64 && !(match.row_stride & 15) &&
                    !(dbg_skip & 32)) {
                    int nw16 = (int)(match.row_stride >> 4);
                    while (row != UINT32_MAX) {
                        const long long* rp =
                            (const long long*)jrow(match, row);
#pragma unroll
                        for (int w = 0; w < 4; w++)
                            if (w < nw16)
                                *(longlong2*)&rec[2 * w] =
                                    *(const longlong2*)(rp + 2 * w);
                        rec_match = false; // rec just overwritten
                        uint32_t alive = (uint32_t)(uint64_t)rec[0];
                        uint32_t nxt = (uint32_t)((uint64_t)rec[0] >> 32);
                        uint32_t vb = (uint32_t)(uint64_t)rec[1];
                        bool eq = alive != 0;
                        for (int i = 0; eq && i < m.KW; i++) {
                            uint8_t col = m.key_cols[1 - S][i];
                            bool valid = (vb >> col) & 1;
                            if (valid == (bool)((nullmask >> i) & 1))
                                eq = false;
                            else if (valid &&
                                     rec_sel(rec + 2, col) != kw[i])
                                eq = false;
                        }
                        if (eq &&
                            join_cond_ok_rec(m, S, b, r, vb, rec + 2)) {
                            my_n++;
                            matched_row = row;
                            rec_match = true; // match is register-resident
                        }
                        row = nxt;
                    }
                } else {
                    while (row != UINT32_MAX) {
                        JoinRowHdr* h = jrow(match, row);
                        if (h->alive &&
                            jhdr_key_eq(h, m.key_cols[1 - S], m.KW, kw,
                                        nullmask, false) &&
                            join_cond_ok(m, S, b, r, h->validbits,
                                         jvals(h))) {
                            my_n++;
                            matched_row = row;
                        }
                        row = h->next;
                    }
                }
            }
        }

        if (!m.append_only) {
            // PRE-ASSIGNED sparse emission: match 0 of probe row r goes to
            // output slot r (ops sentinel 0xFF marks empty slots; the host
            // compacts at drain); matches 2..k append past n via a
            // PER-LANE atomic on the rare multi-match lanes. NO cross-lane
            // ops: the q8-shape ladder measured ANY wave-convergent
            // reservation (ballot or shuffle scan) at +135-155 us per 1M
            // rows — the convergence stalls the wave on its slowest lane's
            // dependent loads — while the pre-assigned form runs at the
            // random-access ceiling (profiles/r02_membw_probes.json).
            uint32_t my_base = r;
            uint32_t extra_base = 0;
            if (my_n > 1) {
                extra_base = atomicAdd(&out.counters[0], my_n - 1);
                if (extra_base + my_n - 1 > out.cap)
                    atomicExch(&out.counters[1], 1u); // overflow
            }
            if (my_n == 1 && !(dbg_skip & 1) && rec_match) {
                // SINGLE-PASS emission from the REGISTER-RESIDENT match
                // (the walk's last-loaded record): zero record re-reads.
                // Store policy A/B (dbg 8 = fully cached, dbg 256 = vals
                // NT + 1-byte ops/nulls cached): measured 0.251 / 0.255 /
                // 0.255 ms/step — full NT is the (narrow) winner, the
                // dense orow=r layout coalesces well under every policy.
                uint32_t orow = my_base;
                uint32_t mvb = (uint32_t)(uint64_t)rec[1];
                bool nt_vals = !(dbg_skip & 8);
                bool nt_meta = !(dbg_skip & (8 | 256));
                if (nt_meta)
                    __builtin_nontemporal_store(op, &out.ops[orow]);
                else
                    out.ops[orow] = op;
                for (int c = 0; c < m.n_out; c++) {
                    bool from_probe = (int)m.out_src[c] == S;
                    uint8_t col = m.out_col[c];
                    int64_t v;
                    uint8_t valid;
                    if (from_probe) {
                        valid = dense || b.col_valid[col][r];
                        v = b.col_vals[col][r];
                    } else {
                        valid = (mvb >> col) & 1;
                        v = rec_sel(rec + 2, col);
                    }
                    if (nt_vals)
                        __builtin_nontemporal_store(
                            valid ? v : 0,
                            &out.vals[(size_t)c * out.cap + orow]);
                    else
                        out.vals[(size_t)c * out.cap + orow] = valid ? v : 0;
                    if (nt_meta)
                        __builtin_nontemporal_store(
                            (uint8_t)!valid,
                            &out.nulls[(size_t)c * out.cap + orow]);
                    else
                        out.nulls[(size_t)c * out.cap + orow] =
                            (uint8_t)!valid;
                }
            } else if (my_n == 1 && !(dbg_skip & 1)) {
                // SINGLE-PASS emission for the dominant <=1-match case:
                // the count walk already identified the row — no re-walk
                JoinRowHdr* h = jrow(match, matched_row);
                uint32_t orow = my_base;
                const long long* mv = jvals(h);
                __builtin_nontemporal_store(op, &out.ops[orow]);
                for (int c = 0; c < m.n_out; c++) {
                    bool from_probe = (int)m.out_src[c] == S;
                    uint8_t col = m.out_col[c];
                    int64_t v;
                    uint8_t valid;
                    if (from_probe) {
                        valid = dense || b.col_valid[col][r];
                        v = b.col_vals[col][r];
                    } else {
                        valid = (h->validbits >> col) & 1;
                        v = mv[col];
                    }
                    __builtin_nontemporal_store(
                        valid ? v : 0, &out.vals[(size_t)c * out.cap + orow]);
                    __builtin_nontemporal_store(
                        (uint8_t)!valid,
                        &out.nulls[(size_t)c * out.cap + orow]);
                }
            } else if (my_n && !(dbg_skip & 1) &&
                !(my_n > 1 && extra_base + my_n - 1 > out.cap)) {
                // second walk: emit (JoinStreamChunkBuilder::append_row).
                // Emit stores are NONTEMPORAL: the rows are written once
                // and only read by later launches, and regular stores pay
                // a read-for-ownership line fetch per partial write
                // (RW_JOIN_NT_OFF=1 re-measures the cached variant).
                uint32_t row = mhead;
                uint32_t k = 0;
                while (row != UINT32_MAX && k < my_n) {
                    JoinRowHdr* h = jrow(match, row);
                    if (h->alive &&
                        jhdr_key_eq(h, m.key_cols[1 - S], m.KW, kw, nullmask,
                                    false) &&
                        join_cond_ok(m, S, b, r, h->validbits, jvals(h))) {
                        uint32_t orow = k == 0 ? my_base
                                               : extra_base + k - 1;
                        const long long* mv = jvals(h);
                        if (dbg_skip & 8) {
                            out.ops[orow] = op;
                            for (int c = 0; c < m.n_out; c++) {
                                bool from_probe = (int)m.out_src[c] == S;
                                uint8_t col = m.out_col[c];
                                int64_t v;
                                uint8_t valid;
                                if (from_probe) {
                                    valid = b.col_valid[col][r];
                                    v = b.col_vals[col][r];
                                } else {
                                    valid = (h->validbits >> col) & 1;
                                    v = mv[col];
                                }
                                out.vals[(size_t)c * out.cap + orow] = valid ? v : 0;
                                out.nulls[(size_t)c * out.cap + orow] = !valid;
                            }
                        } else {
                            __builtin_nontemporal_store(op, &out.ops[orow]);
                            for (int c = 0; c < m.n_out; c++) {
                                bool from_probe = (int)m.out_src[c] == S;
                                uint8_t col = m.out_col[c];
                                int64_t v;
                                uint8_t valid;
                                if (from_probe) {
                                    valid = b.col_valid[col][r];
                                    v = b.col_vals[col][r];
                                } else {
                                    valid = (h->validbits >> col) & 1;
                                    v = mv[col];
                                }
                                __builtin_nontemporal_store(
                                    valid ? v : 0,
                                    &out.vals[(size_t)c * out.cap + orow]);
                                __builtin_nontemporal_store(
                                    (uint8_t)!valid,
                                    &out.nulls[(size_t)c * out.cap + orow]);
                            }
                        }
                        k++;
                    }
                    row = h->next;
                }
            }
        } else if (active && mhead != UINT32_MAX) {
            // append-only path: <=1 match; sc1 reads, per-match atomics
            uint32_t row = mhead;
            while (row != UINT32_MAX) {
                JoinRowHdr* h = jrow(match, row);
                uint32_t vb = ld_u32(&h->validbits);
                if (ld_u32(&h->alive) &&
                    jhdr_key_eq(h, m.key_cols[1 - S], m.KW, kw, nullmask,
                                true) &&
                    join_cond_ok(m, S, b, r, vb, jvals(h))) {
                    uint32_t orow = atomicAdd(&out.counters[0], 1u);
                    if (orow >= out.cap) {
                        atomicExch(&out.counters[1], 1u);
                    } else {
                        out.ops[orow] = op;
                        const long long* mv = jvals(h);
                        for (int c = 0; c < m.n_out; c++) {
                            bool from_probe = (int)m.out_src[c] == S;
                            uint8_t col = m.out_col[c];
                            int64_t v;
                            uint8_t valid;
                            if (from_probe) {
                                valid = b.col_valid[col][r];
                                v = b.col_vals[col][r];
                            } else {
                                valid = (vb >> col) & 1;
                                v = (int64_t)ld_i64((const int64_t*)&mv[col]);
                            }
                            out.vals[(size_t)c * out.cap + orow] = valid ? v : 0;
                            out.nulls[(size_t)c * out.cap + orow] = !valid;
                        }
                    }
                    matched_row = row;
                }
                row = ld_u32(&h->next);
            }
        }

        if (!m.append_only) {
            // PRE-ASSIGNED record slot (row_base + r): inserting rows write
            // their record and link the bucket; every other row marks its
            // slot dead (walks and drains skip by `alive`). No cross-lane
            // cursor — see the emission comment above. dbg 64 = skip this
            // whole section (stage isolation: clean walk timing without
            // the pre-kill store pass the production path replaces with
            // full record writes).
            if (r < r1 && !(dbg_skip & 64)) {
                uint32_t myrow = *row_base + r;
                bool do_insert = active && !(dbg_skip & 2) && is_insert;
                if (myrow < own.row_cap) {
                    JoinRowHdr* hd2 = jrow(own, myrow);
                    if (do_insert) {
                        long long* hv2 = jvals(hd2);
                        uint32_t vb2 = 0;
                        // (Measured dead end: nontemporal record-insert
                        // stores — both 8-B fields and register-assembled
                        // b128s — ran 0.320 vs 0.248 ms/step on q8. The
                        // 64-B lane stride means no single instruction
                        // covers a full line, so NT partial-line writes
                        // reach HBM unmerged, while the cached path's RFO
                        // is amortized by L2 byte-granular write merge.)
                        if (dense) {
                            for (int c = 0; c < m.n_cols[S]; c++)
                                hv2[c] = b.col_vals[c][r];
                            vb2 = (1u << m.n_cols[S]) - 1;
                            hd2->validbits = vb2;
                            hd2->degree = 0;
                            hd2->alive = 1;
                        } else if (b.all_insert) {
                            for (int c = 0; c < m.n_cols[S]; c++) {
                                hv2[c] = b.col_vals[c][r];
                                vb2 |= (uint32_t)(b.col_valid[c][r] != 0) << c;
                            }
                            hd2->validbits = vb2;
                            hd2->degree = 0;
                            hd2->alive = 1;
                        } else {
                            // sc1 payload: same-launch delete walkers read it
                            for (int c = 0; c < m.n_cols[S]; c++) {
                                st_i64((int64_t*)&hv2[c], b.col_vals[c][r]);
                                vb2 |= (uint32_t)(b.col_valid[c][r] != 0) << c;
                            }
                            st_u32(&hd2->validbits, vb2);
                            st_u32(&hd2->degree, 0);
                            st_u32(&hd2->alive, 1);
                        }
                        jbucket_insert(own, h64, myrow, !b.all_insert);
                    } else {
                        hd2->alive = 0;
                    }
                }
                if (active && !(dbg_skip & 2) && !is_insert)
                    jown_delete(own, m, S, b, r, kw, nullmask);
            }
            continue;
        }
        if (!active || (dbg_skip & 2)) continue;

        if (m.append_only && is_insert && matched_row != UINT32_MAX) {
            // append-only optimize (hash_join.rs:1241-1245)
            st_u32(&jrow(match, matched_row)->alive, 0);
            if (match.killed) {
                uint32_t kidx = atomicAdd(match.killed_cursor, 1u);
                if (kidx < match.killed_cap) match.killed[kidx] = matched_row;
            }
            continue;
        }

        // own-side state update (join/hash_join.rs:591-681 without LRU tier)
        if (is_insert)
            jown_insert(own, m, S, b, r, kw, nullmask, out, 0);
        else
            jown_delete(own, m, S, b, r, kw, nullmask);
    }
}

// ============ Partitioned probe/insert pipeline (q8 hot path) ============
//
// The round-1 single-kernel path was random-access-bound: each probe row
// touched ~3 random 64-B lines spread over GBs of slot table + record
// store (PMC: 500–1000 B fetched per row vs the 128-B algorithmic model,
// profiles/r01_q8_pmc_fetch.txt), sustaining ~2 G own-side inserts/s.
// This pipeline partitions an all-Insert unique-key batch by the TOP
// JPART_LOG2 bits of the join-key hash. jslot_start() derives slot
// indices from the same top bits, so partition p's slots are the
// CONTIGUOUS window [p*cap/P, (p+1)*cap/P) of BOTH sides' tables, and its
// records land in one contiguous row-store run per batch (the scatter
// writes each record into its FINAL row-store position). The fused
// probe+insert phase runs one workgroup per partition: its slot window
// (~cap*64/P bytes) stays in the XCD's L2 and its match-record reads fall
// in per-batch partition runs — DRAM-row-local instead of uniformly
// random. eq_join_oneside semantics are unchanged (inner join, all
// visible rows Insert, keys pairwise distinct — the host gates on exactly
// the conditions under which batch-parallel insert+probe commutes with
// the reference's sequential per-row loop; hash_join.rs:949-1075).
//
// Phases (stream-ordered, no host round-trip):
//   1. jpart_count          per-block LDS histograms of partition ids
//   2. jpart_scan           scatter bases + partition ranges + row-store
//                           reservation (single workgroup)
//   3. jpart_scatter        write final row records, partition-clustered
//   4. jpart_probe_insert   per partition: probe match side + emit, then
//                           link own-side chains (plain RMW — unique keys
//                           give every touched slot exactly one writer)

#define JPART_LOG2 12
#define JPART_P (1u << JPART_LOG2)
#define JPART_NBLK 256
#define JPART_MIN_ROWS 131072u

__device__ __forceinline__ uint32_t jpart_of(uint64_t h) {
    return (uint32_t)(h >> (64 - JPART_LOG2));
}

// per-row key extraction from the input batch; returns false for rows that
// carry no state (invisible is excluded by the host gate; never-match NULL
// keys are not inserted and emit nothing under inner join,
// hash_join.rs:1004-1016)
__device__ __forceinline__ bool jpart_row_key(const JoinBatchDev& b,
                                              const JoinMeta& m, int S,
                                              uint32_t r, int64_t* kw,
                                              uint32_t* nullmask) {
    uint32_t nm = 0;
    for (int i = 0; i < m.KW; i++) {
        uint8_t col = m.key_cols[S][i];
        bool valid = b.col_valid[col][r];
        kw[i] = valid ? b.col_vals[col][r] : 0;
        nm |= (uint32_t)(!valid) << i;
    }
    *nullmask = nm;
    return !(nm & ~(uint32_t)m.null_safe_mask);
}

// one counter per 64-B line: same-line atomics from different CUs
// serialize at the line's home (~12.6 ns each, round-1 measurement)
#define JPART_PAD 16

__global__ __launch_bounds__(1024) void jpart_count_kernel(
    JoinBatchDev b, JoinMeta m, int S, uint32_t* ptot) {
    // LDS histogram per block at 1024 threads (4 waves/SIMD — the
    // 256-thread variant ran 1 wave/SIMD and was latency-bound), then one
    // global add per nonzero partition per block (~450K adds total).
    __shared__ uint32_t hist[JPART_P];
    for (uint32_t i = threadIdx.x; i < JPART_P; i += blockDim.x) hist[i] = 0;
    __syncthreads();
    uint32_t n = b.n_rows;
    uint32_t stride = gridDim.x * blockDim.x;
    int64_t kw[MAX_KW];
    uint32_t nm;
    for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < n;
         r += stride) {
        if (!jpart_row_key(b, m, S, r, kw, &nm)) continue;
        atomicAdd(&hist[jpart_of(hash_key(kw, nm, m.KW))], 1u);
    }
    __syncthreads();
    for (uint32_t p = threadIdx.x; p < JPART_P; p += blockDim.x)
        if (hist[p]) atomicAdd(&ptot[(size_t)p * JPART_PAD], hist[p]);
}

// single-workgroup scan: per-partition totals → exclusive bases; rewrites
// pcount[blk][p] into per-block scatter bases; reserves the batch's row
// range on the device cursor (err code 3 = row store full, as the
// wave-aggregated reservation in jown_insert)
__global__ void jpart_scan_kernel(uint32_t* ptot, uint32_t* pcur,
                                  uint32_t* part_base, uint32_t* row_cursor,
                                  uint32_t row_cap, uint32_t* row_base,
                                  uint32_t* err) {
    __shared__ uint32_t tot[JPART_P];
    __shared__ uint32_t excl[JPART_P];
    for (uint32_t p = threadIdx.x; p < JPART_P; p += blockDim.x) {
        uint32_t s = ptot[(size_t)p * JPART_PAD];
        ptot[(size_t)p * JPART_PAD] = 0; // ready for the next batch
        tot[p] = s;
        excl[p] = s;
    }
    __syncthreads();
    // Hillis-Steele inclusive scan over excl[]
    for (uint32_t off = 1; off < JPART_P; off <<= 1) {
        uint32_t v[JPART_P / 1024];
        for (uint32_t i = threadIdx.x, k = 0; i < JPART_P;
             i += blockDim.x, k++)
            v[k] = i >= off ? excl[i - off] : 0;
        __syncthreads();
        for (uint32_t i = threadIdx.x, k = 0; i < JPART_P;
             i += blockDim.x, k++)
            excl[i] += v[k];
        __syncthreads();
    }
    uint32_t total = excl[JPART_P - 1];
    if (threadIdx.x == 0) {
        uint32_t base = atomicAdd(row_cursor, total);
        if ((uint64_t)base + total > row_cap) atomicExch(err, 3u);
        *row_base = base;
        part_base[JPART_P] = total;
    }
    __syncthreads();
    for (uint32_t p = threadIdx.x; p < JPART_P; p += blockDim.x) {
        uint32_t base = excl[p] - tot[p]; // exclusive base of partition p
        part_base[p] = base;
        pcur[(size_t)p * JPART_PAD] = base; // scatter cursors
    }
}

__global__ void jpart_scatter_kernel(JoinBatchDev b, JoinMeta m, int S,
                                     JoinSideDev own, uint32_t* pcur,
                                     const uint32_t* row_base,
                                     const uint32_t* err) {
    if (*err) return;
    uint32_t rb = *row_base;
    uint32_t n = b.n_rows;
    uint32_t stride = gridDim.x * blockDim.x;
    int64_t kw[MAX_KW];
    uint32_t nm;
    for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < n;
         r += stride) {
        if (!jpart_row_key(b, m, S, r, kw, &nm)) continue;
        uint32_t p = jpart_of(hash_key(kw, nm, m.KW));
        uint32_t row = rb + atomicAdd(&pcur[(size_t)p * JPART_PAD], 1u);
        // final row record, built in registers and stored with 16-B vector
        // writes (plain cached stores: the next kernel on the stream
        // observes them through the inter-dispatch cache flush)
        JoinRowHdr* hd = jrow(own, row);
        long long vals[MAX_COLS];
        uint32_t vb = 0;
        int nc = m.n_cols[S];
        for (int c = 0; c < nc; c++) {
            vals[c] = b.col_vals[c][r];
            vb |= (uint32_t)(b.col_valid[c][r] != 0) << c;
        }
        ulonglong2* dst = (ulonglong2*)hd;
        ulonglong2 h0;
        h0.x = ((uint64_t)UINT32_MAX << 32) | 1u;        // alive=1, next=~0
        h0.y = (uint64_t)vb;                              // validbits, degree=0
        dst[0] = h0;
        int c = 0;
        for (; c + 1 < nc; c += 2) {
            ulonglong2 v;
            v.x = (uint64_t)vals[c];
            v.y = (uint64_t)vals[c + 1];
            dst[1 + c / 2] = v;
        }
        if (c < nc) jvals(hd)[c] = vals[c];
    }
}

// inner-join condition over a probe ROW RECORD (vs join_cond_ok's batch row)
__device__ __forceinline__ bool jpart_cond_ok(const JoinMeta& m, int S,
                                              const long long* pv,
                                              uint32_t pvb, uint32_t mvb,
                                              const long long* mv) {
    for (int ci = 0; ci < m.n_cond; ci++) {
        auto& cd = m.cond[ci];
        auto fetch = [&](uint8_t src, uint8_t col, int64_t* v) -> bool {
            if ((int)src == S) {
                if (!((pvb >> col) & 1)) return false;
                *v = pv[col];
            } else {
                if (!((mvb >> col) & 1)) return false;
                *v = mv[col];
            }
            return true;
        };
        int64_t a, c;
        if (!fetch(cd.src_l, cd.col_l, &a)) return false;
        if (!fetch(cd.src_r, cd.col_r, &c)) return false;
        c += cd.rconst;
        bool ok;
        switch (cd.op) {
            case RW_CMP_LT: ok = a < c; break;
            case RW_CMP_LE: ok = a <= c; break;
            case RW_CMP_GT: ok = a > c; break;
            case RW_CMP_GE: ok = a >= c; break;
            default: ok = false;
        }
        if (!ok) return false;
    }
    return true;
}

__global__ __launch_bounds__(256, 8) void jpart_probe_insert_kernel(
    JoinSideDev own, JoinSideDev match, JoinMeta m, int S, JoinOutDev out,
    const uint32_t* part_base, const uint32_t* row_base,
    const uint32_t* err) {
    if (*err) return;
    uint32_t rb = *row_base;
    uint32_t lo = part_base[blockIdx.x];
    uint32_t hi = part_base[blockIdx.x + 1]; // [JPART_P] holds the total
    int lane = threadIdx.x & 63;
    uint32_t n = hi - lo;
    uint32_t iters = (n + blockDim.x - 1) / blockDim.x;
    for (uint32_t it = 0; it < iters; it++) {
        uint32_t i = it * blockDim.x + threadIdx.x;
        bool active = i < n;
        uint32_t row = rb + lo + i;
        int64_t kw[MAX_KW];
        uint32_t nm = 0;
        JoinRowHdr* hd = nullptr;
        long long* hv = nullptr;
        uint32_t pvb = 0;
        if (active) {
            hd = jrow(own, row);
            hv = jvals(hd);
            pvb = hd->validbits;
            for (int k = 0; k < m.KW; k++) {
                uint8_t col = m.key_cols[S][k];
                bool valid = (pvb >> col) & 1;
                kw[k] = valid ? hv[col] : 0;
                nm |= (uint32_t)(!valid) << k;
            }
        }
        // probe the match side (immutable during this launch: the batch
        // mutates only `own`)
        uint32_t mhead = UINT32_MAX;
        uint32_t my_n = 0;
        if (active) {
            mhead = jbucket_head(match, hash_key(kw, nm, m.KW), false);
            uint32_t mr = mhead;
            while (mr != UINT32_MAX) {
                JoinRowHdr* mh = jrow(match, mr);
                if (mh->alive &&
                    jhdr_key_eq(mh, m.key_cols[1 - S], m.KW, kw, nm, false) &&
                    jpart_cond_ok(m, S, hv, pvb, mh->validbits, jvals(mh)))
                    my_n++;
                mr = mh->next;
            }
        }
        // wave-aggregated output reservation + emit (multiset parity; the
        // reference's intra-epoch order is nondeterministic)
        // ballot fast path for my_n<=1 (see join_probe_kernel)
        uint64_t multi = __ballot(my_n > 1);
        uint32_t total, my_base;
        uint32_t base = 0;
        if (!multi) {
            uint64_t got = __ballot(my_n != 0);
            total = (uint32_t)__popcll(got);
            if (lane == 0 && total) base = atomicAdd(&out.counters[0], total);
            base = (uint32_t)__shfl((int)base, 0);
            my_base = base + (uint32_t)__popcll(got & ((1ULL << lane) - 1));
        } else {
            uint32_t incl = my_n;
            for (int d = 1; d < 64; d <<= 1) {
                uint32_t o = __shfl_up(incl, d);
                if (lane >= d) incl += o;
            }
            total = (uint32_t)__shfl((int)incl, 63);
            if (lane == 0 && total) base = atomicAdd(&out.counters[0], total);
            base = (uint32_t)__shfl((int)base, 0);
            my_base = base + incl - my_n;
        }
        if (total && base + total > out.cap) {
            if (lane == 0) atomicExch(&out.counters[1], 1u);
        } else if (my_n) {
            uint32_t mr = mhead;
            uint32_t k = 0;
            while (mr != UINT32_MAX && k < my_n) {
                JoinRowHdr* mh = jrow(match, mr);
                if (mh->alive &&
                    jhdr_key_eq(mh, m.key_cols[1 - S], m.KW, kw, nm, false) &&
                    jpart_cond_ok(m, S, hv, pvb, mh->validbits, jvals(mh))) {
                    uint32_t orow = my_base + k;
                    out.ops[orow] = RW_OP_INSERT;
                    const long long* mv = jvals(mh);
                    for (int c = 0; c < m.n_out; c++) {
                        bool from_probe = (int)m.out_src[c] == S;
                        uint8_t col = m.out_col[c];
                        int64_t v;
                        uint8_t valid;
                        if (from_probe) {
                            valid = (pvb >> col) & 1;
                            v = hv[col];
                        } else {
                            valid = (mh->validbits >> col) & 1;
                            v = mv[col];
                        }
                        out.vals[(size_t)c * out.cap + orow] = valid ? v : 0;
                        out.nulls[(size_t)c * out.cap + orow] = !valid;
                    }
                    k++;
                }
                mr = mh->next;
            }
        }
        // own-side insert: the record is already in place; link the chain.
        // Keys are pairwise distinct in the batch, so each touched slot has
        // exactly one linking thread — plain read-modify-write.
        if (active) {
            // record already in place from the scatter kernel (globally
            // visible across the dispatch boundary); one CAS links it
            jbucket_insert(own, hash_key(kw, nm, m.KW), row, false);
        }
    }
}

// LDS-window variant of the fused probe+insert phase: with chained
// buckets and top-bit slot indexing, partition p's OWN slot window
// [p*cap/P, (p+1)*cap/P) is touched by NO other block during the launch —
// so the block stages the whole window in LDS, performs the bucket pushes
// as LDS CASes, and writes the window back sequentially. This converts the
// insert path's random global CAS traffic (the dominant q8 cost: 290 us of
// the 540 us monolith step by RW_JOIN_SKIP A/B) into two sequential window
// copies + LDS atomics. Dynamic LDS = (cap/P)*8 bytes.
__global__ __launch_bounds__(256) void jpart_probe_insert_lds_kernel(
    JoinSideDev own, JoinSideDev match, JoinMeta m, int S, JoinOutDev out,
    const uint32_t* part_base, const uint32_t* row_base,
    const uint32_t* err) {
    if (*err) return;
    // dynamic LDS: [0, wslots) = the block's own slot window;
    // [wslots, ...) = a per-iteration staging area for blockDim.x records
    // (a lane-per-record read of 64-B records issues 64 line transactions
    // per wave per field — staging streams them COALESCED instead)
    extern __shared__ uint64_t win[];
    uint32_t rb = *row_base;
    uint32_t lo = part_base[blockIdx.x];
    uint32_t hi = part_base[blockIdx.x + 1]; // [JPART_P] holds the total
    uint32_t n = hi - lo;
    if (n == 0) return; // window untouched
    uint32_t wslots = (own.cap_mask + 1) >> JPART_LOG2;
    uint32_t wbase = blockIdx.x * wslots;
    uint32_t rwords = own.row_stride >> 3;
    uint64_t* stage = win + wslots;
    for (uint32_t i = threadIdx.x; i < wslots; i += blockDim.x)
        win[i] = own.slots8[wbase + i];
    int lane = threadIdx.x & 63;
    uint32_t iters = (n + blockDim.x - 1) / blockDim.x;
    for (uint32_t it = 0; it < iters; it++) {
        uint32_t i = it * blockDim.x + threadIdx.x;
        bool active = i < n;
        uint32_t row = rb + lo + i;
        // cooperative coalesced staging of this iteration's records
        {
            uint32_t first = lo + it * blockDim.x;
            uint32_t count = n - it * blockDim.x;
            if (count > blockDim.x) count = blockDim.x;
            const uint64_t* src =
                (const uint64_t*)(own.rows +
                                  (size_t)(rb + first) * own.row_stride);
            uint32_t words = count * rwords;
            __syncthreads();
            for (uint32_t w = threadIdx.x; w < words; w += blockDim.x)
                stage[w] = src[w];
            __syncthreads();
        }
        int64_t kw[MAX_KW];
        uint32_t nm = 0;
        JoinRowHdr* hd = nullptr;   // GLOBAL record (next-link target)
        long long* hv = nullptr;    // LDS copy (reads)
        uint32_t pvb = 0;
        uint64_t h64 = 0;
        if (active) {
            hd = jrow(own, row);
            JoinRowHdr* lh = (JoinRowHdr*)(stage + (size_t)threadIdx.x * rwords);
            hv = (long long*)((uint8_t*)lh + 16);
            pvb = lh->validbits;
            for (int k = 0; k < m.KW; k++) {
                uint8_t col = m.key_cols[S][k];
                bool valid = (pvb >> col) & 1;
                kw[k] = valid ? hv[col] : 0;
                nm |= (uint32_t)(!valid) << k;
            }
            h64 = hash_key(kw, nm, m.KW);
        }
        // probe the match side (immutable during this launch)
        uint32_t mhead = UINT32_MAX;
        uint32_t my_n = 0;
        if (active) {
            mhead = jbucket_head(match, h64, false);
            uint32_t mr = mhead;
            while (mr != UINT32_MAX) {
                JoinRowHdr* mh = jrow(match, mr);
                if (mh->alive &&
                    jhdr_key_eq(mh, m.key_cols[1 - S], m.KW, kw, nm, false) &&
                    jpart_cond_ok(m, S, hv, pvb, mh->validbits, jvals(mh)))
                    my_n++;
                mr = mh->next;
            }
        }
        // ballot fast path for my_n<=1 (see join_probe_kernel)
        uint64_t multi = __ballot(my_n > 1);
        uint32_t total, my_base;
        uint32_t base = 0;
        if (!multi) {
            uint64_t got = __ballot(my_n != 0);
            total = (uint32_t)__popcll(got);
            if (lane == 0 && total) base = atomicAdd(&out.counters[0], total);
            base = (uint32_t)__shfl((int)base, 0);
            my_base = base + (uint32_t)__popcll(got & ((1ULL << lane) - 1));
        } else {
            uint32_t incl = my_n;
            for (int d = 1; d < 64; d <<= 1) {
                uint32_t o = __shfl_up(incl, d);
                if (lane >= d) incl += o;
            }
            total = (uint32_t)__shfl((int)incl, 63);
            if (lane == 0 && total) base = atomicAdd(&out.counters[0], total);
            base = (uint32_t)__shfl((int)base, 0);
            my_base = base + incl - my_n;
        }
        if (total && base + total > out.cap) {
            if (lane == 0) atomicExch(&out.counters[1], 1u);
        } else if (my_n) {
            uint32_t mr = mhead;
            uint32_t k = 0;
            while (mr != UINT32_MAX && k < my_n) {
                JoinRowHdr* mh = jrow(match, mr);
                if (mh->alive &&
                    jhdr_key_eq(mh, m.key_cols[1 - S], m.KW, kw, nm, false) &&
                    jpart_cond_ok(m, S, hv, pvb, mh->validbits, jvals(mh))) {
                    uint32_t orow = my_base + k;
                    out.ops[orow] = RW_OP_INSERT;
                    const long long* mv = jvals(mh);
                    for (int c = 0; c < m.n_out; c++) {
                        bool from_probe = (int)m.out_src[c] == S;
                        uint8_t col = m.out_col[c];
                        int64_t v;
                        uint8_t valid;
                        if (from_probe) {
                            valid = (pvb >> col) & 1;
                            v = hv[col];
                        } else {
                            valid = (mh->validbits >> col) & 1;
                            v = mv[col];
                        }
                        out.vals[(size_t)c * out.cap + orow] = valid ? v : 0;
                        out.nulls[(size_t)c * out.cap + orow] = !valid;
                    }
                    k++;
                }
                mr = mh->next;
            }
        }
        // own-side bucket push in LDS (visibility via the window writeback
        // + the inter-dispatch flush; no drains needed — nothing walks the
        // own side within an all-Insert launch)
        if (active) {
            uint32_t local = jslot_start(h64, own.cap_mask) - wbase;
            uint64_t old = win[local];
            for (;;) {
                hd->next = (uint32_t)old ? jhead_of(old) : UINT32_MAX;
                uint64_t want =
                    ((uint64_t)row << 32) | ((uint32_t)old | jbloom_bit(h64));
                uint64_t prev = atomicCAS((unsigned long long*)&win[local],
                                          old, want);
                if (prev == old) break;
                old = prev;
            }
        }
    }
    __syncthreads();
    for (uint32_t i = threadIdx.x; i < wslots; i += blockDim.x)
        own.slots8[wbase + i] = win[i];
}

// join state restore: append decoded state rows + bucket links + degrees
// (rw_stream.h contract). err: 3 = row store full.
__global__ void join_restore_kernel(JoinBatchDev b, JoinSideDev sd,
                                    JoinMeta m, int S,
                                    const uint32_t* degrees, uint32_t* err) {
    uint32_t stride = gridDim.x * blockDim.x;
    int lane = threadIdx.x & 63;
    uint32_t n = b.n_rows;
    uint32_t iters = (n + stride - 1) / stride;
    for (uint32_t it = 0; it < iters; it++) {
        uint32_t r = it * stride + blockIdx.x * blockDim.x + threadIdx.x;
        bool active = r < n;
        uint64_t wmask = __ballot(active);
        if (!active) continue;
        int leader = 63 - __clzll(wmask);
        uint32_t base = 0;
        if (lane == leader)
            base = atomicAdd(sd.row_cursor, (uint32_t)__popcll(wmask));
        base = (uint32_t)__shfl((int)base, leader);
        uint32_t row = base + (uint32_t)__popcll(wmask & ((1ULL << lane) - 1));
        if (row >= sd.row_cap) {
            atomicExch(err, 3u);
            continue;
        }
        JoinRowHdr* hd = jrow(sd, row);
        long long* hv = jvals(hd);
        uint32_t vb = 0;
        for (int c = 0; c < m.n_cols[S]; c++) {
            hv[c] = b.col_vals[c][r];
            vb |= (uint32_t)(b.col_valid[c][r] != 0) << c;
        }
        hd->validbits = vb;
        hd->degree = degrees ? degrees[r] : 0;
        hd->alive = 1;
        int64_t kw[MAX_KW];
        uint32_t nm = 0;
        for (int i = 0; i < m.KW; i++) {
            uint8_t col = m.key_cols[S][i];
            bool valid = (vb >> col) & 1;
            kw[i] = valid ? hv[col] : 0;
            nm |= (uint32_t)(!valid) << i;
        }
        jbucket_insert(sd, hash_key(kw, nm, m.KW), row, false);
    }
}

// Batched record gather for the checkpoint drains: one kernel + one D2H
// instead of a hipMemcpy per referenced row (~10 us each dominated drain
// time once kill/degree lists grew past a few hundred rows).
__global__ void jgather_rows_kernel(JoinSideDev sd, const uint32_t* ids,
                                    uint32_t n, uint8_t* out) {
    uint32_t stride = gridDim.x * blockDim.x;
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride) {
        const uint8_t* src = sd.rows + (size_t)ids[i] * sd.row_stride;
        uint8_t* dst = out + (size_t)i * sd.row_stride;
        for (uint32_t w = 0; w < sd.row_stride; w += 8)
            *(long long*)(dst + w) = *(const long long*)(src + w);
    }
}

// §8f-4 memory reclamation: rebuild one join side's record store without
// its dead records (retractions and watermark sweeps retire rows in
// place; the reference's LSM compaction reclaims them at the store —
// here a periodic maintenance pass does, between epochs). Pass 1 packs
// alive records into a fresh store (next rebuilt later); pass 2 relinks
// the cleared buckets. Runs only on a DRAINED table (kill list and
// degree deltas consumed, flush mark at the cursor), so row ids are not
// referenced by any pending delta tracking.
__global__ void jcompact_copy_kernel(JoinSideDev oldsd, uint8_t* nrows,
                                     uint32_t* ncursor, uint32_t old_n) {
    uint32_t stride = gridDim.x * blockDim.x;
    int lane = threadIdx.x & 63;
    uint32_t iters = (old_n + stride - 1) / stride;
    for (uint32_t it = 0; it < iters; it++) {
        uint32_t r = it * stride + blockIdx.x * blockDim.x + threadIdx.x;
        bool alive = r < old_n && jrow(oldsd, r)->alive;
        uint64_t wmask = __ballot(alive);
        if (!alive) continue;
        int leader = 63 - __clzll(wmask);
        uint32_t base = 0;
        if (lane == leader)
            base = atomicAdd(ncursor, (uint32_t)__popcll(wmask));
        base = (uint32_t)__shfl((int)base, leader);
        uint32_t nr = base + (uint32_t)__popcll(wmask & ((1ULL << lane) - 1));
        const uint8_t* src = oldsd.rows + (size_t)r * oldsd.row_stride;
        uint8_t* dst = nrows + (size_t)nr * oldsd.row_stride;
        for (uint32_t wq = 0; wq < oldsd.row_stride; wq += 8)
            *(long long*)(dst + wq) = *(const long long*)(src + wq);
        ((JoinRowHdr*)dst)->next = UINT32_MAX;
    }
}

__global__ void jcompact_link_kernel(JoinSideDev sd, JoinMeta m, int S,
                                     uint32_t n) {
    uint32_t stride = gridDim.x * blockDim.x;
    for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < n;
         r += stride) {
        JoinRowHdr* h = jrow(sd, r);
        int64_t kw[MAX_KW];
        uint32_t nm = 0;
        for (int i = 0; i < m.KW; i++) {
            uint8_t col = m.key_cols[S][i];
            bool valid = (h->validbits >> col) & 1;
            kw[i] = valid ? jvals(h)[col] : 0;
            nm |= (uint32_t)(!valid) << i;
        }
        jbucket_insert(sd, hash_key(kw, nm, m.KW), r, false);
    }
}

// watermark TTL sweeps (state_table watermark cleaning):
// rows/groups whose watermarked key column sorts below the value are
// retired in place (slots stay READY so linear probing is undisturbed;
// memory reclamation is compaction work for a later round)
__global__ void join_clean_kernel(JoinSideDev sd, int kcol, long long wm,
                                  int KW) {
    // kcol = RECORD column of the watermarked key position. Buckets mix
    // keys, so the watermark predicate runs PER RECORD; rows below the
    // watermark are retired in place (slot entries stay — probes skip dead
    // records by the alive flag).
    (void)KW;
    size_t cap = (size_t)sd.cap_mask + 1;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (size_t slot = blockIdx.x * blockDim.x + threadIdx.x; slot < cap;
         slot += stride) {
        uint64_t packed = __hip_atomic_load(&sd.slots8[slot], RLX);
        if ((uint32_t)packed == 0) continue; // bloom 0 = empty
        uint32_t row = jhead_of(packed);
        while (row != UINT32_MAX) {
            JoinRowHdr* h = jrow(sd, row);
            if (ld_u32(&h->alive) &&
                ((ld_u32(&h->validbits) >> kcol) & 1) && // NULLs largest
                ld_i64((const int64_t*)&jvals(h)[kcol]) < wm) {
                st_u32(&h->alive, 0);
                // §8f-2 state-cleaning spill deltas (the reference's
                // commit applies the watermark as a range delete on the
                // store, state_table.rs:1707): route retired rows
                // through the kill list so the next drain nets them to
                // DELETE frames — a restore replay must not resurrect
                // them
                if (sd.killed) {
                    uint32_t kidx = atomicAdd(sd.killed_cursor, 1u);
                    if (kidx < sd.killed_cap) sd.killed[kidx] = row;
                }
            }
            row = ld_u32(&h->next);
        }
    }
}

// crecs/crec_n/crec_cap/cover: capture the cleaned PERSISTED groups'
// keys so the host can append DELETE spill frames — the reference's
// commit applies the watermark as a range delete on the store
// (state_table.rs:1707), so the spill stream must net cleaned rows out
// or a restore replay resurrects them. On capture overflow the slot is
// LEFT UNCLEANED (cover set); the host grows the buffer and reruns —
// already-cleaned slots skip via the has_prev they cleared.
__global__ void agg_clean_kernel(AggTableDev t, int kpos, long long wm, int KW,
                                 int n_calls, AggCallDev c0, AggCallDev c1,
                                 AggCallDev c2, AggCallDev c3,
                                 DedupDirtyRec* crecs, uint32_t* crec_n,
                                 uint32_t crec_cap, uint32_t* cover) {
    AggCallDev calls[4] = {c0, c1, c2, c3};
    size_t cap = (size_t)t.cap_mask + 1;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (size_t slot = blockIdx.x * blockDim.x + threadIdx.x; slot < cap;
         slot += stride) {
        if (ld_u32(&t.state[(uint32_t)slot]) != SLOT_READY) continue;
        if ((ld_u32(&t.key_nulls[(uint32_t)slot]) >> kpos) & 1) continue;
        if (ld_i64((const int64_t*)&t.keys[slot * KW + kpos]) >= wm) continue;
        if (t.has_prev[(uint32_t)slot]) {
            uint32_t ci2 = atomicAdd(crec_n, 1u);
            if (ci2 >= crec_cap) {
                atomicExch(cover, 1u);
                continue; // retried after the host grows the buffer
            }
            for (int w = 0; w < KW; w++)
                crecs[ci2].key[w] = t.keys[slot * KW + w];
            crecs[ci2].nulls = t.key_nulls[(uint32_t)slot];
            crecs[ci2].count = 0;
        }
        // reset the group as if freshly created (late rows restart it)
        for (int ci = 0; ci < n_calls; ci++) {
            long long init = 0;
            if (calls[ci].kind == RW_AGG_MIN) init = INT64_MAX;
            if (calls[ci].kind == RW_AGG_MAX) init = INT64_MIN;
            t.acc[(size_t)ci * cap + slot] = init;
            t.has[(size_t)ci * cap + slot] = 0;
            if (calls[ci].minput) {
                uint32_t row = t.mheads[(size_t)calls[ci].mord * cap + slot];
                while (row != UINT32_MAX) {
                    st_u32(&t.malive[row], 0);
                    // minput state-table deltas for the cleaned rows
                    // (netted to DELETE by minput_drain, as apply-path
                    // retractions are)
                    if (t.mkilled) {
                        uint32_t ki = atomicAdd(t.mkilled_cursor, 1u);
                        if (ki < t.mkilled_cap) t.mkilled[ki] = row;
                    }
                    row = ld_u32(&t.mnext[row]);
                }
                t.mheads[(size_t)calls[ci].mord * cap + slot] = UINT32_MAX;
            }
        }
        t.has_prev[(uint32_t)slot] = 0;
        t.dirty_flag[(uint32_t)slot] = 0;
    }
}

// DISTINCT dedup-table state cleaning (the same watermark applies to the
// dedup tables' group-key prefix in the reference): reset counts below
// the watermark and mark the slot dirty so the next dedup drain nets a
// DELETE for persisted rows — and so LATE rows restart visibility
// transitions from 0, as the reference's freshly-deleted state row would.
__global__ void dedup_clean_kernel(JoinSlot* slots, size_t cap, int kpos,
                                   long long wm, uint32_t* dirty_flag,
                                   uint32_t* dirty_list, uint32_t* dirty_n) {
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (size_t slot = blockIdx.x * blockDim.x + threadIdx.x; slot < cap;
         slot += stride) {
        if (slots[slot].state != SLOT_READY) continue;
        if ((slots[slot].nulls >> kpos) & 1) continue;
        if (slots[slot].key[kpos] >= wm) continue;
        if (slots[slot].head == 0) continue; // nothing live to clean
        slots[slot].head = 0;
        if (ld_u32(&dirty_flag[slot]) == 0 &&
            atomicCAS(&dirty_flag[slot], 0u, 1u) == 0u)
            dirty_list[atomicAdd(dirty_n, 1u)] = (uint32_t)slot;
    }
}

// EOWC close (hash_agg.rs:429-474): windows with group-key[0] below the
// watermark emit their FINAL row once (Insert; row_count 0 emits nothing)
// and are reset in place (slot stays READY so linear probing is
// undisturbed; a late row would restart the window). Emission order is
// fixed up host-side (rows sorted by group key, as SortBuffer::consume
// iterates ordered).
// EOWC mid-window state PUTs (hash_agg.rs:429-460): at every barrier the
// dirty groups' CURRENT states upsert into the intermediate table even
// though nothing is emitted until a watermark closes the window. One
// thread per dirty slot; rows carry op=Insert (PUT) and are encoded
// host-side by spill_records (materialized-input states encode None
// there, as in the non-EOWC spill).
__global__ void agg_eowc_dump_kernel(AggTableDev t, int KW, int n_calls,
                                     int row_count_index, AggCallDev c0,
                                     AggCallDev c1, AggCallDev c2,
                                     AggCallDev c3) {
    AggCallDev calls[4] = {c0, c1, c2, c3};
    uint32_t n_dirty = t.counters[0];
    uint32_t stride = gridDim.x * blockDim.x;
    size_t cap = (size_t)t.cap_mask + 1;
    int width = KW + n_calls;
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n_dirty;
         i += stride) {
        uint32_t slot = t.dirty_list[i];
        t.dirty_flag[slot] = 0;
        uint32_t orow = atomicAdd(&t.counters[1], 1u);
        if (orow + 1 > t.out_capacity) {
            atomicExch(&t.counters[2], 2u);
            continue;
        }
        long long rc = t.acc[(size_t)row_count_index * cap + slot];
        if (rc < 0) rc = 0;
        t.out_ops[orow] = RW_OP_INSERT;
        // the window now has a persisted intermediate-table row; the close
        // kernel emits its DELETE marker only for persisted windows (and
        // clears the flag), so re-scanned already-closed slots stay silent
        t.has_prev[slot] = 1;
        for (int k = 0; k < KW; k++) {
            t.out_vals[(size_t)orow * width + k] = t.keys[(size_t)slot * KW + k];
            t.out_nulls[(size_t)orow * width + k] =
                (t.key_nulls[slot] >> k) & 1;
        }
        for (int ci = 0; ci < n_calls; ci++) {
            const AggCallDev& c = calls[ci];
            long long* acc = t.acc + (size_t)ci * cap;
            uint8_t* has = t.has + (size_t)ci * cap;
            if (rc == 0 && !c.minput) {
                // reset value states (agg_state.rs:149-155), as the flush
                // kernel and the oracle's get_outputs do
                switch (c.kind) {
                    case RW_AGG_MIN: acc[slot] = INT64_MAX; break;
                    case RW_AGG_MAX: acc[slot] = INT64_MIN; break;
                    default: acc[slot] = 0;
                }
                has[slot] = 0;
            }
            long long v;
            uint8_t nl;
            if (c.minput) {
                v = 0;
                nl = 1; // spill_records encodes None for minput states
            } else {
                switch (c.kind) {
                    case RW_AGG_COUNT_STAR:
                    case RW_AGG_COUNT:
                    case RW_AGG_SUM0:
                        v = acc[slot];
                        nl = 0;
                        break;
                    default:
                        v = acc[slot];
                        nl = !has[slot];
                }
            }
            t.out_vals[(size_t)orow * width + KW + ci] = v;
            t.out_nulls[(size_t)orow * width + KW + ci] = nl;
        }
    }
}

__global__ void agg_eowc_close_kernel(AggTableDev t, int KW, int n_calls,
                                      int row_count_index, long long wm,
                                      AggCallDev c0, AggCallDev c1,
                                      AggCallDev c2, AggCallDev c3) {
    AggCallDev calls[4] = {c0, c1, c2, c3};
    size_t cap = (size_t)t.cap_mask + 1;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    int width = KW + n_calls;
    for (size_t slot = blockIdx.x * blockDim.x + threadIdx.x; slot < cap;
         slot += stride) {
        if (ld_u32(&t.state[(uint32_t)slot]) != SLOT_READY) continue;
        if ((ld_u32(&t.key_nulls[(uint32_t)slot]) >> 0) & 1) continue;
        if (ld_i64((const int64_t*)&t.keys[slot * KW]) >= wm) continue;
        long long rc = t.acc[(size_t)row_count_index * cap + slot];
        if (rc < 0) rc = 0;
        if (rc > 0) {
            // final outputs (get_outputs, agg_group.rs:431-467)
            long long curr[MAX_CALLS];
            uint8_t curr_null[MAX_CALLS];
            for (int ci = 0; ci < n_calls; ci++) {
                const AggCallDev& c = calls[ci];
                long long* acc = t.acc + (size_t)ci * cap;
                uint8_t* has = t.has + (size_t)ci * cap;
                if (c.minput) {
                    uint32_t row = t.mheads[(size_t)c.mord * cap + slot];
                    bool any = false, any_null = false;
                    long long best = 0;
                    while (row != UINT32_MAX) {
                        if (t.malive[row]) {
                            if (t.mval_null[row]) {
                                any_null = true;
                            } else {
                                long long v = t.mval[row];
                                if (!any) best = v;
                                else if (c.kind == RW_AGG_MIN)
                                    best = v < best ? v : best;
                                else
                                    best = v > best ? v : best;
                                any = true;
                            }
                        }
                        row = t.mnext[row];
                    }
                    if (c.kind == RW_AGG_MAX && any_null) {
                        curr[ci] = 0;
                        curr_null[ci] = 1;
                    } else {
                        curr[ci] = best;
                        curr_null[ci] = !any;
                    }
                    continue;
                }
                switch (c.kind) {
                    case RW_AGG_COUNT_STAR:
                    case RW_AGG_COUNT:
                    case RW_AGG_SUM0:
                        curr[ci] = acc[slot];
                        curr_null[ci] = 0;
                        break;
                    default:
                        curr[ci] = acc[slot];
                        curr_null[ci] = !has[slot];
                }
            }
            uint32_t orow = atomicAdd(&t.counters[1], 1u);
            if (orow + 1 > t.out_capacity) {
                atomicExch(&t.counters[2], 2u);
            } else {
                t.out_ops[orow] = RW_OP_INSERT;
                for (int k = 0; k < KW; k++) {
                    t.out_vals[(size_t)orow * width + k] =
                        t.keys[slot * KW + k];
                    t.out_nulls[(size_t)orow * width + k] =
                        (t.key_nulls[(uint32_t)slot] >> k) & 1;
                }
                for (int ci = 0; ci < n_calls; ci++) {
                    t.out_vals[(size_t)orow * width + KW + ci] = curr[ci];
                    t.out_nulls[(size_t)orow * width + KW + ci] = curr_null[ci];
                }
            }
        } else if (t.has_prev[(uint32_t)slot]) {
            // closed window with row_count 0 but a persisted row: nothing
            // is emitted (OnlyOutputIfHasInput) but the reference still
            // deletes its intermediate-table row — a DELETE-marker row,
            // spilled but filtered out of emission host-side. Slots with
            // no persisted row (already-closed residue) stay silent.
            uint32_t orow = atomicAdd(&t.counters[1], 1u);
            if (orow + 1 > t.out_capacity) {
                atomicExch(&t.counters[2], 2u);
            } else {
                t.out_ops[orow] = RW_OP_DELETE;
                for (int k = 0; k < KW; k++) {
                    t.out_vals[(size_t)orow * width + k] =
                        t.keys[slot * KW + k];
                    t.out_nulls[(size_t)orow * width + k] =
                        (t.key_nulls[(uint32_t)slot] >> k) & 1;
                }
                for (int ci = 0; ci < n_calls; ci++) {
                    t.out_vals[(size_t)orow * width + KW + ci] = 0;
                    t.out_nulls[(size_t)orow * width + KW + ci] = 1;
                }
            }
        }
        // reset the window in place (as agg_clean_kernel does)
        for (int ci = 0; ci < n_calls; ci++) {
            long long init = 0;
            if (calls[ci].kind == RW_AGG_MIN) init = INT64_MAX;
            if (calls[ci].kind == RW_AGG_MAX) init = INT64_MIN;
            t.acc[(size_t)ci * cap + slot] = init;
            t.has[(size_t)ci * cap + slot] = 0;
            if (calls[ci].minput) {
                uint32_t row = t.mheads[(size_t)calls[ci].mord * cap + slot];
                while (row != UINT32_MAX) {
                    st_u32(&t.malive[row], 0);
                    row = ld_u32(&t.mnext[row]);
                }
                t.mheads[(size_t)calls[ci].mord * cap + slot] = UINT32_MAX;
            }
        }
        t.has_prev[(uint32_t)slot] = 0;
        t.dirty_flag[(uint32_t)slot] = 0;
    }
}

// Rescale re-scoping (update_vnode_bitmap, state_table.rs + executor cache
// eviction): state whose distribution-key vnode is no longer owned is
// dropped WITHOUT emitting retractions — the new owner holds it. Dist key =
// group key (agg) / join key (join). CRC feed identical to vnode_kernel.
__device__ __forceinline__ uint32_t crc_key_words(const uint32_t* lut,
                                                  const int64_t* kw,
                                                  uint32_t nulls, int KW,
                                                  const uint8_t* types) {
    uint32_t crc = 0xFFFFFFFFu;
    for (int k = 0; k < KW; k++) {
        if ((nulls >> k) & 1) {
            uint32_t sentinel = 0xfffffff0u;
            crc = crc32_bytes(lut, crc, (const uint8_t*)&sentinel, 4);
        } else if (types[k] == RW_T_I32) {
            int32_t v = (int32_t)kw[k];
            crc = crc32_bytes(lut, crc, (const uint8_t*)&v, 4);
        } else if (types[k] == RW_T_BOOL) {
            uint8_t v = (uint8_t)kw[k];
            crc = crc32_bytes(lut, crc, &v, 1);
        } else {
            int64_t v = kw[k];
            crc = crc32_bytes(lut, crc, (const uint8_t*)&v, 8);
        }
    }
    return crc ^ 0xFFFFFFFFu;
}

__global__ void agg_vnode_scope_kernel(AggTableDev t, int KW, int n_calls,
                                       AggCallDev c0, AggCallDev c1,
                                       AggCallDev c2, AggCallDev c3,
                                       const uint8_t* bitmap,
                                       uint32_t vnode_count, uint8_t t0,
                                       uint8_t t1, uint8_t t2, uint8_t t3) {
    __shared__ uint32_t lut[256];
    stage_crc_lut(lut);
    AggCallDev calls[4] = {c0, c1, c2, c3};
    uint8_t types[4] = {t0, t1, t2, t3};
    size_t cap = (size_t)t.cap_mask + 1;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (size_t slot = blockIdx.x * blockDim.x + threadIdx.x; slot < cap;
         slot += stride) {
        if (ld_u32(&t.state[(uint32_t)slot]) != SLOT_READY) continue;
        uint32_t vn = crc_key_words(lut, &t.keys[slot * KW],
                                    ld_u32(&t.key_nulls[(uint32_t)slot]), KW,
                                    types) %
                      vnode_count;
        if ((bitmap[vn >> 3] >> (vn & 7)) & 1) continue; // still owned
        for (int ci = 0; ci < n_calls; ci++) {
            long long init = 0;
            if (calls[ci].kind == RW_AGG_MIN) init = INT64_MAX;
            if (calls[ci].kind == RW_AGG_MAX) init = INT64_MIN;
            t.acc[(size_t)ci * cap + slot] = init;
            t.has[(size_t)ci * cap + slot] = 0;
            if (calls[ci].minput) {
                uint32_t row = t.mheads[(size_t)calls[ci].mord * cap + slot];
                while (row != UINT32_MAX) {
                    st_u32(&t.malive[row], 0);
                    row = ld_u32(&t.mnext[row]);
                }
                t.mheads[(size_t)calls[ci].mord * cap + slot] = UINT32_MAX;
            }
        }
        t.has_prev[(uint32_t)slot] = 0;
        t.dirty_flag[(uint32_t)slot] = 0;
    }
}

__global__ void join_vnode_scope_kernel(JoinSideDev sd, int KW,
                                        const uint8_t* bitmap,
                                        uint32_t vnode_count, uint8_t t0,
                                        uint8_t t1, uint8_t t2, uint8_t t3,
                                        uint8_t k0, uint8_t k1, uint8_t k2,
                                        uint8_t k3) {
    __shared__ uint32_t lut[256];
    stage_crc_lut(lut);
    uint8_t types[4] = {t0, t1, t2, t3};
    uint8_t kcols[4] = {k0, k1, k2, k3};
    size_t cap = (size_t)sd.cap_mask + 1;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (size_t slot = blockIdx.x * blockDim.x + threadIdx.x; slot < cap;
         slot += stride) {
        uint64_t packed = __hip_atomic_load(&sd.slots8[slot], RLX);
        if ((uint32_t)packed == 0) continue; // bloom 0 = empty
        uint32_t row = jhead_of(packed);
        while (row != UINT32_MAX) { // buckets mix keys: vnode PER RECORD
            JoinRowHdr* h = jrow(sd, row);
            if (ld_u32(&h->alive)) {
                uint32_t vb = ld_u32(&h->validbits);
                const long long* hv = jvals(h);
                int64_t kw[4];
                uint32_t nulls = 0;
                for (int k = 0; k < KW; k++) {
                    bool valid = (vb >> kcols[k]) & 1;
                    kw[k] = valid ? ld_i64((const int64_t*)&hv[kcols[k]]) : 0;
                    nulls |= (uint32_t)(!valid) << k;
                }
                uint32_t vn =
                    crc_key_words(lut, kw, nulls, KW, types) % vnode_count;
                if (!((bitmap[vn >> 3] >> (vn & 7)) & 1))
                    st_u32(&h->alive, 0);
            }
            row = ld_u32(&h->next);
        }
    }
}

struct HashJoin {
    RwHashJoinDesc desc;
    JoinMeta m{};
    JoinSideDev side[2]{};
    JoinOutDev out{};
    hipStream_t stream;
    std::vector<uint8_t> out_types;
    std::vector<uint8_t> types[2];
    // staging
    JoinBatchDev stage[2]{};
    uint32_t stage_cap[2] = {0, 0};
    std::vector<RwChunk*> outq;
    uint8_t* zeros = nullptr; // all-zero null flags for pipeline dummy cols
    uint8_t* d_vnode_bitmap = nullptr; // rescale scope (update_vnode_bitmap)
    uint16_t* d_vnode_hop = nullptr;   // q7-pipeline exchange-hop vnodes
    uint32_t d_vnode_hop_cap = 0;
    // join-key watermark buffering (hash_join.rs:826-867)
    struct Wm { bool has = false; int64_t val = 0; };
    std::vector<uint32_t> wm_pos;
    std::vector<uint8_t> wm_clean;
    std::vector<Wm> wm_side0, wm_side1, wm_out;
    // inequality-pair watermarks (hash_join.rs:869-914): buffered per
    // pair; the min across sides emits for the LARGER side's output
    // columns and (clean flag) sweeps that side's rows by the pair's
    // value column
    std::vector<uint32_t> ineq_col[2]; // [pair] per side
    std::vector<uint8_t> ineq_larger, ineq_do_clean;
    std::vector<Wm> ineq_wm[2], ineq_out;
    // lazy event ring (same rationale as HashAgg: no per-step host sync)
    static constexpr int EV_RING = 256;
    hipEvent_t ev0[EV_RING] = {}, ev1[EV_RING] = {};
    uint8_t ev_pending[EV_RING] = {};
    uint64_t ev_head = 0;
    double probe_ms_total = 0;
    uint64_t probe_launches = 0, probe_rows = 0;

    int ev_harvest(int slot) {
        if (!ev_pending[slot]) return RW_OK;
        HIP_TRY(hipEventSynchronize(ev1[slot]));
        float ms = 0;
        HIP_TRY(hipEventElapsedTime(&ms, ev0[slot], ev1[slot]));
        probe_ms_total += ms;
        ev_pending[slot] = 0;
        return RW_OK;
    }
    int ev_harvest_all() {
        for (int s = 0; s < EV_RING; s++)
            if (ev_harvest(s) != RW_OK) return RW_E_INTERNAL;
        return RW_OK;
    }

    int init(const RwHashJoinDesc* d) {
        if (!gpu_ok()) FAIL(RW_E_NOGPU, "risingwave_amd: no GPU visible (product path has no CPU fallback)");
        desc = *d;
        if (d->join_type > RW_JOIN_RIGHT_ANTI)
            FAIL(RW_E_INVAL, "unknown join type %d", d->join_type);
        if (d->append_only && d->join_type != RW_JOIN_INNER)
            FAIL(RW_E_INVAL, "append-only optimize kernel is inner-only this round");
        if (d->n_key < 1 || d->n_key > MAX_KW) FAIL(RW_E_INVAL, "n_key %u", d->n_key);
        if (d->n_cols_l > MAX_COLS || d->n_cols_r > MAX_COLS)
            FAIL(RW_E_INVAL, "too many columns for round-1 kernels");
        if (d->n_output > MAX_OUT) FAIL(RW_E_INVAL, "too many output columns");
        m.KW = (int)d->n_key;
        m.n_cols[0] = (int)d->n_cols_l;
        m.n_cols[1] = (int)d->n_cols_r;
        m.n_pk[0] = (int)d->n_pk_l;
        m.n_pk[1] = (int)d->n_pk_r;
        types[0].assign(d->types_l, d->types_l + d->n_cols_l);
        types[1].assign(d->types_r, d->types_r + d->n_cols_r);
        for (int s = 0; s < 2; s++)
            for (auto t : types[s])
                if (t != RW_T_I64 && t != RW_T_TS)
                    FAIL(RW_E_INVAL, "column type %d unsupported on GPU (i64/ts only)", t);
        for (uint32_t i = 0; i < d->n_key; i++) {
            m.key_cols[0][i] = (uint8_t)d->key_l[i];
            m.key_cols[1][i] = (uint8_t)d->key_r[i];
            if (d->null_safe[i]) m.null_safe_mask |= 1 << i;
        }
        if (d->n_pk_l > MAX_KW || d->n_pk_r > MAX_KW) FAIL(RW_E_INVAL, "pk too wide");
        for (uint32_t i = 0; i < d->n_pk_l; i++) m.pk_cols[0][i] = (uint8_t)d->pk_l[i];
        for (uint32_t i = 0; i < d->n_pk_r; i++) m.pk_cols[1][i] = (uint8_t)d->pk_r[i];
        m.n_out = (int)d->n_output;
        for (uint32_t i = 0; i < d->n_output; i++) {
            uint32_t idx = d->output_indices[i];
            if (idx < d->n_cols_l) {
                m.out_src[i] = 0;
                m.out_col[i] = (uint8_t)idx;
                out_types.push_back(types[0][idx]);
            } else {
                m.out_src[i] = 1;
                m.out_col[i] = (uint8_t)(idx - d->n_cols_l);
                out_types.push_back(types[1][idx - d->n_cols_l]);
            }
        }
        auto split = [&](uint32_t idx, uint8_t* src, uint8_t* col) {
            if (idx < d->n_cols_l) {
                *src = 0;
                *col = (uint8_t)idx;
            } else {
                *src = 1;
                *col = (uint8_t)(idx - d->n_cols_l);
            }
        };
        m.n_cond = 0;
        if (d->has_cond) {
            m.cond[0].op = d->cond_op;
            split(d->cond_l, &m.cond[0].src_l, &m.cond[0].col_l);
            split(d->cond_r, &m.cond[0].src_r, &m.cond[0].col_r);
            m.cond[0].rconst = d->cond_rconst;
            m.n_cond = 1;
            if (d->has_cond2) {
                m.cond[1].op = d->cond2_op;
                split(d->cond2_l, &m.cond[1].src_l, &m.cond[1].col_l);
                split(d->cond2_r, &m.cond[1].src_r, &m.cond[1].col_r);
                m.cond[1].rconst = d->cond2_rconst;
                m.n_cond = 2;
            }
        }
        m.append_only = d->append_only;
        m.join_type = d->join_type;
        m.need_deg[0] = j_need_degree(d->join_type, RW_SIDE_LEFT);
        m.need_deg[1] = j_need_degree(d->join_type, RW_SIDE_RIGHT);
        if (d->n_wm_jk) {
            wm_pos.assign(d->wm_jk_pos, d->wm_jk_pos + d->n_wm_jk);
            wm_clean.assign(d->wm_jk_clean, d->wm_jk_clean + d->n_wm_jk);
        }
        wm_side0.resize(d->n_key);
        wm_side1.resize(d->n_key);
        wm_out.resize(d->n_key);
        if (d->n_ineq) {
            ineq_col[0].assign(d->ineq_left_col,
                               d->ineq_left_col + d->n_ineq);
            ineq_col[1].assign(d->ineq_right_col,
                               d->ineq_right_col + d->n_ineq);
            ineq_larger.assign(d->ineq_left_larger,
                               d->ineq_left_larger + d->n_ineq);
            ineq_do_clean.assign(d->ineq_clean, d->ineq_clean + d->n_ineq);
            ineq_wm[0].resize(d->n_ineq);
            ineq_wm[1].resize(d->n_ineq);
            ineq_out.resize(d->n_ineq);
        }

        HIP_TRY(hipStreamCreate(&stream));
        uint64_t key_cap = d->state_capacity_hint ? d->state_capacity_hint : (1u << 20);
        uint64_t row_cap = d->row_capacity_hint ? d->row_capacity_hint : (1u << 22);
        for (int s = 0; s < 2; s++) {
            uint32_t cap = 1;
            while (cap < key_cap * 2) cap <<= 1; // ≤50% load factor
            JoinSideDev& js = side[s];
            js.cap_mask = cap - 1;
            js.slots = nullptr; // join tables use compact 8-B slots
            HIP_TRY(hipMalloc(&js.slots8, (size_t)cap * 8));
            HIP_TRY(hipMemsetAsync(js.slots8, 0, (size_t)cap * 8, stream));
            js.row_cap = (uint32_t)row_cap;
            uint32_t rs = 16 + 8 * (uint32_t)m.n_cols[s];
            // line-align records: a stride of 16/32/64 never straddles a
            // 64-B line, so the random record reads of probe/verify fetch
            // exactly ONE line (a 48-B stride fetched ~1.5 lines/record)
            if (rs > 32 && rs <= 64) rs = 64;
            else if (rs > 64) rs = (rs + 31) & ~31u;
            js.row_stride = rs;
            HIP_TRY(hipMalloc(&js.rows, (size_t)row_cap * js.row_stride));
            HIP_TRY(hipMalloc(&js.row_cursor, 4));
            HIP_TRY(hipMemset(js.row_cursor, 0, 4));
            // checkpoint-delta tracking (§8f-2)
            js.killed_cap = 1u << 22;
            HIP_TRY(hipMalloc(&js.killed, (size_t)js.killed_cap * 4));
            HIP_TRY(hipMalloc(&js.killed_cursor, 4));
            HIP_TRY(hipMemset(js.killed_cursor, 0, 4));
            js.deg_dirty_flag = nullptr;
            js.deg_dirty_list = nullptr;
            js.deg_dirty_n = nullptr;
            if (m.need_deg[s]) {
                HIP_TRY(hipMalloc(&js.deg_dirty_flag, (size_t)row_cap * 4));
                HIP_TRY(hipMemset(js.deg_dirty_flag, 0, (size_t)row_cap * 4));
                HIP_TRY(hipMalloc(&js.deg_dirty_list, (size_t)row_cap * 4));
                HIP_TRY(hipMalloc(&js.deg_dirty_n, 4));
                HIP_TRY(hipMemset(js.deg_dirty_n, 0, 4));
            }
        }
        HIP_TRY(hipStreamSynchronize(stream));
        out.cap = 1u << 22;
        HIP_TRY(hipMalloc(&out.vals, (size_t)out.cap * m.n_out * 8));
        HIP_TRY(hipMalloc(&out.nulls, (size_t)out.cap * m.n_out));
        HIP_TRY(hipMalloc(&out.ops, out.cap));
        HIP_TRY(hipMalloc(&out.counters, 8));
        HIP_TRY(hipMemset(out.counters, 0, 8));
        return RW_OK;
    }

    int ensure_stage(int s, uint32_t n) {
        if (stage_cap[s] >= n) return RW_OK;
        for (int c = 0; c < m.n_cols[s]; c++) {
            if (stage[s].col_vals[c]) hipFree(stage[s].col_vals[c]);
            if (stage[s].col_valid[c]) hipFree(stage[s].col_valid[c]);
        }
        if (stage[s].ops) hipFree(stage[s].ops);
        if (stage[s].vis) hipFree(stage[s].vis);
        uint32_t cap = 4096;
        while (cap < n) cap <<= 1;
        for (int c = 0; c < m.n_cols[s]; c++) {
            HIP_TRY(hipMalloc(&stage[s].col_vals[c], (size_t)cap * 8));
            HIP_TRY(hipMalloc(&stage[s].col_valid[c], cap));
        }
        HIP_TRY(hipMalloc(&stage[s].ops, cap));
        HIP_TRY(hipMalloc(&stage[s].vis, cap));
        stage_cap[s] = cap;
        return RW_OK;
    }

    int upload(int s, const RwChunk* c, JoinBatchDev* bout) {
        uint32_t n = c->n_rows;
        int rc = ensure_stage(s, n);
        if (rc != RW_OK) return rc;
        JoinBatchDev b = stage[s];
        for (int ci = 0; ci < m.n_cols[s]; ci++) {
            HIP_TRY(hipMemcpyAsync(b.col_vals[ci], c->cols[ci].data, (size_t)n * 8,
                                   hipMemcpyHostToDevice, stream));
            HIP_TRY(hipMemcpyAsync(b.col_valid[ci], c->cols[ci].valid, n,
                                   hipMemcpyHostToDevice, stream));
        }
        HIP_TRY(hipMemcpyAsync(b.ops, c->ops, n, hipMemcpyHostToDevice, stream));
        if (c->vis)
            HIP_TRY(hipMemcpyAsync(b.vis, c->vis, n, hipMemcpyHostToDevice, stream));
        else
            b.vis = nullptr;
        b.n_rows = n;
        b.all_insert = 1;
        for (uint32_t r = 0; r < n && b.all_insert; r++)
            if (!(c->vis && !c->vis[r]) && c->ops[r] != RW_OP_INSERT &&
                c->ops[r] != RW_OP_UPDATE_INSERT)
                b.all_insert = 0;
        b.all_valid = 1;
        for (int ci = 0; ci < m.n_cols[s] && b.all_valid; ci++)
            for (uint32_t r = 0; r < n && b.all_valid; r++)
                b.all_valid = c->cols[ci].valid[r];
        // chained buckets need no uniqueness pre-pass (round-1's plain
        // head-push lever is gone: inserts are single-CAS regardless)
        b.unique_keys = 0;
        *bout = b;
        return RW_OK;
    }

    // inner-probe pre-assigned layout bookkeeping (see jprobe_setup_kernel)
    uint32_t* d_probe_base = nullptr; // the batch's reserved record base
    unsigned long long* d_emit_count = nullptr;
    uint8_t* d_vis_scratch = nullptr;
    uint32_t d_vis_cap = 0;
    uint32_t last_sparse_rows = 0;
    bool sparse_out = false;

    // partitioned-pipeline device buffers (lazily allocated; ~260 KB)
    uint32_t* d_ptot = nullptr;  // padded per-partition totals (count)
    uint32_t* d_pcur = nullptr;  // padded per-partition scatter cursors
    uint32_t* d_part_base = nullptr;
    uint32_t* d_row_base = nullptr;

    // The partitioned pipeline handles the shape under which the
    // batch-parallel insert commutes with the reference's sequential loop:
    // inner join, every visible row an Insert (same-side rows never match
    // each other, so equal keys within the batch are independent),
    // whole-batch range.
    bool can_partition(const JoinBatchDev& b, uint32_t r0, uint32_t r1) {
        // Default OFF: with LIC-resident bloom-bucket slots the fused
        // monolithic kernel beat the 4-phase pipeline on q8 (0.50 vs
        // 0.61 ms/1M rows — the prelude's ~125 us exceeds the probe
        // phase's locality win; profiles/r02 series). Kept selectable
        // (RW_JOIN_PART=1) and parity-covered.
        const char* e = getenv("RW_JOIN_PART");
        if (!e || atoi(e) == 0) return false;
        if (m.join_type != RW_JOIN_INNER || m.append_only) return false;
        if (!b.all_insert || b.vis) return false;
        if (r0 != 0 || r1 != b.n_rows) return false;
        if (b.n_rows < JPART_MIN_ROWS) return false;
        // partitions must map to multi-slot windows on both sides
        if (side[0].cap_mask < 4 * JPART_P - 1 ||
            side[1].cap_mask < 4 * JPART_P - 1)
            return false;
        return true;
    }

    int ensure_part_bufs() {
        if (d_ptot) return RW_OK;
        HIP_TRY(hipMalloc(&d_ptot, (size_t)JPART_P * JPART_PAD * 4));
        HIP_TRY(hipMemsetAsync(d_ptot, 0, (size_t)JPART_P * JPART_PAD * 4,
                               stream));
        HIP_TRY(hipMalloc(&d_pcur, (size_t)JPART_P * JPART_PAD * 4));
        HIP_TRY(hipMalloc(&d_part_base, (size_t)(JPART_P + 1) * 4));
        HIP_TRY(hipMalloc(&d_row_base, 4));
        return RW_OK;
    }

    int probe(int s, const JoinBatchDev& b, bool timed, uint32_t r0, uint32_t r1) {
        uint32_t blocks = (r1 - r0 + 255) / 256;
        // one row per lane up to 16K blocks: a grid-stride iteration
        // serializes its dependent walk behind the previous row's, while
        // fresh blocks give the scheduler independent work as waves
        // retire (measured on the q8 shape; 2048-cap was the G11 default)
        if (blocks > 16384) blocks = 16384;
        if (!blocks) blocks = 1;
        int slot = -1;
        if (timed) {
            slot = (int)(ev_head++ % EV_RING);
            if (!ev0[slot]) {
                HIP_TRY(hipEventCreate(&ev0[slot]));
                HIP_TRY(hipEventCreate(&ev1[slot]));
            }
            if (ev_harvest(slot) != RW_OK) return RW_E_INTERNAL;
            HIP_TRY(hipEventRecord(ev0[slot], stream));
        }
        const char* dbg_e = getenv("RW_JOIN_SKIP"); // bench A/B only:
        int dbg_skip = dbg_e ? atoi(dbg_e) : 0; // 1=no emits, 2=no own insert
        if (m.join_type == RW_JOIN_INNER && !m.append_only &&
            !can_partition(b, r0, r1) && r0 == 0) {
            // pre-assigned sparse output + one-bump record reservation
            if (!d_probe_base) {
                HIP_TRY(hipMalloc(&d_probe_base, 4));
                HIP_TRY(hipMalloc(&d_emit_count, 8));
            }
            if (b.n_rows > out.cap)
                FAIL(RW_E_INVAL, "probe batch larger than the output block");
            HIP_TRY(hipMemsetAsync(out.ops, 0xFF, b.n_rows, stream));
            jprobe_setup_kernel<<<1, 1, 0, stream>>>(
                out, side[s].row_cursor, side[s].row_cap, d_probe_base,
                b.n_rows);
            last_sparse_rows = b.n_rows;
            sparse_out = true;
        }
        if (can_partition(b, r0, r1)) {
            int rc = ensure_part_bufs();
            if (rc != RW_OK) return rc;
            jpart_count_kernel<<<256, 1024, 0, stream>>>(b, m, s, d_ptot);
            jpart_scan_kernel<<<1, 1024, 0, stream>>>(
                d_ptot, d_pcur, d_part_base, side[s].row_cursor,
                side[s].row_cap, d_row_base, out.counters + 1);
            jpart_scatter_kernel<<<2048, 256, 0, stream>>>(
                b, m, s, side[s], d_pcur, d_row_base, out.counters + 1);
            uint32_t wslots = (side[s].cap_mask + 1) >> JPART_LOG2;
            size_t shmem = (size_t)wslots * 8 + 256 * side[s].row_stride;
            if (shmem <= 64 * 1024) {
                jpart_probe_insert_lds_kernel<<<JPART_P, 256, shmem,
                                                stream>>>(
                    side[s], side[1 - s], m, s, out, d_part_base, d_row_base,
                    out.counters + 1);
            } else {
                jpart_probe_insert_kernel<<<JPART_P, 256, 0, stream>>>(
                    side[s], side[1 - s], m, s, out, d_part_base, d_row_base,
                    out.counters + 1);
            }
        } else if (m.join_type != RW_JOIN_INNER) {
            join_probe_noninner_kernel<<<blocks, 256, 0, stream>>>(
                b, side[s], side[1 - s], m, s, out, r0, r1, dbg_skip);
        } else {
            join_probe_kernel<<<blocks, 256, 0, stream>>>(
                b, side[s], side[1 - s], m, s, out, r0, r1, d_probe_base,
                dbg_skip);
        }
        if (timed) {
            HIP_TRY(hipEventRecord(ev1[slot], stream));
            ev_pending[slot] = 1;
            probe_launches++;
            probe_rows += b.n_rows;
        }
        return RW_OK;
    }

    // §8f-2 checkpoint spill for the join state table: per side, the
    // epoch's KV deltas in the reference's encodings — key = memcomparable
    // (join key ∥ deduped input pk), value = value-encoded full row
    // (state_table.rs:1615-1727 via OrderedRowSerde; record framing matches
    // the agg spill: [put u8][klen u32 LE][k][vlen u32 LE][v], emitted in
    // memcmp key order). Deltas: rows appended since the last drain that
    // are still alive → PUT; rows killed this epoch that predate the last
    // drain → DELETE; kill-after-insert within the epoch nets away (the
    // record is dead, its PUT is skipped). Degree tables spill too (the
    // block below, drained via rw_join_degree_drain); watermark-TTL
    // cleanup is not spilled (the reference cleans via watermark hints).
    uint32_t flush_mark[2] = {0, 0};
    // rw_join_compact ping-pong stores (lazily allocated once per side;
    // a per-call hipMalloc of row_cap*stride dominates compaction cost)
    uint8_t* compact_scratch[2] = {nullptr, nullptr};
    int checkpoint_drain(int s, std::vector<uint8_t>& sp) {
        HIP_TRY(hipStreamSynchronize(stream));
        JoinSideDev& js = side[s];
        uint32_t cur = 0, kcur = 0;
        HIP_TRY(hipMemcpy(&cur, js.row_cursor, 4, hipMemcpyDeviceToHost));
        HIP_TRY(hipMemcpy(&kcur, js.killed_cursor, 4, hipMemcpyDeviceToHost));
        if (kcur > js.killed_cap)
            FAIL(RW_E_INTERNAL, "join kill list overflow (lost deltas)");
        if (cur > js.row_cap) cur = js.row_cap;
        uint32_t mark = flush_mark[s];
        size_t stride = js.row_stride;
        std::vector<uint8_t> fresh((size_t)(cur - mark) * stride);
        if (cur > mark)
            HIP_TRY(hipMemcpy(fresh.data(), js.rows + (size_t)mark * stride,
                              fresh.size(), hipMemcpyDeviceToHost));
        std::vector<uint32_t> kills(kcur);
        if (kcur)
            HIP_TRY(hipMemcpy(kills.data(), js.killed, (size_t)kcur * 4,
                              hipMemcpyDeviceToHost));
        auto encode_key = [&](const uint8_t* rec, std::string& k) {
            const uint32_t vb = ((const uint32_t*)rec)[2]; // validbits
            const int64_t* vals = (const int64_t*)(rec + 16);
            std::vector<uint8_t> kb;
            auto put_datum = [&](uint8_t col) {
                rwcodec::DatumC d{!((vb >> col) & 1), vals[col], 0};
                rwcodec::memcmp_encode_datum(kb, types[s][col], d, {});
            };
            for (int i = 0; i < m.KW; i++) put_datum(m.key_cols[s][i]);
            for (int i = 0; i < m.n_pk[s]; i++) put_datum(m.pk_cols[s][i]);
            k.assign((const char*)kb.data(), kb.size());
        };
        auto encode_val = [&](const uint8_t* rec, std::vector<uint8_t>& v) {
            const uint32_t vb = ((const uint32_t*)rec)[2];
            const int64_t* vals = (const int64_t*)(rec + 16);
            for (int c = 0; c < m.n_cols[s]; c++) {
                rwcodec::DatumC d{!((vb >> c) & 1), vals[c], 0};
                rwcodec::value_encode_datum(v, types[s][c], d);
            }
        };
        // batched gather of every pre-epoch record the kill and degree-
        // dirty lists reference (single kernel + one D2H copy)
        std::vector<uint32_t> dl;
        uint32_t dn = 0;
        if (m.need_deg[s]) {
            HIP_TRY(hipMemcpy(&dn, js.deg_dirty_n, 4,
                              hipMemcpyDeviceToHost));
            dl.resize(dn);
            if (dn)
                HIP_TRY(hipMemcpy(dl.data(), js.deg_dirty_list,
                                  (size_t)dn * 4, hipMemcpyDeviceToHost));
            HIP_TRY(hipMemset(js.deg_dirty_flag, 0, (size_t)js.row_cap * 4));
            HIP_TRY(hipMemset(js.deg_dirty_n, 0, 4));
        }
        std::vector<uint32_t> old_ids;
        for (uint32_t i = 0; i < kcur; i++)
            if (kills[i] < mark) old_ids.push_back(kills[i]);
        size_t n_kill_old = old_ids.size();
        for (uint32_t i = 0; i < dn; i++)
            if (dl[i] < mark) old_ids.push_back(dl[i]);
        std::vector<uint8_t> oldbuf(old_ids.size() * stride);
        if (!old_ids.empty()) {
            uint32_t* d_ids = nullptr;
            uint8_t* d_out = nullptr;
            HIP_TRY(hipMalloc(&d_ids, old_ids.size() * 4));
            HIP_TRY(hipMalloc(&d_out, oldbuf.size()));
            HIP_TRY(hipMemcpy(d_ids, old_ids.data(), old_ids.size() * 4,
                              hipMemcpyHostToDevice));
            uint32_t gblocks = ((uint32_t)old_ids.size() + 255) / 256;
            if (gblocks > 2048) gblocks = 2048;
            jgather_rows_kernel<<<gblocks, 256, 0, stream>>>(
                js, d_ids, (uint32_t)old_ids.size(), d_out);
            int rcg = hipMemcpy(oldbuf.data(), d_out, oldbuf.size(),
                                hipMemcpyDeviceToHost);
            hipFree(d_ids);
            hipFree(d_out);
            if (rcg != hipSuccess)
                FAIL(RW_E_INTERNAL, "drain record gather failed");
        }
        std::map<std::string, std::optional<std::vector<uint8_t>>> delta;
        for (size_t i = 0; i < n_kill_old; i++) {
            std::string k;
            encode_key(oldbuf.data() + i * stride, k);
            delta[k] = std::nullopt; // DELETE (a later PUT overwrites = net)
        }
        for (uint32_t i = mark; i < cur; i++) {
            const uint8_t* rec = fresh.data() + (size_t)(i - mark) * stride;
            if (!((const uint32_t*)rec)[0]) continue; // dead: netted away
            std::string k;
            encode_key(rec, k);
            std::vector<uint8_t> v;
            encode_val(rec, v);
            delta[k] = std::move(v);
        }
        auto put32 = [&](uint32_t x) {
            for (int b = 0; b < 4; b++) sp.push_back((uint8_t)(x >> (8 * b)));
        };
        for (auto& [k, v] : delta) {
            sp.push_back(v.has_value() ? 1 : 0);
            put32((uint32_t)k.size());
            sp.insert(sp.end(), k.begin(), k.end());
            put32(v ? (uint32_t)v->size() : 0);
            if (v) sp.insert(sp.end(), v->begin(), v->end());
        }
        // §8f-2 degree-table deltas (build_degree_row, join/row.rs:99-113:
        // pk = jk ∥ pk as the main table, value = order key ++ degree i64).
        // Keys mirror the main delta (a degree row exists iff its state row
        // does), plus pre-epoch rows whose degree changed during probes
        // (deg_dirty list). Serialized into deg_spill[s], returned by
        // rw_join_degree_drain after this drain.
        if (m.need_deg[s]) {
            auto encode_deg_val = [&](const uint8_t* rec,
                                      std::vector<uint8_t>& v) {
                const uint32_t vb = ((const uint32_t*)rec)[2];
                const int64_t* vals = (const int64_t*)(rec + 16);
                auto put_datum = [&](uint8_t col) {
                    rwcodec::DatumC d{!((vb >> col) & 1), vals[col], 0};
                    rwcodec::value_encode_datum(v, types[s][col], d);
                };
                for (int i = 0; i < m.KW; i++) put_datum(m.key_cols[s][i]);
                for (int i = 0; i < m.n_pk[s]; i++) put_datum(m.pk_cols[s][i]);
                uint32_t deg = ((const uint32_t*)rec)[3];
                rwcodec::value_encode_datum(v, RW_T_I64,
                                            {false, (long long)deg, 0});
            };
            std::map<std::string, std::optional<std::vector<uint8_t>>> dd;
            for (size_t i = 0; i < n_kill_old; i++) {
                std::string k;
                encode_key(oldbuf.data() + i * stride, k);
                dd[k] = std::nullopt;
            }
            for (uint32_t i = mark; i < cur; i++) {
                const uint8_t* rec = fresh.data() + (size_t)(i - mark) * stride;
                if (!((const uint32_t*)rec)[0]) continue;
                std::string k;
                encode_key(rec, k);
                std::vector<uint8_t> v;
                encode_deg_val(rec, v);
                dd[k] = std::move(v);
            }
            // pre-epoch rows whose degree changed (gathered above after
            // the kill entries)
            for (size_t i = n_kill_old; i < old_ids.size(); i++) {
                const uint8_t* rec = oldbuf.data() + i * stride;
                if (!((const uint32_t*)rec)[0]) continue; // killed
                std::string k;
                encode_key(rec, k);
                std::vector<uint8_t> v;
                encode_deg_val(rec, v);
                dd[k] = std::move(v);
            }
            auto& dsp = deg_spill[s];
            auto dput32 = [&](uint32_t x) {
                for (int b = 0; b < 4; b++)
                    dsp.push_back((uint8_t)(x >> (8 * b)));
            };
            for (auto& [k, v] : dd) {
                dsp.push_back(v.has_value() ? 1 : 0);
                dput32((uint32_t)k.size());
                dsp.insert(dsp.end(), k.begin(), k.end());
                dput32(v ? (uint32_t)v->size() : 0);
                if (v) dsp.insert(dsp.end(), v->begin(), v->end());
            }
        }
        HIP_TRY(hipMemset(js.killed_cursor, 0, 4));
        flush_mark[s] = cur;
        return RW_OK;
    }
    std::vector<uint8_t> deg_spill[2];

    int drain_output() {
        uint32_t ctr[2];
        HIP_TRY(hipMemcpy(ctr, out.counters, 8, hipMemcpyDeviceToHost));
        if (ctr[1] == 1) FAIL(RW_E_INTERNAL, "join output buffer overflow");
        if (ctr[1] == 2) FAIL(RW_E_INTERNAL, "join key table full");
        if (ctr[1] == 3) FAIL(RW_E_INTERNAL, "join row store full");
        uint32_t n_out = ctr[0];
        if (n_out) {
            // columnar device block: copy each column's first n_out entries
            std::vector<int64_t> vals((size_t)n_out * m.n_out); // [c][n]
            std::vector<uint8_t> nulls((size_t)n_out * m.n_out);
            std::vector<uint8_t> ops(n_out);
            for (int ci = 0; ci < m.n_out; ci++) {
                HIP_TRY(hipMemcpy(vals.data() + (size_t)ci * n_out,
                                  out.vals + (size_t)ci * out.cap,
                                  (size_t)n_out * 8, hipMemcpyDeviceToHost));
                HIP_TRY(hipMemcpy(nulls.data() + (size_t)ci * n_out,
                                  out.nulls + (size_t)ci * out.cap, n_out,
                                  hipMemcpyDeviceToHost));
            }
            HIP_TRY(hipMemcpy(ops.data(), out.ops, n_out, hipMemcpyDeviceToHost));
            if (sparse_out) {
                // compact the pre-assigned sparse region (sentinel 0xFF)
                std::vector<uint32_t> real;
                real.reserve(n_out);
                for (uint32_t r2 = 0; r2 < n_out; r2++)
                    if (!(r2 < last_sparse_rows && ops[r2] == 0xFF))
                        real.push_back(r2);
                uint32_t nc = (uint32_t)real.size();
                std::vector<int64_t> cv((size_t)nc * m.n_out);
                std::vector<uint8_t> cn((size_t)nc * m.n_out);
                std::vector<uint8_t> co(nc);
                for (uint32_t j2 = 0; j2 < nc; j2++) {
                    co[j2] = ops[real[j2]];
                    for (int ci = 0; ci < m.n_out; ci++) {
                        cv[(size_t)ci * nc + j2] =
                            vals[(size_t)ci * n_out + real[j2]];
                        cn[(size_t)ci * nc + j2] =
                            nulls[(size_t)ci * n_out + real[j2]];
                    }
                }
                vals.swap(cv);
                nulls.swap(cn);
                ops.swap(co);
                n_out = nc;
                sparse_out = false;
            }
            if (!n_out) {
                HIP_TRY(hipMemset(out.counters, 0, 8));
                return RW_OK;
            }
            uint32_t max_rows = desc.chunk_size ? desc.chunk_size : 1024;
            if (max_rows < 2) max_rows = 2;
            for (uint32_t start = 0; start < n_out; start += max_rows) {
                uint32_t take = n_out - start;
                if (take > max_rows) take = max_rows;
                auto* ch = new RwChunk();
                auto* cols = new RwColumn[m.n_out];
                auto* o = new uint8_t[take];
                memcpy(o, ops.data() + start, take);
                for (int ci = 0; ci < m.n_out; ci++) {
                    auto* data = new int64_t[take];
                    auto* valid = new uint8_t[take];
                    for (uint32_t r = 0; r < take; r++) {
                        data[r] = vals[(size_t)ci * n_out + start + r];
                        valid[r] = !nulls[(size_t)ci * n_out + start + r];
                    }
                    cols[ci].type = out_types[ci];
                    cols[ci].valid = valid;
                    cols[ci].data = data;
                }
                ch->n_rows = take;
                ch->n_cols = m.n_out;
                ch->ops = o;
                ch->vis = nullptr;
                ch->cols = cols;
                outq.push_back(ch);
            }
        }
        HIP_TRY(hipMemset(out.counters, 0, 8));
        return RW_OK;
    }

    // The parallel kernel keeps the reference's sequential per-chunk
    // semantics except when a DELETE targets a row also INSERTed in the same
    // chunk (hash_join.rs processes rows in order, :993). Detect those rows
    // host-side and run each such delete in its own single-row segment, in
    // row order — stream ordering makes earlier segments' state visible.
    std::vector<uint32_t> conflict_segments(int s, const RwChunk* c) {
        std::vector<uint32_t> bounds; // segment start offsets (0 implied)
        bool any_delete = false;
        for (uint32_t r = 0; r < c->n_rows; r++) {
            uint8_t op = c->ops[r];
            if (op == RW_OP_DELETE || op == RW_OP_UPDATE_DELETE) {
                any_delete = true;
                break;
            }
        }
        if (!any_delete) return bounds;
        auto row_key = [&](uint32_t r) {
            std::string k;
            k.reserve(m.n_cols[s] * 9);
            for (int ci = 0; ci < m.n_cols[s]; ci++) {
                uint8_t valid = c->cols[ci].valid[r];
                k.push_back((char)valid);
                int64_t v = valid ? ((const int64_t*)c->cols[ci].data)[r] : 0;
                k.append((const char*)&v, 8);
            }
            return k;
        };
        auto join_key = [&](uint32_t r) {
            std::string k;
            for (int i = 0; i < m.KW; i++) {
                uint8_t ci = m.key_cols[s][i];
                uint8_t valid = c->cols[ci].valid[r];
                k.push_back((char)valid);
                int64_t v = valid ? ((const int64_t*)c->cols[ci].data)[r] : 0;
                k.append((const char*)&v, 8);
            }
            return k;
        };
        std::unordered_multiset<std::string> inserts;
        // degree-carrying types: a mixed insert/delete chunk sharing a JOIN
        // KEY races the matched rows' degree transitions — isolate all rows
        // of such keys (sequential semantics, hash_join.rs row order)
        bool deg = m.need_deg[0] || m.need_deg[1];
        std::unordered_map<std::string, int> jk_ops; // 1=ins 2=del bits
        for (uint32_t r = 0; r < c->n_rows; r++) {
            if (c->vis && !c->vis[r]) continue;
            uint8_t op = c->ops[r];
            bool ins = (op == RW_OP_INSERT || op == RW_OP_UPDATE_INSERT);
            if (ins) inserts.insert(row_key(r));
            if (deg) jk_ops[join_key(r)] |= ins ? 1 : 2;
        }
        for (uint32_t r = 0; r < c->n_rows; r++) {
            if (c->vis && !c->vis[r]) continue;
            uint8_t op = c->ops[r];
            bool is_del = (op == RW_OP_DELETE || op == RW_OP_UPDATE_DELETE);
            bool conflict =
                (is_del && !inserts.empty() && inserts.count(row_key(r))) ||
                (deg && jk_ops[join_key(r)] == 3);
            if (conflict) {
                bounds.push_back(r);     // segment ends before this row
                bounds.push_back(r + 1); // the row runs alone, in order
            }
        }
        return bounds;
    }

    // ----- epoch-batched ingestion (the agg's rw_hash_agg_ingest_mode
    // analogue): consecutive SAME-SIDE chunks merge into one staged host
    // batch applied as one launch at the next side switch / barrier /
    // watermark. Exactly order-equivalent: probes read only the MATCH
    // side (which a same-side run never mutates), own-side insert/delete
    // interleavings within the merged batch are the conflict-segment
    // pre-pass's existing job, and cross-SIDE boundaries always flush. -----
    bool epoch_ingest = false;
    int pend_side = -1;
    std::vector<uint8_t> pend_ops;
    std::vector<std::vector<int64_t>> pend_vals;  // per column (16-B decimal n/a)
    std::vector<std::vector<uint8_t>> pend_valid;

    int pending_flush() {
        if (pend_side < 0 || pend_ops.empty()) {
            pend_side = -1;
            return RW_OK;
        }
        int s2 = pend_side;
        uint32_t n = (uint32_t)pend_ops.size();
        std::vector<RwColumn> cols(m.n_cols[s2]);
        for (int c2 = 0; c2 < m.n_cols[s2]; c2++) {
            cols[c2].type = types[s2][c2];
            cols[c2].valid = pend_valid[c2].data();
            cols[c2].data = pend_vals[c2].data();
        }
        RwChunk ch{};
        ch.n_rows = n;
        ch.n_cols = (uint32_t)cols.size();
        ch.ops = pend_ops.data();
        ch.vis = nullptr;
        ch.cols = cols.data();
        pend_side = -1;
        int rc = push_chunk_now(s2, &ch);
        pend_ops.clear();
        for (auto& v : pend_vals) v.clear();
        for (auto& v : pend_valid) v.clear();
        return rc;
    }

    int push_chunk(int s, const RwChunk* c) {
        if (s != 0 && s != 1) FAIL(RW_E_INVAL, "bad side");
        if (epoch_ingest) {
            if (c->vis)
                FAIL(RW_E_INVAL,
                     "epoch-batched ingest requires visibility-compacted "
                     "chunks");
            if (pend_side >= 0 && pend_side != s) {
                int rc = pending_flush();
                if (rc != RW_OK) return rc;
            }
            if (pend_side < 0) {
                pend_side = s;
                pend_vals.assign(m.n_cols[s], {});
                pend_valid.assign(m.n_cols[s], {});
            }
            pend_ops.insert(pend_ops.end(), c->ops, c->ops + c->n_rows);
            for (int c2 = 0; c2 < m.n_cols[s]; c2++) {
                const int64_t* d = (const int64_t*)c->cols[c2].data;
                pend_vals[c2].insert(pend_vals[c2].end(), d, d + c->n_rows);
                pend_valid[c2].insert(pend_valid[c2].end(),
                                      c->cols[c2].valid,
                                      c->cols[c2].valid + c->n_rows);
            }
            return RW_OK;
        }
        return push_chunk_now(s, c);
    }

    int push_chunk_now(int s, const RwChunk* c) {
        JoinBatchDev b;
        int rc = upload(s, c, &b);
        if (rc != RW_OK) return rc;
        std::vector<uint32_t> bounds = conflict_segments(s, c);
        uint32_t start = 0;
        for (uint32_t bnd : bounds) {
            if (bnd > start) {
                rc = probe(s, b, true, start, bnd);
                if (rc != RW_OK) return rc;
            }
            start = bnd;
        }
        if (start < b.n_rows) {
            rc = probe(s, b, true, start, b.n_rows);
            if (rc != RW_OK) return rc;
        }
        HIP_TRY(hipStreamSynchronize(stream));
        return drain_output();
    }

    int flush(uint64_t) {
        if (epoch_ingest) {
            int rc = pending_flush();
            if (rc != RW_OK) return rc;
        }
        HIP_TRY(hipStreamSynchronize(stream));
        return RW_OK;
    }

    int watermark(int s, uint32_t col_idx, int64_t val, uint32_t* out_cols,
                  int64_t* out_vals, int max_out, int* n_out_p) {
        if (epoch_ingest) {
            int rcp = pending_flush();
            if (rcp != RW_OK) return rcp;
        }
        int n_out = 0;
        auto& mine = s == 0 ? wm_side0 : wm_side1;
        for (uint32_t idx = 0; idx < (uint32_t)m.KW; idx++) {
            if (m.key_cols[s][idx] != col_idx) continue;
            mine[idx] = {true, val};
            if (!wm_side0[idx].has || !wm_side1[idx].has) continue;
            int64_t sel = std::min(wm_side0[idx].val, wm_side1[idx].val);
            if (wm_out[idx].has && sel <= wm_out[idx].val) continue;
            wm_out[idx] = {true, sel};
            bool clean = false;
            for (size_t w = 0; w < wm_pos.size(); w++)
                if (wm_pos[w] == idx && wm_clean[w]) clean = true;
            if (clean) {
                for (int sd = 0; sd < 2; sd++)
                    join_clean_kernel<<<2048, 256, 0, stream>>>(
                        side[sd], (int)m.key_cols[sd][idx], sel, m.KW);
                HIP_TRY(hipStreamSynchronize(stream));
            }
            // update side's output columns first, then the match side's
            for (int pass = 0; pass < 2; pass++) {
                int s2 = pass == 0 ? s : 1 - s;
                uint32_t src = m.key_cols[s2][idx] +
                               (s2 == 1 ? (uint32_t)m.n_cols[0] : 0);
                for (int oi = 0; oi < m.n_out; oi++) {
                    uint32_t osrc = (uint32_t)m.out_col[oi] +
                                    (m.out_src[oi] ? (uint32_t)m.n_cols[0] : 0);
                    if (osrc == src && n_out < max_out) {
                        out_cols[n_out] = (uint32_t)oi;
                        out_vals[n_out] = sel;
                        n_out++;
                    }
                }
            }
        }
        // inequality-pair watermarks (hash_join.rs:869-914): buffer per
        // side; the min across sides emits for the LARGER side's output
        // columns and (clean flag) sweeps that side's rows whose pair
        // column sorts below it — through the kill list, so the next
        // drain nets the cleaned rows to DELETE frames
        for (size_t p = 0; p < ineq_col[0].size(); p++) {
            if (ineq_col[s][p] != col_idx) continue;
            ineq_wm[s][p] = {true, val};
            if (!ineq_wm[0][p].has || !ineq_wm[1][p].has) continue;
            int64_t sel = std::min(ineq_wm[0][p].val, ineq_wm[1][p].val);
            if (ineq_out[p].has && sel <= ineq_out[p].val) continue;
            ineq_out[p] = {true, sel};
            int larger = ineq_larger[p] ? 0 : 1;
            if (ineq_do_clean[p]) {
                join_clean_kernel<<<2048, 256, 0, stream>>>(
                    side[larger], (int)ineq_col[larger][p], sel, m.KW);
                HIP_TRY(hipStreamSynchronize(stream));
            }
            uint32_t src = ineq_col[larger][p] +
                           (larger == 1 ? (uint32_t)m.n_cols[0] : 0);
            for (int oi = 0; oi < m.n_out; oi++) {
                uint32_t osrc = (uint32_t)m.out_col[oi] +
                                (m.out_src[oi] ? (uint32_t)m.n_cols[0] : 0);
                if (osrc == src && n_out < max_out) {
                    out_cols[n_out] = (uint32_t)oi;
                    out_vals[n_out] = sel;
                    n_out++;
                }
            }
        }
        *n_out_p = n_out;
        return RW_OK;
    }

    RwChunk* poll() {
        if (outq.empty()) return nullptr;
        RwChunk* c = outq.front();
        outq.erase(outq.begin());
        return c;
    }

    ~HashJoin() {
        for (int s = 0; s < EV_RING; s++)
            if (ev0[s]) {
                hipEventDestroy(ev0[s]);
                hipEventDestroy(ev1[s]);
            }
        if (d_ptot) {
            hipFree(d_ptot);
            hipFree(d_pcur);
            hipFree(d_part_base);
            hipFree(d_row_base);
        }
        if (d_probe_base) {
            hipFree(d_probe_base);
            hipFree(d_emit_count);
        }
        if (d_vis_scratch) hipFree(d_vis_scratch);
        for (int s = 0; s < 2; s++)
            if (compact_scratch[s]) hipFree(compact_scratch[s]);
        for (int s = 0; s < 2; s++) {
            JoinSideDev& js = side[s];
            if (js.slots8) {
                hipFree(js.slots8);
                hipFree(js.rows);
                hipFree(js.row_cursor);
                if (js.killed) {
                    hipFree(js.killed);
                    hipFree(js.killed_cursor);
                }
            }
            for (int c = 0; c < m.n_cols[s]; c++) {
                if (stage[s].col_vals[c]) hipFree(stage[s].col_vals[c]);
                if (stage[s].col_valid[c]) hipFree(stage[s].col_valid[c]);
            }
            if (stage[s].ops) hipFree(stage[s].ops);
            if (stage[s].vis) hipFree(stage[s].vis);
        }
        if (out.vals) {
            hipFree(out.vals);
            hipFree(out.nulls);
            hipFree(out.ops);
            hipFree(out.counters);
        }
        if (zeros) hipFree(zeros);
        if (stream) hipStreamDestroy(stream);
        for (auto* c : outq) rw_chunk_free(c);
    }
};

extern "C" {

void* rw_hash_join_create(const RwHashJoinDesc* d) {
    auto* h = new HashJoin();
    if (h->init(d) != RW_OK) {
        delete h;
        return nullptr;
    }
    return h;
}
int rw_hash_join_push_chunk(void* h, int side, const RwChunk* c) {
    return ((HashJoin*)h)->push_chunk(side, c);
}
int rw_hash_join_flush(void* h, uint64_t epoch) { return ((HashJoin*)h)->flush(epoch); }

// epoch-batched ingest mode (rw_stream.h): same-side chunk runs merge into
// one launch; barriers/watermarks/side switches flush the staged run
int rw_hash_join_ingest_mode(void* h, int epoch_batched) {
    auto* j = (HashJoin*)h;
    if (!epoch_batched && j->pend_side >= 0)
        FAIL(RW_E_INVAL, "staged chunks pending; flush before disabling");
    j->epoch_ingest = epoch_batched != 0;
    return RW_OK;
}

int rw_hash_join_watermark(void* h, int side, uint32_t col_idx, int64_t val,
                           uint32_t* out_cols, int64_t* out_vals, int max_out) {
    int n = 0;
    int rc = ((HashJoin*)h)->watermark(side, col_idx, val, out_cols, out_vals,
                                       max_out, &n);
    return rc == RW_OK ? n : rc;
}

int rw_hash_agg_watermark(void* h, uint32_t group_key_pos, int64_t val) {
    auto* agg = (HashAgg*)h;
    if (group_key_pos >= (uint32_t)agg->KW) FAIL(RW_E_INVAL, "bad group key pos");
    if (agg->eowc) {
        // EOWC: buffer the window watermark; windows close at the next
        // barrier (hash_agg.rs:657-700)
        if (group_key_pos != 0) FAIL(RW_E_INVAL, "EOWC watermark must be on group key 0");
        if (!agg->has_pending_wm || val > agg->pending_wm)
            agg->pending_wm = val;
        agg->has_pending_wm = true;
        return RW_OK;
    }
    // clean + capture persisted cleaned groups for DELETE spill frames
    // (state-table watermark range delete, state_table.rs:1707)
    uint32_t crec_cap = 1u << 18;
    for (;;) {
        DedupDirtyRec* crecs = nullptr;
        uint32_t* cmeta = nullptr; // [0]=n [1]=overflow
        HIP_TRY(hipMalloc(&crecs, (size_t)crec_cap * sizeof(DedupDirtyRec)));
        HIP_TRY(hipMalloc(&cmeta, 8));
        HIP_TRY(hipMemset(cmeta, 0, 8));
        agg_clean_kernel<<<2048, 256, 0, agg->stream>>>(
            agg->t, (int)group_key_pos, val, agg->KW, agg->n_calls,
            agg->cd(0), agg->cd(1), agg->cd(2), agg->cd(3), crecs, cmeta,
            crec_cap, cmeta + 1);
        int rcs = hipStreamSynchronize(agg->stream) == hipSuccess
                      ? RW_OK : RW_E_INTERNAL;
        uint32_t meta[2] = {0, 0};
        if (rcs == RW_OK)
            hipMemcpy(meta, cmeta, 8, hipMemcpyDeviceToHost);
        uint32_t n = meta[0] < crec_cap ? meta[0] : crec_cap;
        std::vector<DedupDirtyRec> recs(n);
        if (rcs == RW_OK && n)
            hipMemcpy(recs.data(), crecs, (size_t)n * sizeof(DedupDirtyRec),
                      hipMemcpyDeviceToHost);
        hipFree(crecs);
        hipFree(cmeta);
        if (rcs != RW_OK) FAIL(RW_E_INTERNAL, "agg clean sync failed");
        auto put32 = [&](uint32_t x) {
            for (int b = 0; b < 4; b++)
                agg->spill.push_back((uint8_t)(x >> (8 * b)));
        };
        for (uint32_t i = 0; i < n; i++) {
            agg->spill.push_back(0); // DELETE
            std::vector<uint8_t> k;
            for (int w = 0; w < agg->KW; w++) {
                rwcodec::DatumC d{((recs[i].nulls >> w) & 1) != 0,
                                  recs[i].key[w], 0};
                rwcodec::memcmp_encode_datum(k, agg->out_types[w], d, {});
            }
            put32((uint32_t)k.size());
            agg->spill.insert(agg->spill.end(), k.begin(), k.end());
            put32(0);
        }
        if (!meta[1]) break; // no overflow: every slot cleaned
        crec_cap *= 4;
    }
    // the same watermark cleans the DISTINCT dedup tables' group prefix
    for (size_t di = 0; di < agg->dedup_slots.size(); di++) {
        size_t dcap = (size_t)agg->dedup_cap_mask + 1;
        dedup_clean_kernel<<<2048, 256, 0, agg->stream>>>(
            agg->dedup_slots[di], dcap, (int)group_key_pos, val,
            agg->ddirty_flag[di], agg->ddirty_list[di], agg->ddirty_n[di]);
    }
    HIP_TRY(hipStreamSynchronize(agg->stream));
    return RW_OK;
}

// rescale re-scope (update_vnode_bitmap): drop state whose dist-key vnode
// (dist key = group key) is no longer owned; no retractions are emitted.
// `bitmap` is vnode_count/8 bytes, LSB-first per byte (common Bitmap layout).
int rw_hash_agg_update_vnode_bitmap(void* h, const uint8_t* bitmap,
                                    uint32_t vnode_count) {
    auto* agg = (HashAgg*)h;
    if (!vnode_count || vnode_count % 8 || vnode_count > 4096)
        FAIL(RW_E_INVAL, "vnode_count");
    if (ensure_crc_table() != RW_OK) FAIL(RW_E_INTERNAL, "crc table");
    if (!agg->d_vnode_bitmap)
        HIP_TRY(hipMalloc(&agg->d_vnode_bitmap, 512));
    HIP_TRY(hipMemcpyAsync(agg->d_vnode_bitmap, bitmap, vnode_count / 8,
                           hipMemcpyHostToDevice, agg->stream));
    uint8_t ty[4] = {RW_T_I64, RW_T_I64, RW_T_I64, RW_T_I64};
    for (int k = 0; k < agg->KW && k < 4; k++) ty[k] = agg->out_types[k];
    agg_vnode_scope_kernel<<<2048, 256, 0, agg->stream>>>(
        agg->t, agg->KW, agg->n_calls, agg->cd(0), agg->cd(1), agg->cd(2),
        agg->cd(3), agg->d_vnode_bitmap, vnode_count, ty[0], ty[1], ty[2],
        ty[3]);
    HIP_TRY(hipStreamSynchronize(agg->stream));
    return RW_OK;
}

// join dist key = join key; both sides are re-scoped together
int rw_hash_join_update_vnode_bitmap(void* h, const uint8_t* bitmap,
                                     uint32_t vnode_count) {
    auto* j = (HashJoin*)h;
    if (!vnode_count || vnode_count % 8 || vnode_count > 4096)
        FAIL(RW_E_INVAL, "vnode_count");
    if (ensure_crc_table() != RW_OK) FAIL(RW_E_INTERNAL, "crc table");
    if (!j->d_vnode_bitmap) HIP_TRY(hipMalloc(&j->d_vnode_bitmap, 512));
    HIP_TRY(hipMemcpyAsync(j->d_vnode_bitmap, bitmap, vnode_count / 8,
                           hipMemcpyHostToDevice, j->stream));
    for (int s = 0; s < 2; s++) {
        uint8_t ty[4] = {RW_T_I64, RW_T_I64, RW_T_I64, RW_T_I64};
        uint8_t kc[4] = {0, 0, 0, 0};
        for (int k = 0; k < j->m.KW && k < 4; k++) {
            ty[k] = j->types[s][j->m.key_cols[s][k]];
            kc[k] = j->m.key_cols[s][k];
        }
        join_vnode_scope_kernel<<<2048, 256, 0, j->stream>>>(
            j->side[s], j->m.KW, j->d_vnode_bitmap, vnode_count, ty[0], ty[1],
            ty[2], ty[3], kc[0], kc[1], kc[2], kc[3]);
    }
    HIP_TRY(hipStreamSynchronize(j->stream));
    return RW_OK;
}
RwChunk* rw_hash_join_poll(void* h) { return ((HashJoin*)h)->poll(); }
void rw_hash_join_destroy(void* h) { delete (HashJoin*)h; }

// --- bench support: device-resident probe batches; outputs stay in HBM
// (the downstream operator is device-resident — DESIGN.md §5; the
// PCIe-inclusive ingest rate is reported separately) ---

void* rw_join_bench_preload(void* h, int side, const RwChunk* c) {
    auto* j = (HashJoin*)h;
    auto* b = new JoinBatchDev{};
    uint32_t n = c->n_rows;
    for (int ci = 0; ci < j->m.n_cols[side]; ci++) {
        if (hipMalloc(&b->col_vals[ci], (size_t)n * 8) != hipSuccess) return nullptr;
        if (hipMalloc(&b->col_valid[ci], n) != hipSuccess) return nullptr;
        hipMemcpy(b->col_vals[ci], c->cols[ci].data, (size_t)n * 8,
                  hipMemcpyHostToDevice);
        hipMemcpy(b->col_valid[ci], c->cols[ci].valid, n, hipMemcpyHostToDevice);
    }
    if (hipMalloc(&b->ops, n) != hipSuccess) return nullptr;
    hipMemcpy(b->ops, c->ops, n, hipMemcpyHostToDevice);
    b->vis = nullptr;
    b->n_rows = n;
    b->all_insert = 1;
    for (uint32_t r = 0; r < n && b->all_insert; r++)
        if (c->ops[r] != RW_OP_INSERT && c->ops[r] != RW_OP_UPDATE_INSERT)
            b->all_insert = 0;
    b->all_valid = 1;
    for (uint32_t ci2 = 0; ci2 < c->n_cols && b->all_valid; ci2++)
        for (uint32_t r = 0; r < n && b->all_valid; r++)
            b->all_valid = c->cols[ci2].valid[r];
    b->unique_keys = 0;
    return b;
}

// launch the probe for a preloaded batch; outputs accumulate in the device
// buffer until rw_join_bench_drain
int rw_join_bench_apply(void* h, int side, void* batch) {
    auto* j = (HashJoin*)h;
    return j->probe(side, *(JoinBatchDev*)batch, true, 0,
                    ((JoinBatchDev*)batch)->n_rows);
}

// device-side drain: return match count, check errors, reset the cursor
__global__ void join_counters_reset_kernel(uint32_t* counters) {
    counters[0] = 0; // out cursor (overflow flag counters[1] persists)
}

// C-side q8 step loop: `steps` probe applies over a cycle of preloaded
// batches with a stream-ordered output-cursor reset after each step (the
// downstream consumes device-resident). No host round-trip per step — the
// Python-side drain cost ~0.15 ms/step of pure sync/copy overhead.
// Overflow flags surface at the next rw_join_bench_drain.
int rw_join_bench_run(void* h, int side, void** batches, int n_batches,
                      int steps) {
    auto* j = (HashJoin*)h;
    for (int i = 0; i < steps; i++) {
        // the inner probe's per-batch setup re-bases the output block, so
        // no per-step reset is needed; the last step's emissions survive
        // for the drain's count
        int rc = j->probe(side, *(JoinBatchDev*)batches[i % n_batches], true,
                          0, ((JoinBatchDev*)batches[i % n_batches])->n_rows);
        if (rc != RW_OK) return rc;
    }
    return RW_OK;
}

long long rw_join_bench_drain(void* h) {
    auto* j = (HashJoin*)h;
    if (hipStreamSynchronize(j->stream) != hipSuccess) return -1;
    uint32_t ctr[2];
    if (hipMemcpy(ctr, j->out.counters, 8, hipMemcpyDeviceToHost) != hipSuccess)
        return -1;
    if (ctr[1] != 0) {
        g_err = "join bench overflow/full (code " + std::to_string(ctr[1]) + ")";
        return -(long long)ctr[1] - 1;
    }
    long long emitted = (long long)ctr[0];
    if (j->sparse_out && j->d_emit_count) {
        // count the sparse region's real emissions (sentinel-aware)
        (void)hipMemset(j->d_emit_count, 0, 8);
        join_count_emitted_kernel<<<2048, 256, 0, j->stream>>>(
            j->out.ops, j->last_sparse_rows, j->out.counters,
            j->d_emit_count);
        unsigned long long cnt = 0;
        if (hipStreamSynchronize(j->stream) != hipSuccess) return -1;
        if (hipMemcpy(&cnt, j->d_emit_count, 8, hipMemcpyDeviceToHost) !=
            hipSuccess)
            return -1;
        emitted = (long long)cnt;
        j->sparse_out = false;
    }
    hipMemset(j->out.counters, 0, 8);
    return emitted;
}

__global__ void join_out_vis_kernel(const uint8_t* ops, uint32_t sparse_n,
                                    uint32_t total, uint8_t* vis) {
    uint32_t stride = gridDim.x * blockDim.x;
    for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < total;
         r += stride)
        vis[r] = !(r < sparse_n && ops[r] == 0xFF);
}

// Device-side pipeline hop: apply the join's accumulated output buffer
// (row-major [row][n_out]) directly as agg input via a strided AggBatch
// view — the fragment edge agg(join(...)) of the TPC-H q3 MV without
// leaving HBM. The agg must have been created with the join's output
// schema as its input schema. Resets the join's output cursor.
// agg-output -> join-input transpose: the agg's flush output block is
// row-major [row][gk ++ calls]; the join consumes columnar batches. The
// map picks which record columns become join input columns.
struct AggOutColMap {
    uint8_t n;
    uint8_t map[MAX_COLS];
};
__global__ void aggout_to_join_kernel(const long long* avals,
                                      const uint8_t* anulls,
                                      const uint8_t* aops, uint32_t n,
                                      int width, AggOutColMap cm,
                                      JoinBatchDev b) {
    uint32_t stride = gridDim.x * blockDim.x;
    for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < n;
         r += stride) {
        for (int i = 0; i < cm.n; i++) {
            b.col_vals[i][r] = avals[(size_t)r * width + cm.map[i]];
            b.col_valid[i][r] = !anulls[(size_t)r * width + cm.map[i]];
        }
        b.ops[r] = aops[r];
    }
}

// Device-resident fragment edge agg -> join (the q7 plan's
// StreamHashJoin right input = the windowed max's change stream,
// nexmark.yaml q7): applies the agg's device-resident flush output
// (rw_agg_flush_device's emitted rows, still in HBM) as one join input
// batch on `side`. Safe without the host conflict pre-pass: a U-/Delete
// record's target row is always a PREVIOUS barrier's U+/Insert (the agg
// emits at most one change pair per group per barrier), so no delete
// targets a same-batch insert (DESIGN §3.2's only inner-join conflict).
int rw_join_apply_aggout(void* join_h, void* agg_h, int side,
                         const uint32_t* col_map, int n_map,
                         uint64_t n_rows) {
    auto* j = (HashJoin*)join_h;
    auto* agg = (HashAgg*)agg_h;
    if (side != 0 && side != 1) FAIL(RW_E_INVAL, "bad side");
    if (n_map != j->m.n_cols[side])
        FAIL(RW_E_INVAL, "col_map size %d != side cols %d", n_map,
             j->m.n_cols[side]);
    if (!n_rows) return RW_OK;
    if (n_rows > UINT32_MAX) FAIL(RW_E_INVAL, "batch too large");
    HIP_TRY(hipStreamSynchronize(agg->stream));
    int rc = j->ensure_stage(side, (uint32_t)n_rows);
    if (rc != RW_OK) return rc;
    AggOutColMap cm{};
    cm.n = (uint8_t)n_map;
    for (int i = 0; i < n_map; i++) {
        if (col_map[i] >= (uint32_t)agg->out_width)
            FAIL(RW_E_INVAL, "col_map[%d]=%u outside agg record", i,
                 col_map[i]);
        cm.map[i] = (uint8_t)col_map[i];
    }
    JoinBatchDev b = j->stage[side];
    b.vis = nullptr;
    b.n_rows = (uint32_t)n_rows;
    b.all_insert = 0; // change stream carries U-/U+ pairs
    b.unique_keys = 0;
    uint32_t blocks = ((uint32_t)n_rows + 255) / 256;
    if (blocks > 2048) blocks = 2048;
    aggout_to_join_kernel<<<blocks, 256, 0, j->stream>>>(
        agg->t.out_vals, agg->t.out_nulls, agg->t.out_ops, (uint32_t)n_rows,
        agg->out_width, cm, b);
    return j->probe(side, b, true, 0, (uint32_t)n_rows);
}

// marshal the join's device-resident output buffer into host chunks on
// the poll queue (what push_chunk does after its probe) — for callers of
// rw_join_apply_aggout that want the emissions as chunks (parity tests)
int rw_join_marshal_output(void* join_h) {
    auto* j = (HashJoin*)join_h;
    HIP_TRY(hipStreamSynchronize(j->stream));
    return j->drain_output();
}

// device-resident VirtualNode::compute_chunk (vnode.rs:146-181) over a
// preloaded join batch's columns, launched on the join's stream — the q7
// pipeline's exchange-hop vnode computation at R=1 (the reference computes
// vnodes in HashDataDispatcher even with one downstream; the N>1 payload
// exchange is librw_exchange's path)
int rw_join_vnode_hop(void* join_h, void* batch, uint32_t col, uint8_t type,
                      uint32_t vnode_count) {
    auto* j = (HashJoin*)join_h;
    auto* b = (JoinBatchDev*)batch;
    if (j->d_vnode_hop_cap < b->n_rows) {
        if (j->d_vnode_hop) hipFree(j->d_vnode_hop);
        HIP_TRY(hipMalloc(&j->d_vnode_hop, (size_t)b->n_rows * 2 + 2));
        j->d_vnode_hop_cap = b->n_rows;
    }
    VnodeBatch vb{};
    vb.col_vals[0] = b->col_vals[col];
    vb.col_valid[0] = b->col_valid[col];
    vb.n_rows = b->n_rows;
    uint32_t blocks = (b->n_rows + 255) / 256;
    if (blocks > 2048) blocks = 2048;
    vnode_kernel<<<blocks, 256, 0, j->stream>>>(vb, 1, vnode_count,
                                                j->d_vnode_hop, type, 0, 0,
                                                0);
    return RW_OK;
}

int rw_agg_apply_joinout(void* agg_h, void* join_h) {
    auto* agg = (HashAgg*)agg_h;
    auto* j = (HashJoin*)join_h;
    if (agg->n_dec)
        FAIL(RW_E_INVAL, "decimal columns not yet in the join output block");
    if (hipStreamSynchronize(j->stream) != hipSuccess)
        FAIL(RW_E_INTERNAL, "join sync failed");
    uint32_t ctr[2];
    HIP_TRY(hipMemcpy(ctr, j->out.counters, 8, hipMemcpyDeviceToHost));
    if (ctr[1] != 0) FAIL(RW_E_INTERNAL, "join overflow %u before pipeline hop", ctr[1]);
    uint32_t n = ctr[0];
    if (!n) return RW_OK;
    if (!j->zeros) {
        HIP_TRY(hipMalloc(&j->zeros, (size_t)j->out.cap * j->m.n_out));
        HIP_TRY(hipMemset(j->zeros, 0, (size_t)j->out.cap * j->m.n_out));
    }
    AggBatch b{};
    b.stride = 1; // columnar output block: plain columnar agg input
    b.valid_inverted = 1;
    auto bind = [&](int slot, int src_col) {
        b.col_vals[slot] = j->out.vals + (size_t)src_col * j->out.cap;
        b.col_valid[slot] = j->out.nulls + (size_t)src_col * j->out.cap;
    };
    for (int i = 0; i < agg->KW; i++) {
        if ((int)agg->group_key[i] >= j->m.n_out)
            FAIL(RW_E_INVAL, "agg group key %u outside join output", agg->group_key[i]);
        bind(i, (int)agg->group_key[i]);
    }
    for (int ci = 0; ci < agg->n_calls; ci++) {
        if (agg->calls[ci].arg >= 0) {
            if (agg->calls[ci].arg >= j->m.n_out)
                FAIL(RW_E_INVAL, "agg arg outside join output");
            bind(agg->KW + ci, agg->calls[ci].arg);
        } else {
            b.col_vals[agg->KW + ci] = j->out.vals; // unread
            b.col_valid[agg->KW + ci] = j->zeros;   // inverted -> all valid
        }
    }
    if (agg->n_minput) {
        for (size_t k = 0; k < agg->stream_key.size(); k++) {
            if ((int)agg->stream_key[k] >= j->m.n_out)
                FAIL(RW_E_INVAL, "agg stream key outside join output");
            bind(agg->KW + agg->n_calls + (int)k, (int)agg->stream_key[k]);
        }
    }
    b.ops = j->out.ops;
    b.vis = nullptr;
    b.n_rows = n;
    if (j->sparse_out) {
        // pre-assigned sparse output: sentinel rows masked by visibility
        if (j->d_vis_cap < n) {
            if (j->d_vis_scratch) hipFree(j->d_vis_scratch);
            HIP_TRY(hipMalloc(&j->d_vis_scratch, n));
            j->d_vis_cap = n;
        }
        uint32_t blocks = (n + 255) / 256;
        if (blocks > 2048) blocks = 2048;
        join_out_vis_kernel<<<blocks, 256, 0, agg->stream>>>(
            j->out.ops, j->last_sparse_rows, n, j->d_vis_scratch);
        b.vis = j->d_vis_scratch;
        b.dense = 0;
        j->sparse_out = false;
    }
    // run the agg on ITS stream after the join's work is drained (synced
    // above), then reset the join cursor
    int rc = agg->apply(b, true);
    if (rc != RW_OK) return rc;
    if (hipStreamSynchronize(agg->stream) != hipSuccess)
        FAIL(RW_E_INTERNAL, "agg sync failed");
    HIP_TRY(hipMemset(j->out.counters, 0, 8));
    return agg->check_overflow();
}

// C-side q7-pipeline step loop (the full q7 stream plan, bench.py
// bench_q7pipe): per step, two vnode exchange hops + the agg apply + the
// join-left probe; every barrier_every steps the agg flush's change
// stream feeds the join's right side device-resident. The Python
// interpreter costs ~30-40 us/step in ctypes dispatch — at a ~0.2 ms
// step that is 20% overhead, so the loop lives here.
int rw_q7pipe_bench_run(void* agg_h, void* join_h, void** agg_batches,
                        void** join_batches, int n_batches, int steps,
                        int barrier_every, const uint32_t* col_map,
                        int n_map, int step0) {
    auto* agg = (HashAgg*)agg_h;
    auto* j = (HashJoin*)join_h;
    // Long-run state maintenance: the agg change stream retires a right-
    // side record per replaced window max (U- then U+), so dead records
    // accumulate and lengthen the probe walks. Every RW_Q7PIPE_COMPACT
    // barriers (default 4; 0 disables) the loop runs the product cadence:
    // checkpoint-drain the right side (the barrier's commit), then
    // rw_join_compact reclaims the dead records. Costs stay inside the
    // timed region — they are part of running the plan.
    const char* ce = getenv("RW_Q7PIPE_COMPACT");
    int compact_every = ce ? atoi(ce) : 4;
    int barriers_done = 0;
    for (int i = 0; i < steps; i++) {
        int bi = (step0 + i) % n_batches;
        auto* jb = (JoinBatchDev*)join_batches[bi];
        int rc = rw_join_vnode_hop(join_h, jb, 3, RW_T_I64, 256);
        if (rc != RW_OK) return rc;
        rc = agg->apply(*(AggBatch*)agg_batches[bi], true);
        if (rc != RW_OK) return rc;
        rc = rw_join_vnode_hop(join_h, jb, 2, RW_T_I64, 256);
        if (rc != RW_OK) return rc;
        rc = j->probe(RW_SIDE_LEFT, *jb, true, 0, jb->n_rows);
        if (rc != RW_OK) return rc;
        if (barrier_every > 0 && (step0 + i + 1) % barrier_every == 0) {
            long long n = rw_agg_flush_device(agg_h, (uint64_t)(step0 + i));
            if (n < 0) return RW_E_INTERNAL;
            rc = rw_join_apply_aggout(join_h, agg_h, RW_SIDE_RIGHT, col_map,
                                      n_map, (uint64_t)n);
            if (rc != RW_OK) return rc;
            long long e = rw_join_bench_drain(join_h);
            if (e < 0) return RW_E_INTERNAL;
            barriers_done++;
            if (compact_every > 0 && barriers_done % compact_every == 0) {
                std::vector<uint8_t> sp;
                rc = j->checkpoint_drain(RW_SIDE_RIGHT, sp);
                if (rc != RW_OK) return rc;
                j->deg_spill[RW_SIDE_RIGHT].clear();
                rc = rw_join_compact(join_h, RW_SIDE_RIGHT, nullptr);
                if (rc != RW_OK) return rc;
            }
        }
    }
    return RW_OK;
}

int rw_join_kernel_stats(void* h, RwKernelStats* out) {
    auto* j = (HashJoin*)h;
    if (j->ev_harvest_all() != RW_OK)
        FAIL(RW_E_INTERNAL, "event harvest failed");
    out->launches = j->probe_launches;
    out->total_ms = j->probe_ms_total;
    out->rows = j->probe_rows;
    return RW_OK;
}

int rw_join_checkpoint_drain(void* h, int side, uint8_t** buf,
                             uint64_t* len) {
    auto* j = (HashJoin*)h;
    if (side != 0 && side != 1) FAIL(RW_E_INVAL, "bad side");
    std::vector<uint8_t> sp;
    int rc = j->checkpoint_drain(side, sp);
    if (rc != RW_OK) return rc;
    return spill_export(sp, buf, len);
}

int rw_hash_join_restore(void* h, int side, const uint8_t* buf,
                         uint64_t len, const uint8_t* deg_buf,
                         uint64_t deg_len) {
    auto* j = (HashJoin*)h;
    if (side != 0 && side != 1) FAIL(RW_E_INVAL, "bad side");
    std::map<std::string, std::vector<uint8_t>> merged;
    bool ok = rwcodec::for_each_frame(
        buf, len,
        [&](uint8_t put, const uint8_t* k, uint32_t klen, const uint8_t* v,
            uint32_t vlen) {
            std::string key((const char*)k, klen);
            if (put)
                merged[key].assign(v, v + vlen);
            else
                merged.erase(key);
        });
    if (!ok) FAIL(RW_E_INVAL, "malformed spill stream");
    std::map<std::string, uint32_t> degs;
    if (deg_buf && deg_len) {
        ok = rwcodec::for_each_frame(
            deg_buf, deg_len,
            [&](uint8_t put, const uint8_t* k, uint32_t klen,
                const uint8_t* v, uint32_t vlen) {
                std::string key((const char*)k, klen);
                if (put && vlen >= 9 && v[vlen - 9] == 1) {
                    uint64_t d = 0;
                    for (int b = 0; b < 8; b++)
                        d |= (uint64_t)v[vlen - 8 + b] << (8 * b);
                    degs[key] = (uint32_t)d;
                } else if (!put) {
                    degs.erase(key);
                }
            });
        if (!ok) FAIL(RW_E_INVAL, "malformed degree spill stream");
    }
    uint32_t n = (uint32_t)merged.size();
    if (!n) return RW_OK;
    int ncols = j->m.n_cols[side];
    std::vector<std::vector<int64_t>> cols(ncols,
                                           std::vector<int64_t>(n));
    std::vector<std::vector<uint8_t>> valid(ncols,
                                            std::vector<uint8_t>(n));
    std::vector<uint32_t> degrees(n, 0);
    uint32_t i = 0;
    for (auto& [kbytes, val] : merged) {
        size_t off = 0;
        for (int c = 0; c < ncols; c++) {
            rwcodec::DatumC d;
            size_t got = rwcodec::value_decode_datum(
                val.data() + off, val.size() - off, j->types[side][c], &d);
            if (!got) FAIL(RW_E_INVAL, "restore: bad row datum");
            off += got;
            cols[c][i] = d.null ? 0 : d.i;
            valid[c][i] = !d.null;
        }
        auto di = degs.find(kbytes);
        if (di != degs.end()) degrees[i] = di->second;
        i++;
    }
    int rc = j->ensure_stage(side, n);
    if (rc != RW_OK) return rc;
    JoinBatchDev b = j->stage[side];
    for (int c = 0; c < ncols; c++) {
        HIP_TRY(hipMemcpy(b.col_vals[c], cols[c].data(), (size_t)n * 8,
                          hipMemcpyHostToDevice));
        HIP_TRY(hipMemcpy(b.col_valid[c], valid[c].data(), n,
                          hipMemcpyHostToDevice));
    }
    b.vis = nullptr;
    b.n_rows = n;
    uint32_t* ddeg = nullptr;
    HIP_TRY(hipMalloc(&ddeg, (size_t)n * 4));
    HIP_TRY(hipMemcpy(ddeg, degrees.data(), (size_t)n * 4,
                      hipMemcpyHostToDevice));
    uint32_t blocks = (n + 255) / 256;
    if (blocks > 2048) blocks = 2048;
    join_restore_kernel<<<blocks, 256, 0, j->stream>>>(
        b, j->side[side], j->m, side, ddeg, j->out.counters + 1);
    int rcs = hipStreamSynchronize(j->stream) == hipSuccess ? RW_OK
                                                            : RW_E_INTERNAL;
    hipFree(ddeg);
    if (rcs != RW_OK) FAIL(RW_E_INTERNAL, "restore sync failed");
    uint32_t ctr[2];
    HIP_TRY(hipMemcpy(ctr, j->out.counters, 8, hipMemcpyDeviceToHost));
    if (ctr[1]) FAIL(RW_E_INTERNAL, "restore overflow (code %u)", ctr[1]);
    // restored rows predate the epoch: the next drain must not re-PUT them
    uint32_t cur = 0;
    HIP_TRY(hipMemcpy(&cur, j->side[side].row_cursor, 4,
                      hipMemcpyDeviceToHost));
    j->flush_mark[side] = cur;
    return RW_OK;
}

// §8f-4 memory reclamation: rebuild one side's record store and buckets
// without its dead records (the reference reclaims them in Hummock
// compaction; here a maintenance pass the embedder schedules between
// epochs, right after the checkpoint drains). Logical state is
// unchanged — subsequent drains, probes, and restores are unaffected.
int rw_join_compact(void* h, int side, uint64_t* reclaimed) {
    auto* j = (HashJoin*)h;
    if (side != 0 && side != 1) FAIL(RW_E_INVAL, "bad side");
    int rcp = j->pending_flush();
    if (rcp != RW_OK) return rcp;
    HIP_TRY(hipStreamSynchronize(j->stream));
    auto& sd = j->side[side];
    uint32_t cur = 0, kcur = 0, dn = 0;
    HIP_TRY(hipMemcpy(&cur, sd.row_cursor, 4, hipMemcpyDeviceToHost));
    HIP_TRY(hipMemcpy(&kcur, sd.killed_cursor, 4, hipMemcpyDeviceToHost));
    if (sd.deg_dirty_n)
        HIP_TRY(hipMemcpy(&dn, sd.deg_dirty_n, 4, hipMemcpyDeviceToHost));
    if (kcur || dn)
        FAIL(RW_E_INVAL,
             "compact requires drained deltas (checkpoint + degree drain "
             "first: %u kills, %u degree deltas pending)", kcur, dn);
    if (j->flush_mark[side] != cur)
        FAIL(RW_E_INVAL, "compact requires a drained state table "
                         "(%u undrained fresh rows)",
             cur - j->flush_mark[side]);
    if (cur > sd.row_cap) cur = sd.row_cap;
    if (!j->compact_scratch[side])
        HIP_TRY(hipMalloc(&j->compact_scratch[side],
                          (size_t)sd.row_cap * sd.row_stride));
    uint8_t* nrows = j->compact_scratch[side];
    uint32_t* ncur = nullptr;
    HIP_TRY(hipMalloc(&ncur, 4));
    HIP_TRY(hipMemsetAsync(ncur, 0, 4, j->stream));
    jcompact_copy_kernel<<<4096, 256, 0, j->stream>>>(sd, nrows, ncur, cur);
    HIP_TRY(hipMemsetAsync(sd.slots8, 0,
                           ((size_t)sd.cap_mask + 1) * 8, j->stream));
    int rcs = hipStreamSynchronize(j->stream) == hipSuccess ? RW_OK
                                                            : RW_E_INTERNAL;
    uint32_t n_alive = 0;
    if (rcs == RW_OK)
        hipMemcpy(&n_alive, ncur, 4, hipMemcpyDeviceToHost);
    hipFree(ncur);
    if (rcs != RW_OK) FAIL(RW_E_INTERNAL, "compact copy failed");
    j->compact_scratch[side] = sd.rows; // ping-pong
    sd.rows = nrows;
    HIP_TRY(hipMemcpy(sd.row_cursor, &n_alive, 4, hipMemcpyHostToDevice));
    jcompact_link_kernel<<<4096, 256, 0, j->stream>>>(sd, j->m, side,
                                                      n_alive);
    if (hipStreamSynchronize(j->stream) != hipSuccess)
        FAIL(RW_E_INTERNAL, "compact relink failed");
    j->flush_mark[side] = n_alive;
    if (reclaimed)
        *reclaimed = (uint64_t)(cur - n_alive) * sd.row_stride;
    return RW_OK;
}

int rw_join_degree_drain(void* h, int side, uint8_t** buf, uint64_t* len) {
    auto* j = (HashJoin*)h;
    if (side != 0 && side != 1) FAIL(RW_E_INVAL, "bad side");
    auto& sp = j->deg_spill[side];
    int rc = spill_export(sp, buf, len);
    if (rc != RW_OK) return rc;
    sp.clear();
    return RW_OK;
}

int rw_join_stats_reset(void* h) {
    auto* j = (HashJoin*)h;
    j->ev_harvest_all();
    j->probe_launches = 0;
    j->probe_ms_total = 0;
    j->probe_rows = 0;
    return RW_OK;
}

} // extern "C"

// ===================== GroupTopN =====================
//
// GPU restatement of stream/src/executor/top_n/group_top_n.rs (WITH_TIES =
// false). Per group the full row set lives in HBM (AoS slot + packed
// records, shared layout with the join side); the observable semantics are
// those of TopNCache<false> (top_n_cache.rs:293-520): the visible window is
// rows [offset, offset+limit) of the group in cache-key order, and each
// pushed chunk emits the ChangeBuffer-compacted window delta
// (change_buffer.rs:76-187: replace = U-pair, equal rows cancel).
//
// Kernel plan per push (stream-ordered):
//   1. touch    — find-or-insert group slots, collect the touched list
//   2. snapshot — per touched group, select the pre-chunk window (row ids)
//   3. apply    — inserts (upsert = kill old + append) / deletes; same-
//                 cache-key conflicts within the chunk run in single-row
//                 segments (host pre-pass, as in the agg/join executors)
//   4. emit     — per touched group, select the new window, merge-diff
//                 against the snapshot, reserve + write output rows
// Parallelism is across groups (2+4) and rows (1+3). Records are never
// overwritten in place, so snapshot row ids stay readable in step 4.

#define TOPN_MAX_CK 8
#define TOPN_MAX_WIN 128

struct TopMeta {
    int KW;     // group key width
    int n_cols; // input schema width
    uint8_t gk_cols[MAX_KW];
    uint8_t gk_float[MAX_KW]; // float group cols: canonicalize key words
                              // (-0.0 -> +0.0, NaN -> one pattern) so the
                              // raw-bit slot compare matches SQL equality
                              // (the reference's HashKey normalizes,
                              // common/src/hash/key.rs:517)
    int n_ck; // cache key = order_by then rest storage-key cols
    uint8_t ck_cols[TOPN_MAX_CK];
    uint8_t ck_desc[TOPN_MAX_CK];
    uint8_t ck_float[TOPN_MAX_CK];
    uint32_t offset, limit;
    uint8_t with_ties; // TopNCache<true>: window extends past limit while
                       // rows tie the limit-th row's order-by prefix
    int n_order;       // cache-key prefix length that defines a tie
};

// cache-key compare on packed records (ordered_cmp semantics: NULLs
// largest in value order, desc reverses, NaN largest among floats)
__device__ __forceinline__ int topn_cmp(const TopMeta& m, const long long* va,
                                        uint32_t nba, const long long* vb,
                                        uint32_t nbb, int ncols) {
    for (int i = 0; i < ncols; i++) {
        int c = m.ck_cols[i];
        bool na = !((nba >> c) & 1), nb = !((nbb >> c) & 1);
        int r;
        if (na || nb) {
            r = (na && nb) ? 0 : (na ? 1 : -1);
        } else if (m.ck_float[i]) {
            double da = __longlong_as_double(va[c]);
            double db = __longlong_as_double(vb[c]);
            bool an = isnan(da), bn = isnan(db);
            if (an || bn) r = an == bn ? 0 : (an ? 1 : -1);
            else r = da < db ? -1 : (da > db ? 1 : 0);
        } else {
            r = va[c] < vb[c] ? -1 : (va[c] > vb[c] ? 1 : 0);
        }
        if (m.ck_desc[i]) r = -r;
        if (r) return r;
    }
    return 0;
}

__device__ __forceinline__ bool topn_ck_eq_batch(const TopMeta& m,
                                                 const JoinBatchDev& b,
                                                 uint32_t r, const long long* v,
                                                 uint32_t nb) {
    for (int i = 0; i < m.n_ck; i++) {
        int c = m.ck_cols[i];
        bool bn = !b.col_valid[c][r];
        bool rn = !((nb >> c) & 1);
        if (bn != rn) return false;
        if (!bn && b.col_vals[c][r] != v[c]) return false;
    }
    return true;
}

// canonicalize a float group-key word: -0.0 -> +0.0, every NaN -> the
// canonical quiet pattern — SQL equality for the raw-bit slot compare
__device__ __forceinline__ int64_t fkey_canon(int64_t w) {
    double d = __longlong_as_double((long long)w);
    if (d == 0.0) return 0;
    if (d != d) return 0x7ff8000000000000LL;
    return w;
}

__device__ __forceinline__ int64_t topn_gkw(const TopMeta& m, int i,
                                            int64_t v) {
    return m.gk_float[i] ? fkey_canon(v) : v;
}

__global__ void topn_touch_kernel(JoinBatchDev b, JoinSideDev sd, TopMeta m,
                                  uint32_t* touched, uint32_t* touched_list,
                                  uint32_t* counters /*0=tcur 1=err*/) {
    uint32_t stride = gridDim.x * blockDim.x;
    for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < b.n_rows;
         r += stride) {
        if (b.vis && !b.vis[r]) continue;
        int64_t kw[MAX_KW];
        uint32_t nullmask = 0;
        for (int i = 0; i < m.KW; i++) {
            uint8_t col = m.gk_cols[i];
            bool valid = b.col_valid[col][r];
            kw[i] = valid ? topn_gkw(m, i, b.col_vals[col][r]) : 0;
            nullmask |= (!valid) << i;
        }
        uint32_t slot =
            jslot_find_or_insert(sd.slots, sd.cap_mask, kw, nullmask, m.KW);
        if (slot == UINT32_MAX) {
            atomicExch(&counters[1], 2u);
            continue;
        }
        if (ld_u32(&touched[slot]) == 0 &&
            atomicCAS(&touched[slot], 0u, 1u) == 0u)
            touched_list[atomicAdd(&counters[0], 1u)] = slot;
    }
}

// select the first (offset+limit) rows of a chain in cache-key order;
// returns count. O(chain * K) insertion selection — the window is small.
__device__ int topn_select(const JoinSideDev& sd, const TopMeta& m,
                           uint32_t slot, uint32_t* sel, int K) {
    int n = 0;
    uint32_t row = sd.slots[slot].head;
    while (row != UINT32_MAX) {
        JoinRowHdr* h = jrow(sd, row);
        if (h->alive) {
            const long long* v = jvals(h);
            int pos = n;
            while (pos > 0) {
                JoinRowHdr* hp = jrow(sd, sel[pos - 1]);
                if (topn_cmp(m, jvals(hp), hp->validbits, v, h->validbits,
                             m.n_ck) <= 0)
                    break;
                pos--;
            }
            if (pos < K) {
                int last = n < K ? n : K - 1;
                for (int k = last; k > pos; k--) sel[k] = sel[k - 1];
                sel[pos] = row;
                if (n < K) n++;
            }
        }
        row = h->next;
    }
    return n;
}

// full visible window incl. WITH TIES extension (ties on the order-by
// prefix of the limit-th row, TopNCache<true>; offset == 0 with ties).
// Safe-overflow argument: sel is sorted by full cache key and the sort
// prefix is a key prefix, so prefixes are non-decreasing along sel and any
// row dropped by the bounded selection is ≥ the last kept row — once the
// tie trim stops before the buffer end, no dropped row can tie. Only a tie
// group consuming the ENTIRE buffer is unrepresentable → error 5.
__device__ int topn_window(const JoinSideDev& sd, const TopMeta& m,
                           uint32_t slot, uint32_t* sel, uint32_t* err) {
    if (!m.with_ties)
        return topn_select(sd, m, slot, sel, (int)(m.offset + m.limit));
    int n = topn_select(sd, m, slot, sel, TOPN_MAX_WIN);
    if (n <= (int)m.limit) return n;
    JoinRowHdr* hc = jrow(sd, sel[m.limit - 1]);
    int nw = (int)m.limit;
    while (nw < n) {
        JoinRowHdr* h = jrow(sd, sel[nw]);
        if (topn_cmp(m, jvals(h), h->validbits, jvals(hc), hc->validbits,
                     m.n_order) != 0)
            break;
        nw++;
    }
    if (nw == TOPN_MAX_WIN) atomicExch(err, 5u); // ties window overflow
    return nw;
}

__global__ void topn_snapshot_kernel(JoinSideDev sd, TopMeta m,
                                     const uint32_t* touched_list,
                                     const uint32_t* counters,
                                     uint32_t* old_win, uint32_t* old_n,
                                     uint32_t* err) {
    int K = m.with_ties ? TOPN_MAX_WIN : (int)(m.offset + m.limit);
    uint32_t nt = counters[0];
    uint32_t stride = gridDim.x * blockDim.x;
    for (uint32_t t = blockIdx.x * blockDim.x + threadIdx.x; t < nt;
         t += stride) {
        uint32_t sel[TOPN_MAX_WIN];
        int n = topn_window(sd, m, touched_list[t], sel, err);
        // visible window = positions [offset, offset+limit) (+ ties)
        int w0 = n < (int)m.offset ? n : (int)m.offset;
        old_n[t] = (uint32_t)(n - w0);
        for (int i = w0; i < n; i++)
            old_win[(size_t)t * K + (i - w0)] = sel[i];
    }
}

__global__ void topn_apply_kernel(JoinBatchDev b, JoinSideDev sd, TopMeta m,
                                  uint32_t r0, uint32_t r1,
                                  uint32_t* counters /*1=err*/) {
    uint32_t stride = gridDim.x * blockDim.x;
    for (uint32_t r = r0 + blockIdx.x * blockDim.x + threadIdx.x; r < r1;
         r += stride) {
        if (b.vis && !b.vis[r]) continue;
        int64_t kw[MAX_KW];
        uint32_t nullmask = 0;
        for (int i = 0; i < m.KW; i++) {
            uint8_t col = m.gk_cols[i];
            bool valid = b.col_valid[col][r];
            kw[i] = valid ? topn_gkw(m, i, b.col_vals[col][r]) : 0;
            nullmask |= (!valid) << i;
        }
        uint32_t slot =
            jslot_find_cached(sd.slots, sd.cap_mask, kw, nullmask, m.KW);
        if (slot == UINT32_MAX) continue; // touch pass created it
        uint8_t op = b.ops[r];
        bool is_insert = (op == RW_OP_INSERT || op == RW_OP_UPDATE_INSERT);
        // find the alive row with this cache key (sc1 loads: the chain
        // mutates within this launch; same-ck rows run in their own
        // segments, so the kill/append below is race-free)
        uint32_t row = ld_u32(&sd.slots[slot].head);
        while (row != UINT32_MAX) {
            JoinRowHdr* h = jrow(sd, row);
            uint32_t vb = ld_u32(&h->validbits);
            if (ld_u32(&h->alive) &&
                topn_ck_eq_batch(m, b, r, jvals(h), vb)) {
                st_u32(&h->alive, 0); // upsert kills the old record
                if (sd.killed) {
                    uint32_t kidx = atomicAdd(sd.killed_cursor, 1u);
                    if (kidx < sd.killed_cap) sd.killed[kidx] = row;
                }
                break;
            }
            row = ld_u32(&h->next);
        }
        if (!is_insert) continue;
        // wave-aggregated reservation (single hot counter, see jown_insert)
        uint64_t wmask = __ballot(true);
        int lane = threadIdx.x & 63;
        int leader = 63 - __clzll(wmask);
        uint32_t base = 0;
        if (lane == leader)
            base = atomicAdd(sd.row_cursor, (uint32_t)__popcll(wmask));
        base = (uint32_t)__shfl((int)base, leader);
        uint32_t nrow = base + (uint32_t)__popcll(wmask & ((1ULL << lane) - 1));
        if (nrow >= sd.row_cap) {
            atomicExch(&counters[1], 3u);
            continue;
        }
        JoinRowHdr* h = jrow(sd, nrow);
        uint32_t vb = 0;
        long long* hv = jvals(h);
        for (int c = 0; c < m.n_cols; c++) {
            st_i64((int64_t*)&hv[c], b.col_vals[c][r]);
            vb |= (uint32_t)(b.col_valid[c][r] != 0) << c;
        }
        st_u32(&h->validbits, vb);
        st_u32(&h->alive, 1);
        uint32_t* headp = &sd.slots[slot].head;
        uint32_t old_head = ld_u32(headp);
        for (;;) {
            st_u32(&h->next, old_head);
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory"); // R1 drain
            uint32_t prev = atomicCAS(headp, old_head, nrow);
            if (prev == old_head) break;
            old_head = prev;
        }
    }
}

__global__ void topn_emit_kernel(JoinSideDev sd, TopMeta m,
                                 const uint32_t* touched_list,
                                 const uint32_t* tcounters, uint32_t* touched,
                                 const uint32_t* old_win,
                                 const uint32_t* old_n, JoinOutDev out,
                                 int n_cols, uint32_t* err) {
    int K = m.with_ties ? TOPN_MAX_WIN : (int)(m.offset + m.limit);
    uint32_t nt = tcounters[0];
    uint32_t stride = gridDim.x * blockDim.x;
    for (uint32_t t = blockIdx.x * blockDim.x + threadIdx.x; t < nt;
         t += stride) {
        uint32_t slot = touched_list[t];
        touched[slot] = 0;
        uint32_t selbuf[TOPN_MAX_WIN];
        int nsel = topn_window(sd, m, slot, selbuf, err);
        // visible window = positions [offset, offset+limit)
        int w0 = nsel < (int)m.offset ? nsel : (int)m.offset;
        const uint32_t* neww = selbuf + w0;
        int nn = nsel - w0;
        const uint32_t* oldw = old_win + (size_t)t * K;
        int no = (int)old_n[t];
        // merge-diff into local delta lists: op + row id (+ paired old id)
        uint8_t dop[2 * TOPN_MAX_WIN];
        uint32_t drow[2 * TOPN_MAX_WIN];
        int nd = 0;
        int i = 0, j = 0;
        while (i < no || j < nn) {
            int c;
            if (i >= no) c = 1;
            else if (j >= nn) c = -1;
            else {
                JoinRowHdr* ho = jrow(sd, oldw[i]);
                JoinRowHdr* hn = jrow(sd, neww[j]);
                c = topn_cmp(m, jvals(ho), ho->validbits, jvals(hn),
                             hn->validbits, m.n_ck);
            }
            if (c == 0) {
                if (oldw[i] != neww[j]) {
                    // same cache key, different record: replaced. Equal
                    // rows cancel (ChangeBuffer no-op filter).
                    JoinRowHdr* ho = jrow(sd, oldw[i]);
                    JoinRowHdr* hn = jrow(sd, neww[j]);
                    bool eq = ho->validbits == hn->validbits;
                    const long long* vo = jvals(ho);
                    const long long* vn = jvals(hn);
                    for (int cc = 0; eq && cc < n_cols; cc++)
                        if (((ho->validbits >> cc) & 1) && vo[cc] != vn[cc])
                            eq = false;
                    if (!eq) {
                        dop[nd] = RW_OP_UPDATE_DELETE;
                        drow[nd++] = oldw[i];
                        dop[nd] = RW_OP_UPDATE_INSERT;
                        drow[nd++] = neww[j];
                    }
                }
                i++;
                j++;
            } else if (c < 0) {
                dop[nd] = RW_OP_DELETE;
                drow[nd++] = oldw[i];
                i++;
            } else {
                dop[nd] = RW_OP_INSERT;
                drow[nd++] = neww[j];
                j++;
            }
        }
        if (!nd) continue;
        uint32_t base = atomicAdd(&out.counters[0], (uint32_t)nd);
        if (base + nd > out.cap) {
            atomicExch(&out.counters[1], 1u);
            continue;
        }
        for (int k = 0; k < nd; k++) {
            uint32_t orow = base + k;
            JoinRowHdr* h = jrow(sd, drow[k]);
            const long long* v = jvals(h);
            out.ops[orow] = dop[k];
            for (int c = 0; c < n_cols; c++) {
                bool valid = (h->validbits >> c) & 1;
                out.vals[(size_t)orow * n_cols + c] = valid ? v[c] : 0;
                out.nulls[(size_t)orow * n_cols + c] = !valid;
            }
        }
    }
}

// GroupTopN watermark state cleaning (group_top_n.rs:266-273: a watermark
// on group_by[0] becomes a state-table watermark = range delete on the
// store). Rows of groups below the watermark are retired in place and
// routed through the kill list so the next drain nets them to DELETE
// frames (slots stay READY; a late row restarts the group).
__global__ void topn_clean_kernel(JoinSideDev sd, int kpos, long long wm) {
    size_t cap = (size_t)sd.cap_mask + 1;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (size_t slot = blockIdx.x * blockDim.x + threadIdx.x; slot < cap;
         slot += stride) {
        if (sd.slots[slot].state != SLOT_READY) continue;
        if ((sd.slots[slot].nulls >> kpos) & 1) continue; // NULLs largest
        if (sd.slots[slot].key[kpos] >= wm) continue;
        uint32_t row = sd.slots[slot].head;
        while (row != UINT32_MAX) {
            JoinRowHdr* h = jrow(sd, row);
            if (h->alive) {
                h->alive = 0;
                if (sd.killed) {
                    uint32_t kidx = atomicAdd(sd.killed_cursor, 1u);
                    if (kidx < sd.killed_cap) sd.killed[kidx] = row;
                }
            }
            row = h->next;
        }
    }
}

// §8f-4 memory reclamation (TopN): relink packed records into freshly
// initialized group slots (jcompact_copy_kernel packs; this rebuilds the
// chains — netted keys need not be unique here, concurrent pushes on one
// group resolve by CAS).
__global__ void topn_relink_kernel(JoinSideDev sd, TopMeta m, uint32_t n,
                                   uint32_t* err) {
    uint32_t stride = gridDim.x * blockDim.x;
    for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < n;
         r += stride) {
        JoinRowHdr* h = jrow(sd, r);
        uint32_t vb = h->validbits;
        int64_t kw[MAX_KW];
        uint32_t nm = 0;
        for (int i = 0; i < m.KW; i++) {
            uint8_t col = m.gk_cols[i];
            bool valid = (vb >> col) & 1;
            kw[i] = valid ? topn_gkw(m, i, jvals(h)[col]) : 0;
            nm |= (uint32_t)(!valid) << i;
        }
        uint32_t slot =
            jslot_find_or_insert(sd.slots, sd.cap_mask, kw, nm, m.KW);
        if (slot == UINT32_MAX) {
            atomicExch(err, 2u);
            continue;
        }
        uint32_t* headp = &sd.slots[slot].head;
        uint32_t old_head = ld_u32(headp);
        for (;;) {
            st_u32(&h->next, old_head);
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory"); // R1 drain
            uint32_t prev = atomicCAS(headp, old_head, r);
            if (prev == old_head) break;
            old_head = prev;
        }
    }
}

// §8f-5 recovery: rebuild the TopN state table from a netted spill replay.
// Each restored row is appended to the record store and chained onto its
// group's slot (cache keys in the netted map are unique, so no same-key
// conflicts; concurrent head pushes on one group resolve by CAS).
__global__ void topn_restore_kernel(JoinBatchDev b, JoinSideDev sd, TopMeta m,
                                    uint32_t* err) {
    uint32_t stride = gridDim.x * blockDim.x;
    int lane = threadIdx.x & 63;
    uint32_t n = b.n_rows;
    uint32_t iters = (n + stride - 1) / stride;
    for (uint32_t it = 0; it < iters; it++) {
        uint32_t r = it * stride + blockIdx.x * blockDim.x + threadIdx.x;
        bool active = r < n;
        uint64_t wmask = __ballot(active);
        if (!active) continue;
        int leader = 63 - __clzll(wmask);
        uint32_t base = 0;
        if (lane == leader)
            base = atomicAdd(sd.row_cursor, (uint32_t)__popcll(wmask));
        base = (uint32_t)__shfl((int)base, leader);
        uint32_t row = base + (uint32_t)__popcll(wmask & ((1ULL << lane) - 1));
        if (row >= sd.row_cap) {
            atomicExch(err, 3u);
            continue;
        }
        JoinRowHdr* h = jrow(sd, row);
        long long* hv = jvals(h);
        uint32_t vb = 0;
        for (int c = 0; c < m.n_cols; c++) {
            hv[c] = b.col_vals[c][r];
            vb |= (uint32_t)(b.col_valid[c][r] != 0) << c;
        }
        h->validbits = vb;
        h->degree = 0;
        h->alive = 1;
        int64_t kw[MAX_KW];
        uint32_t nm = 0;
        for (int i = 0; i < m.KW; i++) {
            uint8_t col = m.gk_cols[i];
            bool valid = (vb >> col) & 1;
            kw[i] = valid ? topn_gkw(m, i, hv[col]) : 0;
            nm |= (uint32_t)(!valid) << i;
        }
        uint32_t slot =
            jslot_find_or_insert(sd.slots, sd.cap_mask, kw, nm, m.KW);
        if (slot == UINT32_MAX) {
            atomicExch(err, 2u);
            continue;
        }
        uint32_t* headp = &sd.slots[slot].head;
        uint32_t old_head = ld_u32(headp);
        for (;;) {
            st_u32(&h->next, old_head);
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory"); // R1 drain
            uint32_t prev = atomicCAS(headp, old_head, row);
            if (prev == old_head) break;
            old_head = prev;
        }
    }
}

struct GroupTopN {
    TopMeta m{};
    JoinSideDev sd{};
    JoinBatchDev stage{};
    uint32_t stage_cap = 0;
    JoinOutDev out{};
    uint32_t* touched = nullptr;     // [slot cap]
    uint32_t* touched_list = nullptr;
    uint32_t* tcounters = nullptr; // 0=tcursor 1=err
    uint32_t* old_win = nullptr;
    uint32_t* old_n = nullptr;
    uint32_t old_cap = 0; // touched capacity for old_win/old_n
    std::vector<uint8_t> types;
    std::vector<uint32_t> ck_col_idx;
    uint32_t chunk_size;
    hipStream_t stream = nullptr;
    std::vector<RwChunk*> outq;

    int init(const RwGroupTopNDesc* d) {
        if (!gpu_ok())
            FAIL(RW_E_NOGPU,
                 "risingwave_amd: no GPU visible (product path has no CPU fallback)");
        if (!d->limit) FAIL(RW_E_INVAL, "limit must be > 0");
        if (d->offset + d->limit > TOPN_MAX_WIN)
            FAIL(RW_E_INVAL, "offset+limit > %d unsupported", TOPN_MAX_WIN);
        if (d->n_group_by > MAX_KW)
            FAIL(RW_E_INVAL, "group_by width 0..%d", MAX_KW);
        // n_group_by == 0 = the plain TopN executor (top_n_plain.rs): every
        // row lands in the single empty-key group
        if (d->n_order_by + d->n_rest > TOPN_MAX_CK)
            FAIL(RW_E_INVAL, "cache key width > %d", TOPN_MAX_CK);
        if (d->n_cols > MAX_COLS) FAIL(RW_E_INVAL, "n_cols > %d", MAX_COLS);
        types.assign(d->types, d->types + d->n_cols);
        for (auto t : types)
            if (t != RW_T_I64 && t != RW_T_TS && t != RW_T_F64)
                FAIL(RW_E_INVAL, "GroupTopN supports 8-byte datum columns");
        m.KW = (int)d->n_group_by;
        m.n_cols = (int)d->n_cols;
        for (uint32_t i = 0; i < d->n_group_by; i++) {
            m.gk_cols[i] = (uint8_t)d->group_by[i];
            m.gk_float[i] = types[d->group_by[i]] == RW_T_F64;
        }
        m.n_ck = (int)(d->n_order_by + d->n_rest);
        int k = 0;
        for (uint32_t i = 0; i < d->n_order_by; i++, k++) {
            m.ck_cols[k] = (uint8_t)d->order_cols[i];
            m.ck_desc[k] = d->order_desc[i];
            m.ck_float[k] = types[d->order_cols[i]] == RW_T_F64;
            ck_col_idx.push_back(d->order_cols[i]);
        }
        for (uint32_t i = 0; i < d->n_rest; i++, k++) {
            m.ck_cols[k] = (uint8_t)d->rest_cols[i];
            m.ck_desc[k] = d->rest_desc[i];
            m.ck_float[k] = types[d->rest_cols[i]] == RW_T_F64;
            ck_col_idx.push_back(d->rest_cols[i]);
        }
        m.offset = (uint32_t)d->offset;
        m.limit = (uint32_t)d->limit;
        m.with_ties = d->with_ties;
        m.n_order = (int)d->n_order_by;
        if (d->with_ties && d->offset)
            FAIL(RW_E_INVAL, "offset unsupported WITH TIES (as in the reference)");
        chunk_size = d->chunk_size ? d->chunk_size : 1024;

        uint64_t cap_hint = d->state_capacity_hint ? d->state_capacity_hint
                                                   : (1ull << 14);
        size_t cap = 1;
        while (cap < cap_hint * 2) cap <<= 1;
        sd.cap_mask = (uint32_t)(cap - 1);
        uint64_t row_cap = d->row_capacity_hint ? d->row_capacity_hint
                                                : (1ull << 20);
        sd.row_stride = 16 + 8u * (uint32_t)m.n_cols;
        sd.row_cap = (uint32_t)row_cap;
        HIP_TRY(hipStreamCreate(&stream));
        HIP_TRY(hipMalloc(&sd.slots, cap * sizeof(JoinSlot)));
        HIP_TRY(hipMalloc(&sd.rows, (size_t)row_cap * sd.row_stride));
        HIP_TRY(hipMalloc(&sd.row_cursor, 4));
        HIP_TRY(hipMemset(sd.row_cursor, 0, 4));
        sd.killed_cap = 1u << 22; // checkpoint-delta tracking (§8f-2)
        HIP_TRY(hipMalloc(&sd.killed, (size_t)sd.killed_cap * 4));
        HIP_TRY(hipMalloc(&sd.killed_cursor, 4));
        HIP_TRY(hipMemset(sd.killed_cursor, 0, 4));
        HIP_TRY(hipMalloc(&touched, cap * 4));
        HIP_TRY(hipMemset(touched, 0, cap * 4));
        HIP_TRY(hipMalloc(&tcounters, 8));
        jslot_init_kernel<<<2048, 256, 0, stream>>>(sd.slots, cap);
        uint32_t out_cap = 1 << 18;
        out.cap = out_cap;
        HIP_TRY(hipMalloc(&out.vals, (size_t)out_cap * m.n_cols * 8));
        HIP_TRY(hipMalloc(&out.nulls, (size_t)out_cap * m.n_cols));
        HIP_TRY(hipMalloc(&out.ops, out_cap));
        HIP_TRY(hipMalloc(&out.counters, 8));
        HIP_TRY(hipMemset(out.counters, 0, 8));
        HIP_TRY(hipStreamSynchronize(stream));
        return RW_OK;
    }

    ~GroupTopN() {
        if (sd.slots) {
            hipFree(sd.slots);
            hipFree(sd.rows);
            hipFree(sd.row_cursor);
            hipFree(sd.killed);
            hipFree(sd.killed_cursor);
            hipFree(touched);
            hipFree(tcounters);
            hipFree(out.vals);
            hipFree(out.nulls);
            hipFree(out.ops);
            hipFree(out.counters);
        }
        if (touched_list) hipFree(touched_list);
        if (old_win) hipFree(old_win);
        if (compact_scratch) hipFree(compact_scratch);
        free_stage();
        if (stream) hipStreamDestroy(stream);
        for (auto* c : outq) {
            for (uint32_t i = 0; i < c->n_cols; i++) {
                delete[] (int64_t*)c->cols[i].data;
                delete[] (uint8_t*)c->cols[i].valid;
            }
            delete[] c->cols;
            delete[] c->ops;
            delete c;
        }
    }

    void free_stage() {
        if (!stage_cap) return;
        for (int i = 0; i < m.n_cols; i++) {
            hipFree(stage.col_vals[i]);
            hipFree(stage.col_valid[i]);
        }
        hipFree(stage.ops);
        hipFree(stage.vis);
        stage_cap = 0;
    }

    int ensure_caps(uint32_t n) {
        if (stage_cap < n) {
            free_stage();
            for (int i = 0; i < m.n_cols; i++) {
                HIP_TRY(hipMalloc(&stage.col_vals[i], (size_t)n * 8));
                HIP_TRY(hipMalloc(&stage.col_valid[i], n));
            }
            HIP_TRY(hipMalloc(&stage.ops, n));
            HIP_TRY(hipMalloc(&stage.vis, n));
            stage_cap = n;
        }
        if (old_cap < n) {
            if (touched_list) hipFree(touched_list);
            if (old_win) hipFree(old_win);
            if (old_n) hipFree(old_n);
            int K = m.with_ties ? TOPN_MAX_WIN : (int)(m.offset + m.limit);
            HIP_TRY(hipMalloc(&touched_list, (size_t)n * 4));
            HIP_TRY(hipMalloc(&old_win, (size_t)n * K * 4));
            HIP_TRY(hipMalloc(&old_n, (size_t)n * 4));
            old_cap = n;
        }
        return RW_OK;
    }

    // same-cache-key rows within one chunk run in their own single-row
    // segments (the reference applies rows in order, group_top_n.rs:168)
    std::vector<uint32_t> conflict_segments(const RwChunk* c) {
        std::vector<uint32_t> bounds;
        std::unordered_map<std::string, int> ck_count;
        auto ck_of = [&](uint32_t r) {
            std::string k;
            for (auto col : ck_col_idx) {
                uint8_t valid = c->cols[col].valid[r];
                k.push_back((char)valid);
                int64_t v = valid ? ((const int64_t*)c->cols[col].data)[r] : 0;
                k.append((const char*)&v, 8);
            }
            return k;
        };
        for (uint32_t r = 0; r < c->n_rows; r++) {
            if (c->vis && !c->vis[r]) continue;
            ck_count[ck_of(r)]++;
        }
        bool any = false;
        for (auto& kv : ck_count)
            if (kv.second > 1) {
                any = true;
                break;
            }
        if (!any) return bounds;
        for (uint32_t r = 0; r < c->n_rows; r++) {
            if (c->vis && !c->vis[r]) continue;
            if (ck_count[ck_of(r)] > 1) {
                bounds.push_back(r);
                bounds.push_back(r + 1);
            }
        }
        return bounds;
    }

    int push_chunk(const RwChunk* c) {
        uint32_t n = c->n_rows;
        if (!n) return RW_OK;
        if (ensure_caps(n) != RW_OK) return RW_E_INTERNAL;
        JoinBatchDev b = stage;
        for (int i = 0; i < m.n_cols; i++) {
            HIP_TRY(hipMemcpyAsync(b.col_vals[i], c->cols[i].data,
                                   (size_t)n * 8, hipMemcpyHostToDevice,
                                   stream));
            HIP_TRY(hipMemcpyAsync(b.col_valid[i], c->cols[i].valid, n,
                                   hipMemcpyHostToDevice, stream));
        }
        HIP_TRY(hipMemcpyAsync(b.ops, c->ops, n, hipMemcpyHostToDevice,
                               stream));
        if (c->vis)
            HIP_TRY(hipMemcpyAsync(b.vis, c->vis, n, hipMemcpyHostToDevice,
                                   stream));
        else
            b.vis = nullptr;
        b.n_rows = n;
        HIP_TRY(hipMemsetAsync(tcounters, 0, 8, stream));

        uint32_t blocks = (n + 255) / 256;
        if (blocks > 2048) blocks = 2048;
        topn_touch_kernel<<<blocks, 256, 0, stream>>>(b, sd, m, touched,
                                                      touched_list, tcounters);
        topn_snapshot_kernel<<<256, 256, 0, stream>>>(
            sd, m, touched_list, tcounters, old_win, old_n, tcounters + 1);
        auto segs = conflict_segments(c);
        uint32_t start = 0;
        auto launch_apply = [&](uint32_t a, uint32_t z) {
            if (z <= a) return;
            uint32_t bl = (z - a + 255) / 256;
            if (bl > 2048) bl = 2048;
            topn_apply_kernel<<<bl, 256, 0, stream>>>(b, sd, m, a, z,
                                                      tcounters);
        };
        for (uint32_t bnd : segs) {
            launch_apply(start, bnd);
            start = bnd;
        }
        launch_apply(start, n);
        topn_emit_kernel<<<256, 256, 0, stream>>>(sd, m, touched_list,
                                                  tcounters, touched, old_win,
                                                  old_n, out, m.n_cols,
                                                  tcounters + 1);
        HIP_TRY(hipStreamSynchronize(stream));
        uint32_t tc[2];
        HIP_TRY(hipMemcpy(tc, tcounters, 8, hipMemcpyDeviceToHost));
        if (tc[1] == 2) FAIL(RW_E_INTERNAL, "group table full");
        if (tc[1] == 3) FAIL(RW_E_INTERNAL, "row store full");
        if (tc[1] == 5)
            FAIL(RW_E_INTERNAL, "WITH TIES window exceeds %d rows", TOPN_MAX_WIN);
        return drain_output();
    }

    int drain_output() {
        uint32_t ctr[2];
        HIP_TRY(hipMemcpy(ctr, out.counters, 8, hipMemcpyDeviceToHost));
        if (ctr[1] == 1) FAIL(RW_E_INTERNAL, "topn output overflow");
        uint32_t n_out = ctr[0];
        if (n_out) {
            std::vector<int64_t> vals((size_t)n_out * m.n_cols);
            std::vector<uint8_t> nulls((size_t)n_out * m.n_cols);
            std::vector<uint8_t> ops(n_out);
            HIP_TRY(hipMemcpy(vals.data(), out.vals, vals.size() * 8,
                              hipMemcpyDeviceToHost));
            HIP_TRY(hipMemcpy(nulls.data(), out.nulls, nulls.size(),
                              hipMemcpyDeviceToHost));
            HIP_TRY(hipMemcpy(ops.data(), out.ops, n_out,
                              hipMemcpyDeviceToHost));
            uint32_t max_rows = chunk_size < 2 ? 2 : chunk_size;
            uint32_t s = 0;
            while (s < n_out) {
                uint32_t take = n_out - s;
                if (take > max_rows) take = max_rows;
                // U-pair adjacency: never split a U-/U+ pair at the edge
                if (s + take < n_out &&
                    ops[s + take - 1] == RW_OP_UPDATE_DELETE)
                    take++;
                auto* ch = new RwChunk();
                auto* cols = new RwColumn[m.n_cols];
                auto* o = new uint8_t[take];
                memcpy(o, ops.data() + s, take);
                for (int ci = 0; ci < m.n_cols; ci++) {
                    auto* data = new int64_t[take];
                    auto* valid = new uint8_t[take];
                    for (uint32_t r = 0; r < take; r++) {
                        data[r] = vals[(size_t)(s + r) * m.n_cols + ci];
                        valid[r] = !nulls[(size_t)(s + r) * m.n_cols + ci];
                    }
                    cols[ci].type = types[ci];
                    cols[ci].valid = valid;
                    cols[ci].data = data;
                }
                ch->n_rows = take;
                ch->n_cols = (uint32_t)m.n_cols;
                ch->ops = o;
                ch->vis = nullptr;
                ch->cols = cols;
                outq.push_back(ch);
                s += take;
            }
        }
        HIP_TRY(hipMemset(out.counters, 0, 8));
        return RW_OK;
    }

    // §8f-4 memory reclamation: pack alive records, reinit the slot
    // table, relink chains. Requires a drained table (kill list consumed,
    // flush mark at the cursor); logical state is unchanged.
    uint8_t* compact_scratch = nullptr;
    int compact(uint64_t* reclaimed) {
        HIP_TRY(hipStreamSynchronize(stream));
        uint32_t cur = 0, kcur = 0;
        HIP_TRY(hipMemcpy(&cur, sd.row_cursor, 4, hipMemcpyDeviceToHost));
        HIP_TRY(hipMemcpy(&kcur, sd.killed_cursor, 4,
                          hipMemcpyDeviceToHost));
        if (kcur)
            FAIL(RW_E_INVAL,
                 "compact requires a drained kill list (%u pending)", kcur);
        if (flush_mark != cur)
            FAIL(RW_E_INVAL, "compact requires a drained state table "
                             "(%u undrained fresh rows)", cur - flush_mark);
        if (cur > sd.row_cap) cur = sd.row_cap;
        if (!compact_scratch)
            HIP_TRY(hipMalloc(&compact_scratch,
                              (size_t)sd.row_cap * sd.row_stride));
        uint32_t* ncur = nullptr;
        HIP_TRY(hipMalloc(&ncur, 4));
        HIP_TRY(hipMemsetAsync(ncur, 0, 4, stream));
        jcompact_copy_kernel<<<4096, 256, 0, stream>>>(sd, compact_scratch,
                                                       ncur, cur);
        jslot_init_kernel<<<2048, 256, 0, stream>>>(
            sd.slots, (size_t)sd.cap_mask + 1);
        int rcs = hipStreamSynchronize(stream) == hipSuccess
                      ? RW_OK : RW_E_INTERNAL;
        uint32_t n_alive = 0;
        if (rcs == RW_OK)
            hipMemcpy(&n_alive, ncur, 4, hipMemcpyDeviceToHost);
        hipFree(ncur);
        if (rcs != RW_OK) FAIL(RW_E_INTERNAL, "topn compact copy failed");
        uint8_t* old = sd.rows;
        sd.rows = compact_scratch;
        compact_scratch = old; // ping-pong
        HIP_TRY(hipMemcpy(sd.row_cursor, &n_alive, 4,
                          hipMemcpyHostToDevice));
        HIP_TRY(hipMemsetAsync(tcounters, 0, 8, stream));
        uint32_t blocks = (n_alive + 255) / 256;
        if (blocks > 2048) blocks = 2048;
        if (!blocks) blocks = 1;
        topn_relink_kernel<<<blocks, 256, 0, stream>>>(sd, m, n_alive,
                                                       tcounters + 1);
        if (hipStreamSynchronize(stream) != hipSuccess)
            FAIL(RW_E_INTERNAL, "topn compact relink failed");
        uint32_t tc[2];
        HIP_TRY(hipMemcpy(tc, tcounters, 8, hipMemcpyDeviceToHost));
        if (tc[1]) FAIL(RW_E_INTERNAL, "topn compact overflow (%u)", tc[1]);
        flush_mark = n_alive;
        if (reclaimed)
            *reclaimed = (uint64_t)(cur - n_alive) * sd.row_stride;
        return RW_OK;
    }

    // §8f-2 checkpoint spill: the TopN state table's per-epoch KV deltas —
    // key = memcomparable storage key (group cols ASC, then the cache-key
    // cols with their declared orders), value = value-encoded full row;
    // same kill-list netting and record framing as the join drain.
    uint32_t flush_mark = 0;
    std::vector<uint8_t> ck_desc_host;
    int checkpoint_drain(std::vector<uint8_t>& sp) {
        HIP_TRY(hipStreamSynchronize(stream));
        uint32_t cur = 0, kcur = 0;
        HIP_TRY(hipMemcpy(&cur, sd.row_cursor, 4, hipMemcpyDeviceToHost));
        HIP_TRY(hipMemcpy(&kcur, sd.killed_cursor, 4, hipMemcpyDeviceToHost));
        if (kcur > sd.killed_cap)
            FAIL(RW_E_INTERNAL, "topn kill list overflow (lost deltas)");
        if (cur > sd.row_cap) cur = sd.row_cap;
        size_t stride = sd.row_stride;
        std::vector<uint8_t> fresh((size_t)(cur - flush_mark) * stride);
        if (cur > flush_mark)
            HIP_TRY(hipMemcpy(fresh.data(),
                              sd.rows + (size_t)flush_mark * stride,
                              fresh.size(), hipMemcpyDeviceToHost));
        std::vector<uint32_t> kills(kcur);
        if (kcur)
            HIP_TRY(hipMemcpy(kills.data(), sd.killed, (size_t)kcur * 4,
                              hipMemcpyDeviceToHost));
        auto encode_key = [&](const uint8_t* rec, std::string& k) {
            const uint32_t vb = ((const uint32_t*)rec)[2];
            const int64_t* vals = (const int64_t*)(rec + 16);
            std::vector<uint8_t> kb;
            for (int i = 0; i < m.KW; i++) {
                uint8_t col = m.gk_cols[i];
                rwcodec::DatumC d = rwcodec::datum_of_word(
                    types[col], !((vb >> col) & 1), vals[col]);
                rwcodec::memcmp_encode_datum(kb, types[col], d, {});
            }
            for (int i = 0; i < m.n_ck; i++) {
                uint8_t col = m.ck_cols[i];
                rwcodec::DatumC d = rwcodec::datum_of_word(
                    types[col], !((vb >> col) & 1), vals[col]);
                rwcodec::OrderType ot;
                ot.desc = m.ck_desc[i] != 0;
                rwcodec::memcmp_encode_datum(kb, types[col], d, ot);
            }
            k.assign((const char*)kb.data(), kb.size());
        };
        auto encode_val = [&](const uint8_t* rec, std::vector<uint8_t>& v) {
            const uint32_t vb = ((const uint32_t*)rec)[2];
            const int64_t* vals = (const int64_t*)(rec + 16);
            for (int c = 0; c < m.n_cols; c++) {
                rwcodec::DatumC d = rwcodec::datum_of_word(
                    types[c], !((vb >> c) & 1), vals[c]);
                rwcodec::value_encode_datum(v, types[c], d);
            }
        };
        std::map<std::string, std::optional<std::vector<uint8_t>>> delta;
        std::vector<uint8_t> oldrec(stride);
        for (uint32_t i = 0; i < kcur; i++) {
            if (kills[i] >= flush_mark) continue;
            HIP_TRY(hipMemcpy(oldrec.data(),
                              sd.rows + (size_t)kills[i] * stride, stride,
                              hipMemcpyDeviceToHost));
            std::string k;
            encode_key(oldrec.data(), k);
            delta[k] = std::nullopt;
        }
        for (uint32_t i = flush_mark; i < cur; i++) {
            const uint8_t* rec = fresh.data() + (size_t)(i - flush_mark) * stride;
            if (!((const uint32_t*)rec)[0]) continue;
            std::string k;
            encode_key(rec, k);
            std::vector<uint8_t> v;
            encode_val(rec, v);
            delta[k] = std::move(v);
        }
        auto put32 = [&](uint32_t x) {
            for (int b2 = 0; b2 < 4; b2++) sp.push_back((uint8_t)(x >> (8 * b2)));
        };
        for (auto& [k, v] : delta) {
            sp.push_back(v.has_value() ? 1 : 0);
            put32((uint32_t)k.size());
            sp.insert(sp.end(), k.begin(), k.end());
            put32(v ? (uint32_t)v->size() : 0);
            if (v) sp.insert(sp.end(), v->begin(), v->end());
        }
        HIP_TRY(hipMemset(sd.killed_cursor, 0, 4));
        flush_mark = cur;
        return RW_OK;
    }

    RwChunk* poll() {
        if (outq.empty()) return nullptr;
        RwChunk* c = outq.front();
        outq.erase(outq.begin());
        return c;
    }
};

extern "C" {

int rw_topn_checkpoint_drain(void* h, uint8_t** buf, uint64_t* len) {
    std::vector<uint8_t> sp;
    int rc = ((GroupTopN*)h)->checkpoint_drain(sp);
    if (rc != RW_OK) return rc;
    return spill_export(sp, buf, len);
}

// §8f-5 recovery: replay one or more concatenated epoch drains into a fresh
// GroupTopN executor. PUT/DELETE frames net host-side (last write wins, as a
// KV store compaction would); the surviving rows rebuild the device state
// and are marked persisted so the next drain does not re-PUT them.
int rw_topn_restore(void* h, const uint8_t* buf, uint64_t len) {
    auto* t = (GroupTopN*)h;
    std::map<std::string, std::vector<uint8_t>> merged;
    bool ok = rwcodec::for_each_frame(
        buf, len,
        [&](uint8_t put, const uint8_t* k, uint32_t klen, const uint8_t* v,
            uint32_t vlen) {
            std::string key((const char*)k, klen);
            if (put)
                merged[key].assign(v, v + vlen);
            else
                merged.erase(key);
        });
    if (!ok) FAIL(RW_E_INVAL, "malformed spill stream");
    uint32_t n = (uint32_t)merged.size();
    if (!n) return RW_OK;
    int ncols = t->m.n_cols;
    std::vector<std::vector<int64_t>> cols(ncols, std::vector<int64_t>(n));
    std::vector<std::vector<uint8_t>> valid(ncols, std::vector<uint8_t>(n));
    uint32_t i = 0;
    for (auto& [kbytes, val] : merged) {
        (void)kbytes;
        size_t off = 0;
        for (int c = 0; c < ncols; c++) {
            rwcodec::DatumC d;
            size_t got = rwcodec::value_decode_datum(
                val.data() + off, val.size() - off, t->types[c], &d);
            if (!got) FAIL(RW_E_INVAL, "restore: bad row datum");
            off += got;
            cols[c][i] = d.null ? 0 : rwcodec::word_of_datum(t->types[c], d);
            valid[c][i] = !d.null;
        }
        i++;
    }
    int rc = t->ensure_caps(n);
    if (rc != RW_OK) return rc;
    JoinBatchDev b = t->stage;
    for (int c = 0; c < ncols; c++) {
        HIP_TRY(hipMemcpy(b.col_vals[c], cols[c].data(), (size_t)n * 8,
                          hipMemcpyHostToDevice));
        HIP_TRY(hipMemcpy(b.col_valid[c], valid[c].data(), n,
                          hipMemcpyHostToDevice));
    }
    b.vis = nullptr;
    b.n_rows = n;
    HIP_TRY(hipMemset(t->tcounters, 0, 8));
    uint32_t blocks = (n + 255) / 256;
    if (blocks > 2048) blocks = 2048;
    topn_restore_kernel<<<blocks, 256, 0, t->stream>>>(b, t->sd, t->m,
                                                       t->tcounters + 1);
    if (hipStreamSynchronize(t->stream) != hipSuccess)
        FAIL(RW_E_INTERNAL, "restore sync failed");
    uint32_t tc[2];
    HIP_TRY(hipMemcpy(tc, t->tcounters, 8, hipMemcpyDeviceToHost));
    if (tc[1]) FAIL(RW_E_INTERNAL, "restore overflow (code %u)", tc[1]);
    uint32_t cur = 0;
    HIP_TRY(hipMemcpy(&cur, t->sd.row_cursor, 4, hipMemcpyDeviceToHost));
    t->flush_mark = cur;
    return RW_OK;
}

void* rw_group_top_n_create(const RwGroupTopNDesc* d) {
    auto* t = new GroupTopN();
    if (t->init(d) != RW_OK) {
        delete t;
        return nullptr;
    }
    return t;
}

int rw_topn_compact(void* h, uint64_t* reclaimed) {
    return ((GroupTopN*)h)->compact(reclaimed);
}

// handle_watermark (group_top_n.rs:266-273): a watermark on the FIRST
// group-by column cleans the state table below it and is forwarded
// (returns 1); any other column's watermark is absorbed (returns 0).
int rw_group_top_n_watermark(void* h, uint32_t col_idx, int64_t val) {
    auto* t = (GroupTopN*)h;
    if (t->m.KW == 0 || t->m.gk_cols[0] != (uint8_t)col_idx) return 0;
    topn_clean_kernel<<<2048, 256, 0, t->stream>>>(t->sd, 0, val);
    if (hipStreamSynchronize(t->stream) != hipSuccess)
        FAIL(RW_E_INTERNAL, "topn clean sync failed");
    return 1;
}
int rw_group_top_n_push_chunk(void* h, const RwChunk* c) {
    return ((GroupTopN*)h)->push_chunk(c);
}
int rw_group_top_n_flush(void* h, uint64_t epoch) {
    (void)h;
    (void)epoch;
    return RW_OK; // emission is per push; checkpoint spill is a later row
}
RwChunk* rw_group_top_n_poll(void* h) { return ((GroupTopN*)h)->poll(); }
void rw_group_top_n_destroy(void* h) { delete (GroupTopN*)h; }

} // extern "C"

// ---------------------------------------------------------------------------
// HBM read-bandwidth probe: pins the "achievable" read rate the roofline
// `frac` is quoted against — the same b128 grid-stride load
// pattern as dense_load, reduced into a sink so nothing is eliminated.
// Measurement infrastructure, not a product path.
__global__ void membw_probe_kernel(const ulonglong2* __restrict__ p, size_t n,
                                   unsigned long long* sink) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    unsigned long long acc = 0;
    for (; i + 3 * stride < n; i += 4 * stride) {
        ulonglong2 a = p[i], b = p[i + stride], c = p[i + 2 * stride],
                   d = p[i + 3 * stride];
        acc += a.x + a.y + b.x + b.y + c.x + c.y + d.x + d.y;
    }
    for (; i < n; i += stride) acc += p[i].x + p[i].y;
    if (acc == 0xdeadbeefdeadbeefULL) *sink = acc; // never taken
}

// RANDOM-access ceiling probe: each lane chases splitmix-scattered 64-B
// lines over `bytes` of HBM — the achievable rate for the hash-join's
// probe/insert access pattern (one random line per touch), the honest
// denominator for q8's roofline discussion. Reads the first
// u64 of a random line per step; reported as GB/s of 64-B lines touched.
__global__ void membw_rand_probe_kernel(const unsigned long long* __restrict__ p,
                                        size_t n_lines, int steps_per_thread,
                                        unsigned long long* sink) {
    uint64_t x = (uint64_t)(blockIdx.x * blockDim.x + threadIdx.x) *
                     0x9e3779b97f4a7c15ULL +
                 0x243f6a8885a308d3ULL;
    unsigned long long acc = 0;
    for (int i = 0; i < steps_per_thread; i++) {
        x += 0x9e3779b97f4a7c15ULL;
        uint64_t h = x;
        h = (h ^ (h >> 30)) * 0xbf58476d1ce4e5b9ULL;
        h = (h ^ (h >> 27)) * 0x94d049bb133111ebULL;
        h ^= h >> 31;
        acc += p[(h % n_lines) * 8]; // 64-B line stride (8 u64)
    }
    if (acc == 0xdeadbeefdeadbeefULL) *sink = acc;
}

// q8-SHAPE ladder: a stripped reproduction of the inner probe kernel's
// per-row structure with stages toggled by `mode` bits, against synthetic
// tables of the bench's sizes — pinpoints which stage carries the gap
// between the ~30 us/1M-row random-access ceiling and the ~490 us kernel.
// mode bits: 1 = wave-scan output reservation, 2 = emit writes (8 cols,
// columnar), 4 = own-side insert (slot CAS + 64-B record write + link).
__global__ void q8shape_kernel(const int64_t* __restrict__ k0,
                               const int64_t* __restrict__ k1,
                               const int64_t* __restrict__ k2,
                               const uint64_t* __restrict__ mslots,
                               uint32_t mcap_mask,
                               const uint64_t* __restrict__ mrows,
                               uint32_t mrow_mask, uint64_t* oslots,
                               uint32_t ocap_mask, uint64_t* orows,
                               uint32_t* ocursor, int64_t* out_vals,
                               uint8_t* out_ops, uint32_t out_cap,
                               uint32_t n, int mode, uint32_t* sink) {
    uint32_t stride = gridDim.x * blockDim.x;
    uint32_t iters = (n + stride - 1) / stride;
    int lane = threadIdx.x & 63;
    for (uint32_t it = 0; it < iters; it++) {
        uint32_t r = it * stride + blockIdx.x * blockDim.x + threadIdx.x;
        bool active = r < n;
        int64_t kw[3] = {0, 0, 0};
        uint64_t h = 0x20210401u;
        if (active) {
            kw[0] = k0[r];
            kw[1] = k1[r];
            kw[2] = k2[r];
            for (int i = 0; i < 3; i++) h = mix64(h ^ (uint64_t)kw[i]);
        }
        if (mode & 16) { // byte streams like the real kernel (ops + valid)
            if (active) {
                h ^= out_ops[r & (out_cap - 1)]; // stand-in ops stream read
                for (int i = 0; i < 3; i++)
                    h += ((const uint8_t*)k0)[r]; // validity-byte stand-ins
            }
        }
        uint32_t my_n = 0;
        uint64_t mv0 = 0;
        if (active) {
            uint32_t slot = (uint32_t)(h >> (64 - __popc(mcap_mask)));
            uint64_t packed = mslots[slot];
            if ((uint32_t)packed & (1u << ((uint32_t)(h >> 32) & 31))) {
                uint32_t head = (uint32_t)(packed >> 32) & mrow_mask;
                if (mode & 8) {
                    // the real kernel's walk SHAPE: header fields, per-key
                    // branchy compares, next chase with alive checks
                    uint32_t row2 = head;
                    int guard = 0;
                    while (row2 != UINT32_MAX && guard++ < 4) {
                        const uint64_t* rec = mrows + (size_t)row2 * 8;
                        uint64_t hdr = rec[0]; // alive | next
                        uint64_t vb = rec[1];
                        bool eq = true;
                        for (int i = 0; i < 3 && eq; i++)
                            eq = ((int64_t)rec[2 + i] ^ kw[i]) != 1; // ~always true
                        if ((hdr & 1) && eq && (vb | 1)) {
                            my_n++;
                            mv0 += rec[5];
                        }
                        // synthetic records have garbage next: terminate
                        row2 = UINT32_MAX;
                    }
                } else {
                    // one 64-B record line: read all 8 words
                    const uint64_t* rec = mrows + (size_t)head * 8;
                    uint64_t acc = 0;
                    for (int w = 0; w < 8; w++) acc += rec[w];
                    mv0 = acc;
                    my_n = 1;
                }
            }
        }
        uint32_t my_base = 0;
        if (mode & 1) {
            uint32_t incl = my_n;
            for (int d = 1; d < 64; d <<= 1) {
                uint32_t o = __shfl_up(incl, d);
                if (lane >= d) incl += o;
            }
            uint32_t total = (uint32_t)__shfl((int)incl, 63);
            uint32_t base = 0;
            if (lane == 0 && total) base = atomicAdd(ocursor, total);
            base = (uint32_t)__shfl((int)base, 0);
            my_base = base + incl - my_n;
        }
        if (mode & 64) { // ballot-form reservation (1 ballot + popcounts)
            uint64_t got = __ballot(my_n != 0);
            uint32_t total = (uint32_t)__popcll(got);
            uint32_t base = 0;
            if (lane == 0 && total) base = atomicAdd(ocursor, total);
            base = (uint32_t)__shfl((int)base, 0);
            my_base = base + (uint32_t)__popcll(got & ((1ULL << lane) - 1));
        }
        if (mode & 128) { // pre-assigned output slot: NO cross-lane ops
            my_base = r;
        }
        if ((mode & 2) && my_n) {
            uint32_t orow = my_base & (out_cap - 1);
            out_ops[orow] = 0;
            for (int c = 0; c < 8; c++)
                out_vals[(size_t)c * out_cap + orow] =
                    (int64_t)(mv0 + kw[c % 3]);
        }
        if ((mode & 4) && active) {
            uint32_t oslot = (uint32_t)(h >> (64 - __popc(ocap_mask)));
            // record append at a wave-aggregated cursor position
            uint64_t wmask = __ballot(true);
            int leader = 63 - __clzll(wmask);
            uint32_t base2 = 0;
            if (lane == leader)
                base2 = atomicAdd(ocursor + 16, (uint32_t)__popcll(wmask));
            base2 = (uint32_t)__shfl((int)base2, leader);
            uint32_t row =
                (base2 + (uint32_t)__popcll(wmask & ((1ULL << lane) - 1))) &
                mrow_mask;
            uint64_t* rec = orows + (size_t)row * 8;
            for (int w = 0; w < 8; w++) rec[w] = h + w;
            // bucket push: CAS on the slot
            uint64_t old = __hip_atomic_load(&oslots[oslot], RLX);
            for (;;) {
                rec[1] = (uint32_t)old ? (old >> 32) : 0xFFFFFFFFull;
                asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
                uint64_t want = ((uint64_t)row << 32) |
                                ((uint32_t)old | (1u << ((uint32_t)(h >> 32) & 31)));
                uint64_t prev = atomicCAS((unsigned long long*)&oslots[oslot],
                                          old, want);
                if (prev == old) break;
                old = prev;
            }
        }
        if (active && mv0 == 0xdeadbeefdeadbeefULL) *sink = 1;
    }
}

__global__ void q8shape_slot_fill_kernel(uint64_t* slots, uint32_t cap,
                                         uint32_t rmask) {
    uint32_t stride = gridDim.x * blockDim.x;
    for (uint32_t s2 = blockIdx.x * blockDim.x + threadIdx.x; s2 < cap;
         s2 += stride)
        slots[s2] = ((uint64_t)(mix64(s2) & rmask) << 32) | 0xFFFFFFFFull;
}

__global__ void q8shape_fill_kernel(int64_t* k0, int64_t* k1, int64_t* k2,
                                    uint32_t n) {
    uint32_t stride = gridDim.x * blockDim.x;
    for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < n;
         r += stride) {
        k0[r] = (int64_t)mix64(r * 2654435761u + 1);
        k1[r] = (int64_t)mix64(r * 40503u + 2);
        k2[r] = (int64_t)mix64(r * 2246822519u + 3);
    }
}

extern "C" int rw_q8shape_probe(uint32_t n_rows, int mode, double* us_out) {
    // bench-sized synthetic state: 2^24-slot tables (134 MB), 16M-row
    // record stores (1 GiB), q8-like 50%-hit blooms
    uint32_t mcap = 1u << 24, rmask = (1u << 24) - 1;
    size_t rbytes = (size_t)(rmask + 1) * 64;
    int64_t *k0, *k1, *k2;
    uint64_t *mslots, *mrows, *oslots, *orows;
    uint32_t* ocursor;
    int64_t* out_vals;
    uint8_t* out_ops;
    uint32_t out_cap = 1u << 22;
    if (hipMalloc(&k0, (size_t)n_rows * 8) != hipSuccess) return RW_E_INTERNAL;
    (void)hipMalloc(&k1, (size_t)n_rows * 8);
    (void)hipMalloc(&k2, (size_t)n_rows * 8);
    (void)hipMalloc(&mslots, (size_t)mcap * 8);
    (void)hipMalloc(&mrows, rbytes);
    (void)hipMalloc(&oslots, (size_t)mcap * 8);
    (void)hipMalloc(&orows, rbytes);
    (void)hipMalloc(&ocursor, 256);
    (void)hipMalloc(&out_vals, (size_t)8 * out_cap * 8);
    (void)hipMalloc(&out_ops, out_cap);
    q8shape_fill_kernel<<<2048, 256>>>(k0, k1, k2, n_rows);
    q8shape_slot_fill_kernel<<<2048, 256>>>(mslots, mcap, rmask);
    (void)hipMemset(mrows, 1, rbytes);
    (void)hipMemset(oslots, 0, (size_t)mcap * 8);
    (void)hipMemset(ocursor, 0, 256);
    uint32_t* sink;
    (void)hipMalloc(&sink, 4);
    uint32_t blocks = (n_rows + 255) / 256;
    if (blocks > 2048) blocks = 2048;
    hipEvent_t e0, e1;
    (void)hipEventCreate(&e0);
    (void)hipEventCreate(&e1);
    q8shape_kernel<<<blocks, 256>>>(k0, k1, k2, mslots, mcap - 1, mrows,
                                    rmask, oslots, mcap - 1, orows, ocursor,
                                    out_vals, out_ops, out_cap, n_rows, mode,
                                    sink);
    (void)hipEventRecord(e0);
    for (int i = 0; i < 8; i++)
        q8shape_kernel<<<blocks, 256>>>(k0, k1, k2, mslots, mcap - 1, mrows,
                                        rmask, oslots, mcap - 1, orows,
                                        ocursor, out_vals, out_ops, out_cap,
                                        n_rows, mode, sink);
    (void)hipEventRecord(e1);
    if (hipEventSynchronize(e1) != hipSuccess) return RW_E_INTERNAL;
    float ms = 0;
    (void)hipEventElapsedTime(&ms, e0, e1);
    *us_out = ms * 1000.0 / 8;
    hipFree(k0); hipFree(k1); hipFree(k2); hipFree(mslots); hipFree(mrows);
    hipFree(oslots); hipFree(orows); hipFree(ocursor); hipFree(out_vals);
    hipFree(out_ops); hipFree(sink);
    (void)hipEventDestroy(e0);
    (void)hipEventDestroy(e1);
    return RW_OK;
}

// random atomicAdd throughput (the insert path's slot-CAS analogue)
__global__ void membw_rand_atomic_kernel(unsigned long long* p, size_t n_lines,
                                         int steps_per_thread) {
    uint64_t x = (uint64_t)(blockIdx.x * blockDim.x + threadIdx.x) *
                     0x9e3779b97f4a7c15ULL +
                 0x243f6a8885a308d3ULL;
    for (int i = 0; i < steps_per_thread; i++) {
        x += 0x9e3779b97f4a7c15ULL;
        uint64_t h = x;
        h = (h ^ (h >> 30)) * 0xbf58476d1ce4e5b9ULL;
        h = (h ^ (h >> 27)) * 0x94d049bb133111ebULL;
        h ^= h >> 31;
        atomicAdd(&p[(h % n_lines) * 8], 1ull);
    }
}

// DEPENDENT random chase: each step's address depends on the previous
// load's value (the probe kernel's hash->slot->record chain shape);
// `mlp` independent chains per thread expose memory-level parallelism
__global__ void membw_rand_chase_kernel(const unsigned long long* __restrict__ p,
                                        size_t n_lines, int steps, int mlp,
                                        unsigned long long* sink) {
    uint64_t x[8];
    for (int j = 0; j < mlp && j < 8; j++)
        x[j] = (uint64_t)(blockIdx.x * blockDim.x + threadIdx.x) *
                   0x9e3779b97f4a7c15ULL +
               j * 0x94d049bb133111ebULL + 1;
    unsigned long long acc = 0;
    for (int i = 0; i < steps; i++) {
        for (int j = 0; j < mlp && j < 8; j++) {
            uint64_t h = x[j];
            h = (h ^ (h >> 30)) * 0xbf58476d1ce4e5b9ULL;
            h = (h ^ (h >> 27)) * 0x94d049bb133111ebULL;
            h ^= h >> 31;
            unsigned long long v = p[(h % n_lines) * 8];
            x[j] += v + 0x9e3779b97f4a7c15ULL; // DEPENDS on the load
            acc += v;
        }
    }
    if (acc == 0xdeadbeefdeadbeefULL) *sink = acc;
}

extern "C" int rw_membw_rand_atomic(uint64_t bytes, int steps_per_thread,
                                    double* gops_out) {
    size_t n_lines = bytes / 64;
    unsigned long long* p = nullptr;
    if (hipMalloc(&p, n_lines * 64) != hipSuccess) return RW_E_INTERNAL;
    (void)hipMemset(p, 0, n_lines * 64);
    int grid = 2048, blk = 256;
    hipEvent_t e0, e1;
    (void)hipEventCreate(&e0);
    (void)hipEventCreate(&e1);
    membw_rand_atomic_kernel<<<grid, blk>>>(p, n_lines, steps_per_thread);
    (void)hipEventRecord(e0);
    membw_rand_atomic_kernel<<<grid, blk>>>(p, n_lines, steps_per_thread);
    (void)hipEventRecord(e1);
    if (hipEventSynchronize(e1) != hipSuccess) return RW_E_INTERNAL;
    float ms = 0;
    (void)hipEventElapsedTime(&ms, e0, e1);
    *gops_out = (double)grid * blk * steps_per_thread / (ms * 1e-3) / 1e9;
    (void)hipFree(p);
    (void)hipEventDestroy(e0);
    (void)hipEventDestroy(e1);
    return RW_OK;
}

extern "C" int rw_membw_rand_chase(uint64_t bytes, int steps, int mlp,
                                   double* glines_out) {
    size_t n_lines = bytes / 64;
    unsigned long long* p = nullptr;
    unsigned long long* sink = nullptr;
    if (hipMalloc(&p, n_lines * 64) != hipSuccess) return RW_E_INTERNAL;
    (void)hipMalloc(&sink, 8);
    (void)hipMemset(p, 1, n_lines * 64);
    int grid = 2048, blk = 256;
    hipEvent_t e0, e1;
    (void)hipEventCreate(&e0);
    (void)hipEventCreate(&e1);
    membw_rand_chase_kernel<<<grid, blk>>>(p, n_lines, steps, mlp, sink);
    (void)hipEventRecord(e0);
    membw_rand_chase_kernel<<<grid, blk>>>(p, n_lines, steps, mlp, sink);
    (void)hipEventRecord(e1);
    if (hipEventSynchronize(e1) != hipSuccess) return RW_E_INTERNAL;
    float ms = 0;
    (void)hipEventElapsedTime(&ms, e0, e1);
    *glines_out =
        (double)grid * blk * steps * mlp / (ms * 1e-3) / 1e9;
    (void)hipFree(p);
    (void)hipFree(sink);
    (void)hipEventDestroy(e0);
    (void)hipEventDestroy(e1);
    return RW_OK;
}

extern "C" int rw_membw_rand_probe(uint64_t bytes, int steps_per_thread,
                                   double* glines_out, double* gbps_out) {
    size_t n_lines = bytes / 64;
    unsigned long long* p = nullptr;
    unsigned long long* sink = nullptr;
    if (hipMalloc(&p, n_lines * 64) != hipSuccess) return RW_E_INTERNAL;
    (void)hipMalloc(&sink, 8);
    (void)hipMemset(p, 1, n_lines * 64);
    int grid = 2048, blk = 256;
    hipEvent_t e0, e1;
    (void)hipEventCreate(&e0);
    (void)hipEventCreate(&e1);
    membw_rand_probe_kernel<<<grid, blk>>>(p, n_lines, steps_per_thread, sink);
    (void)hipEventRecord(e0);
    membw_rand_probe_kernel<<<grid, blk>>>(p, n_lines, steps_per_thread, sink);
    (void)hipEventRecord(e1);
    if (hipEventSynchronize(e1) != hipSuccess) return RW_E_INTERNAL;
    float ms = 0;
    (void)hipEventElapsedTime(&ms, e0, e1);
    double touches = (double)grid * blk * steps_per_thread;
    *glines_out = touches / (ms * 1e-3) / 1e9;
    *gbps_out = touches * 64 / (ms * 1e-3) / 1e9;
    (void)hipFree(p);
    (void)hipFree(sink);
    (void)hipEventDestroy(e0);
    (void)hipEventDestroy(e1);
    return RW_OK;
}

extern "C" int rw_membw_probe(uint64_t bytes, int iters, double* gbps_out) {
    size_t n = bytes / sizeof(ulonglong2);
    ulonglong2* p = nullptr;
    unsigned long long* sink = nullptr;
    if (hipMalloc(&p, n * sizeof(ulonglong2)) != hipSuccess) return RW_E_INTERNAL;
    (void)hipMalloc(&sink, 8);
    (void)hipMemset(p, 1, n * sizeof(ulonglong2));
    int grid = 8192; // ≫256 workgroups: fills all 8 XCDs
    hipEvent_t e0, e1;
    (void)hipEventCreate(&e0);
    (void)hipEventCreate(&e1);
    membw_probe_kernel<<<grid, 256>>>(p, n, sink); // warmup
    (void)hipEventRecord(e0);
    for (int it = 0; it < iters; it++) membw_probe_kernel<<<grid, 256>>>(p, n, sink);
    (void)hipEventRecord(e1);
    if (hipEventSynchronize(e1) != hipSuccess) return RW_E_INTERNAL;
    float ms = 0;
    (void)hipEventElapsedTime(&ms, e0, e1);
    *gbps_out = (double)bytes * iters / (ms * 1e-3) / 1e9;
    (void)hipFree(p);
    (void)hipFree(sink);
    (void)hipEventDestroy(e0);
    (void)hipEventDestroy(e1);
    return RW_OK;
}
