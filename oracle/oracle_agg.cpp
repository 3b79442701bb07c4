// oracle/oracle_agg.cpp — CPU restatement of HashAggExecutor.
// ORACLE — TEST INFRASTRUCTURE ONLY (see common.hpp header note).
//
// Restates (reference under /root/reference):
//  - HashAggExecutor::apply_chunk / flush_data
//    (src/stream/src/executor/aggregate/hash_agg.rs:332-514). Per-row apply in
//    row order is equivalent to the reference's per-group visibility loop
//    (:241-258,366-398): each row belongs to exactly one group and a group's
//    state depends only on its own rows' order.
//  - AggGroup::get_outputs / build_outputs_change with OnlyOutputIfHasInput
//    (aggregate/agg_group.rs:131-165,431-467,545-610): reset value states at
//    row_count==0, 0→n Insert, n→0 Delete(prev row), changed U−/U+,
//    unchanged nothing; negative row_count clamps to 0 (:55-79).
//  - Value-state aggregates (expr/impl/src/aggregate/general.rs): sum with
//    checked add/sub (:18-41), count ±1 (:154-162), append-only min/max
//    (:90-125). NULL inputs are skipped by sum/count(col)/min/max; count(*)
//    counts rows. Sum state stays Some(0) if all inputs retracted (macro
//    Option<S> semantics) — outputs 0, not NULL, while row_count > 0.
//  - Materialized-input min/max (aggregate/minput.rs:170-248): state ordered
//    by [value ASC(min)/DESC(max), stream_key ASC], NULLs largest
//    (test_utils/agg_executor.rs:72-121); output = first entry's value.
#include <algorithm>
#include <cstdlib>
#include <cstring>
#include <stdexcept>
#include <unordered_map>

#include "../include/rw_codec.hpp"
#include "../include/rw_stream.h"
#include "common.hpp"

namespace orc {

thread_local std::string g_err;

#define FAIL(code, ...)                         \
    do {                                        \
        char _b[256];                           \
        snprintf(_b, sizeof(_b), __VA_ARGS__);  \
        g_err = _b;                             \
        return code;                            \
    } while (0)

struct ValueState {
    // count / sum: has=false ⇔ Rust state None (sum) — count init 0 has=true
    bool has = false;
    bool is_float = false;
    int64_t i = 0;
    double d = 0;
};

struct MInputState {
    // ordered multiset: key = [value, stream_key...], value = arg datum
    std::map<Row, Datum, RowOrderLess> entries;
    MInputState() = default;
    explicit MInputState(RowOrderLess less) : entries(std::move(less)) {}
};

// exact decimal-sum accumulator (see common.hpp's decimal restatement)
struct DecSumState {
    I256 sum;
    uint32_t maxscale = 0;
    int64_t nonnull = 0, nan = 0, pinf = 0, ninf = 0;
};

struct AggGroupState {
    std::vector<ValueState> vstates;
    std::vector<DecSumState> dstates; // per call; used for decimal sums
    std::vector<MInputState> mstates; // indexed per call; unused for value calls
    bool has_prev = false;
    Row prev_outputs; // agg outputs only (no group key)
};

struct KeyHash {
    size_t operator()(const Row& r) const {
        size_t h = 0xcbf29ce484222325ULL;
        for (auto& d : r) {
            uint64_t v = d.null ? 0x9e3779b97f4a7c15ULL : (uint64_t)d.i;
            h ^= v + 0x9e3779b97f4a7c15ULL + (h << 6) + (h >> 2);
        }
        return h;
    }
};
struct KeyEq {
    const std::vector<uint8_t>* types;
    bool operator()(const Row& a, const Row& b) const {
        for (size_t i = 0; i < a.size(); i++)
            if (!datum_eq(a[i], b[i], (*types)[i])) return false;
        return true;
    }
};

struct HashAggOracle {
    // descriptor
    std::vector<uint8_t> input_types;
    std::vector<uint32_t> group_key;
    std::vector<RwAggCall> calls;
    uint32_t row_count_index;
    std::vector<uint32_t> stream_key;
    size_t chunk_size;
    bool append_only;
    bool eowc = false; // emit-on-window-close (hash_agg.rs:421-474)
    bool has_pending_wm = false;
    int64_t pending_wm = 0;
    std::vector<uint8_t> group_key_types;
    std::vector<uint8_t> out_types; // group key types ++ ret types
    std::vector<bool> call_is_minput;

    std::unordered_map<Row, AggGroupState, KeyHash, KeyEq> groups;
    std::vector<Row> dirty_order; // first-touch order of dirty group keys
    std::unordered_map<Row, bool, KeyHash, KeyEq> dirty;

    ChunkBuilder builder;
    std::vector<std::unique_ptr<OwnedChunk>> outputs;
    // checkpoint spill buffer (§8f-2): intermediate-state-table KV deltas in
    // memcomparable-key / value-encoded form, accumulated per flush
    std::vector<uint8_t> spill;

    HashAggOracle(const RwHashAggDesc* d, std::vector<uint8_t> out_ts)
        : builder(d->chunk_size, out_ts) {
        input_types.assign(d->input_types, d->input_types + d->n_input_cols);
        group_key.assign(d->group_key_indices, d->group_key_indices + d->n_group_key);
        calls.assign(d->calls, d->calls + d->n_calls);
        row_count_index = d->row_count_index;
        stream_key.assign(d->stream_key, d->stream_key + d->n_stream_key);
        chunk_size = d->chunk_size;
        append_only = d->append_only;
        eowc = d->emit_on_window_close != 0;
        // decimal scope this round: sum/count ARGUMENTS only
        for (auto k : group_key)
            if (input_types[k] == RW_T_DECIMAL)
                throw std::invalid_argument("decimal group keys unsupported");
        for (auto k : stream_key)
            if (input_types[k] == RW_T_DECIMAL)
                throw std::invalid_argument("decimal stream keys unsupported");
        for (auto& c : calls)
            if (c.arg >= 0 && input_types[c.arg] == RW_T_DECIMAL &&
                (c.distinct ||
                 !(c.kind == RW_AGG_SUM || c.kind == RW_AGG_SUM0 ||
                   c.kind == RW_AGG_COUNT)))
                throw std::invalid_argument(
                    "decimal supported as non-DISTINCT sum/count "
                    "arguments only");
        for (auto k : group_key) group_key_types.push_back(input_types[k]);
        out_types = out_ts;
        for (auto& c : calls) {
            bool minput = (c.kind == RW_AGG_MIN || c.kind == RW_AGG_MAX) && !append_only;
            call_is_minput.push_back(minput);
        }
        // DISTINCT dedup (aggregate/distinct.rs:67-198): one counter map per
        // distinct column (the calls here carry no per-call filters, so all
        // calls distinct on a column share one count and one visibility)
        call_dedup_idx.assign(calls.size(), -1);
        for (size_t ci = 0; ci < calls.size(); ci++) {
            if (!calls[ci].distinct || calls[ci].arg < 0) continue;
            uint32_t col = (uint32_t)calls[ci].arg;
            size_t di = 0;
            for (; di < distinct_cols.size(); di++)
                if (distinct_cols[di] == col) break;
            if (di == distinct_cols.size()) {
                distinct_cols.push_back(col);
                RowOrderLess less;
                for (auto t : group_key_types) less.order.push_back({t, false});
                less.order.push_back({input_types[col], false});
                dedup_counts.emplace_back(less);
                dedup_touched.emplace_back(less);
                dedup_persisted.emplace_back(less);
            }
            call_dedup_idx[ci] = (int)di;
        }
        groups = decltype(groups)(16, KeyHash{}, KeyEq{&group_key_types});
        dirty = decltype(dirty)(16, KeyHash{}, KeyEq{&group_key_types});
    }

    std::vector<uint32_t> distinct_cols;
    std::vector<std::map<Row, int64_t, RowOrderLess>> dedup_counts;
    std::vector<int> call_dedup_idx; // per call: index into dedup_counts or -1
    // §8f-2 dedup-table spill: (group, datum) keys touched this epoch and
    // the set persisted at the last checkpoint (distinct.rs:158-185)
    std::vector<std::set<Row, RowOrderLess>> dedup_touched;
    std::vector<std::set<Row, RowOrderLess>> dedup_persisted;

    // §8f-2 round-2: materialized-input state-TABLE spill. One table per
    // retractable min/max call in the reference (AggStateStorage::
    // MaterializedInput, test_utils/agg_executor.rs:63-121): columns =
    // group key ++ arg value ++ stream key, pk ordered [group ASC, value
    // ASC(min)/DESC(max), stream key ASC], NULLs largest. Per-epoch
    // deltas keyed by the memcomparable pk with mem-table netting.
    std::vector<std::map<std::string, std::pair<uint8_t, std::vector<uint8_t>>>>
        minput_delta;

    int minput_ordinal(size_t ci) const {
        int o = 0;
        for (size_t j = 0; j < ci; j++) o += call_is_minput[j];
        return o;
    }

    void minput_encode(size_t ci, const Row& gkey, const Row& entry_key,
                       std::string* kout, std::vector<uint8_t>* vout) {
        const auto& c = calls[ci];
        std::vector<uint8_t> k;
        for (size_t i = 0; i < gkey.size(); i++) {
            rwcodec::DatumC d{gkey[i].null, gkey[i].i, gkey[i].d, gkey[i].i2};
            rwcodec::memcmp_encode_datum(k, group_key_types[i], d, {});
        }
        rwcodec::DatumC dv{entry_key[0].null, entry_key[0].i, entry_key[0].d,
                           entry_key[0].i2};
        rwcodec::memcmp_encode_datum(k, input_types[c.arg], dv,
                                     {c.kind == RW_AGG_MAX, true});
        for (size_t j = 0; j < stream_key.size(); j++) {
            const Datum& d0 = entry_key[1 + j];
            rwcodec::DatumC d{d0.null, d0.i, d0.d, d0.i2};
            rwcodec::memcmp_encode_datum(k, input_types[stream_key[j]], d, {});
        }
        kout->assign((const char*)k.data(), k.size());
        if (vout) {
            for (size_t i = 0; i < gkey.size(); i++) {
                rwcodec::DatumC d{gkey[i].null, gkey[i].i, gkey[i].d,
                                  gkey[i].i2};
                rwcodec::value_encode_datum(*vout, group_key_types[i], d);
            }
            rwcodec::DatumC dv2{entry_key[0].null, entry_key[0].i,
                                entry_key[0].d, entry_key[0].i2};
            rwcodec::value_encode_datum(*vout, input_types[c.arg], dv2);
            for (size_t j = 0; j < stream_key.size(); j++) {
                const Datum& d0 = entry_key[1 + j];
                rwcodec::DatumC d{d0.null, d0.i, d0.d, d0.i2};
                rwcodec::value_encode_datum(*vout, input_types[stream_key[j]],
                                            d);
            }
        }
    }

    int minput_drain(int mi, std::vector<uint8_t>& out) {
        if (mi < 0 || mi >= n_minput_total())
            FAIL(RW_E_INVAL, "minput table index %d", mi);
        if ((size_t)mi >= minput_delta.size()) return RW_OK;
        auto put32 = [&](uint32_t x) {
            for (int b = 0; b < 4; b++) out.push_back((uint8_t)(x >> (8 * b)));
        };
        for (auto& [k, pv] : minput_delta[mi]) {
            out.push_back(pv.first);
            put32((uint32_t)k.size());
            out.insert(out.end(), k.begin(), k.end());
            put32((uint32_t)pv.second.size());
            out.insert(out.end(), pv.second.begin(), pv.second.end());
        }
        minput_delta[mi].clear();
        return RW_OK;
    }

    int n_minput_total() const {
        int n = 0;
        for (auto m : call_is_minput) n += m;
        return n;
    }

    int minput_restore(int mi, const uint8_t* buf, uint64_t len) {
        if (mi < 0 || mi >= n_minput_total())
            FAIL(RW_E_INVAL, "minput table index %d", mi);
        // locate the call with this minput ordinal
        size_t ci = 0;
        int o = -1;
        for (; ci < calls.size(); ci++) {
            if (call_is_minput[ci]) o++;
            if (o == mi) break;
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
        for (auto& [kb, val] : merged) {
            (void)kb;
            size_t off = 0;
            auto rd = [&](uint8_t ty, Datum* out2) -> bool {
                rwcodec::DatumC d;
                size_t n2 = rwcodec::value_decode_datum(val.data() + off,
                                                        val.size() - off, ty,
                                                        &d);
                if (!n2) return false;
                off += n2;
                *out2 = d.null ? Datum()
                               : (ty == RW_T_DECIMAL
                                      ? Datum::of_dec(d.i, d.i2)
                                      : (type_is_float(ty) ? Datum::of_d(d.d)
                                                           : Datum::of_i(d.i)));
                return true;
            };
            Row gkey(group_key_types.size());
            for (size_t i = 0; i < gkey.size(); i++)
                if (!rd(group_key_types[i], &gkey[i]))
                    FAIL(RW_E_INVAL, "minput restore: bad group datum");
            Row ek(1 + stream_key.size());
            if (!rd(input_types[calls[ci].arg], &ek[0]))
                FAIL(RW_E_INVAL, "minput restore: bad value datum");
            for (size_t j = 0; j < stream_key.size(); j++)
                if (!rd(input_types[stream_key[j]], &ek[1 + j]))
                    FAIL(RW_E_INVAL, "minput restore: bad stream-key datum");
            AggGroupState& g = touch(gkey);
            g.mstates[ci].entries.emplace(ek, ek[0]);
        }
        // restored rows predate the epoch: clean, and no deltas
        dirty_order.clear();
        dirty.clear();
        return RW_OK;
    }

    RowOrderLess minput_order(const RwAggCall& c) const {
        // pk of the materialized-input table: value (ASC min / DESC max),
        // then stream key ASC (agg_executor.rs:90-105)
        RowOrderLess less;
        less.order.push_back({input_types[c.arg], c.kind == RW_AGG_MAX});
        for (auto sk : stream_key) less.order.push_back({input_types[sk], false});
        return less;
    }

    bool call_is_decimal(size_t ci) const {
        return !call_is_minput[ci] && calls[ci].arg >= 0 &&
               (size_t)calls[ci].arg < input_types.size() &&
               input_types[calls[ci].arg] == RW_T_DECIMAL;
    }

    AggGroupState& touch(const Row& key) {
        auto it = groups.find(key);
        if (it == groups.end()) {
            AggGroupState g;
            for (auto& c : calls) {
                ValueState v;
                if (c.kind == RW_AGG_COUNT_STAR || c.kind == RW_AGG_COUNT ||
                    c.kind == RW_AGG_SUM0) {
                    v.has = true; // init_state = 0
                }
                v.is_float = (c.ret_type == RW_T_F64 || c.ret_type == RW_T_F32);
                g.vstates.push_back(v);
                g.mstates.emplace_back(minput_order(c));
            }
            g.dstates.resize(calls.size());
            it = groups.emplace(key, std::move(g)).first;
        }
        if (!dirty.count(key)) {
            dirty.emplace(key, true);
            dirty_order.push_back(key);
        }
        return it->second;
    }

    int apply_row(AggGroupState& g, const Row& gkey, const ChunkView& cv,
                  size_t r, bool retract, const std::vector<bool>& hidden) {
        for (size_t ci = 0; ci < calls.size(); ci++) {
            const auto& c = calls[ci];
            if (!hidden.empty() && hidden[ci]) continue; // DISTINCT dup row
            if (call_is_minput[ci]) {
                // minput materializes the row keyed by [value, stream_key...]
                Row key;
                key.push_back(cv.at(r, c.arg));
                for (auto sk : stream_key) key.push_back(cv.at(r, sk));
                auto& m = g.mstates[ci].entries;
                int mi = minput_ordinal(ci);
                if ((size_t)mi >= minput_delta.size())
                    minput_delta.resize(mi + 1);
                auto& delta = minput_delta[mi];
                if (!retract) {
                    std::string kb;
                    std::vector<uint8_t> vb;
                    minput_encode(ci, gkey, key, &kb, &vb);
                    delta[kb] = {1, std::move(vb)};
                    m.emplace(std::move(key), cv.at(r, c.arg));
                } else {
                    auto it = m.find(key);
                    if (it != m.end()) {
                        std::string kb;
                        minput_encode(ci, gkey, key, &kb, nullptr);
                        auto di = delta.find(kb);
                        if (di != delta.end() && di->second.first == 1)
                            delta.erase(di); // created+died this epoch
                        else
                            delta[kb] = {0, {}};
                        m.erase(it);
                    }
                }
                continue;
            }
            auto& v = g.vstates[ci];
            switch (c.kind) {
                case RW_AGG_COUNT_STAR:
                    v.i += retract ? -1 : 1; // general.rs:159-162 (unchecked)
                    break;
                case RW_AGG_COUNT: {
                    Datum a = cv.at(r, c.arg);
                    if (!a.null) v.i += retract ? -1 : 1; // :154-157
                    break;
                }
                case RW_AGG_SUM:
                case RW_AGG_SUM0: {
                    Datum a = cv.at(r, c.arg);
                    if (a.null) break;
                    if (input_types[c.arg] == RW_T_DECIMAL) {
                        // exact-domain decimal sum (common.hpp restatement)
                        DecSumState& ds = g.dstates[ci];
                        DecVal dv = dec_parse(a.i, a.i2);
                        int64_t delta = retract ? -1 : 1;
                        ds.nonnull += delta;
                        if (dv.special == 1) ds.nan += delta;
                        else if (dv.special == 2) ds.pinf += delta;
                        else if (dv.special == 3) ds.ninf += delta;
                        else {
                            I256 add = dec_addend(dv);
                            if (retract) add = add.negated();
                            ds.sum.add(add);
                            if (dv.scale > ds.maxscale)
                                ds.maxscale = dv.scale;
                        }
                        v.has = true;
                        break;
                    }
                    if (v.is_float) {
                        double x = type_is_float(input_types[c.arg]) ? a.d : (double)a.i;
                        v.d = retract ? v.d - x : v.d + x;
                        v.has = true;
                    } else {
                        int64_t out;
                        bool ovf = retract ? __builtin_sub_overflow(v.i, a.i, &out)
                                           : __builtin_add_overflow(v.i, a.i, &out);
                        if (ovf) {
                            g_err = "sum out of range (general.rs:33-40)";
                            return RW_E_OVERFLOW;
                        }
                        v.i = out;
                        v.has = true;
                    }
                    break;
                }
                case RW_AGG_MIN:
                case RW_AGG_MAX: {
                    // value state — append-only only (agg_state.rs:49-56)
                    Datum a = cv.at(r, c.arg);
                    if (a.null) break;
                    uint8_t t = input_types[c.arg];
                    Datum curd;
                    if (v.has) {
                        curd = v.is_float ? Datum::of_d(v.d) : Datum::of_i(v.i);
                        int cmpv = datum_cmp(a, curd, t);
                        bool take = (c.kind == RW_AGG_MIN) ? (cmpv < 0) : (cmpv > 0);
                        if (!take) break;
                    }
                    v.has = true;
                    if (type_is_float(t)) v.d = a.d;
                    else v.i = a.i;
                    break;
                }
            }
        }
        return RW_OK;
    }

    int push_chunk(const RwChunk* chunk) {
        ChunkView cv{chunk};
        std::vector<bool> hidden;
        std::vector<bool> col_hidden(distinct_cols.size());
        for (size_t r = 0; r < cv.n_rows(); r++) {
            if (!cv.visible(r)) continue;
            Row key;
            key.reserve(group_key.size());
            for (auto k : group_key) key.push_back(cv.at(r, k));
            auto& g = touch(key);
            uint8_t op = cv.op(r);
            bool retract = (op == RW_OP_DELETE || op == RW_OP_UPDATE_DELETE);
            if (!distinct_cols.empty()) {
                // DISTINCT dedup (distinct.rs:131-158): count per
                // (group, datum); insert visible iff count 0→1, delete iff
                // 1→0; a count that drops to 0 is removed (:185-191)
                for (size_t di = 0; di < distinct_cols.size(); di++) {
                    Row dk = key;
                    dk.push_back(cv.at(r, distinct_cols[di]));
                    dedup_touched[di].insert(dk);
                    auto& m = dedup_counts[di];
                    if (!retract) {
                        int64_t cnt = ++m[dk];
                        col_hidden[di] = cnt > 1;
                    } else {
                        auto it = m.find(dk);
                        int64_t cnt = it == m.end() ? -1 : --it->second;
                        col_hidden[di] = cnt > 0;
                        if (it != m.end() && cnt == 0) m.erase(it);
                    }
                }
                hidden.assign(calls.size(), false);
                for (size_t ci = 0; ci < calls.size(); ci++)
                    if (call_dedup_idx[ci] >= 0)
                        hidden[ci] = col_hidden[call_dedup_idx[ci]];
            }
            int rc = apply_row(g, key, cv, r, retract, hidden);
            if (rc != RW_OK) return rc;
        }
        return RW_OK;
    }

    int64_t row_count_of(const Row& outputs) const {
        // row_count_of (agg_group.rs:55-79): clamp negatives to 0
        const Datum& d = outputs[row_count_index];
        if (d.null) return 0; // "should not be NULL"; tolerate like release mode
        return d.i < 0 ? 0 : d.i;
    }

    // get_outputs (agg_group.rs:431-467)
    // Rebuild state from drained spill records (rw_stream.h restore
    // contract): net PUT/DELETE frames by key, decode the value-encoded
    // row (group key ++ outputs), and seed the value states + prev
    // outputs so the next flush emits only real changes.
    // For executors with materialized-input aggregates, call
    // rw_agg_minput_restore for EVERY minput table BEFORE this (the
    // reference restores prev outputs from the hydrated minput tables,
    // agg_group.rs:219-221 + minput.rs first-entry output).
    int restore(const uint8_t* buf, uint64_t len) {
        if (!dirty.empty())
            FAIL(RW_E_INVAL, "restore requires a fresh executor");
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
        for (auto& [kbytes, val] : merged) {
            (void)kbytes;
            // decode group key ++ per-call outputs
            size_t off = 0;
            Row key(group_key_types.size());
            for (size_t i = 0; i < group_key_types.size(); i++) {
                rwcodec::DatumC d;
                size_t n = rwcodec::value_decode_datum(
                    val.data() + off, val.size() - off, group_key_types[i],
                    &d);
                if (!n) FAIL(RW_E_INVAL, "restore: bad group datum");
                off += n;
                key[i] = d.null ? Datum()
                                : (type_is_float(group_key_types[i])
                                       ? Datum::of_d(d.d)
                                       : Datum::of_i(d.i));
            }
            Row outs(calls.size());
            for (size_t ci = 0; ci < calls.size(); ci++) {
                rwcodec::DatumC d;
                size_t n = rwcodec::value_decode_datum(
                    val.data() + off, val.size() - off, calls[ci].ret_type,
                    &d);
                if (!n) FAIL(RW_E_INVAL, "restore: bad state datum");
                off += n;
                outs[ci] = d.null ? Datum()
                                  : (calls[ci].ret_type == RW_T_DECIMAL
                                         ? Datum::of_dec(d.i, d.i2)
                                         : (type_is_float(calls[ci].ret_type)
                                                ? Datum::of_d(d.d)
                                                : Datum::of_i(d.i)));
            }
            AggGroupState& g = touch(key);
            for (size_t ci = 0; ci < calls.size(); ci++) {
                if (call_is_minput[ci]) {
                    // prev output = first entry of the (already restored)
                    // minput table; the intermediate record stores None
                    const auto& m = g.mstates[ci].entries;
                    outs[ci] = m.empty() ? Datum() : m.begin()->second;
                    continue;
                }
                ValueState& v2 = g.vstates[ci];
                const Datum& o = outs[ci];
                v2.has = !o.null;
                if (!o.null && call_is_decimal(ci)) {
                    DecSumState& ds = g.dstates[ci];
                    ds = DecSumState{};
                    DecVal dv = dec_parse(o.i, o.i2);
                    if (dv.special == 1) ds.nan = 1;
                    else if (dv.special == 2) ds.pinf = 1;
                    else if (dv.special == 3) ds.ninf = 1;
                    else {
                        ds.sum = dec_addend(dv);
                        ds.maxscale = dv.scale;
                    }
                    continue;
                }
                if (!o.null) {
                    if (v2.is_float) v2.d = o.d;
                    else v2.i = o.i;
                }
                // count/sum0 state is always Some
                if (calls[ci].kind == RW_AGG_COUNT_STAR ||
                    calls[ci].kind == RW_AGG_COUNT ||
                    calls[ci].kind == RW_AGG_SUM0)
                    v2.has = true;
            }
            if (!eowc) {
                g.has_prev = true;
                g.prev_outputs = outs;
            }
        }
        // restored groups are clean (they predate the epoch)
        dirty_order.clear();
        dirty.clear();
        return RW_OK;
    }

    Row get_outputs(AggGroupState& g) {
        // current row count from the count(*) value state
        int64_t rc = g.vstates[row_count_index].i;
        if (rc < 0) rc = 0;
        if (rc == 0) {
            // reset value states only (agg_state.rs:149-155)
            for (size_t ci = 0; ci < calls.size(); ci++) {
                if (call_is_minput[ci]) continue;
                ValueState v;
                const auto& c = calls[ci];
                if (c.kind == RW_AGG_COUNT_STAR || c.kind == RW_AGG_COUNT ||
                    c.kind == RW_AGG_SUM0)
                    v.has = true;
                v.is_float = g.vstates[ci].is_float;
                g.vstates[ci] = v;
                g.dstates[ci] = DecSumState{};
            }
        }
        Row out(calls.size());
        for (size_t ci = 0; ci < calls.size(); ci++) {
            const auto& c = calls[ci];
            if (call_is_minput[ci]) {
                const auto& m = g.mstates[ci].entries;
                out[ci] = m.empty() ? Datum() : m.begin()->second;
                continue;
            }
            const auto& v = g.vstates[ci];
            switch (c.kind) {
                case RW_AGG_COUNT_STAR:
                case RW_AGG_COUNT:
                case RW_AGG_SUM0: out[ci] = Datum::of_i(v.i); break;
                case RW_AGG_SUM:
                    if (call_is_decimal(ci)) {
                        if (!v.has) {
                            out[ci] = Datum();
                            break;
                        }
                        const DecSumState& ds = g.dstates[ci];
                        DecVal r2;
                        if (ds.nan > 0 || (ds.pinf > 0 && ds.ninf > 0)) {
                            r2.special = 1; // NaN (decimal.rs:259-276)
                        } else if (ds.pinf > 0) {
                            r2.special = 2;
                        } else if (ds.ninf > 0) {
                            r2.special = 3;
                        } else if (!dec_from_sum(ds.sum, ds.maxscale, &r2)) {
                            throw std::runtime_error(
                                "decimal sum outside the exact 96-bit "
                                "domain (the reference's order-dependent "
                                "rescale path; unsupported)");
                        }
                        int64_t a2, b2;
                        dec_serialize(r2, &a2, &b2);
                        out[ci] = Datum::of_dec(a2, b2);
                        break;
                    }
                    out[ci] = v.has ? (v.is_float ? Datum::of_d(v.d) : Datum::of_i(v.i))
                                    : Datum();
                    break;
                case RW_AGG_MIN:
                case RW_AGG_MAX:
                    if (!v.has) out[ci] = Datum();
                    else out[ci] = v.is_float ? Datum::of_d(v.d) : Datum::of_i(v.i);
                    break;
            }
        }
        return out;
    }

    // append one state-table KV delta (StateTable::commit spill boundary,
    // state_table.rs:1718): PUT(key,row) for Insert/Update, DELETE(key) for
    // Delete. Row = group key ++ encoded states (value-state calls carry the
    // value; materialized-input states encode None — agg_group.rs:417-421).
    void spill_record(uint8_t put, const Row& key, const Row& outputs) {
        spill.push_back(put);
        std::vector<uint8_t> k, v;
        for (size_t i = 0; i < key.size(); i++) {
            rwcodec::DatumC d{key[i].null, key[i].i, key[i].d};
            rwcodec::memcmp_encode_datum(k, group_key_types[i], d, {});
        }
        if (put) {
            for (size_t i = 0; i < key.size(); i++) {
                rwcodec::DatumC d{key[i].null, key[i].i, key[i].d};
                rwcodec::value_encode_datum(v, group_key_types[i], d);
            }
            for (size_t ci = 0; ci < calls.size(); ci++) {
                if (call_is_minput[ci]) {
                    rwcodec::value_encode_datum(v, calls[ci].ret_type,
                                                {true, 0, 0});
                } else {
                    const Datum& o = outputs[ci];
                    rwcodec::value_encode_datum(v, calls[ci].ret_type,
                                                {o.null, o.i, o.d, o.i2});
                }
            }
        }
        auto put32 = [&](uint32_t x) {
            for (int b = 0; b < 4; b++) spill.push_back((uint8_t)(x >> (8 * b)));
        };
        put32((uint32_t)k.size());
        spill.insert(spill.end(), k.begin(), k.end());
        put32((uint32_t)v.size());
        spill.insert(spill.end(), v.begin(), v.end());
    }

    // §8f-2 drain for one dedup table: one record per (group, datum)
    // touched this epoch — PUT with the current counts, DELETE when the
    // count dropped to 0 and the row was persisted, nothing when it was
    // created and died within the epoch (mem-table netting). Sorted by pk
    // (the canonical form; the reference iterates a HashMap).
    int dedup_drain(int di, std::vector<uint8_t>& out) {
        if (di < 0 || (size_t)di >= distinct_cols.size()) return RW_E_INVAL;
        std::vector<uint8_t> key_types(group_key_types);
        key_types.push_back(input_types[distinct_cols[di]]);
        auto put32 = [&](uint32_t x) {
            for (int b = 0; b < 4; b++) out.push_back((uint8_t)(x >> (8 * b)));
        };
        auto& counts = dedup_counts[di];
        auto& persisted = dedup_persisted[di];
        for (const Row& dk : dedup_touched[di]) {
            auto it = counts.find(dk);
            int64_t cnt = it == counts.end() ? 0 : it->second;
            uint8_t put;
            if (cnt > 0) {
                put = 1;
                persisted.insert(dk);
            } else if (persisted.count(dk)) {
                put = 0;
                persisted.erase(dk);
            } else {
                continue;
            }
            std::vector<uint8_t> k, v;
            for (size_t c = 0; c < dk.size(); c++) {
                rwcodec::DatumC d{dk[c].null, dk[c].i, dk[c].d};
                rwcodec::memcmp_encode_datum(k, key_types[c], d, {});
            }
            if (put) {
                for (size_t c = 0; c < dk.size(); c++) {
                    rwcodec::DatumC d{dk[c].null, dk[c].i, dk[c].d};
                    rwcodec::value_encode_datum(v, key_types[c], d);
                }
                for (size_t ci = 0; ci < calls.size(); ci++)
                    if (call_dedup_idx[ci] == di)
                        rwcodec::value_encode_datum(v, RW_T_I64,
                                                    {false, cnt, 0});
            }
            out.push_back(put);
            put32((uint32_t)k.size());
            out.insert(out.end(), k.begin(), k.end());
            put32((uint32_t)v.size());
            out.insert(out.end(), v.begin(), v.end());
        }
        dedup_touched[di].clear();
        return RW_OK;
    }

    // §8f-5 recovery for one dedup table: net concatenated drain frames
    // and seed the counter map; restored keys are persisted (a later drop
    // to 0 drains as DELETE, exactly as uninterrupted).
    int dedup_restore(int di, const uint8_t* buf, uint64_t len) {
        if (di < 0 || (size_t)di >= distinct_cols.size()) return RW_E_INVAL;
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
        if (!ok) {
            g_err = "malformed spill stream";
            return RW_E_INVAL;
        }
        std::vector<uint8_t> key_types(group_key_types);
        key_types.push_back(input_types[distinct_cols[di]]);
        for (auto& [kbytes, val] : merged) {
            (void)kbytes;
            Row dk;
            size_t off = 0;
            for (size_t c = 0; c < key_types.size(); c++) {
                rwcodec::DatumC d;
                size_t got = rwcodec::value_decode_datum(
                    val.data() + off, val.size() - off, key_types[c], &d);
                if (!got) {
                    g_err = "dedup restore: bad key datum";
                    return RW_E_INVAL;
                }
                off += got;
                Datum dm;
                dm.null = d.null;
                if (key_types[c] == RW_T_F64) dm.d = d.d;
                else dm.i = d.i;
                dk.push_back(dm);
            }
            rwcodec::DatumC d;
            size_t got = rwcodec::value_decode_datum(
                val.data() + off, val.size() - off, RW_T_I64, &d);
            if (!got || d.null) {
                g_err = "dedup restore: bad count datum";
                return RW_E_INVAL;
            }
            dedup_counts[di][dk] = d.i;
            dedup_persisted[di].insert(dk);
        }
        return RW_OK;
    }

    void emit(uint8_t op, const Row& key, const Row& outputs) {
        Row row;
        row.reserve(key.size() + outputs.size());
        for (auto& d : key) row.push_back(d);
        for (auto& d : outputs) row.push_back(d);
        std::unique_ptr<OwnedChunk> full;
        if (builder.append_row(op, row, &full)) outputs_push(std::move(full));
    }
    void outputs_push(std::unique_ptr<OwnedChunk> c) {
        if (c) outputs.push_back(std::move(c));
    }

    // flush_data, emit-on-update branch (hash_agg.rs:475-501)
    int flush(uint64_t epoch) {
        try {
            return flush_inner(epoch);
        } catch (const std::runtime_error& e) {
            g_err = e.what();
            return RW_E_OVERFLOW;
        }
    }

    int flush_inner(uint64_t /*epoch*/) {
        if (eowc) return flush_eowc();
        for (auto& key : dirty_order) {
            auto& g = groups[key];
            Row curr = get_outputs(g);
            int64_t prev_rc = g.has_prev ? row_count_of(g.prev_outputs) : 0;
            int64_t curr_rc = row_count_of(curr);
            // OnlyOutputIfHasInput::infer_change_type (agg_group.rs:131-165)
            if (prev_rc == 0 && curr_rc == 0) {
                // nothing
            } else if (prev_rc == 0) {
                emit(RW_OP_INSERT, key, curr);
                spill_record(1, key, curr);
                g.prev_outputs = curr;
                g.has_prev = true;
            } else if (curr_rc == 0) {
                emit(RW_OP_DELETE, key, g.prev_outputs);
                spill_record(0, key, g.prev_outputs);
                g.has_prev = false;
                g.prev_outputs.clear();
            } else {
                std::vector<uint8_t> call_types;
                for (auto& c : calls) call_types.push_back(c.ret_type);
                if (!row_eq(g.prev_outputs, curr, call_types)) {
                    emit(RW_OP_UPDATE_DELETE, key, g.prev_outputs);
                    emit(RW_OP_UPDATE_INSERT, key, curr);
                    spill_record(1, key, curr);
                    g.prev_outputs = curr;
                }
            }
        }
        dirty_order.clear();
        dirty.clear();
        outputs_push(builder.take());
        return RW_OK;
    }

    // EOWC flush (hash_agg.rs:429-474): windows with group-key[0] below the
    // buffered watermark emit their FINAL row once (Insert; row_count 0
    // emits nothing — OnlyOutputIfHasInput with no prev) in group-key-sorted
    // order (SortBuffer::consume iterates ordered) and are removed. Dirty
    // state rows spill as PUTs; closed groups spill DELETEs.
    int flush_eowc() {
        // mid-window state PUTs (hash_agg.rs:429-460): every dirty group's
        // CURRENT states upsert into the intermediate table at the barrier
        // even though nothing is emitted until the window closes
        for (auto& key : dirty_order) {
            auto& g = groups[key];
            Row curr = get_outputs(g);
            spill_record(1, key, curr);
        }
        dirty_order.clear();
        dirty.clear();
        if (has_pending_wm) {
            std::vector<Row> closing;
            for (auto& [key, g] : groups) {
                const Datum& d = key[0];
                if (!d.null && d.i < pending_wm) closing.push_back(key);
            }
            RowOrderLess less;
            for (auto t : group_key_types) less.order.push_back({t, false});
            std::sort(closing.begin(), closing.end(), less);
            for (auto& key : closing) {
                auto& g = groups[key];
                Row curr = get_outputs(g);
                // every closed window leaves the intermediate table (the
                // reference deletes all rows < wm); rc==0 windows delete
                // their mid-window row without emitting
                spill_record(0, key, curr);
                if (row_count_of(curr) != 0) emit(RW_OP_INSERT, key, curr);
                groups.erase(key);
            }
            has_pending_wm = false;
        }
        outputs_push(builder.take());
        return RW_OK;
    }

    // watermark TTL cleaning (hash_agg.rs:503-507 → update_watermark):
    // groups whose watermarked group-key column sorts below the value are
    // dropped (NULLs largest, kept)
    int watermark(uint32_t pos, int64_t val) {
        if (pos >= group_key.size()) return RW_E_INVAL;
        if (eowc) {
            // EOWC: buffer the window watermark; windows close at the next
            // barrier (hash_agg.rs:657-700 buffers into window_watermark)
            if (pos != 0) return RW_E_INVAL;
            if (!has_pending_wm || val > pending_wm) pending_wm = val;
            has_pending_wm = true;
            return RW_OK;
        }
        for (auto it = groups.begin(); it != groups.end();) {
            const Datum& d = it->first[pos];
            if (!d.null && d.i < val) {
                // state-cleaning spill deltas (the reference's commit
                // applies the watermark as a range delete on the store,
                // state_table.rs:1707): persisted groups net to DELETE
                // so a restore replay cannot resurrect them
                if (it->second.has_prev)
                    spill_record(0, it->first, it->second.prev_outputs);
                // minput state-table rows of the cleaned group
                for (size_t ci = 0; ci < calls.size(); ci++) {
                    if (!call_is_minput[ci] ||
                        ci >= it->second.mstates.size())
                        continue;
                    int mi = minput_ordinal(ci);
                    if ((size_t)mi >= minput_delta.size())
                        minput_delta.resize(mi + 1);
                    auto& delta = minput_delta[mi];
                    for (auto& [ekey, dat] : it->second.mstates[ci].entries) {
                        (void)dat;
                        std::string kb;
                        minput_encode(ci, it->first, ekey, &kb, nullptr);
                        auto di = delta.find(kb);
                        if (di != delta.end() && di->second.first == 1)
                            delta.erase(di); // created+died this epoch
                        else
                            delta[kb] = {0, {}};
                    }
                }
                dirty.erase(it->first);
                it = groups.erase(it);
            } else {
                ++it;
            }
        }
        for (auto it = dirty_order.begin(); it != dirty_order.end();)
            if (!groups.count(*it)) it = dirty_order.erase(it);
            else ++it;
        // the same watermark cleans the DISTINCT dedup tables' group
        // prefix: counts reset (late rows restart visibility from 0) and
        // the next dedup drain nets persisted rows to DELETE
        for (size_t di = 0; di < dedup_counts.size(); di++) {
            auto& counts = dedup_counts[di];
            for (auto it2 = counts.begin(); it2 != counts.end();) {
                const Datum& d2 = it2->first[pos];
                if (!d2.null && d2.i < val) {
                    dedup_touched[di].insert(it2->first);
                    it2 = counts.erase(it2);
                } else {
                    ++it2;
                }
            }
        }
        return RW_OK;
    }

    RwChunk* poll() {
        if (outputs.empty()) return nullptr;
        auto c = std::move(outputs.front());
        outputs.erase(outputs.begin());
        return chunk_to_c(*c);
    }

    // rescale re-scope (update_vnode_bitmap): drop groups whose dist-key
    // vnode (dist key = group key) is no longer owned; no retractions —
    // the new owner holds the state
    int update_vnode_bitmap(const uint8_t* bm, uint32_t vnode_count) {
        for (auto it = groups.begin(); it != groups.end();) {
            uint32_t vn = vnode_of_key_row(it->first, group_key_types,
                                           vnode_count);
            if ((bm[vn >> 3] >> (vn & 7)) & 1) {
                ++it;
            } else {
                dirty.erase(it->first);
                it = groups.erase(it);
            }
        }
        for (auto it = dirty_order.begin(); it != dirty_order.end();)
            if (!groups.count(*it)) it = dirty_order.erase(it);
            else ++it;
        return RW_OK;
    }
};

} // namespace orc

using namespace orc;

extern "C" {

const char* rw_last_error(void) { return g_err.c_str(); }
void rw_chunk_free(RwChunk* c) { chunk_free_c(c); }

void* rw_hash_agg_create(const RwHashAggDesc* d) {
    std::vector<uint8_t> out_types;
    for (uint32_t i = 0; i < d->n_group_key; i++)
        out_types.push_back(d->input_types[d->group_key_indices[i]]);
    for (uint32_t i = 0; i < d->n_calls; i++) out_types.push_back(d->calls[i].ret_type);
    try {
        return new HashAggOracle(d, out_types);
    } catch (const std::invalid_argument& e) {
        g_err = e.what();
        return nullptr;
    }
}
int rw_hash_agg_push_chunk(void* h, const RwChunk* c) {
    return ((HashAggOracle*)h)->push_chunk(c);
}
int rw_hash_agg_flush(void* h, uint64_t epoch) { return ((HashAggOracle*)h)->flush(epoch); }
int rw_hash_agg_watermark(void* h, uint32_t pos, int64_t val) {
    return ((HashAggOracle*)h)->watermark(pos, val);
}
int rw_hash_agg_update_vnode_bitmap(void* h, const uint8_t* bitmap,
                                    uint32_t vnode_count) {
    if (!vnode_count || vnode_count % 8) return RW_E_INVAL;
    return ((HashAggOracle*)h)->update_vnode_bitmap(bitmap, vnode_count);
}

// drain the checkpoint spill buffer (caller frees with rw_spill_free)
int rw_agg_checkpoint_drain(void* h, uint8_t** buf, uint64_t* len) {
    auto* a = (HashAggOracle*)h;
    // memcmp-key order (canonical drain form; matches the GPU library)
    orc::sort_spill_frames(a->spill);
    *len = a->spill.size();
    *buf = (uint8_t*)malloc(a->spill.size() ? a->spill.size() : 1);
    memcpy(*buf, a->spill.data(), a->spill.size());
    a->spill.clear();
    return RW_OK;
}

int rw_hash_agg_restore(void* h, const uint8_t* buf, uint64_t len) {
    return ((HashAggOracle*)h)->restore(buf, len);
}

int rw_agg_n_minput_tables(void* h) {
    return ((HashAggOracle*)h)->n_minput_total();
}
int rw_agg_minput_drain(void* h, int mi, uint8_t** buf, uint64_t* len) {
    auto* a = (HashAggOracle*)h;
    std::vector<uint8_t> sp;
    int rc = a->minput_drain(mi, sp);
    if (rc != RW_OK) return rc;
    *len = sp.size();
    *buf = (uint8_t*)malloc(sp.size() ? sp.size() : 1);
    if (!*buf) return RW_E_INTERNAL;
    memcpy(*buf, sp.data(), sp.size());
    return RW_OK;
}
int rw_agg_minput_restore(void* h, int mi, const uint8_t* buf, uint64_t len) {
    return ((HashAggOracle*)h)->minput_restore(mi, buf, len);
}

int rw_agg_n_dedup_tables(void* h) {
    return (int)((HashAggOracle*)h)->distinct_cols.size();
}

int rw_agg_dedup_drain(void* h, int di, uint8_t** buf, uint64_t* len) {
    auto* a = (HashAggOracle*)h;
    std::vector<uint8_t> sp;
    int rc = a->dedup_drain(di, sp);
    if (rc != RW_OK) return rc;
    *len = sp.size();
    *buf = (uint8_t*)malloc(sp.size() ? sp.size() : 1);
    memcpy(*buf, sp.data(), sp.size());
    return RW_OK;
}

int rw_agg_dedup_restore(void* h, int di, const uint8_t* buf, uint64_t len) {
    return ((HashAggOracle*)h)->dedup_restore(di, buf, len);
}

int rw_agg_minput_compact(void* h, uint64_t* reclaimed) {
    // the oracle's ordered maps hold no dead rows — no-op (see
    // rw_join_compact)
    if (!h) return RW_E_INVAL;
    if (reclaimed) *reclaimed = 0;
    return RW_OK;
}

void rw_spill_free(uint8_t* buf) { free(buf); }
RwChunk* rw_hash_agg_poll(void* h) { return ((HashAggOracle*)h)->poll(); }
void rw_hash_agg_destroy(void* h) { delete (HashAggOracle*)h; }

} // extern "C"
