// oracle/oracle_join.cpp — CPU restatement of HashJoinExecutor.
// ORACLE — TEST INFRASTRUCTURE ONLY (see common.hpp header note).
//
// Restates (reference under /root/reference):
//  - eq_join_oneside / handle_match_rows / handle_match_row
//    (src/stream/src/executor/hash_join.rs:949-1374): per visible probe row —
//    null-safe NeverMatch check (:1004-1016; NeverMatch rows are forwarded
//    and NOT written to own state, :1143-1152), matched-row loop in
//    memcomparable deduped-pk order (cache BTreeMap join/join_row_set.rs /
//    state-table prefix scan join/hash_join.rs:263-288 iterate identically),
//    non-equi condition (:1294-1302, NULL ⇒ false :1388-1403), degree
//    bookkeeping: on Insert emit BEFORE the matched row's degree increment,
//    on Delete emit AFTER the decrement (:1311-1342; update_degree
//    join/hash_join.rs:355-380), probe-row forwarding by its own match count
//    (:1216-1227), append-only optimize: delete the single matched row and
//    skip own insert (:1241-1245,1359-1364).
//  - Output op selection per join type: JoinChunkBuilder
//    (join/builder.rs:153-318) incl. outer-side NULL transitions and
//    eliminate_adjacent_noop_update on every yielded chunk (:166-168).
//  - State mutation: JoinHashMap::insert/delete keyed by (join key,
//    memcmp deduped pk) (join/hash_join.rs:578-681). Duplicate-pk inserts
//    overwrite (non-strict consistency mode, src/stream/src/consistency.rs).
#include <algorithm>
#include <map>
#include <optional>
#include <memory>
#include <string>
#include <vector>

#include "../include/rw_stream.h"
#include "common.hpp"
#include "../include/rw_codec.hpp"

namespace orc {

extern thread_local std::string g_err;

#define FAIL(code, ...)                         \
    do {                                        \
        char _b[256];                           \
        snprintf(_b, sizeof(_b), __VA_ARGS__);  \
        g_err = _b;                             \
        return code;                            \
    } while (0) // defined in oracle_agg.cpp

// join type predicates (executor/join/mod.rs:103-165)
static bool is_outer_side(uint8_t t, int side) {
    return t == RW_JOIN_FULL_OUTER || (t == RW_JOIN_LEFT_OUTER && side == RW_SIDE_LEFT) ||
           (t == RW_JOIN_RIGHT_OUTER && side == RW_SIDE_RIGHT);
}
static bool outer_side_null(uint8_t t, int side) {
    return t == RW_JOIN_FULL_OUTER || (t == RW_JOIN_LEFT_OUTER && side == RW_SIDE_RIGHT) ||
           (t == RW_JOIN_RIGHT_OUTER && side == RW_SIDE_LEFT);
}
static bool forward_exactly_once(uint8_t t, int side) {
    return ((t == RW_JOIN_LEFT_SEMI || t == RW_JOIN_LEFT_ANTI) && side == RW_SIDE_LEFT) ||
           ((t == RW_JOIN_RIGHT_SEMI || t == RW_JOIN_RIGHT_ANTI) && side == RW_SIDE_RIGHT);
}
static bool only_forward_matched_side(uint8_t t, int side) {
    return ((t == RW_JOIN_LEFT_SEMI || t == RW_JOIN_LEFT_ANTI) && side == RW_SIDE_RIGHT) ||
           ((t == RW_JOIN_RIGHT_SEMI || t == RW_JOIN_RIGHT_ANTI) && side == RW_SIDE_LEFT);
}
static bool is_semi(uint8_t t) { return t == RW_JOIN_LEFT_SEMI || t == RW_JOIN_RIGHT_SEMI; }
static bool is_anti(uint8_t t) { return t == RW_JOIN_LEFT_ANTI || t == RW_JOIN_RIGHT_ANTI; }
static bool need_degree(uint8_t t, int side) {
    // need_left_degree / need_right_degree (join/mod.rs:153-165)
    if (side == RW_SIDE_LEFT)
        return t == RW_JOIN_FULL_OUTER || t == RW_JOIN_LEFT_OUTER || t == RW_JOIN_LEFT_ANTI ||
               t == RW_JOIN_LEFT_SEMI;
    return t == RW_JOIN_FULL_OUTER || t == RW_JOIN_RIGHT_OUTER || t == RW_JOIN_RIGHT_ANTI ||
           t == RW_JOIN_RIGHT_SEMI;
}

struct JoinEntry {
    Row row;
    uint64_t degree = 0;
};

struct JoinSideState {
    std::vector<uint32_t> key_idx;
    std::vector<uint32_t> pk_idx; // deduped pk indices into the input row
    std::vector<uint8_t> types;
    bool need_deg = false;
    // jk → (pk → entry); both in memcomparable (ASC, NULLs largest) order
    std::map<Row, std::map<Row, JoinEntry, RowOrderLess>, RowOrderLess> table;
    RowOrderLess pk_less;
};

// The join-type/side-dependent output builder (join/builder.rs:153-318).
struct JoinOutBuilder {
    ChunkBuilder b;
    std::vector<std::pair<uint32_t, uint32_t>> update_to_output, matched_to_output;
    std::vector<std::unique_ptr<OwnedChunk>>* outputs;
    JoinOutBuilder(size_t chunk_size, std::vector<uint8_t> out_types,
                   std::vector<std::pair<uint32_t, uint32_t>> u2o,
                   std::vector<std::pair<uint32_t, uint32_t>> m2o,
                   std::vector<std::unique_ptr<OwnedChunk>>* out)
        // chunk size forced >= 2 (join/builder.rs:44-47)
        : b(std::max<size_t>(chunk_size, 2), out_types),
          update_to_output(std::move(u2o)),
          matched_to_output(std::move(m2o)),
          outputs(out) {}

    size_t width() const { return b.types.size(); }
    void post(std::unique_ptr<OwnedChunk> c) {
        if (!c) return;
        eliminate_adjacent_noop_update(*c); // JoinChunkBuilder::post_process
        outputs->push_back(std::move(c));
    }
    void append_row(uint8_t op, const Row& upd, const Row& match) {
        Row out(width());
        for (auto& [ui, oi] : update_to_output) out[oi] = upd[ui];
        for (auto& [mi, oi] : matched_to_output) out[oi] = match[mi];
        std::unique_ptr<OwnedChunk> full;
        if (b.append_row(op, out, &full)) post(std::move(full));
    }
    void append_row_update(uint8_t op, const Row& upd) {
        Row out(width()); // other side NULL
        for (auto& [ui, oi] : update_to_output) out[oi] = upd[ui];
        std::unique_ptr<OwnedChunk> full;
        if (b.append_row(op, out, &full)) post(std::move(full));
    }
    void append_row_matched(uint8_t op, const Row& match) {
        Row out(width());
        for (auto& [mi, oi] : matched_to_output) out[oi] = match[mi];
        std::unique_ptr<OwnedChunk> full;
        if (b.append_row(op, out, &full)) post(std::move(full));
    }
    void take() { post(b.take()); }
};

struct WmEntry {
    bool has = false;
    int64_t val = 0;
};

struct HashJoinOracle {
    RwHashJoinDesc d_store;
    // join-key watermark buffering (hash_join.rs:826-867; BufferedWatermarks
    // emits the min across sides when it advances)
    std::vector<uint32_t> wm_pos;
    std::vector<uint8_t> wm_clean;
    std::vector<WmEntry> wm_side[2];
    std::vector<WmEntry> wm_out;
    // inequality-pair watermarks (hash_join.rs:869-914)
    std::vector<uint32_t> ineq_col[2];
    std::vector<uint8_t> ineq_larger, ineq_do_clean;
    std::vector<WmEntry> ineq_wm[2], ineq_out;
    uint8_t T;
    bool append_only;
    std::vector<uint8_t> null_safe;
    std::vector<uint8_t> types_l, types_r, concat_types, out_types;
    std::vector<uint32_t> output_indices;
    JoinSideState side[2];
    std::vector<std::pair<uint32_t, uint32_t>> l2o, m_r2o; // left/right → output
    std::vector<std::unique_ptr<OwnedChunk>> outputs;

    HashJoinOracle(const RwHashJoinDesc* d) {
        T = d->join_type;
        append_only = d->append_only;
        null_safe.assign(d->null_safe, d->null_safe + d->n_key);
        types_l.assign(d->types_l, d->types_l + d->n_cols_l);
        types_r.assign(d->types_r, d->types_r + d->n_cols_r);
        concat_types = types_l;
        concat_types.insert(concat_types.end(), types_r.begin(), types_r.end());
        output_indices.assign(d->output_indices, d->output_indices + d->n_output);
        for (auto i : output_indices) out_types.push_back(concat_types[i]);
        d_store = *d;
        if (d->n_wm_jk) {
            wm_pos.assign(d->wm_jk_pos, d->wm_jk_pos + d->n_wm_jk);
            wm_clean.assign(d->wm_jk_clean, d->wm_jk_clean + d->n_wm_jk);
        }
        wm_side[0].resize(d->n_key);
        wm_side[1].resize(d->n_key);
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

        side[0].key_idx.assign(d->key_l, d->key_l + d->n_key);
        side[1].key_idx.assign(d->key_r, d->key_r + d->n_key);
        side[0].pk_idx.assign(d->pk_l, d->pk_l + d->n_pk_l);
        side[1].pk_idx.assign(d->pk_r, d->pk_r + d->n_pk_r);
        side[0].types = types_l;
        side[1].types = types_r;
        side[0].need_deg = need_degree(T, RW_SIDE_LEFT);
        side[1].need_deg = need_degree(T, RW_SIDE_RIGHT);
        for (int s = 0; s < 2; s++) {
            RowOrderLess key_less, pk_less;
            for (auto k : side[s].key_idx) key_less.order.push_back({side[s].types[k], false});
            for (auto p : side[s].pk_idx) pk_less.order.push_back({side[s].types[p], false});
            side[s].table = decltype(side[s].table)(key_less);
            side[s].pk_less = pk_less;
        }
        // get_i2o_mapping (join/builder.rs:63-81)
        for (uint32_t oi = 0; oi < output_indices.size(); oi++) {
            uint32_t idx = output_indices[oi];
            if (idx < types_l.size()) l2o.push_back({idx, oi});
            else m_r2o.push_back({idx - (uint32_t)types_l.size(), oi});
        }
    }

    bool cond_ok(const Row& upd, const Row& match, int probe_side) const {
        if (!d_store.has_cond) return true;
        // row_concat (hash_join.rs:917-932): left part at 0, right at n_cols_l
        const Row& lrow = probe_side == RW_SIDE_LEFT ? upd : match;
        const Row& rrow = probe_side == RW_SIDE_LEFT ? match : upd;
        auto at = [&](uint32_t i) -> const Datum& {
            return i < types_l.size() ? lrow[i] : rrow[i - types_l.size()];
        };
        auto one = [&](uint8_t op, uint32_t li, uint32_t ri,
                       long long rconst) -> bool {
            const Datum& a = at(li);
            Datum b = at(ri);
            if (a.null || b.null) return false; // NULL comparison ⇒ false
            b.i += rconst; // constant offset on the right operand
            uint8_t t = concat_types[li];
            int c = datum_cmp(a, b, t);
            switch (op) {
                case RW_CMP_LT: return c < 0;
                case RW_CMP_LE: return c <= 0;
                case RW_CMP_GT: return c > 0;
                case RW_CMP_GE: return c >= 0;
            }
            return false;
        };
        if (!one(d_store.cond_op, d_store.cond_l, d_store.cond_r,
                 d_store.cond_rconst))
            return false;
        if (d_store.has_cond2 &&
            !one(d_store.cond2_op, d_store.cond2_l, d_store.cond2_r,
                 d_store.cond2_rconst))
            return false;
        return true;
    }

    // Rebuild one side's state from drained spill records (rw_stream.h
    // restore contract): net PUT/DELETE frames by key, decode the
    // value-encoded full row, re-derive jk/pk, and attach degrees from the
    // degree-table drain stream (matched by the shared memcmp(jk ∥ pk)
    // key encoding, join/row.rs:99-113).
    int restore(int S, const uint8_t* buf, uint64_t len, const uint8_t* dbuf,
                uint64_t dlen) {
        JoinSideState& sd = side[S];
        if (!sd.table.empty())
            FAIL(RW_E_INVAL, "restore requires a fresh executor side");
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
        std::map<std::string, uint64_t> degs;
        if (dbuf && dlen) {
            ok = rwcodec::for_each_frame(
                dbuf, dlen,
                [&](uint8_t put, const uint8_t* k, uint32_t klen,
                    const uint8_t* v, uint32_t vlen) {
                    std::string key((const char*)k, klen);
                    if (put && vlen >= 9 && v[vlen - 9] == 1) {
                        uint64_t d = 0;
                        for (int b = 0; b < 8; b++)
                            d |= (uint64_t)v[vlen - 8 + b] << (8 * b);
                        degs[key] = d;
                    } else if (!put) {
                        degs.erase(key);
                    }
                });
            if (!ok) FAIL(RW_E_INVAL, "malformed degree spill stream");
        }
        for (auto& [kbytes, val] : merged) {
            Row row(sd.types.size());
            size_t off = 0;
            for (size_t c = 0; c < sd.types.size(); c++) {
                rwcodec::DatumC d;
                size_t n = rwcodec::value_decode_datum(
                    val.data() + off, val.size() - off, sd.types[c], &d);
                if (!n) FAIL(RW_E_INVAL, "restore: bad row datum");
                off += n;
                row[c] = d.null ? Datum()
                                : (type_is_float(sd.types[c])
                                       ? Datum::of_d(d.d)
                                       : Datum::of_i(d.i));
            }
            Row jk(sd.key_idx.size()), pk(sd.pk_idx.size());
            for (size_t i = 0; i < sd.key_idx.size(); i++)
                jk[i] = row[sd.key_idx[i]];
            for (size_t i = 0; i < sd.pk_idx.size(); i++)
                pk[i] = row[sd.pk_idx[i]];
            JoinEntry e;
            e.row = std::move(row);
            auto di = degs.find(kbytes);
            if (di != degs.end()) e.degree = di->second;
            side_entry(sd, jk).emplace(std::move(pk), std::move(e));
        }
        return RW_OK;
    }

    int push_chunk(int S, const RwChunk* chunk) {
        ChunkView cv{chunk};
        JoinSideState& upd_side = side[S];
        JoinSideState& match_side = side[1 - S];
        // builder: update = probe side mapping, matched = other side
        JoinOutBuilder b(d_store.chunk_size, out_types, S == RW_SIDE_LEFT ? l2o : m_r2o,
                         S == RW_SIDE_LEFT ? m_r2o : l2o, &outputs);

        for (size_t r = 0; r < cv.n_rows(); r++) {
            if (!cv.visible(r)) continue;
            uint8_t in_op = cv.op(r);
            bool is_insert = (in_op == RW_OP_INSERT || in_op == RW_OP_UPDATE_INSERT);
            uint8_t op = is_insert ? RW_OP_INSERT : RW_OP_DELETE; // U± downgraded
            Row row = cv.row(r);

            // null-safe check (hash_join.rs:1004-1016)
            bool never_match = false;
            Row key(upd_side.key_idx.size());
            for (size_t i = 0; i < upd_side.key_idx.size(); i++) {
                key[i] = row[upd_side.key_idx[i]];
                if (key[i].null && !null_safe[i]) never_match = true;
            }
            if (never_match) {
                if (is_anti(T) && forward_exactly_once(T, S)) b.append_row_update(op, row);
                else if (is_outer_side(T, S)) b.append_row_update(op, row);
                continue; // no state write (hash_join.rs:1143-1152)
            }

            uint64_t degree = 0;
            bool have_ao_match = false;
            Row ao_match_pk;
            auto mit = match_side.table.find(key);
            if (mit != match_side.table.end()) {
                for (auto& [mpk, entry] : mit->second) {
                    if (cond_ok(row, entry.row, S)) {
                        degree += 1;
                        if (is_insert) {
                            // emit BEFORE degree update (hash_join.rs:1311-1317)
                            if (!forward_exactly_once(T, S)) with_match(b, op, row, entry);
                            if (match_side.need_deg) entry.degree += 1;
                        } else {
                            // degree update BEFORE emit (hash_join.rs:1334-1342)
                            if (match_side.need_deg) entry.degree -= 1;
                            if (!forward_exactly_once(T, S)) with_match(b, op, row, entry);
                        }
                        if (match_side.need_deg)
                            deg_touched[1 - S][enc_key(1 - S, key, mpk)] = {key,
                                                                            mpk};
                    }
                    if (append_only) {
                        // hash_join.rs:1359-1364: jk ⊇ pk ⇒ at most one match
                        have_ao_match = true;
                        ao_match_pk = mpk;
                    }
                }
            }

            if (degree == 0) {
                // forward_if_not_matched (join/builder.rs:303-312)
                if ((is_anti(T) && forward_exactly_once(T, S)) || is_outer_side(T, S))
                    b.append_row_update(op, row);
            } else if (is_semi(T) && forward_exactly_once(T, S)) {
                // forward_exactly_once_if_matched (join/builder.rs:287-300)
                b.append_row_update(op, row);
            }

            if (append_only && have_ao_match && is_insert) {
                // delete matched row, skip own insert (hash_join.rs:1241-1245)
                mit->second.erase(ao_match_pk);
                delta_delete(1 - S, key, ao_match_pk);
                if (mit->second.empty()) match_side.table.erase(mit);
                continue;
            }

            // own-state update (join/hash_join.rs:578-681)
            Row pk(upd_side.pk_idx.size());
            for (size_t i = 0; i < upd_side.pk_idx.size(); i++) pk[i] = row[upd_side.pk_idx[i]];
            if (is_insert) {
                auto& m = side_entry(upd_side, key);
                m[pk] = JoinEntry{row, degree};
                delta_insert(S, key, pk, row);
            } else {
                auto it = upd_side.table.find(key);
                if (it != upd_side.table.end()) {
                    if (it->second.erase(pk)) delta_delete(S, key, pk);
                    if (it->second.empty()) upd_side.table.erase(it);
                }
            }
        }
        b.take();
        return RW_OK;
    }

    std::map<Row, JoinEntry, RowOrderLess>& side_entry(JoinSideState& s, const Row& key) {
        auto it = s.table.find(key);
        if (it == s.table.end())
            it = s.table.emplace(key, std::map<Row, JoinEntry, RowOrderLess>(s.pk_less)).first;
        return it->second;
    }

    // with_match_on_insert / with_match_on_delete (join/builder.rs:173-284).
    // `op` is RW_OP_INSERT or RW_OP_DELETE; `entry.degree` is pre-increment
    // for insert, post-decrement for delete, so is_zero_degree == (degree==0).
    void with_match(JoinOutBuilder& b, uint8_t op, const Row& row, const JoinEntry& entry) {
        bool zero = entry.degree == 0;
        if (op == RW_OP_INSERT) {
            if (is_anti(T)) {
                if (zero && only_forward_matched_side(T, cur_side))
                    b.append_row_matched(RW_OP_DELETE, entry.row);
            } else if (is_semi(T)) {
                if (zero && only_forward_matched_side(T, cur_side))
                    b.append_row_matched(RW_OP_INSERT, entry.row);
            } else if (zero && outer_side_null(T, cur_side)) {
                b.append_row_matched(RW_OP_DELETE, entry.row);
                b.append_row(RW_OP_INSERT, row, entry.row);
            } else {
                b.append_row(RW_OP_INSERT, row, entry.row);
            }
        } else {
            if (is_anti(T)) {
                if (zero && only_forward_matched_side(T, cur_side))
                    b.append_row_matched(RW_OP_INSERT, entry.row);
            } else if (is_semi(T)) {
                if (zero && only_forward_matched_side(T, cur_side))
                    b.append_row_matched(RW_OP_DELETE, entry.row);
            } else if (zero && outer_side_null(T, cur_side)) {
                b.append_row(RW_OP_DELETE, row, entry.row);
                b.append_row_matched(RW_OP_INSERT, entry.row);
            } else {
                b.append_row(RW_OP_DELETE, row, entry.row);
            }
        }
    }

    // §8f-2 checkpoint spill deltas per side: key = memcmp(jk ∥ pk),
    // tri-state net per key (DEL / fresh PUT / PUT over a pre-epoch row) —
    // mirrors the GPU's kill-list netting. Degree tables spill via
    // rw_join_degree_drain (computed in checkpoint_drain below).
    struct DeltaEnt {
        int st; // 0 = DEL, 1 = PUT (fresh), 2 = PUT (over pre-epoch row)
        std::vector<uint8_t> v;
        Row jk, pk; // for the degree-table drain's current-degree lookup
    };
    std::map<std::string, DeltaEnt> delta[2];
    // §8f-2 degree-table deltas: pre-epoch rows whose degree changed this
    // epoch via probes (key → (jk, pk) for lookup at drain)
    std::map<std::string, std::pair<Row, Row>> deg_touched[2];
    std::vector<uint8_t> deg_spill[2];

    std::string enc_key(int s, const Row& key, const Row& pk) {
        std::vector<uint8_t> kb;
        for (size_t i = 0; i < key.size(); i++) {
            uint8_t t = side[s].types[side[s].key_idx[i]];
            rwcodec::DatumC d{key[i].null, key[i].i, key[i].d};
            rwcodec::memcmp_encode_datum(kb, t, d, {});
        }
        for (size_t i = 0; i < pk.size(); i++) {
            uint8_t t = side[s].types[side[s].pk_idx[i]];
            rwcodec::DatumC d{pk[i].null, pk[i].i, pk[i].d};
            rwcodec::memcmp_encode_datum(kb, t, d, {});
        }
        return std::string((const char*)kb.data(), kb.size());
    }

    void delta_insert(int s, const Row& key, const Row& pk, const Row& row) {
        std::vector<uint8_t> v;
        for (size_t c = 0; c < row.size(); c++) {
            rwcodec::DatumC d{row[c].null, row[c].i, row[c].d};
            rwcodec::value_encode_datum(v, side[s].types[c], d);
        }
        std::string k = enc_key(s, key, pk);
        auto it = delta[s].find(k);
        if (it != delta[s].end() && it->second.st == 0)
            it->second = {2, std::move(v), key, pk}; // PUT over a pre-epoch DEL
        else
            delta[s][k] = {1, std::move(v), key, pk};
    }

    void delta_delete(int s, const Row& key, const Row& pk) {
        std::string k = enc_key(s, key, pk);
        auto it = delta[s].find(k);
        if (it == delta[s].end()) {
            delta[s][k] = {0, {}}; // pre-epoch row deleted
        } else if (it->second.st == 1) {
            delta[s].erase(it); // fresh insert netted away
        } else {
            it->second = {0, {}};
        }
    }

    void checkpoint_drain(int s, std::vector<uint8_t>& sp) {
        auto put32 = [&](uint32_t x) {
            for (int b = 0; b < 4; b++) sp.push_back((uint8_t)(x >> (8 * b)));
        };
        for (auto& [k, e] : delta[s]) {
            sp.push_back(e.st ? 1 : 0);
            put32((uint32_t)k.size());
            sp.insert(sp.end(), k.begin(), k.end());
            put32(e.st ? (uint32_t)e.v.size() : 0);
            if (e.st) sp.insert(sp.end(), e.v.begin(), e.v.end());
        }
        // §8f-2 degree-table deltas (build_degree_row, join/row.rs:99-113):
        // keys mirror the main delta (a degree row exists iff its state row
        // does) plus pre-epoch rows whose degree changed during probes;
        // PUT value = order key (jk ∥ pk) ++ current degree i64.
        if (side[s].need_deg) {
            auto deg_val = [&](const Row& jk, const Row& pk, uint64_t deg,
                               std::vector<uint8_t>& v) {
                for (size_t i = 0; i < jk.size(); i++) {
                    uint8_t t = side[s].types[side[s].key_idx[i]];
                    rwcodec::DatumC d{jk[i].null, jk[i].i, jk[i].d};
                    rwcodec::value_encode_datum(v, t, d);
                }
                for (size_t i = 0; i < pk.size(); i++) {
                    uint8_t t = side[s].types[side[s].pk_idx[i]];
                    rwcodec::DatumC d{pk[i].null, pk[i].i, pk[i].d};
                    rwcodec::value_encode_datum(v, t, d);
                }
                rwcodec::value_encode_datum(v, RW_T_I64,
                                            {false, (long long)deg, 0});
            };
            auto lookup_deg = [&](const Row& jk, const Row& pk,
                                  uint64_t& deg) {
                auto ti = side[s].table.find(jk);
                if (ti == side[s].table.end()) return false;
                auto ei = ti->second.find(pk);
                if (ei == ti->second.end()) return false;
                deg = ei->second.degree;
                return true;
            };
            std::map<std::string, std::optional<std::vector<uint8_t>>> dd;
            for (auto& [k, e] : delta[s]) {
                if (e.st == 0) {
                    dd[k] = std::nullopt;
                } else {
                    uint64_t deg = 0;
                    lookup_deg(e.jk, e.pk, deg); // st>0 ⇒ live
                    std::vector<uint8_t> v;
                    deg_val(e.jk, e.pk, deg, v);
                    dd[k] = std::move(v);
                }
            }
            for (auto& [k, jp] : deg_touched[s]) {
                if (dd.count(k)) continue; // covered by the main delta
                uint64_t deg = 0;
                if (!lookup_deg(jp.first, jp.second, deg)) continue;
                std::vector<uint8_t> v;
                deg_val(jp.first, jp.second, deg, v);
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
        deg_touched[s].clear();
        delta[s].clear();
    }

    int cur_side = 0;
    int push(int S, const RwChunk* chunk) {
        cur_side = S;
        return push_chunk(S, chunk);
    }

    RwChunk* poll() {
        if (outputs.empty()) return nullptr;
        auto c = std::move(outputs.front());
        outputs.erase(outputs.begin());
        return chunk_to_c(*c);
    }

    // state cleaning below the selected watermark (update_watermark →
    // state_table watermark cleaning, join/hash_join.rs:521-527): drop every
    // key whose watermarked key column sorts below the value (NULLs are
    // largest, never cleaned)
    void clean_below(size_t jk_idx, int64_t sel) {
        for (int s = 0; s < 2; s++) {
            auto& tab = side[s].table;
            for (auto it = tab.begin(); it != tab.end();) {
                const Datum& kd = it->first[jk_idx];
                if (!kd.null && kd.i < sel) {
                    // state-cleaning spill deltas: the reference's commit
                    // applies the watermark as a range delete on the
                    // store (state_table.rs:1707), so the spill stream
                    // must carry DELETEs for the cleaned rows — a netted
                    // restore replay would otherwise resurrect them
                    for (auto& [pk, entry] : it->second)
                        delta_delete(s, it->first, pk);
                    it = tab.erase(it);
                } else {
                    ++it;
                }
            }
        }
    }

    // rescale re-scope (update_vnode_bitmap): dist key = join key; both
    // sides drop unowned keys, no retractions
    int update_vnode_bitmap(const uint8_t* bm, uint32_t vnode_count) {
        for (int s = 0; s < 2; s++) {
            std::vector<uint8_t> kt;
            for (auto k : side[s].key_idx) kt.push_back(side[s].types[k]);
            auto& tab = side[s].table;
            for (auto it = tab.begin(); it != tab.end();) {
                uint32_t vn = vnode_of_key_row(it->first, kt, vnode_count);
                if ((bm[vn >> 3] >> (vn & 7)) & 1) ++it;
                else it = tab.erase(it);
            }
        }
        return RW_OK;
    }

    // inequality-pair state cleaning: retire ROWS of side `s2` whose
    // value column `col` sorts below the selected watermark (NULLs
    // largest, kept); per-row spill DELETEs as clean_below
    void clean_rows_below(int s2, uint32_t col, int64_t sel) {
        auto& tab = side[s2].table;
        for (auto it = tab.begin(); it != tab.end();) {
            for (auto pit = it->second.begin(); pit != it->second.end();) {
                const Datum& d = pit->second.row[col];
                if (!d.null && d.i < sel) {
                    delta_delete(s2, it->first, pit->first);
                    pit = it->second.erase(pit);
                } else {
                    ++pit;
                }
            }
            if (it->second.empty()) it = tab.erase(it);
            else ++it;
        }
    }

    int watermark(int s, uint32_t col_idx, int64_t val, uint32_t* out_cols,
                  int64_t* out_vals, int max_out) {
        int n_out = 0;
        for (size_t idx = 0; idx < side[s].key_idx.size(); idx++) {
            if (side[s].key_idx[idx] != col_idx) continue;
            wm_side[s][idx] = {true, val};
            if (!wm_side[0][idx].has || !wm_side[1][idx].has) continue;
            int64_t sel = std::min(wm_side[0][idx].val, wm_side[1][idx].val);
            if (wm_out[idx].has && sel <= wm_out[idx].val) continue;
            wm_out[idx] = {true, sel};
            for (size_t w = 0; w < wm_pos.size(); w++)
                if (wm_pos[w] == idx && wm_clean[w]) clean_below(idx, sel);
            // emit for the update side's mapped output columns, then the
            // match side's (hash_join.rs:852-866)
            for (int s2 : {s, 1 - s}) {
                uint32_t src = side[s2].key_idx[idx] +
                               (s2 == RW_SIDE_RIGHT ? (uint32_t)types_l.size() : 0);
                for (uint32_t oi = 0; oi < output_indices.size(); oi++) {
                    if (output_indices[oi] == src && n_out < max_out) {
                        out_cols[n_out] = oi;
                        out_vals[n_out] = sel;
                        n_out++;
                    }
                }
            }
        }
        // inequality-pair watermarks (hash_join.rs:869-914): min across
        // sides; emit for the LARGER side's output columns; clean that
        // side's rows below the selection when flagged
        for (size_t p = 0; p < ineq_col[0].size(); p++) {
            if (ineq_col[s][p] != col_idx) continue;
            ineq_wm[s][p] = {true, val};
            if (!ineq_wm[0][p].has || !ineq_wm[1][p].has) continue;
            int64_t sel = std::min(ineq_wm[0][p].val, ineq_wm[1][p].val);
            if (ineq_out[p].has && sel <= ineq_out[p].val) continue;
            ineq_out[p] = {true, sel};
            int larger = ineq_larger[p] ? 0 : 1;
            if (ineq_do_clean[p])
                clean_rows_below(larger, ineq_col[larger][p], sel);
            uint32_t src = ineq_col[larger][p] +
                           (larger == 1 ? (uint32_t)types_l.size() : 0);
            for (uint32_t oi = 0; oi < output_indices.size(); oi++) {
                if (output_indices[oi] == src && n_out < max_out) {
                    out_cols[n_out] = oi;
                    out_vals[n_out] = sel;
                    n_out++;
                }
            }
        }
        return n_out;
    }
};

} // namespace orc

using namespace orc;

extern "C" {

int rw_hash_join_restore(void* h, int side, const uint8_t* buf, uint64_t len,
                         const uint8_t* deg_buf, uint64_t deg_len) {
    if (side != 0 && side != 1) return RW_E_INVAL;
    return ((HashJoinOracle*)h)->restore(side, buf, len, deg_buf, deg_len);
}

void* rw_hash_join_create(const RwHashJoinDesc* d) { return new HashJoinOracle(d); }
int rw_hash_join_push_chunk(void* h, int side, const RwChunk* c) {
    return ((HashJoinOracle*)h)->push(side, c);
}
int rw_join_checkpoint_drain(void* h, int side, uint8_t** buf,
                             uint64_t* len) {
    if (side != 0 && side != 1) return RW_E_INVAL;
    std::vector<uint8_t> sp;
    ((HashJoinOracle*)h)->checkpoint_drain(side, sp);
    *len = sp.size();
    *buf = (uint8_t*)malloc(sp.size() ? sp.size() : 1);
    memcpy(*buf, sp.data(), sp.size());
    return RW_OK;
}

int rw_join_degree_drain(void* h, int side, uint8_t** buf, uint64_t* len) {
    if (side != 0 && side != 1) return RW_E_INVAL;
    auto& sp = ((HashJoinOracle*)h)->deg_spill[side];
    *len = sp.size();
    *buf = (uint8_t*)malloc(sp.size() ? sp.size() : 1);
    memcpy(*buf, sp.data(), sp.size());
    sp.clear();
    return RW_OK;
}

int rw_hash_join_update_vnode_bitmap(void* h, const uint8_t* bitmap,
                                     uint32_t vnode_count) {
    if (!vnode_count || vnode_count % 8) return RW_E_INVAL;
    return ((HashJoinOracle*)h)->update_vnode_bitmap(bitmap, vnode_count);
}

int rw_join_compact(void* h, int side, uint64_t* reclaimed) {
    // the oracle's maps hold no dead rows — compaction is a no-op with
    // identical observable state (the GPU build reclaims retired records)
    if (!h || (side != 0 && side != 1)) return RW_E_INVAL;
    if (reclaimed) *reclaimed = 0;
    return RW_OK;
}

int rw_hash_join_watermark(void* h, int side, uint32_t col_idx, int64_t val,
                           uint32_t* out_cols, int64_t* out_vals, int max_out) {
    return ((HashJoinOracle*)h)->watermark(side, col_idx, val, out_cols,
                                           out_vals, max_out);
}

int rw_hash_join_flush(void* h, uint64_t) {
    (void)h; // state commit is a no-op for the in-memory oracle
    return RW_OK;
}
RwChunk* rw_hash_join_poll(void* h) { return ((HashJoinOracle*)h)->poll(); }
void rw_hash_join_destroy(void* h) { delete (HashJoinOracle*)h; }

} // extern "C"
