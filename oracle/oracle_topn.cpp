// oracle/oracle_topn.cpp — CPU restatement of the reference's GroupTopN
// executor, WITH_TIES = false:
//
//  - per-row application and window maintenance:
//    stream/src/executor/top_n/group_top_n.rs:168-239 (apply_chunk: per row,
//    group cache lookup → TopNCache insert/delete → TopNStaging)
//  - window semantics: top_n/top_n_cache.rs:293-520 (TopNCache<false>): the
//    low/middle/high caches are an LRU optimization over the state table;
//    the observable behavior is "the window is rows [offset, offset+limit)
//    of the group's rows in cache-key order", with deltas staged whenever
//    the middle (= window) content changes
//  - cache key order: order_by cols then the remaining storage-key cols
//    (create_cache_key_serde, top_n/utils.rs:130-150), each asc/desc with
//    sort_util.rs default NULLS (ASC last / DESC first — ordered_cmp)
//  - per-chunk compaction: TopNStaging over ChangeBuffer
//    (common/change_buffer.rs:76-187): per cache key, delete+insert with a
//    different row → Update (U-/U+ pair); insert+delete → dropped; no-op
//    updates filtered (into_records:182-190)
//  - output: one StreamChunkBuilder per push (group_top_n.rs:231-239),
//    chunked at chunk_size with U-pair adjacency
//
// This oracle keeps the full ordered set per group (std::map) and computes
// the window delta per push; identical to the incremental cache by
// construction (the reference's caches are exact once synced).
#include <cstring>
#include <map>
#include <optional>
#include <memory>
#include <string>
#include <vector>

#include "../include/rw_chunk.h"
#include "../include/rw_stream.h"
#include "common.hpp"
#include "../include/rw_codec.hpp"

namespace orc {

extern thread_local std::string g_err; // defined in oracle_agg.cpp

struct GroupTopNOracle {
    std::vector<uint8_t> types;
    std::vector<uint32_t> group_by;
    std::vector<uint32_t> ck_cols; // order_by ∥ rest storage-key cols
    RowOrderLess ck_less;          // over the ck_cols projection
    RowOrderLess group_less;
    uint64_t offset, limit;
    uint32_t chunk_size;

    using Group = std::map<Row, Row, RowOrderLess>; // cache key → full row
    std::map<Row, Group, RowOrderLess> groups;
    std::vector<std::unique_ptr<OwnedChunk>> outputs;

    GroupTopNOracle(const RwGroupTopNDesc* d) {
        types.assign(d->types, d->types + d->n_cols);
        group_by.assign(d->group_by, d->group_by + d->n_group_by);
        for (uint32_t i = 0; i < d->n_order_by; i++) {
            ck_cols.push_back(d->order_cols[i]);
            ck_less.order.push_back(
                {types[d->order_cols[i]], d->order_desc[i] != 0});
        }
        for (uint32_t i = 0; i < d->n_rest; i++) {
            ck_cols.push_back(d->rest_cols[i]);
            ck_less.order.push_back(
                {types[d->rest_cols[i]], d->rest_desc[i] != 0});
        }
        for (auto g : group_by) group_less.order.push_back({types[g], false});
        offset = d->offset;
        limit = d->limit;
        with_ties = d->with_ties != 0;
        n_order = d->n_order_by;
        chunk_size = d->chunk_size ? d->chunk_size : 1024;
        groups = decltype(groups)(group_less);
    }

    bool with_ties = false;
    uint32_t n_order = 0;

    // compare only the order-by prefix of two cache keys (sort key,
    // top_n_cache.rs:539-560: ties are on (CacheKey).0)
    int sort_key_cmp(const Row& a, const Row& b) const {
        for (uint32_t i = 0; i < n_order; i++) {
            int c = datum_cmp(a[i], b[i], ck_less.order[i].type);
            if (ck_less.order[i].desc) c = -c;
            if (c) return c;
        }
        return 0;
    }

    // §8f-2 checkpoint spill deltas (same tri-state netting and record
    // framing as the join oracle): key = memcmp storage key (group cols
    // ASC then cache-key cols with their orders), value = full row.
    struct DeltaEnt {
        int st; // 0 = DEL, 1 = PUT (fresh), 2 = PUT (over pre-epoch row)
        std::vector<uint8_t> v;
    };
    std::map<std::string, DeltaEnt> delta;

    std::string enc_key(const Row& gk, const Row& ck) {
        std::vector<uint8_t> kb;
        for (size_t i = 0; i < gk.size(); i++) {
            rwcodec::DatumC d{gk[i].null, gk[i].i, gk[i].d};
            rwcodec::memcmp_encode_datum(kb, types[group_by[i]], d, {});
        }
        for (size_t i = 0; i < ck.size(); i++) {
            rwcodec::DatumC d{ck[i].null, ck[i].i, ck[i].d};
            rwcodec::OrderType ot;
            ot.desc = ck_less.order[i].desc;
            rwcodec::memcmp_encode_datum(kb, types[ck_cols[i]], d, ot);
        }
        return std::string((const char*)kb.data(), kb.size());
    }

    void delta_insert(const Row& gk, const Row& ck, const Row& row) {
        std::vector<uint8_t> v;
        for (size_t c = 0; c < row.size(); c++) {
            rwcodec::DatumC d{row[c].null, row[c].i, row[c].d};
            rwcodec::value_encode_datum(v, types[c], d);
        }
        std::string k = enc_key(gk, ck);
        auto it = delta.find(k);
        if (it != delta.end() && it->second.st == 0)
            it->second = {2, std::move(v)};
        else
            delta[k] = {1, std::move(v)};
    }

    void delta_delete(const Row& gk, const Row& ck) {
        std::string k = enc_key(gk, ck);
        auto it = delta.find(k);
        if (it == delta.end()) delta[k] = {0, {}};
        else if (it->second.st == 1) delta.erase(it);
        else it->second = {0, {}};
    }

    void checkpoint_drain(std::vector<uint8_t>& sp) {
        auto put32 = [&](uint32_t x) {
            for (int b = 0; b < 4; b++) sp.push_back((uint8_t)(x >> (8 * b)));
        };
        for (auto& [k, e] : delta) {
            sp.push_back(e.st ? 1 : 0);
            put32((uint32_t)k.size());
            sp.insert(sp.end(), k.begin(), k.end());
            put32(e.st ? (uint32_t)e.v.size() : 0);
            if (e.st) sp.insert(sp.end(), e.v.begin(), e.v.end());
        }
        delta.clear();
    }

    Row project(const Row& row, const std::vector<uint32_t>& idx) const {
        Row out;
        out.reserve(idx.size());
        for (auto i : idx) out.push_back(row[i]);
        return out;
    }

    std::vector<Row> window_of(const Group& g) const {
        std::vector<Row> w; // cache keys only
        uint64_t i = 0;
        const Row* cut = nullptr; // limit-th row's cache key (ties boundary)
        for (auto& kv : g) {
            if (i >= offset + limit) {
                // WITH TIES: keep rows tying the limit-th row's sort key
                // (TopNCache<true>, top_n_cache.rs:539-640; offset == 0)
                if (!with_ties || !cut || sort_key_cmp(kv.first, *cut) != 0)
                    break;
                w.push_back(kv.first);
            } else if (i >= offset) {
                w.push_back(kv.first);
                if (i + 1 == offset + limit) cut = &kv.first;
            }
            i++;
        }
        return w;
    }

    int push_chunk(const RwChunk* c) {
        ChunkView cv{c};
        // group-key first-touch order; per group the pre-chunk window is
        // snapshotted as (cache key, full row) pairs before any mutation
        std::vector<Row> touched;
        std::map<Row, std::vector<std::pair<Row, Row>>, RowOrderLess> old_win(
            group_less);
        for (size_t r = 0; r < cv.n_rows(); r++) {
            if (!cv.visible(r)) continue;
            Row row = cv.row(r);
            Row gk = project(row, group_by);
            if (!old_win.count(gk)) {
                touched.push_back(gk);
                std::vector<std::pair<Row, Row>> w;
                auto it = groups.find(gk);
                if (it != groups.end()) {
                    uint64_t i = 0;
                    const Row* cut = nullptr;
                    for (auto& kv : it->second) {
                        if (i >= offset + limit) {
                            if (!with_ties || !cut ||
                                sort_key_cmp(kv.first, *cut) != 0)
                                break;
                            w.emplace_back(kv.first, kv.second);
                        } else if (i >= offset) {
                            w.emplace_back(kv.first, kv.second);
                            if (i + 1 == offset + limit) cut = &kv.first;
                        }
                        i++;
                    }
                }
                old_win.emplace(gk, std::move(w));
            }
            Row ck = project(row, ck_cols);
            uint8_t op = cv.op(r);
            auto& g = groups.try_emplace(gk, Group(ck_less)).first->second;
            if (op == RW_OP_INSERT || op == RW_OP_UPDATE_INSERT) {
                // upsert over an existing row = delete + insert for the
                // delta (the pre-epoch row leaves the store)
                if (g.count(ck)) delta_delete(gk, ck);
                g[ck] = row;
                delta_insert(gk, ck, row);
            } else {
                auto it = g.find(ck);
                if (it != g.end()) {
                    g.erase(it);
                    delta_delete(gk, ck);
                }
            }
        }
        // per touched group: merge-diff old vs new window with ChangeBuffer
        // merge rules (delete+insert same key, different row -> U-pair)
        ChunkBuilder cb(chunk_size, types);
        std::unique_ptr<OwnedChunk> done;
        auto emit = [&](uint8_t op, const Row& row) {
            if (cb.append_row(op, row, &done)) outputs.push_back(std::move(done));
        };
        for (auto& gk : touched) {
            auto git = groups.find(gk);
            std::vector<Row> neww = window_of(git->second);
            const auto& oldw = old_win.at(gk);
            size_t i = 0, j = 0;
            while (i < oldw.size() || j < neww.size()) {
                int c2;
                if (i >= oldw.size()) c2 = 1;
                else if (j >= neww.size()) c2 = -1;
                else c2 = ordered_cmp(oldw[i].first, neww[j], ck_less.order);
                if (c2 == 0) {
                    const Row& nr = git->second.at(neww[j]);
                    if (!row_eq(oldw[i].second, nr, types)) {
                        emit(RW_OP_UPDATE_DELETE, oldw[i].second);
                        emit(RW_OP_UPDATE_INSERT, nr);
                    }
                    i++;
                    j++;
                } else if (c2 < 0) {
                    emit(RW_OP_DELETE, oldw[i].second);
                    i++;
                } else {
                    emit(RW_OP_INSERT, git->second.at(neww[j]));
                    j++;
                }
            }
        }
        if (auto rest = cb.take()) outputs.push_back(std::move(rest));
        return RW_OK;
    }

    RwChunk* poll() {
        if (outputs.empty()) return nullptr;
        auto c = std::move(outputs.front());
        outputs.erase(outputs.begin());
        return chunk_to_c(*c);
    }

    // handle_watermark (group_top_n.rs:266-273): a watermark on
    // group_by[0] cleans the state table below it (range delete on the
    // store) and is forwarded; other columns' watermarks are absorbed.
    // Cleaned rows net to DELETE spill frames — a restore replay must
    // not resurrect them.
    int watermark(uint32_t col_idx, int64_t val) {
        if (group_by.empty() || group_by[0] != col_idx) return 0;
        for (auto it = groups.begin(); it != groups.end();) {
            const Datum& d = it->first[0];
            if (!d.null && d.i < val) {
                for (auto& [ck, row] : it->second) {
                    (void)row;
                    delta_delete(it->first, ck);
                }
                it = groups.erase(it);
            } else {
                ++it;
            }
        }
        return 1;
    }

    // §8f-5 recovery: net concatenated drain frames (PUT last-write-wins,
    // DELETE removes — KV compaction semantics) and rebuild `groups` from
    // the surviving value-encoded full rows. Restored rows predate the
    // epoch, so they leave no delta entries.
    int restore(const uint8_t* buf, uint64_t len) {
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
        for (auto& [kbytes, val] : merged) {
            (void)kbytes;
            Row row;
            size_t off = 0;
            for (size_t c = 0; c < types.size(); c++) {
                rwcodec::DatumC d;
                size_t got = rwcodec::value_decode_datum(
                    val.data() + off, val.size() - off, types[c], &d);
                if (!got) {
                    g_err = "restore: bad row datum";
                    return RW_E_INVAL;
                }
                off += got;
                Datum dm;
                dm.null = d.null;
                if (types[c] == RW_T_F64) dm.d = d.d;
                else dm.i = d.i;
                row.push_back(dm);
            }
            Row gk = project(row, group_by);
            Row ck = project(row, ck_cols);
            auto& g = groups.try_emplace(gk, Group(ck_less)).first->second;
            g[ck] = row;
        }
        return RW_OK;
    }
};

} // namespace orc

using namespace orc;

extern "C" {

void* rw_group_top_n_create(const RwGroupTopNDesc* d) {
    if (!d || !d->limit) return nullptr;
    if (d->with_ties && d->offset) return nullptr; // reference asserts too
    return new GroupTopNOracle(d);
}
int rw_group_top_n_push_chunk(void* h, const RwChunk* c) {
    return ((GroupTopNOracle*)h)->push_chunk(c);
}
int rw_group_top_n_flush(void* h, uint64_t epoch) {
    (void)h;
    (void)epoch;
    return RW_OK; // emission is per push; state commit is a no-op here
}
RwChunk* rw_group_top_n_poll(void* h) { return ((GroupTopNOracle*)h)->poll(); }
int rw_topn_restore(void* h, const uint8_t* buf, uint64_t len) {
    return ((GroupTopNOracle*)h)->restore(buf, len);
}
int rw_group_top_n_watermark(void* h, uint32_t col_idx, int64_t val) {
    return ((GroupTopNOracle*)h)->watermark(col_idx, val);
}
int rw_topn_compact(void* h, uint64_t* reclaimed) {
    // no-op: the oracle's maps hold no dead rows (see rw_join_compact)
    if (!h) return RW_E_INVAL;
    if (reclaimed) *reclaimed = 0;
    return RW_OK;
}
int rw_topn_checkpoint_drain(void* h, uint8_t** buf, uint64_t* len) {
    std::vector<uint8_t> sp;
    ((GroupTopNOracle*)h)->checkpoint_drain(sp);
    *len = sp.size();
    *buf = (uint8_t*)malloc(sp.size() ? sp.size() : 1);
    memcpy(*buf, sp.data(), sp.size());
    return RW_OK;
}
void rw_group_top_n_destroy(void* h) { delete (GroupTopNOracle*)h; }

} // extern "C"
