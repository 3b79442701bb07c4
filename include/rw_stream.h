/* rw_stream.h — C-ABI executor boundary for the RisingWave stream hot path.
 *
 * This is the drop-in boundary of SURVEY.md §8b: each entry point replaces
 * the construction/driving of a reference executor:
 *
 *   rw_hash_agg_create  ⇔ HashAggExecutorBuilder::new_boxed_executor
 *                         (src/stream/src/from_proto/hash_agg.rs:52-120;
 *                          node params proto/stream_plan.proto:548-560)
 *   rw_hash_join_create ⇔ HashJoinExecutorBuilder::new_boxed_executor
 *                         (src/stream/src/from_proto/hash_join.rs:37-229;
 *                          node params proto/stream_plan.proto:639-682)
 *   *_push_chunk        ⇔ Message::Chunk into Execute::execute's stream
 *                         (executor/mod.rs:244-291,1332-1349); for the join,
 *                         `side` selects the barrier-aligned input
 *                         (executor/barrier_align.rs:45-160)
 *   *_flush             ⇔ Message::Barrier(epoch): state commit + (agg)
 *                         change emission (hash_agg.rs:651-658,
 *                          hash_join.rs:737-748; StateTable::commit
 *                          state_table.rs:1718)
 *   *_poll              ⇔ the Message::Chunk outputs yielded since the last
 *                         poll, in yield order; NULL when drained
 *
 * Errors: negative return = error code; rw_last_error() gives a message
 * (⇔ StreamExecutorResult). Ownership: chunks passed in are COPIED by the
 * callee during the call; chunks returned by *_poll are callee-allocated and
 * must be freed with rw_chunk_free.
 */
#ifndef RW_STREAM_H
#define RW_STREAM_H

#include "rw_chunk.h"

#ifdef __cplusplus
extern "C" {
#endif

/* ----- errors ----- */
#define RW_OK 0
#define RW_E_INVAL (-1)
#define RW_E_OVERFLOW (-2)       /* checked_add/checked_sub failed (general.rs:28-41) */
#define RW_E_INCONSISTENT (-3)   /* e.g. duplicate pk insert (join/hash_join.rs:760-799) */
#define RW_E_NOGPU (-4)          /* product library without a visible GPU */
#define RW_E_INTERNAL (-5)

const char* rw_last_error(void);
void rw_chunk_free(RwChunk* c);

/* ----- HashAgg ----- */

/* Agg kinds on the path (expr/impl/src/aggregate/general.rs):
 * count(*) :154-157 (counts rows, no arg) · count(col) (skips NULL arg)
 * sum      :18-41 (checked, NULL-skipping; NULL result when no input)
 * sum0     :27 (init 0, used for 2-phase count)
 * min/max  :90-125 (value state when append-only, else materialized input —
 *           chosen as in frontend generic/agg.rs:394-461) */
enum RwAggKind {
    RW_AGG_COUNT_STAR = 0,
    RW_AGG_COUNT = 1,
    RW_AGG_SUM = 2,
    RW_AGG_SUM0 = 3,
    RW_AGG_MIN = 4,
    RW_AGG_MAX = 5,
};

typedef struct RwAggCall {
    uint8_t kind;     /* RwAggKind */
    int32_t arg;      /* input column index, -1 for count(*) */
    uint8_t ret_type; /* RwTypeId of the output */
    uint8_t distinct; /* DISTINCT dedup on `arg` (aggregate/distinct.rs:67-
                         198): per (group, datum) counts gate the call's row
                         visibility — an insert is visible iff the count
                         rises 0→1, a delete iff it falls 1→0 */
} RwAggCall;

typedef struct RwHashAggDesc {
    uint32_t n_input_cols;
    const uint8_t* input_types; /* RwTypeId per input column */
    uint32_t n_group_key;
    const uint32_t* group_key_indices;
    uint32_t n_calls;
    const RwAggCall* calls;
    uint32_t row_count_index; /* index of the count(*) call (hash_agg.rs:97-98) */
    uint32_t n_stream_key;    /* input stream key, orders materialized input */
    const uint32_t* stream_key;
    uint32_t chunk_size;      /* output chunk rows (config/mod.rs:222-224) */
    uint8_t append_only;      /* value-state min/max allowed */
    uint8_t emit_on_window_close; /* EOWC (hash_agg.rs:421-474): barriers
        emit NOTHING until a watermark on group-key position 0 closes
        windows; then each group with window key < watermark emits its
        FINAL row once (Insert, rows sorted by group key; groups with
        row_count 0 emit nothing) and is removed. rw_hash_agg_watermark
        buffers the watermark instead of cleaning. Output watermark
        forwarding is the caller's job (the value passes through). */
    uint64_t state_capacity_hint; /* expected group count (0 = default);
                                     GPU sizes the HBM table from this */
} RwHashAggDesc;

void* rw_hash_agg_create(const RwHashAggDesc* desc);
int rw_hash_agg_push_chunk(void* h, const RwChunk* chunk);
int rw_hash_agg_flush(void* h, uint64_t epoch); /* barrier */
/* Epoch-batched ingestion: with mode 1, push_chunk STAGES chunks into a
 * device-resident epoch buffer (no apply launch) and the barrier flush
 * applies the whole epoch as ONE kernel launch — the engine's
 * one-launch-per-epoch execution shape, reachable from
 * the reference-side binding. Legal for order-free value
 * states only; rejects materialized-input / DISTINCT aggregates and
 * chunks with visibility bitmaps. */
int rw_hash_agg_ingest_mode(void* h, int epoch_batched);
RwChunk* rw_hash_agg_poll(void* h);
void rw_hash_agg_destroy(void* h);

/* ----- HashJoin ----- */

/* Join type constants, same values as executor/join/mod.rs:42-52. */
enum RwJoinType {
    RW_JOIN_INNER = 0,
    RW_JOIN_LEFT_OUTER = 1,
    RW_JOIN_RIGHT_OUTER = 2,
    RW_JOIN_FULL_OUTER = 3,
    RW_JOIN_LEFT_SEMI = 4,
    RW_JOIN_LEFT_ANTI = 5,
    RW_JOIN_RIGHT_SEMI = 6,
    RW_JOIN_RIGHT_ANTI = 7,
};

/* Non-equi condition restricted to the comparison shapes the reference tests
 * and the q7 plan use (a NonStrictExpression comparing two columns of the
 * concatenated row, hash_join.rs:1381-1403). */
enum RwCmpOp {
    RW_CMP_LT = 0,
    RW_CMP_LE = 1,
    RW_CMP_GT = 2,
    RW_CMP_GE = 3,
};

typedef struct RwHashJoinDesc {
    uint8_t join_type;   /* RwJoinType */
    uint8_t append_only; /* append_only_optimize (hash_join.rs:186-187) */
    uint32_t n_key;
    const uint32_t* key_l; /* join key column indices per side */
    const uint32_t* key_r;
    const uint8_t* null_safe; /* n_key flags (stream_plan.proto:648) */
    uint32_t n_cols_l;
    const uint8_t* types_l;
    uint32_t n_cols_r;
    const uint8_t* types_r;
    uint32_t n_pk_l; /* deduped pk indices = JoinParams.deduped_pk_indices */
    const uint32_t* pk_l;
    uint32_t n_pk_r;
    const uint32_t* pk_r;
    uint32_t n_output; /* output_indices into [left cols ‖ right cols] */
    const uint32_t* output_indices;
    uint8_t has_cond; /* optional non-equi condition */
    uint8_t cond_op;  /* RwCmpOp */
    uint32_t cond_l;  /* column indices into the concatenated row */
    uint32_t cond_r;
    /* Constant added to the RIGHT operand, and an optional second
     * conjunct — enough for the reference q7 plan's fused
     * `date_time BETWEEN $expr1 - 10s AND $expr1` predicate
     * (nexmark.yaml q7: StreamFilter over StreamHashJoin; fusing a
     * post-join filter into inner-join emission is semantics-preserving
     * because inner joins carry no degrees). */
    int64_t cond_rconst;
    uint8_t has_cond2;
    uint8_t cond2_op;
    uint32_t cond2_l;
    uint32_t cond2_r;
    int64_t cond2_rconst;
    uint32_t chunk_size;
    uint64_t state_capacity_hint; /* expected distinct keys per side (0 = default) */
    uint64_t row_capacity_hint;   /* expected resident rows per side (0 = default) */
    /* join-key watermark positions (stream_plan.proto watermark handling;
     * hash_join.rs watermark_indices_in_jk): positions into the join key
     * with a per-position state-cleaning flag */
    uint32_t n_wm_jk;
    const uint32_t* wm_jk_pos;
    const uint8_t* wm_jk_clean;
    /* inequality pairs (hash_join.rs InequalityPairInfo / the non-equi
     * condition's watermark derivation): per pair the left/right input
     * column, whether the LEFT side is the larger one (left >= right),
     * and whether the larger side's state is cleaned below the selected
     * watermark. Watermarks on these columns buffer per side; when the
     * min across sides advances it is emitted for the LARGER side's
     * output columns and (if clean) sweeps that side's rows. */
    uint32_t n_ineq;
    const uint32_t* ineq_left_col;
    const uint32_t* ineq_right_col;
    const uint8_t* ineq_left_larger;
    const uint8_t* ineq_clean;
} RwHashJoinDesc;

enum RwJoinSide { RW_SIDE_LEFT = 0, RW_SIDE_RIGHT = 1 };

void* rw_hash_join_create(const RwHashJoinDesc* desc);
int rw_hash_join_push_chunk(void* h, int side, const RwChunk* chunk);
int rw_hash_join_flush(void* h, uint64_t epoch); /* aligned barrier */
/* Epoch-batched ingestion (the agg mode's analogue): consecutive
 * SAME-SIDE chunks merge into one staged batch applied as one launch at
 * the next side switch, watermark or barrier. Exactly order-equivalent:
 * probes read only the match side, which a same-side run never mutates;
 * own-side interleavings are the conflict-segment pre-pass's job. */
int rw_hash_join_ingest_mode(void* h, int epoch_batched);
RwChunk* rw_hash_join_poll(void* h);
void rw_hash_join_destroy(void* h);

/* Watermark on an input column (hash_join.rs:815-914 restricted to join-key
 * watermarks): buffers per side, emits min across sides when it advances
 * (BufferedWatermarks), cleans both sides' state below the selected value
 * when the position's clean flag is set (TTL, state_table.rs:1707).
 * Emitted output watermarks (update side's columns first, then the match
 * side's — hash_join.rs:852-866) are returned through out_cols/out_vals
 * (capacity max_out); returns the count or a negative error. */
int rw_hash_join_watermark(void* h, int side, uint32_t col_idx, int64_t val,
                           uint32_t* out_cols, int64_t* out_vals, int max_out);

/* Watermark on an agg group-key position: state cleaning only (groups with
 * key below the watermark are reset — hash_agg.rs:503-507 /
 * update_watermark); non-EOWC, so nothing is emitted. */
int rw_hash_agg_watermark(void* h, uint32_t group_key_pos, int64_t val);

/* Rescale re-scoping (ExecutorParams vnode bitmap updates on scale-in/out,
 * state_table.rs update_vnode_bitmap + executor cache eviction): state
 * whose distribution-key vnode (dist key = group key for agg, join key for
 * join) is no longer owned is dropped WITHOUT emitting retractions — the
 * new owner holds it after recovery. `bitmap` is vnode_count/8 bytes,
 * LSB-first per byte (common/src/bitmap.rs layout); vnode_count matches
 * VirtualNode::COUNT (256 default). */
int rw_hash_agg_update_vnode_bitmap(void* h, const uint8_t* bitmap,
                                    uint32_t vnode_count);
int rw_hash_join_update_vnode_bitmap(void* h, const uint8_t* bitmap,
                                     uint32_t vnode_count);

/* ----- GroupTopN (top_n/group_top_n.rs) -----
 *
 * Per group (group_by cols), maintain the full ordered row set; the visible
 * window is rows [offset, offset+limit) in cache-key order (cache key =
 * order_by cols then the remaining storage-key cols, each asc/desc with the
 * default NULLS treatment — sort_util.rs defaults: ASC NULLS LAST / DESC
 * NULLS FIRST). Each push emits the ChangeBuffer-compacted window delta for
 * the chunk (group_top_n.rs:168-239 + common/change_buffer.rs:76-187):
 * delete+insert of the same cache key with a different row becomes a U-/U+
 * pair; equal rows cancel. WITH_TIES is not implemented (the reference's
 * default instantiation is WITH_TIES = false). */
typedef struct RwGroupTopNDesc {
    uint32_t n_cols;
    const uint8_t* types; /* RwTypeId per input column */
    uint32_t n_group_by;
    const uint32_t* group_by;
    uint32_t n_order_by;
    const uint32_t* order_cols;
    const uint8_t* order_desc; /* per order col: 1 = descending */
    uint32_t n_rest; /* remaining storage-key cols (after group + order) */
    const uint32_t* rest_cols;
    const uint8_t* rest_desc;
    uint64_t offset;
    uint64_t limit; /* > 0 */
    uint8_t with_ties; /* TopNCache<true> (top_n_cache.rs:539-758): the
                          window extends past `limit` while rows TIE with
                          the limit-th row on the order-by prefix (offset
                          must be 0, as in the reference) */
    uint32_t chunk_size;
    uint64_t state_capacity_hint; /* expected group count */
    uint64_t row_capacity_hint;   /* expected resident rows */
} RwGroupTopNDesc;

void* rw_group_top_n_create(const RwGroupTopNDesc* desc);
int rw_group_top_n_push_chunk(void* h, const RwChunk* chunk);
int rw_group_top_n_flush(void* h, uint64_t epoch); /* state commit only */
RwChunk* rw_group_top_n_poll(void* h);
void rw_group_top_n_destroy(void* h);

/* ----- §8f-2 checkpoint spill (state-store boundary) -----
 *
 * Per-epoch KV deltas of the executor state tables in the reference's
 * encodings — key = memcomparable pk (OrderedRowSerde,
 * memcmp_encoding.rs:35-70), value = value-encoded row
 * (value_encoding/mod.rs:151-215) — so the drained bytes are what
 * StateTable::commit would hand the state store. Record framing:
 * [put u8][klen u32 LE][key][vlen u32 LE][value], emitted in memcmp key
 * order. Agg: intermediate-state table (group key → outputs). Join: per
 * side, pk = join key ∥ deduped input pk, value = full row (degree
 * tables via rw_join_degree_drain below). Caller frees with
 * rw_spill_free. */
int rw_agg_checkpoint_drain(void* h, uint8_t** buf, uint64_t* len);
/* ----- state restore (crash recovery / cache rehydration) -----
 *
 * Rebuild device state from spill records previously produced by the
 * drains (the reference restores executor state from its state tables on
 * recovery — src/meta/src/barrier/worker.rs:1074 — and on cache miss,
 * join/hash_join.rs:232-260). `buf` = concatenated drain outputs in epoch
 * order (PUTs and DELETEs; later frames win — the state store's merged
 * view). Must be called on a freshly created executor, before any input.
 * Restored rows are NOT re-emitted by the next drain (they predate the
 * epoch). Aggregates with materialized-input state (retractable min/max)
 * reject restore loudly: their row sets live in minput state tables that
 * are not yet spilled. Join: `deg_buf` = the degree-table
 * drain stream (same key encoding), required for join types that keep
 * degrees; pass NULL/0 otherwise. */
int rw_hash_agg_restore(void* h, const uint8_t* buf, uint64_t len);
/* Materialized-input (retractable min/max) state TABLES — one per minput
 * call in the reference (AggStateStorage::MaterializedInput,
 * test_utils/agg_executor.rs:63-121): pk = group key ∥ value (ASC min /
 * DESC max) ∥ stream key, value = the full row. Same record framing and
 * per-epoch delta semantics as the other drains. On recovery, restore
 * EVERY minput table BEFORE rw_hash_agg_restore (prev outputs are
 * recomputed from the hydrated chains, agg_group.rs:219-221). */
int rw_agg_n_minput_tables(void* h);
int rw_agg_minput_drain(void* h, int mi, uint8_t** buf, uint64_t* len);
int rw_agg_minput_restore(void* h, int mi, const uint8_t* buf, uint64_t len);
int rw_hash_join_restore(void* h, int side, const uint8_t* buf, uint64_t len,
                         const uint8_t* deg_buf, uint64_t deg_len);
int rw_join_checkpoint_drain(void* h, int side, uint8_t** buf, uint64_t* len);
/* §8f-4 memory reclamation: rebuild one side's record store and buckets
 * without dead records (retractions / watermark sweeps retire rows in
 * place; the reference reclaims them in Hummock compaction). Call
 * between epochs AFTER the side's checkpoint + degree drains (fails
 * loudly on pending deltas or undrained rows). Logical state, drains and
 * restore are unaffected; `reclaimed` (optional) reports freed bytes. */
int rw_join_compact(void* h, int side, uint64_t* reclaimed);
/* Same reclamation for the agg's materialized-input row store and the
 * GroupTopN record store (retractions retire rows in place). Same
 * contract: call after the drains; fails loudly on pending deltas. */
int rw_agg_minput_compact(void* h, uint64_t* reclaimed);
int rw_topn_compact(void* h, uint64_t* reclaimed);
int rw_topn_checkpoint_drain(void* h, uint8_t** buf, uint64_t* len);
/* Replay concatenated rw_topn_checkpoint_drain streams into a freshly
 * created GroupTopN executor (PUT/DELETE frames net host-side; the
 * surviving full rows rebuild the per-group state, marked persisted so the
 * next drain does not re-PUT them). Must run before any input. */
int rw_topn_restore(void* h, const uint8_t* buf, uint64_t len);
/* handle_watermark (group_top_n.rs:266-273): a watermark on input column
 * `col_idx` cleans state below `val` and is forwarded iff col_idx is the
 * FIRST group-by column. Returns 1 = forwarded (state cleaned; retired
 * rows net to DELETE in the next drain), 0 = absorbed, <0 = error. */
int rw_group_top_n_watermark(void* h, uint32_t col_idx, int64_t val);
/* DISTINCT dedup tables (one StateTable per distinct column in the
 * reference, from_proto/hash_agg.rs distinct_dedup_tables): pk = group key
 * ∥ datum, value = full row ++ one i64 count per call distincting on the
 * column (distinct.rs:89-93,158-185). Same record framing; one record per
 * (group, datum) touched since the last drain, sorted by pk; DELETE when
 * the count dropped to 0. `di` indexes tables in first-use order;
 * rw_agg_n_dedup_tables returns how many exist. */
int rw_agg_n_dedup_tables(void* h);
int rw_agg_dedup_drain(void* h, int di, uint8_t** buf, uint64_t* len);
/* Replay concatenated rw_agg_dedup_drain streams for table `di` into a
 * freshly created executor (before any input; independent of the
 * intermediate-table restore order). Restored counts are persisted: a
 * later drop to 0 drains as DELETE, exactly as uninterrupted. */
int rw_agg_dedup_restore(void* h, int di, const uint8_t* buf, uint64_t len);
/* Join degree tables (join/row.rs:99-113 build_degree_row): per side
 * needing degrees (join/mod.rs:153-165), pk = jk ∥ pk as the main table,
 * value = order key ++ degree i64. The deltas are computed by
 * rw_join_checkpoint_drain for that side (rows inserted/killed this epoch
 * plus pre-epoch rows whose degree changed during probes) — call it first,
 * then this to collect the degree records. */
int rw_join_degree_drain(void* h, int side, uint8_t** buf, uint64_t* len);
void rw_spill_free(uint8_t* buf);

#ifdef __cplusplus
}
#endif
#endif /* RW_STREAM_H */
