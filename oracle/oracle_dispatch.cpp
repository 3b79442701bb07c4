// oracle/oracle_dispatch.cpp — CPU restatement of the vnode hash + hash
// dispatch step. ORACLE — TEST INFRASTRUCTURE ONLY (see common.hpp).
//
// Restates:
//  - VirtualNode::compute_chunk (common/src/hash/consistent_hash/
//    vnode.rs:146-181): vnode = Crc32(row-hash of dist keys) % vnode_count,
//    where the row hash feeds each datum via hash_scalar — native-endian
//    bytes of the primitive (types/scalar_impl.rs:47-55, std write_i64), and
//    NULL as the u32 sentinel 0xfffffff0 (array/mod.rs:99,288). crc32fast
//    1.5.0 is standard IEEE CRC-32 (poly 0xEDB88320 reflected); the
//    reference pins no hashed-key vnode fixtures, so ours are
//    cross-checked against an independent IEEE implementation in the tests.
//  - HashDataDispatcher::dispatch_data (stream/src/executor/
//    dispatch.rs:949-1050): per-output visibility = row visible && routed to
//    that output; U−/U+ pairs whose dist-key changed are downgraded to
//    Delete/Insert (:985-1010). (output_mapping here is identity — no
//    projection on this path's fragment edges in the bench plans.)
#include <cstring>
#include <vector>

#include "../include/rw_stream.h"
#include "common.hpp"

namespace orc {

static uint32_t crc_table[256];
static bool crc_init_done = false;
static void crc_init() {
    if (crc_init_done) return;
    for (uint32_t i = 0; i < 256; i++) {
        uint32_t c = i;
        for (int k = 0; k < 8; k++) c = (c & 1) ? 0xEDB88320u ^ (c >> 1) : c >> 1;
        crc_table[i] = c;
    }
    crc_init_done = true;
}

static uint32_t crc32_update(uint32_t crc, const uint8_t* p, size_t n) {
    for (size_t i = 0; i < n; i++) crc = crc_table[(crc ^ p[i]) & 0xFF] ^ (crc >> 8);
    return crc;
}

// feed one datum exactly as hash_datum does (types/mod.rs:1227-1233)
static uint32_t crc32_datum(uint32_t crc, const ChunkView& cv, size_t r, uint32_t col) {
    const RwColumn& c = cv.c->cols[col];
    if (!c.valid[r]) {
        uint32_t sentinel = 0xfffffff0u; // NULL_VAL_FOR_HASH (array/mod.rs:99)
        return crc32_update(crc, (const uint8_t*)&sentinel, 4);
    }
    switch (c.type) {
        case RW_T_I64:
        case RW_T_TS: { // TS here is i64 micros; see header note on chrono
            int64_t v = ((const int64_t*)c.data)[r];
            return crc32_update(crc, (const uint8_t*)&v, 8);
        }
        case RW_T_I32: {
            int32_t v = ((const int32_t*)c.data)[r];
            return crc32_update(crc, (const uint8_t*)&v, 4);
        }
        case RW_T_BOOL: {
            uint8_t v = ((const uint8_t*)c.data)[r];
            return crc32_update(crc, &v, 1);
        }
        case RW_T_F64: {
            double v = ((const double*)c.data)[r];
            return crc32_update(crc, (const uint8_t*)&v, 8);
        }
        case RW_T_F32: {
            float v = ((const float*)c.data)[r];
            return crc32_update(crc, (const uint8_t*)&v, 4);
        }
    }
    return crc;
}

} // namespace orc

using namespace orc;

extern "C" {

typedef struct {
    uint32_t n_keys;
    const uint32_t* key_indices;
    uint32_t vnode_count;
} RwVnodeDesc;

int rw_vnode_compute(const RwVnodeDesc* d, const RwChunk* chunk, uint16_t* out) {
    crc_init();
    ChunkView cv{chunk};
    for (size_t r = 0; r < cv.n_rows(); r++) {
        uint32_t crc = 0xFFFFFFFFu;
        for (uint32_t k = 0; k < d->n_keys; k++)
            crc = crc32_datum(crc, cv, r, d->key_indices[k]);
        uint32_t h = crc ^ 0xFFFFFFFFu; // crc32fast finish
        out[r] = (uint16_t)((uint64_t)h % d->vnode_count);
    }
    return RW_OK;
}

typedef struct {
    RwVnodeDesc v;
    uint32_t n_outputs;
    const uint32_t* vnode_to_output; // [vnode_count] -> output index
} RwDispatchDesc;

// Produces n_outputs chunks sharing rewritten ops/columns with per-output
// visibility (dispatch.rs:949-1050). Caller frees each with rw_chunk_free.
int rw_dispatch_compute(const RwDispatchDesc* d, const RwChunk* chunk,
                        RwChunk** outs) {
    ChunkView cv{chunk};
    size_t n = cv.n_rows();
    std::vector<uint16_t> vnodes(n);
    int rc = rw_vnode_compute(&d->v, chunk, vnodes.data());
    if (rc != RW_OK) return rc;

    // ops rewrite: U-pair downgrade on dist-key change (dispatch.rs:985-1010)
    std::vector<uint8_t> ops(chunk->ops, chunk->ops + n);
    long last_ud = -1;
    for (size_t r = 0; r < n; r++) {
        if (!cv.visible(r)) continue;
        if (ops[r] == RW_OP_UPDATE_DELETE) {
            last_ud = (long)r;
        } else if (ops[r] == RW_OP_UPDATE_INSERT && last_ud >= 0) {
            bool changed = false;
            for (uint32_t k = 0; k < d->v.n_keys && !changed; k++) {
                uint32_t col = d->v.key_indices[k];
                changed = !datum_eq(cv.at(last_ud, col), cv.at(r, col),
                                    cv.type(col));
            }
            if (changed) {
                ops[last_ud] = RW_OP_DELETE;
                ops[r] = RW_OP_INSERT;
            }
            last_ud = -1;
        }
    }

    for (uint32_t o = 0; o < d->n_outputs; o++) {
        OwnedChunk oc;
        oc.types.resize(cv.n_cols());
        for (size_t c = 0; c < cv.n_cols(); c++) oc.types[c] = cv.type(c);
        oc.cols.assign(cv.n_cols(), {});
        oc.ops.assign(ops.begin(), ops.end());
        oc.vis.resize(n);
        for (size_t r = 0; r < n; r++) {
            oc.vis[r] =
                cv.visible(r) && d->vnode_to_output[vnodes[r]] == o ? 1 : 0;
            for (size_t c = 0; c < cv.n_cols(); c++)
                oc.cols[c].push_back(cv.at(r, c));
        }
        outs[o] = chunk_to_c(oc);
    }
    return RW_OK;
}

} // extern "C"
