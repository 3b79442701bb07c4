/* rw_chunk.h — C-ABI mirror of RisingWave's StreamChunk.
 *
 * Reference layout being mirrored (semantics, not bytes):
 *   StreamChunk = Op[] + DataChunk          (common/src/array/stream_chunk.rs:104-108)
 *   DataChunk   = columns + visibility      (common/src/array/data_chunk.rs:65-68)
 *   PrimitiveArray = null bitmap + values   (common/src/array/primitive_array.rs:137-139)
 *
 * Deviation (documented in DESIGN.md §1): validity and visibility are one
 * byte per row (1 = valid/visible) instead of packed bitmaps — the GPU path
 * reads them coalesced; parity is on values, not encodings.
 */
#ifndef RW_CHUNK_H
#define RW_CHUNK_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* Op codes, same order as the Rust enum (stream_chunk.rs `enum Op`). */
enum RwOp {
    RW_OP_INSERT = 0,
    RW_OP_DELETE = 1,
    RW_OP_UPDATE_DELETE = 2,
    RW_OP_UPDATE_INSERT = 3,
};

/* Scalar type ids for the types on the hot path.
 * TS = timestamp as i64 microseconds (common/src/types: Timestamp micros). */
enum RwTypeId {
    RW_T_I64 = 0,
    RW_T_I32 = 1,
    RW_T_F64 = 2,
    RW_T_F32 = 3,
    RW_T_BOOL = 4,
    RW_T_TS = 5,
    /* 16-byte decimal in rust_decimal 1.40.0's serialize layout (the
     * reference's Decimal::unordered_serialize, types/decimal.rs:583-592):
     * u32 flags (scale in bits 16-23, sign in bit 31; byte 0 = 1/2/3 for
     * NaN/+Inf/-Inf specials) ++ u32 lo ++ u32 mid ++ u32 hi, all LE. */
    RW_T_DECIMAL = 6,
};

typedef struct RwColumn {
    uint8_t type;          /* RwTypeId */
    const uint8_t* valid;  /* n_rows bytes, 1 = non-NULL; never NULL ptr */
    const void* data;      /* n_rows values of native width */
} RwColumn;

typedef struct RwChunk {
    uint32_t n_rows; /* capacity, including invisible rows */
    uint32_t n_cols;
    const uint8_t* ops; /* n_rows RwOp codes */
    const uint8_t* vis; /* n_rows bytes, 1 = visible; NULL ⇒ all visible */
    const RwColumn* cols;
} RwChunk;

static inline uint32_t rw_type_size(uint8_t t) {
    switch (t) {
        case RW_T_I64: return 8;
        case RW_T_I32: return 4;
        case RW_T_F64: return 8;
        case RW_T_F32: return 4;
        case RW_T_BOOL: return 1;
        case RW_T_TS: return 8;
        case RW_T_DECIMAL: return 16;
        default: return 0;
    }
}

#ifdef __cplusplus
}
#endif
#endif /* RW_CHUNK_H */
