// rw_codec.hpp — header-only restatement of the reference's state-row
// encodings (the §8f-2 spill boundary):
//
//  - memcomparable key encoding (memcomparable 0.2.0 via OrderedRowSerde,
//    common/src/util/memcmp_encoding.rs:35-70): per datum a null tag byte
//    ((none,some) = (1,0) when NULLs are largest, else (0,1)) followed by the
//    order-preserving payload — integers as sign-flipped big-endian, floats
//    as total-ordered bits, bool as u8 — with EVERY byte complemented for
//    descending order (Deserializer::set_reverse). Pinned by the ordering
//    assertions of memcmp_encoding.rs:346-666 (transcribed in
//    tests/test_codec.py) — the crate itself is a Cargo.lock dependency not
//    vendored under /root/reference.
//  - value encoding (common/src/util/value_encoding/mod.rs:151-215): per
//    datum a presence byte (1/0) followed by little-endian bytes.
//
// Deviation: timestamps are encoded as their i64 microsecond value (this
// build's representation); the reference memcomparable-serializes chrono's
// (secs, nanos) — byte-compatible restore into real Hummock needs that form
// (a later-round item, noted in DESIGN.md).
#pragma once
#include <cstdint>
#include <cstring>
#include <vector>

#include "rw_chunk.h"

namespace rwcodec {

struct OrderType {
    bool desc = false;
    bool nulls_largest = true; // default ASC NULLS LAST / DESC NULLS FIRST
};

struct DatumC {
    bool null;
    int64_t i;
    double d;
};

inline void put_be(std::vector<uint8_t>& buf, uint64_t v, int n) {
    for (int k = n - 1; k >= 0; k--) buf.push_back((uint8_t)(v >> (8 * k)));
}

inline void memcmp_encode_datum(std::vector<uint8_t>& buf, uint8_t type,
                                const DatumC& dat, OrderType order) {
    size_t start = buf.size();
    uint8_t tag_none = order.nulls_largest ? 1 : 0;
    uint8_t tag_some = order.nulls_largest ? 0 : 1;
    if (dat.null) {
        buf.push_back(tag_none);
    } else {
        buf.push_back(tag_some);
        switch (type) {
            case RW_T_I64:
            case RW_T_TS:
                put_be(buf, (uint64_t)dat.i ^ 0x8000000000000000ULL, 8);
                break;
            case RW_T_I32:
                put_be(buf, ((uint32_t)(int32_t)dat.i) ^ 0x80000000u, 4);
                break;
            case RW_T_BOOL:
                buf.push_back((uint8_t)dat.i);
                break;
            case RW_T_F64: {
                uint64_t bits;
                std::memcpy(&bits, &dat.d, 8);
                bits = (bits & 0x8000000000000000ULL)
                           ? ~bits
                           : bits | 0x8000000000000000ULL;
                put_be(buf, bits, 8);
                break;
            }
            case RW_T_F32: {
                float f = (float)dat.d;
                uint32_t bits;
                std::memcpy(&bits, &f, 4);
                bits = (bits & 0x80000000u) ? ~bits : bits | 0x80000000u;
                put_be(buf, bits, 4);
                break;
            }
        }
    }
    if (order.desc)
        for (size_t k = start; k < buf.size(); k++) buf[k] = ~buf[k];
}

inline void value_encode_datum(std::vector<uint8_t>& buf, uint8_t type,
                               const DatumC& dat) {
    if (dat.null) {
        buf.push_back(0);
        return;
    }
    buf.push_back(1);
    switch (type) {
        case RW_T_I64:
        case RW_T_TS: {
            uint64_t v = (uint64_t)dat.i;
            for (int k = 0; k < 8; k++) buf.push_back((uint8_t)(v >> (8 * k)));
            break;
        }
        case RW_T_I32: {
            uint32_t v = (uint32_t)(int32_t)dat.i;
            for (int k = 0; k < 4; k++) buf.push_back((uint8_t)(v >> (8 * k)));
            break;
        }
        case RW_T_BOOL:
            buf.push_back((uint8_t)dat.i);
            break;
        case RW_T_F64: {
            uint64_t bits;
            std::memcpy(&bits, &dat.d, 8);
            for (int k = 0; k < 8; k++)
                buf.push_back((uint8_t)(bits >> (8 * k)));
            break;
        }
        case RW_T_F32: {
            float f = (float)dat.d;
            uint32_t bits;
            std::memcpy(&bits, &f, 4);
            for (int k = 0; k < 4; k++)
                buf.push_back((uint8_t)(bits >> (8 * k)));
            break;
        }
    }
}

} // namespace rwcodec
