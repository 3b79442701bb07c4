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
    int64_t i2 = 0; // decimal datums: high half of the 16-byte image
};

// Build a DatumC from a RAW 8-byte device word: float types reinterpret
// the bits into .d (a plain {null, word, 0} initializer leaves .d = 0.0
// and silently encodes every float as zero).
inline DatumC datum_of_word(uint8_t type, bool null, int64_t w,
                            int64_t w2 = 0) {
    DatumC d{null, w, 0, w2};
    if (!null && (type == RW_T_F64 || type == RW_T_F32))
        std::memcpy(&d.d, &w, 8);
    return d;
}

// Inverse: the raw 8-byte device word for a decoded datum (float types
// take .d's bits; .i is 0 for them after value_decode_datum).
inline int64_t word_of_datum(uint8_t type, const DatumC& d) {
    if (type == RW_T_F64 || type == RW_T_F32) {
        int64_t w;
        std::memcpy(&w, &d.d, 8);
        return w;
    }
    return d.i;
}

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
                // ordered-float canonicalization before the sign-flip
                // transform (memcmp_encoding.rs:543-590 via OrderedFloat):
                // -0.0 encodes as +0.0 and every NaN (incl. -NaN) as one
                // canonical largest pattern — they compare equal, so their
                // storage keys must byte-compare equal too
                double dv = dat.d;
                if (dv == 0.0) dv = 0.0; // -0.0 -> +0.0
                uint64_t bits;
                if (dv != dv)
                    bits = 0x7ff8000000000000ULL; // canonical quiet NaN
                else
                    std::memcpy(&bits, &dv, 8);
                bits = (bits & 0x8000000000000000ULL)
                           ? ~bits
                           : bits | 0x8000000000000000ULL;
                put_be(buf, bits, 8);
                break;
            }
            case RW_T_F32: {
                float f = (float)dat.d;
                if (f == 0.0f) f = 0.0f; // -0.0 -> +0.0 (see F64)
                uint32_t bits;
                if (f != f)
                    bits = 0x7fc00000u; // canonical quiet NaN
                else
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
        case RW_T_DECIMAL: {
            // serialize_decimal (value_encoding/mod.rs:344-350): the raw
            // 16-byte unordered_serialize image
            uint64_t a = (uint64_t)dat.i, b = (uint64_t)dat.i2;
            for (int k = 0; k < 8; k++) buf.push_back((uint8_t)(a >> (8 * k)));
            for (int k = 0; k < 8; k++) buf.push_back((uint8_t)(b >> (8 * k)));
            break;
        }
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

// inverse of value_encode_datum: reads one datum, returns bytes consumed
// (0 on underflow/unknown type)
inline size_t value_decode_datum(const uint8_t* p, size_t avail, uint8_t type,
                                 DatumC* out) {
    if (avail < 1) return 0;
    if (p[0] == 0) {
        *out = {true, 0, 0};
        return 1;
    }
    auto le = [&](int n) {
        uint64_t v = 0;
        for (int k = 0; k < n; k++) v |= (uint64_t)p[1 + k] << (8 * k);
        return v;
    };
    switch (type) {
        case RW_T_DECIMAL: {
            if (avail < 17) return 0;
            uint64_t a = 0, b = 0;
            for (int k = 0; k < 8; k++) a |= (uint64_t)p[1 + k] << (8 * k);
            for (int k = 0; k < 8; k++) b |= (uint64_t)p[9 + k] << (8 * k);
            *out = {false, (int64_t)a, 0, (int64_t)b};
            return 17;
        }
        case RW_T_I64:
        case RW_T_TS:
            if (avail < 9) return 0;
            *out = {false, (int64_t)le(8), 0};
            return 9;
        case RW_T_I32:
            if (avail < 5) return 0;
            *out = {false, (int32_t)(uint32_t)le(4), 0};
            return 5;
        case RW_T_BOOL:
            if (avail < 2) return 0;
            *out = {false, (int64_t)p[1], 0};
            return 2;
        case RW_T_F64: {
            if (avail < 9) return 0;
            uint64_t bits = le(8);
            double d;
            std::memcpy(&d, &bits, 8);
            *out = {false, 0, d};
            return 9;
        }
        case RW_T_F32: {
            if (avail < 5) return 0;
            uint32_t bits = (uint32_t)le(4);
            float f;
            std::memcpy(&f, &bits, 4);
            *out = {false, 0, (double)f};
            return 5;
        }
    }
    return 0;
}

// iterate spill frames ([put u8][klen u32 LE][key][vlen u32 LE][value]);
// calls fn(put, key_ptr, klen, val_ptr, vlen); returns false on malformed
template <typename F>
inline bool for_each_frame(const uint8_t* buf, uint64_t len, F&& fn) {
    uint64_t off = 0;
    auto rd32 = [&](uint64_t o) {
        return (uint32_t)buf[o] | ((uint32_t)buf[o + 1] << 8) |
               ((uint32_t)buf[o + 2] << 16) | ((uint32_t)buf[o + 3] << 24);
    };
    while (off + 5 <= len) {
        uint8_t put = buf[off];
        uint32_t klen = rd32(off + 1);
        if (off + 5 + klen + 4 > len) return false;
        const uint8_t* k = buf + off + 5;
        uint32_t vlen = rd32(off + 5 + klen);
        if (off + 5 + klen + 4 + vlen > len) return false;
        const uint8_t* v = buf + off + 5 + klen + 4;
        fn(put, k, klen, v, vlen);
        off += 5 + (uint64_t)klen + 4 + vlen;
    }
    return off == len;
}

} // namespace rwcodec
