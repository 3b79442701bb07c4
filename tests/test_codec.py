"""Checkpoint spill encoding (§8f-2 foundation): memcomparable keys +
value-encoded rows, drained from the agg at flush.

Pinning: the memcomparable crate is a Cargo.lock dependency with no vendored
source; the reference's own test_memcomparable assertions
(util/memcmp_encoding.rs:346-430) are transcribed here as ordering
properties, plus explicit byte vectors derived from the published encoding
(sign-flipped big-endian + null tag; value encoding = presence byte + LE,
value_encoding/mod.rs:151-215).
"""
import ctypes
import struct

from rwtest import ffi
from rwtest.ffi import AGG_COUNT_STAR, AGG_SUM, T_I64, from_pretty, oracle


def drain(lib, agg):
    L = lib.lib
    L.rw_agg_checkpoint_drain.restype = ctypes.c_int
    L.rw_agg_checkpoint_drain.argtypes = [ctypes.c_void_p,
                                          ctypes.POINTER(ctypes.c_void_p),
                                          ctypes.POINTER(ctypes.c_uint64)]
    L.rw_spill_free.argtypes = [ctypes.c_void_p]
    buf = ctypes.c_void_p()
    ln = ctypes.c_uint64()
    assert L.rw_agg_checkpoint_drain(agg.h, ctypes.byref(buf), ctypes.byref(ln)) == 0
    data = ctypes.string_at(buf, ln.value)
    L.rw_spill_free(buf)
    return parse(data)


def dedup_drain(lib, agg, di):
    L = lib.lib
    L.rw_agg_dedup_drain.restype = ctypes.c_int
    L.rw_agg_dedup_drain.argtypes = [ctypes.c_void_p, ctypes.c_int,
                                     ctypes.POINTER(ctypes.c_void_p),
                                     ctypes.POINTER(ctypes.c_uint64)]
    L.rw_spill_free.argtypes = [ctypes.c_void_p]
    buf = ctypes.c_void_p()
    ln = ctypes.c_uint64()
    assert L.rw_agg_dedup_drain(agg.h, di, ctypes.byref(buf),
                                ctypes.byref(ln)) == 0
    data = ctypes.string_at(buf, ln.value)
    L.rw_spill_free(buf)
    return parse(data)


def n_dedup_tables(lib, agg):
    L = lib.lib
    L.rw_agg_n_dedup_tables.restype = ctypes.c_int
    L.rw_agg_n_dedup_tables.argtypes = [ctypes.c_void_p]
    return L.rw_agg_n_dedup_tables(agg.h)


def parse(data):
    recs = []
    off = 0
    while off < len(data):
        put = data[off]
        off += 1
        klen = struct.unpack_from("<I", data, off)[0]
        off += 4
        k = data[off:off + klen]
        off += klen
        vlen = struct.unpack_from("<I", data, off)[0]
        off += 4
        v = data[off:off + vlen]
        off += vlen
        recs.append((put, k, v))
    return recs


def memcmp_i64(v):
    return b"\x00" + struct.pack(">Q", (v & (2**64 - 1)) ^ (1 << 63))


def value_i64(v):
    return b"\x01" + struct.pack("<q", v)


def run_epoch(lib, chunks):
    calls = [(AGG_COUNT_STAR, -1, T_I64), (AGG_SUM, 1, T_I64)]
    agg = ffi.HashAgg(lib, [T_I64, T_I64], [0], calls, 0)
    for c in chunks:
        agg.push(c)
    agg.flush(1)
    agg.poll_all()
    recs = drain(lib, agg)
    agg.close()
    return agg, recs


def test_spill_record_bytes():
    _, recs = run_epoch(oracle(), [from_pretty(" I I\n + 5 7")])
    assert len(recs) == 1
    put, k, v = recs[0]
    assert put == 1
    assert k == memcmp_i64(5)
    # value = group key 5, count 1, sum 7 (value-encoded)
    assert v == value_i64(5) + value_i64(1) + value_i64(7)


def test_spill_delete_record():
    calls = [(AGG_COUNT_STAR, -1, T_I64), (AGG_SUM, 1, T_I64)]
    agg = ffi.HashAgg(oracle(), [T_I64, T_I64], [0], calls, 0)
    agg.push(from_pretty(" I I\n + 3 9"))
    agg.flush(1)
    agg.poll_all()
    drain(oracle(), agg)
    agg.push(from_pretty(" I I\n - 3 9"))
    agg.flush(2)
    agg.poll_all()
    recs = drain(oracle(), agg)
    agg.close()
    assert recs == [(0, memcmp_i64(3), b"")]


def test_memcmp_ordering_properties():
    # util/memcmp_encoding.rs:346-390 (ASC NULLS LAST): the encoded bytes
    # sort like the values, NULL largest
    _, recs = run_epoch(
        oracle(),
        [from_pretty(" I I\n + -1 1\n + 3874 1\n + 45745 1\n "
                     "+ -9223372036854775808 1\n + 9223372036854775807 1\n + . 1")],
    )
    keys = {tuple(r[1]): None for r in recs}
    ordered = sorted(keys)
    import struct as st

    def dec(k):
        kb = bytes(k)
        if kb[0] == 1:
            return None
        return st.unpack(">Q", kb[1:])[0] ^ (1 << 63)

    vals = [dec(k) for k in ordered]
    nonnull = [v - 2**64 if v >= 2**63 else v for v in vals if v is not None]
    assert nonnull == sorted(nonnull)
    assert vals[-1] is None  # NULL sorts largest


def test_join_spill_bytes():
    # §8f-2 join-state spill: record framing + memcmp(jk ∥ pk) key +
    # value-encoded row, with in-epoch netting (insert+delete cancels;
    # delete of a pre-epoch row emits a DELETE record)
    import struct

    from rwtest.ffi import JOIN_INNER, SIDE_LEFT, T_I64, from_pretty, oracle
    from rwtest import ffi

    o = ffi.HashJoin(oracle(), JOIN_INNER, [T_I64, T_I64], [T_I64, T_I64],
                     key_l=[0], key_r=[0], pk_l=[1], pk_r=[1])
    o.push(SIDE_LEFT, from_pretty(" I I\n + 5 100\n + 7 200\n - 7 200"))
    o.poll_all()
    sp = ffi.join_checkpoint_drain(oracle(), o.h, SIDE_LEFT)
    # only (5,100) survives the epoch: one PUT record
    def memcmp_i64(v):
        return b"\x00" + struct.pack(">q", v ^ -(1 << 63))
    def val_i64(v):
        return b"\x01" + struct.pack("<q", v)
    want_key = memcmp_i64(5) + memcmp_i64(100)
    want_val = val_i64(5) + val_i64(100)
    want = (b"\x01" + struct.pack("<I", len(want_key)) + want_key +
            struct.pack("<I", len(want_val)) + want_val)
    assert sp == want, f"{sp.hex()} vs {want.hex()}"
    # next epoch: delete the pre-epoch row -> one DELETE record
    o.push(SIDE_LEFT, from_pretty(" I I\n - 5 100"))
    o.poll_all()
    sp2 = ffi.join_checkpoint_drain(oracle(), o.h, SIDE_LEFT)
    want2 = (b"\x00" + struct.pack("<I", len(want_key)) + want_key +
             struct.pack("<I", 0))
    assert sp2 == want2, f"{sp2.hex()} vs {want2.hex()}"
    o.close()


def test_join_spill_tristate_netting():
    # across-epoch edge: pre-epoch row X; epoch does del X, ins X', del X'
    # -> net DELETE (X is gone); and del X, ins X' -> net PUT (overwrite)
    import struct

    from rwtest.ffi import JOIN_INNER, SIDE_LEFT, T_I64, from_pretty, oracle
    from rwtest import ffi

    def memcmp_i64(v):
        return b"\x00" + struct.pack(">q", v ^ -(1 << 63))

    def val_i64(v):
        return b"\x01" + struct.pack("<q", v)

    o = ffi.HashJoin(oracle(), JOIN_INNER, [T_I64, T_I64], [T_I64, T_I64],
                     key_l=[0], key_r=[0], pk_l=[1], pk_r=[1])
    o.push(SIDE_LEFT, from_pretty(" I I\n + 1 10\n + 2 20"))
    o.poll_all()
    ffi.join_checkpoint_drain(oracle(), o.h, SIDE_LEFT)  # epoch 1: 2 PUTs
    # epoch 2: key 1: del, re-ins (same pk), del again -> DELETE;
    #          key 2: del, re-ins -> PUT
    o.push(SIDE_LEFT, from_pretty(
        " I I\n - 1 10\n + 1 10\n - 1 10\n - 2 20\n + 2 20"))
    o.poll_all()
    sp = ffi.join_checkpoint_drain(oracle(), o.h, SIDE_LEFT)
    k1 = memcmp_i64(1) + memcmp_i64(10)
    k2 = memcmp_i64(2) + memcmp_i64(20)
    v2 = val_i64(2) + val_i64(20)
    want = (b"\x00" + struct.pack("<I", len(k1)) + k1 + struct.pack("<I", 0) +
            b"\x01" + struct.pack("<I", len(k2)) + k2 +
            struct.pack("<I", len(v2)) + v2)
    assert sp == want, f"{sp.hex()}\nvs\n{want.hex()}"
    o.close()


def test_dedup_spill_records():
    # §8f-2 DISTINCT dedup-table drain (distinct.rs:89-93,158-185): pk =
    # group ∥ datum (memcomparable), value = full row ++ one i64 count per
    # call distincting on the column; one record per touched (group, datum),
    # sorted by pk; DELETE at count 0; created+died in the epoch → nothing.
    from rwtest.ffi import AGG_COUNT

    calls = [(AGG_COUNT_STAR, -1, T_I64), (AGG_COUNT, 1, T_I64, 1)]
    o = ffi.HashAgg(oracle(), [T_I64, T_I64], [0], calls, 0)
    from test_codec import dedup_drain, n_dedup_tables  # self-import ok
    assert n_dedup_tables(oracle(), o) == 1
    o.push(from_pretty(""" I I
        + 1 5
        + 1 5
        + 1 7
        + 2 5
        + 1 9
        - 1 9"""))
    o.flush(1)
    o.poll_all()
    recs = dedup_drain(oracle(), o, 0)
    def k(g, d):
        return memcmp_i64(g) + memcmp_i64(d)
    def v(g, d, cnt):
        return value_i64(g) + value_i64(d) + value_i64(cnt)
    assert recs == [
        (1, k(1, 5), v(1, 5, 2)),
        (1, k(1, 7), v(1, 7, 1)),
        (1, k(2, 5), v(2, 5, 1)),
    ], recs
    # epoch 2: (1,5) drops to 1 → PUT; (1,7) drops to 0 → DELETE
    o.push(from_pretty(""" I I
        - 1 5
        - 1 7"""))
    o.flush(2)
    o.poll_all()
    recs = dedup_drain(oracle(), o, 0)
    assert recs == [
        (1, k(1, 5), v(1, 5, 1)),
        (0, k(1, 7), b""),
    ], recs
    # epoch 3: untouched → empty drain
    assert dedup_drain(oracle(), o, 0) == []
    o.close()


def test_eowc_spill_records():
    # §8f-2 EOWC spill (hash_agg.rs:429-474): every barrier PUTs the dirty
    # groups' current states even though nothing is emitted; a window close
    # DELETEs every intermediate row below the watermark, including
    # row_count-0 windows.
    calls = [(AGG_COUNT_STAR, -1, T_I64)]
    o = ffi.HashAgg(oracle(), [T_I64], [0], calls, 0,
                    emit_on_window_close=True)
    o.push(from_pretty(" I\n + 1\n + 2\n + 2"))
    o.flush(1)
    assert o.poll_all() == []  # EOWC: no emission without a watermark
    recs = drain(oracle(), o)
    assert recs == [
        (1, memcmp_i64(1), value_i64(1) + value_i64(1)),
        (1, memcmp_i64(2), value_i64(2) + value_i64(2)),
    ], recs
    # epoch 2: retract window 2 fully, then close windows < 3:
    # mid-window PUT for the dirty (now rc=0) window 2, then close DELETEs
    # for both windows — window 1 emits, window 2 does not. Drains emit in
    # memcmp-key order (stable within a key: window 2's PUT precedes its
    # close DELETE).
    o.push(from_pretty(" I\n - 2\n - 2"))
    o.watermark(0, 3)
    o.flush(2)
    o.poll_all()
    recs = drain(oracle(), o)
    assert recs == [
        (0, memcmp_i64(1), b""),
        (1, memcmp_i64(2), value_i64(2) + value_i64(0)),
        (0, memcmp_i64(2), b""),
    ], recs
    o.close()


def test_join_degree_spill_bytes():
    # §8f-2 degree-table drain (join/row.rs:99-113 build_degree_row): pk =
    # jk ∥ pk as the main table, value = order key ++ degree i64. Deltas =
    # main-table keys (insert/delete) plus pre-epoch rows whose degree
    # changed during probes of the other side.
    from rwtest.ffi import JOIN_LEFT_SEMI, SIDE_LEFT, SIDE_RIGHT

    o = ffi.HashJoin(oracle(), JOIN_LEFT_SEMI, [T_I64, T_I64],
                     [T_I64, T_I64], key_l=[0], key_r=[0], pk_l=[1],
                     pk_r=[1])
    k = memcmp_i64(1) + memcmp_i64(10)

    def deg_rec(put, deg=None):
        import struct
        if put:
            v = value_i64(1) + value_i64(10) + value_i64(deg)
            return (b"\x01" + struct.pack("<I", len(k)) + k +
                    struct.pack("<I", len(v)) + v)
        return b"\x00" + struct.pack("<I", len(k)) + k + struct.pack("<I", 0)

    # epoch 1: left row (1, 10), no right matches -> degree 0 PUT
    o.push(SIDE_LEFT, from_pretty(" I I\n + 1 10"))
    o.poll_all()
    ffi.join_checkpoint_drain(oracle(), o.h, SIDE_LEFT)
    assert ffi.join_degree_drain(oracle(), o.h, SIDE_LEFT) == deg_rec(1, 0)
    # epoch 2: right insert matches -> pre-epoch left row degree 1
    o.push(SIDE_RIGHT, from_pretty(" I I\n + 1 99"))
    o.poll_all()
    ffi.join_checkpoint_drain(oracle(), o.h, SIDE_LEFT)
    assert ffi.join_degree_drain(oracle(), o.h, SIDE_LEFT) == deg_rec(1, 1)
    # right side of a LEFT SEMI needs no degree table -> empty
    ffi.join_checkpoint_drain(oracle(), o.h, SIDE_RIGHT)
    assert ffi.join_degree_drain(oracle(), o.h, SIDE_RIGHT) == b""
    # epoch 3: right delete -> degree back to 0
    o.push(SIDE_RIGHT, from_pretty(" I I\n - 1 99"))
    o.poll_all()
    ffi.join_checkpoint_drain(oracle(), o.h, SIDE_LEFT)
    assert ffi.join_degree_drain(oracle(), o.h, SIDE_LEFT) == deg_rec(1, 0)
    # epoch 4: left delete -> degree-row DELETE
    o.push(SIDE_LEFT, from_pretty(" I I\n - 1 10"))
    o.poll_all()
    ffi.join_checkpoint_drain(oracle(), o.h, SIDE_LEFT)
    assert ffi.join_degree_drain(oracle(), o.h, SIDE_LEFT) == deg_rec(0)
    # epoch 5: untouched -> empty
    ffi.join_checkpoint_drain(oracle(), o.h, SIDE_LEFT)
    assert ffi.join_degree_drain(oracle(), o.h, SIDE_LEFT) == b""
    o.close()


def test_memcmp_f64_ordering_and_zero_identity():
    # util/memcmp_encoding.rs:543-590 (the legacy-2057 ordered-float
    # fixture, f64 projection): encoded keys sort -inf < -1 < 0 < 1 <
    # inf < NaN (NaN largest), and -0.0 encodes identically to +0.0 —
    # exercised through the GroupTopN storage key on an F64 order column
    import numpy as np

    from rwtest.ffi import T_F64, T_I64, topn_checkpoint_drain

    t = ffi.GroupTopN(oracle(), [T_F64, T_I64], [], [(0, False)],
                      [(1, False)], limit=10)
    vals = [float("-inf"), -1.0, 0.0, 1.0, float("inf"), float("nan")]
    n = len(vals)
    t.push(ffi.Chunk([T_F64, T_I64], np.zeros(n, np.uint8),
                     [np.array(vals, np.float64), np.arange(n)],
                     [np.ones(n, np.uint8)] * 2))
    t.poll_all()
    sp = topn_checkpoint_drain(oracle(), t.h)
    keys = []
    off = 0
    while off < len(sp):
        klen = int.from_bytes(sp[off + 1:off + 5], "little")
        keys.append(sp[off + 5:off + 5 + klen])
        off += 5 + klen
        vlen = int.from_bytes(sp[off:off + 4], "little")
        off += 4 + vlen
    # drains are memcmp-key sorted; recover the pk (second col) per key
    # and check it matches the value order of `vals`
    assert keys == sorted(keys)
    order = [int.from_bytes(k[-8:], "big") ^ (1 << 63) for k in keys]
    assert order == [0, 1, 2, 3, 4, 5], order  # -inf..nan ascending
    t.close()

    # -0.0 == +0.0: same storage key, the second insert upserts
    t2 = ffi.GroupTopN(oracle(), [T_F64, T_I64], [], [(0, False)],
                      [(1, False)], limit=10)
    t2.push(ffi.Chunk([T_F64, T_I64], np.zeros(2, np.uint8),
                      [np.array([0.0, -0.0], np.float64),
                       np.array([7, 7], np.int64)],
                      [np.ones(2, np.uint8)] * 2))
    t2.poll_all()
    sp2 = topn_checkpoint_drain(oracle(), t2.h)
    n_frames = 0
    off = 0
    while off < len(sp2):
        klen = int.from_bytes(sp2[off + 1:off + 5], "little")
        off += 5 + klen
        vlen = int.from_bytes(sp2[off:off + 4], "little")
        off += 4 + vlen
        n_frames += 1
    assert n_frames == 1, "-0.0 and +0.0 must share one storage key"
    t2.close()
