"""Vnode hashing + hash-dispatch parity (CPU).

The reference pins no hashed-key vnode fixtures in-repo (SURVEY §8c —
crc32fast is a Cargo.lock dependency), so the oracle's Crc32 is pinned
against Python's zlib.crc32, an independent IEEE CRC-32 implementation, on
the exact byte feed of hash_datum (types/mod.rs:1227-1233: native-endian
primitive bytes, NULL = u32 0xfffffff0 — array/mod.rs:99).

Dispatch semantics (dispatch.rs:949-1050): routing visibility and the
U-pair downgrade on dist-key change.
"""
import ctypes
import struct
import zlib

import numpy as np

from rwtest import ffi
from rwtest.ffi import T_I64, from_pretty, oracle


class RwVnodeDesc(ctypes.Structure):
    _fields_ = [("n_keys", ctypes.c_uint32),
                ("key_indices", ctypes.POINTER(ctypes.c_uint32)),
                ("vnode_count", ctypes.c_uint32)]


class RwDispatchDesc(ctypes.Structure):
    _fields_ = [("v", RwVnodeDesc), ("n_outputs", ctypes.c_uint32),
                ("vnode_to_output", ctypes.POINTER(ctypes.c_uint32))]


def bind(lib):
    L = lib.lib
    L.rw_vnode_compute.restype = ctypes.c_int
    L.rw_vnode_compute.argtypes = [ctypes.POINTER(RwVnodeDesc),
                                   ctypes.POINTER(ffi.RwChunkC),
                                   ctypes.POINTER(ctypes.c_uint16)]
    L.rw_dispatch_compute.restype = ctypes.c_int
    L.rw_dispatch_compute.argtypes = [ctypes.POINTER(RwDispatchDesc),
                                      ctypes.POINTER(ffi.RwChunkC),
                                      ctypes.POINTER(ctypes.POINTER(ffi.RwChunkC))]
    return L


def compute_vnodes(lib, chunk, keys, vnode_count=256):
    L = bind(lib)
    d = RwVnodeDesc()
    ki = (ctypes.c_uint32 * len(keys))(*keys)
    d.n_keys = len(keys)
    d.key_indices = ki
    d.vnode_count = vnode_count
    out = (ctypes.c_uint16 * chunk.n_rows)()
    cc = chunk.to_c()
    rc = L.rw_vnode_compute(ctypes.byref(d), ctypes.byref(cc), out)
    assert rc == 0
    return list(out)


def expected_vnode(row_datums, vnode_count=256):
    """Independent IEEE CRC-32 via zlib over the hash_datum byte feed."""
    buf = b""
    for v in row_datums:
        if v is None:
            buf += struct.pack("<I", 0xFFFFFFF0)
        else:
            buf += struct.pack("<q", v)
    return zlib.crc32(buf) % vnode_count


def test_vnode_against_zlib_crc32():
    rng = np.random.default_rng(11)
    n = 512
    keys = rng.integers(-(2**62), 2**62, n)
    vals = rng.integers(0, 100, n)
    valid = (rng.random(n) > 0.1).astype(np.uint8)
    c = ffi.Chunk([T_I64, T_I64], np.zeros(n, np.uint8), [keys, vals],
                  [valid, np.ones(n, np.uint8)])
    got = compute_vnodes(oracle(), c, [0])
    for r in range(n):
        k = int(keys[r]) if valid[r] else None
        assert got[r] == expected_vnode([k]), f"row {r}"


def test_vnode_multi_column():
    rng = np.random.default_rng(12)
    n = 256
    a = rng.integers(0, 1000, n)
    b = rng.integers(0, 1000, n)
    c = ffi.Chunk([T_I64, T_I64], np.zeros(n, np.uint8), [a, b],
                  [np.ones(n, np.uint8)] * 2)
    got = compute_vnodes(oracle(), c, [0, 1])
    for r in range(n):
        assert got[r] == expected_vnode([int(a[r]), int(b[r])])


def dispatch(lib, chunk, keys, n_outputs, vnode_count=256):
    L = bind(lib)
    d = RwDispatchDesc()
    ki = (ctypes.c_uint32 * len(keys))(*keys)
    d.v.n_keys = len(keys)
    d.v.key_indices = ki
    d.v.vnode_count = vnode_count
    d.n_outputs = n_outputs
    v2o = (ctypes.c_uint32 * vnode_count)(*[v % n_outputs for v in range(vnode_count)])
    d.vnode_to_output = v2o
    outs = (ctypes.POINTER(ffi.RwChunkC) * n_outputs)()
    cc = chunk.to_c()
    rc = L.rw_dispatch_compute(ctypes.byref(d), ctypes.byref(cc), outs)
    assert rc == 0
    return [lib._read_chunk(outs[o]) for o in range(n_outputs)]


def test_dispatch_partitions_rows():
    rng = np.random.default_rng(13)
    n = 1024
    keys = rng.integers(0, 500, n)
    vals = rng.integers(0, 100, n)
    c = ffi.Chunk([T_I64, T_I64], np.zeros(n, np.uint8), [keys, vals],
                  [np.ones(n, np.uint8)] * 2)
    outs = dispatch(oracle(), c, [0], 4)
    # each visible row appears in exactly one output, on the right shard
    total = 0
    for o, oc in enumerate(outs):
        for op, row in oc.visible_rows():
            assert expected_vnode([row[0]]) % 4 == o
            total += 1
    assert total == n
    # same key ⇒ same output (shard-stability)
    seen = {}
    for o, oc in enumerate(outs):
        for op, row in oc.visible_rows():
            assert seen.setdefault(row[0], o) == o


def test_dispatch_update_pair_downgrade():
    # dispatch.rs:985-1010: U−/U+ with changed dist key become Delete/Insert
    c = from_pretty(
        """ I I
        U- 1 10
        U+ 2 10
        U- 3 7
        U+ 3 8"""
    )
    outs = dispatch(oracle(), c, [0], 1)
    rows = list(outs[0].visible_rows())
    ops = [r[0] for r in rows]
    assert ops == ["-", "+", "U-", "U+"], ops


def test_dispatch_shard_union_equals_single():
    """End-to-end: vnode-dispatch a stream to 2 shards, run the agg on each,
    union of outputs == single-executor output (the §8e invariant)."""
    rng = np.random.default_rng(14)
    chunks = []
    for _ in range(4):
        n = 512
        keys = rng.integers(0, 40, n)
        vals = rng.integers(1, 100, n)
        chunks.append(ffi.Chunk([T_I64, T_I64], np.zeros(n, np.uint8),
                                [keys, vals], [np.ones(n, np.uint8)] * 2))

    calls = [(ffi.AGG_COUNT_STAR, -1, T_I64), (ffi.AGG_SUM, 1, T_I64)]

    def run(inputs):
        agg = ffi.HashAgg(oracle(), [T_I64, T_I64], [0], calls, 0)
        out = []
        for e in range(2):
            for ch in inputs[e * 2:(e + 1) * 2]:
                if isinstance(ch, list):
                    for sub in ch:
                        agg.push(sub)
                else:
                    agg.push(ch)
            agg.flush(e + 1)
            out.extend(ffi.rows_multiset(agg.poll_all()))
        agg.close()
        return sorted(out)

    single = run(chunks)
    shard_in = [[], []]
    for ch in chunks:
        outs = dispatch(oracle(), ch, [0], 2)
        shard_in[0].append(outs[0])
        shard_in[1].append(outs[1])
    union = sorted(run(shard_in[0]) + run(shard_in[1]))
    assert union == single


def test_update_vnode_bitmap_oracle():
    # rescale re-scope: groups whose vnode bit is cleared are dropped with
    # NO retraction; a later row for a dropped key restarts the group
    # (row_count 0 -> n => Insert, agg_group.rs:131-165)
    from rwtest.ffi import AGG_COUNT_STAR, AGG_SUM, rows_multiset

    calls = [(AGG_COUNT_STAR, -1, T_I64), (AGG_SUM, 1, T_I64)]
    a = ffi.HashAgg(oracle(), [T_I64, T_I64], [0], calls, 0)
    keys = list(range(16))
    c = from_pretty("I I\n" + "\n".join(f"+ {k} {10*k+1}" for k in keys))
    a.push(c)
    a.flush(1)
    a.poll_all()

    def vnode(k):
        crc = zlib.crc32(struct.pack("<q", k))
        return crc % 256

    owned = bytearray(32)
    keep = set()
    for k in keys:
        if vnode(k) % 2 == 0:  # keep even vnodes
            keep.add(k)
            owned[vnode(k) >> 3] |= 1 << (vnode(k) & 7)
    assert 0 < len(keep) < len(keys)
    a.update_vnode_bitmap(bytes(owned))
    # push one more row per key: kept keys -> Update pair (count 1->2);
    # dropped keys restart -> Insert with count 1
    c2 = from_pretty("I I\n" + "\n".join(f"+ {k} {10*k+2}" for k in keys))
    a.push(c2)
    a.flush(2)
    got = rows_multiset(a.poll_all())
    want = []
    for k in keys:
        if k in keep:
            want.append(("U-", (k, 1, 10 * k + 1)))
            want.append(("U+", (k, 2, 20 * k + 3)))
        else:
            want.append(("+", (k, 1, 10 * k + 2)))
    key = lambda r: (r[0], r[1])
    assert got == sorted(want, key=key)
    a.close()
