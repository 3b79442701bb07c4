"""State restore / crash-recovery parity (rw_stream.h
restore contract). The reference rebuilds executor state from its state
tables on recovery (src/meta/src/barrier/worker.rs:1074) and on cache miss
(join/hash_join.rs:232-260); here the drained §8f-2 spill records are the
state store's view, and rw_hash_{agg,join}_restore must rebuild a fresh
executor so that continuing produces EXACTLY what an uninterrupted run
produces — same emissions, same subsequent drain bytes."""
import numpy as np
import pytest

from rwtest import ffi
from rwtest.ffi import (AGG_COUNT_STAR, AGG_MAX, AGG_SUM, JOIN_INNER,
                        JOIN_LEFT_SEMI, SIDE_LEFT, SIDE_RIGHT, T_I64,
                        agg_checkpoint_drain_bytes, agg_restore,
                        join_checkpoint_drain, join_degree_drain,
                        join_restore, oracle, rows_multiset)


def mk_chunk(types, ops, cols, valids=None):
    cols = [np.asarray(c, np.int64) for c in cols]
    n = len(cols[0])
    if valids is None:
        valids = [np.ones(n, np.uint8) for _ in cols]
    return ffi.Chunk(types, np.asarray(ops, np.uint8), cols, valids)


def _agg_epoch_chunks(rng, epoch, live):
    """Insert/retract mix over a bounded group space; `live` tracks rows
    (group, val) inserted so retracts target real state."""
    chunks = []
    for _ in range(3):
        n = 512
        g = rng.integers(0, 40, n)
        v = rng.integers(1, 1000, n)
        ops = np.zeros(n, np.uint8)
        for r in range(n):
            if live and rng.random() < 0.25:
                j = int(rng.integers(0, len(live)))
                g[r], v[r] = live.pop(j)
                ops[r] = ffi.OP_DELETE
            else:
                live.append((int(g[r]), int(v[r])))
        chunks.append(mk_chunk([T_I64, T_I64], ops, [g, v]))
    return chunks


def _drive_agg(a, chunks, epoch):
    for c in chunks:
        a.push(c)
    a.flush(epoch)
    return rows_multiset(a.poll_all())


def _agg_restore_flow(lib):
    calls = [(AGG_SUM, 1, T_I64), (AGG_COUNT_STAR, -1, T_I64)]
    mk = lambda: ffi.HashAgg(lib, [T_I64, T_I64], [0], calls, 1)
    a = mk()
    rng = np.random.default_rng(1234)
    live = []
    drains = []
    # 3 pre-crash epochs
    pre_inputs = [_agg_epoch_chunks(rng, e, live) for e in range(3)]
    for e, chunks in enumerate(pre_inputs):
        _drive_agg(a, chunks, e + 1)
        drains.append(agg_checkpoint_drain_bytes(lib, a.h))
    # post-crash epochs, same for both executors
    post_inputs = [_agg_epoch_chunks(rng, 3 + e, live) for e in range(3)]
    b = mk()
    agg_restore(lib, b.h, b"".join(drains))
    outs_a, outs_b, dr_a, dr_b = [], [], [], []
    for e, chunks in enumerate(post_inputs):
        outs_a.append(_drive_agg(a, chunks, 4 + e))
        outs_b.append(_drive_agg(b, chunks, 4 + e))
        dr_a.append(agg_checkpoint_drain_bytes(lib, a.h))
        dr_b.append(agg_checkpoint_drain_bytes(lib, b.h))
    a.close()
    b.close()
    assert outs_a == outs_b, "restored agg diverged from uninterrupted run"
    assert dr_a == dr_b, "restored agg spill stream diverged"


def test_agg_restore_oracle():
    _agg_restore_flow(oracle())


@pytest.mark.gpu
def test_agg_restore_gpu():
    import risingwave_amd

    risingwave_amd.load_library()
    _agg_restore_flow(ffi.Lib(risingwave_amd.lib_path()))


def _join_epoch_pushes(rng, live, pk_counter):
    pushes = []
    for _ in range(2):
        for side in (SIDE_LEFT, SIDE_RIGHT):
            n = 384
            k = rng.integers(0, 60, n)
            v = np.arange(pk_counter[0], pk_counter[0] + n)
            pk_counter[0] += n
            ops = np.zeros(n, np.uint8)
            for r in range(n):
                if live[side] and rng.random() < 0.25:
                    j = int(rng.integers(0, len(live[side])))
                    k[r], v[r] = live[side].pop(j)
                    ops[r] = ffi.OP_DELETE
                else:
                    live[side].append((int(k[r]), int(v[r])))
            pushes.append((side, mk_chunk([T_I64, T_I64], ops, [k, v])))
    return pushes


def _join_restore_flow(lib, join_type):
    mk = lambda: ffi.HashJoin(lib, join_type, [T_I64, T_I64],
                              [T_I64, T_I64], key_l=[0], key_r=[0],
                              pk_l=[1], pk_r=[1])
    a = mk()
    rng = np.random.default_rng(99)
    live = {SIDE_LEFT: [], SIDE_RIGHT: []}
    pk_counter = [0]
    state = {s: b"" for s in (SIDE_LEFT, SIDE_RIGHT)}
    degs = {s: b"" for s in (SIDE_LEFT, SIDE_RIGHT)}
    pre = [_join_epoch_pushes(rng, live, pk_counter) for _ in range(3)]
    for e, pushes in enumerate(pre):
        for side, c in pushes:
            a.push(side, c)
            a.poll_all()
        for s in (SIDE_LEFT, SIDE_RIGHT):
            state[s] += join_checkpoint_drain(lib, a.h, s)
            degs[s] += join_degree_drain(lib, a.h, s)
    post = [_join_epoch_pushes(rng, live, pk_counter) for _ in range(3)]
    b = mk()
    for s in (SIDE_LEFT, SIDE_RIGHT):
        join_restore(lib, b.h, s, state[s], degs[s])
    for e, pushes in enumerate(post):
        for side, c in pushes:
            a.push(side, c)
            b.push(side, c)
            ma = rows_multiset(a.poll_all())
            mb = rows_multiset(b.poll_all())
            assert ma == mb, (f"epoch {e}: restored join diverged "
                              f"({len(ma)} vs {len(mb)} rows)")
        for s in (SIDE_LEFT, SIDE_RIGHT):
            da = join_checkpoint_drain(lib, a.h, s)
            db = join_checkpoint_drain(lib, b.h, s)
            assert da == db, f"epoch {e} side {s}: state drain diverged"
            ga = join_degree_drain(lib, a.h, s)
            gb = join_degree_drain(lib, b.h, s)
            assert ga == gb, f"epoch {e} side {s}: degree drain diverged"
    a.close()
    b.close()


def test_join_restore_oracle_inner():
    _join_restore_flow(oracle(), JOIN_INNER)


def test_join_restore_oracle_semi_degrees():
    _join_restore_flow(oracle(), JOIN_LEFT_SEMI)


@pytest.mark.gpu
def test_join_restore_gpu_inner():
    import risingwave_amd

    risingwave_amd.load_library()
    _join_restore_flow(ffi.Lib(risingwave_amd.lib_path()), JOIN_INNER)


@pytest.mark.gpu
def test_join_restore_gpu_semi_degrees():
    import risingwave_amd

    risingwave_amd.load_library()
    _join_restore_flow(ffi.Lib(risingwave_amd.lib_path()), JOIN_LEFT_SEMI)


@pytest.mark.gpu
def test_restore_gpu_matches_oracle():
    # cross-check: a GPU executor restored from GPU drains continues in
    # lockstep with an ORACLE restored from ORACLE drains
    import risingwave_amd

    risingwave_amd.load_library()
    glib = ffi.Lib(risingwave_amd.lib_path())
    # value states only (restore rejects materialized-input aggregates)
    calls = [(AGG_SUM, 1, T_I64), (AGG_COUNT_STAR, -1, T_I64)]
    rng = np.random.default_rng(7)
    execs = {}
    drains = {}
    for name, lib in (("gpu", glib), ("orc", oracle())):
        a = ffi.HashAgg(lib, [T_I64, T_I64], [0], calls, 1)
        r = np.random.default_rng(7)
        live = []
        d = b""
        for e in range(3):
            _drive_agg(a, _agg_epoch_chunks(r, e, live), e + 1)
            d += agg_checkpoint_drain_bytes(lib, a.h)
        a.close()
        b = ffi.HashAgg(lib, [T_I64, T_I64], [0], calls, 1)
        agg_restore(lib, b.h, d)
        execs[name] = (lib, b)
        drains[name] = d
    assert drains["gpu"] == drains["orc"], "pre-crash drains diverged"
    r1 = np.random.default_rng(8)
    r2 = np.random.default_rng(8)
    live1, live2 = [], []
    for e in range(3):
        og = _drive_agg(execs["gpu"][1], _agg_epoch_chunks(r1, e, live1), e)
        oo = _drive_agg(execs["orc"][1], _agg_epoch_chunks(r2, e, live2), e)
        assert og == oo, f"epoch {e}: restored GPU != restored oracle"
    for lib, b in execs.values():
        b.close()


def _agg_minput_restore_flow(lib):
    # retractable min/max (materialized-input state TABLES, spilled per
    # call): restore = minput tables FIRST, then the intermediate table
    from rwtest.ffi import (AGG_MIN, agg_minput_drain_bytes,
                            agg_minput_restore)

    calls = [(ffi.AGG_MIN, 1, T_I64), (AGG_MAX, 1, T_I64),
             (AGG_COUNT_STAR, -1, T_I64)]
    # stream key = col 2 (unique row id) so minput rows have distinct pks
    mk = lambda: ffi.HashAgg(lib, [T_I64, T_I64, T_I64], [0], calls, 2,
                             stream_key=(2,))
    a = mk()
    rng = np.random.default_rng(55)
    live = []
    rid = [0]

    def chunks():
        out = []
        for _ in range(3):
            n = 384
            g = rng.integers(0, 25, n)
            v = rng.integers(-500, 500, n)
            r2 = np.arange(rid[0], rid[0] + n)
            rid[0] += n
            ops = np.zeros(n, np.uint8)
            for r in range(n):
                if live and rng.random() < 0.3:
                    j = int(rng.integers(0, len(live)))
                    g[r], v[r], r2[r] = live.pop(j)
                    ops[r] = ffi.OP_DELETE
                else:
                    live.append((int(g[r]), int(v[r]), int(r2[r])))
            out.append(mk_chunk([T_I64, T_I64, T_I64], ops, [g, v, r2]))
        return out

    inter = b""
    minp = [b"", b""]
    for e in range(3):
        _drive_agg(a, chunks(), e + 1)
        inter += agg_checkpoint_drain_bytes(lib, a.h)
        for mi in range(2):
            minp[mi] += agg_minput_drain_bytes(lib, a.h, mi)
    b = mk()
    for mi in range(2):
        agg_minput_restore(lib, b.h, mi, minp[mi])
    agg_restore(lib, b.h, inter)
    for e in range(3):
        cs = chunks()
        oa = _drive_agg(a, cs, 5 + e)
        ob = _drive_agg(b, cs, 5 + e)
        assert oa == ob, f"epoch {e}: restored minput agg diverged"
        da = agg_checkpoint_drain_bytes(lib, a.h)
        db = agg_checkpoint_drain_bytes(lib, b.h)
        assert da == db, f"epoch {e}: intermediate drain diverged"
        for mi in range(2):
            ma = agg_minput_drain_bytes(lib, a.h, mi)
            mb = agg_minput_drain_bytes(lib, b.h, mi)
            assert ma == mb, f"epoch {e} minput {mi}: drain diverged"
    a.close()
    b.close()


def test_agg_minput_restore_oracle():
    _agg_minput_restore_flow(oracle())


@pytest.mark.gpu
def test_agg_minput_restore_gpu():
    import risingwave_amd

    risingwave_amd.load_library()
    _agg_minput_restore_flow(ffi.Lib(risingwave_amd.lib_path()))


def _topn_epoch_chunks(rng, live, rid):
    """Insert/delete mix over bounded groups; rows keyed by a unique id so
    deletes target real state rows by their full cache key."""
    chunks = []
    for _ in range(3):
        n = 256
        g = rng.integers(0, 16, n)
        v = rng.integers(0, 200, n)
        r2 = np.arange(rid[0], rid[0] + n)
        rid[0] += n
        ops = np.zeros(n, np.uint8)
        for r in range(n):
            if live and rng.random() < 0.3:
                j = int(rng.integers(0, len(live)))
                g[r], v[r], r2[r] = live.pop(j)
                ops[r] = ffi.OP_DELETE
            else:
                live.append((int(g[r]), int(v[r]), int(r2[r])))
        chunks.append(mk_chunk([T_I64, T_I64, T_I64], ops, [g, v, r2]))
    return chunks


def _topn_restore_flow(lib):
    from rwtest.ffi import topn_checkpoint_drain, topn_restore

    mk = lambda: ffi.GroupTopN(lib, [T_I64, T_I64, T_I64], [0],
                               [(1, False)], [(2, False)], offset=1, limit=3)
    a = mk()
    rng = np.random.default_rng(4242)
    live = []
    rid = [0]
    drains = []
    for e in range(3):
        for c in _topn_epoch_chunks(rng, live, rid):
            a.push(c)
        a.poll_all()
        drains.append(topn_checkpoint_drain(lib, a.h))
    b = mk()
    topn_restore(lib, b.h, b"".join(drains))
    for e in range(3):
        cs = _topn_epoch_chunks(rng, live, rid)
        for c in cs:
            a.push(c)
            b.push(c)
            ma = rows_multiset(a.poll_all())
            mb = rows_multiset(b.poll_all())
            assert ma == mb, f"epoch {e}: restored topn diverged"
        da = topn_checkpoint_drain(lib, a.h)
        db = topn_checkpoint_drain(lib, b.h)
        assert da == db, f"epoch {e}: topn state drain diverged"
    a.close()
    b.close()


def test_topn_restore_oracle():
    _topn_restore_flow(oracle())


@pytest.mark.gpu
def test_topn_restore_gpu():
    import risingwave_amd

    risingwave_amd.load_library()
    _topn_restore_flow(ffi.Lib(risingwave_amd.lib_path()))


@pytest.mark.gpu
def test_topn_restore_gpu_matches_oracle_drains():
    # a GPU executor restored from GPU drains and an oracle restored from
    # oracle drains must produce byte-identical subsequent drains
    import risingwave_amd
    from rwtest.ffi import topn_checkpoint_drain, topn_restore

    risingwave_amd.load_library()
    glib = ffi.Lib(risingwave_amd.lib_path())
    streams = {}
    for name, lib in (("gpu", glib), ("orc", oracle())):
        t = ffi.GroupTopN(lib, [T_I64, T_I64, T_I64], [0],
                          [(1, True)], [(2, False)], limit=2)
        rng = np.random.default_rng(31)
        live, rid = [], [0]
        d = b""
        for e in range(2):
            for c in _topn_epoch_chunks(rng, live, rid):
                t.push(c)
            t.poll_all()
            d += topn_checkpoint_drain(lib, t.h)
        t.close()
        t2 = ffi.GroupTopN(lib, [T_I64, T_I64, T_I64], [0],
                           [(1, True)], [(2, False)], limit=2)
        topn_restore(lib, t2.h, d)
        for c in _topn_epoch_chunks(rng, live, rid):
            t2.push(c)
        t2.poll_all()
        streams[name] = (d, topn_checkpoint_drain(lib, t2.h))
        t2.close()
    assert streams["gpu"] == streams["orc"]


def _agg_dedup_restore_flow(lib):
    # COUNT(DISTINCT col) + plain SUM: the dedup counter table spills per
    # epoch and must restore so duplicate-visibility transitions (0→1, 1→0)
    # continue exactly as uninterrupted
    from rwtest.ffi import (AGG_COUNT, agg_dedup_drain_bytes,
                            agg_dedup_restore)

    calls = [(AGG_COUNT, 1, T_I64, 1), (AGG_SUM, 1, T_I64)]
    mk = lambda: ffi.HashAgg(lib, [T_I64, T_I64], [0], calls, 1)
    a = mk()
    rng = np.random.default_rng(777)
    live = []

    def chunks():
        # small value domain (0..12) so distinct counts really exercise
        # duplicate suppression both ways
        out = []
        for _ in range(3):
            n = 256
            g = rng.integers(0, 10, n)
            v = rng.integers(0, 12, n)
            ops = np.zeros(n, np.uint8)
            for r in range(n):
                if live and rng.random() < 0.35:
                    j = int(rng.integers(0, len(live)))
                    g[r], v[r] = live.pop(j)
                    ops[r] = ffi.OP_DELETE
                else:
                    live.append((int(g[r]), int(v[r])))
            out.append(mk_chunk([T_I64, T_I64], ops, [g, v]))
        return out

    inter = b""
    ded = b""
    for e in range(3):
        _drive_agg(a, chunks(), e + 1)
        inter += agg_checkpoint_drain_bytes(lib, a.h)
        ded += agg_dedup_drain_bytes(lib, a.h, 0)
    b = mk()
    agg_dedup_restore(lib, b.h, 0, ded)
    agg_restore(lib, b.h, inter)
    for e in range(3):
        cs = chunks()
        oa = _drive_agg(a, cs, 5 + e)
        ob = _drive_agg(b, cs, 5 + e)
        assert oa == ob, f"epoch {e}: restored distinct agg diverged"
        da = agg_checkpoint_drain_bytes(lib, a.h)
        db = agg_checkpoint_drain_bytes(lib, b.h)
        assert da == db, f"epoch {e}: intermediate drain diverged"
        xa = agg_dedup_drain_bytes(lib, a.h, 0)
        xb = agg_dedup_drain_bytes(lib, b.h, 0)
        assert xa == xb, f"epoch {e}: dedup drain diverged"
    a.close()
    b.close()


def test_agg_dedup_restore_oracle():
    _agg_dedup_restore_flow(oracle())


@pytest.mark.gpu
def test_agg_dedup_restore_gpu():
    import risingwave_amd

    risingwave_amd.load_library()
    _agg_dedup_restore_flow(ffi.Lib(risingwave_amd.lib_path()))


def _join_clean_restore_flow(lib, join_type):
    """Watermark state-cleaning must surface in the spill stream: cleaned
    rows net to DELETE frames so a restore replay cannot resurrect them
    (the reference's commit applies the watermark as a range delete,
    state_table.rs:1707). Returns all drain streams for cross-impl
    byte-comparison."""
    mk = lambda: ffi.HashJoin(lib, join_type, [T_I64, T_I64],
                              [T_I64, T_I64], key_l=[0], key_r=[0],
                              pk_l=[1], pk_r=[1], wm_jk=((0, True),))
    rng = np.random.default_rng(616)
    pk = [0]

    def epoch_pushes(e):
        pushes = []
        for side in (SIDE_LEFT, SIDE_RIGHT):
            n = 200
            k = rng.integers(e * 10, e * 10 + 25, n)
            v = np.arange(pk[0], pk[0] + n)
            pk[0] += n
            pushes.append((side, mk_chunk([T_I64, T_I64],
                                          np.zeros(n, np.uint8), [k, v])))
        return pushes

    a = mk()
    state = {s: b"" for s in (SIDE_LEFT, SIDE_RIGHT)}
    degs = {s: b"" for s in (SIDE_LEFT, SIDE_RIGHT)}
    streams = []
    for e in range(3):
        for side, c in epoch_pushes(e):
            a.push(side, c)
            a.poll_all()
        # watermark advances on both sides -> clean below e*10
        a.watermark(SIDE_LEFT, 0, e * 10)
        a.watermark(SIDE_RIGHT, 0, e * 10)
        for s in (SIDE_LEFT, SIDE_RIGHT):
            d = join_checkpoint_drain(lib, a.h, s)
            state[s] += d
            streams.append(d)
            g = join_degree_drain(lib, a.h, s)
            degs[s] += g
            streams.append(g)
    # crash + restore
    b = mk()
    for s in (SIDE_LEFT, SIDE_RIGHT):
        join_restore(lib, b.h, s, state[s], degs[s])
    # replay the SAME watermark state on the restored executor (the
    # embedder re-establishes watermarks on recovery, as the reference's
    # barrier/actor context does)
    b.watermark(SIDE_LEFT, 0, 2 * 10)
    b.watermark(SIDE_RIGHT, 0, 2 * 10)
    for e in range(3, 6):
        for side, c in epoch_pushes(e):
            a.push(side, c)
            b.push(side, c)
            ma = rows_multiset(a.poll_all())
            mb = rows_multiset(b.poll_all())
            assert ma == mb, f"epoch {e}: restored join diverged after clean"
        a.watermark(SIDE_LEFT, 0, e * 10)
        a.watermark(SIDE_RIGHT, 0, e * 10)
        b.watermark(SIDE_LEFT, 0, e * 10)
        b.watermark(SIDE_RIGHT, 0, e * 10)
        for s in (SIDE_LEFT, SIDE_RIGHT):
            da = join_checkpoint_drain(lib, a.h, s)
            db = join_checkpoint_drain(lib, b.h, s)
            assert da == db, f"epoch {e} side {s}: drain diverged after clean"
            streams.append(da)
            ga = join_degree_drain(lib, a.h, s)
            gb = join_degree_drain(lib, b.h, s)
            assert ga == gb, f"epoch {e} side {s}: degree drain diverged"
            streams.append(ga)
    a.close()
    b.close()
    return streams


def _join_compact_flow(lib, join_type):
    """rw_join_compact between epochs: logical state, emissions, drains
    and restores are unchanged while dead records are reclaimed."""
    from rwtest.ffi import join_compact

    mk = lambda: ffi.HashJoin(lib, join_type, [T_I64, T_I64],
                              [T_I64, T_I64], key_l=[0], key_r=[0],
                              pk_l=[1], pk_r=[1], wm_jk=((0, True),))
    rng = np.random.default_rng(2727)
    pk = [0]
    live = {SIDE_LEFT: [], SIDE_RIGHT: []}

    def epoch_pushes(e):
        pushes = []
        for side in (SIDE_LEFT, SIDE_RIGHT):
            n = 250
            k = rng.integers(e * 10, e * 10 + 30, n)
            v = np.arange(pk[0], pk[0] + n)
            pk[0] += n
            ops = np.zeros(n, np.uint8)
            for r in range(n):
                if live[side] and rng.random() < 0.3:
                    jx = int(rng.integers(0, len(live[side])))
                    k[r], v[r] = live[side].pop(jx)
                    ops[r] = ffi.OP_DELETE
                else:
                    live[side].append((int(k[r]), int(v[r])))
            pushes.append((side, mk_chunk([T_I64, T_I64], ops, [k, v])))
        return pushes

    # a compacts every epoch, c never does; both track the oracle flow
    a, c = mk(), mk()
    freed_total = 0
    state = {s: b"" for s in (SIDE_LEFT, SIDE_RIGHT)}
    degs = {s: b"" for s in (SIDE_LEFT, SIDE_RIGHT)}
    wm = 0
    for e in range(5):
        for side, ch in epoch_pushes(e):
            for x in (a, c):
                x.push(side, ch)
            ma = rows_multiset(a.poll_all())
            mc = rows_multiset(c.poll_all())
            assert ma == mc, f"epoch {e}: compacted join diverged"
        wm = e * 10
        for x in (a, c):
            x.watermark(SIDE_LEFT, 0, wm)
            x.watermark(SIDE_RIGHT, 0, wm)
        for s in (SIDE_LEFT, SIDE_RIGHT):
            da = join_checkpoint_drain(lib, a.h, s)
            dc = join_checkpoint_drain(lib, c.h, s)
            assert da == dc, f"epoch {e} side {s}: drain diverged"
            ga = join_degree_drain(lib, a.h, s)
            gc = join_degree_drain(lib, c.h, s)
            assert ga == gc, f"epoch {e} side {s}: degree drain diverged"
            state[s] += da
            degs[s] += ga
            freed_total += join_compact(lib, a.h, s)
    # a restore from streams spanning the compactions must continue in
    # lockstep with the compacted executor
    b = mk()
    for s in (SIDE_LEFT, SIDE_RIGHT):
        join_restore(lib, b.h, s, state[s], degs[s])
    b.watermark(SIDE_LEFT, 0, wm)
    b.watermark(SIDE_RIGHT, 0, wm)
    for side, ch in epoch_pushes(5):
        a.push(side, ch)
        b.push(side, ch)
        ma = rows_multiset(a.poll_all())
        mb = rows_multiset(b.poll_all())
        assert ma == mb, "restore across compactions diverged"
    for s in (SIDE_LEFT, SIDE_RIGHT):
        da = join_checkpoint_drain(lib, a.h, s)
        db = join_checkpoint_drain(lib, b.h, s)
        assert da == db, f"side {s}: post-compaction restore drain diverged"
    a.close()
    b.close()
    c.close()
    return freed_total


def _minput_topn_compact_flow(lib):
    """rw_agg_minput_compact / rw_topn_compact: reclamation with no
    observable change (emissions + drains match a non-compacting twin)."""
    from rwtest.ffi import (AGG_MIN, agg_minput_compact,
                            agg_minput_drain_bytes, topn_checkpoint_drain,
                            topn_compact)

    calls = [(AGG_MIN, 1, T_I64), (AGG_COUNT_STAR, -1, T_I64)]
    mka = lambda: ffi.HashAgg(lib, [T_I64, T_I64, T_I64], [0], calls, 1,
                              stream_key=(2,))
    mkt = lambda: ffi.GroupTopN(lib, [T_I64, T_I64, T_I64], [0],
                                [(1, False)], [(2, False)], limit=2)
    a1, a2 = mka(), mka()
    t1, t2 = mkt(), mkt()
    rng = np.random.default_rng(888)
    live_a, live_t, rid = [], [], [0]
    freed = 0
    for e in range(4):
        n = 200
        g = rng.integers(0, 12, n)
        v = rng.integers(-50, 50, n)
        r2 = np.arange(rid[0], rid[0] + n)
        rid[0] += n
        ops = np.zeros(n, np.uint8)
        for r in range(n):
            if live_a and rng.random() < 0.4:
                j = int(rng.integers(0, len(live_a)))
                g[r], v[r], r2[r] = live_a.pop(j)
                ops[r] = ffi.OP_DELETE
            else:
                live_a.append((int(g[r]), int(v[r]), int(r2[r])))
        c = mk_chunk([T_I64] * 3, ops, [g, v, r2])
        oa = _drive_agg(a1, [c], e + 1)
        ob = _drive_agg(a2, [c], e + 1)
        assert oa == ob, f"epoch {e}: compacted minput agg diverged"
        for x in (a1, a2):
            agg_checkpoint_drain_bytes(lib, x.h)
        d1 = agg_minput_drain_bytes(lib, a1.h, 0)
        d2 = agg_minput_drain_bytes(lib, a2.h, 0)
        assert d1 == d2, f"epoch {e}: minput drain diverged"
        freed += agg_minput_compact(lib, a1.h)
        # TopN twin flow with deletes
        gt = rng.integers(0, 10, n)
        vt = rng.integers(0, 60, n)
        rt = np.arange(rid[0], rid[0] + n)
        rid[0] += n
        opst = np.zeros(n, np.uint8)
        for r in range(n):
            if live_t and rng.random() < 0.4:
                j = int(rng.integers(0, len(live_t)))
                gt[r], vt[r], rt[r] = live_t.pop(j)
                opst[r] = ffi.OP_DELETE
            else:
                live_t.append((int(gt[r]), int(vt[r]), int(rt[r])))
        ct = mk_chunk([T_I64] * 3, opst, [gt, vt, rt])
        t1.push(ct)
        t2.push(ct)
        m1 = rows_multiset(t1.poll_all())
        m2 = rows_multiset(t2.poll_all())
        assert m1 == m2, f"epoch {e}: compacted topn diverged"
        s1 = topn_checkpoint_drain(lib, t1.h)
        s2 = topn_checkpoint_drain(lib, t2.h)
        assert s1 == s2, f"epoch {e}: topn drain diverged"
        freed += topn_compact(lib, t1.h)
    for x in (a1, a2, t1, t2):
        x.close()
    return freed


def test_minput_topn_compact_oracle_noop():
    assert _minput_topn_compact_flow(oracle()) == 0


@pytest.mark.gpu
def test_minput_topn_compact_gpu():
    import risingwave_amd

    risingwave_amd.load_library()
    glib = ffi.Lib(risingwave_amd.lib_path())
    freed = _minput_topn_compact_flow(glib)
    assert freed > 0, "compaction reclaimed nothing despite retractions"


def test_join_compact_oracle_noop():
    assert _join_compact_flow(oracle(), JOIN_INNER) == 0


@pytest.mark.gpu
def test_join_compact_gpu():
    import risingwave_amd

    risingwave_amd.load_library()
    glib = ffi.Lib(risingwave_amd.lib_path())
    # deletes + watermark sweeps retire rows, so compaction must free
    freed = _join_compact_flow(glib, JOIN_INNER)
    assert freed > 0, "compaction reclaimed nothing despite retired rows"
    _join_compact_flow(glib, JOIN_LEFT_SEMI)


@pytest.mark.gpu
def test_join_compact_requires_drained():
    import risingwave_amd
    from rwtest.ffi import join_compact

    risingwave_amd.load_library()
    glib = ffi.Lib(risingwave_amd.lib_path())
    j = ffi.HashJoin(glib, JOIN_INNER, [T_I64, T_I64], [T_I64, T_I64],
                     key_l=[0], key_r=[0], pk_l=[1], pk_r=[1])
    j.push(SIDE_LEFT, mk_chunk([T_I64, T_I64], [0], [[1], [2]]))
    j.poll_all()
    with pytest.raises(RuntimeError, match="undrained|pending"):
        join_compact(glib, j.h, SIDE_LEFT)
    join_checkpoint_drain(glib, j.h, SIDE_LEFT)
    join_compact(glib, j.h, SIDE_LEFT)  # drained: succeeds
    j.close()


def test_join_clean_restore_oracle():
    _join_clean_restore_flow(oracle(), JOIN_INNER)


def test_join_clean_restore_oracle_semi():
    _join_clean_restore_flow(oracle(), JOIN_LEFT_SEMI)


@pytest.mark.gpu
def test_join_clean_restore_gpu_and_parity():
    import risingwave_amd

    risingwave_amd.load_library()
    glib = ffi.Lib(risingwave_amd.lib_path())
    for jt in (JOIN_INNER, JOIN_LEFT_SEMI):
        sg = _join_clean_restore_flow(glib, jt)
        so = _join_clean_restore_flow(oracle(), jt)
        assert sg == so, f"join type {jt}: clean drain streams diverged"


def _agg_clean_restore_flow(lib):
    """Agg watermark cleaning: intermediate DELETE frames, minput row
    deltas, and dedup count resets must all reach the spill streams."""
    from rwtest.ffi import (AGG_COUNT, AGG_MIN, agg_dedup_drain_bytes,
                            agg_dedup_restore, agg_minput_drain_bytes,
                            agg_minput_restore)

    # group col 0 is watermarked; MIN is materialized-input; COUNT
    # DISTINCT on col 2 exercises the dedup table
    calls = [(AGG_SUM, 1, T_I64), (AGG_MIN, 1, T_I64),
             (AGG_COUNT, 2, T_I64, 1), (AGG_COUNT_STAR, -1, T_I64)]
    mk = lambda: ffi.HashAgg(lib, [T_I64, T_I64, T_I64, T_I64], [0], calls,
                             3, stream_key=(3,))
    rng = np.random.default_rng(909)
    rid = [0]

    def epoch_chunks(e):
        out = []
        for _ in range(2):
            n = 200
            g = rng.integers(e * 10, e * 10 + 25, n)
            v = rng.integers(-100, 100, n)
            d = rng.integers(0, 6, n)
            r2 = np.arange(rid[0], rid[0] + n)
            rid[0] += n
            out.append(mk_chunk([T_I64] * 4, np.zeros(n, np.uint8),
                                [g, v, d, r2]))
        return out

    a = mk()
    inter, minp, ded = b"", b"", b""
    streams = []
    for e in range(3):
        _drive_agg(a, epoch_chunks(e), e + 1)
        a.watermark(0, e * 10)
        d1 = agg_checkpoint_drain_bytes(lib, a.h)
        d2 = agg_minput_drain_bytes(lib, a.h, 0)
        d3 = agg_dedup_drain_bytes(lib, a.h, 0)
        inter += d1
        minp += d2
        ded += d3
        streams += [d1, d2, d3]
    b = mk()
    agg_minput_restore(lib, b.h, 0, minp)
    agg_dedup_restore(lib, b.h, 0, ded)
    agg_restore(lib, b.h, inter)
    for e in range(3, 6):
        cs = epoch_chunks(e)
        oa = _drive_agg(a, cs, e + 1)
        ob = _drive_agg(b, cs, e + 1)
        assert oa == ob, f"epoch {e}: restored agg diverged after clean"
        a.watermark(0, e * 10)
        b.watermark(0, e * 10)
        for fn in (agg_checkpoint_drain_bytes,
                   lambda l, h: agg_minput_drain_bytes(l, h, 0),
                   lambda l, h: agg_dedup_drain_bytes(l, h, 0)):
            da = fn(lib, a.h)
            db = fn(lib, b.h)
            assert da == db, f"epoch {e}: drain diverged after clean"
            streams.append(da)
    a.close()
    b.close()
    return streams


def _topn_clean_restore_flow(lib):
    """GroupTopN watermark cleaning (group_top_n.rs:266-273): rows of
    groups below the watermark net to DELETE spill frames; restore +
    continue matches an uninterrupted run."""
    from rwtest.ffi import topn_checkpoint_drain, topn_restore

    mk = lambda: ffi.GroupTopN(lib, [T_I64, T_I64, T_I64], [0],
                               [(1, False)], [(2, False)], limit=3)
    rng = np.random.default_rng(515)
    rid = [0]

    def epoch_chunks(e):
        n = 150
        g = rng.integers(e * 8, e * 8 + 20, n)
        v = rng.integers(0, 100, n)
        r2 = np.arange(rid[0], rid[0] + n)
        rid[0] += n
        return [mk_chunk([T_I64] * 3, np.zeros(n, np.uint8), [g, v, r2])]

    a = mk()
    state = b""
    streams = []
    wm = 0
    for e in range(3):
        for c in epoch_chunks(e):
            a.push(c)
        a.poll_all()
        wm = e * 8
        assert a.watermark(0, wm) == 1  # group col -> forwarded
        assert a.watermark(1, 10**9) == 0  # non-group col -> absorbed
        d = topn_checkpoint_drain(lib, a.h)
        state += d
        streams.append(d)
    b = mk()
    topn_restore(lib, b.h, state)
    b.watermark(0, wm)
    for e in range(3, 6):
        cs = epoch_chunks(e)
        for c in cs:
            a.push(c)
            b.push(c)
            ma = rows_multiset(a.poll_all())
            mb = rows_multiset(b.poll_all())
            assert ma == mb, f"epoch {e}: restored topn diverged after clean"
        wm = e * 8
        a.watermark(0, wm)
        b.watermark(0, wm)
        da = topn_checkpoint_drain(lib, a.h)
        db = topn_checkpoint_drain(lib, b.h)
        assert da == db, f"epoch {e}: topn drain diverged after clean"
        streams.append(da)
    a.close()
    b.close()
    return streams


def _agg_eowc_restore_flow(lib):
    """EOWC executors restore from their drain stream (mid-window PUTs +
    close DELETEs are the full state-table view, hash_agg.rs:429-474)."""
    calls = [(AGG_COUNT_STAR, -1, T_I64), (AGG_SUM, 1, T_I64)]
    mk = lambda: ffi.HashAgg(lib, [T_I64, T_I64], [0], calls, 0,
                             emit_on_window_close=True)
    rng = np.random.default_rng(42)
    a = mk()
    d = b""
    wm = 0
    outs = []
    for ep in range(3):
        n = 200
        k = rng.integers(wm, wm + 40, n)
        v = rng.integers(1, 9, n)
        a.push(mk_chunk([T_I64] * 2, np.zeros(n, np.uint8), [k, v]))
        wm += 20
        a.watermark(0, wm)
        a.flush(ep + 1)
        a.poll_all()
        d += agg_checkpoint_drain_bytes(lib, a.h)
    b = mk()
    agg_restore(lib, b.h, d)
    for ep in range(3, 6):
        n = 200
        k = rng.integers(wm, wm + 40, n)
        v = rng.integers(1, 9, n)
        c = mk_chunk([T_I64] * 2, np.zeros(n, np.uint8), [k, v])
        wm += 20
        for x in (a, b):
            x.push(c)
            x.watermark(0, wm)
            x.flush(ep + 1)
        ma = rows_multiset(a.poll_all())
        mb = rows_multiset(b.poll_all())
        assert ma == mb, f"epoch {ep}: restored EOWC agg diverged"
        da = agg_checkpoint_drain_bytes(lib, a.h)
        db = agg_checkpoint_drain_bytes(lib, b.h)
        assert da == db, f"epoch {ep}: EOWC drain diverged"
        outs.append((ma, da))
    a.close()
    b.close()
    return outs


def test_agg_eowc_restore_oracle():
    _agg_eowc_restore_flow(oracle())


@pytest.mark.gpu
def test_agg_eowc_restore_gpu_and_parity():
    import risingwave_amd

    risingwave_amd.load_library()
    glib = ffi.Lib(risingwave_amd.lib_path())
    og = _agg_eowc_restore_flow(glib)
    oo = _agg_eowc_restore_flow(oracle())
    assert og == oo, "EOWC restore flow diverged from oracle"


def test_topn_clean_restore_oracle():
    _topn_clean_restore_flow(oracle())


@pytest.mark.gpu
def test_topn_clean_restore_gpu_and_parity():
    import risingwave_amd

    risingwave_amd.load_library()
    glib = ffi.Lib(risingwave_amd.lib_path())
    sg = _topn_clean_restore_flow(glib)
    so = _topn_clean_restore_flow(oracle())
    assert sg == so, "topn clean drain streams diverged from oracle"


def test_agg_clean_restore_oracle():
    _agg_clean_restore_flow(oracle())


@pytest.mark.gpu
def test_agg_clean_restore_gpu_and_parity():
    import risingwave_amd

    risingwave_amd.load_library()
    glib = ffi.Lib(risingwave_amd.lib_path())
    sg = _agg_clean_restore_flow(glib)
    so = _agg_clean_restore_flow(oracle())
    assert sg == so, "agg clean drain streams diverged from oracle"


def test_agg_minput_drain_parity_cpu_noop():
    # sanity: a fresh executor drains empty minput tables
    from rwtest.ffi import agg_minput_drain_bytes

    a = ffi.HashAgg(oracle(), [T_I64, T_I64, T_I64], [0],
                    [(ffi.AGG_MIN, 1, T_I64), (AGG_COUNT_STAR, -1, T_I64)],
                    1, stream_key=(2,))
    import ctypes

    a.lib.lib.rw_agg_n_minput_tables.restype = ctypes.c_int
    a.lib.lib.rw_agg_n_minput_tables.argtypes = [ctypes.c_void_p]
    assert a.lib.lib.rw_agg_n_minput_tables(a.h) == 1
    assert agg_minput_drain_bytes(oracle(), a.h, 0) == b""
    a.close()


def test_agg_float_state_restore_oracle():
    # float SUM/MIN/MAX states round-trip the drain encoding (the raw-word
    # datum bugs collapsed floats to 0.0 in drains and restores)
    from rwtest.ffi import AGG_MAX, AGG_MIN, T_F64

    calls = [(AGG_SUM, 1, T_F64), (AGG_COUNT_STAR, -1, T_I64)]
    lib = oracle()
    mk = lambda: ffi.HashAgg(lib, [T_I64, T_F64], [0], calls, 1)
    a = mk()
    rng = np.random.default_rng(11)
    d = b""
    for e in range(2):
        n = 100
        g = rng.integers(0, 8, n)
        v = rng.uniform(-5, 5, n)
        a.push(ffi.Chunk([T_I64, T_F64], np.zeros(n, np.uint8),
                         [g, np.array(v, np.float64)],
                         [np.ones(n, np.uint8)] * 2))
        a.flush(e + 1)
        a.poll_all()
        d += agg_checkpoint_drain_bytes(lib, a.h)
    b = mk()
    agg_restore(lib, b.h, d)
    n = 100
    g = rng.integers(0, 8, n)
    v = rng.uniform(-5, 5, n)
    c = ffi.Chunk([T_I64, T_F64], np.zeros(n, np.uint8),
                  [g, np.array(v, np.float64)], [np.ones(n, np.uint8)] * 2)
    oa = _drive_agg(a, [c], 5)
    ob = _drive_agg(b, [c], 5)
    assert oa == ob, "restored float agg diverged"
    assert agg_checkpoint_drain_bytes(lib, a.h) == \
        agg_checkpoint_drain_bytes(lib, b.h)
    a.close()
    b.close()


def test_topn_float_restore_oracle():
    # float order/rest columns round-trip the TopN drain + restore
    from rwtest.ffi import T_F64, topn_checkpoint_drain, topn_restore

    lib = oracle()
    mk = lambda: ffi.GroupTopN(lib, [T_I64, T_F64, T_I64], [0],
                               [(1, True)], [(2, False)], limit=2)
    a = mk()
    rng = np.random.default_rng(13)
    rid = [0]
    d = b""
    for e in range(2):
        n = 80
        g = rng.integers(0, 6, n)
        v = rng.uniform(-3, 3, n)
        r2 = np.arange(rid[0], rid[0] + n)
        rid[0] += n
        a.push(ffi.Chunk([T_I64, T_F64, T_I64], np.zeros(n, np.uint8),
                         [g, np.array(v, np.float64), r2],
                         [np.ones(n, np.uint8)] * 3))
        a.poll_all()
        d += topn_checkpoint_drain(lib, a.h)
    b = mk()
    topn_restore(lib, b.h, d)
    n = 80
    g = rng.integers(0, 6, n)
    v = rng.uniform(-3, 3, n)
    r2 = np.arange(rid[0], rid[0] + n)
    c = ffi.Chunk([T_I64, T_F64, T_I64], np.zeros(n, np.uint8),
                  [g, np.array(v, np.float64), r2],
                  [np.ones(n, np.uint8)] * 3)
    a.push(c)
    b.push(c)
    assert rows_multiset(a.poll_all()) == rows_multiset(b.poll_all())
    assert topn_checkpoint_drain(lib, a.h) == topn_checkpoint_drain(lib, b.h)
    a.close()
    b.close()


def test_agg_float_distinct_and_minput_restore_oracle():
    # float DISTINCT keys and float materialized-input values round-trip
    # their dedup/minput drains (pins the raw-word float paths)
    from rwtest.ffi import (AGG_COUNT, AGG_MIN, T_F64, agg_dedup_drain_bytes,
                            agg_dedup_restore, agg_minput_drain_bytes,
                            agg_minput_restore)

    calls = [(AGG_COUNT, 1, T_I64, 1), (AGG_MIN, 1, T_F64),
             (AGG_COUNT_STAR, -1, T_I64)]
    lib = oracle()
    mk = lambda: ffi.HashAgg(lib, [T_I64, T_F64, T_I64], [0], calls, 2,
                             stream_key=(2,))
    a = mk()
    rng = np.random.default_rng(21)
    rid = [0]

    def chunk():
        n = 120
        g = rng.integers(0, 6, n)
        v = np.round(rng.uniform(-2, 2, n), 1)  # repeats -> real dedup
        r2 = np.arange(rid[0], rid[0] + n)
        rid[0] += n
        return ffi.Chunk([T_I64, T_F64, T_I64], np.zeros(n, np.uint8),
                         [g, np.array(v, np.float64), r2],
                         [np.ones(n, np.uint8)] * 3)

    inter, ded, minp = b"", b"", b""
    for e in range(2):
        _drive_agg(a, [chunk()], e + 1)
        inter += agg_checkpoint_drain_bytes(lib, a.h)
        ded += agg_dedup_drain_bytes(lib, a.h, 0)
        minp += agg_minput_drain_bytes(lib, a.h, 0)
    b = mk()
    agg_minput_restore(lib, b.h, 0, minp)
    agg_dedup_restore(lib, b.h, 0, ded)
    agg_restore(lib, b.h, inter)
    c = chunk()
    oa = _drive_agg(a, [c], 5)
    ob = _drive_agg(b, [c], 5)
    assert oa == ob, "restored float distinct/minput agg diverged"
    for fn in (agg_checkpoint_drain_bytes,
               lambda l, h: agg_dedup_drain_bytes(l, h, 0),
               lambda l, h: agg_minput_drain_bytes(l, h, 0)):
        assert fn(lib, a.h) == fn(lib, b.h)
    a.close()
    b.close()
