"""GPU parity tests: the HIP executors vs the CPU oracle on identical seeded
inputs. The oracle is pinned by the reference's golden vectors
in test_oracle_*.py; these tests pin the GPU path to the oracle.

All tests are @pytest.mark.gpu — run on a real MI355X via gpurun.
"""
import numpy as np
import pytest

from rwtest import ffi
from rwtest.ffi import (
    AGG_COUNT, AGG_COUNT_STAR, AGG_MAX, AGG_MIN, AGG_SUM, CMP_LT, JOIN_INNER,
    SIDE_LEFT, SIDE_RIGHT, T_I64, T_TS, from_pretty, oracle, rows_multiset,
)

pytestmark = pytest.mark.gpu


def gpu():
    import risingwave_amd

    return ffi.Lib(risingwave_amd.lib_path())


def mk_chunk(types, ops, cols, valids=None, vis=None):
    n = len(ops)
    if valids is None:
        valids = [np.ones(n, np.uint8) for _ in types]
    return ffi.Chunk(types, ops, cols, valids, vis)


def rand_insert_chunk(rng, n, key_space, key_scale=1):
    keys = rng.integers(0, key_space, n) * key_scale
    vals = rng.integers(1, 10**7, n)
    return mk_chunk([T_I64, T_I64], np.zeros(n, np.uint8), [keys, vals])


# ---------------- HashAgg ----------------


def agg_pair(calls, row_count_index, append_only, input_types=(T_I64, T_I64),
             group_key=(0,), cap=0):
    g = ffi.HashAgg(gpu(), list(input_types), list(group_key), calls,
                    row_count_index, append_only=append_only,
                    state_capacity_hint=cap)
    o = ffi.HashAgg(oracle(), list(input_types), list(group_key), calls,
                    row_count_index, append_only=append_only)
    return g, o


def run_and_compare(g, o, epochs, push=lambda a, c: a.push(c)):
    for e, chunks in enumerate(epochs):
        for c in chunks:
            push(g, c)
            push(o, c)
        g.flush(e + 1)
        o.flush(e + 1)
        mg = rows_multiset(g.poll_all())
        mo = rows_multiset(o.poll_all())
        assert mg == mo, (
            f"epoch {e + 1}: GPU {len(mg)} rows vs oracle {len(mo)};"
            f" first diff: {next(((a, b) for a, b in zip(mg, mo) if a != b), None)}"
        )
    g.close()
    o.close()


def test_agg_q7_max_count_append_only():
    # q7 shape: group by window (ts), max(price) + count (nexmark.yaml q7:
    # StreamHashAgg [append_only] group_key [$expr1] aggs [max, count])
    rng = np.random.default_rng(1)
    calls = [(AGG_MAX, 1, T_I64), (AGG_COUNT_STAR, -1, T_I64)]
    g, o = agg_pair(calls, 1, append_only=True)
    epochs = [[rand_insert_chunk(rng, 4096, 64, 10_000_000) for _ in range(8)]
              for _ in range(3)]
    run_and_compare(g, o, epochs)


def test_agg_count_sum_retract():
    # retractable count/sum with a delete mix, multiple epochs
    rng = np.random.default_rng(2)
    calls = [(AGG_COUNT_STAR, -1, T_I64), (AGG_SUM, 1, T_I64)]
    g, o = agg_pair(calls, 0, append_only=False)

    inserted = []
    epochs = []
    for _ in range(4):
        chunks = []
        for _ in range(4):
            n = 1024
            keys = rng.integers(0, 300, n)
            vals = rng.integers(1, 1000, n)
            ops = np.zeros(n, np.uint8)
            # delete ~20% previously-inserted rows
            for i in range(n):
                if inserted and rng.random() < 0.2:
                    j = rng.integers(0, len(inserted))
                    keys[i], vals[i] = inserted.pop(int(j))
                    ops[i] = ffi.OP_DELETE
                else:
                    inserted.append((int(keys[i]), int(vals[i])))
            chunks.append(mk_chunk([T_I64, T_I64], ops, [keys, vals]))
        epochs.append(chunks)
    run_and_compare(g, o, epochs)


def test_agg_nulls_and_visibility():
    rng = np.random.default_rng(3)
    calls = [(AGG_COUNT_STAR, -1, T_I64), (AGG_COUNT, 1, T_I64), (AGG_SUM, 1, T_I64)]
    g, o = agg_pair(calls, 0, append_only=False)
    n = 2048
    keys = rng.integers(0, 50, n)
    vals = rng.integers(1, 100, n)
    valid = (rng.random(n) > 0.3).astype(np.uint8)
    vis = (rng.random(n) > 0.1).astype(np.uint8)
    c = ffi.Chunk([T_I64, T_I64], np.zeros(n, np.uint8), [keys, vals],
                  [np.ones(n, np.uint8), valid], vis)
    run_and_compare(g, o, [[c]])


def test_agg_multiword_key():
    # q8-agg shape: 3-part group key (seller i64, ws ts, we ts), count
    rng = np.random.default_rng(4)
    calls = [(AGG_COUNT_STAR, -1, T_I64)]
    n = 4096
    seller = rng.integers(0, 1000, n)
    ws = rng.integers(0, 16, n) * 10_000_000
    we = ws + 10_000_000
    c = mk_chunk([T_I64, T_TS, T_TS], np.zeros(n, np.uint8), [seller, ws, we])
    g = ffi.HashAgg(gpu(), [T_I64, T_TS, T_TS], [0, 1, 2], calls, 0)
    o = ffi.HashAgg(oracle(), [T_I64, T_TS, T_TS], [0, 1, 2], calls, 0)
    run_and_compare(g, o, [[c]])


def test_agg_group_reappears():
    # delete-to-zero then reinsert across epochs (reset-at-zero semantics)
    calls = [(AGG_COUNT_STAR, -1, T_I64), (AGG_SUM, 1, T_I64)]
    g, o = agg_pair(calls, 0, append_only=False)
    e1 = from_pretty(" I I\n + 7 10\n + 8 5")
    e2 = from_pretty(" I I\n - 7 10")
    e3 = from_pretty(" I I\n + 7 99\n - 8 5")
    run_and_compare(g, o, [[e1], [e2], [e3]])


def test_agg_min_append_only_golden():
    # the reference golden fixture (hash_agg.rs:177-256) on the GPU path
    calls = [(AGG_COUNT_STAR, -1, T_I64), (AGG_MIN, 1, T_I64)]
    g = ffi.HashAgg(gpu(), [T_I64, T_I64, T_I64], [0], calls, 0,
                    stream_key=[2], append_only=True)
    o = ffi.HashAgg(oracle(), [T_I64, T_I64, T_I64], [0], calls, 0,
                    stream_key=[2], append_only=True)
    e1 = from_pretty(" I I I\n + 2 5 1000\n + 1 15 1001\n + 1 8 1002\n + 2 5 1003\n + 2 10 1004")
    e2 = from_pretty(" I I I\n + 1 20 1005\n + 1 1 1006\n + 2 10 1007\n + 2 20 1008")
    run_and_compare(g, o, [[e1], [e2]])


def test_agg_dense_tail_and_runs():
    # dense-kernel edge cases: chunk sizes that leave partial tail tiles
    # (not multiples of the 4-rows/lane x 64-lane wave tile), giant
    # single-key runs spanning many waves (q7's monotone windows), and the
    # count-star-as-run-length path (rw_amd.hip agg_apply_dense4_body CS).
    rng = np.random.default_rng(7)
    calls = [(AGG_MAX, 1, T_I64), (AGG_COUNT_STAR, -1, T_I64)]
    for n, n_keys in [(1031, 1), (1537, 2), (4099, 3), (2047, 500)]:
        g, o = agg_pair(calls, 1, append_only=True)
        # sorted keys => giant runs; the tail tile ends mid-run
        keys = np.sort(rng.integers(0, n_keys, n))
        vals = rng.integers(1, 10**7, n)
        c = mk_chunk([T_I64, T_I64], np.zeros(n, np.uint8), [keys, vals])
        run_and_compare(g, o, [[c]])


# ---------------- HashJoin ----------------


def join_pair(**kw):
    g = ffi.HashJoin(gpu(), JOIN_INNER, [T_I64, T_I64], [T_I64, T_I64],
                     key_l=[0], key_r=[0], pk_l=[1], pk_r=[1], **kw)
    o = ffi.HashJoin(oracle(), JOIN_INNER, [T_I64, T_I64], [T_I64, T_I64],
                     key_l=[0], key_r=[0], pk_l=[1], pk_r=[1], **kw)
    return g, o


def run_join_and_compare(g, o, pushes):
    """pushes: list of (side, chunk); compare output multiset per push."""
    for i, (side, c) in enumerate(pushes):
        g.push(side, c)
        o.push(side, c)
        mg = rows_multiset(g.poll_all())
        mo = rows_multiset(o.poll_all())
        assert mg == mo, f"push {i}: GPU {len(mg)} rows vs oracle {len(mo)}"
    g.close()
    o.close()


def test_join_inner_golden():
    g, o = join_pair()
    pushes = [
        (SIDE_LEFT, from_pretty(" I I\n + 1 4\n + 2 5\n + 3 6")),
        (SIDE_LEFT, from_pretty(" I I\n + 3 8\n - 3 8")),
        (SIDE_RIGHT, from_pretty(" I I\n + 2 7\n + 4 8\n + 6 9")),
        (SIDE_RIGHT, from_pretty(" I I\n + 3 10\n + 6 11")),
    ]
    run_join_and_compare(g, o, pushes)


def test_join_inner_random():
    rng = np.random.default_rng(5)
    g, o = join_pair()
    pushes = []
    pk = 0
    for i in range(12):
        n = 2048
        keys = rng.integers(0, 500, n)
        vals = np.arange(pk, pk + n)
        pk += n
        pushes.append((int(rng.integers(0, 2)),
                       mk_chunk([T_I64, T_I64], np.zeros(n, np.uint8), [keys, vals])))
    run_join_and_compare(g, o, pushes)


def test_join_inner_deletes():
    rng = np.random.default_rng(6)
    g, o = join_pair()
    # insert rows with unique pk per side, then delete a sample of them
    side_rows = {0: [], 1: []}
    pushes = []
    pk = 0
    for step in range(10):
        s = int(rng.integers(0, 2))
        n = 512
        keys = rng.integers(0, 40, n)
        vals = np.arange(pk, pk + n)
        pk += n
        ops = np.zeros(n, np.uint8)
        # deletes only target rows from EARLIER chunks: within one chunk the
        # GPU kernel is parallel, so it keeps the reference's sequential
        # semantics only for inter-chunk dependencies
        deletable = list(side_rows[s])
        new_rows = []
        for i in range(n):
            if deletable and rng.random() < 0.3:
                j = int(rng.integers(0, len(deletable)))
                keys[i], vals[i] = deletable.pop(j)
                side_rows[s].remove((int(keys[i]), int(vals[i])))
                ops[i] = ffi.OP_DELETE
            else:
                new_rows.append((int(keys[i]), int(vals[i])))
        side_rows[s].extend(new_rows)
        pushes.append((s, mk_chunk([T_I64, T_I64], ops, [keys, vals])))
    run_join_and_compare(g, o, pushes)


def test_join_inner_nonequi():
    g, o = join_pair(cond=(CMP_LT, 1, 3))
    pushes = [
        (SIDE_LEFT, from_pretty(" I I\n + 1 4\n + 2 10\n + 3 6")),
        (SIDE_LEFT, from_pretty(" I I\n + 3 8\n - 3 8")),
        (SIDE_RIGHT, from_pretty(" I I\n + 2 7\n + 4 8\n + 6 9")),
        (SIDE_RIGHT, from_pretty(" I I\n + 3 10\n + 6 11")),
    ]
    run_join_and_compare(g, o, pushes)


def test_join_null_keys_never_match():
    g, o = join_pair()
    n = 64
    keys = np.arange(n)
    vals = np.arange(n)
    valid = np.ones(n, np.uint8)
    valid[::4] = 0
    c = ffi.Chunk([T_I64, T_I64], np.zeros(n, np.uint8), [keys, vals],
                  [valid, np.ones(n, np.uint8)])
    run_join_and_compare(g, o, [(SIDE_LEFT, c), (SIDE_RIGHT, c)])


def test_join_q8_shape_multiword_key():
    # q8 join shape: key = (id i64, ws ts, we ts) plus a unique row-id pk
    rng = np.random.default_rng(8)
    t4 = [T_I64, T_TS, T_TS, T_I64]
    kw = dict(key_l=[0, 1, 2], key_r=[0, 1, 2], pk_l=[3], pk_r=[3])
    g = ffi.HashJoin(gpu(), JOIN_INNER, t4, t4, **kw)
    o = ffi.HashJoin(oracle(), JOIN_INNER, t4, t4, **kw)
    pushes = []
    rowid = 0
    for i in range(6):
        n = 2048
        ids = rng.integers(0, 3000, n)
        ws = rng.integers(0, 4, n) * 10_000_000
        we = ws + 10_000_000
        rid = np.arange(rowid, rowid + n)
        rowid += n
        pushes.append(
            (i % 2, mk_chunk(t4, np.zeros(n, np.uint8), [ids, ws, we, rid]))
        )
    run_join_and_compare(g, o, pushes)


def test_vnode_gpu_matches_oracle():
    # GPU crc32 vnode kernel vs the oracle (itself pinned against zlib in
    # test_vnode_dispatch.py)
    from test_vnode_dispatch import compute_vnodes

    rng = np.random.default_rng(21)
    n = 4096
    keys = rng.integers(-(2**62), 2**62, n)
    ws = rng.integers(0, 1 << 40, n)
    valid = (rng.random(n) > 0.1).astype(np.uint8)
    c = ffi.Chunk([T_I64, T_TS], np.zeros(n, np.uint8), [keys, ws],
                  [valid, np.ones(n, np.uint8)])
    got_gpu = compute_vnodes(gpu(), c, [0, 1])
    got_orc = compute_vnodes(ffi.oracle(), c, [0, 1])
    assert got_gpu == got_orc


def test_agg_min_retractable_golden():
    # the reference golden fixture (hash_agg.rs:100-174): retractable min
    # via materialized-input state — now on the GPU chain-store path
    calls = [(AGG_COUNT_STAR, -1, T_I64), (AGG_MIN, 1, T_I64)]
    g = ffi.HashAgg(gpu(), [T_I64, T_I64, T_I64], [0], calls, 0, stream_key=[2])
    o = ffi.HashAgg(oracle(), [T_I64, T_I64, T_I64], [0], calls, 0, stream_key=[2])
    e1 = from_pretty(" I I I\n + 1 233 1001\n + 1 23333 1002\n + 2 2333 1003")
    e2 = from_pretty(" I I I\n - 1 233 1001\n - 1 23333 1002 D\n - 2 2333 1003")
    run_and_compare(g, o, [[e1], [e2]])


def test_agg_minmax_retractable_random():
    rng = np.random.default_rng(31)
    calls = [(AGG_COUNT_STAR, -1, T_I64), (AGG_MIN, 1, T_I64), (AGG_MAX, 1, T_I64)]
    g = ffi.HashAgg(gpu(), [T_I64, T_I64, T_I64], [0], calls, 0, stream_key=[2])
    o = ffi.HashAgg(oracle(), [T_I64, T_I64, T_I64], [0], calls, 0, stream_key=[2])
    live = []
    sk = 0
    epochs = []
    for _ in range(4):
        chunks = []
        for _ in range(3):
            n = 512
            keys = rng.integers(0, 40, n)
            vals = rng.integers(0, 10000, n)
            sks = np.zeros(n, np.int64)
            ops = np.zeros(n, np.uint8)
            # deletes target rows from EARLIER chunks only (DESIGN §3.1 —
            # same-chunk conflicts exercised separately below)
            deletable = list(live)
            new_rows = []
            for i in range(n):
                if deletable and rng.random() < 0.25:
                    jx = int(rng.integers(0, len(deletable)))
                    keys[i], vals[i], sks[i] = deletable.pop(jx)
                    live.remove((int(keys[i]), int(vals[i]), int(sks[i])))
                    ops[i] = ffi.OP_DELETE
                else:
                    sks[i] = sk
                    sk += 1
                    new_rows.append((int(keys[i]), int(vals[i]), int(sks[i])))
            live.extend(new_rows)
            chunks.append(mk_chunk([T_I64, T_I64, T_I64], ops, [keys, vals, sks]))
        epochs.append(chunks)
    run_and_compare(g, o, epochs)


def test_agg_min_retractable_same_chunk_conflict():
    # insert-then-delete of the same row within ONE chunk (the segmented
    # launch path)
    calls = [(AGG_COUNT_STAR, -1, T_I64), (AGG_MIN, 1, T_I64)]
    g = ffi.HashAgg(gpu(), [T_I64, T_I64, T_I64], [0], calls, 0, stream_key=[2])
    o = ffi.HashAgg(oracle(), [T_I64, T_I64, T_I64], [0], calls, 0, stream_key=[2])
    e1 = from_pretty(
        " I I I\n + 1 10 1\n + 1 5 2\n - 1 5 2\n + 2 7 3\n - 1 10 1\n + 1 20 4")
    run_and_compare(g, o, [[e1]])


def test_pipeline_join_agg_q3_shape():
    # the q3 fragment edge: inner join feeding sum/count agg — the GPU runs
    # the hop device-side (bench path: rw_join_bench_apply keeps outputs in
    # HBM, rw_agg_apply_joinout consumes them); the oracle pushes the join's
    # output chunks into its agg. Flush outputs must agree per epoch.
    import ctypes

    from rwtest.ffi import AGG_COUNT_STAR, AGG_SUM

    rng = np.random.default_rng(41)
    tl = [T_I64, T_I64, T_I64]
    tr = [T_I64, T_I64, T_I64, T_I64]

    gl = gpu()
    L = gl.lib
    L.rw_join_bench_preload.restype = ctypes.c_void_p
    L.rw_join_bench_preload.argtypes = [ctypes.c_void_p, ctypes.c_int,
                                        ctypes.POINTER(ffi.RwChunkC)]
    L.rw_join_bench_apply.restype = ctypes.c_int
    L.rw_join_bench_apply.argtypes = [ctypes.c_void_p, ctypes.c_int,
                                      ctypes.c_void_p]
    L.rw_join_bench_drain.restype = ctypes.c_longlong
    L.rw_join_bench_drain.argtypes = [ctypes.c_void_p]
    L.rw_agg_apply_joinout.restype = ctypes.c_int
    L.rw_agg_apply_joinout.argtypes = [ctypes.c_void_p, ctypes.c_void_p]

    gj = ffi.HashJoin(gl, JOIN_INNER, tl, tr, key_l=[0], key_r=[0],
                      pk_l=[2], pk_r=[3])
    ga = ffi.HashAgg(gl, [T_I64] * 7, [3, 4, 5],
                     [(AGG_SUM, 1, T_I64), (AGG_COUNT_STAR, -1, T_I64)],
                     row_count_index=1)
    oj = ffi.HashJoin(ffi.oracle(), JOIN_INNER, tl, tr, key_l=[0], key_r=[0],
                      pk_l=[2], pk_r=[3])
    oa = ffi.HashAgg(ffi.oracle(), [T_I64] * 7, [3, 4, 5],
                     [(AGG_SUM, 1, T_I64), (AGG_COUNT_STAR, -1, T_I64)],
                     row_count_index=1)

    def gpu_apply(side, chunk):
        cc = chunk.to_c()
        h = L.rw_join_bench_preload(gj.h, side, ctypes.byref(cc))
        assert h, gl.last_error()
        assert L.rw_join_bench_apply(gj.h, side, h) == 0, gl.last_error()

    # orders build side
    n_orders = 2000
    ok = np.arange(n_orders, dtype=np.int64)
    orders = mk_chunk(tr, np.zeros(n_orders, np.uint8), [ok, ok % 31, ok % 3, ok])
    gpu_apply(SIDE_RIGHT, orders)
    assert L.rw_join_bench_drain(gj.h) == 0, gl.last_error()
    oj.push(SIDE_RIGHT, orders)
    assert oj.poll_all() == []

    rowid = 0
    prev = None
    for epoch in range(3):
        for _ in range(2):
            n = 1024
            okey = rng.integers(0, n_orders, n)
            rev = rng.integers(1, 1000, n)
            rid = np.arange(rowid, rowid + n)
            rowid += n
            ops = np.zeros(n, np.uint8)
            if prev is not None:
                nd = n // 8
                sel = rng.choice(len(prev[0]), nd, replace=False)
                okey[:nd], rev[:nd], rid[:nd] = (prev[0][sel], prev[1][sel],
                                                 prev[2][sel])
                ops[:nd] = ffi.OP_DELETE
            prev = (okey[n // 8:].copy(), rev[n // 8:].copy(),
                    rid[n // 8:].copy())
            li = mk_chunk(tl, ops, [okey, rev, rid])
            gpu_apply(SIDE_LEFT, li)
            assert L.rw_agg_apply_joinout(ga.h, gj.h) == 0, gl.last_error()
            oj.push(SIDE_LEFT, li)
            for oc in oj.poll_all():
                oa.push(oc)
        ga.flush(epoch + 1)
        oa.flush(epoch + 1)
        mg = rows_multiset(ga.poll_all())
        mo = rows_multiset(oa.poll_all())
        assert mg == mo, (f"pipeline epoch {epoch + 1}: GPU {len(mg)} rows "
                          f"vs oracle {len(mo)}")
    gj.close()
    ga.close()
    oj.close()
    oa.close()


def test_join_watermark_gpu():
    # emission order/values (hash_join.rs:3667-3737) + TTL cleaning effect
    g = ffi.HashJoin(gpu(), JOIN_INNER, [T_I64, T_I64], [T_I64, T_I64],
                     key_l=[0], key_r=[0], pk_l=[1], pk_r=[1], wm_jk=[(0, True)])
    assert g.watermark(SIDE_LEFT, 0, 100) == []
    assert g.watermark(SIDE_LEFT, 0, 200) == []
    assert g.watermark(SIDE_RIGHT, 0, 50) == [(2, 50), (0, 50)]
    assert g.watermark(SIDE_RIGHT, 0, 100) == [(2, 100), (0, 100)]
    g.close()

    g = ffi.HashJoin(gpu(), JOIN_INNER, [T_I64, T_I64], [T_I64, T_I64],
                     key_l=[0], key_r=[0], pk_l=[1], pk_r=[1], wm_jk=[(0, True)])
    o = ffi.HashJoin(oracle(), JOIN_INNER, [T_I64, T_I64], [T_I64, T_I64],
                     key_l=[0], key_r=[0], pk_l=[1], pk_r=[1], wm_jk=[(0, True)])
    c = from_pretty(" I I\n + 2 1\n + 6 2")
    for j in (g, o):
        j.push(SIDE_LEFT, c)
        j.poll_all()
        j.watermark(SIDE_LEFT, 0, 5)
        j.watermark(SIDE_RIGHT, 0, 5)
    probe = from_pretty(" I I\n + 2 10\n + 6 11")
    g.push(SIDE_RIGHT, probe)
    o.push(SIDE_RIGHT, probe)
    assert rows_multiset(g.poll_all()) == rows_multiset(o.poll_all())
    g.close()
    o.close()


def test_agg_watermark_gpu():
    calls = [(AGG_COUNT_STAR, -1, T_I64), (AGG_SUM, 1, T_I64)]
    g = ffi.HashAgg(gpu(), [T_I64, T_I64], [0], calls, 0)
    o = ffi.HashAgg(oracle(), [T_I64, T_I64], [0], calls, 0)
    c1 = from_pretty(" I I\n + 1 5\n + 9 7")
    c2 = from_pretty(" I I\n + 1 3\n + 9 1")
    for a in (g, o):
        a.push(c1)
        a.flush(1)
    assert rows_multiset(g.poll_all()) == rows_multiset(o.poll_all())
    for a in (g, o):
        a.watermark(0, 5)
        a.push(c2)
        a.flush(2)
    assert rows_multiset(g.poll_all()) == rows_multiset(o.poll_all())
    g.close()
    o.close()


def test_checkpoint_spill_parity():
    # §8f-2: GPU and oracle drain byte-identical state-table KV deltas
    # (sorted by memcomparable key — the KV store's view) for the same epochs
    from test_codec import drain

    rng = np.random.default_rng(51)
    calls = [(AGG_COUNT_STAR, -1, T_I64), (AGG_SUM, 1, T_I64)]
    g = ffi.HashAgg(gpu(), [T_I64, T_I64], [0], calls, 0)
    o = ffi.HashAgg(ffi.oracle(), [T_I64, T_I64], [0], calls, 0)
    inserted = []
    for epoch in range(3):
        n = 1024
        keys = rng.integers(0, 200, n)
        vals = rng.integers(1, 1000, n)
        ops = np.zeros(n, np.uint8)
        for i in range(n):
            if inserted and rng.random() < 0.3:
                jx = int(rng.integers(0, len(inserted)))
                keys[i], vals[i] = inserted.pop(jx)
                ops[i] = ffi.OP_DELETE
            else:
                inserted.append((int(keys[i]), int(vals[i])))
        c = mk_chunk([T_I64, T_I64], ops, [keys, vals])
        for a in (g, o):
            a.push(c)
            a.flush(epoch + 1)
            a.poll_all()
        rg = sorted(drain(gpu(), g))
        ro = sorted(drain(ffi.oracle(), o))
        assert rg == ro, f"epoch {epoch+1}: {len(rg)} vs {len(ro)} records"
    g.close()
    o.close()


def test_exchange_partition_parity():
    # vnode partition (slice-by-8 CRC count/scan/scatter kernels) + self-loop
    # exchange + payload apply vs the oracle on the same chunk. R=1 uses the
    # device-copy bypass; the RCCL collective itself is exercised by
    # tests/debug_exchange.py under RW_EXCHANGE_FORCE_NCCL=1.
    import ctypes

    import bench
    import risingwave_amd

    L = gpu().lib
    L.rw_agg_bench_preload.restype = ctypes.c_void_p
    L.rw_agg_bench_preload.argtypes = [ctypes.c_void_p,
                                       ctypes.POINTER(ffi.RwChunkC)]
    L.rw_agg_apply_payload.restype = ctypes.c_int
    L.rw_agg_apply_payload.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                       ctypes.POINTER(ctypes.c_uint64),
                                       ctypes.c_int, ctypes.c_int,
                                       ctypes.c_int]
    L.rw_agg_n_batch_slots.restype = ctypes.c_int
    L.rw_agg_n_batch_slots.argtypes = [ctypes.c_void_p]

    calls = [(ffi.AGG_MAX, 1, T_I64), (ffi.AGG_COUNT_STAR, -1, T_I64)]
    agg = ffi.HashAgg(gpu(), [T_I64, T_I64], [0], calls, 1, append_only=True)
    exch = bench.setup_exchange(ffi, 0, 1, None)
    assert exch is not None, "exchange init failed"
    rng = np.random.default_rng(3)
    n = 65536
    c = bench.make_q7_chunk(ffi, rng, n, 0, 32)
    cc = c.to_c()
    batch = L.rw_agg_bench_preload(agg.h, ctypes.byref(cc))
    assert batch
    xb = exch.make_buffers(n * 32 * 4)
    nslots = L.rw_agg_n_batch_slots(agg.h)
    recv_counts = exch.run(agg.h, batch, xb, n_cols=nslots)
    assert sum(recv_counts) == n
    rc = L.rw_agg_apply_payload(agg.h, ctypes.c_void_p(xb.recv), recv_counts,
                                1, nslots, 1)
    assert rc == 0
    agg.flush(1)
    got = ffi.rows_multiset(agg.poll_all())
    o = ffi.HashAgg(ffi.oracle(), [T_I64, T_I64], [0], calls, 1,
                    append_only=True)
    o.push(c)
    o.flush(1)
    want = ffi.rows_multiset(o.poll_all())
    assert got == want
    agg.close()
    o.close()


def test_update_vnode_bitmap_parity():
    # rescale re-scope (update_vnode_bitmap): GPU vs oracle — drop half the
    # vnodes mid-stream on both executors, keep pushing rows (including for
    # dropped keys, which must restart), compare per-epoch multisets
    import struct
    import zlib

    def owned_bitmap(mod):
        bm = bytearray(32)
        for v in range(256):
            if v % 2 == mod:
                bm[v >> 3] |= 1 << (v & 7)
        return bytes(bm)

    # --- agg ---
    calls = [(AGG_COUNT_STAR, -1, T_I64), (AGG_SUM, 1, T_I64)]
    ga = ffi.HashAgg(gpu(), [T_I64, T_I64], [0], calls, 0)
    oa = ffi.HashAgg(ffi.oracle(), [T_I64, T_I64], [0], calls, 0)
    rng = np.random.default_rng(77)
    keys = rng.integers(0, 4000, 4096)
    vals = rng.integers(1, 100, 4096)
    c = mk_chunk([T_I64, T_I64], np.zeros(4096, np.uint8), [keys, vals])
    for a in (ga, oa):
        a.push(c)
        a.flush(1)
        a.poll_all()
        a.update_vnode_bitmap(owned_bitmap(0))
    keys2 = rng.integers(0, 4000, 4096)
    vals2 = rng.integers(1, 100, 4096)
    c2 = mk_chunk([T_I64, T_I64], np.zeros(4096, np.uint8), [keys2, vals2])
    outs = []
    for a in (ga, oa):
        a.push(c2)
        a.flush(2)
        outs.append(rows_multiset(a.poll_all()))
    assert outs[0] == outs[1]
    ga.close()
    oa.close()

    # --- join ---
    g, o = join_pair()
    lk = rng.integers(0, 300, 1024)
    lv = np.arange(1024)
    rk = rng.integers(0, 300, 1024)
    rv = np.arange(1024, 2048)
    cl = mk_chunk([T_I64, T_I64], np.zeros(1024, np.uint8), [lk, lv])
    cr = mk_chunk([T_I64, T_I64], np.zeros(1024, np.uint8), [rk, rv])
    for j in (g, o):
        j.push(SIDE_LEFT, cl)
        j.push(SIDE_RIGHT, cr)
        j.poll_all()
        j.update_vnode_bitmap(owned_bitmap(1))
    lk2 = rng.integers(0, 300, 1024)
    lv2 = np.arange(2048, 3072)
    cl2 = mk_chunk([T_I64, T_I64], np.zeros(1024, np.uint8), [lk2, lv2])
    outs = []
    for j in (g, o):
        j.push(SIDE_LEFT, cl2)
        outs.append(rows_multiset(j.poll_all()))
    assert outs[0] == outs[1]
    g.close()
    o.close()


def _topn_pair(offset, limit, group_by, order_by, rest, types=None):
    types = types or [T_I64, T_I64, T_I64]
    g = ffi.GroupTopN(gpu(), types, group_by, order_by, rest,
                      offset=offset, limit=limit)
    o = ffi.GroupTopN(ffi.oracle(), types, group_by, order_by, rest,
                      offset=offset, limit=limit)
    return g, o


def test_topn_golden():
    # the reference's group_top_n.rs fixtures, GPU vs oracle (the oracle is
    # pinned to the transcribed outputs in tests/test_oracle_topn.py)
    from test_oracle_topn import chunks_0_3

    for off, lim, gb, ob in [(0, 2, [1], [(2, False)]),
                             (1, 2, [1], [(2, False)]),
                             (0, 2, [1, 2], [(0, False)])]:
        rest = [(c, False) for c in [1, 2, 0]
                if c not in gb and c not in [o for o, _ in ob]]
        g, o = _topn_pair(off, lim, gb, ob, rest)
        for c in chunks_0_3():
            g.push(c)
            o.push(c)
            mg = rows_multiset(g.poll_all())
            mo = rows_multiset(o.poll_all())
            assert mg == mo, f"off={off} lim={lim}: {mg} vs {mo}"
        g.close()
        o.close()


def test_topn_random():
    # randomized insert/delete mix incl. same-ck replacements and desc order
    rng = np.random.default_rng(91)
    g, o = _topn_pair(1, 3, [0], [(1, True)], [(2, False)])
    live = []
    for i in range(10):
        n = 512
        gk = rng.integers(0, 40, n)
        ordv = rng.integers(0, 200, n)
        pk = rng.integers(0, 10**6, n)
        ops = np.zeros(n, np.uint8)
        for r in range(n):
            if live and rng.random() < 0.35:
                jx = int(rng.integers(0, len(live)))
                gk[r], ordv[r], pk[r] = live.pop(jx)
                ops[r] = ffi.OP_DELETE
            else:
                live.append((int(gk[r]), int(ordv[r]), int(pk[r])))
        c = mk_chunk([T_I64, T_I64, T_I64], ops, [gk, ordv, pk])
        g.push(c)
        o.push(c)
        mg = rows_multiset(g.poll_all())
        mo = rows_multiset(o.poll_all())
        assert mg == mo, f"push {i}: {len(mg)} vs {len(mo)} rows"
    g.close()
    o.close()


@pytest.mark.parametrize("jt", [
    ffi.JOIN_LEFT_OUTER, ffi.JOIN_RIGHT_OUTER, ffi.JOIN_FULL_OUTER,
    ffi.JOIN_LEFT_SEMI, ffi.JOIN_LEFT_ANTI, ffi.JOIN_RIGHT_SEMI,
    ffi.JOIN_RIGHT_ANTI,
])
def test_join_noninner_random(jt):
    # all non-inner types: randomized insert/delete mix on both sides,
    # GPU vs oracle multiset per push (degree transitions, NULL-side rows,
    # forward_if_not_matched, semi/anti matched-side emissions)
    rng = np.random.default_rng(1000 + jt)
    g = ffi.HashJoin(gpu(), jt, [T_I64, T_I64], [T_I64, T_I64],
                     key_l=[0], key_r=[0], pk_l=[1], pk_r=[1])
    o = ffi.HashJoin(oracle(), jt, [T_I64, T_I64], [T_I64, T_I64],
                     key_l=[0], key_r=[0], pk_l=[1], pk_r=[1])
    live = {SIDE_LEFT: [], SIDE_RIGHT: []}
    pk = 0
    for i in range(10):
        side = int(rng.integers(0, 2))
        n = 1024
        keys = rng.integers(0, 120, n)
        vals = np.arange(pk, pk + n)
        pk += n
        ops = np.zeros(n, np.uint8)
        for r in range(n):
            if live[side] and rng.random() < 0.3:
                jx = int(rng.integers(0, len(live[side])))
                keys[r], vals[r] = live[side].pop(jx)
                ops[r] = ffi.OP_DELETE
            else:
                live[side].append((int(keys[r]), int(vals[r])))
        c = mk_chunk([T_I64, T_I64], ops, [keys, vals])
        g.push(side, c)
        o.push(side, c)
        mg = net_rows(rows_multiset(g.poll_all()))
        mo = net_rows(rows_multiset(o.poll_all()))
        assert mg == mo, (f"type {jt} push {i} side {side}: "
                          f"{len(mg)} vs {len(mo)} rows")
    g.close()
    o.close()


def net_rows(rows):
    """Cancel equal +/- (and U-/U+) pairs: the reference's
    eliminate_adjacent_noop_update (stream_chunk.rs:331-384) removes such
    pairs when they happen to be ADJACENT in its sequential emission order —
    an order-dependent micro-optimization a parallel emitter cannot
    reproduce bit-exactly. Netting ALL equal pairs on both sides makes the
    comparison order-independent while still catching every semantic
    difference (a canceled pair is a no-op for any downstream consumer)."""
    from collections import Counter

    cnt = Counter(rows)
    for pos, neg in (("+", "-"), ("U+", "U-")):
        for (op, val) in list(cnt):
            if op != pos:
                continue
            k = min(cnt[(pos, val)], cnt.get((neg, val), 0))
            if k:
                cnt[(pos, val)] -= k
                cnt[(neg, val)] -= k
    out = []
    for key, c in cnt.items():
        out.extend([key] * c)
    sk = lambda r: (r[0], tuple((v is None, v if v is not None else 0)
                                for v in r[1]))
    return sorted(out, key=sk)


def test_join_left_outer_golden():
    # transcription anchor: the oracle's left-outer behavior is pinned by
    # tests/test_oracle_join.py golden vectors; here GPU vs oracle on the
    # same shaped stream incl. NULL-side transitions
    g = ffi.HashJoin(gpu(), ffi.JOIN_LEFT_OUTER, [T_I64, T_I64],
                     [T_I64, T_I64], key_l=[0], key_r=[0], pk_l=[1], pk_r=[1])
    o = ffi.HashJoin(oracle(), ffi.JOIN_LEFT_OUTER, [T_I64, T_I64],
                     [T_I64, T_I64], key_l=[0], key_r=[0], pk_l=[1], pk_r=[1])
    pushes = [
        (SIDE_LEFT, from_pretty(" I I\n + 1 4\n + 2 5\n + 3 6")),
        (SIDE_RIGHT, from_pretty(" I I\n + 2 7\n + 4 8")),
        (SIDE_RIGHT, from_pretty(" I I\n + 2 9\n - 2 7")),
        (SIDE_LEFT, from_pretty(" I I\n - 2 5")),
        (SIDE_RIGHT, from_pretty(" I I\n - 2 9")),
    ]
    for i, (side, c) in enumerate(pushes):
        g.push(side, c)
        o.push(side, c)
        mg = rows_multiset(g.poll_all())
        mo = rows_multiset(o.poll_all())
        assert mg == mo, f"push {i}: {mg} vs {mo}"
    g.close()
    o.close()


def test_join_checkpoint_spill_parity():
    # §8f-2 join-state spill: GPU vs oracle byte-identical per epoch
    # (memcmp(jk ∥ pk) keys sorted, value-encoded rows, in-epoch netting)
    rng = np.random.default_rng(123)
    g, o = join_pair()
    live = {SIDE_LEFT: [], SIDE_RIGHT: []}
    pk = 0
    for epoch in range(4):
        for side in (SIDE_LEFT, SIDE_RIGHT):
            n = 1024
            keys = rng.integers(0, 200, n)
            vals = np.arange(pk, pk + n)
            pk += n
            ops = np.zeros(n, np.uint8)
            for r in range(n):
                if live[side] and rng.random() < 0.3:
                    jx = int(rng.integers(0, len(live[side])))
                    keys[r], vals[r] = live[side].pop(jx)
                    ops[r] = ffi.OP_DELETE
                else:
                    live[side].append((int(keys[r]), int(vals[r])))
            c = mk_chunk([T_I64, T_I64], ops, [keys, vals])
            g.push(side, c)
            o.push(side, c)
            g.poll_all()
            o.poll_all()
        for side in (SIDE_LEFT, SIDE_RIGHT):
            sg = ffi.join_checkpoint_drain(gpu(), g.h, side)
            so = ffi.join_checkpoint_drain(ffi.oracle(), o.h, side)
            assert sg == so, (f"epoch {epoch} side {side}: "
                              f"{len(sg)} vs {len(so)} bytes")
    g.close()
    o.close()


def test_join_degree_spill_parity():
    # §8f-2 degree-table spill: GPU vs oracle byte-identical per epoch for
    # degree-carrying join types (main-delta keys + pre-epoch rows whose
    # degree changed during probes)
    from rwtest.ffi import JOIN_FULL_OUTER, JOIN_LEFT_OUTER, JOIN_LEFT_SEMI

    for jt in (JOIN_LEFT_OUTER, JOIN_FULL_OUTER, JOIN_LEFT_SEMI):
        rng = np.random.default_rng(77 + jt)
        kw = dict(key_l=[0], key_r=[0], pk_l=[1], pk_r=[1])
        g = ffi.HashJoin(gpu(), jt, [T_I64, T_I64], [T_I64, T_I64], **kw)
        o = ffi.HashJoin(oracle(), jt, [T_I64, T_I64], [T_I64, T_I64], **kw)
        live = {SIDE_LEFT: [], SIDE_RIGHT: []}
        pk = 0
        for epoch in range(4):
            for side in (SIDE_LEFT, SIDE_RIGHT):
                n = 768
                keys = rng.integers(0, 60, n)  # dense keys: many matches
                vals = np.arange(pk, pk + n)
                pk += n
                ops = np.zeros(n, np.uint8)
                for r in range(n):
                    if live[side] and rng.random() < 0.3:
                        jx = int(rng.integers(0, len(live[side])))
                        keys[r], vals[r] = live[side].pop(jx)
                        ops[r] = ffi.OP_DELETE
                    else:
                        live[side].append((int(keys[r]), int(vals[r])))
                c = mk_chunk([T_I64, T_I64], ops, [keys, vals])
                g.push(side, c)
                o.push(side, c)
                g.poll_all()
                o.poll_all()
            for side in (SIDE_LEFT, SIDE_RIGHT):
                mg = ffi.join_checkpoint_drain(gpu(), g.h, side)
                mo = ffi.join_checkpoint_drain(ffi.oracle(), o.h, side)
                assert mg == mo, f"jt {jt} epoch {epoch} side {side}: main"
                dg = ffi.join_degree_drain(gpu(), g.h, side)
                do = ffi.join_degree_drain(ffi.oracle(), o.h, side)
                assert dg == do, (f"jt {jt} epoch {epoch} side {side}: "
                                  f"degree {len(dg)} vs {len(do)} bytes")
        g.close()
        o.close()


def test_count_distinct_parity():
    # DISTINCT dedup (aggregate/distinct.rs): GPU vs oracle — hand case +
    # randomized duplicate-heavy insert/delete mix
    from rwtest.ffi import AGG_COUNT

    calls = [(AGG_COUNT_STAR, -1, T_I64), (AGG_COUNT, 1, T_I64, 1),
             (AGG_SUM, 1, T_I64, 1)]
    g = ffi.HashAgg(gpu(), [T_I64, T_I64], [0], calls, 0)
    o = ffi.HashAgg(oracle(), [T_I64, T_I64], [0], calls, 0)
    rng = np.random.default_rng(321)
    live = []
    for ep in range(5):
        n = 2048
        keys = rng.integers(0, 50, n)
        vals = rng.integers(0, 8, n)  # heavy duplication per (group, datum)
        ops = np.zeros(n, np.uint8)
        for r in range(n):
            if live and rng.random() < 0.4:
                jx = int(rng.integers(0, len(live)))
                keys[r], vals[r] = live.pop(jx)
                ops[r] = ffi.OP_DELETE
            else:
                live.append((int(keys[r]), int(vals[r])))
        c = mk_chunk([T_I64, T_I64], ops, [keys, vals])
        outs = []
        for a in (g, o):
            a.push(c)
            a.flush(ep + 1)
            outs.append(rows_multiset(a.poll_all()))
        assert outs[0] == outs[1], f"epoch {ep}"
        # §8f-2 dedup-table spill: byte-identical drains (both sorted by pk)
        from test_codec import dedup_drain, n_dedup_tables
        assert n_dedup_tables(gpu(), g) == n_dedup_tables(oracle(), o) == 1
        dg = dedup_drain(gpu(), g, 0)
        do = dedup_drain(oracle(), o, 0)
        assert dg == do, (f"epoch {ep}: dedup spill {len(dg)} vs {len(do)};"
                          f" first diff: "
                          f"{next(((a, b) for a, b in zip(dg, do) if a != b), None)}")
    g.close()
    o.close()


def test_topn_with_ties_parity():
    # WITH TIES: the two transcribed top_n_plain.rs fixtures (oracle pinned
    # in tests/test_oracle_topn.py) replayed GPU vs oracle, plus a
    # randomized heavy-tie mix
    from test_oracle_topn import I3, _prepend_group

    for limit, chunks in [
        (4, [""" I I\n + 1 0\n + 2 1\n + 3 2\n + 10 3\n + 9 4\n + 8 5""",
             """ I I\n + 7 6\n - 3 2\n - 1 0\n + 5 7\n - 2 1\n + 11 8""",
             """ I I\n + 6 9\n + 12 10\n + 13 11\n + 14 12""",
             """ I I\n - 5 7\n - 6 9\n - 11 8"""]),
        (3, [""" I I\n + 1 0\n + 2 1\n + 3 2\n + 10 3\n + 9 4\n + 8 5""",
             """ I I\n + 3 6\n + 3 7\n + 1 8\n + 2 9\n + 10 10""",
             """ I I\n - 1 0""",
             """ I I\n - 1 8"""]),
    ]:
        g = ffi.GroupTopN(gpu(), I3, [0], [(1, False)], [(2, False)],
                          offset=0, limit=limit, with_ties=True)
        o = ffi.GroupTopN(oracle(), I3, [0], [(1, False)], [(2, False)],
                          offset=0, limit=limit, with_ties=True)
        for i, pretty in enumerate(chunks):
            c = _prepend_group(pretty)
            g.push(c)
            o.push(c)
            mg = rows_multiset(g.poll_all())
            mo = rows_multiset(o.poll_all())
            assert mg == mo, f"limit={limit} push {i}: {mg} vs {mo}"
        g.close()
        o.close()

    rng = np.random.default_rng(77)
    g = ffi.GroupTopN(gpu(), I3, [0], [(1, False)], [(2, False)],
                      offset=0, limit=3, with_ties=True)
    o = ffi.GroupTopN(oracle(), I3, [0], [(1, False)], [(2, False)],
                      offset=0, limit=3, with_ties=True)
    live = []
    for i in range(8):
        n = 512
        gk = rng.integers(0, 20, n)
        ordv = rng.integers(0, 6, n)  # heavy ties
        pk = rng.integers(0, 10**7, n)
        ops = np.zeros(n, np.uint8)
        for r in range(n):
            if live and rng.random() < 0.4:
                jx = int(rng.integers(0, len(live)))
                gk[r], ordv[r], pk[r] = live.pop(jx)
                ops[r] = ffi.OP_DELETE
            else:
                live.append((int(gk[r]), int(ordv[r]), int(pk[r])))
        c = mk_chunk(I3, ops, [gk, ordv, pk])
        g.push(c)
        o.push(c)
        mg = rows_multiset(g.poll_all())
        mo = rows_multiset(o.poll_all())
        assert mg == mo, f"random push {i}: {len(mg)} vs {len(mo)}"
    g.close()
    o.close()


def test_eowc_parity():
    # emit-on-window-close: the transcribed golden sequence + a randomized
    # multi-window stream, GPU vs oracle (order-exact within a flush — both
    # emit group-key-sorted)
    ga = ffi.HashAgg(gpu(), [T_I64], [0], [(AGG_COUNT_STAR, -1, T_I64)], 0,
                     emit_on_window_close=True)
    oa = ffi.HashAgg(ffi.oracle(), [T_I64], [0], [(AGG_COUNT_STAR, -1, T_I64)],
                     0, emit_on_window_close=True)
    steps = [
        ("push", " I\n + 1\n + 2\n + 3"),
        ("flush", None),
        ("push", " I\n - 2\n + 4"),
        ("wm", 3), ("flush", None),
        ("wm", 4), ("flush", None),
        ("wm", 10), ("flush", None),
        ("wm", 20), ("flush", None),
    ]
    ep = 0
    for kind, arg in steps:
        outs = []
        for a in (ga, oa):
            if kind == "push":
                a.push(from_pretty(arg))
            elif kind == "wm":
                a.watermark(0, arg)
            else:
                ep += 1
                a.flush(ep)
            outs.append(ffi.rows_ordered(a.poll_all()))
        assert outs[0] == outs[1], f"{kind} {arg}: {outs[0]} vs {outs[1]}"
    ga.close()
    oa.close()

    # randomized: windows 0..300, watermark advances between epochs
    rng = np.random.default_rng(55)
    calls = [(AGG_COUNT_STAR, -1, T_I64), (ffi.AGG_SUM, 1, T_I64)]
    g = ffi.HashAgg(gpu(), [T_I64, T_I64], [0], calls, 0,
                    emit_on_window_close=True)
    o = ffi.HashAgg(ffi.oracle(), [T_I64, T_I64], [0], calls, 0,
                    emit_on_window_close=True)
    wm = 0
    for ep in range(6):
        n = 4096
        keys = rng.integers(wm, wm + 80, n)
        vals = rng.integers(1, 100, n)
        c = mk_chunk([T_I64, T_I64], np.zeros(n, np.uint8), [keys, vals])
        wm += 40
        outs = []
        for a in (g, o):
            a.push(c)
            a.watermark(0, wm)
            a.flush(ep + 1)
            outs.append(ffi.rows_ordered(a.poll_all()))
        assert outs[0] == outs[1], f"epoch {ep}: {len(outs[0])} vs {len(outs[1])}"
        # §8f-2 EOWC spill: mid-window state PUTs + window-close DELETEs —
        # same record multiset per epoch (sorted; dirty-list order differs)
        from test_codec import drain
        sg = sorted(drain(gpu(), g))
        so = sorted(drain(ffi.oracle(), o))
        assert sg == so, (f"epoch {ep}: eowc spill {len(sg)} vs {len(so)}; "
                          f"first diff: "
                          f"{next(((a, b) for a, b in zip(sg, so) if a != b), None)}")
    g.close()
    o.close()


def test_plain_topn_parity():
    # plain TopN (zero group columns): randomized stream, GPU vs oracle
    t2 = [T_I64, T_I64]
    rng = np.random.default_rng(42)
    g = ffi.GroupTopN(gpu(), t2, [], [(0, False)], [(1, False)],
                      offset=0, limit=5)
    o = ffi.GroupTopN(oracle(), t2, [], [(0, False)], [(1, False)],
                      offset=0, limit=5)
    live = []
    for i in range(8):
        n = 512
        ordv = rng.integers(0, 500, n)
        pk = rng.integers(0, 10**7, n)
        ops = np.zeros(n, np.uint8)
        for r in range(n):
            if live and rng.random() < 0.35:
                jx = int(rng.integers(0, len(live)))
                ordv[r], pk[r] = live.pop(jx)
                ops[r] = ffi.OP_DELETE
            else:
                live.append((int(ordv[r]), int(pk[r])))
        c = mk_chunk(t2, ops, [ordv, pk])
        g.push(c)
        o.push(c)
        mg = rows_multiset(g.poll_all())
        mo = rows_multiset(o.poll_all())
        assert mg == mo, f"push {i}: {mg} vs {mo}"
    g.close()
    o.close()


def test_topn_checkpoint_spill_parity():
    # GroupTopN §8f-2 spill: byte-identical GPU vs oracle per epoch,
    # incl. upserts (same storage key replaced) and delete mixes
    rng = np.random.default_rng(17)
    t3 = [T_I64, T_I64, T_I64]
    g = ffi.GroupTopN(gpu(), t3, [0], [(1, True)], [(2, False)],
                      offset=0, limit=3)
    o = ffi.GroupTopN(oracle(), t3, [0], [(1, True)], [(2, False)],
                      offset=0, limit=3)
    live = []
    for ep in range(4):
        n = 1024
        gk = rng.integers(0, 30, n)
        ordv = rng.integers(0, 40, n)
        pk = rng.integers(0, 50, n)  # small pk space -> frequent upserts
        ops = np.zeros(n, np.uint8)
        for r in range(n):
            if live and rng.random() < 0.3:
                jx = int(rng.integers(0, len(live)))
                gk[r], ordv[r], pk[r] = live.pop(jx)
                ops[r] = ffi.OP_DELETE
            else:
                live.append((int(gk[r]), int(ordv[r]), int(pk[r])))
        c = mk_chunk(t3, ops, [gk, ordv, pk])
        g.push(c)
        o.push(c)
        g.poll_all()
        o.poll_all()
        sg = ffi.topn_checkpoint_drain(gpu(), g.h)
        so = ffi.topn_checkpoint_drain(ffi.oracle(), o.h)
        assert sg == so, f"epoch {ep}: {len(sg)} vs {len(so)} bytes"
    g.close()
    o.close()


def test_join_partitioned_pipeline_parity():
    # The hash-prefix-partitioned probe/insert pipeline (jpart_* kernels)
    # engages only for all-Insert unique-key inner batches of >= 131072
    # rows (rw_amd.hip can_partition). Drive it with q8-shaped batches on
    # BOTH sides — so partitioned inserts build the chains that later
    # partitioned probes walk — and compare output multisets AND the
    # checkpoint spill bytes (which read back the scattered row records)
    # against the oracle.
    import os
    os.environ["RW_JOIN_PART"] = "1"  # opt into the partitioned pipeline
    rng = np.random.default_rng(77)
    t4 = [T_I64, T_TS, T_TS, T_I64]
    kw = dict(key_l=[0, 1, 2], key_r=[0, 1, 2], pk_l=[3], pk_r=[3])
    g = ffi.HashJoin(gpu(), JOIN_INNER, t4, t4, **kw)
    o = ffi.HashJoin(oracle(), JOIN_INNER, t4, t4, **kw)
    N = 140_000  # > JPART_MIN_ROWS
    rowid = 0

    def batch(ids, null_every=0):
        nonlocal rowid
        n = len(ids)
        ws = (ids % 7) * 10_000_000
        we = ws + 10_000_000
        rid = np.arange(rowid, rowid + n)
        rowid += n
        valid = np.ones(n, np.uint8)
        if null_every:
            # null join keys (never-match under non-null-safe): distinct
            # (ws, we) keeps the batch's keys pairwise distinct so the
            # partitioned path still engages
            valid[::null_every] = 0
        return ffi.Chunk(t4, np.zeros(n, np.uint8), [ids, ws, we, rid],
                         [valid, np.ones(n, np.uint8), np.ones(n, np.uint8),
                          np.ones(n, np.uint8)])

    # build side: two partitioned insert batches (distinct ids, no matches)
    perm = rng.permutation(400_000).astype(np.int64)
    pushes = [
        (SIDE_RIGHT, batch(perm[:N])),
        (SIDE_RIGHT, batch(perm[N:2 * N])),
        # probe batches: ~half the keys hit the build side; some null keys
        (SIDE_LEFT, batch(np.concatenate([perm[:N // 2],
                                          perm[2 * N:2 * N + N // 2]]),
                          null_every=997)),
        # second probe batch repeats earlier LEFT keys -> left chains of
        # length 2, then a RIGHT batch probes those chains
        (SIDE_LEFT, batch(perm[:N // 2 + N // 2])),
        (SIDE_RIGHT, batch(perm[2 * N + N: 2 * N + N + N // 4])),
    ]
    for i, (side, c) in enumerate(pushes):
        g.push(side, c)
        o.push(side, c)
        mg = rows_multiset(g.poll_all())
        mo = rows_multiset(o.poll_all())
        assert mg == mo, f"push {i}: GPU {len(mg)} rows vs oracle {len(mo)}"
    for side in (SIDE_LEFT, SIDE_RIGHT):
        sg = ffi.join_checkpoint_drain(gpu(), g.h, side)
        so = ffi.join_checkpoint_drain(ffi.oracle(), o.h, side)
        assert sg == so, (f"side {side}: spill {len(sg)} vs {len(so)} bytes")
    g.close()
    o.close()
    del os.environ["RW_JOIN_PART"]


def test_q7_pipeline_parity():
    # Full q7 plan chain: HashAgg max(price) by window
    # -> change stream -> inner join on price = max(price) with the BETWEEN
    # filter fused into emission (reference stream_plan nexmark.yaml q7).
    # GPU chain: device-resident hop (rw_agg_flush_device +
    # rw_join_apply_aggout); oracle chain: agg poll -> project -> join push.
    # Join outputs compared as multisets per barrier.
    import ctypes

    from rwtest.ffi import AGG_COUNT_STAR, AGG_MAX, CMP_GE, CMP_LE, \
        JOIN_INNER, SIDE_LEFT, SIDE_RIGHT

    rng = np.random.default_rng(41)
    W = 10_000_000
    calls = [(AGG_MAX, 1, T_I64), (AGG_COUNT_STAR, -1, T_I64)]
    ga = ffi.HashAgg(gpu(), [T_I64, T_I64], [0], calls, 1, append_only=True)
    oa = ffi.HashAgg(oracle(), [T_I64, T_I64], [0], calls, 1,
                     append_only=True)
    t5 = [T_I64] * 5
    jkw = dict(key_l=[2], key_r=[1], pk_l=[4], pk_r=[0],
               cond=(CMP_GE, 3, 5, -W), cond2=(CMP_LE, 3, 5, 0))
    gj = ffi.HashJoin(gpu(), JOIN_INNER, t5, [T_I64, T_I64], **jkw)
    oj = ffi.HashJoin(oracle(), JOIN_INNER, t5, [T_I64, T_I64], **jkw)

    L = gpu().lib
    L.rw_agg_flush_device.restype = ctypes.c_longlong
    L.rw_agg_flush_device.argtypes = [ctypes.c_void_p, ctypes.c_uint64]
    L.rw_join_apply_aggout.restype = ctypes.c_int
    L.rw_join_apply_aggout.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                       ctypes.c_int,
                                       ctypes.POINTER(ctypes.c_uint32),
                                       ctypes.c_int, ctypes.c_uint64]
    L.rw_join_marshal_output.restype = ctypes.c_int
    L.rw_join_marshal_output.argtypes = [ctypes.c_void_p]
    cmap = (ctypes.c_uint32 * 2)(0, 1)

    rowid = 0
    for epoch in range(5):
        for push in range(3):
            n = 1024
            dt = np.sort(rng.integers(epoch * 4 * W, (epoch * 4 + 4) * W, n))
            w = (dt // W + 1) * W
            # prices in a SMALL space so join matches and max-updates happen
            price = rng.integers(1, 50, n)
            auction = rng.integers(0, 100, n)
            bidder = rng.integers(0, 100, n)
            rid = np.arange(rowid, rowid + n)
            rowid += n
            ca = mk_chunk([T_I64, T_I64], np.zeros(n, np.uint8), [w, price])
            cj = mk_chunk(t5, np.zeros(n, np.uint8),
                          [auction, bidder, price, dt, rid])
            ga.push(ca)
            oa.push(ca)
            gj.push(SIDE_LEFT, cj)
            oj.push(SIDE_LEFT, cj)
        # barrier: agg flush; change stream feeds the join's right side
        n_changes = L.rw_agg_flush_device(ga.h, epoch + 1)
        assert n_changes >= 0, gpu().last_error()
        rc = L.rw_join_apply_aggout(gj.h, ga.h, SIDE_RIGHT, cmap, 2,
                                    ctypes.c_uint64(n_changes))
        assert rc == 0, gpu().last_error()
        assert L.rw_join_marshal_output(gj.h) == 0, gpu().last_error()

        oa.flush(epoch + 1)
        ochunks = oa.poll_all()
        ops, rows = [], []
        for c in ochunks:
            for op_tok, vals in c.visible_rows():
                ops.append(ffi.OP_BY_TOKEN[op_tok])
                rows.append(vals)
        if rows:
            # project the change stream to [window_end, maxprice]
            wcol = np.array([r[0] for r in rows], dtype=np.int64)
            mcol = np.array([r[1] for r in rows], dtype=np.int64)
            assert not any(r[0] is None or r[1] is None for r in rows)
            oj.push(SIDE_RIGHT,
                    mk_chunk([T_I64, T_I64], np.array(ops, np.uint8),
                             [wcol, mcol]))
        mg = rows_multiset(gj.poll_all())
        mo = rows_multiset(oj.poll_all())
        assert mg == mo, (f"epoch {epoch}: GPU {len(mg)} join rows vs "
                          f"oracle {len(mo)}")
    for x in (ga, oa, gj, oj):
        x.close()


def test_epoch_ingest_mode_parity():
    # product-reachable epoch-batched ingestion (rw_hash_agg_ingest_mode):
    # staged pushes + one apply launch at the barrier must equal the
    # per-push apply path AND the oracle on the same inputs
    import ctypes

    from rwtest.ffi import AGG_COUNT_STAR, AGG_MAX, AGG_SUM

    rng = np.random.default_rng(31)
    # value states only (retractable min/max is materialized-input, which
    # epoch-batched ingest rejects by contract)
    calls = [(AGG_SUM, 1, T_I64), (AGG_COUNT_STAR, -1, T_I64)]
    a_epoch = ffi.HashAgg(gpu(), [T_I64, T_I64], [0], calls, 1)
    a_plain = ffi.HashAgg(gpu(), [T_I64, T_I64], [0], calls, 1)
    a_orc = ffi.HashAgg(ffi.oracle(), [T_I64, T_I64], [0], calls, 1)
    L = gpu().lib
    L.rw_hash_agg_ingest_mode.restype = ctypes.c_int
    L.rw_hash_agg_ingest_mode.argtypes = [ctypes.c_void_p, ctypes.c_int]
    assert L.rw_hash_agg_ingest_mode(a_epoch.h, 1) == 0, gpu().last_error()
    live = []
    for epoch in range(4):
        for _ in range(5):
            n = 2048
            g = rng.integers(0, 100, n)
            v = rng.integers(1, 10_000, n)
            ops = np.zeros(n, np.uint8)
            for r in range(n):
                if live and rng.random() < 0.2:
                    j = int(rng.integers(0, len(live)))
                    g[r], v[r] = live.pop(j)
                    ops[r] = ffi.OP_DELETE
                else:
                    live.append((int(g[r]), int(v[r])))
            c = mk_chunk([T_I64, T_I64], ops, [g, v])
            for a in (a_epoch, a_plain, a_orc):
                a.push(c)
        outs = []
        for a in (a_epoch, a_plain, a_orc):
            a.flush(epoch + 1)
            outs.append(rows_multiset(a.poll_all()))
        assert outs[0] == outs[1] == outs[2], f"epoch {epoch} diverged"
    for a in (a_epoch, a_plain, a_orc):
        a.close()


def test_join_inner_multimatch_sparse_overflow():
    # the pre-assigned sparse emission's EXTRAS path: probe rows matching
    # MANY build rows (my_n up to 16) append past the sparse region via
    # per-lane atomics; drain compaction must reassemble the multiset
    rng = np.random.default_rng(91)
    g, o = join_pair()
    # build: 16 rows per key for 64 keys
    keys = np.repeat(np.arange(64), 16)
    vals = np.arange(len(keys))
    c = mk_chunk([T_I64, T_I64], np.zeros(len(keys), np.uint8), [keys, vals])
    g.push(SIDE_RIGHT, c)
    o.push(SIDE_RIGHT, c)
    g.poll_all()
    o.poll_all()
    # probe: hits interleaved with misses
    n = 4096
    pk = rng.integers(0, 128, n)  # half miss
    pv = np.arange(10_000, 10_000 + n)
    c = mk_chunk([T_I64, T_I64], np.zeros(n, np.uint8), [pk, pv])
    g.push(SIDE_LEFT, c)
    o.push(SIDE_LEFT, c)
    mg = rows_multiset(g.poll_all())
    mo = rows_multiset(o.poll_all())
    assert mg == mo and len(mg) > 30_000, f"{len(mg)} vs {len(mo)}"
    g.close()
    o.close()


def test_join_epoch_ingest_mode_parity():
    # rw_hash_join_ingest_mode: same-side chunk runs merged into one launch
    # must equal the per-push path AND the oracle
    import ctypes

    rng = np.random.default_rng(61)
    g_epoch, o = join_pair()
    g_plain = ffi.HashJoin(gpu(), JOIN_INNER, [T_I64, T_I64],
                           [T_I64, T_I64], key_l=[0], key_r=[0], pk_l=[1],
                           pk_r=[1])
    L = gpu().lib
    L.rw_hash_join_ingest_mode.restype = ctypes.c_int
    L.rw_hash_join_ingest_mode.argtypes = [ctypes.c_void_p, ctypes.c_int]
    assert L.rw_hash_join_ingest_mode(g_epoch.h, 1) == 0, gpu().last_error()
    live = {SIDE_LEFT: [], SIDE_RIGHT: []}
    pk = 0
    for epoch in range(3):
        # runs of same-side chunks with inserts + deletes
        for side in (SIDE_LEFT, SIDE_LEFT, SIDE_RIGHT, SIDE_RIGHT, SIDE_LEFT):
            n = 768
            keys = rng.integers(0, 120, n)
            vals = np.arange(pk, pk + n)
            pk += n
            ops = np.zeros(n, np.uint8)
            for r in range(n):
                if live[side] and rng.random() < 0.25:
                    j = int(rng.integers(0, len(live[side])))
                    keys[r], vals[r] = live[side].pop(j)
                    ops[r] = ffi.OP_DELETE
                else:
                    live[side].append((int(keys[r]), int(vals[r])))
            c = mk_chunk([T_I64, T_I64], ops, [keys, vals])
            for x in (g_epoch, g_plain, o):
                x.push(side, c)
        outs = []
        for x in (g_epoch, g_plain, o):
            x.flush(epoch + 1)
            outs.append(rows_multiset(x.poll_all()))
        assert outs[0] == outs[1] == outs[2], f"epoch {epoch} diverged"
        d0 = ffi.join_checkpoint_drain(gpu(), g_epoch.h, SIDE_LEFT)
        d1 = ffi.join_checkpoint_drain(gpu(), g_plain.h, SIDE_LEFT)
        d2 = ffi.join_checkpoint_drain(ffi.oracle(), o.h, SIDE_LEFT)
        assert d0 == d1 == d2, f"epoch {epoch}: left drains diverged"
    for x in (g_epoch, g_plain, o):
        x.close()


@pytest.mark.gpu
def test_inequality_join_watermark_gpu():
    # hash_join.rs:1740-1829 on the GPU path, plus drain byte-parity with
    # the oracle after the inequality state-cleaning sweep
    import risingwave_amd
    from rwtest.ffi import (CMP_GE, JOIN_INNER, SIDE_LEFT, SIDE_RIGHT,
                            T_I64, from_pretty, join_checkpoint_drain,
                            oracle, rows_ordered)

    risingwave_amd.load_library()
    glib = ffi.Lib(risingwave_amd.lib_path())
    I2 = [T_I64, T_I64]
    drains = {}
    for name, lib in (("gpu", glib), ("orc", oracle())):
        j = ffi.HashJoin(lib, JOIN_INNER, I2, I2, key_l=[0], key_r=[0],
                         pk_l=[1], pk_r=[1], cond=(CMP_GE, 1, 3),
                         wm_ineq=((1, 1, True, True),))
        j.push(SIDE_LEFT, from_pretty(" I I\n + 2 4\n + 2 7\n + 3 8"))
        assert rows_ordered(j.poll_all()) == []
        assert j.watermark(SIDE_LEFT, 1, 10) == []
        assert j.watermark(SIDE_RIGHT, 1, 6) == [(1, 6)]
        j.push(SIDE_RIGHT, from_pretty(" I I\n + 2 6"))
        assert rows_ordered(j.poll_all()) == [("+", (2, 7, 2, 6))]
        j.push(SIDE_RIGHT, from_pretty(" I I\n + 2 3"))
        assert rows_ordered(j.poll_all()) == [("+", (2, 7, 2, 3))]
        drains[name] = tuple(join_checkpoint_drain(lib, j.h, s)
                             for s in (SIDE_LEFT, SIDE_RIGHT))
        j.close()
    assert drains["gpu"] == drains["orc"]


@pytest.mark.gpu
def test_topn_recovery_fixtures_gpu():
    # the reference's own recovery golden vectors
    # (top_n_plain.rs:811-906 and :1113-1211) replayed on the GPU build
    import risingwave_amd
    from test_oracle_topn import (NEW_CHUNKS, NEW_EXPECT, TIES_CHUNKS,
                                  TIES_EXPECT)
    from rwtest.ffi import (T_I64, from_pretty, rows_multiset,
                            topn_checkpoint_drain, topn_restore)

    risingwave_amd.load_library()
    glib = ffi.Lib(risingwave_amd.lib_path())

    def expect(t, pretty):
        got = rows_multiset(t.poll_all())
        want = rows_multiset([from_pretty(pretty)]) if pretty else []
        assert got == want, f"got {got}\nwant {want}"

    I4 = [T_I64] * 4
    mk4 = lambda: ffi.GroupTopN(glib, I4, [], [(0, False), (3, False)], [],
                                offset=1, limit=3)
    a = mk4()
    for c, e in zip(NEW_CHUNKS[:2], NEW_EXPECT[:2]):
        a.push(from_pretty(c))
        expect(a, e)
    sp = topn_checkpoint_drain(glib, a.h)
    a.close()
    b = mk4()
    topn_restore(glib, b.h, sp)
    for c, e in zip(NEW_CHUNKS[2:], NEW_EXPECT[2:]):
        b.push(from_pretty(c))
        expect(b, e)
    b.close()

    mkt = lambda: ffi.GroupTopN(glib, [T_I64, T_I64], [], [(0, False)],
                                [(1, False)], offset=0, limit=3,
                                with_ties=True)
    a = mkt()
    for c, e in zip(TIES_CHUNKS[:2], TIES_EXPECT[:2]):
        a.push(from_pretty(c))
        expect(a, e)
    sp = topn_checkpoint_drain(glib, a.h)
    a.close()
    b = mkt()
    topn_restore(glib, b.h, sp)
    for c, e in zip(TIES_CHUNKS[2:], TIES_EXPECT[2:]):
        b.push(from_pretty(c))
        expect(b, e)
    b.close()


@pytest.mark.gpu
def test_distinct_dedup_invisible_rows_gpu():
    # the distinct deduplicater fixture's chunks (distinct.rs:373-471)
    # replayed at the executor level: the D (invisible) row must not
    # touch the dedup tables; emissions and every drain byte-compare
    # across builds, including a recovery replay between the chunks
    import risingwave_amd
    from rwtest.ffi import (AGG_COUNT, AGG_COUNT_STAR, AGG_SUM, T_I64,
                            agg_checkpoint_drain_bytes, agg_dedup_drain_bytes,
                            agg_dedup_restore, agg_restore, from_pretty,
                            oracle, rows_multiset)

    risingwave_amd.load_library()
    glib = ffi.Lib(risingwave_amd.lib_path())
    # count(distinct a), sum(distinct a), count(distinct b), count(*)
    # (the fixture's plain count(a) dropped: 4-call kernel limit);
    # group by c (single-group constant)
    calls = [(AGG_COUNT, 0, T_I64, 1), (AGG_SUM, 0, T_I64, 1),
             (AGG_COUNT, 1, T_I64, 1), (AGG_COUNT_STAR, -1, T_I64)]
    chunks = [
        " I  I  I\n + 1 10 0\n + 1 11 0",
        " I  I  I\n + 1 11 0\n + 2 12 0 D\n + 2 12 0",
    ]
    results = {}
    for name, lib in (("gpu", glib), ("orc", oracle())):
        a = ffi.HashAgg(lib, [T_I64, T_I64, T_I64], [2], calls, 3)
        outs, drains = [], []
        for e, c in enumerate(chunks):
            a.push(from_pretty(c))
            a.flush(e + 1)
            outs.append(rows_multiset(a.poll_all()))
            inter = agg_checkpoint_drain_bytes(lib, a.h)
            d0 = agg_dedup_drain_bytes(lib, a.h, 0)
            d1 = agg_dedup_drain_bytes(lib, a.h, 1)
            drains.append((inter, d0, d1))
        a.close()
        # recovery: replay all drains into a fresh executor, push one
        # more chunk exercising both dedup tables
        b = ffi.HashAgg(lib, [T_I64, T_I64, T_I64], [2], calls, 3)
        agg_dedup_restore(lib, b.h, 0, b"".join(d[1] for d in drains))
        agg_dedup_restore(lib, b.h, 1, b"".join(d[2] for d in drains))
        agg_restore(lib, b.h, b"".join(d[0] for d in drains))
        b.push(from_pretty(" I  I  I\n + 2 13 0\n + 3 10 0"))
        b.flush(9)
        outs.append(rows_multiset(b.poll_all()))
        drains.append((agg_checkpoint_drain_bytes(lib, b.h),
                       agg_dedup_drain_bytes(lib, b.h, 0),
                       agg_dedup_drain_bytes(lib, b.h, 1)))
        b.close()
        results[name] = (outs, drains)
    assert results["gpu"] == results["orc"]
    # count(distinct a)=1, sum(distinct a)=1, count(distinct b)=2, n=2
    first = results["gpu"][0][0]
    assert first == rows_multiset([from_pretty(
        " I I I I I\n + 0 1 1 2 2")])


@pytest.mark.gpu
def test_topn_float_group_key_zero_identity_gpu():
    # SQL equality for float group keys: -0.0, +0.0 (and every NaN) are
    # ONE group (the reference's HashKey normalizes floats,
    # common/src/hash/key.rs:517); the GPU's raw-bit slot compare
    # canonicalizes the key words. GPU vs oracle on emissions + drains +
    # a restore replay.
    import numpy as np

    import risingwave_amd
    from rwtest.ffi import (T_F64, T_I64, oracle, rows_multiset,
                            topn_checkpoint_drain, topn_restore)

    risingwave_amd.load_library()
    glib = ffi.Lib(risingwave_amd.lib_path())
    nan = float("nan")
    g = [0.0, -0.0, nan, -nan, 1.5, 0.0]
    v = [1.0, 2.0, 3.0, 4.0, 5.0, 6.0]
    pk = np.arange(6)
    results = {}
    for name, lib in (("gpu", glib), ("orc", oracle())):
        t = ffi.GroupTopN(lib, [T_F64, T_F64, T_I64], [0], [(1, False)],
                          [(2, False)], limit=2)
        t.push(ffi.Chunk([T_F64, T_F64, T_I64], np.zeros(6, np.uint8),
                         [np.array(g, np.float64), np.array(v, np.float64),
                          pk], [np.ones(6, np.uint8)] * 3))
        out = rows_multiset(t.poll_all())
        sp = topn_checkpoint_drain(lib, t.h)
        t.close()
        b = ffi.GroupTopN(lib, [T_F64, T_F64, T_I64], [0], [(1, False)],
                          [(2, False)], limit=2)
        topn_restore(lib, b.h, sp)
        # a late row joins the zero group after restore
        b.push(ffi.Chunk([T_F64, T_F64, T_I64], np.zeros(1, np.uint8),
                         [np.array([-0.0], np.float64),
                          np.array([0.5], np.float64),
                          np.array([9], np.int64)],
                         [np.ones(1, np.uint8)] * 3))
        out2 = rows_multiset(b.poll_all())
        sp2 = topn_checkpoint_drain(lib, b.h)
        b.close()
        def norm(ms):
            return sorted((op, tuple(repr(c) for c in row))
                          for op, row in ms)

        results[name] = (norm(out), sp, norm(out2), sp2)
    assert results["gpu"] == results["orc"]
    # zero group limit 2: of v=1,2,6 only the two smallest stay
    ops = [r[0] for r in results["gpu"][0]]
    assert ops.count("+") >= 4  # zero group 2 + nan group 2 + 1.5 group 1
