"""Decimal sum parity (north star: integer/DECIMAL
values bit-identical). The reference's decimal is rust_decimal 1.40.0
(96-bit mantissa, scale 0..28) wrapped with NaN/±Inf variants
(types/decimal.rs:36-44); sum uses checked_add (expr general.rs:23). The
build restates the EXACT domain — sums accumulated as exact 256-bit
integers at scale 28 — and pins it against python's `decimal` module as an
independent exact-arithmetic oracle (sum of decimals with result scale =
max input scale is pure rational arithmetic, identical across correct
implementations). The reference's silent precision-loss rescale path is
order-dependent even in the reference and raises loudly instead
(DESIGN.md §9)."""
import decimal

import numpy as np
import pytest

from rwtest import ffi
from rwtest.ffi import (AGG_COUNT, AGG_COUNT_STAR, AGG_SUM, T_DECIMAL,
                        T_I64, dec16, dec16_value, dec_col, oracle,
                        rows_multiset)


def mk_dec_chunk(groups, decs, ops=None, valids=None):
    n = len(groups)
    if ops is None:
        ops = np.zeros(n, np.uint8)
    if valids is None:
        valids = [np.ones(n, np.uint8)] * 2
    return ffi.Chunk([T_I64, T_DECIMAL], np.asarray(ops, np.uint8),
                     [np.asarray(groups, np.int64), dec_col(decs)], valids)


def mk_agg(lib):
    calls = [(AGG_SUM, 1, T_DECIMAL), (AGG_COUNT_STAR, -1, T_I64)]
    return ffi.HashAgg(lib, [T_I64, T_DECIMAL], [0], calls, 1)


def _rand_dec(rng):
    scale = int(rng.integers(0, 6))
    m = int(rng.integers(-10**9, 10**9))
    return decimal.Decimal(m).scaleb(-scale)


def _ground_truth_flow(lib):
    """insert/retract mix vs python-Decimal exact sums (no full
    retractions, so the group's max scale is monotone)."""
    rng = np.random.default_rng(17)
    a = mk_agg(lib)
    state = {}  # g -> (list of live decimals, max scale ever, count)
    for epoch in range(3):
        live = [(g, d) for g, (ds, _, _) in state.items() for d in ds]
        for _ in range(3):
            n = 256
            gs, ds, ops = [], [], []
            for r in range(n):
                if live and rng.random() < 0.2:
                    j = int(rng.integers(0, len(live)))
                    g, d = live.pop(j)
                    st = state[g]
                    st[0].remove(d)
                    state[g] = (st[0], st[1], st[2] - 1)
                    ops.append(ffi.OP_DELETE)
                else:
                    g = int(rng.integers(0, 20))
                    d = _rand_dec(rng)
                    st = state.setdefault(g, ([], 0, 0))
                    st[0].append(d)
                    state[g] = (st[0], max(st[1], -d.as_tuple().exponent),
                                st[2] + 1)
                    live.append((g, d))
                    ops.append(ffi.OP_INSERT)
                gs.append(g)
                ds.append(d)
            a.push(mk_dec_chunk(gs, ds, ops))
        a.flush(epoch + 1)
        out = {}
        for c in a.poll_all():
            for op, vals in c.visible_rows():
                if op in ("+", "U+"):
                    out[vals[0]] = vals[1]
        for g, sumbytes in out.items():
            ds, maxscale, cnt = state[g]
            if cnt == 0:
                continue
            expect = sum(ds, decimal.Decimal(0))
            got = dec16_value(sumbytes)
            assert got == expect, f"group {g}: {got} != {expect}"
            # scale (trailing zeros) must match the reference's max-scale
            # semantics exactly — bit-identity, not just value equality
            assert -got.as_tuple().exponent == maxscale, (
                f"group {g}: scale {-got.as_tuple().exponent} != {maxscale}")
    a.close()


def test_decimal_sum_vs_python_decimal_oracle():
    _ground_truth_flow(oracle())


def test_decimal_specials():
    a = mk_agg(oracle())
    # Inf dominates; +Inf + -Inf -> NaN; NaN -> NaN; retraction restores
    a.push(mk_dec_chunk([1, 1, 2, 2, 3], ["1.5", "Inf", "Inf", "-Inf", "NaN"]))
    a.flush(1)
    out = {v[0]: v[1] for op, v in rows_multiset(a.poll_all()) if op == "+"}
    assert dec16_value(out[1]) == "Inf"
    assert dec16_value(out[2]) == "NaN"
    assert dec16_value(out[3]) == "NaN"
    # retract the Inf: group 1 returns to the finite sum
    a.push(mk_dec_chunk([1], ["Inf"], ops=[ffi.OP_DELETE]))
    a.flush(2)
    rows = rows_multiset(a.poll_all())
    upd = {v[0]: v[1] for op, v in rows if op == "U+"}
    assert dec16_value(upd[1]) == decimal.Decimal("1.5")
    a.close()


def test_decimal_value_equality_no_noop_update():
    # rust_decimal Eq compares VALUES (1.2 == 1.20): adding 0.00 changes the
    # state's scale but not its value -> count changes, sum compares equal;
    # a U-/U+ pair IS emitted (count differs) and the sum halves are
    # value-equal
    a = mk_agg(oracle())
    a.push(mk_dec_chunk([7], ["1.2"]))
    a.flush(1)
    a.poll_all()
    a.push(mk_dec_chunk([7], ["0.00"]))
    a.flush(2)
    rows = rows_multiset(a.poll_all())
    ups = [v for op, v in rows if op == "U+"]
    assert len(ups) == 1
    assert dec16_value(ups[0][1]) == decimal.Decimal("1.2")
    assert ups[0][2] == 2  # count changed
    a.close()


def test_decimal_overflow_raises():
    a = mk_agg(oracle())
    big = decimal.Decimal((1 << 96) - 1)  # max 96-bit mantissa, scale 0
    a.push(mk_dec_chunk([1, 1], [big, big]))
    with pytest.raises(RuntimeError, match="exact"):
        a.flush(1)
    a.close()


def test_decimal_rejections():
    with pytest.raises(Exception):
        ffi.HashAgg(oracle(), [T_DECIMAL, T_I64], [0],
                    [(AGG_COUNT_STAR, -1, T_I64)], 0)  # decimal group key
    with pytest.raises(Exception):
        ffi.HashAgg(oracle(), [T_I64, T_DECIMAL], [0],
                    [(ffi.AGG_MIN, 1, T_DECIMAL),
                     (AGG_COUNT_STAR, -1, T_I64)], 1)  # decimal min


def test_decimal_count_arg():
    # count(decimal col) counts non-NULL rows; NULL decimals skipped by sum
    a = ffi.HashAgg(oracle(), [T_I64, T_DECIMAL], [0],
                    [(AGG_COUNT, 1, T_I64), (AGG_COUNT_STAR, -1, T_I64)], 1)
    valids = [np.ones(3, np.uint8), np.array([1, 0, 1], np.uint8)]
    a.push(mk_dec_chunk([1, 1, 1], ["1", "2", "3"], valids=valids))
    a.flush(1)
    rows = rows_multiset(a.poll_all())
    assert rows == [("+", (1, 2, 3))]
    a.close()


@pytest.mark.gpu
def test_decimal_gpu_parity():
    import risingwave_amd

    risingwave_amd.load_library()
    glib = ffi.Lib(risingwave_amd.lib_path())
    _ground_truth_flow(glib)
    # GPU vs oracle: emissions AND spill bytes equal on a shared stream
    from rwtest.ffi import agg_checkpoint_drain_bytes

    rng = np.random.default_rng(23)
    g, o = mk_agg(glib), mk_agg(oracle())
    for epoch in range(3):
        for _ in range(3):
            n = 512
            gs = rng.integers(0, 30, n)
            ds = [_rand_dec(rng) for _ in range(n)]
            c = mk_dec_chunk(gs, ds)
            g.push(c)
            o.push(c)
        outs = []
        for a in (g, o):
            a.flush(epoch + 1)
            outs.append(rows_multiset(a.poll_all()))
        assert outs[0] == outs[1], f"epoch {epoch} diverged"
        dg = agg_checkpoint_drain_bytes(glib, g.h)
        do = agg_checkpoint_drain_bytes(oracle(), o.h)
        assert dg == do, f"epoch {epoch}: spill bytes diverged"
    g.close()
    o.close()


def _decimal_restore_flow(lib):
    from rwtest.ffi import agg_checkpoint_drain_bytes, agg_restore

    rng = np.random.default_rng(77)
    a = mk_agg(lib)
    live = []
    drains = b""

    def chunks():
        out = []
        for _ in range(2):
            n = 256
            gs, ds, ops = [], [], []
            for r in range(n):
                if live and rng.random() < 0.2:
                    j = int(rng.integers(0, len(live)))
                    g, d = live.pop(j)
                    ops.append(ffi.OP_DELETE)
                else:
                    g = int(rng.integers(0, 15))
                    d = _rand_dec(rng)
                    live.append((g, d))
                    ops.append(ffi.OP_INSERT)
                gs.append(g)
                ds.append(d)
            out.append(mk_dec_chunk(gs, ds, ops))
        return out

    for e in range(3):
        for c in chunks():
            a.push(c)
        a.flush(e + 1)
        a.poll_all()
        drains += agg_checkpoint_drain_bytes(lib, a.h)
    b = mk_agg(lib)
    agg_restore(lib, b.h, drains)
    for e in range(3):
        cs = chunks()
        outs = []
        for x in (a, b):
            for c in cs:
                x.push(c)
            x.flush(5 + e)
            outs.append(rows_multiset(x.poll_all()))
        assert outs[0] == outs[1], f"epoch {e}: restored decimal agg diverged"
        da = agg_checkpoint_drain_bytes(lib, a.h)
        db = agg_checkpoint_drain_bytes(lib, b.h)
        assert da == db, f"epoch {e}: decimal drain diverged"
    a.close()
    b.close()


def test_decimal_restore_oracle():
    _decimal_restore_flow(oracle())


@pytest.mark.gpu
def test_decimal_restore_gpu():
    import risingwave_amd

    risingwave_amd.load_library()
    _decimal_restore_flow(ffi.Lib(risingwave_amd.lib_path()))
