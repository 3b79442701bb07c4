"""GroupTopN oracle vs the reference's own test fixtures (CPU).

Golden vectors transcribed from
stream/src/executor/top_n/group_top_n.rs tests (sort_rows comparison =
per-push multiset):
  - test_without_offset_and_with_limits (storage_key [1,2,0] asc,
    group_by [1], order_by [2], offset 0 limit 2)
  - test_with_offset_and_with_limits (offset 1 limit 2)
  - test_multi_group_key (group_by [1,2], order_by [0], offset 0 limit 2)
  - test_compact_changes (ChangeBuffer merge incl. the no-output chunk)
"""
import numpy as np
import pytest

from rwtest import ffi
from rwtest.ffi import T_I64, from_pretty, oracle, rows_multiset

I3 = [T_I64, T_I64, T_I64]


def chunks_0_3():
    return [
        from_pretty("""  I I I
            + 10 9 1
            +  8 8 2
            +  7 8 2
            +  9 1 1
            + 10 1 1
            +  8 1 3"""),
        from_pretty("""  I I I
            - 10 9 1
            -  8 8 2
            - 10 1 1"""),
        from_pretty(""" I I I
            - 7 8 2
            - 8 1 3
            - 9 1 1"""),
        from_pretty("""  I I I
            +  5 1 1
            +  2 1 1
            +  3 1 2
            +  4 1 3"""),
    ]


def expect(t, pretty):
    got = rows_multiset(t.poll_all())
    want = rows_multiset([from_pretty(pretty)]) if pretty else []
    assert got == want, f"got {got}\nwant {want}"


def make(lib, offset, limit, group_by, order_by):
    # storage key = group_by ++ order_by ++ rest of stream key [1,2,0]
    sk = [1, 2, 0]
    rest = [(c, False) for c in sk if c not in group_by
            and c not in [o for o, _ in order_by]]
    return ffi.GroupTopN(lib, I3, group_by, order_by, rest,
                         offset=offset, limit=limit)


def test_without_offset_and_with_limits():
    t = make(oracle(), 0, 2, [1], [(2, False)])
    cs = chunks_0_3()
    t.push(cs[0])
    expect(t, """  I I I
        + 10 9 1
        +  8 8 2
        +  7 8 2
        +  9 1 1
        + 10 1 1""")
    t.push(cs[1])
    expect(t, """  I I I
        - 10 9 1
        -  8 8 2
        - 10 1 1
        +  8 1 3""")
    t.push(cs[2])
    expect(t, """ I I I
        - 7 8 2
        - 8 1 3
        - 9 1 1""")
    t.push(cs[3])
    expect(t, """ I I I
        + 5 1 1
        + 2 1 1""")
    t.close()


def test_with_offset_and_with_limits():
    t = make(oracle(), 1, 2, [1], [(2, False)])
    cs = chunks_0_3()
    t.push(cs[0])
    expect(t, """  I I I
        +  8 8 2
        + 10 1 1
        +  8 1 3""")
    t.push(cs[1])
    expect(t, """  I I I
        -  8 8 2
        - 10 1 1""")
    t.push(cs[2])
    expect(t, """ I I I
        - 8 1 3""")
    t.push(cs[3])
    expect(t, """ I I I
        + 5 1 1
        + 3 1 2""")
    t.close()


def test_multi_group_key():
    t = make(oracle(), 0, 2, [1, 2], [(0, False)])
    cs = chunks_0_3()
    t.push(cs[0])
    expect(t, """  I I I
        + 10 9 1
        +  8 8 2
        +  7 8 2
        +  9 1 1
        + 10 1 1
        +  8 1 3""")
    t.push(cs[1])
    expect(t, """  I I I
        - 10 9 1
        -  8 8 2
        - 10 1 1""")
    t.push(cs[2])
    expect(t, """  I I I
        - 7 8 2
        - 8 1 3
        - 9 1 1""")
    t.push(cs[3])
    expect(t, """  I I I
        +  5 1 1
        +  2 1 1
        +  3 1 2
        +  4 1 3""")
    t.close()


def test_compact_changes():
    # group_by [0,1], order_by [2]; storage key = [0,1,2]
    t = ffi.GroupTopN(oracle(), I3, [0, 1], [(2, False)], [],
                      offset=0, limit=2)
    t.push(from_pretty("""  I I I
        +  0 0 9
        +  0 0 8
        +  0 0 7
        +  0 0 6
        +  0 1 15
        +  0 1 14"""))
    expect(t, """  I I I
        +  0 0 7
        +  0 0 6
        +  0 1 15
        +  0 1 14""")
    t.push(from_pretty("""  I I I
        -  0 0 6
        -  0 0 8
        +  0 0 4
        +  0 0 3
        +  0 1 12
        +  0 2 26
        -  0 1 12
        +  0 1 11"""))
    expect(t, """  I I I
        -  0 0 6
        -  0 0 7
        +  0 0 4
        +  0 0 3
        -  0 1 15
        +  0 1 11
        +  0 2 26""")
    t.push(from_pretty("""  I I I
        +  0 0 11"""))
    expect(t, "")  # no chunk output
    t.close()


def test_topn_delete_update_pair():
    # delete old + insert new with the same cache key in one chunk → U-pair
    # (ChangeBuffer delete-then-insert merge, change_buffer.rs:76-120); needs
    # a non-key payload column: schema (k, ord, payload), storage key (ord),
    # group (k)
    t = ffi.GroupTopN(oracle(), I3, [0], [(1, False)], [], offset=0, limit=2)
    t.push(from_pretty(" I I I\n + 1 5 100\n + 1 6 200"))
    t.poll_all()
    t.push(from_pretty(" I I I\n - 1 5 100\n + 1 5 999"))
    outs = t.poll_all()
    rows = []
    for c in outs:
        rows.extend(c.visible_rows())
    assert ("U-", (1, 5, 100)) in rows and ("U+", (1, 5, 999)) in rows
    # adjacency: U- immediately followed by U+
    i = rows.index(("U-", (1, 5, 100)))
    assert rows[i + 1] == ("U+", (1, 5, 999))
    t.close()


def test_topn_desc_and_deep_offset():
    # DESC order key + offset beyond group size + NULL order values
    # (NULLS FIRST under DESC — sort_util.rs defaults)
    t = ffi.GroupTopN(oracle(), I3, [0], [(1, True)], [(2, False)],
                      offset=2, limit=2)
    t.push(from_pretty(""" I I I
        + 1 10 1
        + 1 30 2
        + 1 20 3
        + 1 . 4"""))
    # DESC NULLS FIRST: order = NULL(4), 30(2), 20(3), 10(1);
    # window [2,4) = 20(3), 10(1)
    got = rows_multiset(t.poll_all())
    want = rows_multiset([from_pretty(" I I I\n + 1 20 3\n + 1 10 1")])
    assert got == want, got
    # delete the NULL head: order becomes 30, 20, 10; window [2,4) = {10};
    # old window {20, 10} -> delta = -20 only (10 stays)
    t.push(from_pretty(" I I I\n - 1 . 4"))
    got = rows_multiset(t.poll_all())
    want = rows_multiset([from_pretty(" I I I\n - 1 20 3")])
    assert got == want, got
    t.close()


def _prepend_group(pretty):
    # plain-TopN fixtures → GroupTopN with a constant group column
    lines = [l.strip() for l in pretty.strip().split("\n") if l.strip()]
    out = ["I " + lines[0]]
    for l in lines[1:]:
        toks = l.split()
        if toks[0] in ("+", "-", "U-", "U+"):
            out.append(f"{toks[0]} 0 " + " ".join(toks[1:]))
        else:
            out.append("0 " + l)
    return from_pretty("\n".join(out))


def _ties_exec(limit):
    # schema (g, a, b); storage key (a asc, b asc); order by (a asc)
    return ffi.GroupTopN(oracle(), I3, [0], [(1, False)], [(2, False)],
                         offset=0, limit=limit, with_ties=True)


def test_with_limit_with_ties():
    # transcribed: top_n_plain.rs test_top_n_executor_with_limit_with_ties
    # (limit 4, WITH TIES), per-push sort_rows comparison
    t = _ties_exec(4)
    t.push(_prepend_group(""" I I
        +  1 0
        +  2 1
        +  3 2
        + 10 3
        +  9 4
        +  8 5"""))
    expect(t, """ I I I
        + 0 1 0
        + 0 2 1
        + 0 3 2
        + 0 8 5""")
    t.push(_prepend_group(""" I I
        +  7 6
        -  3 2
        -  1 0
        +  5 7
        -  2 1
        + 11 8"""))
    expect(t, """ I I I
        + 0 7 6
        - 0 3 2
        - 0 1 0
        + 0 5 7
        - 0 2 1
        + 0 9 4""")
    t.push(_prepend_group("""  I  I
        +  6  9
        + 12 10
        + 13 11
        + 14 12"""))
    expect(t, """ I I I
        - 0 9 4
        + 0 6 9""")
    t.push(_prepend_group("""  I  I
        -  5  7
        -  6  9
        - 11  8"""))
    expect(t, """ I I I
        - 0 5 7
        + 0 9 4
        - 0 6 9
        + 0 10 3""")
    t.close()


def test_with_ties():
    # transcribed: top_n_plain.rs test_with_ties (limit 3, real tie groups)
    t = _ties_exec(3)
    t.push(_prepend_group("""  I I
        +  1 0
        +  2 1
        +  3 2
        + 10 3
        +  9 4
        +  8 5"""))
    expect(t, """ I I I
        + 0 1 0
        + 0 2 1
        + 0 3 2""")
    t.push(_prepend_group("""  I I
        +  3 6
        +  3 7
        +  1 8
        +  2 9
        + 10 10"""))
    expect(t, """ I I I
        - 0 3 2
        + 0 1 8
        + 0 2 9""")
    t.push(_prepend_group(" I I\n - 1 0"))
    expect(t, " I I I\n - 0 1 0")
    t.push(_prepend_group(" I I\n - 1 8"))
    expect(t, """ I I I
        - 0 1 8
        + 0 3 2
        + 0 3 6
        + 0 3 7""")
    t.close()


def test_plain_topn_no_group():
    # plain TopN = GroupTopN with zero group columns (top_n_plain.rs): the
    # limit-4 WITH TIES fixture replayed on the bare 2-col schema
    t2 = [T_I64, T_I64]
    t = ffi.GroupTopN(oracle(), t2, [], [(0, False)], [(1, False)],
                      offset=0, limit=4, with_ties=True)
    t.push(from_pretty(""" I I
        +  1 0
        +  2 1
        +  3 2
        + 10 3
        +  9 4
        +  8 5"""))
    got = rows_multiset(t.poll_all())
    want = rows_multiset([from_pretty(" I I\n + 1 0\n + 2 1\n + 3 2\n + 8 5")])
    assert got == want, got
    t.push(from_pretty(""" I I
        +  7 6
        -  3 2
        -  1 0
        +  5 7
        -  2 1
        + 11 8"""))
    got = rows_multiset(t.poll_all())
    want = rows_multiset([from_pretty(
        " I I\n + 7 6\n - 3 2\n - 1 0\n + 5 7\n - 2 1\n + 9 4")])
    assert got == want, got
    t.close()


I4 = [T_I64, T_I64, T_I64, T_I64]


def make_plain4(offset, limit):
    # plain TopN (no group), order_by (c0 asc, c3 asc) == storage key
    # (top_n_plain.rs:732-741 "new" family)
    return ffi.GroupTopN(oracle(), I4, [], [(0, False), (3, False)], [],
                         offset=offset, limit=limit)


NEW_CHUNKS = [
    " I I I I\n + 1 1 4 1001",
    " I I I I\n + 5 1 4 1002",
    " I I I I\n + 1 9 1 1003\n + 9 8 1 1004\n + 0 2 3 1005",
    " I I I I\n + 1 0 2 1006",
]
NEW_EXPECT = [
    "",
    " I I I I\n + 5 1 4 1002",
    " I I I I\n + 1 9 1 1003\n + 1 1 4 1001",
    " I I I I\n - 5 1 4 1002\n + 1 0 2 1006",
]


def test_plain_offset_and_limit_new():
    # top_n_plain.rs:748-807 test_top_n_executor_with_offset_and_limit_new
    # (offset 1, limit 3)
    t = make_plain4(1, 3)
    for c, e in zip(NEW_CHUNKS, NEW_EXPECT):
        t.push(from_pretty(c))
        expect(t, e)
    t.close()


def test_plain_offset_and_limit_new_after_recovery():
    # top_n_plain.rs:811-906: chunks 0-1 before the crash, state drained,
    # a fresh executor hydrated, chunks 2-3 after — same outputs as the
    # uninterrupted fixture
    from rwtest.ffi import topn_checkpoint_drain, topn_restore

    a = make_plain4(1, 3)
    for c, e in zip(NEW_CHUNKS[:2], NEW_EXPECT[:2]):
        a.push(from_pretty(c))
        expect(a, e)
    sp = topn_checkpoint_drain(oracle(), a.h)
    a.close()
    b = make_plain4(1, 3)
    topn_restore(oracle(), b.h, sp)
    for c, e in zip(NEW_CHUNKS[2:], NEW_EXPECT[2:]):
        b.push(from_pretty(c))
        expect(b, e)
    b.close()


TIES_CHUNKS = [
    "  I I\n + 1 0\n + 2 1\n + 3 2\n + 10 3\n + 9 4\n + 8 5",
    "  I I\n + 3 6\n + 3 7\n + 1 8\n + 2 9\n + 10 10",
    " I I\n - 1 0",
    " I I\n - 1 8",
]
TIES_EXPECT = [
    " I I\n + 1 0\n + 2 1\n + 3 2",
    " I I\n - 3 2\n + 1 8\n + 2 9",
    " I I\n - 1 0",
    " I I\n - 1 8\n + 3 2\n + 3 6\n + 3 7",
]


def test_with_ties_recovery():
    # top_n_plain.rs:1113-1211 test_with_ties_recovery: the ties fixture
    # split at the barrier; a fresh executor hydrates from the drained
    # state table and continues identically
    from rwtest.ffi import topn_checkpoint_drain, topn_restore

    mk = lambda: ffi.GroupTopN(oracle(), [T_I64, T_I64], [], [(0, False)],
                               [(1, False)], offset=0, limit=3,
                               with_ties=True)
    a = mk()
    for c, e in zip(TIES_CHUNKS[:2], TIES_EXPECT[:2]):
        a.push(from_pretty(c))
        expect(a, e)
    sp = topn_checkpoint_drain(oracle(), a.h)
    a.close()
    b = mk()
    topn_restore(oracle(), b.h, sp)
    for c, e in zip(TIES_CHUNKS[2:], TIES_EXPECT[2:]):
        b.push(from_pretty(c))
        expect(b, e)
    b.close()


OLD_CHUNKS = [
    "  I I\n + 1 0\n + 2 1\n + 3 2\n + 10 3\n + 9 4\n + 8 5",
    "  I I\n + 7 6\n - 3 2\n - 1 0\n + 5 7\n - 2 1\n + 11 8",
    "  I I\n + 6 9\n + 12 10\n + 13 11\n + 14 12",
    "  I I\n - 5 7\n - 6 9\n - 11 8",
]


def make_plain2(offset, limit):
    # plain TopN, order_by c0 asc, storage key = order_by ++ [c1 asc]
    # (top_n_plain.rs:266-278 old family)
    return ffi.GroupTopN(oracle(), [T_I64, T_I64], [], [(0, False)],
                         [(1, False)], offset=offset, limit=limit)


def _run_old(offset, limit, expects):
    t = make_plain2(offset, limit)
    for c, e in zip(OLD_CHUNKS, expects):
        t.push(from_pretty(c))
        expect(t, e)
    t.close()


def test_plain_with_offset():
    # top_n_plain.rs:298-376 test_top_n_executor_with_offset (3, 1000);
    # limit 125 stands in for the reference's 1000 (TOPN_MAX_WIN is 128;
    # at most 12 rows are ever alive, so the windows are identical)
    _run_old(3, 125, [
        "  I I\n + 10 3\n + 9 4\n + 8 5",
        "  I I\n - 8 5\n + 11 8",
        "  I I\n + 8 5\n + 12 10\n + 13 11\n + 14 12",
        "  I I\n - 8 5\n - 9 4\n - 11 8",
    ])


def test_plain_with_limit():
    # top_n_plain.rs:378-463 test_top_n_executor_with_limit (0, 4)
    _run_old(0, 4, [
        "  I I\n + 1 0\n + 2 1\n + 3 2\n + 8 5",
        "  I I\n + 7 6\n - 3 2\n - 1 0\n + 5 7\n - 2 1\n + 9 4",
        "  I I\n - 9 4\n + 6 9",
        "  I I\n - 5 7\n + 9 4\n - 6 9\n + 10 3",
    ])


def test_plain_with_offset_and_limit():
    # top_n_plain.rs:551-630 test_top_n_executor_with_offset_and_limit
    # (3, 4)
    _run_old(3, 4, [
        "  I I\n + 10 3\n + 9 4\n + 8 5",
        "  I I\n - 8 5\n + 11 8",
        "  I I\n + 8 5",
        "  I I\n - 8 5\n + 12 10\n - 9 4\n + 13 11\n - 11 8\n + 14 12",
    ])
