"""Oracle HashJoin vs the reference's own golden vectors.

Fixtures transcribed verbatim from the in-module tests of
/root/reference/src/stream/src/executor/hash_join.rs (line refs per test).
The "classical" executor (hash_join.rs:1538-1655): two [i64,i64] inputs,
join key [0], deduped pk [1], chunk_size 1024, output = all columns
(one side only for semi/anti); condition default `$1 < $3` (:1530-1536).
The "append_only" executor (:1657-1737): [i64,i64,i64] inputs, key [0,1],
deduped pk [], append_only_optimize on.

These single-side-input sequences are deterministic in the reference
(matched rows iterate in memcomparable pk order), so comparisons are
order-exact — stronger than the multiset bar. Rows marked D in the
expected chunks are invisible (noop-update-eliminated) and thus absent
from visible_rows().
"""
from rwtest import ffi
from rwtest.ffi import (
    CMP_LT, JOIN_FULL_OUTER, JOIN_INNER, JOIN_LEFT_ANTI, JOIN_LEFT_OUTER,
    JOIN_LEFT_SEMI, JOIN_RIGHT_OUTER, JOIN_RIGHT_SEMI, SIDE_LEFT, SIDE_RIGHT,
    T_I64, from_pretty, oracle, rows_ordered,
)

I2 = [T_I64, T_I64]
I3 = [T_I64, T_I64, T_I64]


def classical(join_type, null_safe=False, cond=False):
    return ffi.HashJoin(
        oracle(), join_type, I2, I2, key_l=[0], key_r=[0], pk_l=[1], pk_r=[1],
        null_safe=[1 if null_safe else 0],
        cond=(CMP_LT, 1, 3) if cond else None,
    )


def append_only(join_type):
    return ffi.HashJoin(
        oracle(), join_type, I3, I3, key_l=[0, 1], key_r=[0, 1],
        pk_l=[], pk_r=[], null_safe=[0, 0], append_only=True,
    )


def push(j, side, pretty):
    j.push(side, from_pretty(pretty))
    return rows_ordered(j.poll_all())


def rows(spec):
    """spec: list of (op, *values) tuples."""
    return [(s[0], tuple(s[1:])) for s in spec]


def test_inner_join():
    # hash_join.rs:1831-1898
    j = classical(JOIN_INNER)
    assert push(j, SIDE_LEFT, " I I\n + 1 4\n + 2 5\n + 3 6") == []
    assert push(j, SIDE_LEFT, " I I\n + 3 8\n - 3 8") == []
    assert push(j, SIDE_RIGHT, " I I\n + 2 7\n + 4 8\n + 6 9") == rows(
        [("+", 2, 5, 2, 7)]
    )
    assert push(j, SIDE_RIGHT, " I I\n + 3 10\n + 6 11") == rows([("+", 3, 6, 3, 10)])
    j.close()


def test_null_safe_inner_join():
    # hash_join.rs:1901-1968
    j = classical(JOIN_INNER, null_safe=True)
    assert push(j, SIDE_LEFT, " I I\n + 1 4\n + 2 5\n + . 6") == []
    assert push(j, SIDE_LEFT, " I I\n + . 8\n - . 8") == []
    assert push(j, SIDE_RIGHT, " I I\n + 2 7\n + 4 8\n + 6 9") == rows(
        [("+", 2, 5, 2, 7)]
    )
    out = push(j, SIDE_RIGHT, " I I\n + . 10\n + 6 11")
    assert out == [("+", (None, 6, None, 10))]
    j.close()


def test_non_null_safe_null_keys_never_match():
    # the null-bitmap subset check (hash_join.rs:1004-1016): NULL keys on a
    # non-null-safe join never match and are not stored
    j = classical(JOIN_INNER)
    assert push(j, SIDE_LEFT, " I I\n + . 6") == []
    assert push(j, SIDE_RIGHT, " I I\n + . 10") == []
    j.close()


def test_left_semi_join():
    # hash_join.rs:1971-2078
    j = classical(JOIN_LEFT_SEMI)
    assert push(j, SIDE_LEFT, " I I\n + 1 4\n + 2 5\n + 3 6") == []
    assert push(j, SIDE_LEFT, " I I\n + 3 8\n - 3 8") == []
    assert push(j, SIDE_RIGHT, " I I\n + 2 7\n + 4 8\n + 6 9") == rows([("+", 2, 5)])
    assert push(j, SIDE_RIGHT, " I I\n + 3 10\n + 6 11") == rows([("+", 3, 6)])
    assert push(j, SIDE_LEFT, " I I\n + 6 10") == rows([("+", 6, 10)])
    assert push(j, SIDE_RIGHT, " I I\n - 6 11") == []
    assert push(j, SIDE_RIGHT, " I I\n - 6 9") == rows([("-", 6, 10)])
    j.close()


def test_right_semi_join():
    # hash_join.rs:2410-2517 (mirror of left semi)
    j = classical(JOIN_RIGHT_SEMI)
    assert push(j, SIDE_RIGHT, " I I\n + 1 4\n + 2 5\n + 3 6") == []
    assert push(j, SIDE_RIGHT, " I I\n + 3 8\n - 3 8") == []
    assert push(j, SIDE_LEFT, " I I\n + 2 7\n + 4 8\n + 6 9") == rows([("+", 2, 5)])
    assert push(j, SIDE_LEFT, " I I\n + 3 10\n + 6 11") == rows([("+", 3, 6)])
    assert push(j, SIDE_RIGHT, " I I\n + 6 10") == rows([("+", 6, 10)])
    assert push(j, SIDE_LEFT, " I I\n - 6 11") == []
    assert push(j, SIDE_LEFT, " I I\n - 6 9") == rows([("-", 6, 10)])
    j.close()


def test_left_anti_join():
    # hash_join.rs:2520-2647
    j = classical(JOIN_LEFT_ANTI)
    assert push(j, SIDE_LEFT, " I I\n + 1 4\n + 2 5\n + 3 6") == rows(
        [("+", 1, 4), ("+", 2, 5), ("+", 3, 6)]
    )
    assert push(j, SIDE_LEFT, " I I\n + 3 8\n - 3 8") == []  # D D eliminated
    assert push(j, SIDE_RIGHT, " I I\n + 2 7\n + 4 8\n + 6 9") == rows([("-", 2, 5)])
    assert push(j, SIDE_RIGHT, " I I\n + 3 10\n + 6 11\n + 1 2\n + 1 3") == rows(
        [("-", 3, 6), ("-", 1, 4)]
    )
    assert push(j, SIDE_LEFT, " I I\n + 9 10") == rows([("+", 9, 10)])
    assert push(j, SIDE_RIGHT, " I I\n - 1 2") == []
    assert push(j, SIDE_RIGHT, " I I\n - 1 3") == rows([("+", 1, 4)])
    j.close()


def test_inner_join_append_only():
    # hash_join.rs:2191-2261
    j = append_only(JOIN_INNER)
    assert push(j, SIDE_LEFT, " I I I\n + 1 4 1\n + 2 5 2\n + 3 6 3") == []
    assert push(j, SIDE_LEFT, " I I I\n + 4 9 4\n + 5 10 5") == []
    assert push(j, SIDE_RIGHT, " I I I\n + 2 5 1\n + 4 9 2\n + 6 9 3") == rows(
        [("+", 2, 5, 2, 2, 5, 1), ("+", 4, 9, 4, 4, 9, 2)]
    )
    assert push(j, SIDE_RIGHT, " I I I\n + 1 4 4\n + 3 6 5") == rows(
        [("+", 1, 4, 1, 1, 4, 4), ("+", 3, 6, 3, 3, 6, 5)]
    )
    j.close()


def test_left_semi_join_append_only():
    # hash_join.rs:2264-2334
    j = append_only(JOIN_LEFT_SEMI)
    assert push(j, SIDE_LEFT, " I I I\n + 1 4 1\n + 2 5 2\n + 3 6 3") == []
    assert push(j, SIDE_LEFT, " I I I\n + 4 9 4\n + 5 10 5") == []
    assert push(j, SIDE_RIGHT, " I I I\n + 2 5 1\n + 4 9 2\n + 6 9 3") == rows(
        [("+", 2, 5, 2), ("+", 4, 9, 4)]
    )
    assert push(j, SIDE_RIGHT, " I I I\n + 1 4 4\n + 3 6 5") == rows(
        [("+", 1, 4, 1), ("+", 3, 6, 3)]
    )
    j.close()


def test_inner_join_with_barrier():
    # hash_join.rs:2780-2872 — processed order after barrier alignment:
    # l1, r1, <barrier>, l2, r2
    j = classical(JOIN_INNER)
    assert push(j, SIDE_LEFT, " I I\n + 1 4\n + 2 5\n + 3 6") == []
    assert push(j, SIDE_RIGHT, " I I\n + 2 7\n + 4 8\n + 6 9") == rows(
        [("+", 2, 5, 2, 7)]
    )
    j.flush(2)
    assert push(j, SIDE_LEFT, " I I\n + 6 8\n + 3 8") == rows([("+", 6, 8, 6, 9)])
    assert push(j, SIDE_RIGHT, " I I\n + 3 10\n + 6 11") == rows(
        [("+", 3, 6, 3, 10), ("+", 3, 8, 3, 10), ("+", 6, 8, 6, 11)]
    )
    j.close()


def test_inner_join_with_null_and_barrier():
    # hash_join.rs:2875-2967 — NULLs in value columns
    j = classical(JOIN_INNER)
    assert push(j, SIDE_LEFT, " I I\n + 1 4\n + 2 .\n + 3 .") == []
    out = push(j, SIDE_RIGHT, " I I\n + 2 7\n + 4 8\n + 6 9")
    assert out == [("+", (2, None, 2, 7))]
    j.flush(2)
    out = push(j, SIDE_LEFT, " I I\n + 6 .\n + 3 8")
    assert out == [("+", (6, None, 6, 9))]
    out = push(j, SIDE_RIGHT, " I I\n + 3 10\n + 6 11")
    assert out == [
        ("+", (3, 8, 3, 10)),
        ("+", (3, None, 3, 10)),
        ("+", (6, None, 6, 11)),
    ]
    j.close()


def test_left_outer_join():
    # hash_join.rs:2970-3051
    j = classical(JOIN_LEFT_OUTER)
    assert push(j, SIDE_LEFT, " I I\n + 1 4\n + 2 5\n + 3 6") == [
        ("+", (1, 4, None, None)),
        ("+", (2, 5, None, None)),
        ("+", (3, 6, None, None)),
    ]
    assert push(j, SIDE_LEFT, " I I\n + 3 8\n - 3 8") == []  # D D
    assert push(j, SIDE_RIGHT, " I I\n + 2 7\n + 4 8\n + 6 9") == [
        ("-", (2, 5, None, None)),
        ("+", (2, 5, 2, 7)),
    ]
    assert push(j, SIDE_RIGHT, " I I\n + 3 10\n + 6 11") == [
        ("-", (3, 6, None, None)),
        ("+", (3, 6, 3, 10)),
    ]
    j.close()


def test_null_safe_left_outer_join():
    # hash_join.rs:3054-3135
    j = classical(JOIN_LEFT_OUTER, null_safe=True)
    assert push(j, SIDE_LEFT, " I I\n + 1 4\n + 2 5\n + . 6") == [
        ("+", (1, 4, None, None)),
        ("+", (2, 5, None, None)),
        ("+", (None, 6, None, None)),
    ]
    assert push(j, SIDE_LEFT, " I I\n + . 8\n - . 8") == []
    assert push(j, SIDE_RIGHT, " I I\n + 2 7\n + 4 8\n + 6 9") == [
        ("-", (2, 5, None, None)),
        ("+", (2, 5, 2, 7)),
    ]
    assert push(j, SIDE_RIGHT, " I I\n + . 10\n + 6 11") == [
        ("-", (None, 6, None, None)),
        ("+", (None, 6, None, 10)),
    ]
    j.close()


def test_right_outer_join():
    # hash_join.rs:3138-3203
    j = classical(JOIN_RIGHT_OUTER)
    assert push(j, SIDE_LEFT, " I I\n + 1 4\n + 2 5\n + 3 6") == []
    assert push(j, SIDE_LEFT, " I I\n + 3 8\n - 3 8") == []
    assert push(j, SIDE_RIGHT, " I I\n + 2 7\n + 4 8\n + 6 9") == [
        ("+", (2, 5, 2, 7)),
        ("+", (None, None, 4, 8)),
        ("+", (None, None, 6, 9)),
    ]
    assert push(j, SIDE_RIGHT, " I I\n + 5 10\n - 5 10") == []  # D D
    j.close()


def test_full_outer_join():
    # hash_join.rs:3366-3449
    j = classical(JOIN_FULL_OUTER)
    assert push(j, SIDE_LEFT, " I I\n + 1 4\n + 2 5\n + 3 6") == [
        ("+", (1, 4, None, None)),
        ("+", (2, 5, None, None)),
        ("+", (3, 6, None, None)),
    ]
    assert push(j, SIDE_LEFT, " I I\n + 3 8\n - 3 8") == []
    assert push(j, SIDE_RIGHT, " I I\n + 2 7\n + 4 8\n + 6 9") == [
        ("-", (2, 5, None, None)),
        ("+", (2, 5, 2, 7)),
        ("+", (None, None, 4, 8)),
        ("+", (None, None, 6, 9)),
    ]
    assert push(j, SIDE_RIGHT, " I I\n + 5 10\n - 5 10") == []
    j.close()


def test_full_outer_join_update():
    # hash_join.rs:3452-3510 — NULL transitions + noop elimination leave
    # exactly a delete+insert pair
    j = classical(JOIN_FULL_OUTER)
    assert push(j, SIDE_LEFT, " I I\n + 1 1") == [("+", (1, 1, None, None))]
    assert push(j, SIDE_RIGHT, " I I\n + 1 1") == [
        ("-", (1, 1, None, None)),
        ("+", (1, 1, 1, 1)),
    ]
    assert push(j, SIDE_LEFT, " I I\n - 1 1\n + 1 2") == [
        ("-", (1, 1, 1, 1)),
        ("+", (1, 2, 1, 1)),
    ]
    j.close()


def test_full_outer_join_nonequi():
    # hash_join.rs:3513-3606 — incl. regression #2420 (forward once on
    # multiple condition-failing matches; forward on empty entry)
    j = classical(JOIN_FULL_OUTER, cond=True)
    assert push(j, SIDE_LEFT, " I I\n + 1 4\n + 2 5\n + 3 6\n + 3 7") == [
        ("+", (1, 4, None, None)),
        ("+", (2, 5, None, None)),
        ("+", (3, 6, None, None)),
        ("+", (3, 7, None, None)),
    ]
    assert push(j, SIDE_LEFT, " I I\n + 3 8\n - 3 8\n - 1 4") == [
        ("-", (1, 4, None, None))
    ]
    assert push(j, SIDE_RIGHT, " I I\n + 2 6\n + 4 8\n + 3 4") == [
        ("-", (2, 5, None, None)),
        ("+", (2, 5, 2, 6)),
        ("+", (None, None, 4, 8)),
        ("+", (None, None, 3, 4)),
    ]
    assert push(j, SIDE_RIGHT, " I I\n + 5 10\n - 5 10\n + 1 2") == [
        ("+", (None, None, 1, 2))
    ]
    j.close()


def test_inner_join_nonequi():
    # hash_join.rs:3609-3664
    j = classical(JOIN_INNER, cond=True)
    assert push(j, SIDE_LEFT, " I I\n + 1 4\n + 2 10\n + 3 6") == []
    assert push(j, SIDE_LEFT, " I I\n + 3 8\n - 3 8") == []
    assert push(j, SIDE_RIGHT, " I I\n + 2 7\n + 4 8\n + 6 9") == []
    assert push(j, SIDE_RIGHT, " I I\n + 3 10\n + 6 11") == rows([("+", 3, 6, 3, 10)])
    j.close()


def test_join_watermark_emission():
    # hash_join.rs:3667-3737 (test_streaming_hash_join_watermark): watermarks
    # buffer per side; the min across sides is emitted when it advances, for
    # the update side's output column first, then the match side's
    j = ffi.HashJoin(
        oracle(), JOIN_INNER, I2, I2, key_l=[0], key_r=[0], pk_l=[1], pk_r=[1],
        cond=(CMP_LT, 1, 3), wm_jk=[(0, True)],
    )
    assert j.watermark(SIDE_LEFT, 0, 100) == []
    assert j.watermark(SIDE_LEFT, 0, 200) == []
    # right 50 -> selected min(200,50)=50; update side (right) col 0 maps to
    # output 2, then left col 0 -> output 0
    assert j.watermark(SIDE_RIGHT, 0, 50) == [(2, 50), (0, 50)]
    assert j.watermark(SIDE_RIGHT, 0, 100) == [(2, 100), (0, 100)]
    j.close()


def test_join_watermark_state_cleaning():
    # rows below the selected watermark are cleaned from both sides and no
    # longer match (hash_join.rs:843-848 -> update_watermark TTL)
    j = ffi.HashJoin(oracle(), JOIN_INNER, I2, I2, key_l=[0], key_r=[0],
                     pk_l=[1], pk_r=[1], wm_jk=[(0, True)])
    assert push(j, SIDE_LEFT, " I I\n + 2 1\n + 6 2") == []
    j.watermark(SIDE_LEFT, 0, 5)
    j.watermark(SIDE_RIGHT, 0, 5)  # selected = 5: left row (2,1) cleaned
    assert push(j, SIDE_RIGHT, " I I\n + 2 10\n + 6 11") == rows([("+", 6, 2, 6, 11)])
    j.close()


def test_right_anti_join():
    # hash_join.rs:2650-2777 (LeftAnti with swapped senders)
    j = classical(ffi.JOIN_RIGHT_ANTI)
    assert push(j, SIDE_RIGHT, " I I\n + 1 4\n + 2 5\n + 3 6") == rows(
        [("+", 1, 4), ("+", 2, 5), ("+", 3, 6)]
    )
    assert push(j, SIDE_RIGHT, " I I\n + 3 8\n - 3 8") == []
    assert push(j, SIDE_LEFT, " I I\n + 2 7\n + 4 8\n + 6 9") == rows([("-", 2, 5)])
    assert push(j, SIDE_LEFT, " I I\n + 3 10\n + 6 11\n + 1 2\n + 1 3") == rows(
        [("-", 3, 6), ("-", 1, 4)]
    )
    assert push(j, SIDE_RIGHT, " I I\n + 9 10") == rows([("+", 9, 10)])
    assert push(j, SIDE_LEFT, " I I\n - 1 2") == []
    assert push(j, SIDE_LEFT, " I I\n - 1 3") == rows([("+", 1, 4)])
    j.close()


def test_left_outer_join_append_only():
    # hash_join.rs:3206-3292
    j = append_only(JOIN_LEFT_OUTER)
    assert push(j, SIDE_LEFT, " I I I\n + 1 4 1\n + 2 5 2\n + 3 6 3") == [
        ("+", (1, 4, 1, None, None, None)),
        ("+", (2, 5, 2, None, None, None)),
        ("+", (3, 6, 3, None, None, None)),
    ]
    assert push(j, SIDE_LEFT, " I I I\n + 4 9 4\n + 5 10 5") == [
        ("+", (4, 9, 4, None, None, None)),
        ("+", (5, 10, 5, None, None, None)),
    ]
    assert push(j, SIDE_RIGHT, " I I I\n + 2 5 1\n + 4 9 2\n + 6 9 3") == [
        ("-", (2, 5, 2, None, None, None)),
        ("+", (2, 5, 2, 2, 5, 1)),
        ("-", (4, 9, 4, None, None, None)),
        ("+", (4, 9, 4, 4, 9, 2)),
    ]
    assert push(j, SIDE_RIGHT, " I I I\n + 1 4 4\n + 3 6 5") == [
        ("-", (1, 4, 1, None, None, None)),
        ("+", (1, 4, 1, 1, 4, 4)),
        ("-", (3, 6, 3, None, None, None)),
        ("+", (3, 6, 3, 3, 6, 5)),
    ]
    j.close()


def test_right_outer_join_append_only():
    # hash_join.rs:3295-3363
    j = append_only(JOIN_RIGHT_OUTER)
    assert push(j, SIDE_LEFT, " I I I\n + 1 4 1\n + 2 5 2\n + 3 6 3") == []
    assert push(j, SIDE_LEFT, " I I I\n + 4 9 4\n + 5 10 5") == []
    assert push(j, SIDE_RIGHT, " I I I\n + 2 5 1\n + 4 9 2\n + 6 9 3") == [
        ("+", (2, 5, 2, 2, 5, 1)),
        ("+", (4, 9, 4, 4, 9, 2)),
        ("+", (None, None, None, 6, 9, 3)),
    ]
    assert push(j, SIDE_RIGHT, " I I I\n + 1 4 4\n + 3 6 5\n + 7 7 6") == [
        ("+", (1, 4, 1, 1, 4, 4)),
        ("+", (3, 6, 3, 3, 6, 5)),
        ("+", (None, None, None, 7, 7, 6)),
    ]
    j.close()


def test_null_safe_left_semi_join():
    # hash_join.rs test_streaming_null_safe_hash_left_semi_join
    j = classical(JOIN_LEFT_SEMI, null_safe=True)
    assert push(j, SIDE_LEFT, " I I\n + 1 4\n + 2 5\n + . 6") == []
    assert push(j, SIDE_LEFT, " I I\n + . 8\n - . 8") == []
    assert push(j, SIDE_RIGHT, " I I\n + 2 7\n + 4 8\n + 6 9") == rows([("+", 2, 5)])
    assert push(j, SIDE_RIGHT, " I I\n + . 10\n + 6 11") == rows([("+", None, 6)])
    assert push(j, SIDE_LEFT, " I I\n + 6 10") == rows([("+", 6, 10)])
    assert push(j, SIDE_RIGHT, " I I\n - 6 11") == []
    assert push(j, SIDE_RIGHT, " I I\n - 6 9") == rows([("-", 6, 10)])
    j.close()


def test_right_semi_join_append_only():
    # hash_join.rs test_streaming_hash_right_semi_join_append_only
    j = append_only(JOIN_RIGHT_SEMI)
    assert push(j, SIDE_LEFT, " I I I\n + 1 4 1\n + 2 5 2\n + 3 6 3") == []
    assert push(j, SIDE_LEFT, " I I I\n + 4 9 4\n + 5 10 5") == []
    assert push(j, SIDE_RIGHT, " I I I\n + 2 5 1\n + 4 9 2\n + 6 9 3") == rows(
        [("+", 2, 5, 1), ("+", 4, 9, 2)])
    assert push(j, SIDE_RIGHT, " I I I\n + 1 4 4\n + 3 6 5") == rows(
        [("+", 1, 4, 4), ("+", 3, 6, 5)])
    j.close()


def test_watermark_reference_fixture():
    # VERBATIM transcription of test_streaming_hash_join_watermark
    # (src/stream/src/executor/hash_join.rs:3667-3737): watermarks on the
    # join-key column of both sides; the executor emits the advanced MIN
    # for the update side's concat column first, then the match side's
    # (hash_join.rs:852-866). Inner Key64 join, state cleaning enabled
    # (`vec![(0, true)]` in the fixture).
    from rwtest.ffi import JOIN_INNER, SIDE_LEFT, SIDE_RIGHT

    o = ffi.HashJoin(oracle(), JOIN_INNER, [T_I64, T_I64], [T_I64, T_I64],
                     key_l=[0], key_r=[0], pk_l=[1], pk_r=[1],
                     wm_jk=[(0, True)])
    assert o.watermark(SIDE_LEFT, 0, 100) == []
    assert o.watermark(SIDE_LEFT, 0, 200) == []
    assert o.watermark(SIDE_RIGHT, 0, 50) == [(2, 50), (0, 50)]
    assert o.watermark(SIDE_RIGHT, 0, 100) == [(2, 100), (0, 100)]
    o.close()


def test_inequality_join_watermark():
    # hash_join.rs:1740-1829 test_inequality_join_watermark: condition
    # left.col1 >= right.col1 with InequalityPairInfo{left_idx:1,
    # right_idx:1, clean_left_state} — value-column watermarks buffer per
    # side, the min emits for the LARGER (left) side's output column, and
    # the left rows below it are state-cleaned
    from rwtest.ffi import CMP_GE

    j = ffi.HashJoin(oracle(), JOIN_INNER, I2, I2, key_l=[0], key_r=[0],
                     pk_l=[1], pk_r=[1], cond=(CMP_GE, 1, 3),
                     wm_ineq=((1, 1, True, True),))
    assert push(j, SIDE_LEFT, " I I\n + 2 4\n + 2 7\n + 3 8") == []
    assert j.watermark(SIDE_LEFT, 1, 10) == []
    # min(10, 6) = 6 emitted on the left col1 output position
    assert j.watermark(SIDE_RIGHT, 1, 6) == [(1, 6)]
    # (2,4) cleaned (4 < 6); (2,7) and (3,8) remain
    assert push(j, SIDE_RIGHT, " I I\n + 2 6") == rows([("+", 2, 7, 2, 6)])
    assert push(j, SIDE_RIGHT, " I I\n + 2 3") == rows([("+", 2, 7, 2, 3)])
    j.close()


def test_inequality_join_watermark_right_larger():
    # the mirror of hash_join.rs:1740-1829: condition left.col1 <=
    # right.col1 (right side larger, clean_right_state) — the min emits
    # for the RIGHT side's output column and cleans right rows below it
    from rwtest.ffi import CMP_LE

    j = ffi.HashJoin(oracle(), JOIN_INNER, I2, I2, key_l=[0], key_r=[0],
                     pk_l=[1], pk_r=[1], cond=(CMP_LE, 1, 3),
                     wm_ineq=((1, 1, False, True),))
    assert push(j, SIDE_RIGHT, " I I\n + 2 4\n + 2 7\n + 3 8") == []
    # arrival order right-then-left also buffers correctly
    assert j.watermark(SIDE_RIGHT, 1, 10) == []
    assert j.watermark(SIDE_LEFT, 1, 6) == [(3, 6)]
    # right rows with col1 < 6 cleaned: (2,4) gone, (2,7)/(3,8) remain
    assert push(j, SIDE_LEFT, " I I\n + 2 6") == rows([("+", 2, 6, 2, 7)])
    # a stale watermark (not advancing the min) emits nothing
    assert j.watermark(SIDE_LEFT, 1, 5) == []
    assert push(j, SIDE_LEFT, " I I\n + 2 3") == rows([("+", 2, 3, 2, 7)])
    j.close()
