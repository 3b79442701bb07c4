"""ctypes bindings for the rw_stream C-ABI (include/rw_stream.h).

Drives either the CPU oracle (oracle/liboracle.so — test infrastructure) or
the product library (risingwave_amd/librw_amd.so — the MI355X path). The two
export identical symbols; tests load each via its own CDLL handle.
"""
import ctypes as C
import os

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

# type ids (rw_chunk.h)
T_I64, T_I32, T_F64, T_F32, T_BOOL, T_TS = 0, 1, 2, 3, 4, 5
T_DECIMAL = 6
OP_INSERT, OP_DELETE, OP_UPDATE_DELETE, OP_UPDATE_INSERT = 0, 1, 2, 3
OP_BY_TOKEN = {"+": OP_INSERT, "-": OP_DELETE, "U-": OP_UPDATE_DELETE, "U+": OP_UPDATE_INSERT}
TOKEN_BY_OP = {v: k for k, v in OP_BY_TOKEN.items()}

NP_BY_TYPE = {
    T_I64: np.int64,
    T_I32: np.int32,
    T_F64: np.float64,
    T_F32: np.float32,
    T_BOOL: np.uint8,
    T_TS: np.int64,
    T_DECIMAL: np.dtype("V16"),  # raw rust_decimal serialize image
}

AGG_COUNT_STAR, AGG_COUNT, AGG_SUM, AGG_SUM0, AGG_MIN, AGG_MAX = 0, 1, 2, 3, 4, 5

JOIN_INNER, JOIN_LEFT_OUTER, JOIN_RIGHT_OUTER, JOIN_FULL_OUTER = 0, 1, 2, 3
JOIN_LEFT_SEMI, JOIN_LEFT_ANTI, JOIN_RIGHT_SEMI, JOIN_RIGHT_ANTI = 4, 5, 6, 7
CMP_LT, CMP_LE, CMP_GT, CMP_GE = 0, 1, 2, 3
SIDE_LEFT, SIDE_RIGHT = 0, 1


class RwColumn(C.Structure):
    _fields_ = [("type", C.c_uint8), ("valid", C.c_void_p), ("data", C.c_void_p)]


class RwChunkC(C.Structure):
    _fields_ = [
        ("n_rows", C.c_uint32),
        ("n_cols", C.c_uint32),
        ("ops", C.POINTER(C.c_uint8)),
        ("vis", C.POINTER(C.c_uint8)),
        ("cols", C.POINTER(RwColumn)),
    ]


class RwAggCall(C.Structure):
    _fields_ = [("kind", C.c_uint8), ("arg", C.c_int32), ("ret_type", C.c_uint8),
                ("distinct", C.c_uint8)]


class RwHashAggDesc(C.Structure):
    _fields_ = [
        ("n_input_cols", C.c_uint32),
        ("input_types", C.POINTER(C.c_uint8)),
        ("n_group_key", C.c_uint32),
        ("group_key_indices", C.POINTER(C.c_uint32)),
        ("n_calls", C.c_uint32),
        ("calls", C.POINTER(RwAggCall)),
        ("row_count_index", C.c_uint32),
        ("n_stream_key", C.c_uint32),
        ("stream_key", C.POINTER(C.c_uint32)),
        ("chunk_size", C.c_uint32),
        ("append_only", C.c_uint8),
        ("emit_on_window_close", C.c_uint8),
        ("state_capacity_hint", C.c_uint64),
    ]


class RwHashJoinDesc(C.Structure):
    _fields_ = [
        ("join_type", C.c_uint8),
        ("append_only", C.c_uint8),
        ("n_key", C.c_uint32),
        ("key_l", C.POINTER(C.c_uint32)),
        ("key_r", C.POINTER(C.c_uint32)),
        ("null_safe", C.POINTER(C.c_uint8)),
        ("n_cols_l", C.c_uint32),
        ("types_l", C.POINTER(C.c_uint8)),
        ("n_cols_r", C.c_uint32),
        ("types_r", C.POINTER(C.c_uint8)),
        ("n_pk_l", C.c_uint32),
        ("pk_l", C.POINTER(C.c_uint32)),
        ("n_pk_r", C.c_uint32),
        ("pk_r", C.POINTER(C.c_uint32)),
        ("n_output", C.c_uint32),
        ("output_indices", C.POINTER(C.c_uint32)),
        ("has_cond", C.c_uint8),
        ("cond_op", C.c_uint8),
        ("cond_l", C.c_uint32),
        ("cond_r", C.c_uint32),
        ("cond_rconst", C.c_int64),
        ("has_cond2", C.c_uint8),
        ("cond2_op", C.c_uint8),
        ("cond2_l", C.c_uint32),
        ("cond2_r", C.c_uint32),
        ("cond2_rconst", C.c_int64),
        ("chunk_size", C.c_uint32),
        ("state_capacity_hint", C.c_uint64),
        ("row_capacity_hint", C.c_uint64),
        ("n_wm_jk", C.c_uint32),
        ("wm_jk_pos", C.POINTER(C.c_uint32)),
        ("wm_jk_clean", C.POINTER(C.c_uint8)),
        ("n_ineq", C.c_uint32),
        ("ineq_left_col", C.POINTER(C.c_uint32)),
        ("ineq_right_col", C.POINTER(C.c_uint32)),
        ("ineq_left_larger", C.POINTER(C.c_uint8)),
        ("ineq_clean", C.POINTER(C.c_uint8)),
    ]


def _u32arr(vals):
    return (C.c_uint32 * len(vals))(*vals)


def _u8arr(vals):
    return (C.c_uint8 * len(vals))(*vals)


class Chunk:
    """Python-side chunk: ops list, vis list (or None), typed numpy columns."""

    def __init__(self, types, ops, cols, valids, vis=None):
        self.types = list(types)
        self.ops = np.asarray(ops, dtype=np.uint8)
        self.cols = [np.ascontiguousarray(c, dtype=NP_BY_TYPE[t]) for c, t in zip(cols, types)]
        self.valids = [np.ascontiguousarray(v, dtype=np.uint8) for v in valids]
        self.vis = None if vis is None else np.asarray(vis, dtype=np.uint8)

    @property
    def n_rows(self):
        return len(self.ops)

    def to_c(self):
        n = self.n_rows
        cols = (RwColumn * len(self.cols))()
        self._keep = []
        for i, (t, d, v) in enumerate(zip(self.types, self.cols, self.valids)):
            cols[i].type = t
            cols[i].valid = v.ctypes.data
            cols[i].data = d.ctypes.data
            self._keep.extend([d, v])
        ch = RwChunkC()
        ch.n_rows = n
        ch.n_cols = len(self.cols)
        ops = self.ops
        ch.ops = C.cast(ops.ctypes.data, C.POINTER(C.c_uint8))
        if self.vis is not None:
            ch.vis = C.cast(self.vis.ctypes.data, C.POINTER(C.c_uint8))
        else:
            ch.vis = None
        ch.cols = cols
        self._keep.extend([ops, cols, self.vis])
        return ch

    def visible_rows(self):
        """Yield (op_token, row-tuple) for visible rows; NULL -> None."""
        for r in range(self.n_rows):
            if self.vis is not None and not self.vis[r]:
                continue
            vals = []
            for t, d, v in zip(self.types, self.cols, self.valids):
                if not v[r]:
                    vals.append(None)
                elif t == T_DECIMAL:
                    vals.append(bytes(d[r]))
                elif t in (T_F64, T_F32):
                    vals.append(float(d[r]))
                else:
                    vals.append(int(d[r]))
            yield (TOKEN_BY_OP[int(self.ops[r])], tuple(vals))


TYPE_BY_TOKEN = {"I": T_I64, "i": T_I32, "F": T_F64, "f": T_F32, "B": T_BOOL, "TS": T_TS}


def from_pretty(s):
    """Parse the reference test fixture format
    (common/src/array/data_chunk.rs:646-760 / stream_chunk.rs from_pretty):
    header of type tokens, then rows `<op> v v v [D]`, `.` = NULL."""
    lines = [l.strip() for l in s.strip().split("\n") if l.strip()]
    types = [TYPE_BY_TOKEN[tok] for tok in lines[0].split()]
    ops, vis = [], []
    cols = [[] for _ in types]
    valids = [[] for _ in types]
    for line in lines[1:]:
        toks = line.split()
        op = toks[0]
        if op in ("U-", "U+", "+", "-"):
            ops.append(OP_BY_TOKEN[op])
            toks = toks[1:]
        else:  # data-chunk style: all inserts
            ops.append(OP_INSERT)
        row_vis = 1
        if toks and toks[-1] == "D":
            row_vis = 0
            toks = toks[:-1]
        vis.append(row_vis)
        assert len(toks) == len(types), f"bad row: {line}"
        for i, tok in enumerate(toks):
            if tok == ".":
                valids[i].append(0)
                cols[i].append(0)
            else:
                valids[i].append(1)
                cols[i].append(float(tok) if types[i] in (T_F64, T_F32) else int(tok))
    if all(v for v in vis):
        vis = None
    return Chunk(types, ops, cols, valids, vis)


class Lib:
    """Wrapper over one .so implementing the rw_stream C-ABI."""

    def __init__(self, path):
        self.lib = C.CDLL(path)
        L = self.lib
        L.rw_hash_agg_create.restype = C.c_void_p
        L.rw_hash_agg_create.argtypes = [C.POINTER(RwHashAggDesc)]
        L.rw_hash_agg_push_chunk.restype = C.c_int
        L.rw_hash_agg_push_chunk.argtypes = [C.c_void_p, C.POINTER(RwChunkC)]
        L.rw_hash_agg_flush.restype = C.c_int
        L.rw_hash_agg_flush.argtypes = [C.c_void_p, C.c_uint64]
        L.rw_hash_agg_poll.restype = C.POINTER(RwChunkC)
        L.rw_hash_agg_poll.argtypes = [C.c_void_p]
        L.rw_hash_agg_destroy.argtypes = [C.c_void_p]
        L.rw_hash_join_create.restype = C.c_void_p
        L.rw_hash_join_create.argtypes = [C.POINTER(RwHashJoinDesc)]
        L.rw_hash_join_push_chunk.restype = C.c_int
        L.rw_hash_join_push_chunk.argtypes = [C.c_void_p, C.c_int, C.POINTER(RwChunkC)]
        L.rw_hash_join_flush.restype = C.c_int
        L.rw_hash_join_flush.argtypes = [C.c_void_p, C.c_uint64]
        L.rw_hash_join_poll.restype = C.POINTER(RwChunkC)
        L.rw_hash_join_poll.argtypes = [C.c_void_p]
        L.rw_hash_join_destroy.argtypes = [C.c_void_p]
        L.rw_chunk_free.argtypes = [C.POINTER(RwChunkC)]
        L.rw_last_error.restype = C.c_char_p
        try:
            L.rw_hash_join_watermark.restype = C.c_int
            L.rw_hash_join_watermark.argtypes = [
                C.c_void_p, C.c_int, C.c_uint32, C.c_int64,
                C.POINTER(C.c_uint32), C.POINTER(C.c_int64), C.c_int]
            L.rw_hash_agg_watermark.restype = C.c_int
            L.rw_hash_agg_watermark.argtypes = [C.c_void_p, C.c_uint32, C.c_int64]
            L.rw_hash_agg_update_vnode_bitmap.restype = C.c_int
            L.rw_hash_agg_update_vnode_bitmap.argtypes = [
                C.c_void_p, C.POINTER(C.c_uint8), C.c_uint32]
            L.rw_hash_join_update_vnode_bitmap.restype = C.c_int
            L.rw_hash_join_update_vnode_bitmap.argtypes = [
                C.c_void_p, C.POINTER(C.c_uint8), C.c_uint32]
        except AttributeError:
            pass

    def last_error(self):
        return self.lib.rw_last_error().decode()

    def _read_chunk(self, p):
        ch = p.contents
        n, m = ch.n_rows, ch.n_cols
        types, cols, valids = [], [], []
        for ci in range(m):
            col = ch.cols[ci]
            t = col.type
            types.append(t)
            dt = NP_BY_TYPE[t]
            nbytes = n * np.dtype(dt).itemsize
            data = np.frombuffer(C.string_at(col.data, nbytes), dtype=dt).copy() if n else np.array([], dt)
            valid = np.frombuffer(C.string_at(col.valid, n), dtype=np.uint8).copy() if n else np.array([], np.uint8)
            cols.append(data)
            valids.append(valid)
        ops = np.frombuffer(C.string_at(ch.ops, n), dtype=np.uint8).copy() if n else np.array([], np.uint8)
        vis = None
        if ch.vis:
            vis = np.frombuffer(C.string_at(ch.vis, n), dtype=np.uint8).copy()
        self.lib.rw_chunk_free(p)
        return Chunk(types, ops, cols, valids, vis)


class HashAgg:
    def __init__(self, lib: Lib, input_types, group_key, calls, row_count_index,
                 stream_key=(), chunk_size=1024, append_only=False,
                 emit_on_window_close=False, state_capacity_hint=0):
        """calls: list of (kind, arg, ret_type)."""
        self.lib = lib
        d = RwHashAggDesc()
        d.n_input_cols = len(input_types)
        self._it = _u8arr(input_types)
        d.input_types = self._it
        d.n_group_key = len(group_key)
        self._gk = _u32arr(group_key)
        d.group_key_indices = self._gk
        d.n_calls = len(calls)
        self._calls = (RwAggCall * len(calls))()
        for i, call in enumerate(calls):
            # (kind, arg, ret_type[, distinct])
            self._calls[i].kind = call[0]
            self._calls[i].arg = call[1]
            self._calls[i].ret_type = call[2]
            self._calls[i].distinct = call[3] if len(call) > 3 else 0
        d.calls = self._calls
        d.row_count_index = row_count_index
        d.n_stream_key = len(stream_key)
        self._sk = _u32arr(stream_key)
        d.stream_key = self._sk
        d.chunk_size = chunk_size
        d.append_only = 1 if append_only else 0
        d.emit_on_window_close = 1 if emit_on_window_close else 0
        d.state_capacity_hint = state_capacity_hint
        self.h = lib.lib.rw_hash_agg_create(C.byref(d))
        if not self.h:
            raise RuntimeError(f"rw_hash_agg_create failed: {lib.last_error()}")

    def push(self, chunk: Chunk):
        c = chunk.to_c()
        rc = self.lib.lib.rw_hash_agg_push_chunk(self.h, C.byref(c))
        if rc != 0:
            raise RuntimeError(f"push failed {rc}: {self.lib.last_error()}")

    def flush(self, epoch):
        rc = self.lib.lib.rw_hash_agg_flush(self.h, epoch)
        if rc != 0:
            raise RuntimeError(f"flush failed {rc}: {self.lib.last_error()}")

    def poll_all(self):
        out = []
        while True:
            p = self.lib.lib.rw_hash_agg_poll(self.h)
            if not p:
                return out
            out.append(self.lib._read_chunk(p))

    def watermark(self, group_key_pos, val):
        rc = self.lib.lib.rw_hash_agg_watermark(self.h, group_key_pos, val)
        if rc != 0:
            raise RuntimeError(f"agg watermark failed {rc}")

    def update_vnode_bitmap(self, bitmap_bytes, vnode_count=256):
        buf = (C.c_uint8 * len(bitmap_bytes))(*bitmap_bytes)
        rc = self.lib.lib.rw_hash_agg_update_vnode_bitmap(self.h, buf,
                                                          vnode_count)
        if rc != 0:
            raise RuntimeError(f"agg vnode bitmap failed {rc}")

    def close(self):
        if self.h:
            self.lib.lib.rw_hash_agg_destroy(self.h)
            self.h = None


class HashJoin:
    def __init__(self, lib: Lib, join_type, types_l, types_r, key_l, key_r,
                 pk_l, pk_r, output_indices=None, null_safe=None, cond=None,
                 cond2=None,
                 chunk_size=1024, append_only=False, state_capacity_hint=0,
                 row_capacity_hint=0, wm_jk=(), wm_ineq=()):
        """cond: (op, cond_l, cond_r) into the concatenated row, or None.
        wm_ineq: (left_col, right_col, left_larger, clean) tuples (the
        reference's InequalityPairInfo)."""
        self.lib = lib
        d = RwHashJoinDesc()
        d.join_type = join_type
        d.append_only = 1 if append_only else 0
        d.n_key = len(key_l)
        self._kl = _u32arr(key_l)
        self._kr = _u32arr(key_r)
        d.key_l, d.key_r = self._kl, self._kr
        ns = null_safe or [0] * len(key_l)
        self._ns = _u8arr(ns)
        d.null_safe = self._ns
        d.n_cols_l = len(types_l)
        self._tl = _u8arr(types_l)
        d.types_l = self._tl
        d.n_cols_r = len(types_r)
        self._tr = _u8arr(types_r)
        d.types_r = self._tr
        d.n_pk_l = len(pk_l)
        self._pl = _u32arr(pk_l)
        d.pk_l = self._pl
        d.n_pk_r = len(pk_r)
        self._pr = _u32arr(pk_r)
        d.pk_r = self._pr
        if output_indices is None:
            if join_type in (JOIN_LEFT_SEMI, JOIN_LEFT_ANTI):
                output_indices = list(range(len(types_l)))
            elif join_type in (JOIN_RIGHT_SEMI, JOIN_RIGHT_ANTI):
                output_indices = list(range(len(types_l), len(types_l) + len(types_r)))
            else:
                output_indices = list(range(len(types_l) + len(types_r)))
        d.n_output = len(output_indices)
        self._oi = _u32arr(output_indices)
        d.output_indices = self._oi
        if cond is not None:
            d.has_cond = 1
            if len(cond) == 3:
                d.cond_op, d.cond_l, d.cond_r = cond
            else:
                d.cond_op, d.cond_l, d.cond_r, d.cond_rconst = cond
        else:
            d.has_cond = 0
            d.cond_op = d.cond_l = d.cond_r = 0
        if cond2 is not None:
            d.has_cond2 = 1
            if len(cond2) == 3:
                d.cond2_op, d.cond2_l, d.cond2_r = cond2
            else:
                d.cond2_op, d.cond2_l, d.cond2_r, d.cond2_rconst = cond2
        d.chunk_size = chunk_size
        d.state_capacity_hint = state_capacity_hint
        d.row_capacity_hint = row_capacity_hint
        d.n_wm_jk = len(wm_jk)
        self._wp = _u32arr([p for p, _ in wm_jk])
        self._wc = _u8arr([1 if c else 0 for _, c in wm_jk])
        d.wm_jk_pos = self._wp
        d.wm_jk_clean = self._wc
        d.n_ineq = len(wm_ineq)
        self._il = _u32arr([l for l, _, _, _ in wm_ineq])
        self._ir = _u32arr([r for _, r, _, _ in wm_ineq])
        self._ig = _u8arr([1 if g else 0 for _, _, g, _ in wm_ineq])
        self._ic = _u8arr([1 if c2 else 0 for _, _, _, c2 in wm_ineq])
        d.ineq_left_col = self._il
        d.ineq_right_col = self._ir
        d.ineq_left_larger = self._ig
        d.ineq_clean = self._ic
        self.h = lib.lib.rw_hash_join_create(C.byref(d))
        if not self.h:
            raise RuntimeError(f"rw_hash_join_create failed: {lib.last_error()}")

    def push(self, side, chunk: Chunk):
        c = chunk.to_c()
        rc = self.lib.lib.rw_hash_join_push_chunk(self.h, side, C.byref(c))
        if rc != 0:
            raise RuntimeError(f"push failed {rc}: {self.lib.last_error()}")

    def flush(self, epoch):
        rc = self.lib.lib.rw_hash_join_flush(self.h, epoch)
        if rc != 0:
            raise RuntimeError(f"flush failed {rc}: {self.lib.last_error()}")

    def poll_all(self):
        out = []
        while True:
            p = self.lib.lib.rw_hash_join_poll(self.h)
            if not p:
                return out
            out.append(self.lib._read_chunk(p))

    def watermark(self, side, col_idx, val, max_out=16):
        cols = (C.c_uint32 * max_out)()
        vals = (C.c_int64 * max_out)()
        n = self.lib.lib.rw_hash_join_watermark(self.h, side, col_idx, val,
                                                cols, vals, max_out)
        if n < 0:
            raise RuntimeError(f"watermark failed {n}: {self.lib.last_error()}")
        return [(int(cols[i]), int(vals[i])) for i in range(n)]

    def update_vnode_bitmap(self, bitmap_bytes, vnode_count=256):
        buf = (C.c_uint8 * len(bitmap_bytes))(*bitmap_bytes)
        rc = self.lib.lib.rw_hash_join_update_vnode_bitmap(self.h, buf,
                                                           vnode_count)
        if rc != 0:
            raise RuntimeError(f"join vnode bitmap failed {rc}")

    def close(self):
        if self.h:
            self.lib.lib.rw_hash_join_destroy(self.h)
            self.h = None


def rows_multiset(chunks):
    """All visible (op, row) across chunks, sorted — the per-epoch multiset
    (parity bar of SURVEY.md: tests/integration_tests/snapshot.rs sorts too)."""
    rows = []
    for c in chunks:
        rows.extend(c.visible_rows())
    key = lambda r: (r[0], tuple((v is None, v if v is not None else 0) for v in r[1]))
    return sorted(rows, key=key)


def rows_ordered(chunks):
    """All visible (op, row) across chunks, in yield order (order-exact tests)."""
    rows = []
    for c in chunks:
        rows.extend(c.visible_rows())
    return rows


_oracle = None


def oracle():
    global _oracle
    if _oracle is None:
        path = os.path.join(REPO, "oracle", "liboracle.so")
        if not os.path.exists(path):
            import subprocess

            subprocess.run(["make", "-C", os.path.join(REPO, "oracle")], check=True)
        _oracle = Lib(path)
    return _oracle


class RwGroupTopNDesc(C.Structure):
    _fields_ = [
        ("n_cols", C.c_uint32), ("types", C.POINTER(C.c_uint8)),
        ("n_group_by", C.c_uint32), ("group_by", C.POINTER(C.c_uint32)),
        ("n_order_by", C.c_uint32), ("order_cols", C.POINTER(C.c_uint32)),
        ("order_desc", C.POINTER(C.c_uint8)),
        ("n_rest", C.c_uint32), ("rest_cols", C.POINTER(C.c_uint32)),
        ("rest_desc", C.POINTER(C.c_uint8)),
        ("offset", C.c_uint64), ("limit", C.c_uint64),
        ("with_ties", C.c_uint8),
        ("chunk_size", C.c_uint32),
        ("state_capacity_hint", C.c_uint64), ("row_capacity_hint", C.c_uint64),
    ]


class GroupTopN:
    """GroupTopN executor wrapper (group_top_n.rs; WITH_TIES=false).

    order_by / rest: lists of (col_idx, desc) pairs; rest = the remaining
    storage-key columns after group_by and order_by."""

    def __init__(self, lib: Lib, input_types, group_by, order_by, rest,
                 offset=0, limit=1, with_ties=False, chunk_size=1024,
                 state_capacity_hint=0, row_capacity_hint=0):
        self.lib = lib
        L = lib.lib
        L.rw_group_top_n_create.restype = C.c_void_p
        L.rw_group_top_n_create.argtypes = [C.POINTER(RwGroupTopNDesc)]
        L.rw_group_top_n_push_chunk.restype = C.c_int
        L.rw_group_top_n_push_chunk.argtypes = [C.c_void_p,
                                                C.POINTER(RwChunkC)]
        L.rw_group_top_n_flush.restype = C.c_int
        L.rw_group_top_n_flush.argtypes = [C.c_void_p, C.c_uint64]
        L.rw_group_top_n_poll.restype = C.POINTER(RwChunkC)
        L.rw_group_top_n_poll.argtypes = [C.c_void_p]
        L.rw_group_top_n_destroy.restype = None
        L.rw_group_top_n_destroy.argtypes = [C.c_void_p]
        d = RwGroupTopNDesc()
        self._keep = []

        def u32s(v):
            a = (C.c_uint32 * max(len(v), 1))(*v)
            self._keep.append(a)
            return a

        def u8s(v):
            a = (C.c_uint8 * max(len(v), 1))(*v)
            self._keep.append(a)
            return a

        d.n_cols = len(input_types)
        d.types = u8s(input_types)
        d.n_group_by = len(group_by)
        d.group_by = u32s(group_by)
        d.n_order_by = len(order_by)
        d.order_cols = u32s([c for c, _ in order_by])
        d.order_desc = u8s([1 if desc else 0 for _, desc in order_by])
        d.n_rest = len(rest)
        d.rest_cols = u32s([c for c, _ in rest])
        d.rest_desc = u8s([1 if desc else 0 for _, desc in rest])
        d.offset = offset
        d.limit = limit
        d.with_ties = 1 if with_ties else 0
        d.chunk_size = chunk_size
        d.state_capacity_hint = state_capacity_hint
        d.row_capacity_hint = row_capacity_hint
        self.h = L.rw_group_top_n_create(C.byref(d))
        if not self.h:
            raise RuntimeError(f"group_top_n create failed: {lib.last_error()}")

    def push(self, chunk):
        cc = chunk.to_c()
        rc = self.lib.lib.rw_group_top_n_push_chunk(self.h, C.byref(cc))
        if rc != 0:
            raise RuntimeError(f"push failed {rc}: {self.lib.last_error()}")

    def watermark(self, col_idx, val):
        L = self.lib.lib
        L.rw_group_top_n_watermark.restype = C.c_int
        L.rw_group_top_n_watermark.argtypes = [C.c_void_p, C.c_uint32,
                                               C.c_int64]
        rc = L.rw_group_top_n_watermark(self.h, col_idx, val)
        if rc < 0:
            raise RuntimeError(
                f"topn watermark failed {rc}: {self.lib.last_error()}")
        return rc  # 1 = forwarded, 0 = absorbed

    def flush(self, epoch):
        rc = self.lib.lib.rw_group_top_n_flush(self.h, epoch)
        if rc != 0:
            raise RuntimeError(f"flush failed {rc}: {self.lib.last_error()}")

    def poll_all(self):
        out = []
        while True:
            p = self.lib.lib.rw_group_top_n_poll(self.h)
            if not p:
                return out
            out.append(self.lib._read_chunk(p))

    def close(self):
        if self.h:
            self.lib.lib.rw_group_top_n_destroy(self.h)
            self.h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


def join_degree_drain(lib, h, side):
    """Collect one side's degree-table records (computed by the preceding
    join_checkpoint_drain call for that side); returns bytes."""
    L = lib.lib
    L.rw_join_degree_drain.restype = C.c_int
    L.rw_join_degree_drain.argtypes = [C.c_void_p, C.c_int,
                                       C.POINTER(C.POINTER(C.c_uint8)),
                                       C.POINTER(C.c_uint64)]
    buf = C.POINTER(C.c_uint8)()
    ln = C.c_uint64()
    rc = L.rw_join_degree_drain(h, side, C.byref(buf), C.byref(ln))
    if rc != 0:
        raise RuntimeError(f"degree drain failed {rc}: {lib.last_error()}")
    out = bytes(bytearray(buf[i] for i in range(ln.value)))
    L.rw_spill_free.argtypes = [C.c_void_p]
    L.rw_spill_free(C.cast(buf, C.c_void_p))
    return out


def join_checkpoint_drain(lib, h, side):
    """Drain one side's §8f-2 checkpoint spill buffer; returns bytes."""
    L = lib.lib
    L.rw_join_checkpoint_drain.restype = C.c_int
    L.rw_join_checkpoint_drain.argtypes = [C.c_void_p, C.c_int,
                                           C.POINTER(C.POINTER(C.c_uint8)),
                                           C.POINTER(C.c_uint64)]
    buf = C.POINTER(C.c_uint8)()
    ln = C.c_uint64()
    rc = L.rw_join_checkpoint_drain(h, side, C.byref(buf), C.byref(ln))
    if rc != 0:
        raise RuntimeError(f"join drain failed {rc}: {lib.last_error()}")
    out = bytes(bytearray(buf[i] for i in range(ln.value)))
    L.rw_spill_free.argtypes = [C.c_void_p]
    L.rw_spill_free(C.cast(buf, C.c_void_p))
    return out


def topn_checkpoint_drain(lib, h):
    """Drain the GroupTopN §8f-2 checkpoint spill buffer; returns bytes."""
    L = lib.lib
    L.rw_topn_checkpoint_drain.restype = C.c_int
    L.rw_topn_checkpoint_drain.argtypes = [C.c_void_p,
                                           C.POINTER(C.POINTER(C.c_uint8)),
                                           C.POINTER(C.c_uint64)]
    buf = C.POINTER(C.c_uint8)()
    ln = C.c_uint64()
    rc = L.rw_topn_checkpoint_drain(h, C.byref(buf), C.byref(ln))
    if rc != 0:
        raise RuntimeError(f"topn drain failed {rc}: {lib.last_error()}")
    out = bytes(bytearray(buf[i] for i in range(ln.value)))
    L.rw_spill_free.argtypes = [C.c_void_p]
    L.rw_spill_free(C.cast(buf, C.c_void_p))
    return out


def join_compact(lib, h, side):
    """Reclaim one join side's dead records; returns freed bytes."""
    L = lib.lib
    L.rw_join_compact.restype = C.c_int
    L.rw_join_compact.argtypes = [C.c_void_p, C.c_int,
                                  C.POINTER(C.c_uint64)]
    freed = C.c_uint64()
    rc = L.rw_join_compact(h, side, C.byref(freed))
    if rc != 0:
        raise RuntimeError(f"join compact failed {rc}: {lib.last_error()}")
    return freed.value


def _compact1(lib, h, sym):
    L = lib.lib
    fn = getattr(L, sym)
    fn.restype = C.c_int
    fn.argtypes = [C.c_void_p, C.POINTER(C.c_uint64)]
    freed = C.c_uint64()
    rc = fn(h, C.byref(freed))
    if rc != 0:
        raise RuntimeError(f"{sym} failed {rc}: {lib.last_error()}")
    return freed.value


def agg_minput_compact(lib, h):
    """Reclaim the agg's dead materialized-input rows; returns bytes."""
    return _compact1(lib, h, "rw_agg_minput_compact")


def topn_compact(lib, h):
    """Reclaim the GroupTopN store's dead records; returns bytes."""
    return _compact1(lib, h, "rw_topn_compact")


def topn_restore(lib, h, buf):
    """Rebuild GroupTopN state from concatenated drain bytes (rw_stream.h)."""
    L = lib.lib
    L.rw_topn_restore.restype = C.c_int
    L.rw_topn_restore.argtypes = [C.c_void_p, C.c_char_p, C.c_uint64]
    rc = L.rw_topn_restore(h, buf, len(buf))
    if rc != 0:
        raise RuntimeError(f"topn restore failed {rc}: {lib.last_error()}")


def agg_checkpoint_drain_bytes(lib, h):
    """Drain the agg's §8f-2 checkpoint spill buffer; returns raw bytes."""
    L = lib.lib
    L.rw_agg_checkpoint_drain.restype = C.c_int
    L.rw_agg_checkpoint_drain.argtypes = [C.c_void_p,
                                          C.POINTER(C.POINTER(C.c_uint8)),
                                          C.POINTER(C.c_uint64)]
    buf = C.POINTER(C.c_uint8)()
    ln = C.c_uint64()
    rc = L.rw_agg_checkpoint_drain(h, C.byref(buf), C.byref(ln))
    if rc != 0:
        raise RuntimeError(f"agg drain failed {rc}: {lib.last_error()}")
    out = bytes(bytearray(buf[i] for i in range(ln.value)))
    L.rw_spill_free.argtypes = [C.c_void_p]
    L.rw_spill_free(C.cast(buf, C.c_void_p))
    return out


def agg_restore(lib, h, buf):
    """Rebuild agg state from concatenated drain bytes (rw_stream.h)."""
    L = lib.lib
    L.rw_hash_agg_restore.restype = C.c_int
    L.rw_hash_agg_restore.argtypes = [C.c_void_p, C.c_char_p, C.c_uint64]
    rc = L.rw_hash_agg_restore(h, buf, len(buf))
    if rc != 0:
        raise RuntimeError(f"agg restore failed {rc}: {lib.last_error()}")


def join_restore(lib, h, side, buf, deg_buf=b""):
    """Rebuild one join side from concatenated drain bytes (rw_stream.h)."""
    L = lib.lib
    L.rw_hash_join_restore.restype = C.c_int
    L.rw_hash_join_restore.argtypes = [C.c_void_p, C.c_int, C.c_char_p,
                                       C.c_uint64, C.c_char_p, C.c_uint64]
    rc = L.rw_hash_join_restore(h, side, buf, len(buf),
                                deg_buf if deg_buf else None,
                                len(deg_buf) if deg_buf else 0)
    if rc != 0:
        raise RuntimeError(f"join restore failed {rc}: {lib.last_error()}")


# ---- decimal helpers (rust_decimal 1.40.0 serialize layout; see
# include/rw_chunk.h RW_T_DECIMAL) ----
import decimal as _pydec
import struct as _struct


def dec16(v):
    """16-byte rust_decimal image from a python Decimal / str / int.
    "NaN" / "Inf" / "-Inf" map to the reference's special encodings
    (types/decimal.rs:583-592)."""
    if isinstance(v, str) and v in ("NaN", "Inf", "-Inf"):
        b0 = {"NaN": 1, "Inf": 2, "-Inf": 3}[v]
        return bytes([b0]) + b"\x00" * 15
    d = _pydec.Decimal(v)
    sign, digits, exp = d.as_tuple()
    assert exp <= 0 and -exp <= 28, f"scale {-exp} out of range"
    m = int("".join(map(str, digits)))
    assert m < (1 << 96), "mantissa exceeds 96 bits"
    flags = ((-exp) << 16) | (0x80000000 if sign else 0)
    return _struct.pack("<IIII", flags, m & 0xFFFFFFFF,
                        (m >> 32) & 0xFFFFFFFF, (m >> 64) & 0xFFFFFFFF)


def dec16_value(b):
    """Python Decimal (or 'NaN'/'Inf'/'-Inf') from a 16-byte image."""
    if b[0] in (1, 2, 3):
        return {1: "NaN", 2: "Inf", 3: "-Inf"}[b[0]]
    flags, lo, mid, hi = _struct.unpack("<IIII", b)
    m = (hi << 64) | (mid << 32) | lo
    scale = (flags >> 16) & 0xFF
    sign = -1 if (flags >> 31) & 1 else 1
    return _pydec.Decimal(sign * m).scaleb(-scale)


def dec_col(values):
    """numpy V16 column from an iterable of dec16-able values."""
    raw = b"".join(dec16(v) for v in values)
    return np.frombuffer(raw, dtype="V16").copy()


def agg_minput_drain_bytes(lib, h, mi):
    """Drain one materialized-input state table; returns raw bytes."""
    L = lib.lib
    L.rw_agg_minput_drain.restype = C.c_int
    L.rw_agg_minput_drain.argtypes = [C.c_void_p, C.c_int,
                                      C.POINTER(C.POINTER(C.c_uint8)),
                                      C.POINTER(C.c_uint64)]
    buf = C.POINTER(C.c_uint8)()
    ln = C.c_uint64()
    rc = L.rw_agg_minput_drain(h, mi, C.byref(buf), C.byref(ln))
    if rc != 0:
        raise RuntimeError(f"minput drain failed {rc}: {lib.last_error()}")
    out = bytes(bytearray(buf[i] for i in range(ln.value)))
    L.rw_spill_free.argtypes = [C.c_void_p]
    L.rw_spill_free(C.cast(buf, C.c_void_p))
    return out


def agg_dedup_drain_bytes(lib, h, di):
    """Drain one DISTINCT dedup table's spill buffer; returns raw bytes."""
    L = lib.lib
    L.rw_agg_dedup_drain.restype = C.c_int
    L.rw_agg_dedup_drain.argtypes = [C.c_void_p, C.c_int,
                                     C.POINTER(C.POINTER(C.c_uint8)),
                                     C.POINTER(C.c_uint64)]
    buf = C.POINTER(C.c_uint8)()
    ln = C.c_uint64()
    rc = L.rw_agg_dedup_drain(h, di, C.byref(buf), C.byref(ln))
    if rc != 0:
        raise RuntimeError(f"dedup drain failed {rc}: {lib.last_error()}")
    out = bytes(bytearray(buf[i] for i in range(ln.value)))
    L.rw_spill_free.argtypes = [C.c_void_p]
    L.rw_spill_free(C.cast(buf, C.c_void_p))
    return out


def agg_dedup_restore(lib, h, di, buf):
    """Rebuild one dedup table from concatenated drain bytes (rw_stream.h)."""
    L = lib.lib
    L.rw_agg_dedup_restore.restype = C.c_int
    L.rw_agg_dedup_restore.argtypes = [C.c_void_p, C.c_int, C.c_char_p,
                                       C.c_uint64]
    rc = L.rw_agg_dedup_restore(h, di, buf, len(buf))
    if rc != 0:
        raise RuntimeError(f"dedup restore failed {rc}: {lib.last_error()}")


def agg_minput_restore(lib, h, mi, buf):
    L = lib.lib
    L.rw_agg_minput_restore.restype = C.c_int
    L.rw_agg_minput_restore.argtypes = [C.c_void_p, C.c_int, C.c_char_p,
                                        C.c_uint64]
    rc = L.rw_agg_minput_restore(h, mi, buf, len(buf))
    if rc != 0:
        raise RuntimeError(f"minput restore failed {rc}: {lib.last_error()}")
