"""Debug repro for the duplicate-output bug (not a pytest test).
Runs the count/sum retract scenario on the GPU, dumps where the duplicate
comes from: table (dup READY keys) vs flush (dup emission)."""
import ctypes
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

import risingwave_amd
from rwtest import ffi
from rwtest.ffi import AGG_COUNT_STAR, AGG_SUM, T_I64, oracle, rows_multiset


class Dbg(ctypes.Structure):
    _fields_ = [("ready_slots", ctypes.c_uint64), ("dup_keys", ctypes.c_uint64),
                ("dirty_count", ctypes.c_uint64), ("out_cursor", ctypes.c_uint64)]


def main():
    gpu = ffi.Lib(risingwave_amd.lib_path())
    gpu.lib.rw_agg_debug_scan.argtypes = [ctypes.c_void_p, ctypes.POINTER(Dbg)]

    rng = np.random.default_rng(2)
    calls = [(AGG_COUNT_STAR, -1, T_I64), (AGG_SUM, 1, T_I64)]
    g = ffi.HashAgg(gpu, [T_I64, T_I64], [0], calls, 0)
    o = ffi.HashAgg(oracle(), [T_I64, T_I64], [0], calls, 0)

    inserted = []
    for epoch in range(4):
        for _ in range(4):
            n = 1024
            keys = rng.integers(0, 300, n)
            vals = rng.integers(1, 1000, n)
            ops = np.zeros(n, np.uint8)
            for i in range(n):
                if inserted and rng.random() < 0.2:
                    jx = rng.integers(0, len(inserted))
                    keys[i], vals[i] = inserted.pop(int(jx))
                    ops[i] = ffi.OP_DELETE
                else:
                    inserted.append((int(keys[i]), int(vals[i])))
            c = ffi.Chunk([T_I64, T_I64], ops, [keys, vals],
                          [np.ones(n, np.uint8)] * 2)
            g.push(c)
            o.push(c)
        d = Dbg()
        gpu.lib.rw_agg_debug_scan(g.h, ctypes.byref(d))
        print(f"epoch {epoch+1} PRE-FLUSH: ready={d.ready_slots} dup_keys={d.dup_keys} dirty={d.dirty_count} out_cursor={d.out_cursor}")
        dump_dirty(gpu, g)
        g.flush(epoch + 1)
        o.flush(epoch + 1)
        gpu.lib.rw_agg_debug_scan(g.h, ctypes.byref(d))
        mg = rows_multiset(g.poll_all())
        mo = rows_multiset(o.poll_all())
        print(f"epoch {epoch+1} POST-FLUSH: ready={d.ready_slots} dup_keys={d.dup_keys} "
              f"gpu_rows={len(mg)} oracle_rows={len(mo)} equal={mg == mo}")
        if mg != mo:
            from collections import Counter

            cg, co = Counter(mg), Counter(mo)
            extra = cg - co
            missing = co - cg
            print("  extra in GPU:", list(extra.items())[:5])
            print("  missing in GPU:", list(missing.items())[:5])
    g.close()
    o.close()


def dump_dirty(gpu, g):
    import ctypes as C

    gpu.lib.rw_agg_debug_dirty.argtypes = [C.c_void_p, C.c_uint32,
        C.POINTER(C.c_uint32), C.POINTER(C.c_uint32), C.POINTER(C.c_int64),
        C.POINTER(C.c_longlong), C.POINTER(C.c_longlong), C.POINTER(C.c_uint32)]
    N = 4096
    slots = (C.c_uint32 * N)(); states = (C.c_uint32 * N)()
    key0 = (C.c_int64 * N)(); a0 = (C.c_longlong * N)(); a1 = (C.c_longlong * N)()
    n = C.c_uint32()
    gpu.lib.rw_agg_debug_dirty(g.h, N, slots, states, key0, a0, a1, C.byref(n))
    rows = [(slots[i], states[i], key0[i], a0[i], a1[i]) for i in range(n.value)]
    # report duplicate slots and non-READY entries
    from collections import Counter

    cnt = Counter(r[0] for r in rows)
    dups = {s: c for s, c in cnt.items() if c > 1}
    nonready = [r for r in rows if r[1] != 2]
    keycnt = Counter(r[2] for r in rows)
    dupkeys = {k: c for k, c in keycnt.items() if c > 1}
    print(f"  dirty={n.value} dup_slots={dups} nonready={nonready[:5]} dup_keys_in_dirty={dupkeys}")
    for r in rows:
        if r[2] in dupkeys:
            print("   entry:", r)


if __name__ == "__main__":
    main()
