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


if __name__ == "__main__":
    main()
