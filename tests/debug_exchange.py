"""Debug: N=1 self-loop exchange (partition kernel + RCCL send/recv to self),
then payload apply — step by step with faulthandler."""
import ctypes
import faulthandler
import os
import sys

faulthandler.enable()
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

import bench
import risingwave_amd
from rwtest import ffi


def main():
    gpu = ffi.Lib(risingwave_amd.lib_path())
    L = gpu.lib
    L.rw_agg_bench_preload.restype = ctypes.c_void_p
    L.rw_agg_bench_preload.argtypes = [ctypes.c_void_p, ctypes.POINTER(ffi.RwChunkC)]
    L.rw_agg_apply_payload.restype = ctypes.c_int
    L.rw_agg_apply_payload.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                       ctypes.POINTER(ctypes.c_uint64),
                                       ctypes.c_int, ctypes.c_int,
                                       ctypes.c_int]

    print("1: create agg + exchange")
    calls = [(ffi.AGG_MAX, 1, ffi.T_I64), (ffi.AGG_COUNT_STAR, -1, ffi.T_I64)]
    agg = ffi.HashAgg(gpu, [ffi.T_I64, ffi.T_I64], [0], calls, 1, append_only=True)
    exch = bench.setup_exchange(ffi, 0, 1, None)
    assert exch is not None, "exchange init failed"
    print("2: preload batch")
    rng = np.random.default_rng(3)
    n = 65536
    c = bench.make_q7_chunk(ffi, rng, n, 0, 32)
    cc = c.to_c()
    batch = L.rw_agg_bench_preload(agg.h, ctypes.byref(cc))
    assert batch
    print("3: buffers")
    xb = exch.make_buffers(n * 32 * 4)
    print("4: run exchange")
    L.rw_agg_n_batch_slots.restype = ctypes.c_int
    L.rw_agg_n_batch_slots.argtypes = [ctypes.c_void_p]
    nslots = L.rw_agg_n_batch_slots(agg.h)
    recv_counts = exch.run(agg.h, batch, xb, n_cols=nslots)
    print("   recv:", list(recv_counts))
    assert sum(recv_counts) == n
    print("5: apply payload")
    rc = L.rw_agg_apply_payload(agg.h, ctypes.c_void_p(xb.recv), recv_counts, 1,
                                nslots, 1)
    assert rc == 0, gpu.last_error()
    print("6: flush + compare vs oracle")
    agg.flush(1)
    got = ffi.rows_multiset(agg.poll_all())
    o = ffi.HashAgg(ffi.oracle(), [ffi.T_I64, ffi.T_I64], [0], calls, 1,
                    append_only=True)
    o.push(c)
    o.flush(1)
    want = ffi.rows_multiset(o.poll_all())
    assert got == want, f"exchange parity failed: {len(got)} vs {len(want)}"
    print("exchange self-loop OK:", len(got), "groups")


if __name__ == "__main__":
    main()
