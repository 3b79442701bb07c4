"""Measures the PCIe-inclusive ingest rate of the normal push_chunk path
(host chunk → async H2D → apply kernel → sync), for DESIGN.md §5's
boundary-rate note. Not part of the bench contract — bench.py's `value` is
the HBM-resident rate."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

import bench
import risingwave_amd
from rwtest import ffi


def main():
    import ctypes

    gpu = ffi.Lib(risingwave_amd.lib_path())
    calls = [(ffi.AGG_MAX, 1, ffi.T_I64), (ffi.AGG_COUNT_STAR, -1, ffi.T_I64)]
    rng = np.random.default_rng(4)
    L = gpu.lib
    L.rw_hash_agg_ingest_mode.restype = ctypes.c_int
    L.rw_hash_agg_ingest_mode.argtypes = [ctypes.c_void_p, ctypes.c_int]
    for mode in ("per-push apply", "epoch-batched ingest (flush/64 chunks)"):
        agg = ffi.HashAgg(gpu, [ffi.T_I64, ffi.T_I64], [0], calls, 1,
                          append_only=True)
        epoch_mode = mode.startswith("epoch")
        if epoch_mode:
            assert L.rw_hash_agg_ingest_mode(agg.h, 1) == 0, gpu.last_error()
        print(f"== {mode} ==")
        for rows_per_chunk in (4096, 65536, 1 << 20):
            chunk = bench.make_q7_chunk(ffi, rng, rows_per_chunk, 0, 64)
            agg.push(chunk)  # warm
            agg.flush(0)
            agg.poll_all()
            n = max(4, min(200, (1 << 24) // rows_per_chunk))
            t0 = time.perf_counter()
            i = 0
            for _ in range(n):
                agg.push(chunk)
                i += 1
                if epoch_mode and i % 64 == 0:
                    agg.flush(i)
                    agg.poll_all()
            if epoch_mode:
                agg.flush(n + 1)
                agg.poll_all()
            dt = time.perf_counter() - t0
            rate = n * rows_per_chunk / dt
            print(f"  rows/chunk={rows_per_chunk:8d}: {rate/1e6:9.1f} M rows/s "
                  f"({rate*19/1e9:.1f} GB/s PCIe-inclusive)")
        agg.close()


if __name__ == "__main__":
    main()
