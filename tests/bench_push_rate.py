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

    # JOIN push path: 4K-row probe chunks against a 1M-key build side,
    # per-push vs epoch-batched (same-side run merging)
    L.rw_hash_join_ingest_mode.restype = ctypes.c_int
    L.rw_hash_join_ingest_mode.argtypes = [ctypes.c_void_p, ctypes.c_int]
    for mode in ("per-push", "epoch-batched (flush/64 chunks)"):
        j = ffi.HashJoin(gpu, ffi.JOIN_INNER, [ffi.T_I64, ffi.T_I64],
                         [ffi.T_I64, ffi.T_I64], key_l=[0], key_r=[0],
                         pk_l=[1], pk_r=[1], state_capacity_hint=1 << 21,
                         row_capacity_hint=1 << 26)
        if mode.startswith("epoch"):
            assert L.rw_hash_join_ingest_mode(j.h, 1) == 0, gpu.last_error()
        rng2 = np.random.default_rng(9)
        build = np.arange(1_000_000, dtype=np.int64)
        for lo in range(0, 1_000_000, 65536):
            ids = build[lo:lo + 65536]
            j.push(ffi.SIDE_RIGHT,
                   ffi.Chunk([ffi.T_I64, ffi.T_I64],
                             np.zeros(len(ids), np.uint8),
                             [ids, ids],
                             [np.ones(len(ids), np.uint8)] * 2))
            j.poll_all()
        j.flush(0)
        n_chunks = 512
        rows = 4096
        chunk = ffi.Chunk([ffi.T_I64, ffi.T_I64], np.zeros(rows, np.uint8),
                          [rng2.integers(0, 1_000_000, rows),
                           np.arange(10_000_000, 10_000_000 + rows)],
                          [np.ones(rows, np.uint8)] * 2)
        j.push(ffi.SIDE_LEFT, chunk)  # warm
        j.flush(0)
        j.poll_all()
        t0 = time.perf_counter()
        for i in range(n_chunks):
            j.push(ffi.SIDE_LEFT, chunk)
            if (i + 1) % 64 == 0:
                j.flush(i)
                j.poll_all()
        j.flush(n_chunks)
        j.poll_all()
        dt = time.perf_counter() - t0
        rate = n_chunks * rows / dt
        print(f"join {mode:34s}: {rate/1e6:9.1f} M probe rows/s "
              f"(4K-row chunks, PCIe-inclusive)")
        j.close()


if __name__ == "__main__":
    main()
