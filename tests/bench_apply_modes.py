"""A/B the agg apply kernel modes on a 1M-row q7 batch (debug tool)."""
import ctypes
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np

import bench
import risingwave_amd
from rwtest import ffi


def run_mode(mode):
    os.environ["RW_AGG_DEBUG_MODE"] = str(mode)
    gpu = ffi.Lib(risingwave_amd.lib_path())
    L = gpu.lib
    L.rw_agg_bench_preload.restype = ctypes.c_void_p
    L.rw_agg_bench_preload.argtypes = [ctypes.c_void_p, ctypes.POINTER(ffi.RwChunkC)]
    L.rw_agg_bench_apply.restype = ctypes.c_int
    L.rw_agg_bench_apply.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
    L.rw_agg_sync.argtypes = [ctypes.c_void_p]
    L.rw_agg_kernel_stats.argtypes = [ctypes.c_void_p, ctypes.POINTER(bench.KernelStats)]
    L.rw_agg_stats_reset.argtypes = [ctypes.c_void_p]
    calls = [(ffi.AGG_MAX, 1, ffi.T_I64), (ffi.AGG_COUNT_STAR, -1, ffi.T_I64)]
    agg = ffi.HashAgg(gpu, [ffi.T_I64, ffi.T_I64], [0], calls, 1, append_only=True)
    rng = np.random.default_rng(9)
    batches = []
    for b in range(16):
        c = bench.make_q7_chunk(ffi, rng, 1 << 20, b * 64 * bench.WINDOW_US, 64)
        cc = c.to_c()
        h = L.rw_agg_bench_preload(agg.h, ctypes.byref(cc))
        batches.append(h)
    for i in range(8):
        L.rw_agg_bench_apply(agg.h, batches[i % 16])
    L.rw_agg_sync(agg.h)
    L.rw_agg_stats_reset(agg.h)
    for i in range(64):
        L.rw_agg_bench_apply(agg.h, batches[i % 16])
    L.rw_agg_sync(agg.h)
    ks = bench.KernelStats()
    L.rw_agg_kernel_stats(agg.h, ctypes.byref(ks))
    print(f"mode {mode}: avg {ks.total_ms / max(ks.launches,1) * 1000:8.1f} us/launch "
          f"({19 * (1<<20) / (ks.total_ms / max(ks.launches,1) / 1e3) / 1e9:7.0f} GB/s algorithmic)")
    agg.close()


if __name__ == "__main__":
    for mode in (0, 3, 4, 1, 2):
        run_mode(mode)
