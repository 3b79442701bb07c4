"""CPU checks of the product C-ABI library: it builds for gfx950, loads,
exports every symbol include/rw_stream.h declares, and refuses to create
executors without a GPU (no silent CPU fallback — DESIGN.md §3.3)."""
import ctypes
import os
import subprocess

import pytest

from rwtest import ffi

REPO = ffi.REPO
LIB = os.path.join(REPO, "risingwave_amd", "librw_amd.so")

SYMBOLS = [
    "rw_last_error",
    "rw_chunk_free",
    "rw_hash_agg_create",
    "rw_hash_agg_push_chunk",
    "rw_hash_agg_flush",
    "rw_hash_agg_poll",
    "rw_hash_agg_destroy",
    "rw_hash_join_create",
    "rw_hash_join_push_chunk",
    "rw_hash_join_flush",
    "rw_hash_join_poll",
    "rw_hash_join_destroy",
    "rw_agg_bench_preload",
    "rw_agg_bench_apply",
    "rw_agg_sync",
    "rw_agg_kernel_stats",
    "rw_hash_agg_watermark",
    "rw_hash_join_watermark",
    "rw_hash_agg_update_vnode_bitmap",
    "rw_hash_join_update_vnode_bitmap",
    "rw_group_top_n_create",
    "rw_group_top_n_push_chunk",
    "rw_group_top_n_flush",
    "rw_group_top_n_poll",
    "rw_group_top_n_destroy",
    "rw_agg_checkpoint_drain",
    "rw_join_checkpoint_drain",
    "rw_topn_checkpoint_drain",
    "rw_spill_free",
    "rw_vnode_compute",
    "rw_dispatch_compute",
]


def ensure_built():
    if not os.path.exists(LIB):
        subprocess.run(["make", "-C", os.path.join(REPO, "risingwave_amd", "csrc")],
                       check=True)


def has_gpu():
    try:
        lib = ctypes.CDLL("libamdhip64.so")
        n = ctypes.c_int(0)
        return lib.hipGetDeviceCount(ctypes.byref(n)) == 0 and n.value > 0
    except OSError:
        return False


def test_symbols_present():
    ensure_built()
    lib = ctypes.CDLL(LIB)
    for s in SYMBOLS:
        assert getattr(lib, s, None) is not None, f"missing symbol {s}"


@pytest.mark.skipif(has_gpu(), reason="only meaningful without a GPU")
def test_no_gpu_fails_loudly():
    ensure_built()
    lib = ffi.Lib(LIB)
    with pytest.raises(RuntimeError, match="no GPU"):
        ffi.HashAgg(lib, [ffi.T_I64], [0], [(ffi.AGG_COUNT_STAR, -1, ffi.T_I64)], 0)


def test_package_raises_without_library(tmp_path, monkeypatch):
    import risingwave_amd

    monkeypatch.setattr(risingwave_amd, "LIB_PATH", str(tmp_path / "nope.so"))
    monkeypatch.setattr(risingwave_amd, "_lib", None)
    with pytest.raises(risingwave_amd.MissingNativeLibrary):
        risingwave_amd.load_library()
