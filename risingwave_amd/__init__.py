"""risingwave_amd — the MI355X-native stream HashAgg/HashJoin executors.

PRODUCT PACKAGE. The compute path is `librw_amd.so` (hand-written HIP for
gfx950, built in-tree by csrc/Makefile) behind the C-ABI of
include/rw_stream.h. There is NO CPU fallback: on a machine with a GPU,
loading failures or missing kernels raise; executor creation on a GPU-less
machine fails with RW_E_NOGPU from the library itself.
"""
import ctypes
import os

_HERE = os.path.dirname(os.path.abspath(__file__))
LIB_PATH = os.path.join(_HERE, "librw_amd.so")


class MissingNativeLibrary(RuntimeError):
    pass


_lib = None


def load_library():
    """dlopen the product library; raises loudly if absent or unloadable."""
    global _lib
    if _lib is None:
        if not os.path.exists(LIB_PATH):
            raise MissingNativeLibrary(
                f"{LIB_PATH} not built — run `make -C {os.path.join(_HERE, 'csrc')}` "
                "(or __graft_entry__.build()); the product path has no fallback"
            )
        _lib = ctypes.CDLL(LIB_PATH)
    return _lib


def lib_path():
    return LIB_PATH
