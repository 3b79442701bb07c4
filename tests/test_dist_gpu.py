"""2-rank RCCL exchange test: the N>1 data-path code
— vnode partition kernel -> ncclSend/Recv all-to-all-v (librw_exchange) ->
receiver-side payload apply — with world_size 2. RCCL enforces ONE RANK
PER DEVICE (ncclCommInitRank rejects duplicate devices, as NCCL does), so
on a single-GPU box the 2-rank communicator cannot form: the test then
verifies the refusal is clean and SKIPS — the RCCL send/recv data path at
world=1 is exercised on hardware by test_exchange_partition_parity, and
the cross-device path by the driver's round-end multi-GPU bench. On a
box with >= 2 visible GPUs the full 2-rank exchange runs and the union of
the two ranks' post-exchange agg outputs must equal one executor over the
union of the inputs (HashDataDispatcher routing, dispatch.rs:949-1050)."""
import ctypes
import os
import pickle
import subprocess
import sys

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _worker_main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    result_dir = os.environ["RESULT_DIR"]
    sys.path.insert(0, REPO)
    sys.path.insert(0, os.path.join(REPO, "tests"))
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    import bench
    import risingwave_amd
    from rwtest import ffi

    risingwave_amd.load_library()
    gpu_lib = ffi.Lib(risingwave_amd.lib_path())
    L = gpu_lib.lib
    L.rw_agg_bench_preload.restype = ctypes.c_void_p
    L.rw_agg_bench_preload.argtypes = [ctypes.c_void_p,
                                       ctypes.POINTER(ffi.RwChunkC)]
    L.rw_agg_apply_payload.restype = ctypes.c_int
    L.rw_agg_apply_payload.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                       ctypes.POINTER(ctypes.c_uint64),
                                       ctypes.c_int, ctypes.c_int,
                                       ctypes.c_int]
    L.rw_agg_n_batch_slots.restype = ctypes.c_int
    L.rw_agg_n_batch_slots.argtypes = [ctypes.c_void_p]

    calls = [(ffi.AGG_MAX, 1, ffi.T_I64), (ffi.AGG_COUNT_STAR, -1, ffi.T_I64)]
    agg = ffi.HashAgg(gpu_lib, [ffi.T_I64, ffi.T_I64], [0], calls, 1,
                      append_only=True)
    exch = bench.setup_exchange(ffi, rank, world, dist)
    if exch is None:
        # clean refusal (duplicate device): report and signal a skip
        try:
            xl = ctypes.CDLL(os.path.join(REPO, "risingwave_amd",
                                          "librw_exchange.so"))
            xl.rw_exchange_last_error.restype = ctypes.c_char_p
            err = xl.rw_exchange_last_error().decode()
        except Exception as e:  # noqa: BLE001
            err = str(e)
        with open(os.path.join(result_dir, f"refused.{rank}"), "w") as f:
            f.write(err)
        dist.barrier()
        dist.destroy_process_group()
        sys.exit(42)
    xb = exch.make_buffers(64 << 20)

    # GLOBAL window space: the exchange routes each window to its vnode
    # owner; inputs deliberately span both ranks' shards
    rng = np.random.default_rng(1000 + rank)
    outs = []
    nslots = L.rw_agg_n_batch_slots(agg.h)
    for epoch in range(2):
        for _ in range(2):
            n = 65536
            c = bench.make_q7_chunk(ffi, rng, n, 0, 64)
            cc = c.to_c()
            h = L.rw_agg_bench_preload(agg.h, ctypes.byref(cc))
            assert h, gpu_lib.last_error()
            recv_blocks = exch.run(agg.h, h, xb, n_cols=nslots)
            rc = L.rw_agg_apply_payload(agg.h, ctypes.c_void_p(xb.recv),
                                        recv_blocks, world, nslots, 1)
            assert rc == 0, gpu_lib.last_error()
        agg.flush(epoch + 1)
        outs.append(ffi.rows_multiset(agg.poll_all()))
    gathered = [None] * world
    dist.all_gather_object(gathered, outs)
    if rank == 0:
        with open(os.path.join(result_dir, "gathered.pkl"), "wb") as f:
            pickle.dump(gathered, f)
    agg.close()
    dist.barrier()
    dist.destroy_process_group()


def test_rccl_exchange_2ranks_1gpu(tmp_path):
    world = 2
    procs = []
    for rank in range(world):
        env = dict(os.environ)
        import torch

        n_dev = max(torch.cuda.device_count(), 1)
        env.update({
            "RANK": str(rank), "WORLD_SIZE": str(world),
            "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29783",
            "RESULT_DIR": str(tmp_path),
            "HIP_VISIBLE_DEVICES": str(rank % n_dev),
            "RW_DIST_GPU_WORKER": "1",
        })
        procs.append(subprocess.Popen(
            [sys.executable, os.path.abspath(__file__)], env=env))
    rcs = []
    try:
        for p in procs:
            rcs.append(p.wait(timeout=180))
    finally:
        for p in procs:
            if p.poll() is None:
                p.kill()
    if all(rc == 42 for rc in rcs):
        err = ""
        for rank in range(world):
            p = tmp_path / f"refused.{rank}"
            if p.exists():
                err = p.read_text()
                break
        pytest.skip("RCCL: one rank per device (clean refusal on a "
                    f"single-GPU box): {err!r}; world=1 RCCL path covered "
                    "by test_exchange_partition_parity")
    assert all(rc == 0 for rc in rcs), f"worker exit codes {rcs}"

    with open(tmp_path / "gathered.pkl", "rb") as f:
        gathered = pickle.load(f)

    # single-executor oracle over the union of both ranks' inputs
    sys.path.insert(0, REPO)
    import bench
    from rwtest import ffi

    calls = [(ffi.AGG_MAX, 1, ffi.T_I64), (ffi.AGG_COUNT_STAR, -1, ffi.T_I64)]
    agg = ffi.HashAgg(ffi.oracle(), [ffi.T_I64, ffi.T_I64], [0], calls, 1,
                      append_only=True)
    rngs = [np.random.default_rng(1000 + r) for r in range(world)]
    single = []
    for epoch in range(2):
        for r in range(world):
            for _ in range(2):
                agg.push(bench.make_q7_chunk(ffi, rngs[r], 65536, 0, 64))
        agg.flush(epoch + 1)
        single.append(ffi.rows_multiset(agg.poll_all()))
    agg.close()
    for e in range(2):
        union = sorted(gathered[0][e] + gathered[1][e])
        assert union == sorted(single[e]), f"epoch {e}: shard union diverged"


if __name__ == "__main__" and os.environ.get("RW_DIST_GPU_WORKER"):
    _worker_main()
