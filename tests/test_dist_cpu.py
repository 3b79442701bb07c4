"""Multi-process (gloo, world_size=2) CPU test of the sharded-execution
semantics bench.py relies on: the path shards by key —
window keys are partitioned across ranks (the reference's vnode hash
dispatch, dispatch.rs:949-1050, routes each key to exactly one actor), so
the union of per-rank executor outputs must equal a single executor's output
over the union of the inputs. Runs entirely on CPU via the oracle."""
import os
import pickle

import numpy as np
import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _make_chunks(rank_keys, seed):
    # built inside workers too; keep pure function of args
    from rwtest import ffi

    rng = np.random.default_rng(seed)
    chunks = []
    for _ in range(4):
        n = 1024
        keys = rng.choice(rank_keys, n)
        vals = rng.integers(1, 10**6, n)
        chunks.append(
            ffi.Chunk(
                [ffi.T_I64, ffi.T_I64],
                np.zeros(n, np.uint8),
                [keys, vals],
                [np.ones(n, np.uint8), np.ones(n, np.uint8)],
            )
        )
    return chunks


def _run_agg(chunks):
    from rwtest import ffi

    calls = [(ffi.AGG_MAX, 1, ffi.T_I64), (ffi.AGG_COUNT_STAR, -1, ffi.T_I64)]
    agg = ffi.HashAgg(ffi.oracle(), [ffi.T_I64, ffi.T_I64], [0], calls, 1,
                      append_only=True)
    out = []
    for e in range(2):
        for c in chunks[e * 2:(e + 1) * 2]:
            agg.push(c)
        agg.flush(e + 1)
        out.append(ffi.rows_multiset(agg.poll_all()))
    agg.close()
    return out


def _worker(rank, world, port, result_dir):
    import sys

    tests_dir = os.path.dirname(os.path.abspath(__file__))
    sys.path.insert(0, tests_dir)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)

    # rank-local key shard (disjoint — the post-exchange invariant)
    rank_keys = np.arange(rank * 100, rank * 100 + 50, dtype=np.int64)
    chunks = _make_chunks(rank_keys, seed=42 + rank)
    out = _run_agg(chunks)

    gathered = [None] * world
    dist.all_gather_object(gathered, out)
    if rank == 0:
        with open(os.path.join(result_dir, "gathered.pkl"), "wb") as f:
            pickle.dump(gathered, f)
    dist.barrier()
    dist.destroy_process_group()


def test_sharded_agg_union_equals_single(tmp_path):
    world = 2
    port = 29781
    mp.spawn(_worker, args=(world, port, str(tmp_path)), nprocs=world, join=True)
    with open(tmp_path / "gathered.pkl", "rb") as f:
        gathered = pickle.load(f)

    # single-executor run over the union of both ranks' inputs
    all_chunks = []
    per_rank_chunks = []
    for rank in range(world):
        rank_keys = np.arange(rank * 100, rank * 100 + 50, dtype=np.int64)
        per_rank_chunks.append(_make_chunks(rank_keys, seed=42 + rank))
    # interleave epoch-wise: epoch e = both ranks' chunks for that epoch
    single_out = []
    from rwtest import ffi

    calls = [(ffi.AGG_MAX, 1, ffi.T_I64), (ffi.AGG_COUNT_STAR, -1, ffi.T_I64)]
    agg = ffi.HashAgg(ffi.oracle(), [ffi.T_I64, ffi.T_I64], [0], calls, 1,
                      append_only=True)
    for e in range(2):
        for rank in range(world):
            for c in per_rank_chunks[rank][e * 2:(e + 1) * 2]:
                agg.push(c)
        agg.flush(e + 1)
        single_out.append(ffi.rows_multiset(agg.poll_all()))
    agg.close()

    for e in range(2):
        union = sorted(gathered[0][e] + gathered[1][e],
                       key=lambda r: (r[0], r[1]))
        single = sorted(single_out[e], key=lambda r: (r[0], r[1]))
        assert union == single
