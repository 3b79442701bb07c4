#!/usr/bin/env python3
"""bench.py — measures the hot path per the driver contract.

BASELINE.json's metric is "input rows/sec/GPU on Nexmark q7+q8"; the default
run emits BOTH lines:
- q7 (configs[1]): append-only HashAgg group_key=[window $expr1],
  aggs=[max(price), count] (reference plan `nexmark.yaml` q7 block) on
  synthetic bid-shaped chunks: price ~ uniform[1,1e7) i64, date_time
  monotone in 10s windows, all-Insert ops, seeded. A step = one checkpoint
  EPOCH: 16 x 1M-row chunk-batches (= 4096 ingested 4K-row chunks,
  DESIGN.md §3.1) already resident in HBM, applied as one launch, plus the
  stream-ordered flush.
- q8 (configs[2]): stream-stream hash-join, 10M-key build side in HBM;
  a step = one 1M-row probe batch.
--workload q3 selects the TPC-H-q3-stream pipeline (configs[4]).

Parity gate: before timing, a small q7 run is compared row-for-row
(multiset per epoch) against the CPU oracle; a mismatch aborts the bench.

Multi-GPU (--gpus N under torchrun): weak scaling, rank-local window key
ranges (the upstream vnode exchange has already routed keys by window —
DESIGN.md §7); no data-path collective; barrier + max-over-ranks timing.
"""
import argparse
import ctypes
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))

import numpy as np


class KernelStats(ctypes.Structure):
    _fields_ = [
        ("launches", ctypes.c_uint64),
        ("total_ms", ctypes.c_double),
        ("rows", ctypes.c_uint64),
    ]


CHUNK_ROWS = 4096
CHUNKS_PER_BATCH = 256   # 1M rows per logical ingest batch
EPOCH_BATCHES = 16       # batches buffered per checkpoint epoch
WINDOW_US = 10_000_000   # 10s tumble (q7)
# Algorithmic bytes per input row for the q7 dense agg_apply kernel: the two
# 8-B input streams (window key, price). The dense path reads no ops/validity
# streams and the few window slots stay cache-resident. PMC counters agree:
# 2x134,747 KB FETCH + 562 KB WRITE per 16,777,216-row epoch launch
# (profiles/r01_q7_pmc_{fetch,write}_v7.txt, x2 = the gfx950 FETCH_SIZE
# correction, MI355X_MICROARCH.md §HBM) = 16.5 B/row measured vs 16.0
# algorithmic.
Q7_BYTES_PER_ROW = 16.0
Q7_TRAFFIC_B_PER_ROW = (2 * 134_747 + 562) * 1024 / (EPOCH_BATCHES * CHUNK_ROWS * CHUNKS_PER_BATCH)
HBM_PEAK_GBS = 8000.0  # spec peak (MI355X_MICROARCH.md)


def make_q7_chunk(ffi, rng, n, window_base, n_windows):
    # date_time-derived window key: monotone within the chunk
    w = window_base + np.sort(rng.integers(0, n_windows, n)) * WINDOW_US
    price = rng.integers(1, 10**7, n)
    return ffi.Chunk(
        [ffi.T_I64, ffi.T_I64],
        np.zeros(n, np.uint8),
        [w, price],
        [np.ones(n, np.uint8), np.ones(n, np.uint8)],
    )


def parity_gate(ffi, gpu_lib, rng):
    from rwtest.ffi import AGG_COUNT_STAR, AGG_MAX, T_I64, oracle, rows_multiset

    calls = [(AGG_MAX, 1, T_I64), (AGG_COUNT_STAR, -1, T_I64)]
    execs = []
    for lib in (gpu_lib, oracle()):
        execs.append(ffi.HashAgg(lib, [T_I64, T_I64], [0], calls, 1, append_only=True))
    for epoch in range(3):
        chunks = [make_q7_chunk(ffi, rng, CHUNK_ROWS, 0, 32) for _ in range(4)]
        outs = []
        for a in execs:
            for c in chunks:
                a.push(c)
            a.flush(epoch + 1)
            outs.append(rows_multiset(a.poll_all()))
        if outs[0] != outs[1]:
            raise SystemExit("PARITY GATE FAILED: GPU != oracle on q7 sample")
    for a in execs:
        a.close()


def cpu_baseline(ffi, rng, target_seconds=10.0):
    """Time the oracle (the CPU restatement, kind 'port') on the same q7
    workload on ALL host cores — one executor instance per thread, the
    reference's actor-parallelism model (one tokio task per actor,
    task/actor_manager.rs:536; window keys shard across actors via the vnode
    exchange). ctypes releases the GIL during the C calls."""
    import threading

    from rwtest.ffi import AGG_COUNT_STAR, AGG_MAX, T_I64, oracle

    ncores = os.cpu_count() or 1
    del threading  # processes, not threads: the per-chunk ctypes dispatch is
    # GIL-bound at 4K-row chunks (8 threads measured BELOW one core)
    import multiprocessing as mp

    ctx = mp.get_context("fork")
    t0 = time.perf_counter()
    # wall-clock-bounded workers: a fixed chunk count misestimates N-way
    # memory contention by 30x on a 256-core host
    with ctx.Pool(ncores) as pool:
        done = pool.map(_cpu_baseline_worker,
                        [(tid, target_seconds) for tid in range(ncores)])
    dt = time.perf_counter() - t0
    rows = sum(done)
    return {
        "value": rows / dt,
        "unit": "rows/s",
        "cores": ncores,
        "kind": "port",
        "sample": f"{rows} q7 rows across {ncores} executor processes "
                  f"({dt:.1f}s, flush every 64 chunks per process; "
                  f"fork overhead included)",
    }


def _cpu_baseline_worker(arg):
    tid, seconds = arg
    from rwtest import ffi
    from rwtest.ffi import AGG_COUNT_STAR, AGG_MAX, T_I64, oracle

    calls = [(AGG_MAX, 1, T_I64), (AGG_COUNT_STAR, -1, T_I64)]
    a = ffi.HashAgg(oracle(), [T_I64, T_I64], [0], calls, 1, append_only=True)
    c = make_q7_chunk(ffi, np.random.default_rng(100 + tid), CHUNK_ROWS,
                      tid * 1_000_000 * WINDOW_US, 32)
    a.push(c)  # warm
    deadline = time.perf_counter() + seconds
    i = 0
    while time.perf_counter() < deadline:
        a.push(c)
        i += 1
        if i % 64 == 0:
            a.flush(i)
            a.poll_all()
    a.close()
    return i * CHUNK_ROWS


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=500)
    ap.add_argument("--warmup", type=int, default=25)
    ap.add_argument("--barrier-every", type=int, default=16)
    ap.add_argument("--windows-per-epoch", type=int, default=64)
    ap.add_argument("--seed", type=int, default=1)
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    ap.add_argument("--workload",
                    choices=["all", "q7", "q7pipe", "q8", "q3"],
                    default="all",
                    help="all (default) = the BASELINE metric 'q7+q8': one "
                         "run emits the q7 line then the q8 line. q7 = "
                         "windowed hash-agg only (configs[1]); q8 = "
                         "stream-stream hash-join, 10M-key build side "
                         "(configs[2]); q3 = TPC-H-stream join+agg pipeline "
                         "with insert/delete mix (configs[4], minus the "
                         "Hummock spill)")
    ap.add_argument("--q3-orders", type=int, default=100_000_000)
    ap.add_argument("--exchange", choices=["auto", "on", "off"], default="auto",
                    help="vnode partition + RCCL all-to-all-v before the agg "
                         "(the reference's HashDataDispatcher hop, SURVEY "
                         "§8e). auto = on when WORLD_SIZE > 1")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if world > 1:
        os.environ.setdefault("HIP_VISIBLE_DEVICES", str(local_rank))

    dist = None
    if world > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        # gloo for control (timing barriers, uniqueId broadcast); the DATA
        # plane collective is our own RCCL communicator (rw_exchange)
        dist.init_process_group(backend="gloo")

    from rwtest import ffi

    import risingwave_amd

    risingwave_amd.load_library()
    gpu_lib = ffi.Lib(risingwave_amd.lib_path())
    L = gpu_lib.lib
    L.rw_agg_bench_preload.restype = ctypes.c_void_p
    L.rw_agg_bench_preload.argtypes = [ctypes.c_void_p, ctypes.POINTER(ffi.RwChunkC)]
    L.rw_agg_bench_apply.restype = ctypes.c_int
    L.rw_agg_bench_apply.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
    L.rw_agg_sync.restype = ctypes.c_int
    L.rw_agg_sync.argtypes = [ctypes.c_void_p]
    L.rw_agg_flush_launch.restype = ctypes.c_int
    L.rw_agg_flush_launch.argtypes = [ctypes.c_void_p, ctypes.c_uint64]
    L.rw_agg_bench_run.restype = ctypes.c_int
    L.rw_agg_bench_run.argtypes = [ctypes.c_void_p,
                                   ctypes.POINTER(ctypes.c_void_p),
                                   ctypes.c_int, ctypes.c_int, ctypes.c_int,
                                   ctypes.c_int]
    L.rw_agg_bench_run_epochs.restype = ctypes.c_int
    L.rw_agg_bench_run_epochs.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                          ctypes.c_uint64, ctypes.c_int,
                                          ctypes.c_int, ctypes.c_int,
                                          ctypes.c_int]
    L.rw_agg_kernel_stats.argtypes = [ctypes.c_void_p, ctypes.POINTER(KernelStats)]
    L.rw_agg_stats_reset.argtypes = [ctypes.c_void_p]
    L.rw_agg_apply_payload.restype = ctypes.c_int
    L.rw_agg_apply_payload.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                       ctypes.POINTER(ctypes.c_uint64),
                                       ctypes.c_int, ctypes.c_int,
                                       ctypes.c_int]
    L.rw_agg_n_batch_slots.restype = ctypes.c_int
    L.rw_agg_n_batch_slots.argtypes = [ctypes.c_void_p]

    rng = np.random.default_rng(args.seed + rank)

    # ---- parity gate (rank 0 only; cheap) ----
    if rank == 0:
        parity_gate(ffi, gpu_lib, np.random.default_rng(99))

    try:
        if args.workload in ("all", "q7"):
            bench_q7(args, ffi, gpu_lib, rng, rank, world, dist)
        if args.workload in ("all", "q7pipe"):
            bench_q7pipe(args, ffi, gpu_lib,
                         np.random.default_rng(args.seed + rank + 7),
                         rank, world, dist)
        if args.workload in ("all", "q8"):
            bench_q8(args, ffi, gpu_lib, np.random.default_rng(args.seed + rank),
                     rank, world, dist)
        if args.workload == "q3":
            bench_q3(args, ffi, gpu_lib, rng, rank, world, dist)
    finally:
        if dist:
            dist.destroy_process_group()


def bench_q7(args, ffi, gpu_lib, rng, rank, world, dist):
    """Nexmark q7 windowed hash-agg (BASELINE configs[1]). A step = one
    checkpoint epoch: EPOCH_BATCHES x 1M-row chunk-batches applied as ONE
    kernel launch (order-free value states) followed by the
    stream-ordered flush — so every timed step is a real launch and small
    --steps counts still average over `steps` launches."""
    import ctypes

    L = gpu_lib.lib
    from rwtest.ffi import AGG_COUNT_STAR, AGG_MAX, T_I64

    calls = [(AGG_MAX, 1, T_I64), (AGG_COUNT_STAR, -1, T_I64)]
    agg = ffi.HashAgg(gpu_lib, [T_I64, T_I64], [0], calls, 1, append_only=True,
                      state_capacity_hint=1 << 20)

    use_exchange = args.exchange == "on" or (args.exchange == "auto" and world > 1)
    exch = None
    if use_exchange:
        exch = setup_exchange(ffi, rank, world, dist)
        if exch is None:
            print(f"# rank {rank}: RCCL exchange init failed "
                  f"({'see stderr' if rank == 0 else ''}) — falling back to "
                  f"pre-routed weak scaling", file=sys.stderr)
            use_exchange = False

    batch_rows = CHUNK_ROWS * CHUNKS_PER_BATCH
    # Without exchange: rank-local window space (weak scaling, upstream
    # exchange already routed keys — DESIGN.md §7). With exchange: a GLOBAL
    # window space; the RCCL all-to-all routes each window to its vnode owner.
    window_base = 0 if use_exchange else rank * 1_000_000 * WINDOW_US
    # 16 × 16 MB ≈ 270 MB of resident input > the 256 MB Infinity Cache, so
    # the timed region streams from HBM (L3-masking gotcha,
    # cdna_hip_programming.md §2)
    n_batches = EPOCH_BATCHES
    batches = []
    giant = None
    if use_exchange:
        for b in range(n_batches):
            c = make_q7_chunk(ffi, rng, batch_rows,
                              window_base + b * args.windows_per_epoch * WINDOW_US,
                              args.windows_per_epoch)
            cc = c.to_c()
            h = L.rw_agg_bench_preload(agg.h, ctypes.byref(cc))
            assert h, gpu_lib.last_error()
            batches.append(h)
    else:
        # one contiguous 16-step region: the engine buffers an epoch's chunks
        # and applies them as ONE launch before each checkpoint flush
        w_all, p_all = [], []
        for b in range(n_batches):
            base = window_base + b * args.windows_per_epoch * WINDOW_US
            w_all.append(base +
                         np.sort(rng.integers(0, args.windows_per_epoch,
                                              batch_rows)) * WINDOW_US)
            p_all.append(rng.integers(1, 10**7, batch_rows))
        total = n_batches * batch_rows
        giant_c = ffi.Chunk(
            [T_I64, T_I64], np.zeros(total, np.uint8),
            [np.concatenate(w_all), np.concatenate(p_all)],
            [np.ones(total, np.uint8), np.ones(total, np.uint8)])
        cc = giant_c.to_c()
        giant = L.rw_agg_bench_preload(agg.h, ctypes.byref(cc))
        assert giant, gpu_lib.last_error()
        del giant_c, w_all, p_all

    if use_exchange:
        payload_cap = int(batch_rows * (1 + 2 * 9) * 4)  # 4x headroom for skew
        xb = exch.make_buffers(payload_cap)

    # A STEP = ONE EPOCH in both modes: EPOCH_BATCHES 1M-row batches applied
    # (one fused launch without exchange; per-batch route+apply with it),
    # then the stream-ordered checkpoint flush. The downstream
    # (exchange/sink) consumes device-resident, as in the q3 pipeline — the
    # host-marshalling flush stays the parity-test surface
    # (tests/test_gpu_parity.py).
    def run_steps(n):
        if use_exchange:
            nslots = L.rw_agg_n_batch_slots(agg.h)
            for e in range(n):
                for b in range(n_batches):
                    recv_blocks = exch.run(agg.h, batches[b], xb,
                                           n_cols=nslots)
                    rc = L.rw_agg_apply_payload(
                        agg.h, ctypes.c_void_p(xb.recv), recv_blocks, world,
                        nslots, 1)  # q7 batches: dense all-Insert, non-null
                    assert rc == 0, gpu_lib.last_error()
                rc = L.rw_agg_flush_launch(agg.h, e)
                assert rc == 0, gpu_lib.last_error()
        else:
            # the whole epoch loop runs in C (the Python interpreter costs
            # more per step than the apply kernel itself)
            rc = L.rw_agg_bench_run_epochs(agg.h, giant, batch_rows,
                                           n_batches, n * n_batches,
                                           n_batches, 0)
            assert rc == 0, gpu_lib.last_error()

    # ---- warmup ----
    run_steps(args.warmup)
    rc = L.rw_agg_sync(agg.h)
    assert rc == 0, gpu_lib.last_error()
    L.rw_agg_stats_reset(agg.h)

    if dist:
        dist.barrier()
    _dev_sync()
    t0 = time.perf_counter()
    run_steps(args.steps)
    rc = L.rw_agg_sync(agg.h)
    assert rc == 0, gpu_lib.last_error()
    _dev_sync()
    elapsed = time.perf_counter() - t0
    if dist:
        import torch

        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    ks = KernelStats()
    L.rw_agg_kernel_stats(agg.h, ctypes.byref(ks))

    # ---- untimed extra segment (rank 0, N=1): the checkpoint-INCLUSIVE
    # epoch cost — apply + host-marshalling flush + device→host spill drain
    # (rw_agg_checkpoint_drain), the preserved-checkpoint path the north
    # star requires as a measured number ----
    ckpt = None
    if rank == 0 and world == 1 and not use_exchange:
        L.rw_agg_checkpoint_drain.restype = ctypes.c_int
        L.rw_agg_checkpoint_drain.argtypes = [
            ctypes.c_void_p, ctypes.POINTER(ctypes.POINTER(ctypes.c_uint8)),
            ctypes.POINTER(ctypes.c_uint64)]
        L.rw_spill_free.argtypes = [ctypes.POINTER(ctypes.c_uint8)]
        n_ck = 4
        drained = 0
        t0 = time.perf_counter()
        for e in range(n_ck):
            # the giant region IS one epoch: one fused apply launch
            rc = L.rw_agg_bench_apply(agg.h, giant)
            assert rc == 0, gpu_lib.last_error()
            rc = L.rw_hash_agg_flush(agg.h, 1000 + e)
            assert rc == 0, gpu_lib.last_error()
            while True:
                c = L.rw_hash_agg_poll(agg.h)
                if not c:
                    break
                L.rw_chunk_free(c)
            buf = ctypes.POINTER(ctypes.c_uint8)()
            ln = ctypes.c_uint64()
            rc = L.rw_agg_checkpoint_drain(agg.h, ctypes.byref(buf),
                                           ctypes.byref(ln))
            assert rc == 0, gpu_lib.last_error()
            drained += ln.value
            L.rw_spill_free(buf)
        ck_elapsed = time.perf_counter() - t0
        ckpt = {
            "ms_per_epoch": ck_elapsed * 1000.0 / n_ck,
            "rows_per_s": n_ck * n_batches * batch_rows / ck_elapsed,
            "drained_bytes_per_epoch": drained / n_ck,
            "epochs": n_ck,
            "note": "epoch incl. host-marshalled emission + spill drain "
                    "(checkpoint-inclusive rate; untimed in the headline)",
        }

    if rank == 0:
        epoch_rows = n_batches * batch_rows
        total_rows = args.steps * epoch_rows * world
        value = total_rows / elapsed
        ms_per_step = elapsed * 1000.0 / args.steps
        # roofline: algorithmic bytes per apply launch ÷ measured launch time
        # (HIP events on the executor's stream, inside the C library);
        # traffic = the PMC-measured per-row bytes (see Q7_TRAFFIC_B_PER_ROW)
        # scaled to this run's rows-per-launch
        avg_launch_ms = ks.total_ms / max(ks.launches, 1)
        rows_per_launch = ks.rows / max(ks.launches, 1)
        achieved_gbs = (Q7_BYTES_PER_ROW * rows_per_launch) / (avg_launch_ms * 1e-3) / 1e9
        exch_stats = None
        if use_exchange and exch is not None:
            ems, en = exch.stats()
            exch_stats = {"avg_ms": ems / max(en, 1), "launches": en}
        result = {
            "metric": "input rows/sec/GPU on Nexmark q7 stream",
            "value": value,
            "unit": "rows/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # no published number in-repo
            "dtype": "int64",
            "data": "synthetic",
            "exchange": exch_stats,
            "config": {
                "workload": "nexmark_q7",
                "chunk_rows": CHUNK_ROWS,
                "chunks_per_epoch_step": n_batches * CHUNKS_PER_BATCH,
                "rows_per_step": epoch_rows,
                "windows_per_epoch": args.windows_per_epoch,
                "barriers": "one checkpoint flush per step (a step = one epoch)",
                "agg": "max(price), count group by 10s window (append-only)",
                "parallelism": f"dp{world}",
            },
            "roofline": {
                "bound": "hbm",
                "achieved": achieved_gbs,
                "peak": HBM_PEAK_GBS,
                "unit": "GB/s",
                "frac": achieved_gbs / HBM_PEAK_GBS,
                "traffic": Q7_TRAFFIC_B_PER_ROW * rows_per_launch,
            },
            "checkpoint": ckpt,
            "cpu_baseline": None,
        }
        if not args.skip_cpu_baseline and world == 1:
            result["cpu_baseline"] = cpu_baseline(ffi, np.random.default_rng(5))
        print(json.dumps(result), flush=True)

    agg.close()


class ExchangeCtx:
    """Wrapper over librw_exchange.so: vnode partition kernel + RCCL
    all-to-all-v (DESIGN.md §7 / SURVEY §8e)."""

    class Buffers:
        pass

    def __init__(self, lib, h, world):
        self.lib = lib
        self.h = h
        self.world = world

    def make_buffers(self, cap):
        L = self.lib
        L.rw_xbuf_alloc.restype = ctypes.c_void_p
        L.rw_xbuf_alloc.argtypes = [ctypes.c_uint64]
        b = self.Buffers()
        b.send = L.rw_xbuf_alloc(cap)
        b.recv = L.rw_xbuf_alloc(cap)
        b.cap = cap
        assert b.send and b.recv
        return b

    def run(self, agg_h, batch, xb, n_cols=2):
        L = self.lib
        if not hasattr(self, "_amd"):
            import risingwave_amd

            self._amd = ctypes.CDLL(risingwave_amd.lib_path())
            self._amd.rw_agg_batch_ptrs.argtypes = [
                ctypes.c_void_p, ctypes.POINTER(ctypes.c_void_p),
                ctypes.POINTER(ctypes.c_void_p),
                ctypes.POINTER(ctypes.c_void_p),
                ctypes.POINTER(ctypes.c_uint32)]
        A = self._amd
        NSLOT = 12  # MAX_KW + MAX_CALLS
        vals = (ctypes.c_void_p * NSLOT)()
        valids = (ctypes.c_void_p * NSLOT)()
        ops = ctypes.c_void_p()
        nrows = ctypes.c_uint32()
        A.rw_agg_batch_ptrs(batch, vals, valids, ctypes.byref(ops),
                            ctypes.byref(nrows))
        key_cols = (ctypes.c_uint32 * 1)(0)  # batch slot 0 = window key
        send_counts = (ctypes.c_uint64 * self.world)()
        recv_counts = (ctypes.c_uint64 * self.world)()
        L.rw_exchange_run.restype = ctypes.c_int
        L.rw_exchange_run.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
            ctypes.c_uint32, ctypes.c_int, ctypes.POINTER(ctypes.c_uint32),
            ctypes.c_int, ctypes.c_uint32, ctypes.c_void_p, ctypes.c_uint64,
            ctypes.c_void_p, ctypes.c_uint64, ctypes.POINTER(ctypes.c_uint64),
            ctypes.POINTER(ctypes.c_uint64)]
        rc = L.rw_exchange_run(
            self.h, ctypes.cast(vals, ctypes.c_void_p),
            ctypes.cast(valids, ctypes.c_void_p), ops, nrows, n_cols, key_cols,
            1, 256, ctypes.c_void_p(xb.send), ctypes.c_uint64(xb.cap),
            ctypes.c_void_p(xb.recv), ctypes.c_uint64(xb.cap), send_counts,
            recv_counts)
        if rc != 0:
            L.rw_exchange_last_error.restype = ctypes.c_char_p
            raise RuntimeError(f"exchange failed {rc}: "
                               f"{L.rw_exchange_last_error().decode()}")
        return recv_counts

    def stats(self):
        ms = ctypes.c_double()
        n = ctypes.c_uint64()
        self.lib.rw_exchange_stats.argtypes = [ctypes.c_void_p,
                                               ctypes.POINTER(ctypes.c_double),
                                               ctypes.POINTER(ctypes.c_uint64)]
        self.lib.rw_exchange_stats(self.h, ctypes.byref(ms), ctypes.byref(n))
        return ms.value, n.value


def setup_exchange(ffi, rank, world, dist):
    try:
        lib = ctypes.CDLL(os.path.join(REPO, "risingwave_amd",
                                       "librw_exchange.so"))
    except OSError as e:
        print(f"# librw_exchange.so load failed: {e}", file=sys.stderr)
        return None
    lib.rw_exchange_unique_id_size.restype = ctypes.c_int
    sz = lib.rw_exchange_unique_id_size()
    uid = (ctypes.c_uint8 * sz)()
    if rank == 0:
        if lib.rw_exchange_get_unique_id(uid) != 0:
            return None
    if world > 1 and dist is not None:
        obj = [bytes(bytearray(uid))] if rank == 0 else [None]
        dist.broadcast_object_list(obj, src=0)
        uid = (ctypes.c_uint8 * sz).from_buffer_copy(obj[0])
    lib.rw_exchange_create.restype = ctypes.c_void_p
    h = lib.rw_exchange_create(world, rank, uid)
    if not h:
        return None
    return ExchangeCtx(lib, h, world)


def bench_q7pipe(args, ffi, gpu_lib, rng, rank, world, dist):
    """The FULL Nexmark q7 stream plan, device-resident (reference
    stream_plan, nexmark.yaml q7 block): bid -> project(window,price) ->
    [vnode hop] -> StreamHashAgg[append_only] max(price) by 10s window ->
    flush change stream -> StreamHashJoin Inner on bid.price = max(price)
    (right input = the agg's U-/U+ stream) with the StreamFilter
    `date_time BETWEEN $expr1-10s AND $expr1` fused into inner emission,
    plus the bid -> [vnode hop] -> join-left input. A step = one 1M-row
    bid batch through BOTH branches; agg flush + join-right apply at every
    --barrier-every steps. value = bid input rows/s."""
    import ctypes

    from rwtest.ffi import AGG_COUNT_STAR, AGG_MAX, CMP_GE, CMP_LE, \
        SIDE_LEFT, SIDE_RIGHT, JOIN_INNER, T_I64

    L = gpu_lib.lib
    L.rw_agg_bench_preload.restype = ctypes.c_void_p
    L.rw_agg_bench_preload.argtypes = [ctypes.c_void_p, ctypes.POINTER(ffi.RwChunkC)]
    L.rw_join_bench_preload.restype = ctypes.c_void_p
    L.rw_join_bench_preload.argtypes = [ctypes.c_void_p, ctypes.c_int,
                                        ctypes.POINTER(ffi.RwChunkC)]
    L.rw_join_bench_apply.restype = ctypes.c_int
    L.rw_join_bench_apply.argtypes = [ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p]
    L.rw_join_bench_drain.restype = ctypes.c_longlong
    L.rw_join_bench_drain.argtypes = [ctypes.c_void_p]
    L.rw_agg_bench_apply.restype = ctypes.c_int
    L.rw_agg_bench_apply.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
    L.rw_agg_flush_device.restype = ctypes.c_longlong
    L.rw_agg_flush_device.argtypes = [ctypes.c_void_p, ctypes.c_uint64]
    L.rw_join_apply_aggout.restype = ctypes.c_int
    L.rw_join_apply_aggout.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                       ctypes.c_int,
                                       ctypes.POINTER(ctypes.c_uint32),
                                       ctypes.c_int, ctypes.c_uint64]
    L.rw_join_vnode_hop.restype = ctypes.c_int
    L.rw_join_vnode_hop.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                    ctypes.c_uint32, ctypes.c_uint8,
                                    ctypes.c_uint32]
    L.rw_join_kernel_stats.argtypes = [ctypes.c_void_p, ctypes.POINTER(KernelStats)]
    L.rw_join_stats_reset.argtypes = [ctypes.c_void_p]
    L.rw_agg_kernel_stats.argtypes = [ctypes.c_void_p, ctypes.POINTER(KernelStats)]

    batch_rows = CHUNK_ROWS * CHUNKS_PER_BATCH
    calls = [(AGG_MAX, 1, T_I64), (AGG_COUNT_STAR, -1, T_I64)]
    agg = ffi.HashAgg(gpu_lib, [T_I64, T_I64], [0], calls, 1,
                      append_only=True, state_capacity_hint=1 << 20)
    # join: left = bid [auction,bidder,price,date_time,rowid], right = the
    # agg change stream projected to [window_end, maxprice]. Key: price =
    # max(price). Fused BETWEEN: dt >= w - 10s AND dt <= w.
    t5 = [T_I64] * 5
    j = ffi.HashJoin(gpu_lib, JOIN_INNER, t5, [T_I64, T_I64],
                     key_l=[2], key_r=[1], pk_l=[4], pk_r=[0],
                     cond=(CMP_GE, 3, 5, -WINDOW_US), cond2=(CMP_LE, 3, 5, 0),
                     state_capacity_hint=1 << 23,
                     row_capacity_hint=(args.steps + args.warmup + 4)
                     * batch_rows + 1_000_000)

    n_batches = EPOCH_BATCHES
    ones = lambda n: np.ones(n, np.uint8)
    agg_batches, join_batches = [], []
    rowid = 0
    for b in range(n_batches):
        # monotone date_time: windows_per_epoch windows per epoch
        dt = np.sort(rng.integers(b * args.windows_per_epoch * WINDOW_US,
                                  (b + 1) * args.windows_per_epoch * WINDOW_US,
                                  batch_rows))
        w = (dt // WINDOW_US + 1) * WINDOW_US  # TumbleStart + 10s = $expr1
        price = rng.integers(1, 10**7, batch_rows)
        auction = rng.integers(0, 1_000_000, batch_rows)
        bidder = rng.integers(0, 1_000_000, batch_rows)
        rid = np.arange(rowid, rowid + batch_rows)
        rowid += batch_rows
        ca = ffi.Chunk([T_I64, T_I64], np.zeros(batch_rows, np.uint8),
                       [w, price], [ones(batch_rows)] * 2)
        cc = ca.to_c()
        h = L.rw_agg_bench_preload(agg.h, ctypes.byref(cc))
        assert h, gpu_lib.last_error()
        agg_batches.append(h)
        cj = ffi.Chunk(t5, np.zeros(batch_rows, np.uint8),
                       [auction, bidder, price, dt, rid],
                       [ones(batch_rows)] * 5)
        cc = cj.to_c()
        h = L.rw_join_bench_preload(j.h, SIDE_LEFT, ctypes.byref(cc))
        assert h, gpu_lib.last_error()
        join_batches.append(h)

    cmap = (ctypes.c_uint32 * 2)(0, 1)  # agg record [w, max, count] -> [w, max]
    agg_ms = [0.0]
    # the whole step loop runs in C (rw_q7pipe_bench_run): python-side
    # ctypes dispatch costs ~30-40 us/step, ~20% at this step size
    L.rw_q7pipe_bench_run.restype = ctypes.c_int
    L.rw_q7pipe_bench_run.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.POINTER(ctypes.c_void_p),
        ctypes.POINTER(ctypes.c_void_p), ctypes.c_int, ctypes.c_int,
        ctypes.c_int, ctypes.POINTER(ctypes.c_uint32), ctypes.c_int,
        ctypes.c_int]
    a_arr = (ctypes.c_void_p * n_batches)(*agg_batches)
    j_arr = (ctypes.c_void_p * n_batches)(*join_batches)

    def run_steps(n, step0):
        rc = L.rw_q7pipe_bench_run(agg.h, j.h, a_arr, j_arr, n_batches, n,
                                   args.barrier_every, cmap, 2, step0)
        assert rc == 0, gpu_lib.last_error()

    run_steps(args.warmup, 0)
    # warm the maintenance path: the FIRST rw_join_compact lazily
    # allocates its ping-pong store (tens of GB at this row capacity;
    # hundreds of ms after allocator churn) — a one-time job-lifetime
    # cost that must not land inside the steady-state timed window
    if int(os.environ.get("RW_Q7PIPE_COMPACT", "4")) > 0:
        from rwtest.ffi import (join_checkpoint_drain, join_compact,
                                join_degree_drain)
        join_checkpoint_drain(gpu_lib, j.h, SIDE_RIGHT)
        join_degree_drain(gpu_lib, j.h, SIDE_RIGHT)
        join_compact(gpu_lib, j.h, SIDE_RIGHT)
    L.rw_join_stats_reset(j.h)
    L.rw_agg_stats_reset(agg.h)
    if dist:
        dist.barrier()
    _dev_sync()
    t0 = time.perf_counter()
    run_steps(args.steps, args.warmup)
    assert L.rw_join_bench_drain(j.h) >= 0, gpu_lib.last_error()
    _dev_sync()
    elapsed = time.perf_counter() - t0
    if dist:
        import torch

        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    jks = KernelStats()
    aks = KernelStats()
    L.rw_join_kernel_stats(j.h, ctypes.byref(jks))
    L.rw_agg_kernel_stats(agg.h, ctypes.byref(aks))
    if rank == 0:
        total_rows = args.steps * batch_rows * world
        avg_probe_ms = jks.total_ms / max(jks.launches, 1)
        # dominant kernel = the join-left probe+insert: ~120 B/row
        # algorithmic (5x8 cols + 5 valid + op reads, 64-B record write,
        # 8-B slot CAS; the right table is LIC-resident)
        achieved = (120 * batch_rows) / (avg_probe_ms * 1e-3) / 1e9
        result = {
            "metric": "input rows/sec/GPU on Nexmark q7 full stream plan",
            "value": total_rows / elapsed,
            "unit": "rows/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed * 1000.0 / args.steps,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "int64",
            "data": "synthetic",
            "config": {
                "workload": "nexmark_q7_pipeline",
                "plan": "project -> vnode hop -> hash-agg max(price) by 10s "
                        "window -> change stream -> inner join price = "
                        "maxprice (BETWEEN fused) <- vnode hop <- bid",
                "rows_per_step": batch_rows,
                "windows_per_epoch": args.windows_per_epoch,
                "barrier_every_steps": args.barrier_every,
                "parallelism": f"dp{world}",
            },
            "kernels": {
                "join_probe_avg_ms": avg_probe_ms,
                "join_launches": jks.launches,
                "agg_apply_avg_ms": aks.total_ms / max(aks.launches, 1),
                "barrier_flush_plus_aggout_ms_total": agg_ms[0] * 1000.0,
            },
            "roofline": {
                "bound": "hbm",
                "achieved": achieved,
                "peak": HBM_PEAK_GBS,
                "unit": "GB/s",
                "frac": achieved / HBM_PEAK_GBS,
                "traffic": None,
            },
            "cpu_baseline": None,
        }
        print(json.dumps(result), flush=True)
    j.close()
    agg.close()


def bench_q8(args, ffi, gpu_lib, rng, rank, world, dist):
    """Nexmark q8 stream-stream hash-join (BASELINE configs[2]): build side =
    10M distinct person ids resident in HBM, probe = auction.seller
    zipf(1.1) over the same id space; join key = (id, window_start,
    window_end) = Key256 class. value = probe input rows/s;
    join outputs stay in HBM for the device-resident downstream operator."""
    import ctypes

    from rwtest.ffi import JOIN_INNER, SIDE_LEFT, SIDE_RIGHT, T_I64, T_TS

    L = gpu_lib.lib
    L.rw_join_bench_preload.restype = ctypes.c_void_p
    L.rw_join_bench_preload.argtypes = [ctypes.c_void_p, ctypes.c_int,
                                        ctypes.POINTER(ffi.RwChunkC)]
    L.rw_join_bench_apply.restype = ctypes.c_int
    L.rw_join_bench_apply.argtypes = [ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p]
    L.rw_join_bench_drain.restype = ctypes.c_longlong
    L.rw_join_bench_drain.argtypes = [ctypes.c_void_p]
    L.rw_join_kernel_stats.argtypes = [ctypes.c_void_p, ctypes.POINTER(KernelStats)]

    BUILD_KEYS = 10_000_000
    # PMC-measured HBM traffic of the probe kernel at this config
    # (profiles/r02_q8_pmc_{fetch,write}_hint25.txt: per-1M-row probe
    # launch averages with the 10 build launches subtracted out; raw
    # FETCH+WRITE, random narrow reads are uncalibrated on gfx950 so no
    # x2 correction is applied — MI355X_MICROARCH.md §HBM): the gap to
    # the 128-B/row algorithmic model is 128-B line granularity on the
    # random touches (match record + own insert RFO) plus the
    # pre-assigned layout's sparse-region writes. The register-resident
    # walk removed the former eviction refetches (was 622 B/row fetched;
    # 602 at the LIC-resident hint-23 tables, 551 at the default 2^26
    # low-load tables whose shorter chains re-walk less).
    Q8_TRAFFIC_B_PER_ROW = (276_081 + 288_165) * 1024 / 1_048_576
    batch_rows = CHUNK_ROWS * CHUNKS_PER_BATCH
    t4 = [T_I64, T_TS, T_TS, T_I64]
    # hint 2^23 -> cap 2^24 slots x 8 B = 134 MB per side: BOTH slot tables
    # sit in the 256 MB Infinity Cache (load factor 0.6 at 10M keys)
    # RW_Q8_HINT_LOG2 A/B hook. Default 25 -> cap 2^26 slots/side
    # (512 MB, load factor 0.15): shorter chains beat LIC residency —
    # measured 0.238 vs 0.245 ms/step against hint 23's LIC-resident
    # 2^24 tables (22/23/24/25 -> 0.262/0.245/0.242/0.238).
    hint_log2 = int(os.environ.get("RW_Q8_HINT_LOG2", "25"))
    j = ffi.HashJoin(gpu_lib, JOIN_INNER, t4, t4, key_l=[0, 1, 2],
                     key_r=[0, 1, 2], pk_l=[3], pk_r=[3],
                     state_capacity_hint=1 << hint_log2,
                     row_capacity_hint=BUILD_KEYS + (args.steps + args.warmup + 4)
                     * batch_rows + 1_000_000)

    WS = 1_000 * WINDOW_US
    ones = lambda n: np.ones(n, np.uint8)

    def preload(side, ids, rowid0):
        n = len(ids)
        c = ffi.Chunk(t4, np.zeros(n, np.uint8),
                      [ids, np.full(n, WS), np.full(n, WS + WINDOW_US),
                       np.arange(rowid0, rowid0 + n)],
                      [ones(n)] * 4)
        cc = c.to_c()
        h = L.rw_join_bench_preload(j.h, side, ctypes.byref(cc))
        assert h, gpu_lib.last_error()
        return h

    # build side (right): 10M distinct person ids, ingested untimed
    rowid = 0
    for lo in range(0, BUILD_KEYS, batch_rows):
        ids = np.arange(lo, min(lo + batch_rows, BUILD_KEYS), dtype=np.int64)
        h = preload(SIDE_RIGHT, ids, rowid)
        rowid += len(ids)
        rc = L.rw_join_bench_apply(j.h, SIDE_RIGHT, h)
        assert rc == 0, gpu_lib.last_error()
        assert L.rw_join_bench_drain(j.h) >= 0, gpu_lib.last_error()

    # probe batches (left): the join's left input is the auction-side AGG
    # OUTPUT (q8 plan, nexmark.yaml: StreamHashAgg group_key [seller, ws, we]
    # with noop_update_hint), so keys are UNIQUE per batch — the zipf skew of
    # raw auction.seller is absorbed by that upstream agg. Sample which
    # sellers appear via a shuffled id space. (8 × ~36 MB ≈ 290 MB resident
    # > L3, so probes stream from HBM.)
    n_batches = 8
    perm = rng.permutation(BUILD_KEYS).astype(np.int64)
    batches = []
    for b in range(n_batches):
        lo = (b * batch_rows) % (BUILD_KEYS - batch_rows)
        ids = perm[lo:lo + batch_rows]
        batches.append(preload(SIDE_LEFT, ids, rowid))
        rowid += batch_rows

    # A/B hook: RW_JOIN_SKIP_PROBE applies RW_JOIN_SKIP only to the PROBE
    # phase (the build above ran unskipped, so the tables are populated —
    # setting RW_JOIN_SKIP outside would also skip the build inserts and
    # measure an empty-table probe)
    if os.environ.get("RW_JOIN_SKIP_PROBE"):
        os.environ["RW_JOIN_SKIP"] = os.environ["RW_JOIN_SKIP_PROBE"]
    # the step loop runs in C with a stream-ordered cursor reset per step —
    # the Python-side per-step drain cost ~0.15 ms of pure sync overhead
    L.rw_join_bench_run.restype = ctypes.c_int
    L.rw_join_bench_run.argtypes = [ctypes.c_void_p, ctypes.c_int,
                                    ctypes.POINTER(ctypes.c_void_p),
                                    ctypes.c_int, ctypes.c_int]
    barr = (ctypes.c_void_p * n_batches)(*batches)

    def run_steps(n):
        rc = L.rw_join_bench_run(j.h, SIDE_LEFT, barr, n_batches, n)
        assert rc == 0, gpu_lib.last_error()
        n_emit = L.rw_join_bench_drain(j.h)  # sync + overflow check
        assert n_emit >= 0, gpu_lib.last_error()
        return n_emit

    run_steps(args.warmup)
    L.rw_join_stats_reset.argtypes = [ctypes.c_void_p]
    L.rw_join_stats_reset(j.h)
    if dist:
        dist.barrier()
    t0 = time.perf_counter()
    last = run_steps(args.steps)  # drain returns the LAST step's emits
    elapsed = time.perf_counter() - t0
    matches = last * args.steps  # per-step emit count is deterministic here
    if dist:
        import torch

        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    ks = KernelStats()
    L.rw_join_kernel_stats(j.h, ctypes.byref(ks))
    if rank == 0:
        total_rows = args.steps * batch_rows * world
        avg_launch_ms = ks.total_ms / max(ks.launches, 1)
        # ≈128 B algorithmic per probe row at match-rate 1
        achieved_gbs = (128 * batch_rows) / (avg_launch_ms * 1e-3) / 1e9
        result = {
            "metric": "input rows/sec/GPU on Nexmark q8 stream",
            "value": total_rows / elapsed,
            "unit": "rows/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed * 1000.0 / args.steps,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "int64",
            "data": "synthetic",
            "config": {
                "workload": "nexmark_q8",
                "build_keys": BUILD_KEYS,
                "probe_rows_per_step": batch_rows,
                "probe_dist": "agg-output key-unique per batch (zipf absorbed upstream)",
                "join_key": "(id i64, ws ts, we ts)",
                "match_rate": matches / max(args.steps * batch_rows, 1),
                "parallelism": f"dp{world}",
            },
            "roofline": {
                "bound": "hbm",
                "achieved": achieved_gbs,
                "peak": HBM_PEAK_GBS,
                "unit": "GB/s",
                "frac": achieved_gbs / HBM_PEAK_GBS,
                "traffic": Q8_TRAFFIC_B_PER_ROW * batch_rows,
            },
            "cpu_baseline": None,
        }
        if not args.skip_cpu_baseline and world == 1:
            result["cpu_baseline"] = cpu_baseline_q8(ffi, np.random.default_rng(5))
        print(json.dumps(result))
    j.close()


def bench_q3(args, ffi, gpu_lib, rng, rank, world, dist):
    """TPC-H-stream q3 (BASELINE configs[4], minus the Hummock spill):
    orders ⋈ lineitem inner join feeding sum(revenue)/count group by
    (orderkey, orderdate, shippriority), 100M-key order state in HBM,
    ~10% Delete mix on the lineitem stream, checkpoint flush every
    --barrier-every steps. The join's output block feeds the agg in HBM
    (rw_agg_apply_joinout). value = lineitem input rows/s."""
    import ctypes

    from rwtest.ffi import AGG_COUNT_STAR, AGG_SUM, JOIN_INNER, SIDE_LEFT, \
        SIDE_RIGHT, T_I64

    L = gpu_lib.lib
    L.rw_join_bench_preload.restype = ctypes.c_void_p
    L.rw_join_bench_preload.argtypes = [ctypes.c_void_p, ctypes.c_int,
                                        ctypes.POINTER(ffi.RwChunkC)]
    L.rw_join_bench_apply.restype = ctypes.c_int
    L.rw_join_bench_apply.argtypes = [ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p]
    L.rw_join_bench_drain.restype = ctypes.c_longlong
    L.rw_join_bench_drain.argtypes = [ctypes.c_void_p]
    L.rw_join_kernel_stats.argtypes = [ctypes.c_void_p, ctypes.POINTER(KernelStats)]
    L.rw_join_stats_reset.argtypes = [ctypes.c_void_p]
    L.rw_agg_apply_joinout.restype = ctypes.c_int
    L.rw_agg_apply_joinout.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
    L.rw_agg_flush_device.restype = ctypes.c_longlong
    L.rw_agg_flush_device.argtypes = [ctypes.c_void_p, ctypes.c_uint64]

    N_ORDERS = args.q3_orders
    batch_rows = CHUNK_ROWS * CHUNKS_PER_BATCH
    # lineitem: [l_orderkey, revenue, l_rowid]; orders: [o_orderkey, o_date,
    # o_prio, o_rowid]; join on orderkey; output = all 7 columns
    tl = [T_I64, T_I64, T_I64]
    tr = [T_I64, T_I64, T_I64, T_I64]
    j = ffi.HashJoin(gpu_lib, JOIN_INNER, tl, tr, key_l=[0], key_r=[0],
                     pk_l=[2], pk_r=[3],
                     state_capacity_hint=max(N_ORDERS, 1 << 22),
                     row_capacity_hint=N_ORDERS +
                     (args.steps + args.warmup + 4) * batch_rows + 1_000_000)
    # agg over the join output: group (o_orderkey, o_date, o_prio) = cols
    # 3,4,5 of the concat row; sum(revenue)=col 1; count(*)
    agg = ffi.HashAgg(gpu_lib, [T_I64] * 7, [3, 4, 5],
                      [(AGG_SUM, 1, T_I64), (AGG_COUNT_STAR, -1, T_I64)],
                      row_count_index=1, state_capacity_hint=1 << 23)

    ones = lambda n: np.ones(n, np.uint8)

    def preload(side, cols, ops=None):
        n = len(cols[0])
        c = ffi.Chunk(tl if side == SIDE_LEFT else tr,
                      np.zeros(n, np.uint8) if ops is None else ops, cols,
                      [ones(n)] * len(cols))
        cc = c.to_c()
        h = L.rw_join_bench_preload(j.h, side, ctypes.byref(cc))
        assert h, gpu_lib.last_error()
        return h

    # build orders (untimed)
    for lo in range(0, N_ORDERS, batch_rows):
        hi = min(lo + batch_rows, N_ORDERS)
        ok = np.arange(lo, hi, dtype=np.int64)
        h = preload(SIDE_RIGHT, [ok, ok % 2557, ok % 3, ok])
        rc = L.rw_join_bench_apply(j.h, SIDE_RIGHT, h)
        assert rc == 0, gpu_lib.last_error()
        assert L.rw_join_bench_drain(j.h) >= 0, gpu_lib.last_error()

    # lineitem batches: ~10% deletes of rows inserted by the previous batch,
    # orderkeys in a sliding 512k-order window (bounds dirty groups/epoch)
    n_batches = 8
    WINDOW = 1 << 19
    batches = []
    rowid = 0
    prev_insert = None
    for b in range(n_batches):
        n = batch_rows
        base = (b * WINDOW) % max(N_ORDERS - WINDOW, 1)
        ok = base + rng.integers(0, WINDOW, n)
        rev = rng.integers(1, 100_000, n)
        rid = np.arange(rowid, rowid + n)
        rowid += n
        ops = np.zeros(n, np.uint8)
        if prev_insert is not None:
            n_del = n // 10
            sel = rng.choice(len(prev_insert[0]), n_del, replace=False)
            ok[:n_del] = prev_insert[0][sel]
            rev[:n_del] = prev_insert[1][sel]
            rid[:n_del] = prev_insert[2][sel]
            ops[:n_del] = ffi.OP_DELETE
        prev_insert = (ok[n // 10:].copy(), rev[n // 10:].copy(),
                       rid[n // 10:].copy())
        batches.append(preload(SIDE_LEFT, [ok, rev, rid], ops))

    emitted = [0]

    def step(i):
        rc = L.rw_join_bench_apply(j.h, SIDE_LEFT, batches[i % n_batches])
        assert rc == 0, gpu_lib.last_error()
        rc = L.rw_agg_apply_joinout(agg.h, j.h)
        assert rc == 0, gpu_lib.last_error()
        if (i + 1) % args.barrier_every == 0:
            # checkpoint barrier: change inference + emission into HBM (the
            # downstream operator is device-resident; host marshalling is
            # the parity-test path, not the pipeline)
            n = L.rw_agg_flush_device(agg.h, i)
            assert n >= 0, gpu_lib.last_error()
            emitted[0] += n

    for i in range(args.warmup):
        step(i)
    L.rw_join_stats_reset(j.h)
    L.rw_agg_stats_reset(agg.h)
    if dist:
        dist.barrier()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(i)
    L.rw_agg_sync(agg.h)
    elapsed = time.perf_counter() - t0
    if dist:
        import torch

        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    jks = KernelStats()
    aks = KernelStats()
    L.rw_join_kernel_stats(j.h, ctypes.byref(jks))
    L.rw_agg_kernel_stats(agg.h, ctypes.byref(aks))
    if rank == 0:
        total_rows = args.steps * batch_rows * world
        avg_probe_ms = jks.total_ms / max(jks.launches, 1)
        achieved = (128 * batch_rows) / (avg_probe_ms * 1e-3) / 1e9
        result = {
            "metric": "input rows/sec/GPU on TPC-H-stream q3",
            "value": total_rows / elapsed,
            "unit": "rows/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed * 1000.0 / args.steps,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "int64",
            "data": "synthetic",
            "config": {
                "workload": "tpch_q3_stream",
                "orders": N_ORDERS,
                "lineitem_rows_per_step": batch_rows,
                "delete_mix": 0.1,
                "barrier_every_steps": args.barrier_every,
                "pipeline": "join(orders ⋈ lineitem) → agg(sum,count by "
                            "orderkey,date,prio) in HBM",
                "parallelism": f"dp{world}",
            },
            "roofline": {
                "bound": "hbm",
                "achieved": achieved,
                "peak": HBM_PEAK_GBS,
                "unit": "GB/s",
                "frac": achieved / HBM_PEAK_GBS,
                "traffic": None,
            },
            "kernels": {
                "join_probe_avg_ms": avg_probe_ms,
                "agg_apply_avg_ms": aks.total_ms / max(aks.launches, 1),
                "emitted_rows": emitted[0],
            },
            "cpu_baseline": None,
        }
        print(json.dumps(result))
    j.close()
    agg.close()


def cpu_baseline_q8(ffi, rng, target_seconds=8.0):
    """Oracle join on a scaled-down q8 sample: build 200k keys, probe 4K-row
    chunks with the same zipf shape; single thread, kind 'port'."""
    from rwtest.ffi import JOIN_INNER, SIDE_LEFT, SIDE_RIGHT, T_I64, T_TS, oracle

    BUILD = 200_000
    t4 = [T_I64, T_TS, T_TS, T_I64]
    j = ffi.HashJoin(oracle(), JOIN_INNER, t4, t4, key_l=[0, 1, 2],
                     key_r=[0, 1, 2], pk_l=[3], pk_r=[3])
    WS = 1_000 * WINDOW_US
    ones = np.ones(CHUNK_ROWS, np.uint8)

    def chunk(ids, rowid0):
        n = len(ids)
        return ffi.Chunk(t4, np.zeros(n, np.uint8),
                         [ids, np.full(n, WS), np.full(n, WS + WINDOW_US),
                          np.arange(rowid0, rowid0 + n)],
                         [np.ones(n, np.uint8)] * 4)

    rowid = 0
    for lo in range(0, BUILD, CHUNK_ROWS):
        ids = np.arange(lo, min(lo + CHUNK_ROWS, BUILD), dtype=np.int64)
        j.push(SIDE_RIGHT, chunk(ids, rowid))
        j.poll_all()
        rowid += len(ids)

    probe = chunk(rng.permutation(BUILD)[:CHUNK_ROWS].astype(np.int64), rowid)
    t0 = time.perf_counter()
    j.push(SIDE_LEFT, probe)
    j.poll_all()
    per_chunk = time.perf_counter() - t0
    n = max(8, min(int(target_seconds / max(per_chunk, 1e-9)), 100_000))
    t0 = time.perf_counter()
    for i in range(n):
        j.push(SIDE_LEFT, probe)
        j.poll_all()
    dt = time.perf_counter() - t0
    j.close()
    rows = n * CHUNK_ROWS
    return {
        "value": rows / dt,
        "unit": "rows/s",
        "cores": 1,
        "kind": "port",
        "sample": f"{rows} q8 probe rows over a 200k-key build side "
                  f"({dt:.1f}s single-thread oracle)",
    }


def _cuda_available():
    try:
        import torch

        return torch.cuda.is_available()
    except Exception:
        return False


def _dev_sync():
    # rw_agg_sync drains the executor's stream; also sync the device per the
    # driver contract when torch sees the GPU
    try:
        import torch

        if torch.cuda.is_available():
            torch.cuda.synchronize()
    except Exception:
        pass


if __name__ == "__main__":
    main()
