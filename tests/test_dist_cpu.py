"""Multi-process (gloo, world_size=2) CPU tests of the distributed
TeraSort control flow — the same sample -> splitter -> classify ->
all-to-all -> local sort -> concatenate protocol thrill_amd/pipeline.py
runs on GPUs, exercised here with the oracle as the compute so the
protocol logic (counts, displacements, boundary ordering, tiebreaks) is
covered without a GPU. Mirrors the reference's RunLocalTests style
(api/context.cpp:319-374: multi-worker simulation in one process).
"""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from thrill_amd.pipeline import sample_size, select_splitters

WORLD = 2


def _worker(rank, world, port, n_total, seed, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        from tests._oracle import Oracle
        o = Oracle()

        base = n_total // world
        rem = n_total % world
        n_local = base + (1 if rank < rem else 0)
        gidx0 = rank * base + min(rank, rem)
        recs = o.gen_records(n_local, seed=seed, index0=gidx0)

        # sample (stride) and gather to rank 0 — pipeline._splitters logic
        S = max(1, sample_size(n_total) // world)
        stride = max(1, n_local // S)
        pos = np.arange(0, n_local, stride)[:S]
        samp = recs[pos]
        gidx = (pos + gidx0).astype(np.uint64)
        gathered = [None] * world
        dist.all_gather_object(gathered, (samp, gidx))
        spl = [None]
        if rank == 0:
            all_recs = np.concatenate([g[0] for g in gathered])
            all_idx = np.concatenate([g[1] for g in gathered])
            spl = [select_splitters(all_recs, all_idx, world)]
        dist.broadcast_object_list(spl, src=0)
        spl_recs, spl_idx = spl[0]

        # classify with the oracle (full-record acceptance order:
        # key_len = rec_size) and exchange
        bucket = o.classify_rec(recs, gidx0, 100, spl_recs, spl_idx, world)
        parts = [recs[bucket == b] for b in range(world)]
        recvd = [None] * world
        # all-to-all via per-pair gather_object
        for b in range(world):
            obj = [None] * world
            dist.all_gather_object(obj, parts[b])
            if rank == b:
                recvd = obj
        mine = np.concatenate([r for r in recvd if len(r)]) \
            if any(len(r) for r in recvd) else np.empty((0, 100), np.uint8)
        out = o.sort_records(mine) if len(mine) else mine

        # local checks + hand results to the parent
        if len(out) > 1:
            lex = [tuple(r.tolist()) for r in out[:: max(1, len(out) // 50)]]
            assert lex == sorted(lex)
        q.put((rank, out.tobytes(), len(out)))
        dist.barrier()
        dist.destroy_process_group()
    except Exception as e:  # surface the failure to the parent
        q.put((rank, f"ERROR: {e!r}", -1))
        raise


@pytest.mark.parametrize("n_total", [1000, 5003])
def test_distributed_terasort_protocol(n_total):
    seed = 0x33
    ctxm = mp.get_context("spawn")
    q = ctxm.Queue()
    port = 29511 + n_total % 100
    procs = [ctxm.Process(target=_worker,
                          args=(r, WORLD, port, n_total, seed, q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(WORLD):
        rank, data, cnt = q.get(timeout=180)
        assert cnt >= 0, data
        results[rank] = (np.frombuffer(data, np.uint8).reshape(cnt, 100),
                         cnt)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    # global result: concatenation by rank == oracle total sort
    from tests._oracle import Oracle
    o = Oracle()
    full = o.gen_records(n_total, seed=seed)
    expect = o.sort_records(full)
    got = np.concatenate([results[r][0] for r in range(WORLD)])
    assert got.shape == expect.shape
    assert np.array_equal(got, expect)


def _counts_worker(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        from thrill_amd.pipeline import exchange_counts, displs_of
        # deterministic matrix: rank r sends (r*10 + dest) items to dest
        send = np.array([rank * 10 + d for d in range(world)],
                        dtype=np.int64)
        recv = exchange_counts(dist, send, rank, world)
        expect = np.array([s * 10 + rank for s in range(world)],
                          dtype=np.int64)
        assert np.array_equal(recv, expect), (recv, expect)
        d = displs_of(recv.astype(np.uint64))
        assert d[0] == 0 and np.all(np.diff(d.astype(np.int64))
                                    == recv[:-1])
        q.put((rank, "ok"))
        dist.barrier()
        dist.destroy_process_group()
    except Exception as e:
        q.put((rank, f"ERROR: {e!r}"))
        raise


def test_exchange_counts_gloo_world3():
    """the control-plane count exchange (all-gather matrix, my column)
    that replaces all_to_all_single(counts) — gloo, world 3."""
    ctxm = mp.get_context("spawn")
    q = ctxm.Queue()
    procs = [ctxm.Process(target=_counts_worker, args=(r, 3, 29713, q))
             for r in range(3)]
    for p in procs:
        p.start()
    for _ in range(3):
        rank, msg = q.get(timeout=120)
        assert msg == "ok", msg
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0


def test_distributed_terasort_protocol_world3():
    """world_size=3 with ragged shard sizes (n % 3 != 0)."""
    global WORLD
    seed, n_total = 0x55, 2003
    ctxm = mp.get_context("spawn")
    q = ctxm.Queue()
    procs = [ctxm.Process(target=_worker,
                          args=(r, 3, 29611, n_total, seed, q))
             for r in range(3)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(3):
        rank, data, cnt = q.get(timeout=180)
        assert cnt >= 0, data
        results[rank] = np.frombuffer(data, np.uint8).reshape(cnt, 100)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    from tests._oracle import Oracle
    o = Oracle()
    expect = o.sort_records(o.gen_records(n_total, seed=seed))
    got = np.concatenate([results[r] for r in range(3)])
    assert np.array_equal(got, expect)


def test_distributed_terasort_protocol_world8():
    """world_size=8 — the driver's scaling shape — with a small ragged
    input; full sample->splitter->classify->exchange->sort protocol on
    the oracle, bit-exact concatenation."""
    seed, n_total = 0x88, 4003
    ctxm = mp.get_context("spawn")
    q = ctxm.Queue()
    procs = [ctxm.Process(target=_worker,
                          args=(r, 8, 29817, n_total, seed, q))
             for r in range(8)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(8):
        rank, data, cnt = q.get(timeout=300)
        assert cnt >= 0, data
        results[rank] = np.frombuffer(data, np.uint8).reshape(cnt, 100)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    from tests._oracle import Oracle
    o = Oracle()
    expect = o.sort_records(o.gen_records(n_total, seed=seed))
    got = np.concatenate([results[r] for r in range(8)])
    assert np.array_equal(got, expect)
