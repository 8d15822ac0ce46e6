"""Multi-process (world_size=2, gloo/CPU) coverage of the distributed
sharding pattern bench.py uses for N>1: independent per-rank stripe batches
(weak scaling), barrier-bracketed timing, MAX-over-ranks reduction,
whole-job aggregation.  No GPU: the per-rank compute leg is the oracle.
"""
import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp


def _worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=world)
    import sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sys.path.insert(0, os.path.join(repo, "oracle"))
    import oracle

    k, m, plen, stripes = 4, 2, 4096, 3
    rng = np.random.default_rng(100 + rank)   # per-rank shard
    data = rng.integers(0, 256, (stripes, k, plen), np.uint8)
    parity = np.zeros((stripes, m, plen), np.uint8)
    tbl, _, _ = oracle.rs_make_tables(k, m, (1 << k) - 1, (1 << k) - 1,
                                      ((1 << m) - 1) << k)

    torch.distributed.barrier()
    import time
    t0 = time.perf_counter()
    oracle.encode_stripes(k, m, plen, stripes, tbl, data, parity)
    elapsed = torch.tensor([time.perf_counter() - t0], dtype=torch.float64)
    torch.distributed.barrier()
    torch.distributed.all_reduce(elapsed, op=torch.distributed.ReduceOp.MAX)

    # every rank's shard must be correct (spot check one stripe)
    exp = oracle.rs_encode(k, m, list(data[0]), plen)
    ok = all(np.array_equal(parity[0, l], exp[l]) for l in range(m))

    total_units = stripes * world  # whole-job aggregate
    torch.distributed.destroy_process_group()
    q.put((rank, ok, float(elapsed.item()), total_units))


def test_weak_scaling_pattern_gloo_world2():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29517
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert len(results) == 2
    for rank, ok, elapsed, total in results:
        assert ok, f"rank {rank} shard wrong"
        assert elapsed > 0
        assert total == 6


def _mixed_worker(rank, world, port, q):
    """bench.py --op mixed exchange structure on CPU: per-rank encode +
    recover of erased data parts, then all_gather of the recovered parts
    (SURVEY §8e — the one real exchange in the path); every rank verifies
    BOTH gathered slots bit-exactly by recomputing each rank's
    deterministic shard."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=world)
    import sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sys.path.insert(0, os.path.join(repo, "oracle"))
    import oracle

    k, m, plen, stripes = 4, 2, 4096, 2
    erased = (1, 3)

    def shard(r):
        rng = np.random.default_rng(200 + r)
        return rng.integers(0, 256, (stripes, k, plen), np.uint8)

    data = shard(rank)
    parity = np.zeros((stripes, m, plen), np.uint8)
    tbl, _, _ = oracle.rs_make_tables(k, m, (1 << k) - 1, (1 << k) - 1,
                                      ((1 << m) - 1) << k)
    oracle.encode_stripes(k, m, plen, stripes, tbl, data, parity)
    # recover the erased data parts from the survivors
    nparts = k + m
    present = sum(1 << i for i in range(nparts) if i not in erased)
    needed = sum(1 << i for i in erased)
    rtbl, ic, oc = oracle.rs_make_tables(k, m, present, present, needed)
    parts = np.concatenate([data, parity], axis=1)
    srcs = np.ascontiguousarray(
        parts[:, [i for i in range(nparts) if i not in erased][:ic]])
    rec = np.zeros((stripes, oc, plen), np.uint8)
    oracle.encode_stripes(ic, oc, plen, stripes, rtbl, srcs, rec)

    gathered = [[torch.empty(stripes, plen, dtype=torch.uint8)
                 for _ in range(world)] for _ in erased]
    for gi in range(len(erased)):
        torch.distributed.all_gather(
            gathered[gi], torch.from_numpy(np.ascontiguousarray(rec[:, gi])))

    ok = True
    for r in range(world):
        exp = shard(r)
        for gi, i in enumerate(erased):
            if not np.array_equal(gathered[gi][r].numpy(), exp[:, i]):
                ok = False
    torch.distributed.destroy_process_group()
    q.put((rank, ok))


def test_mixed_allgather_pattern_gloo_world2():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_mixed_worker, args=(r, 2, 29519, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    for rank, ok in results:
        assert ok, f"rank {rank}: gathered recovered parts wrong"
