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
