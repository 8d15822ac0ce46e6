#!/usr/bin/env python3
"""bench.py — headline benchmark for the MI355X EC engine.

Measures BASELINE.json's metric ("GiB/s EC encode+decode, ec(8,2) 64 MiB
stripes") on BASELINE config 2: ec(8,2) encode of a batch of 64 MiB
synthetic stripes, inputs resident in HBM.  A "step" = one encode pass of
the whole batch.  value = whole-job data-in GiB/s across all ranks
(the reference's own throughput convention, reed_solomon_unittest.cc:44-73).

Usage: python bench.py [--gpus N] [--steps K] [--warmup W] [--stripes S]
N>1 is launched by the driver via torch.distributed.run (one rank per GPU,
RCCL); stripes shard across ranks as independent batches (weak scaling — the
reference processes stripes independently too; no data-path collective).

The cpu_baseline leg times the ORACLE (CPU restatement of the reference
path, OpenMP over stripes) on a bounded sample — reported context, not the
roofline target.
"""
import argparse
import ctypes
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "oracle"))

import numpy as np  # noqa: E402
import torch  # noqa: E402

K_EC = 8
M_EC = 2
STRIPE_BYTES = 64 * 1024 * 1024
PART_LEN = STRIPE_BYTES // K_EC


def log(msg):
    print(msg, file=sys.stderr, flush=True)


def cpu_baseline_leg(target_seconds=12.0):
    """Time the oracle's threaded encode on a bounded sample of the same
    workload on this box's host cores.  kind='port' (restatement of
    galois_field_encode.cc:28-47 + reed_solomon.h encode semantics)."""
    import oracle
    cores = int(os.environ.get("OMP_NUM_THREADS", os.cpu_count() or 1))
    tbl, ic, oc = oracle.rs_make_tables(
        K_EC, M_EC, (1 << K_EC) - 1, (1 << K_EC) - 1,
        ((1 << M_EC) - 1) << K_EC)
    # calibrate with one stripe, then size the sample
    rng = np.random.default_rng(42)
    data = rng.integers(0, 256, (1, K_EC, PART_LEN), np.uint8)
    parity = np.zeros((1, M_EC, PART_LEN), np.uint8)
    t0 = time.perf_counter()
    oracle.encode_stripes(K_EC, M_EC, PART_LEN, 1, tbl, data, parity)
    per_stripe = time.perf_counter() - t0
    n = max(2, min(64, int(target_seconds / max(per_stripe, 1e-3))))
    data = rng.integers(0, 256, (n, K_EC, PART_LEN), np.uint8)
    parity = np.zeros((n, M_EC, PART_LEN), np.uint8)
    t0 = time.perf_counter()
    oracle.encode_stripes(K_EC, M_EC, PART_LEN, n, tbl, data, parity)
    dt = time.perf_counter() - t0
    gib = n * STRIPE_BYTES / (1 << 30)
    return {
        "value": round(gib / dt, 3),
        "unit": "GiB/s",
        "cores": cores,
        "kind": "port",
        "sample": f"{n}x64MiB stripes ec(8,2) encode, {dt:.1f}s, OpenMP",
    }


def read_traffic_calibration(workload):
    """PMC-measured HBM bytes per launch, if a matching calibration was
    committed from a rocprofv3 run (profiles/pmc_traffic.json)."""
    path = os.path.join(REPO, "profiles", "pmc_traffic.json")
    try:
        with open(path) as f:
            d = json.load(f)
        if d.get("workload") == workload:
            return d.get("bytes_per_launch")
    except Exception:
        pass
    return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--stripes", type=int, default=1024,
                    help="stripes per rank (weak scaling)")
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    n_gpus = world if world > 1 else args.gpus
    distributed = world > 1

    if not torch.cuda.is_available():
        log("bench.py requires an MI355X (no CPU fallback on the product path)")
        sys.exit(2)

    torch.cuda.set_device(local_rank)
    if distributed:
        torch.distributed.init_process_group("nccl")

    from lizardfs_amd.ec import ReedSolomon

    S = args.stripes
    workload = (f"ec({K_EC},{M_EC}) encode, {S}x64MiB synthetic stripes/GPU, "
                f"device-resident")
    log(f"[rank {rank}] generating {S} stripes "
        f"({S * STRIPE_BYTES / (1 << 30):.0f} GiB data) on cuda:{local_rank}")
    g = torch.Generator(device="cuda").manual_seed(42 + rank)
    data = torch.randint(0, 256, (S, K_EC, PART_LEN), dtype=torch.uint8,
                         device="cuda", generator=g)
    parity = torch.empty((S, M_EC, PART_LEN), dtype=torch.uint8, device="cuda")

    rs = ReedSolomon(K_EC, M_EC, device=local_rank)

    def step():
        rs.encode_batch(data, parity)

    # warmup
    for _ in range(args.warmup):
        step()
    torch.cuda.synchronize()
    if distributed:
        torch.distributed.barrier()
    torch.cuda.synchronize()

    # timed region: K steps, HIP events per step for the kernel roofline
    ev = [(torch.cuda.Event(enable_timing=True),
           torch.cuda.Event(enable_timing=True)) for _ in range(args.steps)]
    t0 = time.perf_counter()
    for i in range(args.steps):
        ev[i][0].record()
        step()
        ev[i][1].record()
    torch.cuda.synchronize()
    if distributed:
        torch.distributed.barrier()
    torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    if distributed:
        t = torch.tensor([elapsed], device="cuda", dtype=torch.float64)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())

    step_ms = [a.elapsed_time(b) for a, b in ev]
    avg_launch_ms = sum(step_ms) / len(step_ms)

    data_gib_per_step = n_gpus * S * STRIPE_BYTES / (1 << 30)
    value = data_gib_per_step * args.steps / elapsed

    if rank != 0:
        return

    # roofline for the dominant kernel (ec_encode_kernel<2>): one launch per
    # step; algorithmic bytes = read k*L + write m*L per stripe (SURVEY §8d:
    # ec(8,2) moves 1.25 B per data byte).
    alg_bytes_per_launch = S * STRIPE_BYTES * (K_EC + M_EC) // K_EC
    achieved_gbps = alg_bytes_per_launch / (avg_launch_ms / 1e3) / 1e9
    peak_gbps = 8000.0  # MI355X_MICROARCH.md: HBM3E 8 TB/s spec peak
    traffic = read_traffic_calibration(workload)

    cpu_baseline = None
    if n_gpus == 1 and not args.skip_cpu_baseline:
        log("[rank 0] timing CPU baseline (oracle, OpenMP)...")
        cpu_baseline = cpu_baseline_leg()

    result = {
        "metric": "GiB/s EC encode, ec(8,2) 64MiB stripes",
        "value": round(value, 2),
        "unit": "GiB/s",
        "n_gpus": n_gpus,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(elapsed * 1e3 / args.steps, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,  # BASELINE.json.published == {}
        "dtype": "u8",
        "data": "synthetic",
        "config": {
            "workload": workload,
            "k": K_EC, "m": M_EC,
            "stripe_bytes": STRIPE_BYTES,
            "stripes_per_gpu": S,
            "parallelism": f"independent stripe batches x{n_gpus}",
        },
        "roofline": {
            "bound": "hbm",
            "achieved": round(achieved_gbps, 1),
            "peak": peak_gbps,
            "unit": "GB/s",
            "frac": round(achieved_gbps / peak_gbps, 4),
            "traffic": traffic,
        },
        "cpu_baseline": cpu_baseline,
    }
    print(json.dumps(result), flush=True)


if __name__ == "__main__":
    main()
