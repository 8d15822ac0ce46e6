#!/usr/bin/env python3
"""bench.py — headline benchmark for the MI355X EC engine.

Measures BASELINE.json's metric ("GiB/s EC encode+decode, ec(8,2) 64 MiB
stripes") on BASELINE config 2: ec(8,2) encode of a batch of 64 MiB
synthetic stripes, inputs resident in HBM.  A "step" = one encode pass of
the whole batch.  value = whole-job data-in GiB/s across all ranks
(the reference's own throughput convention, reed_solomon_unittest.cc:44-73).

Usage: python bench.py [--gpus N] [--steps K] [--warmup W] [--stripes S]
                       [--op encode|decode|crc]
N>1 is launched by the driver via torch.distributed.run (one rank per GPU,
RCCL); stripes shard across ranks as independent batches (weak scaling — the
reference processes stripes independently too; no data-path collective,
SURVEY §8e).

--op decode measures BASELINE config 3 (2 erased data parts); --op crc the
per-64KiB-block CRC32 gate.  The default (encode) is the contract line.

The cpu_baseline leg times the ORACLE (CPU restatement of the reference
path, OpenMP over stripes) on a bounded sample — reported context, not the
roofline target.
"""
import argparse
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
ERASED = (1, 5)   # BASELINE config 3: 2 erased data parts


def log(msg):
    print(msg, file=sys.stderr, flush=True)


def cpu_baseline_leg(op, target_seconds=10.0):
    """Time the oracle's threaded path on a bounded sample of the same
    workload on this box's host cores.  kind='port' (restatement of
    galois_field_encode.cc:28-47 + reed_solomon.h semantics)."""
    import oracle
    cores = int(os.environ.get("OMP_NUM_THREADS", os.cpu_count() or 1))
    rng = np.random.default_rng(42)
    n = 16  # 1 GiB data sample, repeated until ~target_seconds
    data = rng.integers(0, 256, (n, K_EC, PART_LEN), np.uint8)

    if op == "crc":
        buf = np.ascontiguousarray(data.reshape(-1))
        t0 = time.perf_counter()
        oracle.crc32_blocks(buf, 65536)
        once = time.perf_counter() - t0
        reps = max(1, int(target_seconds / max(once, 1e-3)))
        t0 = time.perf_counter()
        for _ in range(reps):
            oracle.crc32_blocks(buf, 65536)
        dt = time.perf_counter() - t0
        gib = reps * buf.size / (1 << 30)
        sample = f"{reps}x{n * 64}MiB CRC32/64KiB blocks, {dt:.1f}s, OpenMP"
        return {"value": round(gib / dt, 3), "unit": "GiB/s", "cores": cores,
                "kind": "port", "sample": sample}

    if op == "decode":
        present = sum(1 << i for i in range(K_EC + M_EC) if i not in ERASED)
        needed = sum(1 << i for i in ERASED)
        tbl, ic, oc = oracle.rs_make_tables(K_EC, M_EC, present, present,
                                            needed)
        # reuse encode_stripes shape: inputs = the ic surviving parts
        parity = np.zeros((n, oc, PART_LEN), np.uint8)
        srcs = np.ascontiguousarray(data[:, :ic])  # ic=8 surviving parts
        enc = lambda: oracle.encode_stripes(ic, oc, PART_LEN, n, tbl,
                                            srcs, parity)
    else:
        tbl, ic, oc = oracle.rs_make_tables(
            K_EC, M_EC, (1 << K_EC) - 1, (1 << K_EC) - 1,
            ((1 << M_EC) - 1) << K_EC)
        parity = np.zeros((n, M_EC, PART_LEN), np.uint8)
        enc = lambda: oracle.encode_stripes(K_EC, M_EC, PART_LEN, n, tbl,
                                            data, parity)

    t0 = time.perf_counter()
    enc()
    once = time.perf_counter() - t0
    reps = max(1, int(target_seconds / max(once, 1e-3)))
    t0 = time.perf_counter()
    for _ in range(reps):
        enc()
    dt = time.perf_counter() - t0
    gib = reps * n * STRIPE_BYTES / (1 << 30)
    sample = f"{reps}x{n}x64MiB stripes ec(8,2) {op}, {dt:.1f}s, OpenMP"
    return {"value": round(gib / dt, 3), "unit": "GiB/s", "cores": cores,
            "kind": "port", "sample": sample}


def read_traffic_calibration(workload):
    """PMC-measured HBM bytes per launch, if a matching calibration was
    committed from a rocprofv3 run (profiles/pmc_traffic.json)."""
    path = os.path.join(REPO, "profiles", "pmc_traffic.json")
    try:
        with open(path) as f:
            entries = json.load(f)
        for d in entries if isinstance(entries, list) else [entries]:
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
    ap.add_argument("--op", choices=("encode", "decode", "crc"),
                    default="encode")
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
    from lizardfs_amd import crc as lcrc

    S = args.stripes
    workload = (f"ec({K_EC},{M_EC}) {args.op}, {S}x64MiB synthetic "
                f"stripes/GPU, device-resident")
    log(f"[rank {rank}] generating {S} stripes "
        f"({S * STRIPE_BYTES / (1 << 30):.0f} GiB data) on cuda:{local_rank}")
    g = torch.Generator(device="cuda").manual_seed(42 + rank)
    data = torch.randint(0, 256, (S, K_EC, PART_LEN), dtype=torch.uint8,
                         device="cuda", generator=g)
    parity = torch.empty((S, M_EC, PART_LEN), dtype=torch.uint8,
                         device="cuda")

    rs = ReedSolomon(K_EC, M_EC, device=local_rank)
    nparts = K_EC + M_EC

    if args.op == "encode":
        def step():
            rs.encode_batch(data, parity)
        # dominant kernel: ec_encode_kernel<2>; 1 launch/step
        alg_bytes_per_launch = S * STRIPE_BYTES * (K_EC + M_EC) // K_EC
        metric = "GiB/s EC encode, ec(8,2) 64MiB stripes"
    elif args.op == "decode":
        rs.encode_batch(data, parity)   # produce real parity first
        rs.sync()
        frags = [None if i in ERASED else
                 (data[:, i, :] if i < K_EC else parity[:, i - K_EC, :])
                 for i in range(nparts)]
        outs = {i: torch.empty((S, PART_LEN), dtype=torch.uint8,
                               device="cuda") for i in ERASED}

        def step():
            rs.recover_batch(frags, erased=ERASED, out=outs)
        alg_bytes_per_launch = S * PART_LEN * (K_EC + len(ERASED))
        metric = "GiB/s EC decode(2 erasures), ec(8,2) 64MiB stripes"
    else:  # crc
        flat = data.reshape(-1)
        crcs = torch.empty(flat.numel() // 65536, dtype=torch.int32,
                           device="cuda")

        def step():
            lcrc.crc32_blocks(flat, 65536, out=crcs)
        alg_bytes_per_launch = S * STRIPE_BYTES + 4 * (S * STRIPE_BYTES // 65536)
        metric = "GiB/s CRC32 per-64KiB-block"

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

    achieved_gbps = alg_bytes_per_launch / (avg_launch_ms / 1e3) / 1e9
    peak_gbps = 8000.0  # MI355X_MICROARCH.md: HBM3E 8 TB/s spec peak
    traffic = read_traffic_calibration(workload)

    cpu_baseline = None
    if n_gpus == 1 and not args.skip_cpu_baseline:
        log("[rank 0] timing CPU baseline (oracle, OpenMP)...")
        cpu_baseline = cpu_baseline_leg(args.op)

    result = {
        "metric": metric,
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
