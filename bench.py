#!/usr/bin/env python3
"""bench.py — headline benchmark for the MI355X EC engine.

Measures BASELINE.json's metric ("GiB/s EC encode+decode, ec(8,2) 64 MiB
stripes") on BASELINE config 2: ec(8,2) encode of a batch of 64 MiB
synthetic stripes, inputs resident in HBM.  A "step" = one pass of the op
over the whole batch.  value = whole-job data GiB/s across all ranks (the
reference's own throughput convention, reed_solomon_unittest.cc:44-73:
bytes of stripe data / time).

Usage: python bench.py [--gpus N] [--steps K] [--warmup W] [--stripes S]
                       [--op encode|decode|crc|encode_crc|mixed]
N>1 is launched by the driver via torch.distributed.run (one rank per GPU,
RCCL); stripes shard across ranks as independent batches (weak scaling —
the reference processes stripes independently too).

Ops map to the BASELINE configs:
  encode      config 2: ec(8,2) encode                       [default]
  decode      config 3: ec(8,2) decode, 2 erased data parts
  crc         the per-64KiB-block CRC32 gate alone
  encode_crc  config 4: ec(16,4) encode + CRC32 verify of all parts
  mixed       config 5: ec(32,6) encode + 3-erasure decode; when
              distributed, recovered parts are all-gathered over
              RCCL/xGMI (the one real exchange in the path)

The cpu_baseline leg times the REFERENCE'S OWN code (oracle/_ref/libref.so
= galois_field_encode.cc AVX2 dispatch + crc.cc, compiled unmodified from
/root/reference by oracle/Makefile), threaded over stripes across all host
cores (ctypes releases the GIL during the foreign calls).  kind =
"reference".  If the prebuilt _ref library is absent it falls back to the
oracle restatement (kind = "port").  Reported context, not the roofline
target.
"""
import argparse
import ctypes
import json
import os
import sys
import time
from concurrent.futures import ThreadPoolExecutor

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "oracle"))

import numpy as np  # noqa: E402
import torch  # noqa: E402

STRIPE_BYTES = 64 * 1024 * 1024

OPS = {
    # op: (k, m, erased_parts, default stripes per rank)
    "encode": (8, 2, (), 1024),
    "decode": (8, 2, (1, 5), 1024),
    "crc": (8, 2, (), 1024),
    "encode_crc": (16, 4, (), 1024),
    "mixed": (32, 6, (2, 9, 30), 512),
}


def log(msg):
    print(msg, file=sys.stderr, flush=True)


def _load_libref():
    path = os.path.join(REPO, "oracle", "_ref", "libref.so")
    if not os.path.exists(path):
        return None
    try:
        L = ctypes.CDLL(path)
    except OSError:
        return None
    u8p = ctypes.POINTER(ctypes.c_uint8)
    pp = ctypes.POINTER(ctypes.c_void_p)
    L.ref_rs_encode.restype = ctypes.c_int
    L.ref_rs_encode.argtypes = [ctypes.c_int, ctypes.c_int, pp, pp,
                                ctypes.c_size_t]
    L.ref_rs_recover.restype = ctypes.c_int
    L.ref_rs_recover.argtypes = [ctypes.c_int, ctypes.c_int, pp,
                                 ctypes.c_uint64, pp, ctypes.c_size_t]
    L.ref_mycrc32.restype = ctypes.c_uint32
    L.ref_mycrc32.argtypes = [ctypes.c_uint32, u8p, ctypes.c_uint32]
    try:
        L.ref_mycrc32_blocks.argtypes = [u8p, ctypes.c_uint64,
                                         ctypes.c_uint32,
                                         ctypes.POINTER(ctypes.c_uint32)]
    except AttributeError:
        pass   # older prebuilt _ref: per-block calls instead
    L.ref_mycrc32_init()
    return L


def cpu_baseline_leg(op, k, m, erased, target_seconds=6.0, max_reps=64):
    """Time the reference's own encode/decode/CRC (unmodified sources from
    /root/reference compiled into oracle/_ref/libref.so — the AVX2 dispatch
    of galois_field_encode.cc:151-225) on a bounded sample of the same
    workload, threaded over stripes on all host cores.  Falls back to the
    oracle restatement (kind='port') if the prebuilt _ref is absent."""
    cores = int(os.environ.get("OMP_NUM_THREADS", os.cpu_count() or 1))
    rng = np.random.default_rng(42)
    part_len = STRIPE_BYTES // k
    ref = _load_libref()
    # enough stripes to occupy every core, capped at 4 GiB of sample data
    n = max(16, min(2 * cores, 64)) if ref else 16
    data = rng.integers(0, 256, (n, k, part_len), np.uint8)

    def timed(fn, unit_gib):
        fn()   # warmup: page-in, thread pool spin-up
        t0 = time.perf_counter()
        fn()
        once = max(time.perf_counter() - t0, 1e-3)
        reps = max(1, min(int(target_seconds / once), max_reps))
        t0 = time.perf_counter()
        for _ in range(reps):
            fn()
        dt = time.perf_counter() - t0
        return reps * unit_gib / dt, reps, dt

    legs = []
    if ref is not None:
        nparts = k + m
        parity = np.zeros((n, m, part_len), np.uint8)
        pool = ThreadPoolExecutor(cores)
        # sub-stripe task granularity so n stripes can still feed all
        # cores (encode/recover/CRC are independent per byte range);
        # segments stay 64 KiB-aligned for the CRC leg
        segs = max(1, -(-4 * cores // n))
        seg_len = max(65536, (part_len // segs) & ~65535)
        seg_offs = list(range(0, part_len, seg_len))
        tasks = [(s, off, min(seg_len, part_len - off))
                 for s in range(n) for off in seg_offs]

        def stripe_ptrs(a, idx, off=0):
            arr = (ctypes.c_void_p * len(idx))()
            for j, i in enumerate(idx):
                arr[j] = a[i].ctypes.data + off
            return arr

        if op in ("encode", "encode_crc", "mixed"):
            srcs = {(s, off): stripe_ptrs(data[s], range(k), off)
                    for s, off, _ in tasks}
            dsts = {(s, off): stripe_ptrs(parity[s], range(m), off)
                    for s, off, _ in tasks}

            def enc_one(t):
                s, off, ln = t
                ref.ref_rs_encode(k, m, srcs[(s, off)], dsts[(s, off)], ln)
            legs.append(lambda: list(pool.map(enc_one, tasks)))
        if op in ("decode", "mixed"):
            # recover the erased parts from the survivors (for decode-only
            # ops, fill parity with one untimed encode pass first)
            if op == "decode":
                s0 = {(s, off): stripe_ptrs(data[s], range(k), off)
                      for s, off, _ in tasks}
                d0 = {(s, off): stripe_ptrs(parity[s], range(m), off)
                      for s, off, _ in tasks}
                list(pool.map(lambda t: ref.ref_rs_encode(
                    k, m, s0[(t[0], t[1])], d0[(t[0], t[1])], t[2]), tasks))
            pad = tuple(range(nparts - (m - len(erased)), nparts))
            erased_full = tuple(erased) + pad
            mask = sum(1 << i for i in erased_full)
            rec = np.zeros((n, len(erased_full), part_len), np.uint8)
            frg = {}
            for s, off, _ in tasks:
                fr = (ctypes.c_void_p * nparts)()
                ou = (ctypes.c_void_p * nparts)()
                for i in range(nparts):
                    if i not in erased_full:
                        fr[i] = (data[s, i].ctypes.data if i < k
                                 else parity[s, i - k].ctypes.data) + off
                for j, i in enumerate(erased_full):
                    ou[i] = rec[s, j].ctypes.data + off
                frg[(s, off)] = (fr, ou)

            def rec_one(t):
                s, off, ln = t
                fr, ou = frg[(s, off)]
                ref.ref_rs_recover(k, m, fr, mask, ou, ln)
            legs.append(lambda: list(pool.map(rec_one, tasks)))
        if op in ("crc", "encode_crc"):
            u8p = ctypes.POINTER(ctypes.c_uint8)
            u32p = ctypes.POINTER(ctypes.c_uint32)
            rows = [data[s].reshape(-1) for s in range(n)]
            crcs = [np.empty(rows[0].size // 65536, np.uint32)
                    for _ in range(n)]
            crc_seg = max(65536, (rows[0].size // segs) & ~65535)
            crc_tasks = [(s, off, min(crc_seg, rows[0].size - off))
                         for s in range(n)
                         for off in range(0, rows[0].size, crc_seg)]
            blocks_fn = getattr(ref, "ref_mycrc32_blocks", None)

            def crc_one(t):
                s, off, ln = t
                base = rows[s].ctypes.data + off
                if blocks_fn is not None:
                    blocks_fn(ctypes.cast(base, u8p), ln // 65536, 65536,
                              ctypes.cast(crcs[s].ctypes.data +
                                          (off // 65536) * 4, u32p))
                else:
                    for o in range(0, ln, 65536):
                        ref.ref_mycrc32(0, ctypes.cast(base + o, u8p), 65536)
            legs.append(lambda: list(pool.map(crc_one, crc_tasks)))
        kind = "reference"
        how = (f"reference AVX2 (oracle/_ref/libref.so), {cores} threads "
               f"over stripes")
    else:
        import oracle
        if op in ("encode", "decode", "encode_crc", "mixed"):
            if op == "decode":
                present = sum(1 << i for i in range(k + m) if i not in erased)
                needed = sum(1 << i for i in erased)
                tbl, ic, oc = oracle.rs_make_tables(k, m, present, present,
                                                    needed)
                srcs = np.ascontiguousarray(data[:, :ic])
            else:
                tbl, ic, oc = oracle.rs_make_tables(
                    k, m, (1 << k) - 1, (1 << k) - 1, ((1 << m) - 1) << k)
                srcs = data
            out = np.zeros((n, oc, part_len), np.uint8)
            legs.append(lambda: oracle.encode_stripes(ic, oc, part_len, n,
                                                      tbl, srcs, out))
        if op in ("crc", "encode_crc"):
            buf = np.ascontiguousarray(data.reshape(-1))
            legs.append(lambda: oracle.crc32_blocks(buf, 65536))
        kind = "port"
        how = "oracle restatement, OpenMP"

    def all_legs():
        for f in legs:
            f()

    gibs, reps, dt = timed(all_legs, n * STRIPE_BYTES / (1 << 30))
    sample = (f"{reps}x{n}x64MiB stripes ec({k},{m}) {op}, {dt:.1f}s, "
              f"{how}")
    return {"value": round(gibs, 3), "unit": "GiB/s", "cores": cores,
            "kind": kind, "sample": sample}


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
    ap.add_argument("--stripes", type=int, default=0,
                    help="stripes per rank (weak scaling); 0 = op default")
    ap.add_argument("--op", choices=tuple(OPS), default="encode")
    ap.add_argument("--backend", choices=("nccl", "gloo"), default="nccl",
                    help="gloo allows exercising the multi-rank path on a "
                         "single GPU (coordination on CPU; compute on GPU)")
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    ap.add_argument("--validate", action="store_true",
                    help="after timing, bit-check recovered parts against "
                         "the original data (decode/mixed round-trip) and "
                         "the all-gather's own-rank slot")
    args = ap.parse_args()

    K, M, ERASED, S_default = OPS[args.op]
    S = args.stripes or S_default
    PART_LEN = STRIPE_BYTES // K
    nparts = K + M

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if args.gpus > 1 and world == 1:
        log(f"--gpus {args.gpus} requires torchrun (one rank per GPU): "
            f"python -m torch.distributed.run --nnodes=1 --nproc-per-node "
            f"{args.gpus} --master-addr 127.0.0.1 bench.py ...")
        sys.exit(2)
    n_gpus = world if world > 1 else args.gpus
    distributed = world > 1

    if not torch.cuda.is_available():
        log("bench.py requires an MI355X (no CPU fallback on the product path)")
        sys.exit(2)

    dev_idx = local_rank % torch.cuda.device_count()
    torch.cuda.set_device(dev_idx)
    if distributed:
        torch.distributed.init_process_group(args.backend)

    from lizardfs_amd.ec import ReedSolomon
    from lizardfs_amd import crc as lcrc

    workload = (f"ec({K},{M}) {args.op}, {S}x64MiB synthetic "
                f"stripes/GPU, device-resident")
    log(f"[rank {rank}] generating {S} stripes "
        f"({S * STRIPE_BYTES / (1 << 30):.0f} GiB data) on cuda:{dev_idx}")
    g = torch.Generator(device="cuda").manual_seed(42 + rank)
    data = torch.randint(0, 256, (S, K, PART_LEN), dtype=torch.uint8,
                         device="cuda", generator=g)
    parity = torch.empty((S, M, PART_LEN), dtype=torch.uint8, device="cuda")

    rs = ReedSolomon(K, M, device=dev_idx)

    if args.op == "encode":
        def step():
            rs.encode_batch(data, parity)
        alg_bytes_per_launch = S * STRIPE_BYTES * (K + M) // K
        metric = "GiB/s EC encode, ec(8,2) 64MiB stripes"
    elif args.op == "decode":
        rs.encode_batch(data, parity)
        rs.sync()
        frags = [None if i in ERASED else
                 (data[:, i, :] if i < K else parity[:, i - K, :])
                 for i in range(nparts)]
        outs = {i: torch.empty((S, PART_LEN), dtype=torch.uint8,
                               device="cuda") for i in ERASED}

        def step():
            rs.recover_batch(frags, erased=ERASED, out=outs)
        alg_bytes_per_launch = S * PART_LEN * (K + len(ERASED))
        metric = "GiB/s EC decode(2 erasures), ec(8,2) 64MiB stripes"
    elif args.op == "crc":
        flat = data.reshape(-1)
        crcs = torch.empty(flat.numel() // 65536, dtype=torch.int32,
                           device="cuda")

        def step():
            lcrc.crc32_blocks(flat, 65536, out=crcs)
        alg_bytes_per_launch = (S * STRIPE_BYTES +
                                4 * (S * STRIPE_BYTES // 65536))
        metric = "GiB/s CRC32 per-64KiB-block"
    elif args.op == "encode_crc":
        # config 4: encode + CRC-verify every 64 KiB block of every part,
        # as the replicator does before hdd_write
        # (chunk_replicator.cc:189, hddspacemgr.cc:1918).  Pipelined: the
        # data CRC does not depend on the encode, so it runs on a second
        # stream concurrent with the encode; the parity CRC follows the
        # encode on the main stream.
        dflat = data.reshape(-1)
        pflat = parity.reshape(-1)
        dcrcs = torch.empty(dflat.numel() // 65536, dtype=torch.int32,
                            device="cuda")
        pcrcs = torch.empty(pflat.numel() // 65536, dtype=torch.int32,
                            device="cuda")
        crc_stream = torch.cuda.Stream()
        overlap = os.environ.get("LIZEC_BENCH_NO_OVERLAP", "0") != "1"

        def step():
            if not overlap:
                rs.encode_batch(data, parity)
                lcrc.crc32_blocks(dflat, 65536, out=dcrcs)
                lcrc.crc32_blocks(pflat, 65536, out=pcrcs)
                return
            main = torch.cuda.current_stream()
            crc_stream.wait_stream(main)
            with torch.cuda.stream(crc_stream):
                lcrc.crc32_blocks(dflat, 65536, out=dcrcs)
            rs.encode_batch(data, parity)
            lcrc.crc32_blocks(pflat, 65536, out=pcrcs)
            main.wait_stream(crc_stream)
        alg_bytes_per_launch = S * STRIPE_BYTES * (K + M) // K  # encode leg
        metric = "GiB/s EC encode+CRC32 pipeline, ec(16,4) 64MiB stripes"
    else:  # mixed — config 5
        rs.encode_batch(data, parity)
        rs.sync()
        # exactly m erased required (reed_solomon.h:95): pad the 3 data
        # erasures with unneeded parity parts, as ec_read_plan.h:126 does
        pad = tuple(range(nparts - (M - len(ERASED)), nparts))
        erased_full = tuple(ERASED) + pad
        frags = [None if i in erased_full else
                 (data[:, i, :] if i < K else parity[:, i - K, :])
                 for i in range(nparts)]
        outs = {i: torch.empty((S, PART_LEN), dtype=torch.uint8,
                               device="cuda") for i in ERASED}
        if distributed:
            gdev = "cuda" if args.backend == "nccl" else "cpu"
            gathered = [
                [torch.empty((S, PART_LEN), dtype=torch.uint8, device=gdev)
                 for _ in range(world)] for _ in ERASED]

        def step():
            rs.encode_batch(data, parity)
            rs.recover_batch(frags, erased=erased_full, want=set(ERASED),
                             out=outs)
            if distributed:
                # all-gather of recovered parts so every rank holds the
                # full recovered set — RCCL over xGMI under
                # nccl; under gloo (single-GPU validation runs) the parts
                # are staged through host memory, as gloo requires
                for gi, i in enumerate(sorted(outs)):
                    src = outs[i] if gdev == "cuda" else outs[i].cpu()
                    torch.distributed.all_gather(gathered[gi], src)
        alg_bytes_per_launch = S * STRIPE_BYTES * (K + M) // K
        metric = ("GiB/s EC mixed encode+3-erasure decode, ec(32,6) "
                  "64MiB stripes")

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
        dev = "cuda" if args.backend == "nccl" else "cpu"
        t = torch.tensor([elapsed], device=dev, dtype=torch.float64)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())

    validated = None
    if args.validate and args.op in ("decode", "mixed"):
        # encode -> erase -> recover must reproduce the original bytes
        # (bit-exact round trip; the GF math is deterministic)
        for i in sorted(outs):
            want_t = data[:, i, :] if i < K else parity[:, i - K, :]
            if not torch.equal(outs[i], want_t):
                log(f"[rank {rank}] VALIDATION FAILED: part {i} mismatch")
                sys.exit(3)
        if distributed and args.op == "mixed":
            for gi, i in enumerate(sorted(outs)):
                own = gathered[gi][rank]
                src = outs[i] if own.is_cuda else outs[i].cpu()
                if not torch.equal(own, src):
                    log(f"[rank {rank}] VALIDATION FAILED: all-gather "
                        f"slot {rank} part {i}")
                    sys.exit(3)
        validated = "recover round-trip bit-exact" + \
            (", all-gather own-slot bit-exact"
             if distributed and args.op == "mixed" else "")
        log(f"[rank {rank}] validation OK: {validated}")

    step_ms = [a.elapsed_time(b) for a, b in ev]
    avg_step_ms = sum(step_ms) / len(step_ms)
    # roofline leg: the dominant kernel is the EC encode launch (1/step)
    # except for --op crc where it is the CRC kernel itself
    launches_per_step = 1
    avg_launch_ms = avg_step_ms / launches_per_step

    data_gib_per_step = n_gpus * S * STRIPE_BYTES / (1 << 30)
    value = data_gib_per_step * args.steps / elapsed

    if rank != 0:
        return

    achieved_gbps = alg_bytes_per_launch / (avg_launch_ms / 1e3) / 1e9
    peak_gbps = 8000.0  # MI355X_MICROARCH.md: HBM3E 8 TB/s spec peak
    traffic = read_traffic_calibration(workload)
    if args.op in ("encode_crc", "mixed"):
        # multiple kernels per step: the per-step event time is not a
        # single-kernel duration; report whole-step rate and mark it
        roofline = {
            "bound": "hbm",
            "achieved": round(alg_bytes_per_launch / (avg_step_ms / 1e3) / 1e9, 1),
            "peak": peak_gbps, "unit": "GB/s",
            "frac": None,   # composite step; per-kernel fracs in profiles/
            "traffic": traffic,
        }
    else:
        roofline = {
            "bound": "hbm",
            "achieved": round(achieved_gbps, 1),
            "peak": peak_gbps,
            "unit": "GB/s",
            "frac": round(achieved_gbps / peak_gbps, 4),
            "traffic": traffic,
        }

    cpu_baseline = None
    if n_gpus == 1 and not args.skip_cpu_baseline:
        log("[rank 0] timing CPU baseline (oracle, OpenMP)...")
        cpu_baseline = cpu_baseline_leg(args.op, K, M, ERASED)

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
            "k": K, "m": M,
            "stripe_bytes": STRIPE_BYTES,
            "stripes_per_gpu": S,
            "parallelism": f"independent stripe batches x{n_gpus}" +
                           (f" + {args.backend} all-gather of recovered "
                            f"parts"
                            if args.op == "mixed" and distributed else ""),
            "backend": args.backend if distributed else None,
        },
        "validated": validated,
        "roofline": roofline,
        "cpu_baseline": cpu_baseline,
    }
    print(json.dumps(result), flush=True)


if __name__ == "__main__":
    main()
