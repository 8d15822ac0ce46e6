#!/usr/bin/env python3
"""End-to-end streaming replication pipeline benchmark.

Measures the full host-to-host rebuild rate of lizec_replicate_run:
pinned-host surviving parts -> H2D -> EC recover -> per-block CRC ->
MooseFS image assembly on-device -> D2H pinned-host images, double
buffered.  PCIe-inclusive by definition — reported SEPARATELY from the
HBM-resident headline (DESIGN.md §8d note), alongside the standalone
staging ceiling (scripts/pcie_pipeline_bench.py).

Prints one JSON line per config with:
  in_gibs   = surviving-part bytes pulled per second (network-pull proxy)
  out_gibs  = image bytes emitted per second (disk-write proxy)
  rebuild_gibs = recovered part-data bytes per second (the replicate rate)
"""
import ctypes
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np  # noqa: E402
import torch  # noqa: E402

from lizardfs_amd import lib as L  # noqa: E402
from lizardfs_amd import scrub, slice_traits as st  # noqa: E402
from lizardfs_amd.replicate import replicate_stream  # noqa: E402
from lizardfs_amd.ec import ReedSolomon  # noqa: E402


def run(k, m, nchunks, reps=3, sub_batch=0):
    plen = (64 * 1024 * 1024) // k
    rng = np.random.default_rng(42)
    erased = tuple(range(m))          # rebuild the first m parts
    surv = [i for i in range(k + m) if i not in erased]

    # synthetic chunk batch: data parts random; parity via the GPU engine
    data = torch.from_numpy(
        rng.integers(0, 256, (nchunks, k, plen), np.uint8)).cuda()
    rs = ReedSolomon(k, m)
    parity = rs.encode_batch(data)
    rs.sync()
    data_np = data.cpu().numpy()
    parity_np = parity.cpu().numpy()
    del data, parity
    torch.cuda.empty_cache()

    host_parts = [None] * (k + m)
    for i in surv:
        a = L.pinned_empty((nchunks, plen))
        a[:] = data_np[:, i, :] if i < k else parity_np[:, i - k, :]
        host_parts[i] = a
    hdr = scrub.header_size(st.ec_slice_type(k, m))
    out = {p: L.pinned_empty((nchunks, hdr + plen)) for p in erased}
    chunk_ids = list(range(1, nchunks + 1))

    replicate_stream(k, m, host_parts, erased, erased, chunk_ids, 3,
                     out=out, sub_batch=sub_batch)  # warmup
    t0 = time.perf_counter()
    for _ in range(reps):
        replicate_stream(k, m, host_parts, erased, erased, chunk_ids, 3,
                         out=out, sub_batch=sub_batch)
    dt = (time.perf_counter() - t0) / reps

    # spot-check: erased data parts really were rebuilt
    for p in erased:
        if p < k:
            assert np.array_equal(out[p][0][hdr:], data_np[0, p]), p

    in_b = nchunks * len(surv) * plen
    out_b = nchunks * len(erased) * (hdr + plen)
    reb_b = nchunks * len(erased) * plen
    print(json.dumps({
        "bench": "replicate_pipeline", "k": k, "m": m, "nchunks": nchunks,
        "part_mib": plen >> 20, "erased": list(erased),
        "sub_batch": sub_batch or "auto", "seconds_per_pass": round(dt, 4),
        "in_gibs": round(in_b / dt / 2**30, 2),
        "out_gibs": round(out_b / dt / 2**30, 2),
        "rebuild_gibs": round(reb_b / dt / 2**30, 2),
        "pcie_inclusive": True, "host_buffers": "pinned",
    }), flush=True)


if __name__ == "__main__":
    nchunks = int(sys.argv[1]) if len(sys.argv) > 1 else 48
    run(8, 2, nchunks)
    run(8, 2, nchunks, sub_batch=4)
    run(8, 2, nchunks, sub_batch=16)
    run(16, 4, nchunks)
