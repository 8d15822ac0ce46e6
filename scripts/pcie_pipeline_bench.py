#!/usr/bin/env python3
"""PCIe-inclusive pipeline rate (DESIGN.md note per SURVEY §8d).

The headline metric is kernel-side GiB/s with device-resident stripes;
this script measures the OTHER regime: stripes handed over as HOST
buffers (the chunkserver's case before any integration deeper than
memcpy), with upload (H2D), ec(8,2) encode, and parity download (D2H)
pipelined across chunks on separate streams, double-buffered.

Run on the GPU box: python scripts/pcie_pipeline_bench.py [total_stripes]
"""
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import torch  # noqa: E402

from lizardfs_amd.ec import ReedSolomon  # noqa: E402

K, M = 8, 2
STRIPE = 64 * 1024 * 1024
PART = STRIPE // K
CHUNK = 32                  # stripes per pipeline stage (2 GiB data)


def main():
    total = int(sys.argv[1]) if len(sys.argv) > 1 else 256
    assert total % CHUNK == 0
    stages = total // CHUNK

    # host-resident source data (pinned for async DMA) + parity destination
    host_data = torch.randint(0, 256, (total, K, PART), dtype=torch.uint8,
                              pin_memory=True)
    host_parity = torch.empty((total, M, PART), dtype=torch.uint8,
                              pin_memory=True)

    # device double buffers
    dbuf = [torch.empty((CHUNK, K, PART), dtype=torch.uint8, device="cuda")
            for _ in range(2)]
    pbuf = [torch.empty((CHUNK, M, PART), dtype=torch.uint8, device="cuda")
            for _ in range(2)]

    rs = ReedSolomon(K, M)
    up = torch.cuda.Stream()
    down = torch.cuda.Stream()
    compute = torch.cuda.current_stream()
    up_done = [torch.cuda.Event() for _ in range(2)]
    enc_done = [torch.cuda.Event() for _ in range(2)]

    def run_once():
        for s in range(stages):
            b = s % 2
            span = slice(s * CHUNK, (s + 1) * CHUNK)
            with torch.cuda.stream(up):
                # wait until the previous encode using this buffer finished
                if s >= 2:
                    up.wait_event(enc_done[b])
                dbuf[b].copy_(host_data[span], non_blocking=True)
                up_done[b].record()
            compute.wait_event(up_done[b])
            rs.encode_batch(dbuf[b], pbuf[b])
            enc_done[b].record()
            with torch.cuda.stream(down):
                down.wait_event(enc_done[b])
                host_parity[span].copy_(pbuf[b], non_blocking=True)
        torch.cuda.synchronize()

    run_once()  # warmup
    t0 = time.perf_counter()
    reps = 3
    for _ in range(reps):
        run_once()
    dt = time.perf_counter() - t0

    gib = reps * total * STRIPE / (1 << 30)
    up_gib = gib                      # data uploaded
    down_gib = gib * M / K            # parity downloaded
    print(f"PCIe-inclusive ec({K},{M}) encode over host buffers:")
    print(f"  {gib / dt:.1f} GiB/s data-in end-to-end "
          f"({total} stripes x {reps}, chunked {CHUNK}, double-buffered)")
    print(f"  H2D volume {up_gib:.0f} GiB, D2H {down_gib:.0f} GiB, "
          f"wall {dt:.2f} s")

    # sanity: spot-check one stripe against the host path
    import numpy as np
    import ctypes
    from lizardfs_amd import lib as L
    tbl = (ctypes.c_uint8 * (32 * K * M))()
    L.check(L.lib().lizec_rs_encode_tables(K, M, tbl))
    s0 = host_data[0].numpy()
    exp = np.zeros((M, PART), np.uint8)
    sp = (ctypes.c_void_p * K)(*[s0[j].ctypes.data_as(ctypes.c_void_p).value
                                 for j in range(K)])
    dp = (ctypes.c_void_p * M)(*[exp[l].ctypes.data_as(ctypes.c_void_p).value
                                 for l in range(M)])
    L.lib().ec_encode_data(PART, K, M, tbl, sp, dp)
    assert np.array_equal(host_parity[0].numpy(), exp), "parity mismatch"
    print("  parity spot-check vs host scalar path: OK")


if __name__ == "__main__":
    main()
