#!/usr/bin/env python3
"""Stability soak for the adopted kernels: continuous encode + 2-erasure
recover + per-block CRC on one batch, output bit-compared against the
first iteration every 50 rounds.  Usage: soak.py [seconds]"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from lizardfs_amd import crc as lcrc
from lizardfs_amd.ec import ReedSolomon

seconds = int(sys.argv[1]) if len(sys.argv) > 1 else 300
k, m, S, plen = 8, 2, 256, 8 * 1024 * 1024 // 8
g = torch.Generator(device="cuda").manual_seed(99)
data = torch.randint(0, 256, (S, k, plen), dtype=torch.uint8, device="cuda",
                     generator=g)
rs = ReedSolomon(k, m)
parity = torch.empty((S, m, plen), dtype=torch.uint8, device="cuda")
erased = (1, 5)
outs = {i: torch.empty((S, plen), dtype=torch.uint8, device="cuda")
        for i in erased}
crcs = torch.empty(data.numel() // 65536, dtype=torch.int32, device="cuda")

rs.encode_batch(data, parity)
frags = [None if i in erased else
         (data[:, i, :] if i < k else parity[:, i - k, :])
         for i in range(k + m)]
rs.recover_batch(frags, erased=erased, out=outs)
lcrc.crc32_blocks(data.reshape(-1), 65536, out=crcs)
torch.cuda.synchronize()
ref_par = parity.clone()
ref_rec = {i: outs[i].clone() for i in erased}
ref_crc = crcs.clone()

t0 = time.time()
it = 0
bytes_moved = 0
while time.time() - t0 < seconds:
    rs.encode_batch(data, parity)
    rs.recover_batch(frags, erased=erased, out=outs)
    lcrc.crc32_blocks(data.reshape(-1), 65536, out=crcs)
    it += 1
    bytes_moved += (S * plen * (k + m)) * 2 + S * plen * k
    if it % 50 == 0:
        torch.cuda.synchronize()
        assert torch.equal(parity, ref_par), f"parity drift at iter {it}"
        for i in erased:
            assert torch.equal(outs[i], ref_rec[i]), f"rec drift {it}"
        assert torch.equal(crcs, ref_crc), f"crc drift at iter {it}"
torch.cuda.synchronize()
dt = time.time() - t0
print(f"soak OK: {it} iterations in {dt:.0f}s, zero output drift, "
      f"~{bytes_moved / dt / 1e12:.2f} TB/s sustained mixed-op traffic")
