#!/usr/bin/env python3
"""A/B the CRC32 kernel implementations on one MI355X.

Times table-slicing vs carry-less-folding variants on the same
device-resident buffer (the LIZEC_* env knobs are read per call inside
liblizec, so one process sweeps every config), checks every variant's
output is bit-identical to the table kernel's (itself pinned by the
oracle/golden tests), and prints GB/s of data read.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from lizardfs_amd import crc as lcrc  # noqa: E402

GIB = int(sys.argv[1]) if len(sys.argv) > 1 else 16  # device buffer size
BLOCK = 65536
REPS = 10

CFGS = [
    # name, impl, chains, nacc, nt, pf, bv
    ("table_c2_s8", "table", "2", "", "0", "0", "0"),
    ("fold_c1_n1 ", "fold", "1", "1", "0", "0", "0"),
    ("fold_pf_bv8", "fold", "1", "1", "0", "1", "0"),
    ("fold_pf_bv4", "fold", "1", "1", "0", "1", "4"),
    ("fold_bv4   ", "fold", "1", "1", "0", "0", "4"),
    ("fold_n2_bv4", "fold", "1", "2", "0", "0", "4"),
    ("fold_n4_bv4", "fold", "1", "4", "0", "0", "4"),
]


def main():
    torch.cuda.init()
    n = GIB << 30
    g = torch.Generator(device="cuda").manual_seed(123)
    buf = torch.randint(0, 256, (n,), dtype=torch.uint8, device="cuda",
                        generator=g)
    nblocks = n // BLOCK
    out = torch.empty(nblocks, dtype=torch.int32, device="cuda")
    ref = None
    e0 = torch.cuda.Event(enable_timing=True)
    e1 = torch.cuda.Event(enable_timing=True)
    print(f"crc_ab buffer {GIB} GiB", flush=True)
    for name, impl, chains, nacc, nt, pf, bv in CFGS:
        os.environ["LIZEC_CRC_IMPL"] = impl
        os.environ["LIZEC_CRC_CHAINS"] = chains
        os.environ["LIZEC_CRC_FOLD_NACC"] = nacc or "2"
        os.environ["LIZEC_CRC_NT"] = nt
        os.environ["LIZEC_CRC_PF"] = pf
        os.environ["LIZEC_CRC_BV"] = bv
        out.zero_()
        torch.cuda.synchronize()
        for _ in range(3):
            lcrc.crc32_blocks(buf, BLOCK, out=out)
        torch.cuda.synchronize()
        e0.record()
        for _ in range(REPS):
            lcrc.crc32_blocks(buf, BLOCK, out=out)
        e1.record()
        torch.cuda.synchronize()
        ms = e0.elapsed_time(e1) / REPS
        gbps = n / (ms / 1e3) / 1e9
        if ref is None:
            ref = out.clone()
            ok = "REF"
        else:
            ok = "OK" if torch.equal(out, ref) else "WRONG"
        frac = gbps / 8000.0
        print(f"crc_ab {name} {gbps:8.1f} GB/s  frac={frac:.3f} "
              f"({ms:.3f} ms)  {ok}", flush=True)

    # seeded + odd-size sanity through the generic path stays correct
    os.environ["LIZEC_CRC_IMPL"] = "fold"
    os.environ["LIZEC_CRC_CHAINS"] = "1"
    os.environ["LIZEC_CRC_NT"] = "0"
    os.environ["LIZEC_CRC_PF"] = "0"
    small = buf[:3 * 1024 * 1024]
    c1 = lcrc.crc32_blocks(small, 3072, seed=0xDEADBEEF)
    os.environ["LIZEC_CRC_IMPL"] = "table"
    c2 = lcrc.crc32_blocks(small, 3072, seed=0xDEADBEEF)
    torch.cuda.synchronize()
    print("crc_ab odd-size seeded:", "OK" if torch.equal(c1, c2) else "WRONG",
          flush=True)


if __name__ == "__main__":
    main()
