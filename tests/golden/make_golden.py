#!/usr/bin/env python3
"""Golden-vector generator for the LizardFS EC hot path.

Runs the REFERENCE's own code (oracle/_ref/libref.so, compiled unmodified
from /root/reference by oracle/Makefile) on deterministic inputs and commits
the outputs as small fixtures under tests/golden/.  The oracle and the GPU
engine are both tested against these vectors; the reference tree itself is
never needed at test run time (it does not exist on the GPU box).

Inputs follow the reference's own test conventions:
  - the uint32 LE counter pattern of src/unittests/plan_tester.h:48-55
    (byte i of the chunk = LE byte of uint32 value 4*(i/4))
  - fixed-seed (42) uniform random bytes (numpy PCG64)
  - all-zero parts / NULL parts (NULL = implicit zeros, reed_solomon.h:79)

Erasure patterns per reed_solomon_unittest.cc:136-166 ({0,2},{0,5},{4,5} on
ec(4,2)) plus the BASELINE.json configs: ec(3,1), ec(8,2) erase {1,5},
ec(16,4), ec(32,6) erase 3 data parts.

Output: tests/golden/golden.npz + golden_meta.json.
"""
import ctypes
import json
import os
import sys

import numpy as np

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.abspath(os.path.join(HERE, "..", ".."))
REF_SO = os.path.join(REPO, "oracle", "_ref", "libref.so")


def load_ref():
    lib = ctypes.CDLL(REF_SO)
    lib.ref_rs_encode.restype = ctypes.c_int
    lib.ref_rs_encode.argtypes = [ctypes.c_int, ctypes.c_int,
                                  ctypes.POINTER(ctypes.c_void_p),
                                  ctypes.POINTER(ctypes.c_void_p),
                                  ctypes.c_size_t]
    lib.ref_rs_recover.restype = ctypes.c_int
    lib.ref_rs_recover.argtypes = [ctypes.c_int, ctypes.c_int,
                                   ctypes.POINTER(ctypes.c_void_p),
                                   ctypes.c_uint64,
                                   ctypes.POINTER(ctypes.c_void_p),
                                   ctypes.c_size_t]
    lib.ref_mycrc32.restype = ctypes.c_uint32
    lib.ref_mycrc32.argtypes = [ctypes.c_uint32, ctypes.c_char_p, ctypes.c_uint32]
    lib.ref_mycrc32_combine.restype = ctypes.c_uint32
    lib.ref_mycrc32_combine.argtypes = [ctypes.c_uint32, ctypes.c_uint32, ctypes.c_uint32]
    lib.ref_mycrc32_init()
    return lib


def ptr_array(arrs, n):
    """Array of n void* from a list of numpy arrays (None -> NULL)."""
    a = (ctypes.c_void_p * n)()
    for i, x in enumerate(arrs):
        a[i] = x.ctypes.data_as(ctypes.c_void_p).value if x is not None else None
    return a


def counter_pattern(nbytes, offset=0):
    """plan_tester.h:48-55: uint32 LE counter 0,4,8,... starting at offset."""
    nwords = (nbytes + 3) // 4
    words = (np.arange(nwords, dtype=np.uint32) * 4 + offset).astype("<u4")
    return words.view(np.uint8)[:nbytes].copy()


def ref_encode(lib, k, m, parts, size):
    data_p = ptr_array(parts, k)
    parity = [np.zeros(size, np.uint8) for _ in range(m)]
    par_p = ptr_array(parity, m)
    r = lib.ref_rs_encode(k, m, data_p, par_p, size)
    assert r == 0
    return parity


def ref_recover(lib, k, m, frags, erased_mask, want, size):
    in_p = ptr_array(frags, k + m)
    outs = [np.zeros(size, np.uint8) if i in want else None for i in range(k + m)]
    out_p = ptr_array(outs, k + m)
    r = lib.ref_rs_recover(k, m, in_p, erased_mask, out_p, size)
    assert r == 0
    return outs


def main():
    lib = load_ref()
    rng = np.random.default_rng(42)
    out = {}
    meta = {"seed": 42, "cases": []}

    SIZE = 4096  # small parity size: oracle finishes in ms; full sizes are
                 # covered by size-independent property tests in tests/

    # --- encode+decode cases ---
    # (k, m, pattern, erasures) ; erasures drawn from data+parity indices
    cases = [
        (4, 2, "counter", (0, 2)),    # reed_solomon_unittest.cc:136-166
        (4, 2, "counter", (0, 5)),
        (4, 2, "counter", (4, 5)),
        (3, 1, "random",  (1,)),      # BASELINE config 1
        (8, 2, "random",  (1, 5)),    # BASELINE configs 2-3
        (8, 2, "zero",    (1, 5)),
        (16, 4, "random", (0, 7, 15, 17)),   # BASELINE config 4
        (32, 6, "random", (2, 9, 30)),       # BASELINE config 5: 3 data erasures
        (32, 6, "random", (0, 1, 2, 3, 4, 5)),
        (22, 4, "random", (0, 21)),   # k>20, m==4 -> Cauchy branch (reed_solomon.h:168)
        (20, 4, "random", (0, 19)),   # m==4, k<=20 -> Vandermonde branch
        (2, 32, "random", tuple(range(2, 34))[:32]),  # max parity width
        (32, 32, "random", (0, 15, 31, 40, 50, 63)),  # k+m = 64 edge
    ]
    for ci, (k, m, pat, erase) in enumerate(cases):
        if pat == "counter":
            # one 'chunk' striped into k parts, block-wise like buildECData
            parts = [counter_pattern(SIZE, offset=i * SIZE) for i in range(k)]
        elif pat == "zero":
            parts = [None if i % 2 == 0 else rng.integers(0, 256, SIZE, np.uint8)
                     for i in range(k)]
        else:
            parts = [rng.integers(0, 256, SIZE, np.uint8) for i in range(k)]

        parity = ref_encode(lib, k, m, parts, SIZE)
        name = f"case{ci}_ec{k}_{m}"
        for i, p in enumerate(parts):
            out[f"{name}_data{i}"] = p if p is not None else np.zeros(0, np.uint8)
        for i, p in enumerate(parity):
            out[f"{name}_parity{i}"] = p

        # decode: erase `erase`, pad to exactly m erasures with highest parts
        # (mirrors ec_read_plan.h:126-133: exactly m erasures required)
        erased = set(erase)
        for i in range(k + m - 1, -1, -1):
            if len(erased) == m:
                break
            if i not in erased:
                erased.add(i)
        # available fragments (erased -> NULL); data parts None stay NULL
        full = [(parts[i] if i < k else parity[i - k]) for i in range(k + m)]
        frags = [None if i in erased else full[i] for i in range(k + m)]
        mask = sum(1 << i for i in erased)
        want = set(erase)
        rec = ref_recover(lib, k, m, frags, mask, want, SIZE)
        for i in want:
            out[f"{name}_rec{i}"] = rec[i]
        meta["cases"].append({"name": name, "k": k, "m": m, "pattern": pat,
                              "size": SIZE, "erased_mask": mask,
                              "wanted": sorted(want)})

    # --- CRC32 vectors ---
    crc_cases = []
    kat = [b"", b"a", b"abc", b"message digest", bytes(range(256)) * 7]
    for i, s in enumerate(kat):
        crc_cases.append({"input": f"kat{i}", "len": len(s), "seed": 0,
                          "crc": int(lib.ref_mycrc32(0, s, len(s)))})
        out[f"crc_kat{i}"] = np.frombuffer(s, np.uint8).copy() if s else np.zeros(0, np.uint8)
    blk = rng.integers(0, 256, 65536, np.uint8)
    out["crc_block64k"] = blk
    crc_cases.append({"input": "block64k", "len": 65536, "seed": 0,
                      "crc": int(lib.ref_mycrc32(0, blk.tobytes(), 65536))})
    crc_cases.append({"input": "block64k", "len": 65536, "seed": 0x12345678,
                      "crc": int(lib.ref_mycrc32(0x12345678, blk.tobytes(), 65536))})
    # zero block (hddspacemgr sparse-file special case)
    crc_cases.append({"input": "zeros64k", "len": 65536, "seed": 0,
                      "crc": int(lib.ref_mycrc32(0, bytes(65536), 65536))})
    # combine: split block64k at several points
    comb = []
    for split in (1, 3, 4096, 30000, 65535):
        c1 = lib.ref_mycrc32(0, blk.tobytes()[:split], split)
        c2 = lib.ref_mycrc32(0, blk.tobytes()[split:], 65536 - split)
        cc = lib.ref_mycrc32_combine(c1, c2, 65536 - split)
        comb.append({"split": split, "c1": int(c1), "c2": int(c2),
                     "combined": int(cc)})
    meta["crc"] = crc_cases
    meta["crc_combine"] = comb

    np.savez_compressed(os.path.join(HERE, "golden.npz"), **out)
    with open(os.path.join(HERE, "golden_meta.json"), "w") as f:
        json.dump(meta, f, indent=1)
    print(f"wrote {len(out)} arrays, {len(meta['cases'])} EC cases, "
          f"{len(crc_cases)} CRC vectors")


if __name__ == "__main__":
    sys.exit(main())
