"""Randomized (k, m, erasure-pattern) sweep: GPU vs oracle, bit-exact.

Covers the full supported range (k in [2,32], m in [1,32], random NULL
zero-parts, random erasures incl. mixed data/parity, random part lengths
incl. ragged tile tails) beyond the fixed golden cases.
"""
import os

import numpy as np
import pytest

import oracle

pytestmark = pytest.mark.gpu

SEED = int(os.environ.get("LIZEC_SWEEP_SEED", "31337"))
TRIALS = int(os.environ.get("LIZEC_SWEEP_TRIALS", "24"))


def test_random_config_sweep():
    import torch
    from lizardfs_amd.ec import ReedSolomon

    rng = np.random.default_rng(SEED)
    for trial in range(TRIALS):
        k = int(rng.integers(2, 33))
        m = int(rng.integers(1, 33))
        nparts = k + m
        plen = int(rng.choice([16, 256, 4096, 16384, 16384 + 16,
                               32768 + 4096 + 64]))
        S = int(rng.integers(1, 4))
        data_np = rng.integers(0, 256, (S, k, plen), np.uint8)

        rs = ReedSolomon(k, m)
        data = torch.from_numpy(data_np).cuda()
        parity = rs.encode_batch(data)
        rs.sync()

        # oracle encode cross-check
        exp_par = [oracle.rs_encode(k, m, list(data_np[s]), plen)
                   for s in range(S)]
        got_par = parity.cpu().numpy()
        for s in range(S):
            for l in range(m):
                assert np.array_equal(got_par[s, l], exp_par[s][l]), \
                    (trial, k, m, s, l)

        # random erasure pattern: exactly m erased, subset wanted
        erased = sorted(rng.choice(nparts, m, replace=False).tolist())
        want = [i for i in erased if rng.random() < 0.7] or [erased[0]]
        # random zero (NULL) surviving data parts
        null_parts = {i for i in range(nparts)
                      if i not in erased and i < k and rng.random() < 0.15}
        dn = data_np.copy()
        for i in null_parts:
            dn[:, i] = 0
        parity2 = rs.encode_batch(torch.from_numpy(dn).cuda())
        rs.sync()
        frags = []
        for i in range(nparts):
            if i in erased or i in null_parts:
                frags.append(None)
            elif i < k:
                frags.append(torch.from_numpy(
                    np.ascontiguousarray(dn[:, i])).cuda())
            else:
                frags.append(parity2[:, i - k].contiguous())
        rec = rs.recover_batch(frags, erased=erased, want=set(want))
        rs.sync()
        par2_np = parity2.cpu().numpy()
        for i in want:
            expect = dn[:, i] if i < k else par2_np[:, i - k]
            assert np.array_equal(rec[i].cpu().numpy(), expect), \
                (trial, k, m, erased, i)


def test_full_size_roundtrip_wide_configs():
    """64 MiB stripes at the BASELINE widths beyond ec(8,2)."""
    import torch
    from lizardfs_amd.ec import ReedSolomon

    for k, m, erase in [(16, 4, (0, 7, 15, 17)), (32, 6, (2, 9, 30))]:
        plen = (64 * 1024 * 1024 // k) & ~15
        S = 1
        g = torch.Generator(device="cuda").manual_seed(k * 100 + m)
        data = torch.randint(0, 256, (S, k, plen), dtype=torch.uint8,
                             device="cuda", generator=g)
        rs = ReedSolomon(k, m)
        parity = rs.encode_batch(data)
        nparts = k + m
        erased = set(erase)
        for i in range(nparts - 1, -1, -1):
            if len(erased) == m:
                break
            if i not in erased:
                erased.add(i)
        frags = [None if i in erased else
                 (data[:, i, :] if i < k else parity[:, i - k, :])
                 for i in range(nparts)]
        rec = rs.recover_batch(frags, erased=erased, want=set(erase))
        rs.sync()
        for i in erase:
            expect = data[:, i] if i < k else parity[:, i - k]
            assert torch.equal(rec[i], expect), (k, m, i)


def test_random_crc_block_sweep():
    """Random block sizes/seeds across the CRC dispatch boundaries
    (fold fast path % 8 KiB, table fast path % 16 KiB, generic % 1 KiB),
    bit-exact vs the oracle."""
    import torch
    from lizardfs_amd import crc as crc_mod

    rng = np.random.default_rng(SEED + 1)
    for trial in range(max(8, TRIALS // 2)):
        blen = int(rng.integers(1, 257)) * 1024
        nblocks = int(rng.integers(1, 9))
        seed = int(rng.integers(0, 2**32)) if trial % 3 else 0
        buf = rng.integers(0, 256, nblocks * blen, np.uint8)
        got = crc_mod.crc32_blocks(
            torch.from_numpy(buf).cuda(), blen,
            seed=seed).cpu().numpy().view(np.uint32)
        exp = oracle.crc32_blocks(buf, blen, seed=seed)
        assert np.array_equal(got, exp), (trial, blen, nblocks, hex(seed))
