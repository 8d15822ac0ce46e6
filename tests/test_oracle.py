"""Pin the CPU oracle against the reference's golden vectors.

Golden vectors were produced by the reference's OWN code (oracle/_ref,
compiled unmodified from /root/reference by oracle/Makefile; generator:
tests/golden/make_golden.py) and are committed under tests/golden/.
These tests run everywhere (no GPU, no reference tree needed).
"""
import json
import os

import numpy as np
import pytest

import oracle

HERE = os.path.dirname(os.path.abspath(__file__))
GOLDEN = np.load(os.path.join(HERE, "golden", "golden.npz"))
META = json.load(open(os.path.join(HERE, "golden", "golden_meta.json")))


def load_case(case):
    k, m, name, size = case["k"], case["m"], case["name"], case["size"]
    parts = []
    for i in range(k):
        p = GOLDEN[f"{name}_data{i}"]
        parts.append(None if p.size == 0 else p)
    parity = [GOLDEN[f"{name}_parity{i}"] for i in range(m)]
    return k, m, name, size, parts, parity


@pytest.mark.parametrize("case", META["cases"], ids=lambda c: c["name"])
def test_encode_matches_reference(case):
    k, m, name, size, parts, parity = load_case(case)
    got = oracle.rs_encode(k, m, parts, size)
    for i in range(m):
        assert np.array_equal(got[i], parity[i]), f"parity {i} differs"


@pytest.mark.parametrize("case", META["cases"], ids=lambda c: c["name"])
def test_recover_matches_reference(case):
    k, m, name, size, parts, parity = load_case(case)
    mask = case["erased_mask"]
    want = set(case["wanted"])
    full = [(parts[i] if i < k else parity[i - k]) for i in range(k + m)]
    frags = [None if (mask >> i) & 1 else full[i] for i in range(k + m)]
    rec = oracle.rs_recover(k, m, frags, mask, want, size)
    for i in want:
        assert np.array_equal(rec[i], GOLDEN[f"{name}_rec{i}"]), \
            f"recovered part {i} differs"


def test_recovered_parts_equal_originals():
    """reed_solomon_unittest.cc:136-166 semantics: recovery returns the
    original bytes, not merely something the reference also computes."""
    rng = np.random.default_rng(7)
    for (k, m, erase) in [(4, 2, (0, 2)), (8, 2, (1, 5)), (8, 2, (3, 9)),
                          (32, 6, (2, 9, 30, 33, 34, 35))]:
        size = 2048
        parts = [rng.integers(0, 256, size, np.uint8) for _ in range(k)]
        parity = oracle.rs_encode(k, m, parts, size)
        full = parts + parity
        erased = set(erase)
        for i in range(k + m - 1, -1, -1):
            if len(erased) == m:
                break
            erased.add(i) if i not in erased else None
        mask = sum(1 << i for i in erased)
        frags = [None if i in erased else full[i] for i in range(k + m)]
        rec = oracle.rs_recover(k, m, frags, mask, set(erase), size)
        for i in erase:
            assert np.array_equal(rec[i], full[i]), (k, m, i)


def test_null_inputs_are_zero_parts():
    """reed_solomon.h:79: NULL input fragment == all-zero part."""
    k, m, size = 6, 2, 1024
    rng = np.random.default_rng(9)
    parts = [rng.integers(0, 256, size, np.uint8) for _ in range(k)]
    zero_idx = [1, 4]
    for i in zero_idx:
        parts[i] = np.zeros(size, np.uint8)
    explicit = oracle.rs_encode(k, m, parts, size)
    parts_null = [None if i in zero_idx else parts[i] for i in range(k)]
    with_null = oracle.rs_encode(k, m, parts_null, size)
    for a, b in zip(explicit, with_null):
        assert np.array_equal(a, b)


def test_matrix_invertibility_sweep():
    """reed_solomon_unittest.cc:252-319: every Vandermonde-derived recovery
    submatrix stays invertible for all 2-erasure combos, k in [2,32], m<=4
    (subset for speed; the full sweep is the reference's own test)."""
    for k in (2, 3, 5, 8, 16, 32):
        for m in (1, 2, 4):
            mat = oracle.gen_rs_matrix(k, m) if not (m >= 5 or (m == 4 and k > 20)) \
                else oracle.gen_cauchy1_matrix(k, m)
            nparts = k + m
            import itertools
            for erased in itertools.combinations(range(nparts), min(m, 2)):
                rows = [i for i in range(nparts) if i not in erased][:k]
                sub = mat[rows, :]
                assert oracle.invert_matrix(sub) is not None, (k, m, erased)


def test_gf_identities():
    for a in range(256):
        assert oracle.gf_mul(a, 1) == a
        assert oracle.gf_mul(1, a) == a
        assert oracle.gf_mul(a, 0) == 0
        if a:
            assert oracle.gf_mul(a, oracle.gf_inv(a)) == 1
    # spot associativity/commutativity
    rng = np.random.default_rng(3)
    for _ in range(200):
        a, b, c = rng.integers(0, 256, 3)
        assert oracle.gf_mul(a, b) == oracle.gf_mul(b, a)
        assert oracle.gf_mul(oracle.gf_mul(a, b), c) == \
            oracle.gf_mul(a, oracle.gf_mul(b, c))


def test_crc_golden():
    for c in META["crc"]:
        if c["input"].startswith("kat"):
            data = GOLDEN[f'crc_{c["input"]}'].tobytes()
        elif c["input"] == "block64k":
            data = GOLDEN["crc_block64k"].tobytes()
        else:
            data = bytes(c["len"])
        assert oracle.crc32(data, c["seed"]) == c["crc"], c


def test_crc_combine_golden():
    for c in META["crc_combine"]:
        got = oracle.crc32_combine(c["c1"], c["c2"], 65536 - c["split"])
        assert got == c["combined"], c


def test_crc_combine_matches_direct():
    rng = np.random.default_rng(11)
    data = rng.integers(0, 256, 10000, np.uint8).tobytes()
    whole = oracle.crc32(data)
    for split in (0, 1, 17, 5000, 9999, 10000):
        c1 = oracle.crc32(data[:split])
        c2 = oracle.crc32(data[split:])
        assert oracle.crc32_combine(c1, c2, 10000 - split) == whole, split


def test_encode_stripes_helper_matches_single():
    """The threaded bench helper must agree with per-stripe encode."""
    k, m, L, S = 4, 2, 512, 8
    rng = np.random.default_rng(5)
    data = rng.integers(0, 256, (S, k, L), np.uint8)
    parity = np.zeros((S, m, L), np.uint8)
    tbls, ic, oc = oracle.rs_make_tables(
        k, m, present_mask=(1 << k) - 1, nonnull_mask=(1 << k) - 1,
        needed_mask=((1 << m) - 1) << k)
    assert (ic, oc) == (k, m)
    oracle.encode_stripes(k, m, L, S, tbls, data, parity)
    for s in range(S):
        exp = oracle.rs_encode(k, m, list(data[s]), L)
        for l in range(m):
            assert np.array_equal(parity[s, l], exp[l])
