"""Randomized property tests (hypothesis) — CPU-only, oracle + product host.

Properties are the domain's own invariants (round trips,
linearity, combine algebra), independent of the fixed golden cases.
"""
import numpy as np
import pytest
from hypothesis import given, settings, strategies as st_

import oracle
from lizardfs_amd import lib as L


@settings(max_examples=40, deadline=None)
@given(st_.binary(min_size=0, max_size=3000),
       st_.binary(min_size=0, max_size=3000),
       st_.integers(min_value=0, max_value=0xFFFFFFFF))
def test_crc_combine_is_concat(a, b, seed):
    """combine(crc(seed,A), crc(0,B), |B|) == crc(seed, A||B) — both
    product and oracle (crc.cc:207-224 semantics)."""
    lib = L.lib()
    ca = lib.lizec_crc32(seed, a, len(a))
    cb = lib.lizec_crc32(0, b, len(b))
    whole = lib.lizec_crc32(seed, a + b, len(a) + len(b))
    assert lib.lizec_crc32_combine(ca, cb, len(b)) == whole
    assert oracle.crc32_combine(ca, cb, len(b)) == whole


@settings(max_examples=30, deadline=None)
@given(st_.integers(min_value=2, max_value=32),
       st_.integers(min_value=1, max_value=32),
       st_.randoms(use_true_random=False))
def test_rs_any_k_of_n_reconstructs(k, m, rnd):
    """Erase any m parts; the survivors reconstruct the originals
    bit-exactly (reed_solomon_unittest.cc:136's property, generalized)."""
    size = 64
    rng = np.random.default_rng(rnd.randrange(2**32))
    parts = [rng.integers(0, 256, size, np.uint8) for _ in range(k)]
    parity = oracle.rs_encode(k, m, parts, size)
    full = parts + parity
    erased = sorted(rng.choice(k + m, m, replace=False).tolist())
    frags = [None if i in erased else full[i] for i in range(k + m)]
    mask = sum(1 << i for i in erased)
    rec = oracle.rs_recover(k, m, frags, mask, set(erased), size)
    for i in erased:
        assert np.array_equal(rec[i], full[i]), (k, m, erased, i)


@settings(max_examples=30, deadline=None)
@given(st_.integers(min_value=2, max_value=16),
       st_.integers(min_value=1, max_value=8),
       st_.integers(min_value=0, max_value=2**32 - 1))
def test_encode_is_gf_linear(k, m, seed):
    """parity(a XOR b) == parity(a) XOR parity(b): the code is linear over
    GF(2), so any implementation disagreement shows up under XOR."""
    size = 48
    rng = np.random.default_rng(seed)
    a = [rng.integers(0, 256, size, np.uint8) for _ in range(k)]
    b = [rng.integers(0, 256, size, np.uint8) for _ in range(k)]
    pa = oracle.rs_encode(k, m, a, size)
    pb = oracle.rs_encode(k, m, b, size)
    pxor = oracle.rs_encode(k, m, [x ^ y for x, y in zip(a, b)], size)
    for l in range(m):
        assert np.array_equal(pxor[l], pa[l] ^ pb[l])


def test_python_api_error_paths():
    """Validation raises before any GPU work (usable on CPU-only hosts)."""
    from lizardfs_amd import slice_traits as st
    with pytest.raises(ValueError):
        st.ec_slice_type(1, 1)
    with pytest.raises(ValueError):
        st.ec_slice_type(33, 1)
    with pytest.raises(ValueError):
        st.ec_slice_type(8, 0)

    import torch
    if torch.cuda.is_available():
        pytest.skip("remaining checks target the CPU-only container")
    from lizardfs_amd.xor import XorSlice
    with pytest.raises(L.LizecError, match="ENOGPU"):
        XorSlice(5)  # engine creation fails loudly on CPU-only hosts
    from lizardfs_amd.ec import ReedSolomon
    with pytest.raises(L.LizecError, match="ENOGPU"):
        ReedSolomon(8, 2)
