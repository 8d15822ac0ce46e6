"""Product host-side (liblizec.so, CPU surfaces) vs the pinned oracle.

Runs everywhere (no GPU): the C-ABI library must load, export every symbol
include/lizec.h declares, and its host matrix/CRC/slice algebra must be
bit-exact vs the oracle (itself pinned against the reference by
test_oracle.py).
"""
import ctypes
import itertools

import numpy as np
import pytest

import oracle
from lizardfs_amd import lib as L
from lizardfs_amd import slice_traits as st

U8P = ctypes.POINTER(ctypes.c_uint8)


def u8p(a):
    return a.ctypes.data_as(U8P)


def test_exports_complete():
    """Every entry point declared in include/lizec.h resolves."""
    lib = L.lib()
    for sym in ["gf_gen_rs_matrix", "gf_gen_cauchy1_matrix",
                "gf_invert_matrix", "ec_init_tables", "ec_encode_data",
                "lizec_crc32", "lizec_crc32_combine", "lizec_crc32_init",
                "lizec_rs_tables", "lizec_rs_encode_tables",
                "lizec_slice_type_ec", "lizec_slice_is_ec",
                "lizec_slice_data_parts", "lizec_slice_parity_parts",
                "lizec_chunk_part_id", "lizec_chunk_part_slice_type",
                "lizec_chunk_part_index", "lizec_chunk_part_length",
                "lizec_gpu_count", "lizec_engine_create",
                "lizec_engine_destroy", "lizec_engine_sync",
                "lizec_ec_encode_batch", "lizec_crc32_batch",
                "lizec_ec_plan_create", "lizec_ec_plan_run",
                "lizec_ec_plan_destroy", "lizec_scrub_batch",
                "lizec_scrub_batch_strided", "lizec_host_alloc",
                "lizec_host_free", "lizec_replicate_run"]:
        assert getattr(lib, sym, None) is not None, sym


def test_gpu_apis_fail_loudly_without_gpu():
    """The product path has no CPU fallback: on a GPU-less host, engine
    creation must raise (LIZEC_ENOGPU), never silently degrade."""
    import torch
    if torch.cuda.is_available():
        pytest.skip("GPU present; the loud-failure path needs a CPU-only host")
    import lizardfs_amd.lib as _l
    h = ctypes.c_void_p()
    rc = _l.lib().lizec_engine_create(ctypes.byref(h), 0)
    assert rc == -3  # LIZEC_ENOGPU
    with pytest.raises(L.LizecError, match="ENOGPU"):
        _l.check(rc, "engine")
    with pytest.raises(L.LizecError, match="ENOGPU"):
        _l.engine(0)


def test_mangled_crc_aliases_present():
    """The reference's C++-mangled mycrc32 symbols (crc.h:25-31) resolve."""
    lib = L.lib()
    for sym in ["_Z7mycrc32jPKhj", "_Z15mycrc32_combinejjj", "_Z12mycrc32_initv"]:
        assert getattr(lib, sym, None) is not None, sym


@pytest.mark.parametrize("k,m", [(2, 1), (3, 1), (8, 2), (16, 4), (20, 4),
                                 (22, 4), (32, 6), (32, 32), (2, 32)])
def test_matrix_generators_match_oracle(k, m):
    lib = L.lib()
    a = np.zeros((k + m, k), np.uint8)
    lib.gf_gen_rs_matrix(u8p(a), k + m, k)
    assert np.array_equal(a, oracle.gen_rs_matrix(k, m))
    lib.gf_gen_cauchy1_matrix(u8p(a), k + m, k)
    assert np.array_equal(a, oracle.gen_cauchy1_matrix(k, m))


def test_invert_matches_oracle():
    lib = L.lib()
    rng = np.random.default_rng(21)
    for n in (1, 2, 5, 16, 32):
        for trial in range(8):
            mat = rng.integers(0, 256, (n, n), np.uint8)
            inp = mat.copy()
            out = np.zeros((n, n), np.uint8)
            r = lib.gf_invert_matrix(u8p(inp), u8p(out), n)
            exp = oracle.invert_matrix(mat)
            if exp is None:
                assert r == -1
            else:
                assert r == 0
                assert np.array_equal(out, exp)
    # deliberately singular
    s = np.zeros((3, 3), np.uint8)
    out = np.zeros((3, 3), np.uint8)
    assert lib.gf_invert_matrix(u8p(s.copy()), u8p(out), 3) == -1


def test_init_tables_matches_oracle():
    lib = L.lib()
    coeffs = np.arange(256, dtype=np.uint8)
    got = np.zeros(32 * 256, np.uint8)
    lib.ec_init_tables(256, 1, u8p(coeffs), u8p(got))
    assert np.array_equal(got, oracle.init_tables(coeffs))


def test_host_ec_encode_data_matches_oracle():
    lib = L.lib()
    rng = np.random.default_rng(33)
    k, m, n = 5, 3, 2000
    srcs = [np.ascontiguousarray(rng.integers(0, 256, n, np.uint8))
            for _ in range(k)]
    coeffs = rng.integers(0, 256, k * m, np.uint8)
    tbl = oracle.init_tables(coeffs)
    dst = [np.zeros(n, np.uint8) for _ in range(m)]
    sp = (ctypes.c_void_p * k)(*[s.ctypes.data_as(ctypes.c_void_p).value
                                 for s in srcs])
    dp = (ctypes.c_void_p * m)(*[d.ctypes.data_as(ctypes.c_void_p).value
                                 for d in dst])
    lib.ec_encode_data(n, k, m, u8p(tbl), sp, dp)
    exp = [np.zeros(n, np.uint8) for _ in range(m)]
    oracle.ec_encode_data(tbl, srcs, exp)
    for a, b in zip(dst, exp):
        assert np.array_equal(a, b)


def test_rs_tables_match_oracle():
    lib = L.lib()
    rng = np.random.default_rng(44)
    cases = [(8, 2), (4, 2), (16, 4), (32, 6), (22, 4), (3, 1), (2, 32)]
    for k, m in cases:
        nparts = k + m
        allm = (1 << nparts) - 1
        for trial in range(12):
            erased = rng.choice(nparts, m, replace=False)
            present = allm & ~sum(1 << int(i) for i in erased)
            # random NULL (=zero) subset of present parts
            nonnull = present
            for i in range(nparts):
                if (present >> i) & 1 and rng.random() < 0.2 and \
                        nonnull != (1 << i):
                    nonnull &= ~(1 << i)
            if nonnull == 0:
                continue
            wanted = [int(i) for i in erased if rng.random() < 0.8] or \
                     [int(erased[0])]
            needed = sum(1 << i for i in wanted)
            exp_t, exp_ic, exp_oc = oracle.rs_make_tables(
                k, m, present, nonnull, needed)
            got = np.zeros(32 * 32 * 32, np.uint8)
            ic = ctypes.c_int()
            oc = ctypes.c_int()
            r = lib.lizec_rs_tables(k, m, present, nonnull, needed,
                                    u8p(got), ctypes.byref(ic),
                                    ctypes.byref(oc))
            assert r == 0, (k, m, present, nonnull, needed)
            assert (ic.value, oc.value) == (exp_ic, exp_oc)
            assert np.array_equal(got[:exp_t.size], exp_t), \
                (k, m, present, nonnull, needed)


def test_host_crc_matches_oracle():
    rng = np.random.default_rng(55)
    for ln in (0, 1, 3, 7, 64, 1000, 65536):
        data = rng.integers(0, 256, ln, np.uint8).tobytes()
        for seed in (0, 0xDEADBEEF):
            got = L.lib().lizec_crc32(seed, data, ln)
            assert got == oracle.crc32(data, seed), (ln, seed)
    # combine
    data = rng.integers(0, 256, 5000, np.uint8).tobytes()
    whole = oracle.crc32(data)
    for split in (0, 1, 2499, 4999, 5000):
        c1 = L.lib().lizec_crc32(0, data[:split], split)
        c2 = L.lib().lizec_crc32(0, data[split:], 5000 - split)
        assert L.lib().lizec_crc32_combine(c1, c2, 5000 - split) == whole


def test_slice_algebra():
    lib = L.lib()
    # goal.h:118: $ec(2,1) is the first EC type, id 10
    assert lib.lizec_slice_type_ec(2, 1) == 10
    assert lib.lizec_slice_type_ec(32, 32) == 10 + 31 * 32 - 1
    assert lib.lizec_slice_type_ec(8, 2) == st.ec_slice_type(8, 2)
    for k, m in itertools.product(range(2, 33), (1, 2, 4, 17, 32)):
        t = lib.lizec_slice_type_ec(k, m)
        assert t == st.ec_slice_type(k, m)
        assert lib.lizec_slice_is_ec(t) == 1
        assert lib.lizec_slice_data_parts(t) == k == st.data_parts(t)
        assert lib.lizec_slice_parity_parts(t) == m == st.parity_parts(t)
        for part in (0, k - 1, k, k + m - 1):
            pid = lib.lizec_chunk_part_id(t, part)
            assert pid == st.chunk_part_id(t, part)
            assert lib.lizec_chunk_part_slice_type(pid) == t
            assert lib.lizec_chunk_part_index(pid) == part
    assert lib.lizec_slice_is_ec(0) == 0
    assert lib.lizec_slice_is_ec(9) == 0


def test_chunk_part_length():
    """slice_traits.h:332-349 restated twice (C and Python) must agree, and
    part lengths must sum to the chunk length for data parts."""
    lib = L.lib()
    for k, m in [(2, 1), (8, 2), (16, 4), (32, 6)]:
        t = st.ec_slice_type(k, m)
        for chunk_len in (0, 1, 65536, 65537, 4 * 65536 + 123,
                          st.CHUNK_SIZE, st.CHUNK_SIZE - 7):
            total = 0
            for part in range(k + m):
                a = lib.lizec_chunk_part_length(t, part, chunk_len)
                b = st.chunk_part_length(t, part, chunk_len)
                assert a == b, (k, m, part, chunk_len)
                if part < k:
                    total += a
            assert total == chunk_len, (k, m, chunk_len)
