"""ctypes wrapper over oracle/liboracle.so — TEST INFRASTRUCTURE ONLY.

Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
import this module.  The product (lizardfs_amd + liblizec) never does.

The C side (gf_oracle.c) restates the reference's GF(2^8)/RS/CRC math;
citations live there.
"""
import ctypes
import os
import subprocess

import numpy as np

HERE = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(HERE, "liboracle.so")


def _build_if_needed():
    if not os.path.exists(_SO):
        subprocess.run(["make", "-C", HERE, "liboracle.so"], check=True,
                       capture_output=True)


_lib = None


def lib():
    global _lib
    if _lib is None:
        _build_if_needed()
        _lib = ctypes.CDLL(_SO)
        L = _lib
        L.oracle_init()
        L.oracle_gf_mul.restype = ctypes.c_uint8
        L.oracle_gf_mul.argtypes = [ctypes.c_uint8, ctypes.c_uint8]
        L.oracle_gf_inv.restype = ctypes.c_uint8
        L.oracle_gf_inv.argtypes = [ctypes.c_uint8]
        L.oracle_gf_invert_matrix.restype = ctypes.c_int
        L.oracle_rs_recover.restype = ctypes.c_int
        L.oracle_rs_encode.restype = ctypes.c_int
        L.oracle_rs_make_tables.restype = ctypes.c_int
        L.oracle_crc32.restype = ctypes.c_uint32
        L.oracle_crc32.argtypes = [ctypes.c_uint32, ctypes.c_char_p, ctypes.c_uint32]
        L.oracle_crc32_combine.restype = ctypes.c_uint32
        L.oracle_crc32_combine.argtypes = [ctypes.c_uint32, ctypes.c_uint32,
                                           ctypes.c_uint32]
    return _lib


def _ptrs(arrs, n):
    a = (ctypes.c_void_p * n)()
    for i, x in enumerate(arrs):
        if x is not None:
            assert x.dtype == np.uint8 and x.flags["C_CONTIGUOUS"]
            a[i] = x.ctypes.data_as(ctypes.c_void_p).value
    return a


def gf_mul(a, b):
    return lib().oracle_gf_mul(a, b)


def gf_inv(a):
    return lib().oracle_gf_inv(a)


def gen_rs_matrix(k, m):
    """(k+m) x k Vandermonde matrix (galois_field_isal.cc:53)."""
    a = np.zeros((k + m, k), np.uint8)
    lib().oracle_gf_gen_rs_matrix(a.ctypes.data_as(ctypes.c_void_p), k + m, k)
    return a


def gen_cauchy1_matrix(k, m):
    a = np.zeros((k + m, k), np.uint8)
    lib().oracle_gf_gen_cauchy1_matrix(a.ctypes.data_as(ctypes.c_void_p), k + m, k)
    return a


def invert_matrix(mat):
    """Returns inverse or None if singular (galois_field_isal.cc:87)."""
    n = mat.shape[0]
    inp = mat.astype(np.uint8).copy()
    out = np.zeros((n, n), np.uint8)
    r = lib().oracle_gf_invert_matrix(inp.ctypes.data_as(ctypes.c_void_p),
                                      out.ctypes.data_as(ctypes.c_void_p), n)
    return out if r == 0 else None


def init_tables(coeffs):
    """Expand flat coefficient array into 32-byte ISA-L tables."""
    c = np.ascontiguousarray(coeffs, np.uint8).ravel()
    t = np.zeros(32 * c.size, np.uint8)
    lib().oracle_ec_init_tables(c.size, 1, c.ctypes.data_as(ctypes.c_void_p),
                                t.ctypes.data_as(ctypes.c_void_p))
    return t


def ec_encode_data(tbls, srcs_list, dests):
    """Raw kernel: dests modified in place (galois_field_encode.cc:28)."""
    n = len(srcs_list)
    d = len(dests)
    size = srcs_list[0].size
    lib().oracle_ec_encode_data(size, n, d,
                                tbls.ctypes.data_as(ctypes.c_void_p),
                                _ptrs(srcs_list, n), _ptrs(dests, d))


def rs_encode(k, m, data_parts, size):
    """ReedSolomon<32,32>::encode.  data_parts: list of k arrays (None=zeros).
    Returns list of m parity arrays."""
    parity = [np.zeros(size, np.uint8) for _ in range(m)]
    r = lib().oracle_rs_encode(k, m, _ptrs(data_parts, k), _ptrs(parity, m),
                               ctypes.c_size_t(size))
    assert r == 0, f"rs_encode failed: {r}"
    return parity


def rs_recover(k, m, fragments, erased_mask, want, size):
    """ReedSolomon<32,32>::recover.  fragments: k+m arrays or None.
    want: set of part indices to reconstruct.  Returns dict idx->array."""
    outs = [np.zeros(size, np.uint8) if i in want else None
            for i in range(k + m)]
    r = lib().oracle_rs_recover(k, m, _ptrs(fragments, k + m),
                                ctypes.c_uint64(erased_mask),
                                _ptrs(outs, k + m), ctypes.c_size_t(size))
    assert r == 0, f"rs_recover failed: {r}"
    return {i: outs[i] for i in want}


def rs_make_tables(k, m, present_mask, nonnull_mask, needed_mask):
    """Expanded gf tables for a recover() call; (tables, in_count, out_count)."""
    t = np.zeros(32 * 32 * 32, np.uint8)
    ic = ctypes.c_int()
    oc = ctypes.c_int()
    r = lib().oracle_rs_make_tables(k, m, ctypes.c_uint64(present_mask),
                                    ctypes.c_uint64(nonnull_mask),
                                    ctypes.c_uint64(needed_mask),
                                    t.ctypes.data_as(ctypes.c_void_p),
                                    ctypes.byref(ic), ctypes.byref(oc))
    assert r == 0, f"rs_make_tables failed: {r}"
    return t[:32 * ic.value * oc.value], ic.value, oc.value


def crc32(data, seed=0):
    b = bytes(data)
    return lib().oracle_crc32(seed, b, len(b))


def crc32_combine(c1, c2, len2):
    return lib().oracle_crc32_combine(c1, c2, len2)


def encode_stripes(k, m, part_len, nstripes, tbls, data, parity):
    """Threaded CPU-baseline encode over [S,k,part_len] / [S,m,part_len]."""
    lib().oracle_encode_stripes(k, m, part_len, nstripes,
                                tbls.ctypes.data_as(ctypes.c_void_p),
                                data.ctypes.data_as(ctypes.c_void_p),
                                parity.ctypes.data_as(ctypes.c_void_p))


def crc32_blocks(buf, block_len, seed=0):
    nblocks = buf.size // block_len
    crcs = np.zeros(nblocks, np.uint32)
    lib().oracle_crc32_blocks(buf.ctypes.data_as(ctypes.c_void_p),
                              block_len, nblocks, seed,
                              crcs.ctypes.data_as(ctypes.c_void_p))
    return crcs
