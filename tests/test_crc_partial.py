"""Partial-block CRC algebra (crc.h:27-29 macros + hdd_write's splice path,
hddspacemgr.cc:1952-2003) — liblizec host helpers vs direct computation and
the pinned oracle.  CPU-only."""
import ctypes

import numpy as np

import oracle
from lizardfs_amd import lib as L


def lib():
    l = L.lib()
    l.lizec_crc32_zeroblock.restype = ctypes.c_uint32
    l.lizec_crc32_zeroblock.argtypes = [ctypes.c_uint32, ctypes.c_uint32]
    l.lizec_crc32_zeroexpanded.restype = ctypes.c_uint32
    l.lizec_crc32_zeroexpanded.argtypes = [ctypes.c_uint32, ctypes.c_char_p,
                                           ctypes.c_uint32, ctypes.c_uint32]
    l.lizec_crc32_xorblocks.restype = ctypes.c_uint32
    l.lizec_crc32_xorblocks.argtypes = [ctypes.c_uint32] * 3 + [ctypes.c_uint32]
    l.lizec_crc32_splice.restype = ctypes.c_uint32
    l.lizec_crc32_splice.argtypes = [ctypes.c_uint32] * 6
    l.lizec_recompute_crc_if_block_empty.argtypes = [
        ctypes.c_char_p, ctypes.c_uint32, ctypes.POINTER(ctypes.c_uint32)]
    return l


def test_zeroblock_equals_crc_of_zeros():
    l = lib()
    for n in (0, 1, 17, 65536):
        assert l.lizec_crc32_zeroblock(0, n) == oracle.crc32(bytes(n)), n
    # seeded: crc.h:27 semantics — combine((crc)^~0, ~0, zeros)
    rng = np.random.default_rng(1)
    data = rng.integers(0, 256, 100, np.uint8).tobytes()
    c = oracle.crc32(data)
    assert l.lizec_crc32_zeroblock(c, 5000) == oracle.crc32(data + bytes(5000))


def test_zeroexpanded():
    l = lib()
    rng = np.random.default_rng(2)
    data = rng.integers(0, 256, 777, np.uint8).tobytes()
    got = l.lizec_crc32_zeroexpanded(0, data, len(data), 1234)
    assert got == oracle.crc32(data + bytes(1234))


def test_xorblocks():
    l = lib()
    rng = np.random.default_rng(3)
    a = rng.integers(0, 256, 4096, np.uint8)
    b = rng.integers(0, 256, 4096, np.uint8)
    got = l.lizec_crc32_xorblocks(0, oracle.crc32(a.tobytes()),
                                  oracle.crc32(b.tobytes()), 4096)
    assert got == oracle.crc32((a ^ b).tobytes())


def test_splice_matches_direct():
    """hdd_write's recombine: replacing [offset, offset+size) of a 64 KiB
    block must yield the direct CRC of the resulting block."""
    l = lib()
    rng = np.random.default_rng(4)
    BL = 65536
    block = rng.integers(0, 256, BL, np.uint8)
    for offset, size in [(0, BL), (0, 1000), (512, 1024), (1, 3),
                         (65000, 536), (65535, 1), (4096, 61440)]:
        newdata = rng.integers(0, 256, size, np.uint8)
        precrc = oracle.crc32(block[:offset].tobytes())
        crc = oracle.crc32(newdata.tobytes())
        postcrc = oracle.crc32(block[offset + size:].tobytes())
        got = l.lizec_crc32_splice(precrc, offset, crc, size, postcrc, BL)
        spliced = block.copy()
        spliced[offset:offset + size] = newdata
        assert got == oracle.crc32(spliced.tobytes()), (offset, size)


def test_recompute_crc_if_block_empty():
    l = lib()
    zeros = bytes(65536)
    c = ctypes.c_uint32(0)
    l.lizec_recompute_crc_if_block_empty(zeros, 65536, ctypes.byref(c))
    assert c.value == oracle.crc32(zeros)
    # non-zero crc untouched
    c = ctypes.c_uint32(123)
    l.lizec_recompute_crc_if_block_empty(zeros, 65536, ctypes.byref(c))
    assert c.value == 123
    # non-empty block untouched
    nz = b"\x01" + bytes(65535)
    c = ctypes.c_uint32(0)
    l.lizec_recompute_crc_if_block_empty(nz, 65536, ctypes.byref(c))
    assert c.value == 0
