"""XOR slice family (xor2..xor9) — host drop-in and GPU parity.

Reference semantics: parity part (slice part 0) = byte-XOR of the level
data parts (xor_read_plan.h:39, chunk_writer.cc:373-381, block_xor.cc:47);
recovery of a missing data part = parity XOR the other data parts.
"""
import ctypes

import numpy as np
import pytest

from lizardfs_amd import lib as L
from lizardfs_amd import slice_traits as st


def test_host_blockxor_matches_numpy():
    lib = L.lib()
    rng = np.random.default_rng(3)
    for size in (0, 1, 7, 8, 9, 4096, 65536, 65537):
        dest = rng.integers(0, 256, max(size, 1), np.uint8)
        src = rng.integers(0, 256, max(size, 1), np.uint8)
        exp = dest[:size] ^ src[:size]
        lib.lizec_blockxor(dest.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
                           src.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
                           size)
        assert np.array_equal(dest[:size], exp), size


def test_mangled_blockxor_alias_present():
    lib = L.lib()
    assert getattr(lib, "_Z8blockXorPhPKhm", None) is not None


def test_xor_slice_types():
    # goal.h:110-117: kXor2=2 .. kXor9=9; parity is slice part 0
    for level in range(2, 10):
        t = st.K_XOR2 + level - 2
        assert st.is_xor(t)
        assert st.data_parts(t) == level
        assert st.parity_parts(t) == 1


@pytest.mark.gpu
@pytest.mark.parametrize("level", [2, 5, 9])
def test_xor_parity_and_recovery_gpu(level):
    import torch
    from lizardfs_amd.xor import XorSlice

    S, plen = 4, 32768
    rng = np.random.default_rng(40 + level)
    data_np = rng.integers(0, 256, (S, level, plen), np.uint8)
    data = torch.from_numpy(data_np).cuda()
    xs = XorSlice(level)
    parity = xs.parity_batch(data)
    torch.cuda.synchronize()
    exp = data_np[:, 0].copy()
    for j in range(1, level):
        exp ^= data_np[:, j]
    assert np.array_equal(parity.cpu().numpy(), exp)

    # recover data part 1 from parity + others
    others = [data[:, j, :] for j in range(level) if j != 1]
    rec = xs.recover_data_batch(parity, others)
    torch.cuda.synchronize()
    assert np.array_equal(rec.cpu().numpy(), data_np[:, 1])
