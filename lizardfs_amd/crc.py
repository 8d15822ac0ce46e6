"""Per-block CRC32 — the chunkserver's read/write/scrub gate
(hddspacemgr.cc:1918-1920, :1525-1594, :2148-2212), batched on GPU.

Host-side scalar crc32/crc32_combine mirror common/crc.h:25-31 for the
protocol edges (packet CRCs, partial-block combines)."""
import ctypes

import numpy as np
import torch

from . import lib as L
from . import slice_traits


def crc32(data, seed=0):
    """Host scalar CRC-32 (zlib-compatible; crc.cc:113-151 semantics)."""
    b = bytes(data)
    return L.lib().lizec_crc32(seed, b, len(b))


def crc32_combine(crc1, crc2, len2):
    """crc.cc:207-224 semantics (crc of concatenation)."""
    return L.lib().lizec_crc32_combine(crc1, crc2, len2)


def crc32_blocks(buf, block_len=slice_traits.BLOCK_SIZE, seed=0, out=None,
                 device=None):
    """CRC32 of consecutive fixed-size blocks of a device buffer.

    buf: uint8 CUDA tensor, flat or any shape with numel % block_len == 0.
    Returns int32 CUDA tensor [nblocks] holding the CRCs (bit pattern; view
    as uint32).  block_len must be a multiple of 1024.
    """
    if buf.dtype != torch.uint8 or not buf.is_cuda or not buf.is_contiguous():
        raise ValueError("buf must be a contiguous CUDA uint8 tensor")
    n = buf.numel()
    if n % block_len:
        raise ValueError("buffer size must be a multiple of block_len")
    nblocks = n // block_len
    dev = buf.device.index or 0
    if out is None:
        out = torch.empty(nblocks, dtype=torch.int32, device=buf.device)
    stream = torch.cuda.current_stream(dev).cuda_stream
    L.check(L.lib().lizec_crc32_batch(
        L.engine(dev), ctypes.c_void_p(buf.data_ptr()), block_len, nblocks,
        ctypes.c_uint32(seed), ctypes.c_void_p(out.data_ptr()),
        ctypes.c_void_p(stream)), "lizec_crc32_batch")
    return out
