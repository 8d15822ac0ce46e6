"""Batched chunk scrub — the chunkserver's hdd_int_test loop
(hddspacemgr.cc:2148-2212) over MooseFS-format chunk-part files, on GPU.

On-disk format (chunkserver/chunk.cc, chunk_signature.cc):
  [0)      1 KiB signature block: 8-byte id "LIZC 1.1" + u64 BE chunk id
           + u32 BE version + u16 BE ChunkPartType id (chunk_signature.cc:87-90)
  [1024)   CRC array: big-endian u32 per block, 4 * maxBlocksInFile bytes
           (chunk.cc getCrcOffset/getCrcBlockSize)
  [hdr)    64 KiB blocks; hdr = 1024 + 4*maxBlocksInFile, rounded up to a
           4 KiB multiple for EC/xor parts (chunk.cc getHeaderSize)
maxBlocksInFile = ceil(1024 / data_parts) (chunk.cc:74-77).
"""
import ctypes
import struct

import numpy as np
import torch

from . import lib as L
from . import slice_traits as st

SIGNATURE_ID = b"LIZC 1.1"        # chunk_signature.cc:30
SIGNATURE_BLOCK = 1024            # chunk.h:156 kMaxSignatureBlockSize
DISK_BLOCK = 4096                 # chunk.h:161 kDiskBlockSize
CLEAN = 0x7FFFFFFF


def max_blocks_in_file(slice_type):
    """chunk.cc:74-77."""
    k = st.data_parts(slice_type)
    return (st.BLOCKS_IN_CHUNK + k - 1) // k


def header_size(slice_type):
    """chunk.cc getHeaderSize: signature + CRC array, rounded to 4 KiB for
    non-standard (EC/xor) slice types."""
    req = SIGNATURE_BLOCK + 4 * max_blocks_in_file(slice_type)
    if slice_type == st.K_STANDARD:
        return req
    return (req + DISK_BLOCK - 1) // DISK_BLOCK * DISK_BLOCK


def build_signature(chunk_id, version, slice_type, part):
    """chunk_signature.cc:87-90 (LIZC 1.1 serialization, big-endian)."""
    pid = st.chunk_part_id(slice_type, part)
    return SIGNATURE_ID + struct.pack(">QIH", chunk_id, version, pid)


def parse_signature(buf):
    """Returns (chunk_id, version, slice_type, part) or None if invalid
    (hdd_int_chunk_readcrc's hasValidSignatureId check)."""
    if bytes(buf[:8]) != SIGNATURE_ID:
        return None
    cid, ver, pid = struct.unpack(">QIH", bytes(buf[8:22]))
    return cid, ver, st.chunk_part_slice_type(pid), st.chunk_part_index(pid)


def build_chunk_image(chunk_id, version, slice_type, part, blocks_data,
                      crc32_fn=None):
    """Assemble a MooseFS part-file image (numpy uint8) from 64 KiB blocks.
    Used by tests/bench; the CRC array is filled with crc32_fn (default:
    liblizec host crc32)."""
    if crc32_fn is None:
        from . import crc as lcrc
        crc32_fn = lcrc.crc32
    hdr = header_size(slice_type)
    nblocks = len(blocks_data)
    assert nblocks <= max_blocks_in_file(slice_type)
    img = np.zeros(hdr + nblocks * st.BLOCK_SIZE, np.uint8)
    sig = build_signature(chunk_id, version, slice_type, part)
    img[:len(sig)] = np.frombuffer(sig, np.uint8)
    for b, blk in enumerate(blocks_data):
        assert blk.size == st.BLOCK_SIZE
        img[hdr + b * st.BLOCK_SIZE:hdr + (b + 1) * st.BLOCK_SIZE] = blk
        crc = crc32_fn(blk.tobytes())
        img[SIGNATURE_BLOCK + 4 * b:SIGNATURE_BLOCK + 4 * b + 4] = \
            np.frombuffer(struct.pack(">I", crc), np.uint8)
    return img


def scrub_batch(images, device=0):
    """Verify a batch of chunk-part images resident on the GPU.

    images: list of (tensor, slice_type) — tensor is the flat uint8 CUDA
    image (full file bytes).  Validates each signature (host-side, 1 KiB
    read, as hdd_int_chunk_readcrc does) then CRC-checks every block on
    the GPU.  Returns list of first-damaged-block index or None if clean;
    a bad signature reports -2 (the reference's LIZARDFS_ERROR_IO class).
    """
    n = len(images)
    dptrs = np.zeros(n, np.uint64)
    doffs = np.zeros(n, np.uint32)
    coffs = np.zeros(n, np.uint32)
    counts = np.zeros(n, np.uint32)
    sig_bad = [False] * n
    for i, (img, slice_type) in enumerate(images):
        if img.dtype != torch.uint8 or not img.is_cuda or \
                not img.is_contiguous():
            raise ValueError("images must be contiguous CUDA uint8 tensors")
        head = img[:SIGNATURE_BLOCK].cpu().numpy()
        parsed = parse_signature(head)
        if parsed is None or parsed[2] != slice_type:
            sig_bad[i] = True
            counts[i] = 0
            continue
        hdr = header_size(slice_type)
        nb = (img.numel() - hdr) // st.BLOCK_SIZE
        dptrs[i] = img.data_ptr()
        doffs[i] = hdr
        coffs[i] = SIGNATURE_BLOCK
        counts[i] = nb
    if counts.max(initial=0) == 0:
        return [-2 if bad else None for bad in sig_bad]

    status = torch.empty(n, dtype=torch.int32, device=f"cuda:{device}")
    stream = torch.cuda.current_stream(device).cuda_stream
    # chunks with bad signatures get count 0 (skipped; stay CLEAN on GPU)
    counts_safe = np.maximum(counts, 0)
    L.check(L.lib().lizec_scrub_batch(
        L.engine(device),
        dptrs.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
        doffs.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
        coffs.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
        counts_safe.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
        n, ctypes.c_void_p(status.data_ptr()), ctypes.c_void_p(stream)),
        "lizec_scrub_batch")
    torch.cuda.synchronize(device)
    out = []
    for i, s in enumerate(status.cpu().numpy()):
        if sig_bad[i]:
            out.append(-2)
        elif s == CLEAN:
            out.append(None)
        else:
            out.append(int(s))
    return out
