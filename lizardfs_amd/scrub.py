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
HDD_BLOCK = st.BLOCK_SIZE + 4     # chunk.h:40 kHddBlockSize (interleaved)
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
                      crc32_fn=None, fmt="moosefs"):
    """Assemble a part-file image (numpy uint8) from 64 KiB blocks in
    either on-disk format (chunk.cc MooseFSChunk / InterleavedChunk).
    Used by tests/bench; CRCs are filled with crc32_fn (default: liblizec
    host crc32) and stored big-endian (put32bit)."""
    if crc32_fn is None:
        from . import crc as lcrc
        crc32_fn = lcrc.crc32
    nblocks = len(blocks_data)
    if fmt == "interleaved":
        # chunk.cc:195-209: no signature; 4-byte CRC inline before each block
        img = np.zeros(nblocks * HDD_BLOCK, np.uint8)
        for b, blk in enumerate(blocks_data):
            assert blk.size == st.BLOCK_SIZE
            crc = crc32_fn(blk.tobytes())
            img[b * HDD_BLOCK:b * HDD_BLOCK + 4] = \
                np.frombuffer(struct.pack(">I", crc), np.uint8)
            img[b * HDD_BLOCK + 4:(b + 1) * HDD_BLOCK] = blk
        return img
    hdr = header_size(slice_type)
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


def _scrub_group(idx, images, fmts, device, results):
    """One lizec_scrub_batch_strided call for images sharing a format."""
    n = len(idx)
    dptrs = np.zeros(n, np.uint64)
    doffs = np.zeros(n, np.uint32)
    coffs = np.zeros(n, np.uint32)
    counts = np.zeros(n, np.uint32)
    fmt = fmts[idx[0]]
    if fmt == "interleaved":
        bstride, cstride = HDD_BLOCK, HDD_BLOCK
    else:
        bstride, cstride = st.BLOCK_SIZE, 4
    for j, i in enumerate(idx):
        img, slice_type = images[i][0], images[i][1]
        if fmt == "interleaved":
            dptrs[j] = img.data_ptr()
            doffs[j] = 4
            coffs[j] = 0
            counts[j] = img.numel() // HDD_BLOCK
        else:
            hdr = header_size(slice_type)
            dptrs[j] = img.data_ptr()
            doffs[j] = hdr
            coffs[j] = SIGNATURE_BLOCK
            counts[j] = (img.numel() - hdr) // st.BLOCK_SIZE
    if counts.max(initial=0) == 0:
        return
    status = torch.empty(n, dtype=torch.int32, device=f"cuda:{device}")
    stream = torch.cuda.current_stream(device).cuda_stream
    L.check(L.lib().lizec_scrub_batch_strided(
        L.engine(device),
        dptrs.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
        doffs.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
        coffs.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
        counts.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
        n, bstride, cstride, ctypes.c_void_p(status.data_ptr()),
        ctypes.c_void_p(stream)), "lizec_scrub_batch_strided")
    torch.cuda.synchronize(device)
    for j, i in enumerate(idx):
        s = int(status[j].item())
        results[i] = None if s == CLEAN else s


def scrub_batch(images, device=0):
    """Verify a batch of chunk-part images resident on the GPU.

    images: list of (tensor, slice_type) or (tensor, slice_type, fmt) with
    fmt in {"moosefs" (default), "interleaved"} — tensor is the flat uint8
    CUDA image (full file bytes).  MooseFS images get a host-side signature
    check (1 KiB read, as hdd_int_chunk_readcrc does); then every block is
    CRC-checked on the GPU.  Returns list of first-damaged-block index or
    None if clean; a bad signature reports -2 (the reference's
    LIZARDFS_ERROR_IO class)."""
    n = len(images)
    images = [t if len(t) == 3 else (t[0], t[1], "moosefs") for t in images]
    fmts = [t[2] for t in images]
    results = [None] * n
    groups = {}
    for i, (img, slice_type, fmt) in enumerate(images):
        if img.dtype != torch.uint8 or not img.is_cuda or \
                not img.is_contiguous():
            raise ValueError("images must be contiguous CUDA uint8 tensors")
        if fmt == "moosefs":
            head = img[:SIGNATURE_BLOCK].cpu().numpy()
            parsed = parse_signature(head)
            if parsed is None or parsed[2] != slice_type:
                results[i] = -2
                continue
        groups.setdefault(fmt, []).append(i)
    for fmt, idx in groups.items():
        _scrub_group(idx, images, fmts, device, results)
    return results
