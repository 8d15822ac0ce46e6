"""Chunkserver-side EC part rebuild — the compute core of
ChunkReplicator::replicate (chunk_replicator.cc:139-196), batched on GPU.

The reference's replicate loop pulls surviving parts from peers, recovers
the missing part via SliceRecoveryPlanner + ReedSolomon, CRCs every
recovered 64 KiB block (chunk_replicator.cc:189), and writes a MooseFS
part file.  Networking/disk stay with the host; this module does the whole
compute pipeline on the GPU: recover -> per-block CRC -> assembled part
image (header + big-endian CRC array + blocks) ready to pwrite.
"""
import ctypes

import numpy as np
import torch

from . import crc as lcrc
from . import lib as L
from . import scrub
from . import slice_traits as st
from .ec import ReedSolomon


def rebuild_part_images(k, m, fragments, erased, want, chunk_ids, version,
                        device=0):
    """Recover the `want` parts of a batch of chunks and build their
    MooseFS part images on the GPU.

    fragments: list of k+m entries ([S, L] uint8 CUDA tensors or None),
      part data only (no headers), L a multiple of 64 KiB; row s belongs
      to chunk chunk_ids[s].
    erased: exactly m part indices (pad like ec_read_plan.h:126-133).
    want: subset of erased to rebuild.
    Returns dict (chunk_index s, part_index) -> image tensor; image layout
    per chunkserver/chunk.cc (scrub.py geometry), CRC array filled by the
    GPU crc32_blocks kernel, signature per chunk_signature.cc:87-90.
    """
    rs = ReedSolomon(k, m, device=device)
    slice_type = st.ec_slice_type(k, m)
    rec = rs.recover_batch(fragments, erased=erased, want=want)

    out = {}
    hdr = scrub.header_size(slice_type)
    for part, data in rec.items():
        S, plen = data.shape
        if plen % st.BLOCK_SIZE:
            raise ValueError("part length must be whole 64 KiB blocks")
        nblocks = plen // st.BLOCK_SIZE
        # per-block CRCs for the whole batch at once, then byte-swap to the
        # on-disk big-endian layout (put32bit convention)
        crcs = lcrc.crc32_blocks(data.reshape(-1), st.BLOCK_SIZE)
        be = crcs.view(torch.uint8).reshape(S, nblocks, 4).flip(dims=(2,))
        for s in range(S):
            img = torch.zeros(hdr + plen, dtype=torch.uint8,
                              device=data.device)
            sig = scrub.build_signature(int(chunk_ids[s]), version,
                                        slice_type, part)
            img[:len(sig)] = torch.from_numpy(
                np.frombuffer(sig, np.uint8).copy()).to(data.device)
            img[scrub.SIGNATURE_BLOCK:
                scrub.SIGNATURE_BLOCK + 4 * nblocks] = be[s].reshape(-1)
            img[hdr:] = data[s]
            out[(s, part)] = img
    return out


def replicate_stream(k, m, host_parts, erased, want, chunk_ids, version,
                     out=None, device=0, sub_batch=0):
    """The full ChunkReplicator::replicate pipeline
    (chunk_replicator.cc:139-196), host-to-host and streaming: surviving
    parts in HOST memory -> H2D -> recover -> per-block CRC -> MooseFS part
    image assembly on-device -> D2H, double-buffered (lizec_replicate_run).

    host_parts: list of k+m entries; entry i is a numpy uint8 [S, L] array
      (surviving part bytes, L a multiple of 64 KiB) or None (erased).
      Use lib.pinned_empty for true copy/compute overlap.
    erased: exactly m part indices; want: subset to rebuild.
    out: optional dict part -> uint8 [S, hdr+L] array to fill (pinned
      recommended); allocated (pinned) if absent.
    Returns dict part -> [S, hdr+L] images (signature + BE CRC array +
    recovered blocks), pwrite-ready.
    """
    nparts = k + m
    if len(host_parts) != nparts:
        raise ValueError(f"need {nparts} part slots")
    erased = sorted(set(erased))
    if len(erased) != m:
        raise ValueError(f"exactly m={m} erased parts required")
    want = sorted(set(want)) if want is not None else erased
    if not set(want).issubset(erased):
        raise ValueError("want must be a subset of erased")
    surv = [i for i in range(nparts) if i not in erased]
    S = plen = None
    for i in surv:
        if host_parts[i] is None:
            raise ValueError(f"surviving part {i} is None (streaming path "
                             f"takes explicit zero buffers)")
        a = host_parts[i]
        if a.dtype != np.uint8 or a.ndim != 2:
            raise ValueError("parts must be uint8 [S, L]")
        if S is None:
            S, plen = a.shape
        elif a.shape != (S, plen):
            raise ValueError("part shapes differ")
    if plen % st.BLOCK_SIZE:
        raise ValueError("part length must be whole 64 KiB blocks")
    if len(chunk_ids) != S:
        raise ValueError("need one chunk id per stripe")

    slice_type = st.ec_slice_type(k, m)
    hdr = scrub.header_size(slice_type)
    present = sum(1 << i for i in surv)
    needed = sum(1 << i for i in want)
    lib = L.lib()
    tbl = np.zeros(32 * 32 * 32, np.uint8)
    ic = ctypes.c_int()
    oc = ctypes.c_int()
    L.check(lib.lizec_rs_tables(
        k, m, present, present, needed,
        tbl.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        ctypes.byref(ic), ctypes.byref(oc)), "lizec_rs_tables")
    ic, oc = ic.value, oc.value
    assert (ic, oc) == (len(surv), len(want))

    if out is None:
        out = {p: L.pinned_empty((S, hdr + plen)) for p in want}
    for p in want:
        if out[p].dtype != np.uint8 or out[p].shape != (S, hdr + plen):
            raise ValueError(f"out[{p}] must be uint8 [S={S}, {hdr + plen}]")

    src = np.empty((S, ic), np.uint64)
    for j, i in enumerate(surv):
        a = host_parts[i]
        src[:, j] = a.ctypes.data + np.arange(S, dtype=np.uint64) * \
            np.uint64(a.strides[0])
    dst = np.empty((S, oc), np.uint64)
    sigs = np.zeros((S, oc, 22), np.uint8)
    for j, p in enumerate(want):
        a = out[p]
        dst[:, j] = a.ctypes.data + np.arange(S, dtype=np.uint64) * \
            np.uint64(a.strides[0])
        for s in range(S):
            sig = scrub.build_signature(int(chunk_ids[s]), version,
                                        slice_type, p)
            sigs[s, j, :len(sig)] = np.frombuffer(sig, np.uint8)

    L.check(lib.lizec_replicate_run(
        L.engine(device), plen, ic, oc,
        tbl.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        np.ascontiguousarray(src.ravel()).ctypes.data_as(
            ctypes.POINTER(ctypes.c_uint64)),
        sigs.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
        22, hdr, scrub.SIGNATURE_BLOCK,
        np.ascontiguousarray(dst.ravel()).ctypes.data_as(
            ctypes.POINTER(ctypes.c_uint64)),
        S, sub_batch), "lizec_replicate_run")
    return out
