"""Chunkserver-side EC part rebuild — the compute core of
ChunkReplicator::replicate (chunk_replicator.cc:139-196), batched on GPU.

The reference's replicate loop pulls surviving parts from peers, recovers
the missing part via SliceRecoveryPlanner + ReedSolomon, CRCs every
recovered 64 KiB block (chunk_replicator.cc:189), and writes a MooseFS
part file.  Networking/disk stay with the host; this module does the whole
compute pipeline on the GPU: recover -> per-block CRC -> assembled part
image (header + big-endian CRC array + blocks) ready to pwrite.
"""
import numpy as np
import torch

from . import crc as lcrc
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
