"""Chunk-format scrub — MooseFS part-file layout and GPU hdd_int_test.

Format citations: chunk.cc:74-77 (maxBlocksInFile), getHeaderSize (1 KiB
signature + CRC array, 4 KiB-rounded for EC parts), chunk_signature.cc:30
("LIZC 1.1"), :87-90 (BE serialization); scrub semantics
hddspacemgr.cc:2148-2212 (first CRC-mismatching block damages the chunk).
"""
import numpy as np
import pytest

import oracle
from lizardfs_amd import scrub
from lizardfs_amd import slice_traits as st


def test_header_geometry():
    # standard: 1024 + 4*1024 = MFSHDRSIZE (MFSCommunication.h:69)
    assert scrub.header_size(st.K_STANDARD) == 5120
    # ec(8,2): maxBlocks = 128 -> 1024 + 512 = 1536 -> round to 4096
    t = st.ec_slice_type(8, 2)
    assert scrub.max_blocks_in_file(t) == 128
    assert scrub.header_size(t) == 4096
    # ec(3,1): ceil(1024/3) = 342 -> 1024 + 1368 = 2392 -> 4096
    t = st.ec_slice_type(3, 1)
    assert scrub.max_blocks_in_file(t) == 342
    assert scrub.header_size(t) == 4096
    # xor2: 512 blocks -> 1024 + 2048 = 3072 -> 4096
    assert scrub.header_size(st.K_XOR2) == 4096


def test_signature_roundtrip():
    sig = scrub.build_signature(0x1122334455667788, 7, st.ec_slice_type(8, 2), 3)
    assert len(sig) == 22
    parsed = scrub.parse_signature(np.frombuffer(sig, np.uint8))
    assert parsed == (0x1122334455667788, 7, st.ec_slice_type(8, 2), 3)
    assert scrub.parse_signature(np.zeros(1024, np.uint8)) is None


def _make_image(slice_type, part, nblocks, rng, chunk_id=42, version=1):
    blocks = [rng.integers(0, 256, st.BLOCK_SIZE, np.uint8)
              for _ in range(nblocks)]
    img = scrub.build_chunk_image(
        chunk_id, version, slice_type, part, blocks,
        crc32_fn=lambda b: oracle.crc32(b))
    return img


def test_build_image_crc_matches_oracle():
    rng = np.random.default_rng(5)
    t = st.ec_slice_type(8, 2)
    img = _make_image(t, 0, 3, rng)
    hdr = scrub.header_size(t)
    for b in range(3):
        stored = int.from_bytes(
            img[1024 + 4 * b:1024 + 4 * b + 4].tobytes(), "big")
        blk = img[hdr + b * 65536:hdr + (b + 1) * 65536]
        assert stored == oracle.crc32(blk.tobytes())


@pytest.mark.gpu
def test_scrub_interleaved_gpu():
    """INTERLEAVED on-disk format (chunk.h:40 kHddBlockSize = 65540: 4-byte
    BE CRC inline before each 64 KiB block, no signature block)."""
    import torch
    rng = np.random.default_rng(8)
    t = st.ec_slice_type(8, 2)

    def make(nblocks):
        blocks = [rng.integers(0, 256, st.BLOCK_SIZE, np.uint8)
                  for _ in range(nblocks)]
        return scrub.build_chunk_image(7, 1, t, 0, blocks,
                                       crc32_fn=lambda b: oracle.crc32(b),
                                       fmt="interleaved")

    clean = make(3)
    bad = make(4)
    bad[2 * 65540 + 4 + 100] ^= 0x10   # corrupt block 2's data
    badcrc = make(2)
    badcrc[1 * 65540 + 3] ^= 0x01      # corrupt block 1's stored CRC
    batch = [(torch.from_numpy(x).cuda(), t, "interleaved")
             for x in (clean, bad, badcrc)]
    # mix formats in one call
    moose = scrub.build_chunk_image(9, 1, t, 1,
                                    [rng.integers(0, 256, st.BLOCK_SIZE,
                                                  np.uint8)],
                                    crc32_fn=lambda b: oracle.crc32(b))
    batch.append((torch.from_numpy(moose).cuda(), t))
    res = scrub.scrub_batch(batch)
    assert res == [None, 2, 1, None], res


@pytest.mark.gpu
def test_scrub_batch_gpu():
    import torch
    rng = np.random.default_rng(6)
    t = st.ec_slice_type(8, 2)

    imgs = []
    # clean image
    imgs.append((_make_image(t, 0, 4, rng), t, None))
    # corrupted data byte in block 2
    bad = _make_image(t, 1, 4, rng)
    hdr = scrub.header_size(t)
    bad[hdr + 2 * 65536 + 1234] ^= 0xFF
    imgs.append((bad, t, 2))
    # corrupted CRC array entry for block 0
    bad2 = _make_image(t, 2, 3, rng)
    bad2[1024 + 1] ^= 0x01
    imgs.append((bad2, t, 0))
    # corruption in blocks 1 AND 3 -> first damaged = 1
    bad3 = _make_image(t, 3, 5, rng)
    bad3[hdr + 1 * 65536] ^= 0x80
    bad3[hdr + 3 * 65536 + 65535] ^= 0x01
    imgs.append((bad3, t, 1))
    # invalid signature
    badsig = _make_image(t, 4, 2, rng)
    badsig[0] = 0
    imgs.append((badsig, t, -2))
    # corruption INSIDE the header padding only -> still clean blocks
    pad = _make_image(t, 5, 2, rng)
    pad[1024 + 4 * scrub.max_blocks_in_file(t) + 8] = 0xEE
    imgs.append((pad, t, None))

    batch = [(torch.from_numpy(i).cuda(), ty) for (i, ty, _) in imgs]
    res = scrub.scrub_batch(batch)
    exp = [e for (_, _, e) in imgs]
    assert res == exp, (res, exp)
