"""End-to-end replication rebuild on GPU: recover erased parts, CRC them,
assemble MooseFS part images — then prove the images are byte-correct by
(a) comparing recovered data with the originals, (b) passing the scrub
(which re-CRCs every block against the stored array), and (c) failing the
scrub after deliberate corruption.  Mirrors the compute pipeline of
chunk_replicator.cc:139-196 + test_crc_error_fixing.sh's intent."""
import numpy as np
import pytest

from lizardfs_amd import slice_traits as st

pytestmark = pytest.mark.gpu


def test_rebuild_part_images_roundtrip():
    import torch
    from lizardfs_amd import scrub
    from lizardfs_amd.ec import ReedSolomon
    from lizardfs_amd.replicate import rebuild_part_images

    k, m, S = 8, 2, 3
    plen = 2 * st.BLOCK_SIZE  # 2 blocks per part
    rng = np.random.default_rng(99)
    data_np = rng.integers(0, 256, (S, k, plen), np.uint8)
    data = torch.from_numpy(data_np).cuda()
    rs = ReedSolomon(k, m)
    parity = rs.encode_batch(data)
    rs.sync()

    erased = (3, 6)
    frags = [None if i in erased else
             (data[:, i, :] if i < k else parity[:, i - k, :])
             for i in range(k + m)]
    chunk_ids = [1000 + s for s in range(S)]
    images = rebuild_part_images(k, m, frags, erased, set(erased),
                                 chunk_ids, version=5)

    t = st.ec_slice_type(k, m)
    hdr = scrub.header_size(t)
    for (s, part), img in sorted(images.items()):
        # recovered data bytes are the originals
        got = img[hdr:].cpu().numpy()
        assert np.array_equal(got, data_np[s, part]), (s, part)
        # signature parses back
        parsed = scrub.parse_signature(img[:1024].cpu().numpy())
        assert parsed == (1000 + s, 5, t, part)

    # every rebuilt image passes the scrub...
    batch = [(img, t) for (_, img) in sorted(images.items())]
    assert scrub.scrub_batch(batch) == [None] * len(batch)

    # ...and a corrupted one is caught at the right block
    (s0, p0), img0 = sorted(images.items())[0]
    img0[hdr + st.BLOCK_SIZE + 7] ^= 0x55   # corrupt block 1
    assert scrub.scrub_batch([(img0, t)]) == [1]


def test_replicate_stream_matches_reference_images():
    """The streaming host-to-host pipeline (lizec_replicate_run) emits
    images byte-identical to the host-assembled reference
    (build_chunk_image) and to the device-resident path."""
    import torch
    from lizardfs_amd import lib as L
    from lizardfs_amd import scrub
    from lizardfs_amd.ec import ReedSolomon
    from lizardfs_amd.replicate import replicate_stream

    k, m, S = 8, 2, 5
    plen = 4 * st.BLOCK_SIZE
    rng = np.random.default_rng(7)
    data_np = rng.integers(0, 256, (S, k, plen), np.uint8)
    data = torch.from_numpy(data_np).cuda()
    rs = ReedSolomon(k, m)
    parity = rs.encode_batch(data)
    rs.sync()
    parity_np = parity.cpu().numpy()

    erased = (1, 4)
    host_parts = [None if i in erased else
                  (np.ascontiguousarray(data_np[:, i, :]) if i < k
                   else np.ascontiguousarray(parity_np[:, i - k, :]))
                  for i in range(k + m)]
    chunk_ids = [2000 + s for s in range(S)]
    # pinned inputs for the overlap path on part 1's fragments
    pinned = L.pinned_empty((S, plen))
    pinned[:] = host_parts[0]
    host_parts[0] = pinned

    out = replicate_stream(k, m, host_parts, erased, erased, chunk_ids,
                           version=9, sub_batch=2)

    t = st.ec_slice_type(k, m)
    for p in erased:
        for s in range(S):
            ref_img = scrub.build_chunk_image(
                chunk_ids[s], 9, t, p,
                [data_np[s, p, b * st.BLOCK_SIZE:(b + 1) * st.BLOCK_SIZE]
                 for b in range(plen // st.BLOCK_SIZE)])
            got = out[p][s]
            assert got.shape == ref_img.shape
            assert np.array_equal(got, ref_img), (p, s)

    # the streamed images also pass the GPU scrub
    batch = [(torch.from_numpy(np.ascontiguousarray(out[p][s])).cuda(), t)
             for p in erased for s in range(S)]
    assert scrub.scrub_batch(batch) == [None] * len(batch)
