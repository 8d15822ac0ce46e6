"""GPU parity tests — the HIP kernels vs the pinned oracle and the
reference's golden vectors.  All tests here need a real MI355X and call
through the C ABI (lizardfs_amd -> liblizec.so); there is no CPU fallback
to accidentally pass on.
"""
import json
import os

import numpy as np
import pytest

import oracle

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu

HERE = os.path.dirname(os.path.abspath(__file__))
GOLDEN = np.load(os.path.join(HERE, "golden", "golden.npz"))
META = json.load(open(os.path.join(HERE, "golden", "golden_meta.json")))


@pytest.fixture(scope="module")
def rs_mod():
    from lizardfs_amd.ec import ReedSolomon
    return ReedSolomon


@pytest.fixture(scope="module")
def crc_mod():
    from lizardfs_amd import crc
    return crc


def to_gpu(a):
    return torch.from_numpy(np.ascontiguousarray(a)).cuda()


@pytest.mark.parametrize("case", META["cases"], ids=lambda c: c["name"])
def test_encode_matches_golden(rs_mod, case):
    """GPU encode == the reference's own output (golden fixtures)."""
    k, m, name, size = case["k"], case["m"], case["name"], case["size"]
    parts = []
    for i in range(k):
        p = GOLDEN[f"{name}_data{i}"]
        parts.append(np.zeros(size, np.uint8) if p.size == 0 else p)
    data = to_gpu(np.stack(parts)[None])  # [1, k, size]
    rs = rs_mod(k, m)
    parity = rs.encode_batch(data)
    rs.sync()
    got = parity.cpu().numpy()[0]
    for i in range(m):
        assert np.array_equal(got[i], GOLDEN[f"{name}_parity{i}"]), \
            f"{name}: parity {i}"


@pytest.mark.parametrize("case", META["cases"], ids=lambda c: c["name"])
def test_recover_matches_golden(rs_mod, case):
    k, m, name, size = case["k"], case["m"], case["name"], case["size"]
    mask = case["erased_mask"]
    want = set(case["wanted"])
    frags = []
    for i in range(k + m):
        if (mask >> i) & 1:
            frags.append(None)
        elif i < k:
            p = GOLDEN[f"{name}_data{i}"]
            frags.append(None if p.size == 0 else to_gpu(p)[None])
        else:
            frags.append(to_gpu(GOLDEN[f"{name}_parity{i - k}"])[None])
    rs = rs_mod(k, m)
    rec = rs.recover_batch(frags, erased={i for i in range(k + m)
                                          if (mask >> i) & 1}, want=want)
    rs.sync()
    for i in want:
        assert np.array_equal(rec[i].cpu().numpy()[0],
                              GOLDEN[f"{name}_rec{i}"]), f"{name}: part {i}"


@pytest.mark.parametrize("k,m,S,plen", [
    (8, 2, 7, 65536),       # headline shape, several stripes
    (8, 2, 3, 16),          # minimum part length
    (8, 2, 2, 16384 + 32),  # tile tail (16 KiB tile + ragged rest)
    (8, 2, 1, 4096 + 16),
    (3, 1, 5, 2048),
    (16, 4, 3, 8192),
    (32, 6, 2, 4096),       # two kernel launch groups (D=4+2)
    (2, 9, 2, 1024),        # wide parity (3 launch groups)
    (31, 3, 2, 1008),       # odd k, length not a tile divisor
])
def test_encode_vs_oracle_shapes(rs_mod, k, m, S, plen):
    rng = np.random.default_rng(1000 + k * 100 + m)
    data_np = rng.integers(0, 256, (S, k, plen), np.uint8)
    rs = rs_mod(k, m)
    parity = rs.encode_batch(to_gpu(data_np))
    rs.sync()
    got = parity.cpu().numpy()
    tbl, ic, oc = oracle.rs_make_tables(k, m, (1 << k) - 1, (1 << k) - 1,
                                        ((1 << m) - 1) << k)
    exp = np.zeros((S, m, plen), np.uint8)
    oracle.encode_stripes(k, m, plen, S, tbl,
                          np.ascontiguousarray(data_np), exp)
    assert np.array_equal(got, exp)


def test_roundtrip_full_size_stripes(rs_mod):
    """BASELINE full size: 64 MiB stripes, ec(8,2) — encode on GPU, erase 2
    data parts, decode on GPU, require the original bytes back (bit-exact,
    size-independent property), plus a direct oracle cross-check of the
    parity at full size."""
    k, m, S = 8, 2, 2
    plen = 64 * 1024 * 1024 // k  # 8 MiB parts -> 64 MiB stripes
    g = torch.Generator(device="cuda").manual_seed(42)
    data = torch.randint(0, 256, (S, k, plen), dtype=torch.uint8,
                         device="cuda", generator=g)
    rs = rs_mod(k, m)
    parity = rs.encode_batch(data)
    frags = [None if i in (1, 5) else
             (data[:, i].contiguous() if i < k else parity[:, i - k].contiguous())
             for i in range(k + m)]
    rec = rs.recover_batch(frags, erased=(1, 5))
    rs.sync()
    for i in (1, 5):
        assert torch.equal(rec[i], data[:, i]), f"part {i} round-trip"
    # full-size oracle cross-check on one stripe's parity
    data0 = data[0].cpu().numpy()
    exp = oracle.rs_encode(k, m, list(data0), plen)
    got = parity[0].cpu().numpy()
    for l in range(m):
        assert np.array_equal(got[l], exp[l])


def test_recover_from_strided_views_with_out(rs_mod):
    """Fragments may be strided views into the [S,k,L] batch (no copies),
    and caller-provided outputs enable the cached-plan path."""
    k, m, S, plen = 8, 2, 5, 32768
    rng = np.random.default_rng(2024)
    data_np = rng.integers(0, 256, (S, k, plen), np.uint8)
    data = to_gpu(data_np)
    rs = rs_mod(k, m)
    parity = rs.encode_batch(data)
    frags = [None if i in (1, 5) else
             (data[:, i, :] if i < k else parity[:, i - k, :])
             for i in range(k + m)]
    outs = {i: torch.empty((S, plen), dtype=torch.uint8, device="cuda")
            for i in (1, 5)}
    for _ in range(3):   # repeated runs exercise the plan cache
        rec = rs.recover_batch(frags, erased=(1, 5), out=outs)
    rs.sync()
    for i in (1, 5):
        assert np.array_equal(rec[i].cpu().numpy(), data_np[:, i, :])


def test_recover_with_null_zero_parts(rs_mod):
    """NULL surviving part = implicit zeros (reed_solomon.h:79) on GPU."""
    k, m, S, plen = 6, 2, 3, 4096
    rng = np.random.default_rng(77)
    data_np = rng.integers(0, 256, (S, k, plen), np.uint8)
    zero_idx = [0, 3]
    for i in zero_idx:
        data_np[:, i] = 0
    rs = rs_mod(k, m)
    parity = rs.encode_batch(to_gpu(data_np))
    rs.sync()
    # decode parts 2 and 4 from the rest, passing None for the zero parts
    erased = (2, 4)
    frags = []
    for i in range(k + m):
        if i in erased or i in zero_idx:
            frags.append(None)
        elif i < k:
            frags.append(to_gpu(data_np[:, i]))
        else:
            frags.append(parity[:, i - k].contiguous())
    rec = rs.recover_batch(frags, erased=erased)
    rs.sync()
    for i in erased:
        assert np.array_equal(rec[i].cpu().numpy(), data_np[:, i])


def test_crc_blocks_matches_golden_and_oracle(crc_mod):
    blk = GOLDEN["crc_block64k"]
    got = crc_mod.crc32_blocks(to_gpu(blk), 65536).cpu().numpy().view(np.uint32)
    ref = [c for c in META["crc"] if c["input"] == "block64k" and c["seed"] == 0]
    assert got[0] == ref[0]["crc"]

    rng = np.random.default_rng(88)
    buf = rng.integers(0, 256, 8 * 65536, np.uint8)
    got = crc_mod.crc32_blocks(to_gpu(buf), 65536).cpu().numpy().view(np.uint32)
    exp = oracle.crc32_blocks(buf, 65536)
    assert np.array_equal(got, exp)
    # other block sizes + seed: generic kernel (1 KiB, 2 KiB) and the
    # fast kernel off the 64 KiB shape (48 KiB)
    for blen in (1024, 2048, 49152):
        got = crc_mod.crc32_blocks(to_gpu(buf[:8 * 49152]), blen,
                                   seed=0xABCD1234) \
            .cpu().numpy().view(np.uint32)
        exp = oracle.crc32_blocks(buf[:8 * 49152], blen, seed=0xABCD1234)
        assert np.array_equal(got, exp), blen
    # all-zero blocks (sparse-chunk special case, crc.cc:235-243 context)
    z = np.zeros(4 * 65536, np.uint8)
    got = crc_mod.crc32_blocks(to_gpu(z), 65536).cpu().numpy().view(np.uint32)
    exp = oracle.crc32_blocks(z, 65536)
    assert np.array_equal(got, exp)


def test_native_library_is_loaded():
    """Guard against silent eager fallback: the loaded compute library must
    be the in-tree liblizec.so and a GPU must be visible to it."""
    from lizardfs_amd import lib as L
    import lizardfs_amd
    so = os.path.join(os.path.dirname(lizardfs_amd.__file__), "liblizec.so")
    assert os.path.exists(so)
    assert L.lib()._name == so
    assert L.lib().lizec_gpu_count() >= 1


def test_crc_fold_shapes_agree(crc_mod):
    """Every fold shape the autotuner or env hooks can pick (no-PF,
    burst-prefetch, BV=4 variants) is bit-identical to the pinned default
    on a batch large enough to trigger autotuning (>= 4096 blocks)."""
    import torch
    g = torch.Generator(device="cuda").manual_seed(5)
    buf = torch.randint(0, 256, (4096 * 65536,), dtype=torch.uint8,
                        device="cuda", generator=g)
    saved = {k: os.environ.get(k) for k in
             ("LIZEC_CRC_AUTOTUNE", "LIZEC_CRC_PF", "LIZEC_CRC_BV")}
    try:
        os.environ["LIZEC_CRC_AUTOTUNE"] = "0"
        os.environ["LIZEC_CRC_PF"] = "0"
        os.environ["LIZEC_CRC_BV"] = "0"
        ref = crc_mod.crc32_blocks(buf, 65536).clone()
        torch.cuda.synchronize()
        for env in ({"LIZEC_CRC_PF": "1"},
                    {"LIZEC_CRC_PF": "1", "LIZEC_CRC_BV": "4"},
                    {"LIZEC_CRC_BV": "4"},
                    {"LIZEC_CRC_AUTOTUNE": "1"}):
            os.environ.update(env)
            got = crc_mod.crc32_blocks(buf, 65536)
            torch.cuda.synchronize()
            assert torch.equal(got, ref), env
            os.environ["LIZEC_CRC_PF"] = "0"
            os.environ["LIZEC_CRC_BV"] = "0"
            os.environ["LIZEC_CRC_AUTOTUNE"] = "0"
    finally:
        for k, v in saved.items():
            if v is None:
                os.environ.pop(k, None)
            else:
                os.environ[k] = v
