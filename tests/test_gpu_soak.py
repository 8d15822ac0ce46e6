"""Engine stability under sustained reuse: repeated batches, plan-cache
eviction churn, mixed ops on one engine.  Guards against leaks/corruption
that single-shot parity tests cannot see."""
import numpy as np
import pytest

pytestmark = pytest.mark.gpu


def test_engine_soak_mixed_ops():
    import torch
    from lizardfs_amd.ec import ReedSolomon
    from lizardfs_amd import crc as lcrc

    rng = np.random.default_rng(123)
    k, m, S, plen = 8, 2, 16, 65536
    data_np = rng.integers(0, 256, (S, k, plen), np.uint8)
    data = torch.from_numpy(data_np).cuda()
    parity = torch.empty((S, m, plen), dtype=torch.uint8, device="cuda")
    rs = ReedSolomon(k, m)

    exp_parity = None
    exp_crc = None
    for it in range(50):
        rs.encode_batch(data, parity)
        crcs = lcrc.crc32_blocks(parity.reshape(-1), 65536)
        if it % 10 == 0:
            rs.sync()
            got = parity.cpu().numpy().copy()
            gotc = crcs.cpu().numpy().copy()
            if exp_parity is None:
                exp_parity, exp_crc = got, gotc
            else:
                assert np.array_equal(got, exp_parity), f"iteration {it}"
                assert np.array_equal(gotc, exp_crc), f"crc iteration {it}"
    # plan-cache churn: >64 distinct batches forces eviction of live plans
    small = torch.from_numpy(
        rng.integers(0, 256, (1, k, 1024), np.uint8)).cuda()
    for it in range(80):
        p = torch.empty((1, m, 1024), dtype=torch.uint8, device="cuda")
        rs.encode_batch(small, p)   # fresh parity => no plan cached
        out = torch.empty((1, m, 1024), dtype=torch.uint8, device="cuda")
        rs.encode_batch(small, out)  # caller buffer => plan cached+evicted
    rs.sync()
    assert len(rs._plans) <= 64
