"""Multi-threaded engine stress — mirrors the chunkserver's bgjobs
threading model (network_main_thread.cc:230-234: several HDD worker
threads issue EC/CRC work concurrently).  One engine per device, one
stream per thread; per-stream scratch in liblizec must keep concurrent
batch calls race-free (lizec_gpu.hip ctx_acquire)."""
import threading

import numpy as np
import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu

N_THREADS = 8
ITERS = 25


def test_concurrent_encode_streams():
    from lizardfs_amd.ec import ReedSolomon

    k, m, S, plen = 8, 2, 8, 64 * 1024
    rs = ReedSolomon(k, m)
    datas, expect = [], []
    g = torch.Generator(device="cuda").manual_seed(7)
    for t in range(N_THREADS):
        d = torch.randint(0, 256, (S, k, plen), dtype=torch.uint8,
                          device="cuda", generator=g)
        datas.append(d)
        p = rs.encode_batch(d)          # trusted single-stream pass
        rs.sync()
        expect.append(p.clone())

    errors = []
    barrier = threading.Barrier(N_THREADS)

    def worker(t):
        try:
            stream = torch.cuda.Stream()
            with torch.cuda.stream(stream):
                barrier.wait()
                for it in range(ITERS):
                    out = torch.empty_like(expect[t])
                    # fresh output buffer each iter -> no plan cache, the
                    # per-call upload path (the racy one before the fix)
                    # runs every time
                    rs.encode_batch(datas[t], parity=out)
                    stream.synchronize()
                    if not torch.equal(out, expect[t]):
                        errors.append((t, it, "encode mismatch"))
                        return
        except Exception as exc:  # pragma: no cover
            errors.append((t, None, repr(exc)))

    threads = [threading.Thread(target=worker, args=(t,))
               for t in range(N_THREADS)]
    for th in threads:
        th.start()
    for th in threads:
        th.join()
    assert not errors, errors


def test_concurrent_mixed_ec_crc():
    """EC encodes and CRC batches in flight on different streams of the
    same engine at once (the config-4 overlap shape, multi-threaded)."""
    import ctypes

    from lizardfs_amd import crc as crc_mod
    from lizardfs_amd.ec import ReedSolomon

    k, m, S, plen = 8, 2, 8, 64 * 1024
    rs = ReedSolomon(k, m)
    g = torch.Generator(device="cuda").manual_seed(11)
    data = torch.randint(0, 256, (S, k, plen), dtype=torch.uint8,
                         device="cuda", generator=g)
    expect_par = rs.encode_batch(data)
    rs.sync()
    flat = data.reshape(-1)
    expect_crc = crc_mod.crc32_blocks(flat, 65536)
    torch.cuda.synchronize()
    expect_par = expect_par.clone()
    expect_crc = expect_crc.clone()

    errors = []
    barrier = threading.Barrier(4)

    def enc_worker(t):
        try:
            stream = torch.cuda.Stream()
            with torch.cuda.stream(stream):
                barrier.wait()
                for it in range(ITERS):
                    out = torch.empty_like(expect_par)
                    rs.encode_batch(data, parity=out)
                    stream.synchronize()
                    if not torch.equal(out, expect_par):
                        errors.append((t, it, "encode mismatch"))
                        return
        except Exception as exc:  # pragma: no cover
            errors.append((t, None, repr(exc)))

    def crc_worker(t):
        try:
            stream = torch.cuda.Stream()
            with torch.cuda.stream(stream):
                barrier.wait()
                for it in range(ITERS):
                    got = crc_mod.crc32_blocks(flat, 65536)
                    stream.synchronize()
                    if not torch.equal(got, expect_crc):
                        errors.append((t, it, "crc mismatch"))
                        return
        except Exception as exc:  # pragma: no cover
            errors.append((t, None, repr(exc)))

    threads = [threading.Thread(target=enc_worker, args=(t,))
               for t in range(2)]
    threads += [threading.Thread(target=crc_worker, args=(t,))
                for t in range(2, 4)]
    for th in threads:
        th.start()
    for th in threads:
        th.join()
    assert not errors, errors
