"""Definitive drop-in proof: compile a probe against the REFERENCE'S OWN
headers (common/galois_field.h + common/crc.h + common/block_xor.h, which
declare the EC backend with C++ linkage) and link it against liblizec.so —
no reference sources, no oracle.  The probe encodes, inverts, CRCs and
XORs through the reference's declarations; outputs are checked against the
pinned oracle.

Runs only where the reference tree is present (this container); the GPU
box never needs it (the symbols themselves are asserted everywhere by
test_host.py::test_mangled_crc_aliases_present + nm checks).
"""
import os
import subprocess

import numpy as np
import pytest

import oracle

REF = "/root/reference"
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

pytestmark = pytest.mark.skipif(
    not os.path.isdir(os.path.join(REF, "src", "common")),
    reason="reference tree not present (GPU box)")

PROBE = r"""
#include <cstdio>
#include <cstdint>
#include <cstring>
#include <stdexcept>
#include "common/galois_field.h"
#include "common/crc.h"
#include "common/block_xor.h"

int main() {
	const int k = 6, m = 3, n = 256;
	uint8_t mat[(6 + 3) * 6];
	gf_gen_rs_matrix(mat, k + m, k);
	for (int i = 0; i < (k + m) * k; ++i) printf("%02x", mat[i]);
	printf("\n");
	gf_gen_cauchy1_matrix(mat, k + m, k);
	for (int i = 0; i < (k + m) * k; ++i) printf("%02x", mat[i]);
	printf("\n");

	/* encode via the C++-linked surface */
	uint8_t coeff[6 * 3];
	gf_gen_rs_matrix(mat, k + m, k);
	memcpy(coeff, mat + k * k, k * m);
	uint8_t tbls[32 * 6 * 3];
	ec_init_tables(k, m, coeff, tbls);
	uint8_t data[6][256], par[3][256];
	for (int j = 0; j < k; ++j)
		for (int i = 0; i < n; ++i) data[j][i] = (uint8_t)(j * 37 + i * 11);
	uint8_t *sp[6], *dp[3];
	for (int j = 0; j < k; ++j) sp[j] = data[j];
	for (int l = 0; l < m; ++l) dp[l] = par[l];
	ec_encode_data(n, k, m, tbls, sp, dp);
	for (int l = 0; l < m; ++l)
		for (int i = 0; i < n; ++i) printf("%02x", par[l][i]);
	printf("\n");

	mycrc32_init();
	printf("%08x\n", mycrc32(0, (const uint8_t *)"a", 1));
	uint8_t xa[64], xb[64];
	for (int i = 0; i < 64; ++i) { xa[i] = (uint8_t)i; xb[i] = (uint8_t)(i * 3); }
	blockXor(xa, xb, 64);
	for (int i = 0; i < 64; ++i) printf("%02x", xa[i]);
	printf("\n");
	return 0;
}
"""


RS_PROBE = r"""
#include <cstdio>
#include <cstdint>
#include <cstring>
#include <stdexcept>
/* ISA-L mode: reed_solomon.h:27-31 pulls isa-l/erasure_code.h — served by
 * liblizec's shim (include/isa-l/erasure_code.h). */
#define LIZARDFS_HAVE_ISA_L_ERASURE_CODE_H 1
#include "common/reed_solomon.h"

int main() {
	const int k = 8, m = 2, n = 512;
	typedef ReedSolomon<32, 32> RS;
	RS rs(k, m);
	uint8_t data[8][512], par[2][512], rec[2][512];
	for (int j = 0; j < k; ++j)
		for (int i = 0; i < n; ++i) data[j][i] = (uint8_t)(j * 53 + i * 7);

	RS::ConstFragmentMap dmap{{0}};
	RS::FragmentMap pmap{{0}};
	for (int j = 0; j < k; ++j) dmap[j] = data[j];
	for (int l = 0; l < m; ++l) pmap[l] = par[l];
	rs.encode(dmap, pmap, n);
	for (int l = 0; l < m; ++l)
		for (int i = 0; i < n; ++i) printf("%02x", par[l][i]);
	printf("\n");

	/* erase data parts 1 and 5, recover through the reference's recover() */
	RS::ConstFragmentMap in{{0}};
	RS::FragmentMap out{{0}};
	RS::ErasedMap erased;
	for (int i = 0; i < k + m; ++i) {
		if (i == 1 || i == 5) { erased.set(i); continue; }
		in[i] = i < k ? data[i] : par[i - k];
	}
	out[1] = rec[0];
	out[5] = rec[1];
	rs.recover(in, erased, out, n);
	int ok1 = memcmp(rec[0], data[1], n) == 0;
	int ok2 = memcmp(rec[1], data[5], n) == 0;
	printf("%d %d\n", ok1, ok2);
	return 0;
}
"""


def test_reference_reed_solomon_runs_on_liblizec(tmp_path):
    """Compile the REFERENCE'S ReedSolomon<32,32> (reed_solomon.h) in its
    ISA-L configuration against liblizec's shim + exports, and run the
    reference's own encode() and recover() on our backend."""
    src = tmp_path / "rs_probe.cc"
    src.write_text(RS_PROBE)
    exe = tmp_path / "rs_probe"
    lib = os.path.join(REPO, "lizardfs_amd", "liblizec.so")
    cmd = ["g++", "-O1", "-std=c++14", str(src), "-o", str(exe),
           f"-I{REF}/src", f"-I{REPO}/include",
           f"-I{REPO}/oracle/ref_config",
           lib, f"-Wl,-rpath,{os.path.dirname(lib)}"]
    r = subprocess.run(cmd, capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    out = subprocess.run([str(exe)], capture_output=True, text=True)
    assert out.returncode == 0, out.stderr
    lines = out.stdout.strip().splitlines()
    k, m, n = 8, 2, 512
    data = [np.array([(j * 53 + i * 7) & 0xFF for i in range(n)], np.uint8)
            for j in range(k)]
    exp = oracle.rs_encode(k, m, data, n)
    got = np.frombuffer(bytes.fromhex(lines[0]), np.uint8).reshape(m, n)
    for l in range(m):
        assert np.array_equal(got[l], exp[l]), f"parity {l}"
    assert lines[1] == "1 1"   # both erased parts recovered bit-exactly


def test_reference_headers_link_against_liblizec(tmp_path):
    src = tmp_path / "probe.cc"
    src.write_text(PROBE)
    exe = tmp_path / "probe"
    lib = os.path.join(REPO, "lizardfs_amd", "liblizec.so")
    cmd = ["g++", "-O1", "-std=c++14", str(src), "-o", str(exe),
           f"-I{REF}/src", f"-I{REPO}/oracle/ref_config",
           lib, f"-Wl,-rpath,{os.path.dirname(lib)}"]
    r = subprocess.run(cmd, capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    out = subprocess.run([str(exe)], capture_output=True, text=True)
    assert out.returncode == 0, out.stderr
    lines = out.stdout.strip().splitlines()
    assert len(lines) == 5

    k, m, n = 6, 3, 256
    rs = bytes.fromhex(lines[0])
    assert np.array_equal(np.frombuffer(rs, np.uint8).reshape(k + m, k),
                          oracle.gen_rs_matrix(k, m))
    cau = bytes.fromhex(lines[1])
    assert np.array_equal(np.frombuffer(cau, np.uint8).reshape(k + m, k),
                          oracle.gen_cauchy1_matrix(k, m))

    data = [np.array([(j * 37 + i * 11) & 0xFF for i in range(n)], np.uint8)
            for j in range(k)]
    exp = oracle.rs_encode(k, m, data, n)
    got = np.frombuffer(bytes.fromhex(lines[2]), np.uint8).reshape(m, n)
    for l in range(m):
        assert np.array_equal(got[l], exp[l]), f"parity {l}"

    assert lines[3] == "e8b7be43"   # crc_unittest.cc:30

    xa = np.array([i for i in range(64)], np.uint8)
    xb = np.array([(i * 3) & 0xFF for i in range(64)], np.uint8)
    assert np.array_equal(np.frombuffer(bytes.fromhex(lines[4]), np.uint8),
                          xa ^ xb)
