/* lizec_demo.cc — standalone C++ host over the C ABI only (no Python, no
 * torch): the chunkserver-shaped end-to-end flow on one MI355X.
 *
 *   1. build ec(8,2) encode tables (host matrix algebra)
 *   2. encode a batch of device-resident 64 MiB stripes
 *   3. CRC32 every 64 KiB block of the parity (the hdd_write gate)
 *   4. erase two data parts, rebuild them, verify bit-exactness
 *
 * Build: make -C lizardfs_amd/csrc demo   (hipcc; links liblizec.so)
 * Run:   ./tools/lizec_demo [stripes]     (GPU box only)
 */
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <cstring>
#include <vector>

#include "../include/lizec.h"

#define CK(x)                                                            \
	do {                                                                 \
		hipError_t e_ = (x);                                             \
		if (e_ != hipSuccess) {                                          \
			fprintf(stderr, "HIP error %s @%d\n", hipGetErrorString(e_), \
			        __LINE__);                                           \
			return 1;                                                    \
		}                                                                \
	} while (0)
#define CKL(x)                                                           \
	do {                                                                 \
		int r_ = (x);                                                    \
		if (r_ != LIZEC_OK) {                                            \
			fprintf(stderr, "lizec error %d @%d\n", r_, __LINE__);       \
			return 1;                                                    \
		}                                                                \
	} while (0)

int main(int argc, char **argv) {
	const int k = 8, m = 2;
	const uint64_t part_len = 8u * 1024 * 1024;   /* 64 MiB stripes */
	int stripes = argc > 1 ? atoi(argv[1]) : 64;

	if (lizec_gpu_count() < 1) {
		fprintf(stderr, "no MI355X visible\n");
		return 2;
	}
	lizec_engine *eng = nullptr;
	CKL(lizec_engine_create(&eng, 0));

	/* device-resident batch: [stripe][part][part_len] */
	uint8_t *d_data, *d_par, *d_rec;
	uint32_t *d_crcs;
	size_t data_bytes = (size_t)stripes * k * part_len;
	CK(hipMalloc(&d_data, data_bytes));
	CK(hipMalloc(&d_par, (size_t)stripes * m * part_len));
	CK(hipMalloc(&d_rec, (size_t)stripes * 2 * part_len));
	uint64_t nblocks = (size_t)stripes * m * part_len / 65536;
	CK(hipMalloc(&d_crcs, nblocks * 4));

	/* deterministic fill (host pattern, uploaded once) */
	std::vector<uint8_t> h(part_len);
	for (size_t i = 0; i < part_len; ++i) h[i] = (uint8_t)(i * 131 + i / 251);
	for (int s = 0; s < stripes; ++s)
		for (int j = 0; j < k; ++j) {
			h[0] = (uint8_t)(s * 31 + j);
			CK(hipMemcpy(d_data + ((size_t)s * k + j) * part_len, h.data(),
			             part_len, hipMemcpyHostToDevice));
		}

	/* 1. host matrix algebra (reed_solomon.h semantics) */
	std::vector<uint8_t> tbls(32 * k * m);
	CKL(lizec_rs_encode_tables(k, m, tbls.data()));

	/* 2. batched encode */
	std::vector<uint64_t> sp(stripes * k), pp(stripes * m);
	for (int s = 0; s < stripes; ++s) {
		for (int j = 0; j < k; ++j)
			sp[s * k + j] = (uint64_t)(d_data + ((size_t)s * k + j) * part_len);
		for (int l = 0; l < m; ++l)
			pp[s * m + l] = (uint64_t)(d_par + ((size_t)s * m + l) * part_len);
	}
	CKL(lizec_ec_encode_batch(eng, part_len, k, m, tbls.data(), sp.data(),
	                          pp.data(), stripes, nullptr));

	/* 3. the CRC gate over the parity */
	CKL(lizec_crc32_batch(eng, d_par, 65536, nblocks, 0, d_crcs, nullptr));

	/* 4. erase data parts 1 and 5, rebuild from the rest */
	uint64_t present = 0, needed = (1ull << 1) | (1ull << 5);
	for (int i = 0; i < k + m; ++i)
		if (i != 1 && i != 5) present |= 1ull << i;
	int ic, oc;
	std::vector<uint8_t> rtbl(32 * 32 * 32);
	CKL(lizec_rs_tables(k, m, present, present, needed, rtbl.data(), &ic, &oc));
	std::vector<uint64_t> rsp((size_t)stripes * ic), rdp((size_t)stripes * oc);
	for (int s = 0; s < stripes; ++s) {
		int w = 0;
		for (int i = 0; i < k + m; ++i) {
			if (i == 1 || i == 5) continue;
			rsp[(size_t)s * ic + w++] =
			    i < k ? sp[s * k + i] : pp[s * m + (i - k)];
		}
		rdp[(size_t)s * oc + 0] = (uint64_t)(d_rec + (size_t)s * 2 * part_len);
		rdp[(size_t)s * oc + 1] =
		    (uint64_t)(d_rec + (size_t)s * 2 * part_len + part_len);
	}
	CKL(lizec_ec_encode_batch(eng, part_len, ic, oc, rtbl.data(), rsp.data(),
	                          rdp.data(), stripes, nullptr));
	CKL(lizec_engine_sync(eng));

	/* verify: rebuilt parts == originals; parity CRCs == host CRCs */
	std::vector<uint8_t> a(part_len), b(part_len);
	for (int s = 0; s < stripes; s += stripes > 4 ? stripes / 4 : 1) {
		int wi = 0;
		for (int i : {1, 5}) {
			CK(hipMemcpy(a.data(), (void *)sp[s * k + i], part_len,
			             hipMemcpyDeviceToHost));
			CK(hipMemcpy(b.data(), (void *)rdp[(size_t)s * oc + wi++],
			             part_len, hipMemcpyDeviceToHost));
			if (memcmp(a.data(), b.data(), part_len) != 0) {
				fprintf(stderr, "MISMATCH stripe %d part %d\n", s, i);
				return 1;
			}
		}
	}
	std::vector<uint32_t> crcs(nblocks);
	CK(hipMemcpy(crcs.data(), d_crcs, nblocks * 4, hipMemcpyDeviceToHost));
	std::vector<uint8_t> pbuf(65536);
	for (uint64_t bI : {(uint64_t)0, nblocks / 2, nblocks - 1}) {
		CK(hipMemcpy(pbuf.data(), d_par + bI * 65536, 65536,
		             hipMemcpyDeviceToHost));
		if (crcs[bI] != lizec_crc32(0, pbuf.data(), 65536)) {
			fprintf(stderr, "CRC MISMATCH block %lu\n", (unsigned long)bI);
			return 1;
		}
	}
	printf("lizec_demo OK: %d stripes ec(%d,%d) encoded, parity CRC'd, "
	       "2-erasure rebuild bit-exact\n", stripes, k, m);
	lizec_engine_destroy(eng);
	return 0;
}
