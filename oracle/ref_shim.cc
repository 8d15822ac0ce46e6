/* oracle/ref_shim.cc — extern "C" surface over the REFERENCE's own EC code,
 * compiled unmodified from /root/reference (see oracle/Makefile; objects land
 * in oracle/_ref/ only).  TEST INFRASTRUCTURE: used to pin the oracle and to
 * generate golden vectors; never part of the product.
 *
 * Exposes: ReedSolomon<32,32>::encode/recover (reference
 * src/common/reed_solomon.h:41-373) and the galois_field.h:35-88 +
 * crc.h:25-31 C surfaces (those are plain functions in the reference
 * objects already; redeclared here for clarity).
 */
#include <cstdint>
#include <cstddef>
#include <stdexcept>  // reed_solomon.h:250 names std::runtime_error without including it

#include "common/reed_solomon.h"
#include "common/crc.h"

typedef ReedSolomon<32, 32> RS;

extern "C" {

/* encode: data/parity are arrays of k (resp. m) pointers; NULL data = zeros. */
int ref_rs_encode(int k, int m, const uint8_t **data, uint8_t **parity,
                  size_t size) {
	RS rs(k, m);
	RS::ConstFragmentMap data_parts{{0}};
	RS::FragmentMap parity_parts{{0}};
	for (int i = 0; i < k; ++i) data_parts[i] = data[i];
	for (int i = 0; i < m; ++i) parity_parts[i] = parity[i];
	rs.encode(data_parts, parity_parts, size);
	return 0;
}

/* recover: fragments/outputs are arrays of k+m pointers; erased_mask must
 * have exactly m bits set (reed_solomon.h:95). */
int ref_rs_recover(int k, int m, const uint8_t **fragments,
                   uint64_t erased_mask, uint8_t **outputs, size_t size) {
	RS rs(k, m);
	RS::ConstFragmentMap in{{0}};
	RS::FragmentMap out{{0}};
	RS::ErasedMap erased;
	for (int i = 0; i < k + m; ++i) {
		in[i] = fragments[i];
		out[i] = outputs[i];
		if ((erased_mask >> i) & 1) erased.set(i);
	}
	rs.recover(in, erased, out, size);
	return 0;
}

/* CRC surface (C++-mangled in the reference; re-exported as C).
 * Reference: crc.h:25-31, crc.cc:68-229. */
uint32_t ref_mycrc32(uint32_t crc, const uint8_t *block, uint32_t leng) {
	return mycrc32(crc, block, leng);
}

/* Per-block CRC loop in C so threaded bench callers are not bound by the
 * Python GIL between 64 KiB calls (one foreign call covers many blocks). */
void ref_mycrc32_blocks(const uint8_t *buf, uint64_t nblocks,
                        uint32_t block_len, uint32_t *out) {
	for (uint64_t b = 0; b < nblocks; ++b)
		out[b] = mycrc32(0, buf + b * (uint64_t)block_len, block_len);
}
uint32_t ref_mycrc32_combine(uint32_t crc1, uint32_t crc2, uint32_t leng2) {
	return mycrc32_combine(crc1, crc2, leng2);
}
void ref_mycrc32_init(void) {
	mycrc32_init();
}

}  /* extern "C" */
