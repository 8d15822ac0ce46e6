/* lizec_host.cpp — host side of the MI355X EC engine: GF(2^8) matrix
 * algebra, ISA-L-shaped drop-in surface, CRC32, slice-type algebra.
 *
 * Semantics follow the reference (lizardfs/lizardfs) interfaces cited in
 * include/lizec.h; the implementation is our own (carry-less polynomial
 * multiply + Fermat inverse instead of log/exp tables, slicing-by-8 CRC).
 * Bit-exactness vs the reference is enforced by tests/ against the oracle
 * and the committed golden vectors.
 */
#include <cstdint>
#include <cstring>

#include "../../include/lizec.h"

/* ------------------------------------------------------------------ */
/* GF(2^8), polynomial 0x11D                                          */
/* ------------------------------------------------------------------ */

namespace {

/* Carry-less 8x8 multiply, then reduce mod x^8+x^4+x^3+x^2+1. */
inline uint8_t gfmul(uint8_t a, uint8_t b) {
	uint32_t r = 0;
	uint32_t aa = a;
	for (int i = 0; i < 8; ++i)
		if ((b >> i) & 1) r ^= aa << i;
	/* reduce bits 15..8 */
	for (int i = 15; i >= 8; --i)
		if ((r >> i) & 1) r ^= 0x11Du << (i - 8);
	return (uint8_t)r;
}

inline uint8_t gfinv(uint8_t a) {
	if (a == 0) return 0;
	/* a^254 = a^-1 in GF(256): square-and-multiply. */
	uint8_t r = 1, p = a;
	int e = 254;
	while (e) {
		if (e & 1) r = gfmul(r, p);
		p = gfmul(p, p);
		e >>= 1;
	}
	return r;
}

}  // namespace

extern "C" void gf_gen_rs_matrix(uint8_t *a, int m, int k) {
	memset(a, 0, (size_t)k * m);
	for (int i = 0; i < k; ++i) a[k * i + i] = 1;
	uint8_t gen = 1;
	for (int i = k; i < m; ++i) {
		uint8_t p = 1;
		for (int j = 0; j < k; ++j) {
			a[k * i + j] = p;
			p = gfmul(p, gen);
		}
		gen = gfmul(gen, 2);
	}
}

extern "C" void gf_gen_cauchy1_matrix(uint8_t *a, int m, int k) {
	memset(a, 0, (size_t)k * m);
	for (int i = 0; i < k; ++i) a[k * i + i] = 1;
	uint8_t *p = &a[k * k];
	for (int i = k; i < m; ++i)
		for (int j = 0; j < k; ++j)
			*p++ = gfinv((uint8_t)(i ^ j));
}

extern "C" int gf_invert_matrix(uint8_t *in_mat, uint8_t *out_mat, const int n) {
	for (int i = 0; i < n * n; ++i) out_mat[i] = 0;
	for (int i = 0; i < n; ++i) out_mat[i * n + i] = 1;

	for (int i = 0; i < n; ++i) {
		if (in_mat[i * n + i] == 0) {
			int j = i + 1;
			for (; j < n; ++j)
				if (in_mat[j * n + i]) break;
			if (j == n) return -1;
			for (int c = 0; c < n; ++c) {
				uint8_t t = in_mat[i * n + c];
				in_mat[i * n + c] = in_mat[j * n + c];
				in_mat[j * n + c] = t;
				t = out_mat[i * n + c];
				out_mat[i * n + c] = out_mat[j * n + c];
				out_mat[j * n + c] = t;
			}
		}
		uint8_t piv = gfinv(in_mat[i * n + i]);
		for (int c = 0; c < n; ++c) {
			in_mat[i * n + c] = gfmul(in_mat[i * n + c], piv);
			out_mat[i * n + c] = gfmul(out_mat[i * n + c], piv);
		}
		for (int r = 0; r < n; ++r) {
			if (r == i) continue;
			uint8_t f = in_mat[r * n + i];
			if (!f) continue;
			for (int c = 0; c < n; ++c) {
				in_mat[r * n + c] ^= gfmul(f, in_mat[i * n + c]);
				out_mat[r * n + c] ^= gfmul(f, out_mat[i * n + c]);
			}
		}
	}
	return 0;
}

extern "C" void ec_init_tables(int k, int rows, uint8_t *a, uint8_t *g_tbls) {
	int n = k * rows;   /* linear expansion; layout [row][col][32] */
	for (int i = 0; i < n; ++i) {
		uint8_t c = a[i];
		uint8_t *t = g_tbls + (size_t)i * 32;
		for (int v = 0; v < 16; ++v) {
			t[v] = gfmul(c, (uint8_t)v);
			t[16 + v] = gfmul(c, (uint8_t)(v << 4));
		}
	}
}

extern "C" void ec_encode_data(int len, int srcs, int dests, uint8_t *v,
                               uint8_t **src, uint8_t **dest) {
	for (int l = 0; l < dests; ++l) {
		uint8_t *vl = v + (size_t)l * srcs * 32;
		uint8_t *d = dest[l];
		for (int i = 0; i < len; ++i) {
			uint8_t s = 0;
			const uint8_t *tbl = vl;
			for (int j = 0; j < srcs; ++j) {
				uint8_t a = src[j][i];
				s ^= tbl[a & 0xF] ^ tbl[16 + (a >> 4)];
				tbl += 32;
			}
			d[i] = s;
		}
	}
}


/* distinct-name C entry points for the C++-mangled alias TU
 * (lizec_abi_aliases.cpp) — avoids linkage conflicts in one TU. */
extern "C" void lizec_impl_gen_rs_matrix(uint8_t *a, int m, int k) {
	gf_gen_rs_matrix(a, m, k);
}
extern "C" void lizec_impl_gen_cauchy1_matrix(uint8_t *a, int m, int k) {
	gf_gen_cauchy1_matrix(a, m, k);
}
extern "C" int lizec_impl_invert_matrix(uint8_t *in_mat, uint8_t *out_mat,
                                        int n) {
	return gf_invert_matrix(in_mat, out_mat, n);
}
extern "C" void lizec_impl_init_tables(int k, int rows, uint8_t *a,
                                       uint8_t *g_tbls) {
	ec_init_tables(k, rows, a, g_tbls);
}
extern "C" void lizec_impl_encode_data(int len, int srcs, int dests,
                                       uint8_t *v, uint8_t **src,
                                       uint8_t **dest) {
	ec_encode_data(len, srcs, dests, v, src, dest);
}

/* ------------------------------------------------------------------ */
/* ReedSolomon table builders (reed_solomon.h:41-373 semantics)       */
/* ------------------------------------------------------------------ */

namespace {

constexpr int MAXK = 32, MAXM = 32, MAXP = 64;

inline int popcount(uint64_t x) { return __builtin_popcountll(x); }

void select_rows(uint8_t *out, const uint8_t *in, int s1, int s2,
                 uint64_t rows) {
	for (int i = 0; i < s1; ++i, in += s2)
		if ((rows >> i) & 1) {
			memcpy(out, in, (size_t)s2);
			out += s2;
		}
}

void select_columns(uint8_t *out, const uint8_t *in, int s1, int s2,
                    uint64_t cols) {
	for (int i = 0; i < s1; ++i, in += s2)
		for (int j = 0; j < s2; ++j)
			if ((cols >> j) & 1) *out++ = in[j];
}

}  // namespace

extern "C" int lizec_rs_tables(int k, int m, uint64_t present_mask,
                               uint64_t nonnull_mask, uint64_t needed_mask,
                               uint8_t *gftbls, int *in_count, int *out_count) {
	if (k < 1 || k > MAXK || m < 1 || m > MAXM) return LIZEC_EINVAL;
	int nparts = k + m;
	uint64_t all = (nparts == 64) ? ~0ULL : (((uint64_t)1 << nparts) - 1);
	present_mask &= all;
	nonnull_mask &= present_mask;
	needed_mask &= all & ~present_mask;
	if (popcount(~present_mask & all) != m) return LIZEC_EINVAL;

	uint8_t rs_matrix[MAXP * MAXK];
	if (m >= 5 || (m == 4 && k > 20))
		gf_gen_cauchy1_matrix(rs_matrix, nparts, k);
	else
		gf_gen_rs_matrix(rs_matrix, nparts, k);

	/* non_zero_input indexed by surviving-part ORDER (reed_solomon.h:103-108) */
	uint64_t non_zero_input = 0;
	int in_with_zero = 0, data_present = 0;
	for (int i = 0; i < nparts; ++i) {
		if ((present_mask >> i) & 1) {
			if ((nonnull_mask >> i) & 1)
				non_zero_input |= (uint64_t)1 << in_with_zero;
			in_with_zero++;
			data_present += (i < k);
		}
	}
	int nz = popcount(non_zero_input);
	int needed_count = popcount(needed_mask);
	int parity_needed = popcount(needed_mask >> k);
	if (needed_count == 0 || nz == 0) return LIZEC_EINVAL;

	uint8_t work[MAXP * MAXK];
	uint8_t recover_m[MAXP * MAXK];

	if (data_present == k) {
		/* createEncodingMatrix (reed_solomon.h:189-217) */
		select_rows(recover_m, rs_matrix, nparts, k, needed_mask);
	} else {
		/* createRecoveryMatrix (reed_solomon.h:229-281) */
		uint8_t decode_m[MAXK * MAXK];
		select_rows(work, rs_matrix, nparts, k, present_mask);
		if (gf_invert_matrix(work, decode_m, k) != 0) return LIZEC_ESINGULAR;
		if (parity_needed > 0) {
			uint8_t sel[MAXP * MAXK];
			select_rows(sel, rs_matrix, nparts, k, needed_mask);
			/* recover = sel x decode  (matrixMultiply, reed_solomon.h:344) */
			for (int i = 0; i < needed_count; ++i)
				for (int c = 0; c < k; ++c) {
					uint8_t s = 0;
					for (int j = 0; j < k; ++j)
						s ^= gfmul(sel[i * k + j], decode_m[j * k + c]);
					recover_m[i * k + c] = s;
				}
		} else {
			select_rows(recover_m, decode_m, k, k, needed_mask);
		}
	}

	if (nz < k) {
		select_columns(work, recover_m, needed_count, k, non_zero_input);
		ec_init_tables(needed_count, nz, work, gftbls);
	} else {
		ec_init_tables(needed_count, k, recover_m, gftbls);
	}
	if (in_count) *in_count = nz;
	if (out_count) *out_count = needed_count;
	return LIZEC_OK;
}

extern "C" int lizec_rs_encode_tables(int k, int m, uint8_t *gftbls) {
	int ic, oc;
	uint64_t data = ((uint64_t)1 << k) - 1;
	uint64_t par = (((uint64_t)1 << m) - 1) << k;
	return lizec_rs_tables(k, m, data, data, par, gftbls, &ic, &oc);
}

/* ------------------------------------------------------------------ */
/* CRC32 (reflected, poly 0xEDB88320; crc.h:25-31 semantics)          */
/* ------------------------------------------------------------------ */

static uint32_t crc8tab[8][256];
static bool crc_ready = false;

extern "C" void lizec_crc32_init(void) {
	if (crc_ready) return;
	for (uint32_t i = 0; i < 256; ++i) {
		uint32_t c = i;
		for (int b = 0; b < 8; ++b)
			c = (c & 1) ? (0xEDB88320u ^ (c >> 1)) : (c >> 1);
		crc8tab[0][i] = c;
	}
	for (uint32_t i = 0; i < 256; ++i)
		for (int t = 1; t < 8; ++t)
			crc8tab[t][i] = crc8tab[0][crc8tab[t - 1][i] & 0xff] ^
			                (crc8tab[t - 1][i] >> 8);
	crc_ready = true;
}

extern "C" uint32_t lizec_crc32(uint32_t crc, const uint8_t *p, uint32_t len) {
	if (!crc_ready) lizec_crc32_init();
	crc ^= 0xFFFFFFFFu;
	while (len && ((uintptr_t)p & 7)) {
		crc = crc8tab[0][(crc ^ *p++) & 0xFF] ^ (crc >> 8);
		len--;
	}
	while (len >= 8) {
		uint64_t w;
		memcpy(&w, p, 8);
		w ^= crc;  /* little-endian host */
		crc = crc8tab[7][w & 0xff] ^ crc8tab[6][(w >> 8) & 0xff] ^
		      crc8tab[5][(w >> 16) & 0xff] ^ crc8tab[4][(w >> 24) & 0xff] ^
		      crc8tab[3][(w >> 32) & 0xff] ^ crc8tab[2][(w >> 40) & 0xff] ^
		      crc8tab[1][(w >> 48) & 0xff] ^ crc8tab[0][w >> 56];
		p += 8;
		len -= 8;
	}
	while (len) {
		crc = crc8tab[0][(crc ^ *p++) & 0xFF] ^ (crc >> 8);
		len--;
	}
	return crc ^ 0xFFFFFFFFu;
}

namespace {

uint32_t gf2_times(const uint32_t *mat, uint32_t vec) {
	uint32_t s = 0;
	for (int i = 0; vec; vec >>= 1, ++i)
		if (vec & 1) s ^= mat[i];
	return s;
}

void gf2_square(uint32_t *sq, const uint32_t *mat) {
	for (int i = 0; i < 32; ++i) sq[i] = gf2_times(mat, mat[i]);
}

}  // namespace

extern "C" uint32_t lizec_crc32_combine(uint32_t crc1, uint32_t crc2,
                                        uint32_t len2) {
	/* Matches the reference's table walk (crc.cc:207-224): advance crc1 by
	 * len2 zero bytes, xor crc2; len2==0 degenerates to crc1^crc2. */
	if (len2 == 0) return crc1 ^ crc2;
	uint32_t even[32], odd[32];
	odd[0] = 0xEDB88320u;
	for (int i = 1; i < 32; ++i) odd[i] = 1u << (i - 1);
	gf2_square(even, odd);
	gf2_square(odd, even);
	do {
		gf2_square(even, odd);
		if (len2 & 1) crc1 = gf2_times(even, crc1);
		len2 >>= 1;
		if (!len2) break;
		gf2_square(odd, even);
		if (len2 & 1) crc1 = gf2_times(odd, crc1);
		len2 >>= 1;
	} while (len2);
	return crc1 ^ crc2;
}

/* Partial-block CRC algebra — the crc.h:27-29 macros and hdd_write's
 * splice logic (hddspacemgr.cc:1952-2003) as first-class functions. */

/* crc.h:27: mycrc32_zeroblock — CRC after appending `zeros` zero bytes. */
extern "C" uint32_t lizec_crc32_zeroblock(uint32_t crc, uint32_t zeros) {
	return lizec_crc32_combine(crc ^ 0xFFFFFFFFu, 0xFFFFFFFFu, zeros);
}

/* crc.h:28: mycrc32_zeroexpanded — CRC of data followed by zeros. */
extern "C" uint32_t lizec_crc32_zeroexpanded(uint32_t crc, const uint8_t *block,
                                             uint32_t leng, uint32_t zeros) {
	return lizec_crc32_zeroblock(lizec_crc32(crc, block, leng), zeros);
}

/* crc.h:29: mycrc32_xorblocks — CRC of the byte-XOR of two equal-length
 * blocks from their CRCs (CRC is affine over GF(2)). */
extern "C" uint32_t lizec_crc32_xorblocks(uint32_t crc, uint32_t crcblock1,
                                          uint32_t crcblock2, uint32_t leng) {
	return crcblock1 ^ crcblock2 ^ lizec_crc32_zeroblock(crc, leng);
}

/* hdd_write's partial-write recombine (hddspacemgr.cc:1995-2003): CRC of a
 * block_len-byte block whose [offset, offset+size) range has CRC `crc`,
 * with precrc = CRC of [0, offset) and postcrc = CRC of the tail. */
extern "C" uint32_t lizec_crc32_splice(uint32_t precrc, uint32_t offset,
                                       uint32_t crc, uint32_t size,
                                       uint32_t postcrc, uint32_t block_len) {
	uint32_t combined;
	if (offset == 0) {
		return lizec_crc32_combine(crc, postcrc, block_len - size);
	}
	combined = lizec_crc32_combine(precrc, crc, size);
	if (offset + size < block_len)
		combined = lizec_crc32_combine(combined, postcrc,
		                               block_len - (offset + size));
	return combined;
}

/* crc.cc:235-243: sparse-file special case — a zero block whose cached
 * CRC is 0 gets the canonical empty-block CRC. */
extern "C" void lizec_recompute_crc_if_block_empty(const uint8_t *block,
                                                   uint32_t block_len,
                                                   uint32_t *crc) {
	if (*crc != 0) return;
	if (block[0] != 0 || memcmp(block, block + 1, block_len - 1) != 0) return;
	*crc = lizec_crc32_zeroblock(0, block_len);
}

/* C++-linkage aliases so a LizardFS build linking against common/crc.h's
 * mangled symbols (crc.h:25-31) resolves them from this library. */
uint32_t mycrc32(uint32_t crc, const uint8_t *block, uint32_t leng) {
	return lizec_crc32(crc, block, leng);
}
uint32_t mycrc32_combine(uint32_t crc1, uint32_t crc2, uint32_t leng2) {
	return lizec_crc32_combine(crc1, crc2, leng2);
}
void mycrc32_init(void) {
	lizec_crc32_init();
}

/* ------------------------------------------------------------------ */
/* XOR family host op (block_xor.h:33 drop-in; xor parity on GPU runs  */
/* through the EC kernel with all-ones coefficients)                   */
/* ------------------------------------------------------------------ */

extern "C" void lizec_blockxor(uint8_t *dest, const uint8_t *source,
                               size_t size) {
	size_t i = 0;
	for (; i + 8 <= size; i += 8) {
		uint64_t a, b;
		memcpy(&a, dest + i, 8);
		memcpy(&b, source + i, 8);
		a ^= b;
		memcpy(dest + i, &a, 8);
	}
	for (; i < size; ++i) dest[i] ^= source[i];
}

/* C++-linkage alias for the reference's mangled symbol (block_xor.h:33). */
void blockXor(uint8_t *dest, const uint8_t *source, size_t size) {
	lizec_blockxor(dest, source, size);
}

/* ------------------------------------------------------------------ */
/* Slice-type algebra (goal.h:108-119, slice_traits.h, chunk_part_type.h) */
/* ------------------------------------------------------------------ */

namespace {
constexpr int kECFirst = 10;               /* goal.h:118 */
constexpr int kECLast = kECFirst + 31 * 32 - 1;
constexpr int kMaxPartsCount = 64;         /* chunk_part_type.h:145 */
constexpr int64_t kBlockSize = 65536;      /* MFSBLOCKSIZE */
}

extern "C" int lizec_slice_type_ec(int k, int m) {
	if (k < 2 || k > 32 || m < 1 || m > 32) return -1;
	return 32 * (k - 2) + (m - 1) + kECFirst;   /* slice_traits.h:148-151 */
}

extern "C" int lizec_slice_is_ec(int t) {
	return t >= kECFirst && t <= kECLast;       /* slice_traits.h:67-70 */
}

extern "C" int lizec_slice_data_parts(int t) {
	if (!lizec_slice_is_ec(t)) return -1;
	return 2 + (t - kECFirst) / 32;             /* slice_traits.h:159-161 */
}

extern "C" int lizec_slice_parity_parts(int t) {
	if (!lizec_slice_is_ec(t)) return -1;
	return 1 + (t - kECFirst) % 32;             /* slice_traits.h:171-173 */
}

extern "C" int lizec_chunk_part_id(int slice_type, int part) {
	return slice_type * kMaxPartsCount + part;  /* chunk_part_type.h:170-174 */
}

extern "C" int lizec_chunk_part_slice_type(int id) {
	return id / kMaxPartsCount;
}

extern "C" int lizec_chunk_part_index(int id) {
	return id % kMaxPartsCount;
}

extern "C" int64_t lizec_chunk_part_length(int slice_type, int part,
                                           int64_t chunk_length) {
	/* slice_traits.h:332-349 */
	int k = lizec_slice_data_parts(slice_type);
	if (k < 0) return -1;
	if (k == 1) return chunk_length;
	int64_t full_stripe = chunk_length / (k * kBlockSize);
	int64_t base_len = full_stripe * kBlockSize;
	int64_t rest = chunk_length - base_len * k;
	int data_part_index = (part < k) ? part : 0;
	int64_t part_rest = rest - (int64_t)data_part_index * kBlockSize;
	if (part_rest < 0) part_rest = 0;
	if (part_rest > kBlockSize) part_rest = kBlockSize;
	return base_len + part_rest;
}
