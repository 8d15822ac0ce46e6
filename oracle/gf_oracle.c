/* oracle/gf_oracle.c — CPU oracle for the LizardFS EC hot path.
 *
 * TEST INFRASTRUCTURE ONLY.  This file is the parity checker for the GPU
 * engine: a plain-C restatement of the reference algorithms.  Only tests/,
 * __graft_entry__.smoke() and bench.py's cpu_baseline leg may call into this
 * library; the product path (lizardfs_amd + liblizec) never links or loads it.
 *
 * Every function cites the reference (lizardfs/lizardfs @ /root/reference)
 * file:line it restates.  The GF(2^8) field is the ISA-L field: polynomial
 * 0x11D (x^8+x^4+x^3+x^2+1), generator 2 — reference
 * src/common/galois_coeff.h:30-32 (gf_mul2) and :68-71 (log/exp tables).
 *
 * Parity pinning: validated bit-exactly against the reference's own code
 * compiled from /root/reference into oracle/_ref (see Makefile), and against
 * the reference's known answers (crc("a")=0xE8B7BE43, crc_unittest.cc:30;
 * recovery bit-equality, reed_solomon_unittest.cc:136-199).
 */
#include <stdint.h>
#include <stddef.h>
#include <string.h>

/* ---------------- GF(2^8) base field ---------------- */

static uint8_t gf_log_tbl[256];
static uint8_t gf_exp_tbl[256];
static int gf_initialized = 0;

/* Multiply by 2 in GF(2^8), poly 0x11D.  Reference: galois_coeff.h:30-32. */
static uint8_t gf_mul2(uint8_t x) {
	return (uint8_t)((x << 1) ^ ((x & 0x80) ? 0x1d : 0));
}

/* Log/exp tables with the reference's conventions: exp[0]=1, exp[n]=2^n
 * (so exp[255]=1), log[0]=0, log[1]=255.  Reference: galois_coeff.h:40-71. */
void oracle_init(void) {
	if (gf_initialized) return;
	uint8_t p = 1;
	gf_exp_tbl[0] = 1;
	gf_log_tbl[0] = 0;
	for (int n = 1; n <= 255; ++n) {
		p = gf_mul2(p);
		gf_exp_tbl[n] = p;
		gf_log_tbl[p] = (uint8_t)n;   /* n=255 sets log[1]=255 */
	}
	gf_initialized = 1;
}

/* Reference: galois_field_isal.cc:37-44. */
uint8_t oracle_gf_mul(uint8_t a, uint8_t b) {
	int i;
	if (a == 0 || b == 0) return 0;
	i = gf_log_tbl[a] + gf_log_tbl[b];
	return gf_exp_tbl[i > 254 ? i - 255 : i];
}

/* Reference: galois_field_isal.cc:46-51. */
uint8_t oracle_gf_inv(uint8_t a) {
	if (a == 0) return 0;
	return gf_exp_tbl[255 - gf_log_tbl[a]];
}

/* ---------------- generator matrices ---------------- */

/* Vandermonde-style matrix: identity on top, then rows of successive powers
 * of gen (gen itself stepping through powers of 2).  m = total rows
 * (k data + parity), k = columns.  Reference: galois_field_isal.cc:53-69. */
void oracle_gf_gen_rs_matrix(uint8_t *a, int m, int k) {
	oracle_init();
	memset(a, 0, (size_t)k * m);
	for (int i = 0; i < k; ++i) a[k * i + i] = 1;
	uint8_t gen = 1;
	for (int i = k; i < m; ++i) {
		uint8_t p = 1;
		for (int j = 0; j < k; ++j) {
			a[k * i + j] = p;
			p = oracle_gf_mul(p, gen);
		}
		gen = oracle_gf_mul(gen, 2);
	}
}

/* Cauchy matrix: identity on top, then rows 1/(i xor j).
 * Reference: galois_field_isal.cc:71-85. */
void oracle_gf_gen_cauchy1_matrix(uint8_t *a, int m, int k) {
	oracle_init();
	memset(a, 0, (size_t)k * m);
	for (int i = 0; i < k; ++i) a[k * i + i] = 1;
	uint8_t *p = &a[k * k];
	for (int i = k; i < m; ++i)
		for (int j = 0; j < k; ++j)
			*p++ = oracle_gf_inv((uint8_t)(i ^ j));
}

/* Gauss-Jordan inversion in GF(2^8); mutates in_mat; -1 if singular.
 * Reference: galois_field_isal.cc:87-139. */
int oracle_gf_invert_matrix(uint8_t *in_mat, uint8_t *out_mat, const int n) {
	int i, j, k;
	uint8_t temp;
	oracle_init();
	for (i = 0; i < n * n; ++i) out_mat[i] = 0;
	for (i = 0; i < n; ++i) out_mat[i * n + i] = 1;

	for (i = 0; i < n; ++i) {
		if (in_mat[i * n + i] == 0) {
			for (j = i + 1; j < n; ++j)
				if (in_mat[j * n + i]) break;
			if (j == n) return -1;
			for (k = 0; k < n; ++k) {
				temp = in_mat[i * n + k];
				in_mat[i * n + k] = in_mat[j * n + k];
				in_mat[j * n + k] = temp;
				temp = out_mat[i * n + k];
				out_mat[i * n + k] = out_mat[j * n + k];
				out_mat[j * n + k] = temp;
			}
		}
		temp = oracle_gf_inv(in_mat[i * n + i]);
		for (j = 0; j < n; ++j) {
			in_mat[i * n + j] = oracle_gf_mul(in_mat[i * n + j], temp);
			out_mat[i * n + j] = oracle_gf_mul(out_mat[i * n + j], temp);
		}
		for (j = 0; j < n; ++j) {
			if (j == i) continue;
			temp = in_mat[j * n + i];
			for (k = 0; k < n; ++k) {
				out_mat[j * n + k] ^= oracle_gf_mul(temp, out_mat[i * n + k]);
				in_mat[j * n + k] ^= oracle_gf_mul(temp, in_mat[i * n + k]);
			}
		}
	}
	return 0;
}

/* ---------------- coefficient expansion (ISA-L table format) ----------------
 *
 * 32-byte table for one coefficient c:
 *   tbl[i]    = c * i          (GF product), i = 0..15   (low-nibble products)
 *   tbl[16+i] = c * (i << 4),  i = 0..15                 (high-nibble products)
 * Reference: galois_field_isal.cc:141-244 (gf_vect_mul_init) — the 64-bit
 * magic-constant construction there produces exactly these products. */
void oracle_gf_vect_mul_init(uint8_t c, uint8_t *tbl) {
	oracle_init();
	for (int i = 0; i < 16; ++i) {
		tbl[i] = oracle_gf_mul(c, (uint8_t)i);
		tbl[16 + i] = oracle_gf_mul(c, (uint8_t)(i << 4));
	}
}

/* Linear expansion of rows*k coefficients into 32-byte tables.
 * Reference: galois_field_isal.cc:246-255.  (Note the reference's arg names
 * are swapped at some call sites — the walk is purely linear, so only the
 * product rows*k matters.) */
void oracle_ec_init_tables(int k, int rows, uint8_t *a, uint8_t *g_tbls) {
	for (int i = 0; i < rows; ++i)
		for (int j = 0; j < k; ++j) {
			oracle_gf_vect_mul_init(*a++, g_tbls);
			g_tbls += 32;
		}
}

/* ---------------- the hot kernel (scalar restatement) ----------------
 *
 * dest[l][i] = XOR over j of tbl(l,j)[src[j][i]] using the lo/hi nibble
 * split.  Reference: galois_field_encode.cc:28-47 (ec_encode_data_default);
 * table layout v + (l*srcs + j)*32. */
void oracle_ec_encode_data(int len, int srcs, int dests, uint8_t *v,
                           uint8_t **src, uint8_t **dest) {
	for (int l = 0; l < dests; ++l) {
		uint8_t *vl = v + (size_t)l * srcs * 32;
		for (int i = 0; i < len; ++i) {
			uint8_t s = 0;
			uint8_t *tbl = vl;
			for (int j = 0; j < srcs; ++j) {
				uint8_t a = src[j][i];
				s ^= tbl[a & 0xF] ^ tbl[16 + (a >> 4)];
				tbl += 32;
			}
			dest[l][i] = s;
		}
	}
}

/* ---------------- ReedSolomon<MAXK,MAXM> semantics ----------------
 *
 * Restates reed_solomon.h:41-373 for arbitrary (k,m) up to (32,32), without
 * the single-entry matrix cache (the oracle recomputes; results identical).
 * Matrix choice: Cauchy iff m>=5 || (m==4 && k>20), else Vandermonde
 * (reed_solomon.h:163-178).
 */

#define ORACLE_MAXK 32
#define ORACLE_MAXM 32
#define ORACLE_MAXP (ORACLE_MAXK + ORACLE_MAXM)

static void rs_matrix_for(int k, int m, uint8_t *rs_matrix /* (k+m)*k */) {
	if (m >= 5 || (m == 4 && k > 20)) {
		oracle_gf_gen_cauchy1_matrix(rs_matrix, k + m, k);
	} else {
		oracle_gf_gen_rs_matrix(rs_matrix, k + m, k);
	}
}

/* selectRows: reed_solomon.h:292-310 */
static void select_rows(uint8_t *out, const uint8_t *in, int s1, int s2,
                        uint64_t required_rows) {
	for (int i = 0; i < s1; ++i) {
		if (!((required_rows >> i) & 1)) { in += s2; continue; }
		memcpy(out, in, (size_t)s2);
		out += s2;
		in += s2;
	}
}

/* selectColumns: reed_solomon.h:321-332 */
static void select_columns(uint8_t *out, const uint8_t *in, int s1, int s2,
                           uint64_t required_columns) {
	for (int i = 0; i < s1; ++i) {
		for (int j = 0; j < s2; ++j)
			if ((required_columns >> j) & 1) *out++ = in[j];
		in += s2;
	}
}

/* matrixMultiply: reed_solomon.h:344-358 — out[i][c] = sum_j a[i][j]*b[j][c],
 * out rows written with stride s2 (as the reference does). */
static void matrix_multiply(uint8_t *out, int s1, int s2, int s3,
                            const uint8_t *a, const uint8_t *b) {
	for (int i = 0; i < s1; ++i)
		for (int c = 0; c < s2; ++c) {
			uint8_t s = 0;
			for (int j = 0; j < s3; ++j)
				s ^= oracle_gf_mul(a[i * s2 + j], b[j * s2 + c]);
			out[i * s2 + c] = s;
		}
}

static int popcount64(uint64_t x) {
	int c = 0;
	while (x) { c += (int)(x & 1); x >>= 1; }
	return c;
}

/* Build the expanded gf table for a recover() call, without running the data
 * pass.  Restates reed_solomon.h:87-121 (recover) + :189-217
 * (createEncodingMatrix) + :229-281 (createRecoveryMatrix).
 *
 * input_present[i] (i in [0,k+m)): part i is NOT erased.
 * input_nonnull[i]: its buffer pointer is non-NULL (NULL = implicit zeros).
 * needed[i]: part i is erased AND has a non-NULL output buffer.
 *
 * Writes gf table (32 * in_count * out_count bytes); returns 0, or -1 if the
 * decode matrix is singular (the reference's missing `throw` at
 * reed_solomon.h:250 means it would proceed on garbage; we refuse instead).
 * in_count_out = number of non-NULL surviving inputs (order of part index);
 * out_count_out = number of needed outputs (order of part index). */
int oracle_rs_make_tables(int k, int m, uint64_t present_mask,
                          uint64_t nonnull_mask, uint64_t needed_mask,
                          uint8_t *gf_tbls, int *in_count_out, int *out_count_out) {
	uint8_t rs_matrix[ORACLE_MAXP * ORACLE_MAXK];
	uint8_t tmp[ORACLE_MAXP * ORACLE_MAXK];
	uint8_t tmp2[ORACLE_MAXP * ORACLE_MAXK];
	uint8_t decode_m[ORACLE_MAXK * ORACLE_MAXK];
	uint8_t recover_m[ORACLE_MAXP * ORACLE_MAXK];
	uint8_t reduced[ORACLE_MAXP * ORACLE_MAXK];
	int nparts = k + m;
	uint64_t all_mask = nparts >= 64 ? ~(uint64_t)0
	                                 : (((uint64_t)1 << nparts) - 1);

	oracle_init();
	rs_matrix_for(k, m, rs_matrix);

	/* recover() bookkeeping, reed_solomon.h:97-111 */
	uint64_t erased = ~present_mask & all_mask;
	uint64_t non_zero_input = 0;  /* indexed by surviving-part order */
	int in_count = 0, out_count = 0, in_with_zero_count = 0;
	int data_part_count = 0, parity_needed = 0;
	for (int i = 0; i < nparts; ++i) {
		if (((erased >> i) & 1) && ((needed_mask >> i) & 1)) {
			out_count++;
			parity_needed += (i >= k);
		}
		if ((present_mask >> i) & 1) {
			if ((nonnull_mask >> i) & 1) {
				non_zero_input |= (uint64_t)1 << in_with_zero_count;
				in_count++;
			}
			in_with_zero_count++;
			data_part_count += (i < k);
		}
	}
	int nz_count = popcount64(non_zero_input);
	int needed_count = out_count;

	if (data_part_count == k) {
		/* createEncodingMatrix, reed_solomon.h:189-217 */
		select_rows(tmp, rs_matrix, nparts, k, needed_mask);
		if (nz_count < k) {
			select_columns(reduced, tmp, needed_count, k, non_zero_input);
			oracle_ec_init_tables(needed_count, nz_count, reduced, gf_tbls);
		} else {
			oracle_ec_init_tables(needed_count, k, tmp, gf_tbls);
		}
	} else {
		/* createRecoveryMatrix, reed_solomon.h:229-281 */
		select_rows(tmp, rs_matrix, nparts, k, present_mask);
		if (oracle_gf_invert_matrix(tmp, decode_m, k) != 0) return -1;
		if (parity_needed > 0) {
			select_rows(tmp2, rs_matrix, nparts, k, needed_mask);
			matrix_multiply(recover_m, needed_count, k, k, tmp2, decode_m);
		} else {
			select_rows(recover_m, decode_m, k, k, needed_mask);
		}
		if (nz_count < k) {
			select_columns(reduced, recover_m, needed_count, k, non_zero_input);
			oracle_ec_init_tables(needed_count, nz_count, reduced, gf_tbls);
		} else {
			oracle_ec_init_tables(needed_count, k, recover_m, gf_tbls);
		}
	}
	*in_count_out = in_count;
	*out_count_out = needed_count;
	return 0;
}

/* ReedSolomon::recover — reed_solomon.h:87-121.
 * fragments: array of k+m pointers (inputs for surviving parts; may be NULL
 * = implicit zeros).  outputs: array of k+m pointers (non-NULL where wanted).
 * erased_mask: exactly m bits set. */
int oracle_rs_recover(int k, int m, const uint8_t **input_fragments,
                      uint64_t erased_mask, uint8_t **output_fragments,
                      size_t data_size) {
	uint8_t gf_tbls[32 * ORACLE_MAXK * ORACLE_MAXM];
	uint8_t *in_parts[ORACLE_MAXP];
	uint8_t *out_parts[ORACLE_MAXP];
	int nparts = k + m;
	uint64_t present = ~erased_mask &
	    (nparts >= 64 ? ~(uint64_t)0 : (((uint64_t)1 << nparts) - 1));
	uint64_t nonnull = 0, needed = 0;
	int in_count = 0, out_count = 0;

	if (popcount64(erased_mask) != m) return -2;
	for (int i = 0; i < nparts; ++i) {
		if (((present >> i) & 1) && input_fragments[i]) {
			nonnull |= (uint64_t)1 << i;
			in_parts[in_count++] = (uint8_t *)input_fragments[i];
		}
		if (((erased_mask >> i) & 1) && output_fragments[i]) {
			needed |= (uint64_t)1 << i;
			out_parts[out_count++] = output_fragments[i];
		}
	}
	int ic, oc;
	if (oracle_rs_make_tables(k, m, present, nonnull, needed, gf_tbls, &ic, &oc) != 0)
		return -1;
	oracle_ec_encode_data((int)data_size, ic, oc, gf_tbls, in_parts, out_parts);
	return 0;
}

/* ReedSolomon::encode — reed_solomon.h:134-155: all m parities erased+needed. */
int oracle_rs_encode(int k, int m, const uint8_t **data_fragments,
                     uint8_t **parity_fragments, size_t data_size) {
	const uint8_t *in[ORACLE_MAXP] = {0};
	uint8_t *out[ORACLE_MAXP] = {0};
	uint64_t erased = 0;
	for (int i = 0; i < k; ++i) in[i] = data_fragments[i];
	for (int i = 0; i < m; ++i) {
		out[k + i] = parity_fragments[i];
		erased |= (uint64_t)1 << (k + i);
	}
	return oracle_rs_recover(k, m, in, erased, out, data_size);
}

/* ---------------- CRC32 ----------------
 *
 * Reflected CRC-32, poly 0xEDB88320 (protocol/MFSCommunication.h:81),
 * zlib-compatible; 4-way slicing.  Reference: crc.cc:68-151.
 * Known answer: crc32(0,"a",1) = 0xE8B7BE43 (crc_unittest.cc:30). */

static uint32_t crc_tab[4][256];
static int crc_initialized = 0;

void oracle_crc32_init(void) {
	if (crc_initialized) return;
	for (uint32_t i = 0; i < 256; ++i) {
		uint32_t c = i;
		for (int b = 0; b < 8; ++b)
			c = (c & 1) ? (0xEDB88320u ^ (c >> 1)) : (c >> 1);
		crc_tab[0][i] = c;
	}
	for (uint32_t i = 0; i < 256; ++i) {
		uint32_t c = crc_tab[0][i];
		for (int t = 1; t < 4; ++t) {
			c = crc_tab[0][c & 0xff] ^ (c >> 8);
			crc_tab[t][i] = c;
		}
	}
	crc_initialized = 1;
}

uint32_t oracle_crc32(uint32_t crc, const uint8_t *block, uint32_t leng) {
	oracle_crc32_init();
	crc ^= 0xFFFFFFFFu;
	while (leng && ((uintptr_t)block & 3)) {
		crc = crc_tab[0][(crc ^ *block++) & 0xFF] ^ (crc >> 8);
		leng--;
	}
	const uint32_t *block4 = (const uint32_t *)block;
	while (leng >= 4) {
		crc ^= *block4++;
		crc = crc_tab[3][crc & 0xff] ^ crc_tab[2][(crc >> 8) & 0xff] ^
		      crc_tab[1][(crc >> 16) & 0xff] ^ crc_tab[0][crc >> 24];
		leng -= 4;
	}
	block = (const uint8_t *)block4;
	while (leng) {
		crc = crc_tab[0][(crc ^ *block++) & 0xFF] ^ (crc >> 8);
		leng--;
	}
	return crc ^ 0xFFFFFFFFu;
}

/* crc32_combine: append leng2 zero bytes to crc1's message, then xor crc2.
 * Implemented with GF(2) matrix squaring (zlib's method); bit-identical to
 * the reference's crc_combine_table walk (crc.cc:153-224). */
static uint32_t gf2_times(const uint32_t *mat, uint32_t vec) {
	uint32_t sum = 0;
	int i = 0;
	while (vec) {
		if (vec & 1) sum ^= mat[i];
		vec >>= 1;
		i++;
	}
	return sum;
}

static void gf2_square(uint32_t *sq, const uint32_t *mat) {
	for (int i = 0; i < 32; ++i) sq[i] = gf2_times(mat, mat[i]);
}

uint32_t oracle_crc32_combine(uint32_t crc1, uint32_t crc2, uint32_t len2) {
	uint32_t even[32], odd[32];
	if (len2 == 0) return crc1 ^ crc2;
	/* odd = matrix for one zero BIT: reflected poly in row 0. */
	odd[0] = 0xEDB88320u;
	uint32_t row = 1;
	for (int i = 1; i < 32; ++i) { odd[i] = row; row <<= 1; }
	gf2_square(even, odd);  /* 2 bits */
	gf2_square(odd, even);  /* 4 bits */
	/* apply len2 (bytes): advance 8*len2 bits = len2 zero bytes.
	 * even/odd now hold "4 zero bits"; squaring from here and applying on
	 * set bits of len2 matches the reference's per-bit byte tables
	 * (crc.cc:169-205 builds combine_table[i] = advance 2^i BYTES). */
	do {
		gf2_square(even, odd);   /* even = advance 2^k bytes (k from 3) */
		if (len2 & 1) crc1 = gf2_times(even, crc1);
		len2 >>= 1;
		if (!len2) break;
		gf2_square(odd, even);
		if (len2 & 1) crc1 = gf2_times(odd, crc1);
		len2 >>= 1;
	} while (len2);
	return crc1 ^ crc2;
}

/* ---------------- bench helper: threaded stripe encode ----------------
 *
 * CPU-baseline leg for bench.py (BASELINE.md plan): encodes `nstripes`
 * independent stripes with the scalar kernel, OpenMP across stripes.
 * Layout: data = nstripes * k parts of part_len bytes, contiguous
 * [stripe][part][byte]; parity likewise [stripe][m][part_len].
 * gf_tbls = 32*k*m encode tables (from oracle_rs_make_tables or
 * oracle_ec_init_tables on the encode matrix). */
void oracle_encode_stripes(int k, int m, int part_len, int nstripes,
                           uint8_t *gf_tbls, uint8_t *data, uint8_t *parity) {
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
	for (int s = 0; s < nstripes; ++s) {
		uint8_t *in_parts[ORACLE_MAXK];
		uint8_t *out_parts[ORACLE_MAXM];
		for (int j = 0; j < k; ++j)
			in_parts[j] = data + ((size_t)s * k + j) * part_len;
		for (int l = 0; l < m; ++l)
			out_parts[l] = parity + ((size_t)s * m + l) * part_len;
		oracle_ec_encode_data(part_len, k, m, gf_tbls, in_parts, out_parts);
	}
}

/* Same shape for CRC: one CRC per block over a contiguous buffer. */
void oracle_crc32_blocks(const uint8_t *buf, int block_len, long nblocks,
                         uint32_t seed, uint32_t *crcs_out) {
	oracle_crc32_init();
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
	for (long b = 0; b < nblocks; ++b)
		crcs_out[b] = oracle_crc32(seed, buf + (size_t)b * block_len, (uint32_t)block_len);
}
