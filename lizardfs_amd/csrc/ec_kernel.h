/* ec_kernel.h — the GF(2^8) EC encode/decode kernel for gfx950 (CDNA4),
 * shared between the product library (lizec_gpu.hip) and the variant
 * benchmark harness (bench_variants.hip).
 *
 * Algorithm: dest[l][i] = XOR_j tbl(l,j)[src[j][i]] over GF(2^8) — the
 * reference's ec_encode_data contract (galois_field_encode.cc:28-47,
 * ISA-L table layout [dest][src][{lo16,hi16}]).  GF multiply by a constant
 * is two in-register 16-byte LUTs (low/high nibble products) built from
 * v_perm_b32; tables are staged in LDS per block and broadcast-read into
 * VGPRs once per (dest, src) per tile.
 *
 * Template knobs:
 *   D    destinations per pass (accumulators in VGPRs)
 *   CH   4-KiB chunks per tile (per-thread bytes = CH*16)
 *   SWZ  XCD-aware block remap (dispatcher puts block b on XCD b%8; the
 *        bijective remap gives each XCD a contiguous tile range)
 *   NTST non-temporal parity stores (streaming, no reuse)
 */
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

constexpr int kThreads = 256;
constexpr uint32_t kChunkBytes = kThreads * 16;   /* 4096 */

/* 16-entry byte LUT on 4 packed nibbles via v_perm_b32 (sel byte 0-3 picks
 * a byte of src1, 4-7 of src0). */
__device__ __forceinline__ uint32_t lut16(uint32_t t0, uint32_t t1,
                                          uint32_t t2, uint32_t t3,
                                          uint32_t nib) {
	uint32_t s3 = nib & 0x07070707u;
	uint32_t lo = __builtin_amdgcn_perm(t1, t0, s3);   /* nib in 0..7  */
	uint32_t hi = __builtin_amdgcn_perm(t3, t2, s3);   /* nib in 8..15 */
	uint32_t msel = 0x03020100u | ((nib >> 1) & 0x04040404u);
	return __builtin_amdgcn_perm(hi, lo, msel);
}

/* GF(2^8) multiply-accumulate of one 32-bit word against one coefficient
 * table (L = products of low nibbles, H = of high nibbles). */
__device__ __forceinline__ uint32_t gf_macc(uint32_t acc, uint32_t w,
                                            const uint4 &L, const uint4 &H) {
	uint32_t nl = w & 0x0f0f0f0fu;
	uint32_t nh = (w >> 4) & 0x0f0f0f0fu;
	return acc ^ lut16(L.x, L.y, L.z, L.w, nl) ^ lut16(H.x, H.y, H.z, H.w, nh);
}

template <int D, int CH>
__device__ __forceinline__ void gf_macc_all(uint4 (&acc)[D][CH], int c,
                                            const uint4 &w, const uint4 (&L)[D],
                                            const uint4 (&H)[D]) {
#pragma unroll
	for (int d = 0; d < D; ++d) {
		acc[d][c].x = gf_macc(acc[d][c].x, w.x, L[d], H[d]);
		acc[d][c].y = gf_macc(acc[d][c].y, w.y, L[d], H[d]);
		acc[d][c].z = gf_macc(acc[d][c].z, w.z, L[d], H[d]);
		acc[d][c].w = gf_macc(acc[d][c].w, w.w, L[d], H[d]);
	}
}

/* Quarter-LUT GF multiply: byte n = f0 + 4*f1 + 16*f2 + 64*f3 (2-bit
 * fields, disjoint bits => XOR), so c*n = c*f0 ^ (4c)*f1 ^ (16c)*f2 ^
 * (64c)*f3 by GF(2^8) linearity.  Each 4-entry product table packs into
 * ONE register and one v_perm (sel 0-3 picks src1 bytes), so a dest costs
 * 4 perms + xor-fold instead of 6 perms + selector math — and the LDS
 * table is 16 bytes per (src,dest) instead of 32.  Packed layout (from
 * the ISA-L 32-byte table): [tbl[0..3], tbl[{0,4,8,12}], tbl[16..19],
 * tbl[16+{0,4,8,12}]]. */
__device__ __forceinline__ uint32_t gf_macc_q(uint32_t acc, const uint4 &T,
                                              uint32_t f0, uint32_t f1,
                                              uint32_t f2, uint32_t f3) {
	uint32_t p0 = __builtin_amdgcn_perm(T.x, T.x, f0);
	uint32_t p1 = __builtin_amdgcn_perm(T.y, T.y, f1);
	uint32_t p2 = __builtin_amdgcn_perm(T.z, T.z, f2);
	uint32_t p3 = __builtin_amdgcn_perm(T.w, T.w, f3);
	return acc ^ p0 ^ p1 ^ p2 ^ p3;
}

template <int D, int CH>
__device__ __forceinline__ void gf_macc_all_q(uint4 (&acc)[D][CH], int c,
                                              const uint4 &w,
                                              const uint4 (&T)[D]) {
	constexpr uint32_t M = 0x03030303u;
	uint32_t f[4][4];
	const uint32_t ws[4] = {w.x, w.y, w.z, w.w};
#pragma unroll
	for (int q = 0; q < 4; ++q) {
		f[q][0] = ws[q] & M;
		f[q][1] = (ws[q] >> 2) & M;
		f[q][2] = (ws[q] >> 4) & M;
		f[q][3] = (ws[q] >> 6) & M;
	}
#pragma unroll
	for (int d = 0; d < D; ++d) {
		acc[d][c].x = gf_macc_q(acc[d][c].x, T[d], f[0][0], f[0][1], f[0][2], f[0][3]);
		acc[d][c].y = gf_macc_q(acc[d][c].y, T[d], f[1][0], f[1][1], f[1][2], f[1][3]);
		acc[d][c].z = gf_macc_q(acc[d][c].z, T[d], f[2][0], f[2][1], f[2][2], f[2][3]);
		acc[d][c].w = gf_macc_q(acc[d][c].w, T[d], f[3][0], f[3][1], f[3][2], f[3][3]);
	}
}

/* Mixed-radix GF multiply: byte n = f0 + 8*f1 + 64*f2 (3+3+2-bit fields,
 * disjoint bits => XOR over GF(2^8) linearity).  An 8-entry product table
 * spans exactly the two source operands of one v_perm (sel 0-3 -> src1
 * bytes, 4-7 -> src0), so a dest costs 3 perms + 3 xors per word instead
 * of the quarter-LUT's 4+4 — at the price of a 32-byte padded LDS table
 * per (src,dest) pair read as b128+b32.  Packed layout:
 *   [0..3]   c*v, v=0..3        [4..7]   c*v, v=4..7
 *   [8..11]  c*(v<<3), v=0..3   [12..15] c*(v<<3), v=4..7
 *   [16..19] c*(v<<6), v=0..3   [20..31] pad */
__device__ __forceinline__ uint32_t gf_macc_mr(uint32_t acc, const uint4 &T,
                                               uint32_t t6, uint32_t f0,
                                               uint32_t f1, uint32_t f2) {
	uint32_t p0 = __builtin_amdgcn_perm(T.y, T.x, f0);
	uint32_t p1 = __builtin_amdgcn_perm(T.w, T.z, f1);
	uint32_t p2 = __builtin_amdgcn_perm(t6, t6, f2);
	return acc ^ p0 ^ p1 ^ p2;
}

template <int D, int CH>
__device__ __forceinline__ void gf_macc_all_mr(uint4 (&acc)[D][CH], int c,
                                               const uint4 &w,
                                               const uint4 (&T)[D],
                                               const uint4 (&T6)[D]) {
	constexpr uint32_t M3 = 0x07070707u, M2 = 0x03030303u;
	uint32_t f[4][3];
	const uint32_t ws[4] = {w.x, w.y, w.z, w.w};
#pragma unroll
	for (int q = 0; q < 4; ++q) {
		f[q][0] = ws[q] & M3;
		f[q][1] = (ws[q] >> 3) & M3;
		f[q][2] = (ws[q] >> 6) & M2;
	}
#pragma unroll
	for (int d = 0; d < D; ++d) {
		acc[d][c].x = gf_macc_mr(acc[d][c].x, T[d], T6[d].x, f[0][0], f[0][1], f[0][2]);
		acc[d][c].y = gf_macc_mr(acc[d][c].y, T[d], T6[d].x, f[1][0], f[1][1], f[1][2]);
		acc[d][c].z = gf_macc_mr(acc[d][c].z, T[d], T6[d].x, f[2][0], f[2][1], f[2][2]);
		acc[d][c].w = gf_macc_mr(acc[d][c].w, T[d], T6[d].x, f[3][0], f[3][1], f[3][2]);
	}
}

/* Bijective XCD remap (8 XCDs): consecutive hardware block ids round-robin
 * the XCDs; remapped ids give each XCD one contiguous range. */
__device__ __forceinline__ uint32_t xcd_remap(uint32_t b, uint32_t n) {
	uint32_t xcd = b & 7u, i = b >> 3;
	uint32_t q = n >> 3, r = n & 7u;
	return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
}

/* non-temporal 16B load helper (streamed sources, no reuse) */
__device__ __forceinline__ uint4 ld_nt(const uint8_t *p) {
	typedef unsigned int u32x4 __attribute__((ext_vector_type(4)));
	u32x4 v = __builtin_nontemporal_load((const u32x4 *)p);
	return make_uint4(v.x, v.y, v.z, v.w);
}

template <int D, int CH, bool SWZ, bool NTST, bool NTLD = false,
          bool TPIPE = false, bool QL = false, bool MR = false>
__global__ __launch_bounds__(kThreads) void ec_encode_kernel(
    uint32_t part_len, int srcs, int dest_base,
    const uint8_t *__restrict__ gftbls_dev,  /* 32*srcs*dests_total */
    const uint64_t *__restrict__ src_ptrs,   /* [stripes][srcs]  */
    const uint64_t *__restrict__ dst_ptrs,   /* [stripes][dests_total] */
    int dests_total, uint32_t tiles_per_part, uint32_t total_tiles) {
	/* QL keeps its packed table in L only; H is sized 1 then, so the
	 * TPIPE double-buffer path (H[1]) must not be combined with it. */
	static_assert(!(QL && TPIPE), "QL and TPIPE are mutually exclusive");
	static_assert(!(MR && (QL || TPIPE)), "MR excludes QL/TPIPE");
	constexpr uint32_t kTile = kChunkBytes * CH;
	extern __shared__ __attribute__((aligned(16))) uint8_t smem[];
	const uint32_t tid = threadIdx.x;

	/* Stage this pass's table rows: [D][srcs][32] bytes (16 when QL —
	 * gftbls_dev then holds the packed quarter-LUT layout). */
	{
		const int tw = QL ? 16 : 32;
		const uint8_t *src_tbl = gftbls_dev + (size_t)dest_base * srcs * tw;
		int nbytes = D * srcs * tw;
		for (int i = tid * 16; i < nbytes; i += kThreads * 16)
			*(uint4 *)(smem + i) = *(const uint4 *)(src_tbl + i);
	}
	__syncthreads();

	uint32_t b0 = SWZ ? xcd_remap(blockIdx.x, gridDim.x) : blockIdx.x;
	for (uint32_t tile = b0; tile < total_tiles; tile += gridDim.x) {
		uint32_t stripe = tile / tiles_per_part;
		uint32_t tin = tile - stripe * tiles_per_part;
		uint32_t base = tin * kTile + tid * 16u;
		const uint64_t *sp_tab = src_ptrs + (uint64_t)stripe * srcs;
		const uint64_t *dp_tab =
		    dst_ptrs + (uint64_t)stripe * dests_total + dest_base;

		uint4 acc[D][CH];
#pragma unroll
		for (int d = 0; d < D; ++d)
#pragma unroll
			for (int c = 0; c < CH; ++c)
				acc[d][c] = make_uint4(0, 0, 0, 0);

		if ((uint64_t)(tin + 1) * kTile <= part_len) {
			/* full tile: branchless; next source's strips — and, with
			 * TPIPE, its coefficient tables — prefetched while the
			 * current source is accumulated (the per-source LDS
			 * broadcast + lgkmcnt wait otherwise serializes large k) */
			uint4 w[CH], wn[CH];
			uint4 L[TPIPE ? 2 : 1][D], H[QL ? 1 : (TPIPE ? 2 : 1)][D];
			auto tbl_read = [&](int j, uint4 (&Lb)[D], uint4 (&Hb)[D]) {
#pragma unroll
				for (int d = 0; d < D; ++d) {
					if (QL) {
						Lb[d] = *(const uint4 *)(smem +
						                         ((size_t)d * srcs + j) * 16);
					} else if (MR) {
						const uint8_t *tb =
						    smem + ((size_t)d * srcs + j) * 32;
						Lb[d] = *(const uint4 *)tb;
						Hb[d].x = *(const uint32_t *)(tb + 16);
					} else {
						const uint8_t *tb =
						    smem + ((size_t)d * srcs + j) * 32;
						Lb[d] = *(const uint4 *)tb;
						Hb[d] = *(const uint4 *)(tb + 16);
					}
				}
			};
			{
				const uint8_t *sp = (const uint8_t *)sp_tab[0];
#pragma unroll
				for (int c = 0; c < CH; ++c)
					w[c] = NTLD ? ld_nt(sp + (base + c * kChunkBytes))
					            : *(const uint4 *)(sp + (base + c * kChunkBytes));
			}
			/* one source step: compute with (Lc,Hc) while prefetching the
			 * next source's strips and (with TPIPE) its tables into
			 * (Ln,Hn) — all buffer indices static so everything stays in
			 * registers */
			auto step = [&](int j, uint4 (&Lc)[D], uint4 (&Hc)[D],
			                uint4 (&Ln)[D], uint4 (&Hn)[D]) {
				if (j + 1 < srcs) {
					const uint8_t *spn = (const uint8_t *)sp_tab[j + 1];
#pragma unroll
					for (int c = 0; c < CH; ++c)
						wn[c] = NTLD ? ld_nt(spn + (base + c * kChunkBytes))
						             : *(const uint4 *)(spn + (base + c * kChunkBytes));
					if (TPIPE) tbl_read(j + 1, Ln, Hn);
				}
				if (!TPIPE) tbl_read(j, Lc, Hc);
#pragma unroll
				for (int c = 0; c < CH; ++c) {
					if (QL)
						gf_macc_all_q<D, CH>(acc, c, w[c], Lc);
					else if (MR)
						gf_macc_all_mr<D, CH>(acc, c, w[c], Lc, Hc);
					else
						gf_macc_all<D, CH>(acc, c, w[c], Lc, Hc);
				}
#pragma unroll
				for (int c = 0; c < CH; ++c) w[c] = wn[c];
			};
			if (TPIPE) tbl_read(0, L[0], H[0]);
			int j = 0;
			for (; j + 1 < srcs; j += 2) {
				step(j, L[0], H[0], L[TPIPE ? 1 : 0], H[TPIPE ? 1 : 0]);
				step(j + 1, L[TPIPE ? 1 : 0], H[TPIPE ? 1 : 0], L[0], H[0]);
			}
			if (j < srcs) step(j, L[0], H[0], L[TPIPE ? 1 : 0], H[TPIPE ? 1 : 0]);
#pragma unroll
			for (int d = 0; d < D; ++d) {
				uint8_t *dp = (uint8_t *)dp_tab[d];
#pragma unroll
				for (int c = 0; c < CH; ++c) {
					uint4 *p = (uint4 *)(dp + (base + c * kChunkBytes));
					if (NTST) {
						typedef unsigned int u32x4
						    __attribute__((ext_vector_type(4)));
						u32x4 v = {acc[d][c].x, acc[d][c].y, acc[d][c].z,
						           acc[d][c].w};
						__builtin_nontemporal_store(v, (u32x4 *)p);
					} else {
						*p = acc[d][c];
					}
				}
			}
		} else {
			/* ragged tail tile: per-strip bounds checks */
			for (int j = 0; j < srcs; ++j) {
				const uint8_t *sp = (const uint8_t *)sp_tab[j];
				uint4 L[D], H[D];
#pragma unroll
				for (int d = 0; d < D; ++d) {
					if (QL) {
						L[d] = *(const uint4 *)(smem +
						                        ((size_t)d * srcs + j) * 16);
					} else if (MR) {
						const uint8_t *tb =
						    smem + ((size_t)d * srcs + j) * 32;
						L[d] = *(const uint4 *)tb;
						H[d].x = *(const uint32_t *)(tb + 16);
					} else {
						const uint8_t *tb =
						    smem + ((size_t)d * srcs + j) * 32;
						L[d] = *(const uint4 *)tb;
						H[d] = *(const uint4 *)(tb + 16);
					}
				}
#pragma unroll
				for (int c = 0; c < CH; ++c) {
					uint32_t off = base + c * kChunkBytes;
					if (off < part_len) {
						uint4 w = *(const uint4 *)(sp + off);
						if (QL)
							gf_macc_all_q<D, CH>(acc, c, w, L);
						else if (MR)
							gf_macc_all_mr<D, CH>(acc, c, w, L, H);
						else
							gf_macc_all<D, CH>(acc, c, w, L, H);
					}
				}
			}
#pragma unroll
			for (int d = 0; d < D; ++d) {
				uint8_t *dp = (uint8_t *)dp_tab[d];
#pragma unroll
				for (int c = 0; c < CH; ++c) {
					uint32_t off = base + c * kChunkBytes;
					if (off < part_len) *(uint4 *)(dp + off) = acc[d][c];
				}
			}
		}
	}
}
