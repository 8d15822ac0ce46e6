/* crc_fold.h — LDS-free carry-less-folding CRC32 core for gfx950 (CDNA4).
 *
 * Replaces the per-byte LDS table lookups of the slicing kernels with pure
 * VALU shift/XOR folding: the round-1 counters showed the table method is
 * bound by LDS bank conflicts (random byte indices, ~2.2x irreducible) and
 * the serial lookup chain, capping it at ~0.39 of the HBM roofline
 * (profiles/ROUND1.md).  Here each lane keeps 128-bit accumulators and
 * folds 16 data bytes per step:
 *
 *     acc' = clmul(lo64(acc), KL) ^ clmul(hi64(acc), KH) ^ data16
 *
 * with clmul against the compile-time constants emitted by
 * tools/gen_crc_fold.py (solved from the mycrc32 semantics, crc.cc:113-151,
 * and minimized to 8-9 set bits each by information-set decoding over the
 * 32-dim solution coset).  The constant tests unroll away, so one fold is
 * ~17 shifted-XOR terms of straight-line VALU code — no LDS in the hot
 * loop, no serial table chain.
 * LDS is used only in the epilogue: a single 256-entry byte table reduces
 * the final 16-byte accumulator (16 lookups per lane segment), and the
 * mycrc32_combine advance matrices (crc.cc:153-224) drive the cross-lane
 * shfl tree exactly as in the table kernels.
 *
 * Semantics are mycrc32 (reflected CRC-32, poly 0xEDB88320, zlib
 * compatible); bit-exactness vs the oracle/golden vectors is enforced by
 * tests/test_gpu_parity.py and tests/test_crc_partial.py.
 */
#pragma once
#include <hip/hip_runtime.h>

#include <cstdint>

#include "crc_fold_consts.h"

/* Apply advance-by-len2-zero-bytes to crc (mycrc32_combine semantics,
 * crc.cc:207-224) using the LDS matrix bank (M_i = append 2^i zero bytes). */
__device__ __forceinline__ uint32_t crc_advance(uint32_t crc, uint32_t len2,
                                                const uint32_t *mats) {
	int i = 0;
	while (len2) {
		if (len2 & 1) {
			const uint32_t *M = mats + i * 32;
			uint32_t r = 0, v = crc;
#pragma unroll
			for (int j = 0; j < 32; ++j) {
				r ^= (v & 1) ? M[j] : 0u;
				v >>= 1;
			}
			crc = r;
		}
		len2 >>= 1;
		++i;
	}
	return crc;
}

/* XOR the carry-less product (a1:a0) * K into r0..r3.  K is a compile-time
 * constant: the bit tests fold away and each set bit costs ~3 shifts (the
 * middle word compiles to v_alignbit) + 3 XORs. */
template <uint64_t K>
__device__ __forceinline__ void clmul_acc(uint32_t a0, uint32_t a1,
                                          uint32_t &r0, uint32_t &r1,
                                          uint32_t &r2, uint32_t &r3) {
#pragma unroll
	for (int s = 0; s < 64; ++s) {
		if ((K >> s) & 1) {
			const int t = s & 31;
			uint32_t w0 = t ? (a0 << t) : a0;
			uint32_t w1 = t ? ((a1 << t) | (a0 >> (32 - t))) : a1;
			uint32_t w2 = t ? (a1 >> (32 - t)) : 0u;
			if (s < 32) {
				r0 ^= w0;
				r1 ^= w1;
				if (t) r2 ^= w2;
			} else {
				r1 ^= w0;
				r2 ^= w1;
				if (t) r3 ^= w2;
			}
		}
	}
}

/* One fold step: acc = clmul(lo64, KL) ^ clmul(hi64, KH) ^ data16.
 * (The products stay below 2^128 because KL/KH are < 2^64.) */
template <uint64_t KL, uint64_t KH>
__device__ __forceinline__ void crc_fold16(uint32_t (&acc)[4], uint32_t d0,
                                           uint32_t d1, uint32_t d2,
                                           uint32_t d3) {
	uint32_t r0 = d0, r1 = d1, r2 = d2, r3 = d3;
	clmul_acc<KL>(acc[0], acc[1], r0, r1, r2, r3);
	clmul_acc<KH>(acc[2], acc[3], r0, r1, r2, r3);
	acc[0] = r0;
	acc[1] = r1;
	acc[2] = r2;
	acc[3] = r3;
}

/* Fold with the constants matching the accumulator interleave depth
 * (NACC accumulators -> fold distance NACC*16 bytes). */
template <int NACC>
__device__ __forceinline__ void crc_fold_step(uint32_t (&acc)[4],
                                              const uint4 &d) {
	static_assert(NACC == 1 || NACC == 2 || NACC == 4, "NACC in {1,2,4}");
	if constexpr (NACC == 1)
		crc_fold16<kCrcFoldKL1, kCrcFoldKH1>(acc, d.x, d.y, d.z, d.w);
	else if constexpr (NACC == 2)
		crc_fold16<kCrcFoldKL2, kCrcFoldKH2>(acc, d.x, d.y, d.z, d.w);
	else
		crc_fold16<kCrcFoldKL4, kCrcFoldKH4>(acc, d.x, d.y, d.z, d.w);
}

/* Reduce a 16-byte accumulator to the raw (un-inverted) CRC state with the
 * plain byte table — 16 LDS lookups per lane segment, epilogue only. */
__device__ __forceinline__ uint32_t crc_reduce16(const uint32_t (&acc)[4],
                                                 const uint32_t *T0) {
	uint32_t s = 0;
#pragma unroll
	for (int w = 0; w < 4; ++w) {
		uint32_t v = acc[w];
#pragma unroll
		for (int b = 0; b < 4; ++b)
			s = (s >> 8) ^ T0[(s ^ (v >> (8 * b))) & 0xFFu];
	}
	return s;
}

/* 16-byte load; AL16=false uses dword loads for 4-mod-16 bases (the
 * INTERLEAVED chunk format puts block data at offset 4); NT streams
 * through non-temporal loads (read-once data). */
template <bool AL16, bool NT = false>
__device__ __forceinline__ uint4 crc_ld16(const uint8_t *p) {
	if constexpr (AL16 && NT) {
		typedef unsigned int u32x4 __attribute__((ext_vector_type(4)));
		u32x4 v = __builtin_nontemporal_load((const u32x4 *)p);
		return make_uint4(v.x, v.y, v.z, v.w);
	} else if constexpr (AL16) {
		return *(const uint4 *)p;
	} else {
		const uint32_t *u = (const uint32_t *)p;
		return make_uint4(u[0], u[1], u[2], u[3]);
	}
}

/* Per-wave CRC of one block via carry-less folding.
 *
 * Layout mirrors the table kernel (crc_block_wave): the block is cut into
 * C spans; lane l owns segment l of every span; each (lane, span) chain
 * keeps NACC interleaved 128-bit accumulators (ILP on the serial fold
 * dependency) and bursts BV*16 contiguous bytes per iteration so every
 * fetched line is consumed while resident.  Per-lane segment CRCs fold in
 * a shfl tree with the advance matrices; lane 0 splices the C span CRCs.
 * Host guarantees block_len % (C * 64 * 16 * BV) == 0 (BV = 8, or 4 when
 * C = 4).  Returns the block CRC (lane 0's value is authoritative). */
template <int C, int NACC, bool AL16 = true, bool NT = false,
          bool PF = false, int BVO = 0>
__device__ uint32_t crc_block_wave_fold(const uint8_t *__restrict__ block,
                                        uint32_t block_len, uint32_t seed,
                                        const uint32_t *T0,
                                        const uint32_t *mats, int lane) {
	constexpr int BV = BVO ? BVO : (C >= 4 ? 4 : 8);
	static_assert(BV % NACC == 0, "burst must cover whole interleave groups");
	const uint32_t span = block_len / C;
	const uint32_t seg = span >> 6; /* bytes per lane, % (16*BV) == 0 */
	const uint8_t *base = block + (uint32_t)lane * seg;
	uint32_t acc[C][NACC][4];
	uint4 w[C][BV];
	auto ldburst = [&](uint4 (&buf)[C][BV], uint32_t i) {
#pragma unroll
		for (int c = 0; c < C; ++c)
#pragma unroll
			for (int q = 0; q < BV; ++q)
				buf[c][q] = crc_ld16<AL16, NT>(base + c * span + i + q * 16);
	};
	auto foldburst = [&](uint4 (&buf)[C][BV]) {
#pragma unroll
		for (int q = 0; q < BV; ++q)
#pragma unroll
			for (int c = 0; c < C; ++c)
				crc_fold_step<NACC>(acc[c][q % NACC], buf[c][q]);
	};
	/* burst 0: seed the accumulators from the first NACC 16B groups */
	ldburst(w, 0);
#pragma unroll
	for (int c = 0; c < C; ++c) {
		const uint32_t raw0 =
		    ((c == 0 && lane == 0) ? seed : 0u) ^ 0xFFFFFFFFu;
#pragma unroll
		for (int a = 0; a < NACC; ++a) {
			acc[c][a][0] = w[c][a].x ^ (a == 0 ? raw0 : 0u);
			acc[c][a][1] = w[c][a].y;
			acc[c][a][2] = w[c][a].z;
			acc[c][a][3] = w[c][a].w;
		}
#pragma unroll
		for (int q = NACC; q < BV; ++q)
			crc_fold_step<NACC>(acc[c][q % NACC], w[c][q]);
	}
	if (PF) {
		/* software pipeline: the next burst's loads are in flight while
		 * the current burst folds (two static buffers, 2 bursts/iter) */
		uint4 w2[C][BV];
		uint32_t i = 16 * BV;
		if (i < seg) ldburst(w2, i);
		for (; i < seg; i += 2 * 16 * BV) {
			if (i + 16 * BV < seg) ldburst(w, i + 16 * BV);
			foldburst(w2);
			if (i + 16 * BV < seg) {
				if (i + 2 * 16 * BV < seg) ldburst(w2, i + 2 * 16 * BV);
				foldburst(w);
			}
		}
	} else {
		for (uint32_t i = 16 * BV; i < seg; i += 16 * BV) {
			ldburst(w, i);
			foldburst(w);
		}
	}
	/* per-chain epilogue: merge the NACC interleaved accumulators with
	 * 16-byte-distance folds, table-reduce, invert */
	uint32_t crc[C];
#pragma unroll
	for (int c = 0; c < C; ++c) {
		uint32_t r[4] = {acc[c][0][0], acc[c][0][1], acc[c][0][2],
		                 acc[c][0][3]};
#pragma unroll
		for (int a = 1; a < NACC; ++a)
			crc_fold16<kCrcFoldKL1, kCrcFoldKH1>(
			    r, acc[c][a][0], acc[c][a][1], acc[c][a][2], acc[c][a][3]);
		crc[c] = crc_reduce16(r, T0) ^ 0xFFFFFFFFu;
	}
	/* cross-lane fold + span splice — same algebra as the table kernels */
	uint32_t len = seg;
#pragma unroll
	for (int s = 0; s < 6; ++s) {
		uint32_t olen = __shfl_down(len, 1 << s, 64);
#pragma unroll
		for (int c = 0; c < C; ++c) {
			uint32_t o = __shfl_down(crc[c], 1 << s, 64);
			crc[c] = crc_advance(crc[c], olen, mats) ^ o;
		}
		len += olen;
	}
	uint32_t a = crc[0];
#pragma unroll
	for (int c = 1; c < C; ++c) a = crc_advance(a, span, mats) ^ crc[c];
	return a;
}
