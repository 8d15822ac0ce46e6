/* lizec_gpu.hip — MI355X (gfx950/CDNA4) kernels + batch engine for the
 * LizardFS EC hot path.
 *
 * Replaces the reference's SIMD byte loop (galois_field_encode.cc:151-201,
 * AVX2 pshufb) with a CDNA4-native design:
 *  - GF(2^8) multiply via in-register mixed-radix (3+3+2 bit field)
 *    product LUTs selected with v_perm_b32 (__builtin_amdgcn_perm):
 *    3 perms + 3 XORs per destination word, tables re-read per tile as
 *    wave-uniform LDS broadcasts (ec_kernel.h gf_macc_mr);
 *  - 16-byte vectorized HBM loads/stores laid out so every wave instruction
 *    is a fully-coalesced 1 KiB access; full tiles run a branchless
 *    software-pipelined path (next source's strips prefetched while the
 *    current one is accumulated), ragged tails take a guarded path;
 *  - per-64KiB-block CRC32 (hddspacemgr.cc:1918 gate) with one wave per
 *    block via LDS-free carry-less folding (crc_fold.h) and a shfl-based
 *    combine tree using the same GF(2) "advance by N zero bytes" matrices
 *    as mycrc32_combine (crc.cc:153-224); slicing-table kernels remain
 *    for odd block sizes.
 *
 * This file is the product compute path: it REQUIRES a GPU.  There is no
 * CPU fallback here by design (parity claims are void otherwise).
 */
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <cstring>
#include <cstdlib>
#include <atomic>
#include <mutex>
#include <unordered_map>

#include "../../include/lizec.h"

#define LIZEC_CHECK(x)                                                   \
	do {                                                                 \
		hipError_t _e = (x);                                             \
		if (_e != hipSuccess) return LIZEC_EHIP;                         \
	} while (0)

#include "ec_kernel.h"
#include "crc_fold.h"

/* Product kernel configuration (chosen by the variant A/B harness,
 * bench_variants.hip; numbers in profiles/ROUND1.md + ROUND2.md):
 *  - XCD-bijective block remap + non-temporal parity stores everywhere;
 *  - tile size per destination-group width D: D<=2 uses 4-chunk (16 KiB)
 *    tiles; D>=3 uses 2-chunk (8 KiB) tiles (fewer accumulator VGPRs ->
 *    4 waves/SIMD; ec(16,4) D4: 3953 -> 4113 GB/s, ec(32,6) D6: 3001);
 *  - mixed-radix (3+3+2) GF tables: 3 v_perms + 3 XORs per dest word
 *    instead of the quarter-LUT's 4+4 — r2a/r2b A/B: ec(16,4) 4446 ->
 *    5392 GB/s (+21%), ec(32,6) 3239 -> 3953 (+22%), ec(8,2) tie. */
constexpr bool kECSwz = true;
constexpr bool kECNtStore = true;
constexpr bool kECNtLoad = true;   /* +3.6%: 5720 -> 5924 GB/s (profiles) */
constexpr int ec_chunks_for(int d) { return d <= 4 ? 4 : 2; }
constexpr int kECTblBytes = 32;    /* mixed-radix padded layout */

/* repack one 32-byte ISA-L coefficient table into the padded 32-byte
 * mixed-radix layout the MR kernel stages (see gf_macc_mr in ec_kernel.h):
 * 8-entry product tables for bit fields [0:3) and [3:6), 4-entry for
 * [6:8) — derived from the ISA-L lo/hi-nibble tables by GF linearity
 * over disjoint bits. */
static void pack_mixed_radix(const uint8_t *t, uint8_t *q) {
	for (int v = 0; v < 8; ++v) {
		q[v] = t[v];                                   /* c*v      */
		q[8 + v] = t[(8 * v) & 15] ^ t[16 + (v >> 1)]; /* c*(v<<3) */
	}
	for (int v = 0; v < 4; ++v)
		q[16 + v] = t[16 + (v << 2)];                  /* c*(v<<6) */
	memset(q + 20, 0, 12);
}

/* ------------------------------------------------------------------ */
/* CRC32 kernel                                                       */
/* ------------------------------------------------------------------ */

/* Device-constant block uploaded once per engine:
 *  [0..16*256)  u32: slicing tables T0..T15 (reflected, poly 0xEDB88320;
 *               the generic kernel uses T0..T3, the fast path T0..T7 by
 *               default or T0..T15 under LIZEC_CRC_SLICE=16)
 *  [then]       u32[26][32]: advance matrices M_i = "append 2^i zero
 *               BYTES", i = 0..25 (supports block_len < 64 MiB)          */
constexpr int kCrcTabWords = 16 * 256;  /* slicing-by-16 tables T0..T15 */
constexpr int kCrcMatCount = 26;
constexpr int kCrcConstWords = kCrcTabWords + kCrcMatCount * 32;

/* One wave per block: lane l owns segment [l*seg, l*seg+seg) of the block
 * (seg = block_len/64, multiple of 16 enforced by the host).  Lane 0 seeds;
 * the shfl tree folds 64 segment CRCs into the block CRC. */
__global__ __launch_bounds__(kThreads) void crc32_blocks_kernel(
    const uint8_t *__restrict__ buf, uint32_t block_len, uint64_t nblocks,
    uint32_t seed, const uint32_t *__restrict__ crc_const,
    uint32_t *__restrict__ out) {
	__shared__ __attribute__((aligned(16))) uint32_t stabs[kCrcConstWords];
	for (int i = threadIdx.x; i < kCrcConstWords; i += kThreads)
		stabs[i] = crc_const[i];
	__syncthreads();
	const uint32_t *T0 = stabs;
	const uint32_t *T1 = stabs + 256;
	const uint32_t *T2 = stabs + 512;
	const uint32_t *T3 = stabs + 768;
	const uint32_t *mats = stabs + kCrcTabWords;

	const int wave = threadIdx.x >> 6;
	const int lane = threadIdx.x & 63;
	const uint32_t seg = block_len >> 6;   /* bytes per lane */

	for (uint64_t blk = (uint64_t)blockIdx.x * 4 + wave; blk < nblocks;
	     blk += (uint64_t)gridDim.x * 4) {
		const uint8_t *p = buf + blk * block_len + (uint32_t)lane * seg;
		uint32_t crc = (lane == 0 ? seed : 0u) ^ 0xFFFFFFFFu;
		for (uint32_t i = 0; i < seg; i += 16) {
			uint4 w = *(const uint4 *)(p + i);
#define LIZEC_CRC_WORD(x)                                              \
	do {                                                               \
		uint32_t u = crc ^ (x);                                        \
		crc = T3[u & 0xff] ^ T2[(u >> 8) & 0xff] ^                     \
		      T1[(u >> 16) & 0xff] ^ T0[u >> 24];                      \
	} while (0)
			LIZEC_CRC_WORD(w.x);
			LIZEC_CRC_WORD(w.y);
			LIZEC_CRC_WORD(w.z);
			LIZEC_CRC_WORD(w.w);
#undef LIZEC_CRC_WORD
		}
		crc ^= 0xFFFFFFFFu;

		/* fold: combine(cl, cr, len_r) = advance(cl, len_r) ^ cr */
		uint32_t len = seg;
#pragma unroll
		for (int s = 0; s < 6; ++s) {
			uint32_t ocrc = __shfl_down(crc, 1 << s, 64);
			uint32_t olen = __shfl_down(len, 1 << s, 64);
			crc = crc_advance(crc, olen, mats) ^ ocrc;
			len += olen;
		}
		if (lane == 0) out[blk] = crc;
	}
}

/* Fast path (block_len % 16384 == 0, e.g. the standard 64 KiB block):
 * one wave per block, C independent CRC chains per lane.  The block is cut
 * into C equal spans; lane l owns segment l of every span, so each lane
 * advances C serial CRC recurrences at once (ILP) and bursts BV*16
 * contiguous bytes per chain per iteration so every fetched line is
 * consumed while resident (the naive strided walk over-fetched HBM 2.2x
 * per the PMC counters).  Slicing-by-8 tables halve the dependent LDS
 * steps; per-lane CRCs fold in a shfl tree using the GF(2) "advance by N
 * zero bytes" matrices of mycrc32_combine (crc.cc:153-224), then lane 0
 * splices the C span CRCs. */
/* Per-wave CRC of one block: C chains x 64 lane segments, BV*16-byte
 * bursts, slicing-by-8; returns the block CRC on every lane (lane 0's
 * value is authoritative). */
template <int C, int BV, int SL = 16>
__device__ uint32_t crc_block_wave(const uint8_t *__restrict__ block,
                                   uint32_t block_len, uint32_t seed,
                                   const uint32_t *T, const uint32_t *mats,
                                   int lane) {
	const uint32_t span = block_len / C;
	const uint32_t seg = span >> 6;
	const uint8_t *base = block + (uint32_t)lane * seg;
	uint32_t crc[C];
#pragma unroll
	for (int c = 0; c < C; ++c)
		crc[c] = ((c == 0 && lane == 0) ? seed : 0u) ^ 0xFFFFFFFFu;
	for (uint32_t i = 0; i < seg; i += 16 * BV) {
		uint4 w[C][BV];
#pragma unroll
		for (int c = 0; c < C; ++c)
#pragma unroll
			for (int q = 0; q < BV; ++q)
				w[c][q] = *(const uint4 *)(base + c * span + i + q * 16);
/* one 16-byte step: 16 independent lookups per chain (slicing-by-16 —
 * the serial dependency is one LDS round trip per 16 bytes) */
#define LIZEC_CRC16(crc, v)                                              \
	do {                                                                 \
		uint32_t u0 = (crc) ^ (v).x, u1 = (v).y, u2 = (v).z, u3 = (v).w; \
		(crc) = T[15 * 256 + (u0 & 0xff)] ^                              \
		        T[14 * 256 + ((u0 >> 8) & 0xff)] ^                       \
		        T[13 * 256 + ((u0 >> 16) & 0xff)] ^                      \
		        T[12 * 256 + (u0 >> 24)] ^                               \
		        T[11 * 256 + (u1 & 0xff)] ^                              \
		        T[10 * 256 + ((u1 >> 8) & 0xff)] ^                       \
		        T[9 * 256 + ((u1 >> 16) & 0xff)] ^                       \
		        T[8 * 256 + (u1 >> 24)] ^                                \
		        T[7 * 256 + (u2 & 0xff)] ^                               \
		        T[6 * 256 + ((u2 >> 8) & 0xff)] ^                        \
		        T[5 * 256 + ((u2 >> 16) & 0xff)] ^                       \
		        T[4 * 256 + (u2 >> 24)] ^                                \
		        T[3 * 256 + (u3 & 0xff)] ^                               \
		        T[2 * 256 + ((u3 >> 8) & 0xff)] ^                        \
		        T[1 * 256 + ((u3 >> 16) & 0xff)] ^ T[u3 >> 24];          \
	} while (0)
#define LIZEC_CRC8(crc, lo, hi)                                          \
	do {                                                                 \
		uint32_t u0 = (crc) ^ (lo), u1 = (hi);                           \
		(crc) = T[7 * 256 + (u0 & 0xff)] ^                               \
		        T[6 * 256 + ((u0 >> 8) & 0xff)] ^                        \
		        T[5 * 256 + ((u0 >> 16) & 0xff)] ^                       \
		        T[4 * 256 + (u0 >> 24)] ^                                \
		        T[3 * 256 + (u1 & 0xff)] ^                               \
		        T[2 * 256 + ((u1 >> 8) & 0xff)] ^                        \
		        T[1 * 256 + ((u1 >> 16) & 0xff)] ^ T[u1 >> 24];          \
	} while (0)
#pragma unroll
		for (int q = 0; q < BV; ++q)
#pragma unroll
			for (int c = 0; c < C; ++c) {
				if (SL == 16) {
					LIZEC_CRC16(crc[c], w[c][q]);
				} else {
					LIZEC_CRC8(crc[c], w[c][q].x, w[c][q].y);
					LIZEC_CRC8(crc[c], w[c][q].z, w[c][q].w);
				}
			}
#undef LIZEC_CRC16
#undef LIZEC_CRC8
	}
#pragma unroll
	for (int c = 0; c < C; ++c) crc[c] ^= 0xFFFFFFFFu;

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
	uint32_t acc = crc[0];
#pragma unroll
	for (int c = 1; c < C; ++c)
		acc = crc_advance(acc, span, mats) ^ crc[c];
	return acc;
}

template <int C, int BV, int SL = 16>
__global__ __launch_bounds__(kThreads) void crc32_blocks_kernel_multi(
    const uint8_t *__restrict__ buf, uint32_t block_len, uint64_t nblocks,
    uint32_t seed, const uint32_t *__restrict__ crc_const,
    uint32_t *__restrict__ out) {
	__shared__ __attribute__((aligned(16))) uint32_t stabs[kCrcConstWords];
	for (int i = threadIdx.x; i < kCrcConstWords; i += kThreads)
		stabs[i] = crc_const[i];
	__syncthreads();
	const uint32_t *T = stabs;
	const uint32_t *mats = stabs + kCrcTabWords;
	const int wave = threadIdx.x >> 6;
	const int lane = threadIdx.x & 63;
	for (uint64_t blk = (uint64_t)blockIdx.x * 4 + wave; blk < nblocks;
	     blk += (uint64_t)gridDim.x * 4) {
		uint32_t crc = crc_block_wave<C, BV, SL>(buf + blk * block_len,
		                                         block_len, seed, T, mats,
		                                         lane);
		if (lane == 0) out[blk] = crc;
	}
}


/* LDS-free folding CRC kernel (crc_fold.h): the hot loop is pure VALU;
 * LDS holds only the epilogue byte table T0 and the combine matrices. */
constexpr int kCrcFoldLdsWords = 256 + kCrcMatCount * 32;

template <int C, int NACC, bool AL16 = true, bool NT = false,
          bool PF = false, int BVO = 0>
__global__ __launch_bounds__(kThreads) void crc32_blocks_kernel_fold(
    const uint8_t *__restrict__ buf, uint32_t block_len, uint64_t nblocks,
    uint32_t seed, const uint32_t *__restrict__ crc_const,
    uint32_t *__restrict__ out) {
	__shared__ __attribute__((aligned(16))) uint32_t stabs[kCrcFoldLdsWords];
	for (int i = threadIdx.x; i < 256; i += kThreads)
		stabs[i] = crc_const[i];
	for (int i = threadIdx.x; i < kCrcMatCount * 32; i += kThreads)
		stabs[256 + i] = crc_const[kCrcTabWords + i];
	__syncthreads();
	const int wave = threadIdx.x >> 6;
	const int lane = threadIdx.x & 63;
	for (uint64_t blk = (uint64_t)blockIdx.x * 4 + wave; blk < nblocks;
	     blk += (uint64_t)gridDim.x * 4) {
		uint32_t crc = crc_block_wave_fold<C, NACC, AL16, NT, PF, BVO>(
		    buf + blk * block_len, block_len, seed, stabs, stabs + 256, lane);
		if (lane == 0) out[blk] = crc;
	}
}

/* Batched chunk scrub / image-CRC fill.
 *
 * WRITE=false: hdd_int_test semantics (hddspacemgr.cc:2148-2212): for
 * every 64 KiB block of every chunk-part image, compare mycrc32(0, block,
 * MFSBLOCKSIZE) against the stored CRC array entry (big-endian u32 at
 * crc_off + crc_stride*b, cf. get32bit at hddspacemgr.cc:2183 and
 * chunk.cc:183-188).  status[c] collects the FIRST damaged block index
 * via atomicMin (host pre-fills INT32_MAX = clean).
 *
 * WRITE=true: image assembly for the replication pipeline
 * (chunk_replicator.cc:189 + chunk.cc:126-188): compute the same CRCs and
 * STORE them big-endian into the image's CRC array (status unused). */
template <bool WRITE>
__global__ __launch_bounds__(kThreads) void scrub_chunks_kernel(
    const uint64_t *__restrict__ chunk_dptrs,
    const uint32_t *__restrict__ data_offs,
    const uint32_t *__restrict__ crc_offs,
    const uint32_t *__restrict__ block_counts, uint32_t nchunks,
    uint32_t max_blocks, uint32_t block_stride, uint32_t crc_stride,
    const uint32_t *__restrict__ crc_const, int32_t *__restrict__ status) {
	__shared__ __attribute__((aligned(16))) uint32_t stabs[kCrcFoldLdsWords];
	for (int i = threadIdx.x; i < 256; i += kThreads)
		stabs[i] = crc_const[i];
	for (int i = threadIdx.x; i < kCrcMatCount * 32; i += kThreads)
		stabs[256 + i] = crc_const[kCrcTabWords + i];
	__syncthreads();
	const uint32_t *T0 = stabs;
	const uint32_t *mats = stabs + 256;
	const int wave = threadIdx.x >> 6;
	const int lane = threadIdx.x & 63;
	const uint64_t total = (uint64_t)nchunks * max_blocks;
	for (uint64_t flat = (uint64_t)blockIdx.x * 4 + wave; flat < total;
	     flat += (uint64_t)gridDim.x * 4) {
		uint32_t c = (uint32_t)(flat / max_blocks);
		uint32_t b = (uint32_t)(flat - (uint64_t)c * max_blocks);
		if (b >= block_counts[c]) continue;
		const uint8_t *img = (const uint8_t *)chunk_dptrs[c];
		const uint8_t *blockp = img + data_offs[c] + b * block_stride;
		/* INTERLEAVED-format blocks sit at 4 mod 16 — use the dword-load
		 * instantiation there (uint4 loads would be misaligned UB) */
		uint32_t crc =
		    (((uintptr_t)blockp & 15) == 0)
		        ? crc_block_wave_fold<1, 1, true>(blockp, 65536u, 0u, T0,
		                                          mats, lane)
		        : crc_block_wave_fold<1, 1, false>(blockp, 65536u, 0u, T0,
		                                           mats, lane);
		if (lane == 0) {
			uint8_t *p = (uint8_t *)img + crc_offs[c] + crc_stride * b;
			if (WRITE) {
				p[0] = (uint8_t)(crc >> 24);
				p[1] = (uint8_t)(crc >> 16);
				p[2] = (uint8_t)(crc >> 8);
				p[3] = (uint8_t)crc;
			} else {
				uint32_t stored = ((uint32_t)p[0] << 24) |
				                  ((uint32_t)p[1] << 16) |
				                  ((uint32_t)p[2] << 8) | p[3];
				if (stored != crc) atomicMin(&status[c], (int32_t)b);
			}
		}
	}
}

/* ------------------------------------------------------------------ */
/* Engine + plans                                                     */
/* ------------------------------------------------------------------ */

/* Per-stream call scratch.  The batch entry points upload tables/pointer
 * arrays into device scratch; keying the scratch by stream makes
 * concurrent calls on different streams of one engine race-free (the
 * chunkserver's bgjobs threading model is exactly multi-threaded,
 * network_main_thread.cc:230-234).  Within one stream, device scratch
 * reuse is stream-ordered; the pinned host staging buffer is guarded by
 * an event recorded after the H2D copies are enqueued. */
struct lizec_stream_ctx {
	uint8_t *d_gftbls = nullptr;    /* 16*32*32 (packed quarter-LUT) */
	uint64_t *d_ptrs = nullptr;     /* grows */
	size_t ptrs_cap = 0;            /* in elements */
	uint8_t *h_staging = nullptr;   /* pinned: tables + pointer arrays */
	size_t h_cap = 0;               /* staging bytes */
	hipEvent_t uploaded = nullptr;  /* staging consumed up to here */
	bool ev_recorded = false;
};

struct lizec_engine {
	int device;
	hipStream_t stream;        /* default stream for calls passing NULL */
	uint32_t *d_crc_const;     /* kCrcConstWords */
	std::mutex mu;             /* guards ctxs */
	std::unordered_map<void *, lizec_stream_ctx> ctxs;
	/* CRC fold shape, autotuned on the first large batch: -1 undecided,
	 * 0 = no-prefetch (5 waves/SIMD, robust), 1 = burst-prefetch
	 * (2 waves; +6-7% on most boxes, -20% on some — profiles/ROUND2.md) */
	std::atomic<int> crc_shape{-1};
};

static void ctx_free(lizec_stream_ctx &c) {
	(void)hipFree(c.d_gftbls);
	(void)hipFree(c.d_ptrs);
	(void)hipHostFree(c.h_staging);
	if (c.uploaded) (void)hipEventDestroy(c.uploaded);
	c = lizec_stream_ctx();
}

/* Get (or create) the scratch for `s`, sized for `ptrs` pointer slots and
 * `staging` staging bytes; waits out any still-pending staging use.
 * Thread contract: concurrent calls must use distinct streams (each
 * stream's scratch is touched by one call at a time — the map lookup is
 * the only globally locked step, so streams never stall each other). */
static int ctx_acquire(lizec_engine *e, hipStream_t s, size_t ptrs,
                       size_t staging, lizec_stream_ctx **out) {
	lizec_stream_ctx *cp;
	{
		std::lock_guard<std::mutex> lk(e->mu);
		cp = &e->ctxs[(void *)s];   /* element refs survive rehash */
	}
	lizec_stream_ctx &c = *cp;
	if (!c.uploaded) {
		if (hipEventCreateWithFlags(&c.uploaded, hipEventDisableTiming) !=
		    hipSuccess)
			return LIZEC_EHIP;
	}
	if (c.ev_recorded) {
		/* previous call on this stream may still be reading h_staging (and
		 * a grow below would free device scratch it reads) — wait it out */
		LIZEC_CHECK(hipEventSynchronize(c.uploaded));
		c.ev_recorded = false;
	}
	if (!c.d_gftbls)
		LIZEC_CHECK(hipMalloc(&c.d_gftbls, (size_t)kECTblBytes * 32 * 32));
	if (c.ptrs_cap < ptrs) {
		size_t cap = c.ptrs_cap ? c.ptrs_cap : (1 << 16);
		while (cap < ptrs) cap *= 2;
		(void)hipFree(c.d_ptrs);
		c.d_ptrs = nullptr;
		c.ptrs_cap = 0;
		LIZEC_CHECK(hipMalloc(&c.d_ptrs, cap * sizeof(uint64_t)));
		c.ptrs_cap = cap;
	}
	size_t want = staging + (size_t)kECTblBytes * 32 * 32;
	if (c.h_cap < want) {
		size_t cap = c.h_cap ? c.h_cap : (1 << 20);
		while (cap < want) cap *= 2;
		(void)hipHostFree(c.h_staging);
		c.h_staging = nullptr;
		c.h_cap = 0;
		LIZEC_CHECK(hipHostMalloc(&c.h_staging, cap));
		c.h_cap = cap;
	}
	*out = &c;
	return LIZEC_OK;
}

/* A plan = a prepared batch: device-resident tables + pointer arrays.
 * Mirrors the reference's cached-matrix + read-plan structure
 * (reed_solomon.h:194-198, read_plan.h): build once per (erasure pattern,
 * batch), run many times with zero host->device traffic. */
struct lizec_plan {
	lizec_engine *e;
	uint64_t part_len;
	int srcs, dests, num_stripes;
	uint8_t *d_tbls;
	uint64_t *d_src;
	uint64_t *d_dst;
};

extern "C" int lizec_gpu_count(void) {
	int n = 0;
	if (hipGetDeviceCount(&n) != hipSuccess) return 0;
	return n;
}

/* Build the CRC constant block on the host (product-side tables; bit-exact
 * semantics enforced by tests vs the oracle). */
static void build_crc_const(uint32_t *w) {
	for (uint32_t i = 0; i < 256; ++i) {
		uint32_t c = i;
		for (int b = 0; b < 8; ++b)
			c = (c & 1) ? (0xEDB88320u ^ (c >> 1)) : (c >> 1);
		w[i] = c;
	}
	for (uint32_t i = 0; i < 256; ++i)
		for (int t = 1; t < 16; ++t) {
			uint32_t c = w[(t - 1) * 256 + i];
			w[t * 256 + i] = w[c & 0xff] ^ (c >> 8);
		}
	/* advance matrices: M_0 = 1 zero byte; M_{i+1} = M_i^2 */
	uint32_t *mats = w + kCrcTabWords;
	uint32_t odd[32], even[32];
	odd[0] = 0xEDB88320u;                 /* 1 zero BIT */
	for (int i = 1; i < 32; ++i) odd[i] = 1u << (i - 1);
	auto times = [](const uint32_t *m, uint32_t v) {
		uint32_t s = 0;
		for (int i = 0; v; v >>= 1, ++i)
			if (v & 1) s ^= m[i];
		return s;
	};
	auto square = [&](uint32_t *sq, const uint32_t *m) {
		for (int i = 0; i < 32; ++i) sq[i] = times(m, m[i]);
	};
	square(even, odd);    /* 2 bits */
	square(odd, even);    /* 4 bits */
	square(even, odd);    /* 8 bits = 1 byte -> M_0 */
	memcpy(mats, even, 32 * 4);
	for (int i = 1; i < kCrcMatCount; ++i) {
		square(odd, (uint32_t *)(mats + (i - 1) * 32));
		memcpy(mats + i * 32, odd, 32 * 4);
	}
}

extern "C" int lizec_engine_create(lizec_engine **out, int device_id) {
	*out = nullptr;
	int n = lizec_gpu_count();
	if (n <= 0 || device_id >= n) return LIZEC_ENOGPU;
	LIZEC_CHECK(hipSetDevice(device_id));
	lizec_engine *e = new lizec_engine();
	e->device = device_id;
	if (hipStreamCreate(&e->stream) != hipSuccess ||
	    hipMalloc(&e->d_crc_const, kCrcConstWords * 4) != hipSuccess) {
		lizec_engine_destroy(e);   /* frees whatever was allocated */
		return LIZEC_ENOMEM;
	}
	uint32_t *host_const = (uint32_t *)malloc(kCrcConstWords * 4);
	build_crc_const(host_const);
	hipError_t err = hipMemcpy(e->d_crc_const, host_const, kCrcConstWords * 4,
	                           hipMemcpyHostToDevice);
	free(host_const);
	if (err != hipSuccess) {
		lizec_engine_destroy(e);
		return LIZEC_EHIP;
	}
	*out = e;
	return LIZEC_OK;
}

extern "C" void lizec_engine_destroy(lizec_engine *e) {
	if (!e) return;
	for (auto &kv : e->ctxs) ctx_free(kv.second);
	(void)hipFree(e->d_crc_const);
	(void)hipStreamDestroy(e->stream);
	delete e;
}

extern "C" int lizec_engine_sync(lizec_engine *e) {
	LIZEC_CHECK(hipSetDevice(e->device));
	LIZEC_CHECK(hipStreamSynchronize(e->stream));
	return LIZEC_OK;
}

static int check_batch_args(uint64_t part_len, int srcs, int dests,
                            int num_stripes) {
	if (srcs < 1 || srcs > 32 || dests < 1 || dests > 32 || num_stripes < 1)
		return LIZEC_EINVAL;
	if (part_len == 0 || (part_len & 15) || part_len > (uint64_t)1 << 31)
		return LIZEC_EINVAL;
	return LIZEC_OK;
}

template <int D>
static void launch_ec(uint32_t part_len, int srcs, int dest_base,
                      const uint8_t *d_tbls, const uint64_t *d_src,
                      const uint64_t *d_dst, int dests_total,
                      uint32_t tiles_per_part_unused, uint32_t nstripes,
                      hipStream_t s) {
	constexpr int CH = ec_chunks_for(D);
	uint32_t tile_bytes = kChunkBytes * CH;
	uint32_t tiles_per_part = (part_len + tile_bytes - 1) / tile_bytes;
	uint32_t total_tiles = tiles_per_part * nstripes;
	uint32_t grid = total_tiles < 1048576u ? total_tiles : 1048576u;   /* exact grid (1 tile/block) measured +0.7% */
	size_t lds = (size_t)D * srcs * kECTblBytes;
	(void)tiles_per_part_unused;
	hipLaunchKernelGGL(HIP_KERNEL_NAME(ec_encode_kernel<D, CH, kECSwz, kECNtStore, kECNtLoad, false, false, true>),
	                   dim3(grid), dim3(kThreads), lds, s,
	                   part_len, srcs, dest_base, d_tbls, d_src, d_dst,
	                   dests_total, tiles_per_part, total_tiles);
}

static int run_batch(uint64_t part_len, int srcs, int dests,
                     const uint8_t *d_tbls, const uint64_t *d_src,
                     const uint64_t *d_dst, uint32_t num_stripes,
                     hipStream_t s) {
	/* One pass computes up to kMaxDestsPerPass destinations; wider m splits
	 * into passes (each pass re-reads the sources, so fewer passes wins on
	 * this HBM-bound kernel — D<=8 measured best, profiles/). */
	for (int base = 0; base < dests;) {
		int d = dests - base;
		if (d > 8) d = 8;
		switch (d) {
		case 1: launch_ec<1>((uint32_t)part_len, srcs, base, d_tbls, d_src,
		                     d_dst, dests, 0, num_stripes, s);
			break;
		case 2: launch_ec<2>((uint32_t)part_len, srcs, base, d_tbls, d_src,
		                     d_dst, dests, 0, num_stripes, s);
			break;
		case 3: launch_ec<3>((uint32_t)part_len, srcs, base, d_tbls, d_src,
		                     d_dst, dests, 0, num_stripes, s);
			break;
		case 4: launch_ec<4>((uint32_t)part_len, srcs, base, d_tbls, d_src,
		                     d_dst, dests, 0, num_stripes, s);
			break;
		case 5: launch_ec<5>((uint32_t)part_len, srcs, base, d_tbls, d_src,
		                     d_dst, dests, 0, num_stripes, s);
			break;
		case 6: launch_ec<6>((uint32_t)part_len, srcs, base, d_tbls, d_src,
		                     d_dst, dests, 0, num_stripes, s);
			break;
		case 7: launch_ec<7>((uint32_t)part_len, srcs, base, d_tbls, d_src,
		                     d_dst, dests, 0, num_stripes, s);
			break;
		default: launch_ec<8>((uint32_t)part_len, srcs, base, d_tbls, d_src,
		                      d_dst, dests, 0, num_stripes, s);
			break;
		}
		base += d;
	}
	LIZEC_CHECK(hipGetLastError());
	return LIZEC_OK;
}

extern "C" int lizec_ec_encode_batch(lizec_engine *e, uint64_t part_len,
                                     int srcs, int dests,
                                     const uint8_t *gftbls,
                                     const uint64_t *src_dptrs,
                                     const uint64_t *dst_dptrs,
                                     int num_stripes, void *stream) {
	if (!e) return LIZEC_EINVAL;
	int r = check_batch_args(part_len, srcs, dests, num_stripes);
	if (r != LIZEC_OK) return r;
	hipStream_t s = stream ? (hipStream_t)stream : e->stream;
	LIZEC_CHECK(hipSetDevice(e->device));

	size_t nsrc = (size_t)num_stripes * srcs;
	size_t ndst = (size_t)num_stripes * dests;
	lizec_stream_ctx *c;
	r = ctx_acquire(e, s, nsrc + ndst, (nsrc + ndst) * 8, &c);
	if (r != LIZEC_OK) return r;
	uint64_t *d_src = c->d_ptrs;
	uint64_t *d_dst = c->d_ptrs + nsrc;
	/* stage tables + pointer arrays through this stream's pinned buffer
	 * (ctx_acquire waited until any previous copies from it finished) */
	uint8_t *tstage = c->h_staging;
	uint64_t *pstage = (uint64_t *)(c->h_staging + (size_t)kECTblBytes * 32 * 32);
	for (int i = 0; i < srcs * dests; ++i)
		pack_mixed_radix(gftbls + (size_t)i * 32, tstage + (size_t)i * kECTblBytes);
	memcpy(pstage, src_dptrs, nsrc * 8);
	memcpy(pstage + nsrc, dst_dptrs, ndst * 8);
	LIZEC_CHECK(hipMemcpyAsync(c->d_gftbls, tstage, (size_t)kECTblBytes * srcs * dests,
	                           hipMemcpyHostToDevice, s));
	LIZEC_CHECK(hipMemcpyAsync(d_src, pstage, (nsrc + ndst) * 8,
	                           hipMemcpyHostToDevice, s));
	LIZEC_CHECK(hipEventRecord(c->uploaded, s));
	c->ev_recorded = true;

	return run_batch(part_len, srcs, dests, c->d_gftbls, d_src, d_dst,
	                 (uint32_t)num_stripes, s);
}

extern "C" int lizec_ec_plan_create(lizec_engine *e, uint64_t part_len,
                                    int srcs, int dests,
                                    const uint8_t *gftbls,
                                    const uint64_t *src_dptrs,
                                    const uint64_t *dst_dptrs,
                                    int num_stripes, lizec_plan **out) {
	*out = nullptr;
	if (!e) return LIZEC_EINVAL;
	int r = check_batch_args(part_len, srcs, dests, num_stripes);
	if (r != LIZEC_OK) return r;
	LIZEC_CHECK(hipSetDevice(e->device));
	lizec_plan *p = new lizec_plan();
	p->e = e;
	p->part_len = part_len;
	p->srcs = srcs;
	p->dests = dests;
	p->num_stripes = num_stripes;
	size_t nsrc = (size_t)num_stripes * srcs;
	size_t ndst = (size_t)num_stripes * dests;
	if (hipMalloc(&p->d_tbls, (size_t)kECTblBytes * srcs * dests) != hipSuccess ||
	    hipMalloc(&p->d_src, nsrc * 8) != hipSuccess ||
	    hipMalloc(&p->d_dst, ndst * 8) != hipSuccess) {
		lizec_ec_plan_destroy(p);
		return LIZEC_ENOMEM;
	}
	uint8_t packed[kECTblBytes * 32 * 32];
	for (int i = 0; i < srcs * dests; ++i)
		pack_mixed_radix(gftbls + (size_t)i * 32, packed + (size_t)i * kECTblBytes);
	if (hipMemcpy(p->d_tbls, packed, (size_t)kECTblBytes * srcs * dests,
	              hipMemcpyHostToDevice) != hipSuccess ||
	    hipMemcpy(p->d_src, src_dptrs, nsrc * 8,
	              hipMemcpyHostToDevice) != hipSuccess ||
	    hipMemcpy(p->d_dst, dst_dptrs, ndst * 8,
	              hipMemcpyHostToDevice) != hipSuccess) {
		lizec_ec_plan_destroy(p);
		return LIZEC_EHIP;
	}
	*out = p;
	return LIZEC_OK;
}

extern "C" int lizec_ec_plan_run(lizec_plan *p, void *stream) {
	if (!p) return LIZEC_EINVAL;
	hipStream_t s = stream ? (hipStream_t)stream : p->e->stream;
	LIZEC_CHECK(hipSetDevice(p->e->device));
	return run_batch(p->part_len, p->srcs, p->dests, p->d_tbls, p->d_src,
	                 p->d_dst, (uint32_t)p->num_stripes, s);
}

extern "C" void lizec_ec_plan_destroy(lizec_plan *p) {
	if (!p) return;
	(void)hipFree(p->d_tbls);
	(void)hipFree(p->d_src);
	(void)hipFree(p->d_dst);
	delete p;
}

extern "C" int lizec_crc32_batch(lizec_engine *e, const void *dev_buf,
                                 uint32_t block_len, uint64_t nblocks,
                                 uint32_t seed, uint32_t *dev_crcs_out,
                                 void *stream) {
	if (!e || !dev_buf || !dev_crcs_out || nblocks < 1) return LIZEC_EINVAL;
	/* one wave per block, 64 equal lane segments, 16B vector loads */
	if (block_len == 0 || (block_len & 1023)) return LIZEC_EINVAL;
	hipStream_t s = stream ? (hipStream_t)stream : e->stream;
	LIZEC_CHECK(hipSetDevice(e->device));
	uint64_t groups = (nblocks + 3) / 4;
	uint32_t grid = (uint32_t)(groups < 131072 ? groups : 131072);
	const char *ch = getenv("LIZEC_CRC_CHAINS");   /* A/B hooks */
	const char *sl = getenv("LIZEC_CRC_SLICE");
	const char *im = getenv("LIZEC_CRC_IMPL");
	const char *fn = getenv("LIZEC_CRC_FOLD_NACC");
	const char *nt = getenv("LIZEC_CRC_NT");
	const char *pf = getenv("LIZEC_CRC_PF");
	/* fold C=1 NACC=1 measured best (r2b: 5112 GB/s = 0.64 of spec peak
	 * vs table 3511; NT loads defeat the L1 line-burst reuse, -57%) */
	int chains = ch ? atoi(ch) : 1;
	int slice = sl ? atoi(sl) : 8;   /* slice-16 measured -16% within-box (profiles) */
	bool fold = !(im && strcmp(im, "table") == 0);
	int nacc = fn ? atoi(fn) : 1;
	bool ntld = nt && atoi(nt) != 0;
	/* burst prefetch measured box-DEPENDENT: +7% on one box (r2d,
	 * 5147 GB/s) but -20% on another (r2e, 4154 vs 5197 no-PF) — its
	 * 2-waves/SIMD occupancy is fragile, the 5-wave no-PF shape is
	 * robust across boxes.  Default off; env opt-in for A/B. */
	bool pfld = pf && atoi(pf) != 0;
	bool al16 = (((uintptr_t)dev_buf | block_len) & 15) == 0;
	/* carry-less-folding path (default): block must split into C spans of
	 * whole 64-lane x BV*16-byte bursts (BV=8) */
	if (fold && block_len % ((uint32_t)chains * 64 * 16 * 8) == 0 &&
	    (chains == 1 || chains == 2)) {
		const uint8_t *b = (const uint8_t *)dev_buf;
#define LIZEC_LAUNCH_FOLD(C, N, A, NT)                                       \
	hipLaunchKernelGGL(HIP_KERNEL_NAME(crc32_blocks_kernel_fold<C, N, A, NT>), \
	                   dim3(grid), dim3(kThreads), 0, s, b, block_len,       \
	                   nblocks, seed, e->d_crc_const, dev_crcs_out)
		if (!al16) {
			if (chains == 2) LIZEC_LAUNCH_FOLD(2, 2, false, false);
			else LIZEC_LAUNCH_FOLD(1, 1, false, false);
		} else if (chains == 2) {
			if (nacc == 1) LIZEC_LAUNCH_FOLD(2, 1, true, false);
			else LIZEC_LAUNCH_FOLD(2, 2, true, false);
		} else if (pfld) {
			const char *bv = getenv("LIZEC_CRC_BV");
			if (bv && atoi(bv) == 4)
				hipLaunchKernelGGL(HIP_KERNEL_NAME(crc32_blocks_kernel_fold<1, 1, true, false, true, 4>),
				                   dim3(grid), dim3(kThreads), 0, s, b,
				                   block_len, nblocks, seed, e->d_crc_const,
				                   dev_crcs_out);
			else
				hipLaunchKernelGGL(HIP_KERNEL_NAME(crc32_blocks_kernel_fold<1, 1, true, false, true>),
				                   dim3(grid), dim3(kThreads), 0, s, b,
				                   block_len, nblocks, seed, e->d_crc_const,
				                   dev_crcs_out);
		} else if (ntld) {
			if (nacc == 1) LIZEC_LAUNCH_FOLD(1, 1, true, true);
			else if (nacc == 4) LIZEC_LAUNCH_FOLD(1, 4, true, true);
			else LIZEC_LAUNCH_FOLD(1, 2, true, true);
		} else {
			const char *bv = getenv("LIZEC_CRC_BV");
			const char *at = getenv("LIZEC_CRC_AUTOTUNE");
			bool tune = !(at && atoi(at) == 0);
			if (bv && atoi(bv) == 4 && nacc == 1) {
				hipLaunchKernelGGL(HIP_KERNEL_NAME(crc32_blocks_kernel_fold<1, 1, true, false, false, 4>),
				                   dim3(grid), dim3(kThreads), 0, s, b,
				                   block_len, nblocks, seed, e->d_crc_const,
				                   dev_crcs_out);
			} else if (nacc == 4) {
				if (bv && atoi(bv) == 4)
					hipLaunchKernelGGL(HIP_KERNEL_NAME(crc32_blocks_kernel_fold<1, 4, true, false, false, 4>),
					                   dim3(grid), dim3(kThreads), 0, s, b,
					                   block_len, nblocks, seed,
					                   e->d_crc_const, dev_crcs_out);
				else
					LIZEC_LAUNCH_FOLD(1, 4, true, false);
			} else if (nacc == 2) {
				if (bv && atoi(bv) == 4)
					hipLaunchKernelGGL(HIP_KERNEL_NAME(crc32_blocks_kernel_fold<1, 2, true, false, false, 4>),
					                   dim3(grid), dim3(kThreads), 0, s, b,
					                   block_len, nblocks, seed,
					                   e->d_crc_const, dev_crcs_out);
				else
					LIZEC_LAUNCH_FOLD(1, 2, true, false);
			} else {
				int shape = e->crc_shape.load(std::memory_order_relaxed);
				if (shape < 0 && tune && nblocks >= 4096) {
					/* Autotune once per engine: time both shapes on this
					 * batch.  Both kernels emit the correct CRCs, so the
					 * probe costs one extra pass and the call stays
					 * correct even if timing fails. */
					hipEvent_t ev[3];
					int made = 0;
					for (; made < 3; ++made)
						if (hipEventCreate(&ev[made]) != hipSuccess) break;
					if (made == 3) {
						(void)hipEventRecord(ev[0], s);
						LIZEC_LAUNCH_FOLD(1, 1, true, false);
						(void)hipEventRecord(ev[1], s);
						hipLaunchKernelGGL(HIP_KERNEL_NAME(crc32_blocks_kernel_fold<1, 1, true, false, true>),
						                   dim3(grid), dim3(kThreads), 0, s,
						                   b, block_len, nblocks, seed,
						                   e->d_crc_const, dev_crcs_out);
						(void)hipEventRecord(ev[2], s);
						if (hipEventSynchronize(ev[2]) == hipSuccess) {
							float t0 = 0.f, t1 = 0.f;
							(void)hipEventElapsedTime(&t0, ev[0], ev[1]);
							(void)hipEventElapsedTime(&t1, ev[1], ev[2]);
							shape = (t1 > 0.f && t1 < t0) ? 1 : 0;
							e->crc_shape.store(shape,
							                   std::memory_order_relaxed);
						}
					} else {
						LIZEC_LAUNCH_FOLD(1, 1, true, false);
					}
					for (int i = 0; i < made; ++i)
						(void)hipEventDestroy(ev[i]);
					LIZEC_CHECK(hipGetLastError());
					return LIZEC_OK;
				}
				if (shape == 1)
					hipLaunchKernelGGL(HIP_KERNEL_NAME(crc32_blocks_kernel_fold<1, 1, true, false, true>),
					                   dim3(grid), dim3(kThreads), 0, s, b,
					                   block_len, nblocks, seed,
					                   e->d_crc_const, dev_crcs_out);
				else
					LIZEC_LAUNCH_FOLD(1, 1, true, false);
			}
		}
#undef LIZEC_LAUNCH_FOLD
		LIZEC_CHECK(hipGetLastError());
		return LIZEC_OK;
	}
	if (block_len % 32768 == 0 && chains >= 4)
		hipLaunchKernelGGL(HIP_KERNEL_NAME(crc32_blocks_kernel_multi<4, 8>),
		                   dim3(grid), dim3(kThreads), 0, s,
		                   (const uint8_t *)dev_buf, block_len, nblocks,
		                   seed, e->d_crc_const, dev_crcs_out);
	else if (block_len % 16384 == 0 && slice == 8)
		hipLaunchKernelGGL(HIP_KERNEL_NAME(crc32_blocks_kernel_multi<2, 8, 8>),
		                   dim3(grid), dim3(kThreads), 0, s,
		                   (const uint8_t *)dev_buf, block_len, nblocks,
		                   seed, e->d_crc_const, dev_crcs_out);
	else if (block_len % 16384 == 0)
		hipLaunchKernelGGL(HIP_KERNEL_NAME(crc32_blocks_kernel_multi<2, 8>),
		                   dim3(grid), dim3(kThreads), 0, s,
		                   (const uint8_t *)dev_buf, block_len, nblocks,
		                   seed, e->d_crc_const, dev_crcs_out);
	else
		hipLaunchKernelGGL(crc32_blocks_kernel, dim3(grid), dim3(kThreads),
		                   0, s, (const uint8_t *)dev_buf, block_len, nblocks,
		                   seed, e->d_crc_const, dev_crcs_out);
	LIZEC_CHECK(hipGetLastError());
	return LIZEC_OK;
}

extern "C" int lizec_scrub_batch_strided(
    lizec_engine *e, const uint64_t *chunk_dptrs, const uint32_t *data_offs,
    const uint32_t *crc_offs, const uint32_t *block_counts, int nchunks,
    uint32_t block_stride, uint32_t crc_stride, int32_t *dev_status_out,
    void *stream) {
	if (!e || !chunk_dptrs || nchunks < 1 || !dev_status_out)
		return LIZEC_EINVAL;
	hipStream_t s = stream ? (hipStream_t)stream : e->stream;
	LIZEC_CHECK(hipSetDevice(e->device));
	/* upload per-chunk tables into stream scratch (u64 slots reused):
	 * [dptrs u64 x n][doffs u32 x n][coffs][counts] */
	size_t words = (size_t)nchunks;               /* dptrs */
	size_t meta = words + (3 * words * 4 + 7) / 8;
	size_t bytes = meta * 8;
	lizec_stream_ctx *c;
	int r = ctx_acquire(e, s, meta + 8, bytes, &c);
	if (r != LIZEC_OK) return r;
	uint64_t *d_ptrs = c->d_ptrs;
	uint32_t *d_doffs = (uint32_t *)(d_ptrs + nchunks);
	uint32_t *d_coffs = d_doffs + nchunks;
	uint32_t *d_counts = d_coffs + nchunks;
	uint8_t *st = c->h_staging + (size_t)kECTblBytes * 32 * 32;
	memcpy(st, chunk_dptrs, (size_t)nchunks * 8);
	memcpy(st + (size_t)nchunks * 8, data_offs, (size_t)nchunks * 4);
	memcpy(st + (size_t)nchunks * 12, crc_offs, (size_t)nchunks * 4);
	memcpy(st + (size_t)nchunks * 16, block_counts, (size_t)nchunks * 4);
	LIZEC_CHECK(hipMemcpyAsync(d_ptrs, st, (size_t)nchunks * 20,
	                           hipMemcpyHostToDevice, s));
	LIZEC_CHECK(hipEventRecord(c->uploaded, s));
	c->ev_recorded = true;
	uint32_t max_blocks = 0;
	for (int i = 0; i < nchunks; ++i)
		if (block_counts[i] > max_blocks) max_blocks = block_counts[i];
	if (max_blocks == 0) return LIZEC_EINVAL;
	/* INT32_MAX sentinel = clean */
	LIZEC_CHECK(hipMemsetD32Async((hipDeviceptr_t)dev_status_out, 0x7FFFFFFF,
	                              nchunks, s));
	uint64_t total = (uint64_t)nchunks * max_blocks;
	uint64_t groups = (total + 3) / 4;
	uint32_t grid = (uint32_t)(groups < 131072 ? groups : 131072);
	hipLaunchKernelGGL(HIP_KERNEL_NAME(scrub_chunks_kernel<false>),
	                   dim3(grid), dim3(kThreads), 0, s,
	                   d_ptrs, d_doffs, d_coffs, d_counts, (uint32_t)nchunks,
	                   max_blocks, block_stride, crc_stride, e->d_crc_const,
	                   dev_status_out);
	LIZEC_CHECK(hipGetLastError());
	return LIZEC_OK;
}

/* Launch the WRITE variant on already-device-resident meta arrays (used by
 * the replication pipeline, which stages meta itself). */
static int image_crc_launch(lizec_engine *e, const uint64_t *d_ptrs,
                            const uint32_t *d_doffs, const uint32_t *d_coffs,
                            const uint32_t *d_counts, int nimages,
                            uint32_t max_blocks, hipStream_t s) {
	uint64_t total = (uint64_t)nimages * max_blocks;
	uint64_t groups = (total + 3) / 4;
	uint32_t grid = (uint32_t)(groups < 131072 ? groups : 131072);
	hipLaunchKernelGGL(HIP_KERNEL_NAME(scrub_chunks_kernel<true>),
	                   dim3(grid), dim3(kThreads), 0, s, d_ptrs, d_doffs,
	                   d_coffs, d_counts, (uint32_t)nimages, max_blocks,
	                   65536u, 4u, e->d_crc_const, nullptr);
	LIZEC_CHECK(hipGetLastError());
	return LIZEC_OK;
}

/* MooseFS layout: 64 KiB blocks after the header, 4-byte CRC array. */
extern "C" int lizec_scrub_batch(lizec_engine *e, const uint64_t *chunk_dptrs,
                                 const uint32_t *data_offs,
                                 const uint32_t *crc_offs,
                                 const uint32_t *block_counts, int nchunks,
                                 int32_t *dev_status_out, void *stream) {
	return lizec_scrub_batch_strided(e, chunk_dptrs, data_offs, crc_offs,
	                                 block_counts, nchunks, 65536u, 4u,
	                                 dev_status_out, stream);
}

/* ------------------------------------------------------------------ */
/* Streaming replication pipeline                  */
/* ------------------------------------------------------------------ */

extern "C" int lizec_host_alloc(void **ptr, uint64_t bytes) {
	*ptr = nullptr;
	LIZEC_CHECK(hipHostMalloc(ptr, bytes));
	return LIZEC_OK;
}

extern "C" void lizec_host_free(void *p) {
	(void)hipHostFree(p);
}

/* One double-buffer slot of the replication pipeline. */
struct repl_slot {
	hipStream_t stream = nullptr;
	hipEvent_t done = nullptr;
	uint8_t *d_in = nullptr;    /* B * ic * part_len           */
	uint8_t *d_img = nullptr;   /* B * oc * img_bytes          */
	uint64_t *d_meta = nullptr; /* ptr tables + image meta     */
	uint8_t *h_stage = nullptr; /* pinned: meta + signatures   */
	bool used = false;
};

static void repl_slot_free(repl_slot &sl) {
	(void)hipFree(sl.d_in);
	(void)hipFree(sl.d_img);
	(void)hipFree(sl.d_meta);
	(void)hipHostFree(sl.h_stage);
	if (sl.done) (void)hipEventDestroy(sl.done);
	if (sl.stream) (void)hipStreamDestroy(sl.stream);
	sl = repl_slot();
}

/* The ChunkReplicator::replicate loop (chunk_replicator.cc:139-196) as a
 * host-to-host streaming pipeline: pull surviving parts from host memory
 * (the reference pulls them from peer sockets), recover the erased parts
 * on-GPU, CRC every recovered 64 KiB block (chunk_replicator.cc:189), and
 * emit complete MooseFS part images (chunk.cc:126-188 geometry: caller's
 * signature bytes at 0, BE CRC array at crc_off, blocks at header_size)
 * into caller host buffers — H2D, compute and D2H double-buffered on two
 * streams.  Use lizec_host_alloc'd (pinned) buffers for real overlap;
 * pageable buffers degrade to synchronous copies but stay correct.
 *
 * host_src:  nchunks*ic host addresses (surviving part bytes, part_len each)
 * gftbls:    32*ic*oc recover tables (lizec_rs_tables)
 * sigs:      nchunks*oc signatures, sig_len bytes each
 * host_dst:  nchunks*oc host addresses (header_size + part_len each)
 * sub_batch: chunks per pipeline stage (0 = auto) */
extern "C" int lizec_replicate_run(lizec_engine *e, uint64_t part_len,
                                   int ic, int oc, const uint8_t *gftbls,
                                   const uint64_t *host_src,
                                   const uint8_t *sigs, uint32_t sig_len,
                                   uint32_t header_size, uint32_t crc_off,
                                   const uint64_t *host_dst, int nchunks,
                                   int sub_batch) {
	if (!e || !gftbls || !host_src || !host_dst || nchunks < 1)
		return LIZEC_EINVAL;
	if (ic < 1 || ic > 32 || oc < 1 || oc > 32) return LIZEC_EINVAL;
	if (part_len == 0 || part_len % 65536 || part_len > (uint64_t)1 << 31)
		return LIZEC_EINVAL;
	if (sig_len > header_size || crc_off + 4 * (part_len / 65536) > header_size)
		return LIZEC_EINVAL;
	LIZEC_CHECK(hipSetDevice(e->device));

	const uint64_t img_bytes = header_size + part_len;
	int B = sub_batch;
	if (B <= 0) {
		/* ~2 GiB of device staging per slot (sub_batch 16 at ec(8,2)
		 * measured ~10% over 12; profiles/ROUND2.md) */
		uint64_t per_chunk = (uint64_t)ic * part_len + oc * img_bytes;
		B = (int)(((uint64_t)1 << 31) / per_chunk);
		if (B < 1) B = 1;
		if (B > 64) B = 64;
	}
	if (B > nchunks) B = nchunks;

	/* per-(stripe,part) device pointer tables + per-image CRC meta:
	 * layout in d_meta/h_stage (8-byte aligned blocks):
	 *   [0)              src ptrs   B*ic u64
	 *   [src_end)        dst ptrs   B*oc u64
	 *   [meta_img)       image ptrs B*oc u64
	 *   [meta_off)       doffs/coffs/counts  3 * B*oc u32  */
	const size_t n_src = (size_t)B * ic, n_dst = (size_t)B * oc;
	const size_t meta_words = n_src + n_dst + n_dst + (3 * n_dst * 4 + 7) / 8;
	const size_t stage_bytes = meta_words * 8 + (size_t)B * oc * sig_len;

	uint8_t *d_tbl = nullptr;
	repl_slot sl[2];
	int rc = LIZEC_OK;
	uint8_t packed[kECTblBytes * 32 * 32];
	for (int i = 0; i < ic * oc; ++i)
		pack_mixed_radix(gftbls + (size_t)i * 32, packed + (size_t)i * kECTblBytes);
	if (hipMalloc(&d_tbl, (size_t)kECTblBytes * ic * oc) != hipSuccess)
		return LIZEC_ENOMEM;
	if (hipMemcpy(d_tbl, packed, (size_t)kECTblBytes * ic * oc,
	              hipMemcpyHostToDevice) != hipSuccess) {
		(void)hipFree(d_tbl);
		return LIZEC_EHIP;
	}
	for (int i = 0; i < 2 && rc == LIZEC_OK; ++i) {
		if (hipStreamCreate(&sl[i].stream) != hipSuccess ||
		    hipEventCreateWithFlags(&sl[i].done, hipEventDisableTiming) !=
		        hipSuccess ||
		    hipMalloc(&sl[i].d_in, (size_t)B * ic * part_len) != hipSuccess ||
		    hipMalloc(&sl[i].d_img, (size_t)B * oc * img_bytes) !=
		        hipSuccess ||
		    hipMalloc(&sl[i].d_meta, meta_words * 8) != hipSuccess ||
		    hipHostMalloc(&sl[i].h_stage, stage_bytes) != hipSuccess)
			rc = LIZEC_ENOMEM;
	}

	for (int c0 = 0; c0 < nchunks && rc == LIZEC_OK; c0 += B) {
		int nb = nchunks - c0 < B ? nchunks - c0 : B;
		repl_slot &s = sl[(c0 / B) & 1];
		hipStream_t st = s.stream;
		if (s.used) {
			if (hipEventSynchronize(s.done) != hipSuccess) {
				rc = LIZEC_EHIP;
				break;
			}
		}
		s.used = true;
		/* stage pointer tables + sigs (host side; slot is idle now) */
		uint64_t *h_src = (uint64_t *)s.h_stage;
		uint64_t *h_dst = h_src + n_src;
		uint64_t *h_img = h_dst + n_dst;
		uint32_t *h_off = (uint32_t *)(h_img + n_dst);
		uint32_t *h_coff = h_off + n_dst;
		uint32_t *h_cnt = h_coff + n_dst;
		uint8_t *h_sig = s.h_stage + meta_words * 8;
		const uint32_t nblocks = (uint32_t)(part_len / 65536);
		for (int i = 0; i < nb; ++i)
			for (int j = 0; j < ic; ++j)
				h_src[i * ic + j] = (uint64_t)(s.d_in +
				                               ((size_t)i * ic + j) * part_len);
		for (int i = 0; i < nb; ++i)
			for (int j = 0; j < oc; ++j) {
				size_t n = (size_t)i * oc + j;
				uint64_t img = (uint64_t)(s.d_img + n * img_bytes);
				h_img[n] = img;
				h_dst[n] = img + header_size;
				h_off[n] = header_size;
				h_coff[n] = crc_off;
				h_cnt[n] = nblocks;
			}
		if (sigs)
			memcpy(h_sig, sigs + (size_t)c0 * oc * sig_len,
			       (size_t)nb * oc * sig_len);
		/* enqueue the stage: H2D parts, header init, recover, CRC, D2H */
		for (int i = 0; i < nb && rc == LIZEC_OK; ++i)
			for (int j = 0; j < ic; ++j)
				if (hipMemcpyAsync(s.d_in + ((size_t)i * ic + j) * part_len,
				                   (const void *)host_src[(size_t)(c0 + i) *
				                                          ic + j],
				                   part_len, hipMemcpyHostToDevice,
				                   st) != hipSuccess)
					rc = LIZEC_EHIP;
		if (rc != LIZEC_OK) break;
		if (hipMemcpyAsync(s.d_meta, s.h_stage, meta_words * 8,
		                   hipMemcpyHostToDevice, st) != hipSuccess) {
			rc = LIZEC_EHIP;
			break;
		}
		for (int n = 0; n < nb * oc && rc == LIZEC_OK; ++n) {
			uint8_t *img = s.d_img + (size_t)n * img_bytes;
			if (hipMemsetAsync(img, 0, header_size, st) != hipSuccess)
				rc = LIZEC_EHIP;
			else if (sigs && sig_len &&
			         hipMemcpyAsync(img, h_sig + (size_t)n * sig_len, sig_len,
			                        hipMemcpyHostToDevice, st) != hipSuccess)
				rc = LIZEC_EHIP;
		}
		if (rc != LIZEC_OK) break;
		uint64_t *d_src = s.d_meta;
		uint64_t *d_dst = s.d_meta + n_src;
		uint64_t *d_img_ptrs = d_dst + n_dst;
		uint32_t *d_off = (uint32_t *)(d_img_ptrs + n_dst);
		uint32_t *d_coff = d_off + n_dst;
		uint32_t *d_cnt = d_coff + n_dst;
		rc = run_batch(part_len, ic, oc, d_tbl, d_src, d_dst, (uint32_t)nb,
		               st);
		if (rc != LIZEC_OK) break;
		rc = image_crc_launch(e, d_img_ptrs, d_off, d_coff, d_cnt, nb * oc,
		                      nblocks, st);
		if (rc != LIZEC_OK) break;
		for (int n = 0; n < nb * oc && rc == LIZEC_OK; ++n)
			if (hipMemcpyAsync((void *)host_dst[(size_t)c0 * oc + n],
			                   s.d_img + (size_t)n * img_bytes, img_bytes,
			                   hipMemcpyDeviceToHost, st) != hipSuccess)
				rc = LIZEC_EHIP;
		if (rc != LIZEC_OK) break;
		if (hipEventRecord(s.done, st) != hipSuccess) rc = LIZEC_EHIP;
	}
	for (int i = 0; i < 2; ++i) {
		if (sl[i].used && sl[i].stream)
			(void)hipStreamSynchronize(sl[i].stream);
		repl_slot_free(sl[i]);
	}
	(void)hipFree(d_tbl);
	return rc;
}
