/* bench_variants.hip — A/B harness for the EC encode kernel variants on a
 * real MI355X.  Standalone executable (not part of liblizec.so): allocates
 * device-resident ec(k,m) batches (k, m, stripes, reps from argv), runs
 * each template configuration whose D divides m, checks a sample of the
 * output against the host scalar path, and prints a GB/s table
 * (traffic-based: (k+m)/k bytes per data byte).
 *
 * Build: make bench_variants  ; run on the GPU box only.
 */
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

#include "../../include/lizec.h"
#include "ec_kernel.h"

#define CK(x)                                                           \
	do {                                                                \
		hipError_t e = (x);                                             \
		if (e != hipSuccess) {                                          \
			fprintf(stderr, "HIP error %s at %s:%d\n",                  \
			        hipGetErrorString(e), __FILE__, __LINE__);          \
			exit(1);                                                    \
		}                                                               \
	} while (0)

__global__ void fill_kernel(uint8_t *p, size_t n, uint32_t salt) {
	size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
	size_t stride = (size_t)gridDim.x * blockDim.x;
	for (; i * 4 < n; i += stride) {
		uint32_t v = (uint32_t)(i * 2654435761u) ^ salt;
		v ^= v >> 15;
		v *= 0x2c1b3c6du;
		v ^= v >> 12;
		((uint32_t *)p)[i] = v;
	}
}

template <int D, int CH, bool SWZ, bool NTST, bool NTLD = false,
          bool TPIPE = false, bool QL = false, bool MR = false>
static void launch_var(uint32_t part_len, int srcs, const uint8_t *tbls,
                       const uint64_t *src, const uint64_t *dst, int dests,
                       uint32_t tiles_per_part, uint32_t total_tiles,
                       uint32_t grid_cap, hipStream_t s) {
	uint32_t grid = total_tiles < grid_cap ? total_tiles : grid_cap;
	size_t lds = (size_t)D * srcs * (QL ? 16 : 32);
	for (int base = 0; base + D <= dests; base += D)
		hipLaunchKernelGGL(HIP_KERNEL_NAME(ec_encode_kernel<D, CH, SWZ, NTST, NTLD, TPIPE, QL, MR>),
		                   dim3(grid), dim3(kThreads), lds, s, part_len, srcs,
		                   base, tbls, src, dst, dests, tiles_per_part,
		                   total_tiles);
}

int main(int argc, char **argv) {
	int k = 8, m = 2, stripes = 1024, reps = 6;
	if (argc > 1) k = atoi(argv[1]);
	if (argc > 2) m = atoi(argv[2]);
	if (argc > 3) stripes = atoi(argv[3]);
	if (argc > 4) reps = atoi(argv[4]);
	uint64_t part_len = 64ull * 1024 * 1024 / k;
	part_len &= ~15ull;

	size_t data_bytes = (size_t)stripes * k * part_len;
	size_t par_bytes = (size_t)stripes * m * part_len;
	printf("ec(%d,%d) stripes=%d part=%lu MiB data=%.1f GiB\n", k, m, stripes,
	       part_len >> 20, data_bytes / 1073741824.0);

	uint8_t *d_data, *d_par;
	CK(hipMalloc(&d_data, data_bytes));
	CK(hipMalloc(&d_par, par_bytes));
	hipLaunchKernelGGL(fill_kernel, dim3(4096), dim3(256), 0, 0, d_data,
	                   data_bytes, 0x1234567u);

	static uint8_t tbls[32 * 32 * 32];
	if (lizec_rs_encode_tables(k, m, tbls) != 0) return 1;
	uint8_t *d_tbls;
	CK(hipMalloc(&d_tbls, 32 * k * m));
	CK(hipMemcpy(d_tbls, tbls, 32 * k * m, hipMemcpyHostToDevice));
	/* packed quarter-LUT layout: 16 B per (dest,src) */
	static uint8_t tbls_q[16 * 32 * 32];
	for (int i = 0; i < k * m; ++i) {
		const uint8_t *t = tbls + i * 32;
		uint8_t *q = tbls_q + i * 16;
		for (int b = 0; b < 4; ++b) {
			q[b] = t[b];
			q[4 + b] = t[4 * b];
			q[8 + b] = t[16 + b];
			q[12 + b] = t[16 + 4 * b];
		}
	}
	uint8_t *d_tbls_q;
	CK(hipMalloc(&d_tbls_q, 16 * k * m));
	CK(hipMemcpy(d_tbls_q, tbls_q, 16 * k * m, hipMemcpyHostToDevice));
	/* packed mixed-radix (3+3+2) layout: 32 B padded per (dest,src);
	 * derived from the ISA-L table by GF linearity over disjoint bits */
	static uint8_t tbls_mr[32 * 32 * 32];
	memset(tbls_mr, 0, sizeof(tbls_mr));
	for (int i = 0; i < k * m; ++i) {
		const uint8_t *t = tbls + i * 32;
		uint8_t *q = tbls_mr + i * 32;
		for (int v = 0; v < 8; ++v) {
			q[v] = t[v];                                   /* c*v        */
			q[8 + v] = t[(8 * v) & 15] ^ t[16 + (v >> 1)]; /* c*(v<<3)   */
		}
		for (int v = 0; v < 4; ++v)
			q[16 + v] = t[16 + (v << 2)];                  /* c*(v<<6)   */
	}
	uint8_t *d_tbls_mr;
	CK(hipMalloc(&d_tbls_mr, 32 * k * m));
	CK(hipMemcpy(d_tbls_mr, tbls_mr, 32 * k * m, hipMemcpyHostToDevice));

	std::vector<uint64_t> sp(stripes * k), dp(stripes * m);
	for (int s = 0; s < stripes; ++s) {
		for (int j = 0; j < k; ++j)
			sp[s * k + j] = (uint64_t)(d_data + ((size_t)s * k + j) * part_len);
		for (int l = 0; l < m; ++l)
			dp[s * m + l] = (uint64_t)(d_par + ((size_t)s * m + l) * part_len);
	}
	uint64_t *d_sp, *d_dp;
	CK(hipMalloc(&d_sp, sp.size() * 8));
	CK(hipMalloc(&d_dp, dp.size() * 8));
	CK(hipMemcpy(d_sp, sp.data(), sp.size() * 8, hipMemcpyHostToDevice));
	CK(hipMemcpy(d_dp, dp.data(), dp.size() * 8, hipMemcpyHostToDevice));

	/* host reference for a sample window */
	const int SAMPLE = 4096;
	std::vector<uint8_t> h_src(k * SAMPLE), h_exp(m * SAMPLE), h_got(SAMPLE);
	{
		for (int j = 0; j < k; ++j)
			CK(hipMemcpy(&h_src[j * SAMPLE], (void *)sp[j], SAMPLE,
			             hipMemcpyDeviceToHost));
		uint8_t *srcp[32], *dstp[32];
		for (int j = 0; j < k; ++j) srcp[j] = &h_src[j * SAMPLE];
		for (int l = 0; l < m; ++l) dstp[l] = &h_exp[l * SAMPLE];
		ec_encode_data(SAMPLE, k, m, tbls, srcp, dstp);
	}

	struct Cfg {
		const char *name;
		void (*fn)(uint32_t, int, const uint8_t *, const uint64_t *,
		           const uint64_t *, int, uint32_t, uint32_t, uint32_t,
		           hipStream_t);
		int ch;
		int d;
		uint32_t grid_cap;
		int tbl = 0;   /* 0 = full 32B, 1 = quarter-LUT 16B, 2 = MR 32B */
	};
	Cfg cfgs[] = {
	    {"D2_CH4_mr    ", launch_var<2, 4, true, true, true, false, false, true>, 4, 2, 1048576, 2},
	    {"D4_CH4_mr    ", launch_var<4, 4, true, true, true, false, false, true>, 4, 4, 1048576, 2},
	    {"D4_CH2_mr    ", launch_var<4, 2, true, true, true, false, false, true>, 2, 4, 1048576, 2},
	    {"D6_CH2_mr    ", launch_var<6, 2, true, true, true, false, false, true>, 2, 6, 1048576, 2},
	    {"D6_CH1_mr    ", launch_var<6, 1, true, true, true, false, false, true>, 1, 6, 1048576, 2},
	    {"D3_CH4_mr    ", launch_var<3, 4, true, true, true, false, false, true>, 4, 3, 1048576, 2},
	    {"D6_CH4_mr    ", launch_var<6, 4, true, true, true, false, false, true>, 4, 6, 1048576, 2},
	    {"D8_CH2_mr    ", launch_var<8, 2, true, true, true, false, false, true>, 2, 8, 1048576, 2},
	    {"D1_CH4_swz_nt ", launch_var<1, 4, true, true>, 4, 1, 262144},
	    {"D2_CH4_swz_nt ", launch_var<2, 4, true, true>, 4, 2, 262144},
	    {"D2_CH4_swz_nt_ntld", launch_var<2, 4, true, true, true>, 4, 2, 262144},
	    {"D2_CH4_ntld_gexact", launch_var<2, 4, true, true, true>, 4, 2, 1048576},
	    {"D2_CH4_ql     ", launch_var<2, 4, true, true, true, false, true>, 4, 2, 1048576, true},
	    {"D2_CH4_ntld_tp", launch_var<2, 4, true, true, true, true>, 4, 2, 262144},
	    {"D2_CH6_swz_nt ", launch_var<2, 6, true, true>, 6, 2, 262144},
	    {"D2_CH4_base   ", launch_var<2, 4, false, false>, 4, 2, 262144},
	    {"D3_CH4_swz_nt ", launch_var<3, 4, true, true>, 4, 3, 262144},
	    {"D4_CH4_swz_nt ", launch_var<4, 4, true, true>, 4, 4, 262144},
	    {"D4_CH3_swz_nt ", launch_var<4, 3, true, true>, 3, 4, 262144},
	    {"D4_CH2_swz_nt ", launch_var<4, 2, true, true>, 2, 4, 262144},
	    {"D4_CH2_ntld   ", launch_var<4, 2, true, true, true>, 2, 4, 262144},
	    {"D4_CH2_ql     ", launch_var<4, 2, true, true, true, false, true>, 2, 4, 1048576, true},
	    {"D4_CH4_ql     ", launch_var<4, 4, true, true, true, false, true>, 4, 4, 1048576, true},
	    {"D4_CH2_ntld_tp", launch_var<4, 2, true, true, true, true>, 2, 4, 262144},
	    {"D4_CH4_ntld_tp", launch_var<4, 4, true, true, true, true>, 4, 4, 262144},
	    {"D4_CH4_ntld   ", launch_var<4, 4, true, true, true>, 4, 4, 262144},
	    {"D5_CH2_swz_nt ", launch_var<5, 2, true, true>, 2, 5, 262144},
	    {"D6_CH4_swz_nt ", launch_var<6, 4, true, true>, 4, 6, 262144},
	    {"D6_CH3_swz_nt ", launch_var<6, 3, true, true>, 3, 6, 262144},
	    {"D6_CH2_swz_nt ", launch_var<6, 2, true, true>, 2, 6, 262144},
	    {"D6_CH2_ntld   ", launch_var<6, 2, true, true, true>, 2, 6, 262144},
	    {"D6_CH2_ql     ", launch_var<6, 2, true, true, true, false, true>, 2, 6, 1048576, true},
	    {"D6_CH4_ql     ", launch_var<6, 4, true, true, true, false, true>, 4, 6, 1048576, true},
	    {"D8_CH2_ql     ", launch_var<8, 2, true, true, true, false, true>, 2, 8, 1048576, true},
	    {"D8_CH4_ql     ", launch_var<8, 4, true, true, true, false, true>, 4, 8, 1048576, true},
	    {"D6_CH2_ntld_tp", launch_var<6, 2, true, true, true, true>, 2, 6, 262144},
	    {"D6_CH4_ntld   ", launch_var<6, 4, true, true, true>, 4, 6, 262144},
	    {"D8_CH2_swz_nt ", launch_var<8, 2, true, true>, 2, 8, 262144},
	    {"D8_CH1_swz_nt ", launch_var<8, 1, true, true>, 1, 8, 262144},
	};

	double traffic = (double)stripes * part_len * (k + m);
	hipEvent_t e0, e1;
	CK(hipEventCreate(&e0));
	CK(hipEventCreate(&e1));

	for (auto &c : cfgs) {
		if (m % c.d != 0) { printf("%s  skip (D !| m)\n", c.name); continue; }
		uint32_t tpp = (uint32_t)((part_len + c.ch * kChunkBytes - 1) /
		                          (c.ch * kChunkBytes));
		uint32_t tot = tpp * stripes;
		CK(hipMemset(d_par, 0, 4096));
		/* warmup x2 */
		const uint8_t *tb = c.tbl == 1 ? d_tbls_q
		                  : c.tbl == 2 ? d_tbls_mr : d_tbls;
		for (int r = 0; r < 2; ++r)
			c.fn((uint32_t)part_len, k, tb, d_sp, d_dp, m, tpp, tot,
			     c.grid_cap, 0);
		CK(hipDeviceSynchronize());
		CK(hipEventRecord(e0, 0));
		for (int r = 0; r < reps; ++r)
			c.fn((uint32_t)part_len, k, tb, d_sp, d_dp, m, tpp, tot,
			     c.grid_cap, 0);
		CK(hipEventRecord(e1, 0));
		CK(hipDeviceSynchronize());
		float ms;
		CK(hipEventElapsedTime(&ms, e0, e1));
		double gbps = traffic * reps / (ms / 1e3) / 1e9;
		/* verify sample of parity 0 */
		CK(hipMemcpy(h_got.data(), (void *)dp[0], SAMPLE,
		             hipMemcpyDeviceToHost));
		bool ok = memcmp(h_got.data(), h_exp.data(), SAMPLE) == 0;
		printf("%s  %8.1f GB/s  (%.3f ms/rep)  %s\n", c.name, gbps,
		       ms / reps, ok ? "OK" : "WRONG");
	}
	return 0;
}
