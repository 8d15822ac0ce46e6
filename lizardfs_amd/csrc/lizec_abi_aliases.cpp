/* lizec_abi_aliases.cpp — C++-linkage aliases of the galois_field.h
 * surface (reference common/galois_field.h:35-88).
 *
 * The reference's fallback build references these functions with C++
 * mangling (galois_field.h has no extern "C"); the ISA-L build uses the C
 * names.  liblizec exports both: lizec_host.cpp defines the extern "C"
 * canon, this TU (no extern "C" declarations in scope, so no linkage
 * conflict) adds the mangled forwarders.  Verified end-to-end by
 * tests/test_abi_dropin.py, which compiles a probe against the
 * reference's own headers and links this library.
 */
#include <cstddef>
#include <cstdint>

/* internal C entry points (distinct names; defined in lizec_host.cpp) */
extern "C" {
void lizec_impl_gen_rs_matrix(uint8_t *a, int m, int k);
void lizec_impl_gen_cauchy1_matrix(uint8_t *a, int m, int k);
int lizec_impl_invert_matrix(uint8_t *in_mat, uint8_t *out_mat, int n);
void lizec_impl_init_tables(int k, int rows, uint8_t *a, uint8_t *g_tbls);
void lizec_impl_encode_data(int len, int srcs, int dests, uint8_t *v,
                            uint8_t **src, uint8_t **dest);
}

/* the reference's C++ prototypes (galois_field.h:35-88) */
void gf_gen_rs_matrix(uint8_t *a, int m, int k) {
	lizec_impl_gen_rs_matrix(a, m, k);
}
void gf_gen_cauchy1_matrix(uint8_t *a, int m, int k) {
	lizec_impl_gen_cauchy1_matrix(a, m, k);
}
int gf_invert_matrix(uint8_t *in_mat, uint8_t *out_mat, const int n) {
	return lizec_impl_invert_matrix(in_mat, out_mat, n);
}
void ec_init_tables(int k, int rows, uint8_t *a, uint8_t *g_tbls) {
	lizec_impl_init_tables(k, rows, a, g_tbls);
}
void ec_encode_data(int len, int srcs, int dests, uint8_t *v, uint8_t **src,
                    uint8_t **dest) {
	lizec_impl_encode_data(len, srcs, dests, v, src, dest);
}
