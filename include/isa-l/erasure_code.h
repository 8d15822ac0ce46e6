/* isa-l/erasure_code.h — shim for LizardFS's ISA-L build mode.
 *
 * The reference's reed_solomon.h:27-31 includes this header when
 * LIZARDFS_HAVE_ISA_L_ERASURE_CODE_H is set and expects the five ISA-L
 * entry points; liblizec.so provides them (include/lizec.h).  Point the
 * LizardFS build's include path at this directory's parent and link
 * liblizec.so instead of -lisal (INTEGRATION.md level 1).
 */
#ifndef LIZEC_ISAL_SHIM_H
#define LIZEC_ISAL_SHIM_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

void gf_gen_rs_matrix(uint8_t *a, int m, int k);
void gf_gen_cauchy1_matrix(uint8_t *a, int m, int k);
int gf_invert_matrix(uint8_t *in_mat, uint8_t *out_mat, const int n);
void ec_init_tables(int k, int rows, uint8_t *a, uint8_t *g_tbls);
void ec_encode_data(int len, int srcs, int dests, uint8_t *v, uint8_t **src,
                    uint8_t **dest);

#ifdef __cplusplus
}
#endif

#endif /* LIZEC_ISAL_SHIM_H */
