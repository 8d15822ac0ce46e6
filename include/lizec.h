/* lizec.h — C ABI of the MI355X-native LizardFS erasure-coding engine.
 *
 * This is the drop-in boundary for the LizardFS chunkserver/client EC hot
 * path.  The reference already swaps EC backends at exactly this seam: its
 * reed_solomon.h:27-31 compiles against either Intel ISA-L's
 * <isa-l/erasure_code.h> or the in-tree common/galois_field.h:35-88 — both
 * expose the five functions below.  liblizec.so exports them with the same
 * names and semantics (extern "C", ISA-L calling convention), so a LizardFS
 * build configured for ISA-L links against this library unchanged.  The CRC
 * surface mirrors common/crc.h:25-31 (also exported with C++ linkage from
 * the library so the reference's mangled references resolve).
 *
 * On top of the per-call surface sits the batched GPU engine (lizec_engine,
 * lizec_*_batch): the product proper.  Per-64KiB-call granularity cannot
 * feed a GPU, so the engine takes batches of stripes of device-resident
 * parts; the host-side matrix algebra (reference reed_solomon.h:163-358)
 * stays on the CPU exactly as in the reference.
 *
 * Ownership/threading: caller owns every buffer; no function retains state
 * except the engine object.  The per-call functions are pure and
 * thread-safe (matching bgjobs-worker concurrency in the reference,
 * network_main_thread.cc:230-234).  An engine is NOT thread-safe: use one
 * engine per stream/thread.
 *
 * Error convention: 0 = success; negative = error (see LIZEC_E*).  The GF
 * layer itself only signals singular matrices, like the reference
 * (galois_field_isal.cc:87-139 returns -1).
 */
#ifndef LIZEC_H
#define LIZEC_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ------------------------------------------------------------------ */
/* ISA-L-shaped surface — drop-in for common/galois_field.h:35-88.    */
/* GF(2^8), polynomial 0x11D (galois_coeff.h:30-32).                  */
/* ------------------------------------------------------------------ */

/* (k+m) x k Vandermonde-style generator matrix; identity on top.
 * Replaces gf_gen_rs_matrix (galois_field.h:35, galois_field_isal.cc:53). */
void gf_gen_rs_matrix(uint8_t *a, int m, int k);

/* (k+m) x k Cauchy matrix (1/(i^j)); identity on top.
 * Replaces gf_gen_cauchy1_matrix (galois_field.h:48, galois_field_isal.cc:71). */
void gf_gen_cauchy1_matrix(uint8_t *a, int m, int k);

/* Gauss-Jordan inversion in GF(2^8).  Mutates in_mat.  -1 if singular.
 * Replaces gf_invert_matrix (galois_field.h:58, galois_field_isal.cc:87). */
int gf_invert_matrix(uint8_t *in_mat, uint8_t *out_mat, const int n);

/* Expand rows*k coefficients into 32-byte lo/hi-nibble product tables
 * (tbl[i]=c*i, tbl[16+i]=c*(i<<4)); layout [row][col][32].
 * Replaces ec_init_tables (galois_field.h:71, galois_field_isal.cc:246). */
void ec_init_tables(int k, int rows, uint8_t *a, uint8_t *g_tbls);

/* Host scalar encode: dest[l][i] = XOR_j tbl(l,j)[src[j][i]].  This is the
 * reference's per-64KiB-block CPU entry point (galois_field.h:88,
 * galois_field_encode.cc:28-225); kept for drop-in completeness and for the
 * host-side matrixMultiply.  The measured product path is the batched GPU
 * engine below — this function never runs inside it. */
void ec_encode_data(int len, int srcs, int dests, uint8_t *v, uint8_t **src,
                    uint8_t **dest);

/* ------------------------------------------------------------------ */
/* CRC surface — drop-in for common/crc.h:25-31.                      */
/* Reflected CRC-32, poly 0xEDB88320, zlib-compatible.                */
/* ------------------------------------------------------------------ */

uint32_t lizec_crc32(uint32_t crc, const uint8_t *block, uint32_t leng);
uint32_t lizec_crc32_combine(uint32_t crc1, uint32_t crc2, uint32_t leng2);
void lizec_crc32_init(void);

/* Partial-block CRC algebra (crc.h:27-29 macros + hdd_write's splice,
 * hddspacemgr.cc:1952-2003) — byte-range writes without re-hashing. */
uint32_t lizec_crc32_zeroblock(uint32_t crc, uint32_t zeros);
uint32_t lizec_crc32_zeroexpanded(uint32_t crc, const uint8_t *block,
                                  uint32_t leng, uint32_t zeros);
uint32_t lizec_crc32_xorblocks(uint32_t crc, uint32_t crcblock1,
                               uint32_t crcblock2, uint32_t leng);
uint32_t lizec_crc32_splice(uint32_t precrc, uint32_t offset, uint32_t crc,
                            uint32_t size, uint32_t postcrc,
                            uint32_t block_len);
void lizec_recompute_crc_if_block_empty(const uint8_t *block,
                                        uint32_t block_len, uint32_t *crc);

/* Host byte-XOR (xor-goal parity building block; also exported with the
 * C++ mangling of the reference's blockXor, common/block_xor.h:33).
 * GPU xor parity/recovery runs through the EC kernel with all-ones
 * coefficient tables (lizardfs_amd/xor.py). */
void lizec_blockxor(uint8_t *dest, const uint8_t *source, size_t size);

/* ------------------------------------------------------------------ */
/* Reed-Solomon table builders — host-side restatement of             */
/* ReedSolomon<32,32> (reed_solomon.h:41-373).                        */
/* Parts indexed 0..k+m-1: data 0..k-1, parity k..k+m-1               */
/* (slice_traits.h:183-197).  Matrix: Cauchy iff m>=5 || (m==4 &&     */
/* k>20), else Vandermonde (reed_solomon.h:168).                      */
/* ------------------------------------------------------------------ */

enum {
	LIZEC_EINVAL = -2,
	LIZEC_ESINGULAR = -1,   /* decode matrix not invertible */
	LIZEC_OK = 0,
	LIZEC_ENOGPU = -3,
	LIZEC_EHIP = -4,
	LIZEC_ENOMEM = -5,
};

/* Expanded gf tables for ReedSolomon::recover (reed_solomon.h:87-121):
 *  present_mask: parts NOT erased (bit i = part i available)
 *  nonnull_mask: available parts whose buffer is non-NULL (NULL = zeros,
 *                reed_solomon.h:79); must be a subset of present_mask
 *  needed_mask : erased parts to reconstruct (subset of ~present_mask)
 * gftbls must hold 32 * popcount(nonnull) * popcount(needed) bytes.
 * Outputs the number of inputs (surviving non-NULL parts, ascending part
 * order) and outputs (needed parts, ascending part order).
 * Encode is the special case present=nonnull=data parts,
 * needed=all m parities (reed_solomon.h:134-155). */
int lizec_rs_tables(int k, int m, uint64_t present_mask, uint64_t nonnull_mask,
                    uint64_t needed_mask, uint8_t *gftbls, int *in_count,
                    int *out_count);

/* Convenience: encode tables for all m parities from all k data parts. */
int lizec_rs_encode_tables(int k, int m, uint8_t *gftbls /* 32*k*m */);

/* ------------------------------------------------------------------ */
/* Slice-type algebra — mirror of slice_traits.h / goal.h:108-119.    */
/* EC(k,m) slice type = 32*(k-2) + (m-1) + 10  (slice_traits.h:148).  */
/* ------------------------------------------------------------------ */

int lizec_slice_type_ec(int data_parts, int parity_parts);
int lizec_slice_is_ec(int slice_type);
int lizec_slice_data_parts(int slice_type);    /* k */
int lizec_slice_parity_parts(int slice_type);  /* m */
/* ChunkPartType packing: id = type*64 + part (chunk_part_type.h:145,170). */
int lizec_chunk_part_id(int slice_type, int part);
int lizec_chunk_part_slice_type(int id);
int lizec_chunk_part_index(int id);
/* chunkLengthToChunkPartLength (slice_traits.h:332-349); block = 64 KiB. */
int64_t lizec_chunk_part_length(int slice_type, int part, int64_t chunk_length);

/* ------------------------------------------------------------------ */
/* The batched GPU engine (the product).                              */
/* All device pointers are HIP device addresses on the engine's       */
/* device; `stream` is a hipStream_t (NULL = engine's own stream).    */
/* Calls are asynchronous on that stream.  Fails with LIZEC_ENOGPU    */
/* if no MI355X is present — there is no CPU fallback.                */
/* ------------------------------------------------------------------ */

typedef struct lizec_engine lizec_engine;

int lizec_gpu_count(void);
int lizec_engine_create(lizec_engine **out, int device_id);
void lizec_engine_destroy(lizec_engine *e);
int lizec_engine_sync(lizec_engine *e);

/* Batched EC encode/decode over device-resident stripes.  All stripes in a
 * batch share (srcs, dests, gftbls) — the reference likewise reuses one
 * cached table per (erasure pattern, k, m) (reed_solomon.h:194-198).
 *  part_len : bytes per part
 *  gftbls   : HOST pointer, 32*srcs*dests bytes (from lizec_rs_tables)
 *  src_dptrs: HOST array of num_stripes*srcs device addresses
 *             (stripe-major: stripe s, input j at [s*srcs + j])
 *  dst_dptrs: HOST array of num_stripes*dests device addresses
 * Covers encode (a1/a6: chunk_writer.cc:365-402), degraded-read decode
 * (ec_read_plan.h:113-146) and replicator recovery
 * (chunk_replicator.cc:139-196) — all three reduce to ec_encode_data with
 * different tables. */
int lizec_ec_encode_batch(lizec_engine *e, uint64_t part_len, int srcs,
                          int dests, const uint8_t *gftbls,
                          const uint64_t *src_dptrs, const uint64_t *dst_dptrs,
                          int num_stripes, void *stream);

/* Prepared batches ("plans", after the reference's read-plan/matrix-cache
 * structure, read_plan.h + reed_solomon.h:194-198): device-resident tables
 * and pointer arrays built once, then run repeatedly with no host->device
 * traffic.  Same arguments as lizec_ec_encode_batch. */
typedef struct lizec_plan lizec_plan;
int lizec_ec_plan_create(lizec_engine *e, uint64_t part_len, int srcs,
                         int dests, const uint8_t *gftbls,
                         const uint64_t *src_dptrs, const uint64_t *dst_dptrs,
                         int num_stripes, lizec_plan **out);
int lizec_ec_plan_run(lizec_plan *p, void *stream);
void lizec_ec_plan_destroy(lizec_plan *p);

/* Per-block CRC32 over a contiguous device buffer:
 *  dev_crcs_out[b] = lizec_crc32(seed, dev_buf + b*block_len, block_len)
 * The chunkserver's per-64KiB-block gate (hddspacemgr.cc:1918-1920, scrub
 * :2148-2212, replicate chunk_replicator.cc:189) in batch form. */
int lizec_crc32_batch(lizec_engine *e, const void *dev_buf, uint32_t block_len,
                      uint64_t nblocks, uint32_t seed, uint32_t *dev_crcs_out,
                      void *stream);

/* Batched chunk scrub — hdd_int_test semantics (hddspacemgr.cc:2148-2212)
 * over MooseFS-format chunk-part images (chunk.cc:126-188: 1 KiB signature
 * block, big-endian u32 CRC array at crc_off, 64 KiB blocks at data_off):
 * verifies every block's CRC against the stored array.
 *  chunk_dptrs[i] : device address of part-file image i
 *  data_offs[i]   : header size (offset of block 0)
 *  crc_offs[i]    : offset of the CRC array (kMaxSignatureBlockSize = 1024)
 *  block_counts[i]: blocks in image i
 *  dev_status_out : device int32[nchunks]; first damaged block index, or
 *                   INT32_MAX if the image is clean. */
int lizec_scrub_batch(lizec_engine *e, const uint64_t *chunk_dptrs,
                      const uint32_t *data_offs, const uint32_t *crc_offs,
                      const uint32_t *block_counts, int nchunks,
                      int32_t *dev_status_out, void *stream);

/* Generalized layout: block b's data at data_offs[i] + b*block_stride,
 * its stored CRC at crc_offs[i] + b*crc_stride.  Covers both on-disk
 * formats: MooseFS (stride 65536 / 4; the wrapper above) and INTERLEAVED
 * (kHddBlockSize = 65540, chunk.h:40: 4-byte CRC inline before each
 * block — data_off 4, both strides 65540, crc_off 0). */
int lizec_scrub_batch_strided(lizec_engine *e, const uint64_t *chunk_dptrs,
                              const uint32_t *data_offs,
                              const uint32_t *crc_offs,
                              const uint32_t *block_counts, int nchunks,
                              uint32_t block_stride, uint32_t crc_stride,
                              int32_t *dev_status_out, void *stream);

/* Pinned host memory for the streaming APIs (hipHostMalloc/hipHostFree):
 * pageable buffers work too but degrade the pipeline to synchronous
 * copies. */
int lizec_host_alloc(void **ptr, uint64_t bytes);
void lizec_host_free(void *p);

/* Streaming chunk rebuild — ChunkReplicator::replicate
 * (chunk_replicator.cc:139-196) as a host-to-host pipeline: H2D the
 * surviving parts, EC-recover the erased parts, CRC every recovered
 * 64 KiB block (chunk_replicator.cc:189), assemble complete MooseFS part
 * images (chunk.cc:126-188: signature bytes at 0, big-endian CRC array at
 * crc_off, blocks at header_size) and D2H them — double-buffered over two
 * streams so H2D/compute/D2H overlap.
 *  part_len    : bytes per part, multiple of 65536
 *  ic, oc      : surviving parts in / parts to rebuild per chunk
 *  gftbls      : 32*ic*oc recover tables (lizec_rs_tables)
 *  host_src    : nchunks*ic HOST addresses of surviving part bytes
 *  sigs        : nchunks*oc signatures (sig_len bytes each), or NULL
 *  header_size : image header bytes; crc_off: CRC array offset within it
 *  host_dst    : nchunks*oc HOST addresses, header_size+part_len each
 *  sub_batch   : chunks per pipeline stage (0 = auto, ~1 GiB staging) */
int lizec_replicate_run(lizec_engine *e, uint64_t part_len, int ic, int oc,
                        const uint8_t *gftbls, const uint64_t *host_src,
                        const uint8_t *sigs, uint32_t sig_len,
                        uint32_t header_size, uint32_t crc_off,
                        const uint64_t *host_dst, int nchunks, int sub_batch);

#ifdef __cplusplus
}
#endif

#endif /* LIZEC_H */
