/* ec_mi355x.h — C-ABI of the MI355X-native erasure-coding core
 * (libec_mi355x_core.so).
 *
 * This is the drop-in boundary between host code (C++ plugin glue, Python
 * ctypes, or a real Ceph build) and the GPU backend. It sits exactly where
 * the reference crosses from its plugin glue into the GF library:
 *
 *   reference interface replaced                     where cited
 *   ------------------------------------------------ -------------------------
 *   jerasure_matrix_encode(k,m,w,matrix,data,coding, src/erasure-code/jerasure/
 *     blocksize) / isa ec_encode_data(len,k,m,tbls,    ErasureCodeJerasure.cc:382-387,
 *     data,coding)                                     src/erasure-code/isa/ErasureCodeIsa.cc:289-300
 *     -> ecx_encode()                                  (and the host-batch forms below)
 *   jerasure_matrix_decode(...)/isa_decode(...)      ErasureCodeJerasure.cc:389-396,
 *     -> ecx_decode()                                  ErasureCodeIsa.cc:371-570
 *   galois_region_xor / isa xor_gen (encode_delta)   ErasureCodeJerasure.cc:258-268,
 *     -> ecx_encode_delta()                            ErasureCodeIsa.cc:317-328
 *   galois_w08_region_multiply accumulate /          ErasureCodeJerasure.cc:285-331,
 *     isa ec_encode_data_update (apply_delta)          ErasureCodeIsa.cc:333-366
 *     -> ecx_apply_delta()
 *   minimum_to_decode survivor choice                src/erasure-code/ErasureCode.cc:154-170
 *     -> ecx_minimum_to_decode()
 *   get_chunk_size / get_alignment                   ErasureCodeIsa.cc:65-79,
 *     -> ecx_chunk_size()                              ErasureCodeJerasure.cc:85-108
 *
 * All compute runs on the GPU (hand-written HIP for gfx950). There is NO CPU
 * fallback: every entry point returns ECX_ERR_NO_GPU if no HIP device is
 * available. Host-side work is limited to matrix derivation (integers, once
 * per (k,m,technique)) and decode-table composition with an LRU cache
 * mirroring ErasureCodeIsaTableCache (src/erasure-code/isa/
 * ErasureCodeIsaTableCache.h:46-48).
 *
 * Conventions: int returns, 0 on success, negative errno-style on failure
 * (ErasureCodeInterface.h:29-35). Plain pointers and sizes only — no torch
 * or HIP types cross this boundary. Thread safety: one ecx_ctx may be used
 * from many threads concurrently (mirrors the reference's stateless-
 * after-prepare() plugin contract, ErasureCodeInterface.h:414-449); stream
 * slots are internally synchronised.
 */
#ifndef EC_MI355X_H
#define EC_MI355X_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

#define ECX_API __attribute__((visibility("default")))

/* Techniques (profile "technique" values of the mi355x plugin):
 *   reed_sol_van          -> ECX_T_RS_VAN_ISA  (bit-compatible with plugin=isa
 *                            technique=reed_sol_van matrix construction)
 *   cauchy                -> ECX_T_CAUCHY_ISA  (isa gf_gen_cauchy1_matrix)
 *   jerasure_reed_sol_van -> ECX_T_RS_VAN_JERASURE (jerasure w=8 matrix)
 * Values must match oracle/ec_ref.h enum ecref_technique. */
enum ecx_technique {
  ECX_T_RS_VAN_ISA = 0,
  ECX_T_CAUCHY_ISA = 1,
  ECX_T_RS_VAN_JERASURE = 2,
  /* jerasure cauchy_orig: bitmatrix/packet layout (w=8), profile key
   * packetsize (default 2048, ErasureCodeJerasure.h DEFAULT_PACKETSIZE);
   * chunk sizes must be multiples of w*packetsize
   * (ErasureCodeJerasureCauchy::get_alignment, ErasureCodeJerasure.cc:522-536) */
  ECX_T_CAUCHY_ORIG_JERASURE = 3,
  /* jerasure reed_sol_van with w=16: GF(2^16) over gf-complete's 0x1100B,
   * u16 LE symbols (galois_w16_region_multiply semantics). Select via
   * ecx_create2(..., w=16, ...) with this id. Compute-bound compatibility
   * technique; chunk sizes are multiples of k*w*4 / k = 64 B. */
  ECX_T_RS_VAN_JERASURE_W16 = 4,
  /* jerasure cauchy_good: cauchy_original + the n_ones-minimising improve
   * pass (jerasure cauchy.c cauchy_good_general_coding_matrix general
   * branch; ErasureCodeJerasure.cc:537-555). Same bitmatrix/packet layout
   * and chunk-size rule as ECX_T_CAUCHY_ORIG_JERASURE. m == 2 is refused
   * at this layer: jerasure's m==2 path reads precomputed cbest tables
   * that cannot be faithfully restated here. */
  ECX_T_CAUCHY_GOOD_JERASURE = 5,
};

enum ecx_err {
  ECX_OK = 0,
  ECX_ERR_INVAL = -22,      /* EINVAL */
  ECX_ERR_NOMEM = -12,      /* ENOMEM */
  ECX_ERR_IO = -5,          /* EIO: too many erasures / singular matrix */
  ECX_ERR_NO_GPU = -19,     /* ENODEV: no HIP device — no CPU fallback */
  ECX_ERR_HIP = -71,        /* EPROTO: unexpected HIP runtime failure */
};

typedef struct ecx_ctx ecx_ctx;

/* Version string of this ABI ("ec-mi355x <semver>"). The plugin shim's
 * __erasure_code_version() (ErasureCodePlugin.cc:162-171 gate) is provided
 * by the shim, not here. */
ECX_API const char *ecx_version(void);

/* Number of visible HIP devices (0 => every compute call fails ECX_ERR_NO_GPU). */
ECX_API int ecx_device_count(void);

/* Create a context for one (k, m, technique) on one device.
 * n_streams >= 1 internal HIP streams (round-robin per call).
 * Builds the generator matrix host-side and uploads lookup tables once,
 * mirroring prepare() (ErasureCodeIsa.cc:637-697). */
ECX_API int ecx_create(int k, int m, int technique, int device, int n_streams,
                       ecx_ctx **out);
/* Extended form: w (must be 8) and packetsize (bitmatrix techniques only;
 * ignored for matrix techniques). */
ECX_API int ecx_create2(int k, int m, int technique, int w, int packetsize,
                        int device, int n_streams, ecx_ctx **out);
ECX_API void ecx_destroy(ecx_ctx *ctx);

ECX_API int ecx_k(const ecx_ctx *ctx);
ECX_API int ecx_m(const ecx_ctx *ctx);

/* Generator matrix readback for tests/verification: fills (k+m)*k bytes
 * (identity top, coding rows below), isa-l row-major layout. */
ECX_API int ecx_get_matrix(const ecx_ctx *ctx, uint8_t *out);
/* w=16 variant: fills (k+m)*k u16 entries; EINVAL on w=8 contexts (and
 * vice versa for ecx_get_matrix). */
ECX_API int ecx_get_matrix16(const ecx_ctx *ctx, uint16_t *out);

/* Chunk-size rule of the technique (a7 in SURVEY §8):
 * ECX_T_*_ISA: ceil(width/k) rounded up to 32 (ErasureCodeIsa.cc:65-79);
 * ECX_T_RS_VAN_JERASURE: stripe padded to k*w*4, w=8
 * (ErasureCodeJerasure.cc:85-108). Both are multiples of 16 as the kernels
 * require. */
ECX_API unsigned ecx_chunk_size(const ecx_ctx *ctx, unsigned stripe_width);

/* Survivor selection (ErasureCode.cc:154-170): given bitmask of available
 * chunk ids (bit i = chunk i available) and wanted ids, fill minimum with
 * the chunk ids to fetch; returns count or negative errno. */
ECX_API int ecx_minimum_to_decode(const ecx_ctx *ctx, uint64_t want_mask,
                                  uint64_t avail_mask, uint64_t *minimum_mask);

/* ---------------- device-resident batch API (the hot path) ----------------
 * A batch buffer holds n_stripes stripes, each (k+m) chunks of chunk_bytes:
 * chunk c of stripe s lives at offset (s*(k+m) + c)*chunk_bytes. Chunks
 * 0..k-1 are data, k..k+m-1 parity. chunk_bytes must be a multiple of 16.
 */

/* Allocate/free a device buffer (returned handle is the device pointer). */
ECX_API int ecx_dbuf_alloc(ecx_ctx *ctx, size_t bytes, void **dptr);
ECX_API int ecx_dbuf_free(ecx_ctx *ctx, void *dptr);

/* Host<->device copies on a stream slot (slot < n_streams; blocking=1 syncs). */
ECX_API int ecx_upload(ecx_ctx *ctx, void *dptr, const void *host, size_t bytes,
                       int slot, int blocking);
ECX_API int ecx_download(ecx_ctx *ctx, void *host, const void *dptr,
                         size_t bytes, int slot, int blocking);

/* Fill a device buffer with deterministic pseudo-random bytes (seeded),
 * for bench/tests without 32 GiB PCIe uploads. */
ECX_API int ecx_dbuf_fill_random(ecx_ctx *ctx, void *dptr, size_t bytes,
                                 uint64_t seed, int slot);

/* Encode the batch in place: parity chunks k..k+m-1 of every stripe are
 * computed from data chunks 0..k-1. Replaces ec_encode_data /
 * jerasure_matrix_encode over the whole batch in one (or few) kernel
 * launches. */
ECX_API int ecx_encode_batch(ecx_ctx *ctx, void *dptr, long n_stripes,
                             size_t chunk_bytes, int slot);

/* Decode the batch in place under a uniform erasure pattern:
 * present_mask bit i set => chunk i of every stripe is intact. Erased
 * chunks are reconstructed (bit-exact re-encode for lost parity). Decode
 * rows are composed host-side and LRU-cached per erasure signature
 * (ErasureCodeIsa.cc:460-567). */
ECX_API int ecx_decode_batch(ecx_ctx *ctx, void *dptr, long n_stripes,
                             size_t chunk_bytes, uint64_t present_mask,
                             int slot);

/* Parity-delta batch ops (a6 in SURVEY §8):
 * delta = old ^ new over a device region; */
ECX_API int ecx_encode_delta_dev(ecx_ctx *ctx, const void *d_old,
                                 const void *d_new, void *d_delta,
                                 size_t bytes, int slot);
/* parity ^= M[coding_shard-k][data_shard] * delta (galois region multiply
 * accumulate) over a device region. */
ECX_API int ecx_apply_delta_dev(ecx_ctx *ctx, const void *d_delta,
                                int data_shard, int coding_shard,
                                void *d_parity, size_t bytes, int slot);

/* ---------------- variable-size slice batch (caller-shaped) --
 * The OSD write path calls encode_chunks once per page-aligned slice with
 * varying blocksize (shard_extent_map_t::encode, src/osd/ECUtil.cc:485-514,
 * EC_ALIGN_SIZE=4096) — many small calls. This entry point batches N such
 * slices into ONE kernel launch over device-resident chunk pointers.
 * Arrays are host-side, length n_slices:
 *   d_chunks[s*(k+m)+c] = device pointer to chunk c of slice s (data
 *     0..k-1 may be NULL => zeros);
 *   bytes[s] = slice length (multiple of 16).
 * Encode: parity pointers k..k+m-1 are written.
 */
ECX_API int ecx_encode_slices(ecx_ctx *ctx, void *const *d_chunks,
                              const size_t *bytes, int n_slices, int slot);

/* Same for decode under a per-call uniform erasure pattern. */
ECX_API int ecx_decode_slices(ecx_ctx *ctx, void *const *d_chunks,
                              const size_t *bytes, int n_slices,
                              uint64_t present_mask, int slot);

/* Replace the coding rows of the generator with a custom m x k matrix
 * (clears decode-plan caches). Lets composed codecs — SHEC's shingled
 * matrix (src/erasure-code/shec/ErasureCodeShec.cc:700-768), custom
 * research codes — reuse every standard entry point. Matrix techniques
 * only (not bitmatrix). */
ECX_API int ecx_set_matrix(ecx_ctx *ctx, const uint8_t *coding_rows);

/* Generic GF(2^8) matmul over host chunks: outs[j] = XOR_i rows[j*n_src+i]
 * * srcs[i]. The primitive behind jerasure_matrix_dotprod
 * (used by SHEC decode, ErasureCodeShec.cc:1030-1046); srcs may be NULL
 * (zeros). Stages over PCIe and runs the standard kernel. */
ECX_API int ecx_matmul_chunks_host(ecx_ctx *ctx,
                                   const uint8_t *const *srcs, int n_src,
                                   uint8_t *const *outs, int n_out,
                                   const uint8_t *rows, size_t bytes);

/* SHEC shingled coding matrix (m x k), restated from the reference's
 * in-tree shec_reedsolomon_coding_matrix (ErasureCodeShec.cc:700-768);
 * single != 0 selects the SINGLE technique. */
ECX_API int ecx_shec_matrix(int k, int m, int c, int single, uint8_t *out);

/* CPU-only probes (no GPU context; usable on a GPU-less box for tests):
 * generator readback for any technique id — fills (k+m)*k bytes, identity
 * top — and jerasure cauchy.c's cauchy_n_ones(e, w=8). */
ECX_API int ecx_gen_matrix_probe(int technique, int k, int m, uint8_t *out);
ECX_API int ecx_cauchy_n_ones_probe(int e);
/* decode-plan composition probe: survivors[k], erased[<=m],
 * rows[n_erased*k]; returns n_erased or -errno. */
ECX_API int ecx_decode_rows_probe(int technique, int k, int m,
                                  uint64_t present_mask, int *survivors,
                                  int *erased, uint8_t *rows);

/* Generic device-batch GF(2^8) matmul over the standard batch layout:
 * out chunk ids = XOR_i rows[j*n_src+i] * src chunk ids, per stripe. The
 * composition primitive for layered codes (LRC layers, custom research
 * codes) on device-resident batches; w=8 matrix techniques only. */
ECX_API int ecx_matmul_batch(ecx_ctx *ctx, void *dptr, long n_stripes,
                             size_t chunk_bytes, const int *src_ids,
                             int n_src, const int *out_ids, int n_out,
                             const uint8_t *rows, int slot);

/* Synchronise a stream slot. */
ECX_API int ecx_sync(ecx_ctx *ctx, int slot);

/* Milliseconds of the most recent kernel on this slot (hipEvent pair
 * recorded around the kernel on ITS stream — bench.py's roofline source). */
ECX_API int ecx_last_kernel_ms(ecx_ctx *ctx, int slot, double *ms);

/* ---------------- host-pointer API (plugin path) ----------------
 * Mirrors the reference glue's char** marshalling
 * (ErasureCodeJerasure.cc:124-159, ErasureCodeIsa.cc:118-165): k data
 * pointers (NULL => chunk of zeros, the zeros-buffer convention) and m
 * parity pointers. Stages over PCIe, runs the same kernels, copies back.
 * One stripe per call; for throughput use the batch API. */
ECX_API int ecx_encode_chunks_host(ecx_ctx *ctx,
                                   const uint8_t *const *data /* [k] */,
                                   uint8_t *const *parity /* [m] */,
                                   size_t chunk_bytes);

/* decode_chunks semantics (ErasureCodeIsa.cc:167-243): chunks[] has k+m
 * entries; present_mask marks intact chunks; chunks[i] for erased ids must
 * be valid writable buffers (NULL allowed for intact ids the caller does
 * not have — treated as zeros, matching the reference's invented-zeros
 * buffers at ErasureCodeIsa.cc:212-226). */
ECX_API int ecx_decode_chunks_host(ecx_ctx *ctx, uint8_t *const *chunks,
                                   uint64_t present_mask, size_t chunk_bytes);

ECX_API int ecx_encode_delta_host(ecx_ctx *ctx, const uint8_t *old_data,
                                  const uint8_t *new_data, uint8_t *delta,
                                  size_t bytes);
ECX_API int ecx_apply_delta_host(ecx_ctx *ctx, const uint8_t *delta,
                                 int data_shard, int coding_shard,
                                 uint8_t *parity, size_t bytes);

#ifdef __cplusplus
}
#endif
#endif /* EC_MI355X_H */
