/* ec_cpu.c — CPU baseline for bench.py's cpu_baseline leg.
 *
 * TEST/BASELINE INFRASTRUCTURE ONLY (same scope rules as ec_ref.h).
 *
 * This is the stand-in for the host ISA-L path named by BASELINE.md: the
 * reference's canonical benchmark runs plugin=isa, whose ec_encode_data is a
 * 4-bit split-table pshufb kernel (isa-l erasure_code.h; submodule absent —
 * see ec_ref.h). This file implements the same algorithm with AVX2
 * _mm256_shuffle_epi8 and OpenMP over stripes, with a scalar fallback, so
 * the CPU number reported beside the GPU number is ISA-L-class, not a
 * strawman. Correctness of this file is itself pinned against ec_ref.c by
 * tests/test_oracle_properties.py.
 */
#include "ec_ref.h"

#include <stdint.h>
#include <string.h>

#if defined(__x86_64__)
#include <immintrin.h>
#endif

#ifdef _OPENMP
#include <omp.h>
#endif

/* Build the 32-byte split table for one coefficient: bytes 0..15 = c*x for
 * x in 0..15 (low nibble), bytes 16..31 = c*(x<<4) (high nibble). This is
 * the isa-l ec_init_tables / gf_vect_mul_init layout. */
static void mul_table_32(uint8_t c, uint8_t *t)
{
  for (int x = 0; x < 16; x++) {
    t[x] = ecref_gf_mul(c, (uint8_t)x);
    t[16 + x] = ecref_gf_mul(c, (uint8_t)(x << 4));
  }
}

static void region_mul_acc_scalar(uint8_t c, const uint8_t *d, uint8_t *out,
                                  size_t len)
{
  if (c == 0)
    return;
  if (c == 1) {
    for (size_t i = 0; i < len; i++)
      out[i] ^= d[i];
    return;
  }
  const uint8_t *logt = ecref_gf_log_table();
  const uint8_t *expt = ecref_gf_exp_table();
  uint8_t lc = logt[c];
  for (size_t i = 0; i < len; i++) {
    uint8_t v = d[i];
    if (v) {
      int s = lc + logt[v];
      if (s >= 255)
        s -= 255;
      out[i] ^= expt[s];
    }
  }
}

#if defined(__x86_64__)
__attribute__((target("avx2"))) static void
region_mul_acc_avx2(const uint8_t *tbl, const uint8_t *d, uint8_t *out,
                    size_t len)
{
  const __m256i tlo = _mm256_broadcastsi128_si256(
      _mm_loadu_si128((const __m128i *)tbl));
  const __m256i thi = _mm256_broadcastsi128_si256(
      _mm_loadu_si128((const __m128i *)(tbl + 16)));
  const __m256i mask = _mm256_set1_epi8(0x0f);
  size_t i = 0;
  for (; i + 32 <= len; i += 32) {
    __m256i v = _mm256_loadu_si256((const __m256i *)(d + i));
    __m256i lo = _mm256_and_si256(v, mask);
    __m256i hi = _mm256_and_si256(_mm256_srli_epi64(v, 4), mask);
    __m256i p = _mm256_xor_si256(_mm256_shuffle_epi8(tlo, lo),
                                 _mm256_shuffle_epi8(thi, hi));
    __m256i o = _mm256_loadu_si256((const __m256i *)(out + i));
    _mm256_storeu_si256((__m256i *)(out + i), _mm256_xor_si256(o, p));
  }
  if (i < len) {
    /* scalar tail via the table */
    for (; i < len; i++) {
      uint8_t v = d[i];
      out[i] ^= (uint8_t)(tbl[v & 0x0f] ^ tbl[16 + (v >> 4)]);
    }
  }
}

__attribute__((target("avx2"))) static void
region_xor_acc_avx2(const uint8_t *d, uint8_t *out, size_t len)
{
  size_t i = 0;
  for (; i + 32 <= len; i += 32) {
    __m256i v = _mm256_loadu_si256((const __m256i *)(d + i));
    __m256i o = _mm256_loadu_si256((const __m256i *)(out + i));
    _mm256_storeu_si256((__m256i *)(out + i), _mm256_xor_si256(o, v));
  }
  for (; i < len; i++)
    out[i] ^= d[i];
}

static int have_avx2(void)
{
  static int v = -1;
  if (v < 0)
    v = __builtin_cpu_supports("avx2") ? 1 : 0;
  return v;
}
#else
static int have_avx2(void) { return 0; }
#endif

#define ECCPU_BLK 16384

/* One cache-blocked region: parity[j][off..off+n) for all j. */
static void encode_block(int k, int m, const uint8_t *tables /* m*k*32 */,
                         const uint8_t *coding_rows,
                         const uint8_t *const *data, uint8_t *const *parity,
                         size_t off, size_t n)
{
  for (int j = 0; j < m; j++) {
    uint8_t *out = parity[j] + off;
    memset(out, 0, n);
    for (int i = 0; i < k; i++) {
      if (data[i] == NULL)
        continue;
      uint8_t c = coding_rows[(size_t)j * k + i];
      if (c == 0)
        continue;
#if defined(__x86_64__)
      if (have_avx2()) {
        if (c == 1)
          region_xor_acc_avx2(data[i] + off, out, n);
        else
          region_mul_acc_avx2(tables + ((size_t)j * k + i) * 32,
                              data[i] + off, out, n);
        continue;
      }
#endif
      region_mul_acc_scalar(c, data[i] + off, out, n);
    }
  }
}

/* Parallel first-touch of a batch buffer: page placement follows the
 * first writer (Linux first-touch NUMA policy), so a buffer filled by a
 * single numpy thread lands on ONE node and the OpenMP encode then
 * starves every other socket's cores (measured on a GPU-box host: 101
 * GiB/s at 32 threads vs 46 at 64 with single-thread fill). Touch pages
 * with the same static partitioning the encode loops use BEFORE filling
 * content; later writes do not move pages. */
int eccpu_first_touch(uint8_t *base, size_t bytes)
{
  const size_t blk = 2u << 20; /* 2 MiB */
  long nblk = (long)((bytes + blk - 1) / blk);
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
  for (long b = 0; b < nblk; b++) {
    size_t off = (size_t)b * blk;
    size_t n = bytes - off < blk ? bytes - off : blk;
    memset(base + off, 0, n);
  }
  return 0;
}

int eccpu_threads(void)
{
#ifdef _OPENMP
  return omp_get_max_threads();
#else
  return 1;
#endif
}

/* Encode a batch laid out exactly like the GPU path's device buffer:
 * stripe s, chunk c at base + (s*(k+m)+c)*chunk_bytes; chunks 0..k-1 are
 * data, k..k+m-1 parity. OpenMP-parallel over stripes. */
int eccpu_encode_batch(int technique, int k, int m, uint8_t *base,
                       long n_stripes, size_t chunk_bytes)
{
  uint8_t gen[255 * 255];
  if (ecref_matrix(technique, gen, k, m) != 0)
    return -1;
  const uint8_t *rows = gen + (size_t)k * k;
  uint8_t tables[255 * 32 * 8]; /* m*k*32, bounded by k,m <= 32 */
  for (int j = 0; j < m; j++)
    for (int i = 0; i < k; i++)
      mul_table_32(rows[(size_t)j * k + i], tables + ((size_t)j * k + i) * 32);

  /* flatten (stripe, block) so many threads stay fed even for small
   * stripe counts (the GPU-box host has 256 hardware threads) */
  long bpc = (long)((chunk_bytes + ECCPU_BLK - 1) / ECCPU_BLK);
  long total = n_stripes * bpc;
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic, 4)
#endif
  for (long w = 0; w < total; w++) {
    long s = w / bpc, b = w % bpc;
    const uint8_t *data[255];
    uint8_t *parity[255];
    uint8_t *stripe = base + (size_t)s * (k + m) * chunk_bytes;
    for (int i = 0; i < k; i++)
      data[i] = stripe + (size_t)i * chunk_bytes;
    for (int j = 0; j < m; j++)
      parity[j] = stripe + (size_t)(k + j) * chunk_bytes;
    size_t off = (size_t)b * ECCPU_BLK;
    size_t n = chunk_bytes - off < ECCPU_BLK ? chunk_bytes - off : ECCPU_BLK;
    encode_block(k, m, tables, rows, data, parity, off, n);
  }
  return 0;
}

/* Decode a batch with a uniform erasure pattern (present[] over k+m ids),
 * same layout; erased chunks are reconstructed in place. */
int eccpu_decode_batch(int technique, int k, int m, uint8_t *base,
                       const uint8_t *present, long n_stripes,
                       size_t chunk_bytes)
{
  /* Build the composed decode rows once (same math as ecref_decode). */
  uint8_t gen[255 * 255];
  if (ecref_matrix(technique, gen, k, m) != 0)
    return -1;
  int n = k + m, nerrs = 0;
  int erasures[255], decode_index[255];
  for (int i = 0; i < n; i++)
    if (!present[i])
      erasures[nerrs++] = i;
  if (nerrs == 0)
    return 0;
  if (nerrs > m)
    return -1;
  {
    int r = 0;
    for (int i = 0; i < k; i++, r++) {
      while (r < n && !present[r])
        r++;
      if (r >= n)
        return -1;
      decode_index[i] = r;
    }
  }
  uint8_t b[255 * 255], d[255 * 255], c[255 * 255];
  for (int i = 0; i < k; i++)
    memcpy(&b[(size_t)i * k], &gen[(size_t)decode_index[i] * k], k);
  if (ecref_gf_invert_matrix(b, d, k) != 0)
    return -1;
  for (int p = 0; p < nerrs; p++) {
    if (erasures[p] < k) {
      memcpy(&c[(size_t)p * k], &d[(size_t)erasures[p] * k], k);
    } else {
      for (int i = 0; i < k; i++) {
        uint8_t s = 0;
        for (int j = 0; j < k; j++)
          s ^= ecref_gf_mul(d[(size_t)j * k + i],
                            gen[(size_t)erasures[p] * k + j]);
        c[(size_t)p * k + i] = s;
      }
    }
  }
  uint8_t tables[255 * 32 * 8];
  for (int j = 0; j < nerrs; j++)
    for (int i = 0; i < k; i++)
      mul_table_32(c[(size_t)j * k + i], tables + ((size_t)j * k + i) * 32);

  long bpc = (long)((chunk_bytes + ECCPU_BLK - 1) / ECCPU_BLK);
  long total = n_stripes * bpc;
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic, 4)
#endif
  for (long w = 0; w < total; w++) {
    long s = w / bpc, b = w % bpc;
    const uint8_t *src[255];
    uint8_t *dst[255];
    uint8_t *stripe = base + (size_t)s * (k + m) * chunk_bytes;
    for (int i = 0; i < k; i++)
      src[i] = stripe + (size_t)decode_index[i] * chunk_bytes;
    for (int p = 0; p < nerrs; p++)
      dst[p] = stripe + (size_t)erasures[p] * chunk_bytes;
    size_t off = (size_t)b * ECCPU_BLK;
    size_t n = chunk_bytes - off < ECCPU_BLK ? chunk_bytes - off : ECCPU_BLK;
    encode_block(k, nerrs, tables, c, src, dst, off, n);
  }
  return 0;
}
