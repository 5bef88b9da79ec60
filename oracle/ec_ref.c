/* ec_ref.c — CPU oracle. TEST INFRASTRUCTURE ONLY (see ec_ref.h header for
 * scope, reference citations and the parity-pinning status). */
#include "ec_ref.h"

#include <errno.h>
#include <stdlib.h>
#include <string.h>

/* ---------------- GF(2^8) over 0x11d ---------------- */

static uint8_t gf_log[256];
static uint8_t gf_exp[256];
static int gf_ready = 0;

void ecref_gf_init(void)
{
  if (gf_ready)
    return;
  /* generator 2, primitive polynomial x^8+x^4+x^3+x^2+1 (0x11d): the field
   * of gf-complete w=8 and isa-l (published; submodules absent, ec_ref.h). */
  unsigned v = 1;
  for (int i = 0; i < 255; i++) {
    gf_exp[i] = (uint8_t)v;
    gf_log[v] = (uint8_t)i;
    v <<= 1;
    if (v & 0x100)
      v ^= 0x11d;
  }
  gf_exp[255] = gf_exp[0]; /* convenience wrap */
  gf_log[0] = 0;           /* undefined; callers must special-case 0 */
  gf_ready = 1;
}

uint8_t ecref_gf_mul(uint8_t a, uint8_t b)
{
  if (a == 0 || b == 0)
    return 0;
  int s = gf_log[a] + gf_log[b];
  if (s >= 255)
    s -= 255;
  return gf_exp[s];
}

uint8_t ecref_gf_inv(uint8_t a)
{
  if (a == 0)
    return 0; /* isa-l gf_inv(0) returns 0 */
  return gf_exp[255 - gf_log[a]];
}

const uint8_t *ecref_gf_log_table(void) { ecref_gf_init(); return gf_log; }
const uint8_t *ecref_gf_exp_table(void) { ecref_gf_init(); return gf_exp; }

static uint8_t gf_div(uint8_t a, uint8_t b)
{
  if (a == 0)
    return 0;
  /* b == 0 is a caller bug; mirror gf arithmetic by returning 0 */
  if (b == 0)
    return 0;
  int s = gf_log[a] - gf_log[b];
  if (s < 0)
    s += 255;
  return gf_exp[s];
}

/* ---------------- generator matrices ---------------- */

/* Restates isa-l gf_gen_rs_matrix (used at ErasureCodeIsa.cc:659):
 * identity on top; row k+i = [g^0, g^1, ..., g^(k-1)] with g = 2^i.
 * The first coding row (i=0, g=1) is all ones => parity0 = XOR of data,
 * the structural fact the reference's single-erasure fast path relies on
 * (ErasureCodeIsa.cc:395-456). */
int ecref_matrix_rs_vandermonde_isa(uint8_t *a, int k, int m)
{
  ecref_gf_init();
  if (k < 1 || m < 0 || k + m > 255)
    return -EINVAL;
  memset(a, 0, (size_t)(k + m) * k);
  for (int i = 0; i < k; i++)
    a[(size_t)k * i + i] = 1;
  uint8_t gen = 1;
  for (int i = k; i < k + m; i++) {
    uint8_t p = 1;
    for (int j = 0; j < k; j++) {
      a[(size_t)k * i + j] = p;
      p = ecref_gf_mul(p, gen);
    }
    gen = ecref_gf_mul(gen, 2);
  }
  return 0;
}

/* Restates isa-l gf_gen_cauchy1_matrix (ErasureCodeIsa.cc:661):
 * identity on top; coding element (i,j) = 1/(i XOR j), i in [k, k+m). */
int ecref_matrix_cauchy_isa(uint8_t *a, int k, int m)
{
  ecref_gf_init();
  if (k < 1 || m < 0 || k + m > 255)
    return -EINVAL;
  memset(a, 0, (size_t)(k + m) * k);
  for (int i = 0; i < k; i++)
    a[(size_t)k * i + i] = 1;
  uint8_t *p = a + (size_t)k * k;
  for (int i = k; i < k + m; i++)
    for (int j = 0; j < k; j++)
      *p++ = ecref_gf_inv((uint8_t)(i ^ j));
  return 0;
}

/* Restates jerasure-2.0 reed_sol.c (submodule absent; published algorithm):
 * reed_sol_extended_vandermonde_matrix -> systematic reduction ->
 * first-coding-row normalised to all ones. Used by the reference at
 * ErasureCodeJerasure.cc:431-435 via reed_sol_vandermonde_coding_matrix.
 * w=8 only here. Byte-level agreement with compiled jerasure is an
 * assumption to be spot-verified (ec_ref.h); the structural properties
 * (systematic top identity, all-ones first coding row, MDS) are test-pinned.
 */
int ecref_matrix_rs_vandermonde_jerasure(uint8_t *a, int k, int m)
{
  ecref_gf_init();
  int rows = k + m, cols = k;
  if (k < 1 || m < 0 || rows > 255)
    return -EINVAL;

  /* extended Vandermonde: row 0 = e_0; rows 1..rows-2: [i^0, i^1, ...];
   * last row = e_{cols-1}. */
  for (int j = 0; j < cols; j++)
    a[j] = (j == 0);
  if (rows > 1)
    for (int j = 0; j < cols; j++)
      a[(size_t)(rows - 1) * cols + j] = (j == cols - 1);
  for (int i = 1; i < rows - 1; i++) {
    uint8_t v = 1;
    for (int j = 0; j < cols; j++) {
      a[(size_t)i * cols + j] = v;
      v = ecref_gf_mul(v, (uint8_t)i);
    }
  }

  /* Systematic reduction by column operations (row swap for pivoting),
   * mirroring reed_sol_big_vandermonde_distribution_matrix. */
  for (int i = 1; i < cols; i++) {
    /* pivot: find row j >= i with a[j][i] != 0, swap into row i */
    int j = i;
    while (j < rows && a[(size_t)j * cols + i] == 0)
      j++;
    if (j >= rows)
      return -EDOM; /* cannot happen for valid (k,m,w) */
    if (j != i)
      for (int c = 0; c < cols; c++) {
        uint8_t t = a[(size_t)j * cols + c];
        a[(size_t)j * cols + c] = a[(size_t)i * cols + c];
        a[(size_t)i * cols + c] = t;
      }
    /* scale column i so a[i][i] == 1 */
    uint8_t piv = a[(size_t)i * cols + i];
    if (piv != 1) {
      uint8_t inv = gf_div(1, piv);
      for (int r = 0; r < rows; r++)
        a[(size_t)r * cols + i] = ecref_gf_mul(inv, a[(size_t)r * cols + i]);
    }
    /* eliminate row i outside the pivot column: col j ^= a[i][j] * col i */
    for (int c = 0; c < cols; c++) {
      uint8_t t = a[(size_t)i * cols + c];
      if (c != i && t != 0)
        for (int r = 0; r < rows; r++)
          a[(size_t)r * cols + c] ^=
              ecref_gf_mul(t, a[(size_t)r * cols + i]);
    }
  }

  /* Normalise the first coding row (row cols) to all ones: scale each
   * column j by 1/a[cols][j], then restore the identity diagonal by scaling
   * row j (which holds only the diagonal element) back. */
  for (int j = 0; j < cols; j++) {
    uint8_t t = a[(size_t)cols * cols + j];
    if (t != 0 && t != 1) {
      uint8_t inv = gf_div(1, t);
      for (int r = 0; r < rows; r++)
        a[(size_t)r * cols + j] = ecref_gf_mul(inv, a[(size_t)r * cols + j]);
      for (int c = 0; c < cols; c++)
        a[(size_t)j * cols + c] = ecref_gf_mul(t, a[(size_t)j * cols + c]);
    }
  }
  return 0;
}

int ecref_matrix(int technique, uint8_t *a, int k, int m)
{
  switch (technique) {
  case ECREF_T_RS_VAN_ISA:      return ecref_matrix_rs_vandermonde_isa(a, k, m);
  case ECREF_T_CAUCHY_ISA:      return ecref_matrix_cauchy_isa(a, k, m);
  case ECREF_T_RS_VAN_JERASURE: return ecref_matrix_rs_vandermonde_jerasure(a, k, m);
  default:                      return -EINVAL;
  }
}

/* Restates isa-l gf_invert_matrix (used at ErasureCodeIsa.cc:535):
 * Gauss-Jordan with row-swap pivoting. in_mat clobbered. */
int ecref_gf_invert_matrix(uint8_t *in_mat, uint8_t *out_mat, int k)
{
  ecref_gf_init();
  memset(out_mat, 0, (size_t)k * k);
  for (int i = 0; i < k; i++)
    out_mat[(size_t)k * i + i] = 1;

  for (int i = 0; i < k; i++) {
    if (in_mat[(size_t)k * i + i] == 0) {
      int j = i + 1;
      while (j < k && in_mat[(size_t)k * j + i] == 0)
        j++;
      if (j >= k)
        return -1; /* singular */
      for (int c = 0; c < k; c++) {
        uint8_t t = in_mat[(size_t)k * i + c];
        in_mat[(size_t)k * i + c] = in_mat[(size_t)k * j + c];
        in_mat[(size_t)k * j + c] = t;
        t = out_mat[(size_t)k * i + c];
        out_mat[(size_t)k * i + c] = out_mat[(size_t)k * j + c];
        out_mat[(size_t)k * j + c] = t;
      }
    }
    uint8_t piv = in_mat[(size_t)k * i + i];
    uint8_t inv = ecref_gf_inv(piv);
    for (int c = 0; c < k; c++) {
      in_mat[(size_t)k * i + c] = ecref_gf_mul(inv, in_mat[(size_t)k * i + c]);
      out_mat[(size_t)k * i + c] = ecref_gf_mul(inv, out_mat[(size_t)k * i + c]);
    }
    for (int r = 0; r < k; r++) {
      if (r == i)
        continue;
      uint8_t f = in_mat[(size_t)k * r + i];
      if (f == 0)
        continue;
      for (int c = 0; c < k; c++) {
        in_mat[(size_t)k * r + c] ^= ecref_gf_mul(f, in_mat[(size_t)k * i + c]);
        out_mat[(size_t)k * r + c] ^= ecref_gf_mul(f, out_mat[(size_t)k * i + c]);
      }
    }
  }
  return 0;
}

/* ---------------- region ops ---------------- */

void ecref_encode(int k, int m, const uint8_t *coding_rows,
                  const uint8_t *const *data, uint8_t *const *parity,
                  size_t len)
{
  ecref_gf_init();
  for (int j = 0; j < m; j++) {
    uint8_t *out = parity[j];
    memset(out, 0, len);
    for (int i = 0; i < k; i++) {
      const uint8_t *d = data[i];
      if (d == NULL)
        continue; /* zeros chunk: contributes nothing */
      uint8_t c = coding_rows[(size_t)j * k + i];
      if (c == 0)
        continue;
      if (c == 1) {
        for (size_t b = 0; b < len; b++)
          out[b] ^= d[b];
      } else {
        const uint8_t lc = gf_log[c];
        for (size_t b = 0; b < len; b++) {
          uint8_t v = d[b];
          if (v) {
            int s = lc + gf_log[v];
            if (s >= 255)
              s -= 255;
            out[b] ^= gf_exp[s];
          }
        }
      }
    }
  }
}

int ecref_decode(int technique, int k, int m,
                 uint8_t *const *chunks, const uint8_t *present,
                 size_t len)
{
  ecref_gf_init();
  uint8_t gen[255 * 255];
  if (ecref_matrix(technique, gen, k, m) != 0)
    return -1;

  int n = k + m;
  int nerrs = 0;
  int erasures[255];
  for (int i = 0; i < n; i++)
    if (!present[i])
      erasures[nerrs++] = i;
  if (nerrs == 0)
    return 0;
  if (nerrs > m)
    return -1;

  /* survivor selection: first k present in id order
   * (ErasureCodeIsa.cc:483-494 decode_index / ErasureCode.cc:154-170). */
  int decode_index[255];
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

  /* b = survivor rows of the generator; d = b^-1
   * (ErasureCodeIsa.cc:518-535). */
  uint8_t b[255 * 255], d[255 * 255], c[255 * 255];
  for (int i = 0; i < k; i++)
    memcpy(&b[(size_t)i * k], &gen[(size_t)decode_index[i] * k], k);
  if (ecref_gf_invert_matrix(b, d, k) != 0)
    return -1;

  /* decode rows: data erasure -> row of d; parity erasure -> generator row
   * composed with d (ErasureCodeIsa.cc:540-557). */
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

  const uint8_t *src[255];
  uint8_t *dst[255];
  for (int i = 0; i < k; i++)
    src[i] = chunks[decode_index[i]];
  for (int p = 0; p < nerrs; p++)
    dst[p] = chunks[erasures[p]];
  ecref_encode(k, nerrs, c, src, dst, len);
  return 0;
}

void ecref_xor_region(const uint8_t *a, const uint8_t *b, uint8_t *out, size_t len)
{
  for (size_t i = 0; i < len; i++)
    out[i] = a[i] ^ b[i];
}

void ecref_region_mul_xor(uint8_t coeff, const uint8_t *delta, uint8_t *parity,
                          size_t len)
{
  ecref_gf_init();
  if (coeff == 0)
    return;
  if (coeff == 1) {
    for (size_t i = 0; i < len; i++)
      parity[i] ^= delta[i];
    return;
  }
  const uint8_t lc = gf_log[coeff];
  for (size_t i = 0; i < len; i++) {
    uint8_t v = delta[i];
    if (v) {
      int s = lc + gf_log[v];
      if (s >= 255)
        s -= 255;
      parity[i] ^= gf_exp[s];
    }
  }
}

/* ErasureCodeIsa.cc:65-79: chunk = ceil(width/k) rounded up to 32. */
unsigned ecref_chunk_size_isa(int k, unsigned stripe_width)
{
  unsigned chunk = (stripe_width + k - 1) / k;
  unsigned mod = chunk % 32u;
  if (mod)
    chunk += 32u - mod;
  return chunk;
}

/* ErasureCodeJerasure.cc:85-108 (per_chunk_alignment=false default):
 * pad the stripe to a multiple of get_alignment() = k*w*sizeof(int)
 * (w*sizeof(int) % 16 == 0 for w=8, so no LARGEST_VECTOR_WORDSIZE bump),
 * then divide by k. */
unsigned ecref_chunk_size_jerasure(int k, int w, unsigned stripe_width)
{
  unsigned alignment = (unsigned)k * w * 4u;
  if ((w * 4u) % 16u)
    alignment = (unsigned)k * w * 16u;
  unsigned tail = stripe_width % alignment;
  unsigned padded = stripe_width + (tail ? alignment - tail : 0);
  return padded / k;
}

/* ---------------- GF(2^16) (w=16 jerasure RS-van) ---------------- */

static uint16_t *gf16_log = NULL;  /* 65536 u16 */
static uint16_t *gf16_exp = NULL;
static int gf16_ready = 0;

void ecref_gf16_init(void)
{
  if (gf16_ready)
    return;
  gf16_log = (uint16_t *)malloc(65536 * 2);
  gf16_exp = (uint16_t *)malloc(65536 * 2);
  unsigned v = 1;
  for (int i = 0; i < 65535; i++) {
    gf16_exp[i] = (uint16_t)v;
    gf16_log[v] = (uint16_t)i;
    v <<= 1;
    if (v & 0x10000)
      v ^= 0x1100B; /* gf-complete w=16 default polynomial */
  }
  gf16_exp[65535] = gf16_exp[0];
  gf16_log[0] = 0;
  gf16_ready = 1;
}

uint16_t ecref_gf16_mul(uint16_t a, uint16_t b)
{
  if (!a || !b)
    return 0;
  int s = gf16_log[a] + gf16_log[b];
  if (s >= 65535)
    s -= 65535;
  return gf16_exp[s];
}

uint16_t ecref_gf16_inv(uint16_t a)
{
  if (!a)
    return 0;
  return gf16_exp[65535 - gf16_log[a]];
}

static uint16_t gf16_div(uint16_t a, uint16_t b)
{
  if (!a || !b)
    return 0;
  int s = gf16_log[a] - gf16_log[b];
  if (s < 0)
    s += 65535;
  return gf16_exp[s];
}

/* same jerasure big-Vandermonde algorithm as the w=8 restatement above,
 * in GF(2^16) */
int ecref_matrix_rs_vandermonde_jerasure_w16(uint16_t *a, int k, int m)
{
  ecref_gf16_init();
  int rows = k + m, cols = k;
  if (k < 1 || m < 0 || rows > 65535)
    return -EINVAL;
  for (int j = 0; j < cols; j++)
    a[j] = (j == 0);
  if (rows > 1)
    for (int j = 0; j < cols; j++)
      a[(size_t)(rows - 1) * cols + j] = (j == cols - 1);
  for (int i = 1; i < rows - 1; i++) {
    uint16_t v = 1;
    for (int j = 0; j < cols; j++) {
      a[(size_t)i * cols + j] = v;
      v = ecref_gf16_mul(v, (uint16_t)i);
    }
  }
  for (int i = 1; i < cols; i++) {
    int j = i;
    while (j < rows && a[(size_t)j * cols + i] == 0)
      j++;
    if (j >= rows)
      return -EDOM;
    if (j != i)
      for (int c = 0; c < cols; c++) {
        uint16_t t = a[(size_t)j * cols + c];
        a[(size_t)j * cols + c] = a[(size_t)i * cols + c];
        a[(size_t)i * cols + c] = t;
      }
    uint16_t piv = a[(size_t)i * cols + i];
    if (piv != 1) {
      uint16_t inv = gf16_div(1, piv);
      for (int r = 0; r < rows; r++)
        a[(size_t)r * cols + i] = ecref_gf16_mul(inv, a[(size_t)r * cols + i]);
    }
    for (int c = 0; c < cols; c++) {
      uint16_t t = a[(size_t)i * cols + c];
      if (c != i && t != 0)
        for (int r = 0; r < rows; r++)
          a[(size_t)r * cols + c] ^=
              ecref_gf16_mul(t, a[(size_t)r * cols + i]);
    }
  }
  for (int j = 0; j < cols; j++) {
    uint16_t t = a[(size_t)cols * cols + j];
    if (t != 0 && t != 1) {
      uint16_t inv = gf16_div(1, t);
      for (int r = 0; r < rows; r++)
        a[(size_t)r * cols + j] = ecref_gf16_mul(inv, a[(size_t)r * cols + j]);
      for (int c = 0; c < cols; c++)
        a[(size_t)j * cols + c] = ecref_gf16_mul(t, a[(size_t)j * cols + c]);
    }
  }
  return 0;
}

void ecref_encode16(int k, int m, const uint16_t *coding_rows,
                    const uint8_t *const *data, uint8_t *const *parity,
                    size_t len)
{
  ecref_gf16_init();
  size_t n = len / 2;
  for (int j = 0; j < m; j++) {
    uint16_t *out = (uint16_t *)parity[j];
    memset(out, 0, len);
    for (int i = 0; i < k; i++) {
      if (data[i] == NULL)
        continue;
      uint16_t c = coding_rows[(size_t)j * k + i];
      if (c == 0)
        continue;
      const uint16_t *d = (const uint16_t *)data[i];
      if (c == 1) {
        for (size_t b = 0; b < n; b++)
          out[b] ^= d[b];
      } else {
        uint32_t lc = gf16_log[c];
        for (size_t b = 0; b < n; b++) {
          uint16_t v = d[b];
          if (v) {
            unsigned s = lc + gf16_log[v];
            if (s >= 65535)
              s -= 65535;
            out[b] ^= gf16_exp[s];
          }
        }
      }
    }
  }
}

static int gf16_invert_matrix(uint16_t *in, uint16_t *out, int k)
{
  memset(out, 0, (size_t)k * k * 2);
  for (int i = 0; i < k; i++)
    out[(size_t)i * k + i] = 1;
  for (int i = 0; i < k; i++) {
    if (!in[(size_t)i * k + i]) {
      int j = i + 1;
      while (j < k && !in[(size_t)j * k + i])
        j++;
      if (j >= k)
        return -1;
      for (int c = 0; c < k; c++) {
        uint16_t t = in[(size_t)i * k + c];
        in[(size_t)i * k + c] = in[(size_t)j * k + c];
        in[(size_t)j * k + c] = t;
        t = out[(size_t)i * k + c];
        out[(size_t)i * k + c] = out[(size_t)j * k + c];
        out[(size_t)j * k + c] = t;
      }
    }
    uint16_t inv = ecref_gf16_inv(in[(size_t)i * k + i]);
    for (int c = 0; c < k; c++) {
      in[(size_t)i * k + c] = ecref_gf16_mul(inv, in[(size_t)i * k + c]);
      out[(size_t)i * k + c] = ecref_gf16_mul(inv, out[(size_t)i * k + c]);
    }
    for (int r = 0; r < k; r++) {
      if (r == i)
        continue;
      uint16_t f = in[(size_t)r * k + i];
      if (!f)
        continue;
      for (int c = 0; c < k; c++) {
        in[(size_t)r * k + c] ^= ecref_gf16_mul(f, in[(size_t)i * k + c]);
        out[(size_t)r * k + c] ^= ecref_gf16_mul(f, out[(size_t)i * k + c]);
      }
    }
  }
  return 0;
}

int ecref_decode16(uint8_t *const *chunks, const uint8_t *present,
                   int k, int m, size_t len)
{
  ecref_gf16_init();
  uint16_t *gen = (uint16_t *)malloc((size_t)(k + m) * k * 2);
  if (!gen)
    return -ENOMEM;
  if (ecref_matrix_rs_vandermonde_jerasure_w16(gen, k, m) != 0) {
    free(gen);
    return -1;
  }
  int n = k + m, nerrs = 0, erasures[255], decode_index[255];
  for (int i = 0; i < n; i++)
    if (!present[i])
      erasures[nerrs++] = i;
  if (nerrs == 0) {
    free(gen);
    return 0;
  }
  if (nerrs > m) {
    free(gen);
    return -1;
  }
  {
    int r = 0;
    for (int i = 0; i < k; i++, r++) {
      while (r < n && !present[r])
        r++;
      if (r >= n) {
        free(gen);
        return -1;
      }
      decode_index[i] = r;
    }
  }
  uint16_t *b = (uint16_t *)malloc((size_t)k * k * 2);
  uint16_t *d = (uint16_t *)malloc((size_t)k * k * 2);
  uint16_t *c = (uint16_t *)malloc((size_t)nerrs * k * 2);
  if (!b || !d || !c) {
    free(gen); free(b); free(d); free(c);
    return -ENOMEM;
  }
  for (int i = 0; i < k; i++)
    memcpy(&b[(size_t)i * k], &gen[(size_t)decode_index[i] * k], k * 2);
  if (gf16_invert_matrix(b, d, k) != 0) {
    free(gen); free(b); free(d); free(c);
    return -1;
  }
  for (int p = 0; p < nerrs; p++) {
    if (erasures[p] < k) {
      memcpy(&c[(size_t)p * k], &d[(size_t)erasures[p] * k], k * 2);
    } else {
      for (int i = 0; i < k; i++) {
        uint16_t s = 0;
        for (int j = 0; j < k; j++)
          s ^= ecref_gf16_mul(d[(size_t)j * k + i],
                              gen[(size_t)erasures[p] * k + j]);
        c[(size_t)p * k + i] = s;
      }
    }
  }
  const uint8_t *src[255];
  uint8_t *dst[255];
  for (int i = 0; i < k; i++)
    src[i] = chunks[decode_index[i]];
  for (int p = 0; p < nerrs; p++)
    dst[p] = chunks[erasures[p]];
  ecref_encode16(k, nerrs, c, src, dst, len);
  free(gen); free(b); free(d); free(c);
  return 0;
}

/* ---------------- jerasure bitmatrix (Cauchy-original) family ----------- */

/* cauchy.c cauchy_original_coding_matrix: m[i][j] = 1/(i XOR (m+j)). */
int ecref_matrix_cauchy_orig_jerasure(uint8_t *coding, int k, int m)
{
  ecref_gf_init();
  if (k < 1 || m < 1 || k + m > 255)
    return -EINVAL;
  for (int i = 0; i < m; i++)
    for (int j = 0; j < k; j++) {
      int x = i ^ (m + j);
      if (x == 0)
        return -EDOM;
      coding[(size_t)i * k + j] = ecref_gf_inv((uint8_t)x);
    }
  return 0;
}

/* cauchy.c cauchy_n_ones, w=8: ones in the companion bitmatrix of e,
 * i.e. sum over c of popcount(e * 2^c) (column c of the block is the bit
 * pattern of e*2^c — ecref_matrix_to_bitmatrix below). jerasure computes
 * the same value with an incremental recurrence; this is the direct form. */
int ecref_cauchy_n_ones(uint8_t e)
{
  ecref_gf_init();
  int no = 0;
  uint8_t v = e;
  for (int c = 0; c < 8; c++) {
    no += __builtin_popcount(v);
    v = ecref_gf_mul(v, 2);
  }
  return no;
}

/* cauchy.c cauchy_improve_coding_matrix: (1) scale each column so row 0
 * becomes all ones; (2) for each later row, divide by the first element
 * that strictly minimises the row's total bitmatrix ones. */
void ecref_cauchy_improve_matrix(uint8_t *coding, int k, int m)
{
  ecref_gf_init();
  for (int j = 0; j < k; j++) {
    if (coding[j] != 1) {
      uint8_t tmp = ecref_gf_inv(coding[j]);
      for (int i = 1; i < m; i++)
        coding[(size_t)i * k + j] =
            ecref_gf_mul(coding[(size_t)i * k + j], tmp);
      coding[j] = 1;
    }
  }
  for (int i = 1; i < m; i++) {
    uint8_t *row = coding + (size_t)i * k;
    int bno = 0, bno_index = -1;
    for (int j = 0; j < k; j++) bno += ecref_cauchy_n_ones(row[j]);
    for (int j = 0; j < k; j++) {
      if (row[j] == 1) continue;
      uint8_t tmp = ecref_gf_inv(row[j]);
      int tno = 0;
      for (int col = 0; col < k; col++)
        tno += ecref_cauchy_n_ones(ecref_gf_mul(row[col], tmp));
      if (tno < bno) { bno = tno; bno_index = j; }
    }
    if (bno_index != -1) {
      uint8_t tmp = ecref_gf_inv(row[bno_index]);
      for (int j = 0; j < k; j++) row[j] = ecref_gf_mul(row[j], tmp);
    }
  }
}

/* cauchy.c cauchy_good_general_coding_matrix, general branch (m != 2):
 * cauchy_original + improve. m == 2 would read jerasure's precomputed
 * cbest tables (unsourceable in this container) => -EDOM. */
int ecref_matrix_cauchy_good_jerasure(uint8_t *coding, int k, int m)
{
  if (m == 2)
    return -EDOM;
  int r = ecref_matrix_cauchy_orig_jerasure(coding, k, m);
  if (r)
    return r;
  ecref_cauchy_improve_matrix(coding, k, m);
  return 0;
}

/* jerasure.c jerasure_matrix_to_bitmatrix: block (i,j) column c holds the
 * bit pattern of coeff * 2^c (companion-matrix representation): bit row r
 * of the block = bit r of gf_mul(coeff, 1<<c). */
void ecref_matrix_to_bitmatrix(const uint8_t *coding, int k, int m, int w,
                               uint8_t *bitmat)
{
  ecref_gf_init();
  int W = k * w;
  for (int i = 0; i < m; i++)
    for (int j = 0; j < k; j++) {
      uint8_t v = coding[(size_t)i * k + j];
      for (int c = 0; c < w; c++) {
        for (int r = 0; r < w; r++)
          bitmat[(size_t)(i * w + r) * W + j * w + c] = (v >> r) & 1;
        v = ecref_gf_mul(v, 2);
      }
    }
}

static void bitmatrix_dotprod(int n_src, int w, const uint8_t *rows /* w x n_src*w */,
                              const uint8_t *const *src, uint8_t *dst,
                              size_t size, int packetsize)
{
  size_t super = (size_t)w * packetsize;
  for (size_t off = 0; off < size; off += super) {
    for (int r = 0; r < w; r++) {
      uint8_t *d = dst + off + (size_t)r * packetsize;
      memset(d, 0, packetsize);
      const uint8_t *row = rows + (size_t)r * n_src * w;
      for (int j = 0; j < n_src; j++) {
        if (src[j] == NULL)
          continue; /* zeros chunk */
        for (int c = 0; c < w; c++) {
          if (!row[j * w + c])
            continue;
          const uint8_t *s = src[j] + off + (size_t)c * packetsize;
          for (int b = 0; b < packetsize; b++)
            d[b] ^= s[b];
        }
      }
    }
  }
}

int ecref_bitmatrix_encode(int k, int m, int w, const uint8_t *bitmat,
                           const uint8_t *const *data, uint8_t *const *coding,
                           size_t size, int packetsize)
{
  if (packetsize <= 0 || size % ((size_t)w * packetsize))
    return -EINVAL;
  for (int i = 0; i < m; i++)
    bitmatrix_dotprod(k, w, bitmat + (size_t)(i * w) * k * w, data,
                      coding[i], size, packetsize);
  return 0;
}

/* GF(2) Gauss-Jordan inversion of an n x n bit matrix (bytes 0/1). */
static int gf2_invert(uint8_t *a, uint8_t *inv, int n)
{
  memset(inv, 0, (size_t)n * n);
  for (int i = 0; i < n; i++)
    inv[(size_t)i * n + i] = 1;
  for (int i = 0; i < n; i++) {
    if (!a[(size_t)i * n + i]) {
      int j = i + 1;
      while (j < n && !a[(size_t)j * n + i])
        j++;
      if (j >= n)
        return -1;
      for (int c = 0; c < n; c++) {
        uint8_t t = a[(size_t)i * n + c];
        a[(size_t)i * n + c] = a[(size_t)j * n + c];
        a[(size_t)j * n + c] = t;
        t = inv[(size_t)i * n + c];
        inv[(size_t)i * n + c] = inv[(size_t)j * n + c];
        inv[(size_t)j * n + c] = t;
      }
    }
    for (int r = 0; r < n; r++) {
      if (r == i || !a[(size_t)r * n + i])
        continue;
      for (int c = 0; c < n; c++) {
        a[(size_t)r * n + c] ^= a[(size_t)i * n + c];
        inv[(size_t)r * n + c] ^= inv[(size_t)i * n + c];
      }
    }
  }
  return 0;
}

int ecref_bitmatrix_decode(int k, int m, int w, const uint8_t *bitmat,
                           uint8_t *const *chunks, const uint8_t *present,
                           size_t size, int packetsize)
{
  if (packetsize <= 0 || size % ((size_t)w * packetsize))
    return -EINVAL;
  int n = k + m, W = k * w;
  int nerrs = 0, erasures[255], decode_index[255];
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
  /* survivor bit matrix B (W x W): data survivor -> identity block row,
   * coding survivor -> its bitmatrix rows */
  /* casts keep this file compilable as C and as C++ (the oracle fixture
   * plugin builds it with g++) */
  uint8_t *B = (uint8_t *)malloc((size_t)W * W);
  uint8_t *D = (uint8_t *)malloc((size_t)W * W);
  uint8_t *rows = (uint8_t *)malloc((size_t)m * w * W);
  if (!B || !D || !rows) {
    free(B); free(D); free(rows);
    return -ENOMEM;
  }
  memset(B, 0, (size_t)W * W);
  for (int i = 0; i < k; i++) {
    int id = decode_index[i];
    if (id < k) {
      for (int r = 0; r < w; r++)
        B[(size_t)(i * w + r) * W + id * w + r] = 1;
    } else {
      memcpy(&B[(size_t)(i * w) * W], &bitmat[(size_t)((id - k) * w) * W],
             (size_t)w * W);
    }
  }
  if (gf2_invert(B, D, W) != 0) {
    free(B); free(D); free(rows);
    return -1;
  }
  const uint8_t *src[255];
  for (int i = 0; i < k; i++)
    src[i] = chunks[decode_index[i]];

  for (int p = 0; p < nerrs; p++) {
    int e = erasures[p];
    uint8_t *out_rows = rows + (size_t)p * w * W;
    if (e < k) {
      memcpy(out_rows, &D[(size_t)(e * w) * W], (size_t)w * W);
    } else {
      /* lost coding: compose its bitmatrix rows with D over GF(2) */
      const uint8_t *C = &bitmat[(size_t)((e - k) * w) * W];
      for (int r = 0; r < w; r++)
        for (int c = 0; c < W; c++) {
          uint8_t s = 0;
          for (int t = 0; t < W; t++)
            s ^= C[(size_t)r * W + t] & D[(size_t)t * W + c];
          out_rows[(size_t)r * W + c] = s;
        }
    }
  }
  for (int p = 0; p < nerrs; p++)
    bitmatrix_dotprod(k, w, rows + (size_t)p * w * W, src,
                      chunks[erasures[p]], size, packetsize);
  free(B); free(D); free(rows);
  return 0;
}
