// gf.cpp — see gf.h. Product-side host GF(2^8) + matrix derivation.
#include "gf.h"

#include <cstring>
#include <algorithm>
#include <limits>

namespace ecx {

GF8::GF8() {
  unsigned v = 1;
  for (int i = 0; i < 255; i++) {
    exp[i] = (uint8_t)v;
    log[v] = (uint8_t)i;
    v <<= 1;
    if (v & 0x100) v ^= 0x11d;
  }
  exp[255] = exp[0];
  log[0] = 0;
}

const GF8 &gf8() {
  static const GF8 f;
  return f;
}

// isa-l gf_gen_rs_matrix (ErasureCodeIsa.cc:659): identity ++ power rows,
// row k+i = [(2^i)^j], so the first coding row is all ones.
bool gen_matrix_rs_van_isa(std::vector<uint8_t> &a, int k, int m) {
  const GF8 &f = gf8();
  if (k < 1 || m < 0 || k + m > 255) return false;
  a.assign((size_t)(k + m) * k, 0);
  for (int i = 0; i < k; i++) a[(size_t)k * i + i] = 1;
  uint8_t gen = 1;
  for (int i = k; i < k + m; i++) {
    uint8_t p = 1;
    for (int j = 0; j < k; j++) {
      a[(size_t)k * i + j] = p;
      p = f.mul(p, gen);
    }
    gen = f.mul(gen, 2);
  }
  return true;
}

// isa-l gf_gen_cauchy1_matrix (ErasureCodeIsa.cc:661).
bool gen_matrix_cauchy_isa(std::vector<uint8_t> &a, int k, int m) {
  const GF8 &f = gf8();
  if (k < 1 || m < 0 || k + m > 255) return false;
  a.assign((size_t)(k + m) * k, 0);
  for (int i = 0; i < k; i++) a[(size_t)k * i + i] = 1;
  for (int i = k; i < k + m; i++)
    for (int j = 0; j < k; j++) a[(size_t)k * i + j] = f.inv((uint8_t)(i ^ j));
  return true;
}

// jerasure reed_sol_vandermonde_coding_matrix, w=8 (published jerasure-2.0
// algorithm; used by ErasureCodeJerasure.cc:431-435). Extended Vandermonde
// -> systematic reduction by column ops -> first coding row normalised to
// all ones.
bool gen_matrix_rs_van_jerasure(std::vector<uint8_t> &a, int k, int m) {
  const GF8 &f = gf8();
  int rows = k + m, cols = k;
  if (k < 1 || m < 0 || rows > 255) return false;
  a.assign((size_t)rows * cols, 0);
  a[0] = 1;
  if (rows > 1) a[(size_t)(rows - 1) * cols + (cols - 1)] = 1;
  for (int i = 1; i < rows - 1; i++) {
    uint8_t v = 1;
    for (int j = 0; j < cols; j++) {
      a[(size_t)i * cols + j] = v;
      v = f.mul(v, (uint8_t)i);
    }
  }
  for (int i = 1; i < cols; i++) {
    int j = i;
    while (j < rows && a[(size_t)j * cols + i] == 0) j++;
    if (j >= rows) return false;
    if (j != i)
      for (int c = 0; c < cols; c++)
        std::swap(a[(size_t)j * cols + c], a[(size_t)i * cols + c]);
    uint8_t piv = a[(size_t)i * cols + i];
    if (piv != 1) {
      uint8_t inv = f.div(1, piv);
      for (int r = 0; r < rows; r++)
        a[(size_t)r * cols + i] = f.mul(inv, a[(size_t)r * cols + i]);
    }
    for (int c = 0; c < cols; c++) {
      uint8_t t = a[(size_t)i * cols + c];
      if (c != i && t != 0)
        for (int r = 0; r < rows; r++)
          a[(size_t)r * cols + c] ^= f.mul(t, a[(size_t)r * cols + i]);
    }
  }
  for (int j = 0; j < cols; j++) {
    uint8_t t = a[(size_t)cols * cols + j];
    if (t != 0 && t != 1) {
      uint8_t inv = f.div(1, t);
      for (int r = 0; r < rows; r++)
        a[(size_t)r * cols + j] = f.mul(inv, a[(size_t)r * cols + j]);
      for (int c = 0; c < cols; c++)
        a[(size_t)j * cols + c] = f.mul(t, a[(size_t)j * cols + c]);
    }
  }
  return true;
}

bool gen_matrix(int technique, std::vector<uint8_t> &a, int k, int m) {
  switch (technique) {
    case 0: return gen_matrix_rs_van_isa(a, k, m);
    case 1: return gen_matrix_cauchy_isa(a, k, m);
    case 2: return gen_matrix_rs_van_jerasure(a, k, m);
    case 3: return gen_matrix_cauchy_orig(a, k, m);
    case 5: return gen_matrix_cauchy_good(a, k, m);
    default: return false;
  }
}

// jerasure cauchy.c cauchy_n_ones semantics, w=8: the number of ones in
// the 8x8 companion-basis bitmatrix of e. Column c of that block is the
// bit pattern of e*2^c (matrix_to_bitmatrix above), so the count is
// sum_{c=0..7} popcount(e * 2^c). (jerasure computes the same value with
// an incremental shift-and-correct recurrence; this is the direct form.)
int cauchy_n_ones(uint8_t e) {
  const GF8 &f = gf8();
  int no = 0;
  uint8_t v = e;
  for (int c = 0; c < 8; c++) {
    no += __builtin_popcount(v);
    v = f.mul(v, 2);
  }
  return no;
}

// jerasure cauchy.c cauchy_improve_coding_matrix (the behaviour the
// in-tree plugin documents for technique=cauchy_good,
// ErasureCodeJerasure.cc:537-555), operating on the m x k coding rows:
//  1) divide each COLUMN j by its row-0 element, making row 0 all ones
//     (an all-ones row bitmatrix is pure XOR — zero multiply cost);
//  2) for each row i >= 1, scan elements: dividing the row by element j
//     yields total bitmatrix ones tno = sum_col n_ones(row[col]/row[j]);
//     take the first j strictly improving the current count, then divide
//     the whole row by that element.
// Row/column scaling by nonzero constants preserves the MDS property of
// the Cauchy matrix (every square submatrix determinant scales by a
// nonzero factor).
void improve_cauchy_matrix(uint8_t *coding, int k, int m) {
  const GF8 &f = gf8();
  for (int j = 0; j < k; j++) {
    uint8_t e = coding[j];
    if (e != 1) {
      uint8_t tmp = f.div(1, e);
      for (int i = 1; i < m; i++)
        coding[(size_t)i * k + j] = f.mul(coding[(size_t)i * k + j], tmp);
      coding[j] = 1;
    }
  }
  for (int i = 1; i < m; i++) {
    uint8_t *row = coding + (size_t)i * k;
    int bno = 0;
    for (int j = 0; j < k; j++) bno += cauchy_n_ones(row[j]);
    int bno_index = -1;
    for (int j = 0; j < k; j++) {
      if (row[j] == 1) continue;
      uint8_t tmp = f.div(1, row[j]);
      int tno = 0;
      for (int col = 0; col < k; col++)
        tno += cauchy_n_ones(f.mul(row[col], tmp));
      if (tno < bno) {
        bno = tno;
        bno_index = j;
      }
    }
    if (bno_index != -1) {
      uint8_t tmp = f.div(1, row[bno_index]);
      for (int j = 0; j < k; j++) row[j] = f.mul(row[j], tmp);
    }
  }
}

// jerasure cauchy.c cauchy_good_general_coding_matrix, general branch:
// cauchy_original + improve. NOTE: for m == 2 (RAID-6) jerasure instead
// reads precomputed "cbest" element tables whose values cannot be
// faithfully restated from material available in this container; callers
// that need jerasure-bit-exact m=2 cauchy_good must gate on that
// (plugin_mi355x rejects m==2 by default).
bool gen_matrix_cauchy_good(std::vector<uint8_t> &a, int k, int m) {
  if (!gen_matrix_cauchy_orig(a, k, m)) return false;
  improve_cauchy_matrix(a.data() + (size_t)k * k, k, m);
  return true;
}

// jerasure cauchy.c cauchy_original_coding_matrix: coding (i,j) =
// inv(i XOR (m+j)).
bool gen_matrix_cauchy_orig(std::vector<uint8_t> &a, int k, int m) {
  const GF8 &f = gf8();
  if (k < 1 || m < 1 || k + m > 255) return false;
  a.assign((size_t)(k + m) * k, 0);
  for (int i = 0; i < k; i++) a[(size_t)k * i + i] = 1;
  for (int i = 0; i < m; i++)
    for (int j = 0; j < k; j++) {
      int x = i ^ (m + j);
      if (!x) return false;
      a[(size_t)(k + i) * k + j] = f.inv((uint8_t)x);
    }
  return true;
}

// jerasure.c jerasure_matrix_to_bitmatrix: block (i,j) column c = bits of
// coeff * 2^c (companion basis).
void matrix_to_bitmatrix(const uint8_t *coding_rows, int k, int m, int w,
                         std::vector<uint8_t> &bitmat) {
  const GF8 &f = gf8();
  int W = k * w;
  bitmat.assign((size_t)m * w * W, 0);
  for (int i = 0; i < m; i++)
    for (int j = 0; j < k; j++) {
      uint8_t v = coding_rows[(size_t)i * k + j];
      for (int c = 0; c < w; c++) {
        for (int r = 0; r < w; r++)
          bitmat[(size_t)(i * w + r) * W + j * w + c] = (v >> r) & 1;
        v = f.mul(v, 2);
      }
    }
}

static bool gf2_invert(std::vector<uint8_t> &a, std::vector<uint8_t> &inv,
                       int n) {
  inv.assign((size_t)n * n, 0);
  for (int i = 0; i < n; i++) inv[(size_t)i * n + i] = 1;
  for (int i = 0; i < n; i++) {
    if (!a[(size_t)i * n + i]) {
      int j = i + 1;
      while (j < n && !a[(size_t)j * n + i]) j++;
      if (j >= n) return false;
      for (int c = 0; c < n; c++) {
        std::swap(a[(size_t)i * n + c], a[(size_t)j * n + c]);
        std::swap(inv[(size_t)i * n + c], inv[(size_t)j * n + c]);
      }
    }
    for (int r = 0; r < n; r++) {
      if (r == i || !a[(size_t)r * n + i]) continue;
      for (int c = 0; c < n; c++) {
        a[(size_t)r * n + c] ^= a[(size_t)i * n + c];
        inv[(size_t)r * n + c] ^= inv[(size_t)i * n + c];
      }
    }
  }
  return true;
}

bool compose_bit_decode_rows(const std::vector<uint8_t> &bitmat, int k,
                             int m, int w, uint64_t present_mask,
                             std::vector<int> &survivors,
                             std::vector<int> &erased,
                             std::vector<uint8_t> &rows) {
  int n = k + m, W = k * w;
  survivors.clear();
  erased.clear();
  for (int i = 0; i < n; i++) {
    if (present_mask & (1ull << i)) {
      if ((int)survivors.size() < k) survivors.push_back(i);
    } else {
      erased.push_back(i);
    }
  }
  if ((int)survivors.size() < k || (int)erased.size() > m) return false;

  std::vector<uint8_t> B((size_t)W * W, 0), D;
  for (int i = 0; i < k; i++) {
    int id = survivors[i];
    if (id < k) {
      for (int r = 0; r < w; r++) B[(size_t)(i * w + r) * W + id * w + r] = 1;
    } else {
      std::memcpy(&B[(size_t)(i * w) * W],
                  &bitmat[(size_t)((id - k) * w) * W], (size_t)w * W);
    }
  }
  if (!gf2_invert(B, D, W)) return false;

  rows.assign(erased.size() * (size_t)w * W, 0);
  for (size_t p = 0; p < erased.size(); p++) {
    int e = erased[p];
    uint8_t *out = &rows[p * (size_t)w * W];
    if (e < k) {
      std::memcpy(out, &D[(size_t)(e * w) * W], (size_t)w * W);
    } else {
      const uint8_t *C = &bitmat[(size_t)((e - k) * w) * W];
      for (int r = 0; r < w; r++)
        for (int c = 0; c < W; c++) {
          uint8_t s = 0;
          for (int t = 0; t < W; t++)
            s ^= C[(size_t)r * W + t] & D[(size_t)t * W + c];
          out[(size_t)r * W + c] = s;
        }
    }
  }
  return true;
}

bool gf_invert(const uint8_t *in, uint8_t *out, int k) {
  const GF8 &f = gf8();
  std::vector<uint8_t> w(in, in + (size_t)k * k);
  std::memset(out, 0, (size_t)k * k);
  for (int i = 0; i < k; i++) out[(size_t)k * i + i] = 1;
  for (int i = 0; i < k; i++) {
    if (w[(size_t)k * i + i] == 0) {
      int j = i + 1;
      while (j < k && w[(size_t)k * j + i] == 0) j++;
      if (j >= k) return false;
      for (int c = 0; c < k; c++) {
        std::swap(w[(size_t)k * i + c], w[(size_t)k * j + c]);
        std::swap(out[(size_t)k * i + c], out[(size_t)k * j + c]);
      }
    }
    uint8_t inv = f.inv(w[(size_t)k * i + i]);
    for (int c = 0; c < k; c++) {
      w[(size_t)k * i + c] = f.mul(inv, w[(size_t)k * i + c]);
      out[(size_t)k * i + c] = f.mul(inv, out[(size_t)k * i + c]);
    }
    for (int r = 0; r < k; r++) {
      if (r == i) continue;
      uint8_t t = w[(size_t)k * r + i];
      if (!t) continue;
      for (int c = 0; c < k; c++) {
        w[(size_t)k * r + c] ^= f.mul(t, w[(size_t)k * i + c]);
        out[(size_t)k * r + c] ^= f.mul(t, out[(size_t)k * i + c]);
      }
    }
  }
  return true;
}

bool compose_decode_rows(const std::vector<uint8_t> &gen, int k, int m,
                         uint64_t present_mask,
                         std::vector<int> &survivors,
                         std::vector<int> &erased,
                         std::vector<uint8_t> &rows) {
  const GF8 &f = gf8();
  int n = k + m;
  survivors.clear();
  erased.clear();
  for (int i = 0; i < n; i++) {
    if (present_mask & (1ull << i)) {
      if ((int)survivors.size() < k) survivors.push_back(i);
    } else {
      erased.push_back(i);
    }
  }
  if ((int)survivors.size() < k) return false;
  if ((int)erased.size() > m) return false;

  std::vector<uint8_t> b((size_t)k * k), d((size_t)k * k);
  for (int i = 0; i < k; i++)
    std::memcpy(&b[(size_t)i * k], &gen[(size_t)survivors[i] * k], k);
  if (!gf_invert(b.data(), d.data(), k)) return false;

  rows.assign(erased.size() * (size_t)k, 0);
  for (size_t p = 0; p < erased.size(); p++) {
    int e = erased[p];
    if (e < k) {
      std::memcpy(&rows[p * k], &d[(size_t)e * k], k);
    } else {
      // lost parity: compose generator row with the inverse
      // (ErasureCodeIsa.cc:546-557)
      for (int i = 0; i < k; i++) {
        uint8_t s = 0;
        for (int j = 0; j < k; j++)
          s ^= f.mul(d[(size_t)j * k + i], gen[(size_t)e * k + j]);
        rows[p * k + i] = s;
      }
    }
  }
  return true;
}

// ---- GF(2^16) (w=16 jerasure RS-van) ----

GF16::GF16() : log(65536), exp(65536) {
  unsigned v = 1;
  for (int i = 0; i < 65535; i++) {
    exp[i] = (uint16_t)v;
    log[v] = (uint16_t)i;
    v <<= 1;
    if (v & 0x10000) v ^= 0x1100B;
  }
  exp[65535] = exp[0];
  log[0] = 0;
}

const GF16 &gf16() {
  static const GF16 f;
  return f;
}

bool gen_matrix_rs_van_jerasure_w16(std::vector<uint16_t> &a, int k, int m) {
  const GF16 &f = gf16();
  int rows = k + m, cols = k;
  if (k < 1 || m < 0 || rows > 65535) return false;
  a.assign((size_t)rows * cols, 0);
  a[0] = 1;
  if (rows > 1) a[(size_t)(rows - 1) * cols + (cols - 1)] = 1;
  for (int i = 1; i < rows - 1; i++) {
    uint16_t v = 1;
    for (int j = 0; j < cols; j++) {
      a[(size_t)i * cols + j] = v;
      v = f.mul(v, (uint16_t)i);
    }
  }
  for (int i = 1; i < cols; i++) {
    int j = i;
    while (j < rows && a[(size_t)j * cols + i] == 0) j++;
    if (j >= rows) return false;
    if (j != i)
      for (int c = 0; c < cols; c++)
        std::swap(a[(size_t)j * cols + c], a[(size_t)i * cols + c]);
    uint16_t piv = a[(size_t)i * cols + i];
    if (piv != 1) {
      uint16_t inv = f.div(1, piv);
      for (int r = 0; r < rows; r++)
        a[(size_t)r * cols + i] = f.mul(inv, a[(size_t)r * cols + i]);
    }
    for (int c = 0; c < cols; c++) {
      uint16_t t = a[(size_t)i * cols + c];
      if (c != i && t != 0)
        for (int r = 0; r < rows; r++)
          a[(size_t)r * cols + c] ^= f.mul(t, a[(size_t)r * cols + i]);
    }
  }
  for (int j = 0; j < cols; j++) {
    uint16_t t = a[(size_t)cols * cols + j];
    if (t != 0 && t != 1) {
      uint16_t inv = f.div(1, t);
      for (int r = 0; r < rows; r++)
        a[(size_t)r * cols + j] = f.mul(inv, a[(size_t)r * cols + j]);
      for (int c = 0; c < cols; c++)
        a[(size_t)j * cols + c] = f.mul(t, a[(size_t)j * cols + c]);
    }
  }
  return true;
}

bool gf16_invert(const uint16_t *in, uint16_t *out, int k) {
  const GF16 &f = gf16();
  std::vector<uint16_t> w(in, in + (size_t)k * k);
  std::memset(out, 0, (size_t)k * k * 2);
  for (int i = 0; i < k; i++) out[(size_t)k * i + i] = 1;
  for (int i = 0; i < k; i++) {
    if (!w[(size_t)k * i + i]) {
      int j = i + 1;
      while (j < k && !w[(size_t)k * j + i]) j++;
      if (j >= k) return false;
      for (int c = 0; c < k; c++) {
        std::swap(w[(size_t)k * i + c], w[(size_t)k * j + c]);
        std::swap(out[(size_t)k * i + c], out[(size_t)k * j + c]);
      }
    }
    uint16_t inv = f.inv(w[(size_t)k * i + i]);
    for (int c = 0; c < k; c++) {
      w[(size_t)k * i + c] = f.mul(inv, w[(size_t)k * i + c]);
      out[(size_t)k * i + c] = f.mul(inv, out[(size_t)k * i + c]);
    }
    for (int r = 0; r < k; r++) {
      if (r == i) continue;
      uint16_t t = w[(size_t)k * r + i];
      if (!t) continue;
      for (int c = 0; c < k; c++) {
        w[(size_t)k * r + c] ^= f.mul(t, w[(size_t)k * i + c]);
        out[(size_t)k * r + c] ^= f.mul(t, out[(size_t)k * i + c]);
      }
    }
  }
  return true;
}

bool compose_decode_rows16(const std::vector<uint16_t> &gen, int k, int m,
                           uint64_t present_mask,
                           std::vector<int> &survivors,
                           std::vector<int> &erased,
                           std::vector<uint16_t> &rows) {
  const GF16 &f = gf16();
  int n = k + m;
  survivors.clear();
  erased.clear();
  for (int i = 0; i < n; i++) {
    if (present_mask & (1ull << i)) {
      if ((int)survivors.size() < k) survivors.push_back(i);
    } else {
      erased.push_back(i);
    }
  }
  if ((int)survivors.size() < k || (int)erased.size() > m) return false;
  std::vector<uint16_t> b((size_t)k * k), d((size_t)k * k);
  for (int i = 0; i < k; i++)
    std::memcpy(&b[(size_t)i * k], &gen[(size_t)survivors[i] * k], k * 2);
  if (!gf16_invert(b.data(), d.data(), k)) return false;
  rows.assign(erased.size() * (size_t)k, 0);
  for (size_t p = 0; p < erased.size(); p++) {
    int e = erased[p];
    if (e < k) {
      std::memcpy(&rows[p * k], &d[(size_t)e * k], k * 2);
    } else {
      for (int i = 0; i < k; i++) {
        uint16_t s = 0;
        for (int j = 0; j < k; j++)
          s ^= f.mul(d[(size_t)j * k + i], gen[(size_t)e * k + j]);
        rows[p * k + i] = s;
      }
    }
  }
  return true;
}

// ---- SHEC (restated from the reference's in-tree implementation) ----

// shec_calc_recovery_efficiency1 (ErasureCodeShec.cc:660-697)
static double shec_r_e1(int k, int m1, int m2, int c1, int c2) {
  if (m1 < c1 || m2 < c2) return -1;
  if ((m1 == 0 && c1 != 0) || (m2 == 0 && c2 != 0)) return -1;
  std::vector<int> r_eff_k(k, 100000000);
  double r_e1 = 0;
  for (int rr = 0; rr < m1; rr++) {
    int start = ((rr * k) / m1) % k;
    int end = (((rr + c1) * k) / m1) % k;
    int first = 1;
    for (int cc = start; first || cc != end; cc = (cc + 1) % k) {
      first = 0;
      r_eff_k[cc] =
          std::min(r_eff_k[cc], ((rr + c1) * k) / m1 - (rr * k) / m1);
    }
    r_e1 += ((rr + c1) * k) / m1 - (rr * k) / m1;
  }
  for (int rr = 0; rr < m2; rr++) {
    int start = ((rr * k) / m2) % k;
    int end = (((rr + c2) * k) / m2) % k;
    int first = 1;
    for (int cc = start; first || cc != end; cc = (cc + 1) % k) {
      first = 0;
      r_eff_k[cc] =
          std::min(r_eff_k[cc], ((rr + c2) * k) / m2 - (rr * k) / m2);
    }
    r_e1 += ((rr + c2) * k) / m2 - (rr * k) / m2;
  }
  for (int i = 0; i < k; i++) r_e1 += r_eff_k[i];
  return r_e1 / (k + m1 + m2);
}

// shec_reedsolomon_coding_matrix (ErasureCodeShec.cc:700-768), w=8
bool shec_matrix(std::vector<uint8_t> &coding, int k, int m, int c,
                 bool single) {
  if (k < 1 || m < 1 || c < 1 || c > m) return false;
  int m1, m2, c1, c2;
  if (!single) {
    int c1_best = -1, m1_best = -1;
    double min_r_e1 = 100.0;
    for (int tc1 = 0; tc1 <= c / 2; tc1++) {
      for (int tm1 = 0; tm1 <= m; tm1++) {
        int tc2 = c - tc1, tm2 = m - tm1;
        if (tm1 < tc1 || tm2 < tc2) continue;
        if ((tm1 == 0 && tc1 != 0) || (tm2 == 0 && tc2 != 0)) continue;
        if ((tm1 != 0 && tc1 == 0) || (tm2 != 0 && tc2 == 0)) continue;
        double r = shec_r_e1(k, tm1, tm2, tc1, tc2);
        if (min_r_e1 - r > std::numeric_limits<double>::epsilon() &&
            r < min_r_e1) {
          min_r_e1 = r;
          c1_best = tc1;
          m1_best = tm1;
        }
      }
    }
    m1 = m1_best;
    c1 = c1_best;
    m2 = m - m1_best;
    c2 = c - c1_best;
    if (m1 < 0) return false;
  } else {
    m1 = 0; c1 = 0; m2 = m; c2 = c;
  }
  std::vector<uint8_t> full;
  if (!gen_matrix_rs_van_jerasure(full, k, m)) return false;
  coding.assign(full.begin() + (size_t)k * k, full.end());
  for (int rr = 0; rr < m1; rr++) {
    int end = ((rr * k) / m1) % k;
    int start = (((rr + c1) * k) / m1) % k;
    for (int cc = start; cc != end; cc = (cc + 1) % k)
      coding[(size_t)rr * k + cc] = 0;
  }
  for (int rr = 0; rr < m2; rr++) {
    int end = ((rr * k) / m2) % k;
    int start = (((rr + c2) * k) / m2) % k;
    for (int cc = start; cc != end; cc = (cc + 1) % k)
      coding[(size_t)(rr + m1) * k + cc] = 0;
  }
  return true;
}

// GF(2^8) determinant via elimination (char 2: no sign bookkeeping);
// semantics match shec determinant.c's zero/nonzero answer.
static uint8_t gf_det(std::vector<uint8_t> mat, int n) {
  const GF8 &f = gf8();
  uint8_t det = 1;
  for (int i = 0; i < n; i++) {
    if (!mat[(size_t)i * n + i]) {
      int j = i + 1;
      while (j < n && !mat[(size_t)j * n + i]) j++;
      if (j >= n) return 0;
      for (int cc = 0; cc < n; cc++)
        std::swap(mat[(size_t)i * n + cc], mat[(size_t)j * n + cc]);
    }
    uint8_t piv = mat[(size_t)i * n + i];
    det = f.mul(det, piv);
    uint8_t inv = f.inv(piv);
    for (int r = i + 1; r < n; r++) {
      uint8_t t = f.mul(inv, mat[(size_t)r * n + i]);
      if (!t) continue;
      for (int cc = i; cc < n; cc++)
        mat[(size_t)r * n + cc] ^= f.mul(t, mat[(size_t)i * n + cc]);
    }
  }
  return det;
}

// shec_make_decoding_matrix + shec_matrix_decode marshalling
// (ErasureCodeShec.cc:770-1050). Returns false when unrecoverable.
bool shec_decode_plan(const std::vector<uint8_t> &coding, int k, int m,
                      uint64_t want_mask, uint64_t avail_mask,
                      ShecPlan &plan) {
  std::vector<int> want(k + m, 0), avails(k + m, 0);
  for (int i = 0; i < k + m; i++) {
    want[i] = (want_mask >> i) & 1;
    avails[i] = (avail_mask >> i) & 1;
  }
  // wanting a lost parity pulls in the data chunks it covers (:782-790)
  for (int i = 0; i < m; i++)
    if (want[k + i] && !avails[k + i])
      for (int j = 0; j < k; j++)
        if (coding[(size_t)i * k + j] > 0) want[j] = 1;

  int mindup = k + 1, minp = k + 1;
  std::vector<int> best_row, best_col;
  for (unsigned long long pp = 0; pp < (1ull << m); ++pp) {
    int ek = 0;
    std::vector<int> p(m);
    for (int i = 0; i < m; i++)
      if (pp & (1ull << i)) p[ek++] = i;
    if (ek > minp) continue;
    bool ok = true;
    for (int i = 0; i < ek && ok; i++)
      if (!avails[k + p[i]]) ok = false;
    if (!ok) continue;

    std::vector<int> tmprow(k + m, 0), tmpcol(k, 0);
    for (int i = 0; i < k; i++)
      if (want[i] && !avails[i]) tmpcol[i] = 1;
    for (int i = 0; i < ek; i++) {
      tmprow[k + p[i]] = 1;
      for (int j = 0; j < k; j++) {
        uint8_t el = coding[(size_t)p[i] * k + j];
        if (el != 0) tmpcol[j] = 1;
        if (el != 0 && avails[j] == 1) tmprow[j] = 1;
      }
    }
    int dup_row = 0, dup_col = 0;
    for (int i = 0; i < k + m; i++) dup_row += tmprow[i];
    for (int i = 0; i < k; i++) dup_col += tmpcol[i];
    if (dup_row != dup_col) continue;
    int dup = dup_row;
    if (dup == 0) {
      mindup = 0;
      best_row.clear();
      best_col.clear();
      break;
    }
    if (dup < mindup) {
      std::vector<uint8_t> tmpmat((size_t)dup * dup);
      int row = 0;
      for (int i = 0; i < k + m; i++) {
        if (!tmprow[i]) continue;
        int col = 0;
        for (int j = 0; j < k; j++) {
          if (!tmpcol[j]) continue;
          tmpmat[(size_t)row * dup + col] =
              i < k ? (i == j ? 1 : 0) : coding[(size_t)(i - k) * k + j];
          col++;
        }
        row++;
      }
      if (gf_det(tmpmat, dup) != 0) {
        mindup = dup;
        minp = ek;
        best_row.clear();
        best_col.clear();
        for (int i = 0; i < k + m; i++)
          if (tmprow[i]) best_row.push_back(i);
        for (int i = 0; i < k; i++)
          if (tmpcol[i]) best_col.push_back(i);
      }
    }
  }
  if (mindup == k + 1) return false;

  // minimum set (:957-985)
  plan.minimum.clear();
  std::vector<int> minimum(k + m, 0);
  for (int id : best_row) minimum[id] = 1;
  for (int i = 0; i < k; i++)
    if (want[i] && avails[i]) minimum[i] = 1;
  for (int i = 0; i < m; i++) {
    if (want[k + i] && avails[k + i] && !minimum[k + i]) {
      for (int j = 0; j < k; j++) {
        if (coding[(size_t)i * k + j] > 0 && !want[j]) {
          minimum[k + i] = 1;
          break;
        }
      }
    }
  }
  for (int i = 0; i < k + m; i++)
    if (minimum[i]) plan.minimum.push_back(i);

  plan.src_ids.clear();
  plan.out_ids.clear();
  plan.rows.clear();
  if (mindup > 0) {
    std::vector<uint8_t> tmpmat((size_t)mindup * mindup),
        inv((size_t)mindup * mindup);
    for (int i = 0; i < mindup; i++)
      for (int j = 0; j < mindup; j++)
        tmpmat[(size_t)i * mindup + j] =
            best_row[i] < k
                ? (best_row[i] == best_col[j] ? 1 : 0)
                : coding[(size_t)(best_row[i] - k) * k + best_col[j]];
    if (!gf_invert(tmpmat.data(), inv.data(), mindup)) return false;
    plan.src_ids = best_row;  // original chunk ids (data or parity)
    for (int i = 0; i < mindup; i++) {
      if (avails[best_col[i]]) continue;  // only erased columns computed
      plan.out_ids.push_back(best_col[i]);
      plan.rows.insert(plan.rows.end(), inv.begin() + (size_t)i * mindup,
                       inv.begin() + (size_t)(i + 1) * mindup);
    }
  }
  // phase 2: re-encode wanted lost parity from (recovered) data (:1040-1046)
  plan.parity_out.clear();
  plan.parity_rows.clear();
  for (int i = 0; i < m; i++) {
    if (want[k + i] && !avails[k + i]) {
      plan.parity_out.push_back(k + i);
      plan.parity_rows.insert(plan.parity_rows.end(),
                              coding.begin() + (size_t)i * k,
                              coding.begin() + (size_t)(i + 1) * k);
    }
  }
  return true;
}

}  // namespace ecx
