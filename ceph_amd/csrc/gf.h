// gf.h — product-side GF(2^8) host arithmetic for libec_mi355x_core.
//
// Host work on the EC hot path is integers only: generator-matrix derivation
// once per (k,m,technique) (mirrors prepare(), src/erasure-code/isa/
// ErasureCodeIsa.cc:637-697 and src/erasure-code/jerasure/
// ErasureCodeJerasure.cc:431-435) and per-erasure-signature decode-row
// composition (ErasureCodeIsa.cc:510-567). The byte/stripe work itself is on
// the GPU (ec_core.hip). Independent implementation from oracle/ec_ref.c —
// the oracle is the checker, this is the product.
#pragma once

#include <cstdint>
#include <vector>

namespace ecx {

// GF(2^8) over the primitive polynomial 0x11d (the field of gf-complete w=8
// and isa-l).
struct GF8 {
  uint8_t log[256];
  uint8_t exp[256];
  GF8();
  uint8_t mul(uint8_t a, uint8_t b) const {
    if (!a || !b) return 0;
    int s = log[a] + log[b];
    if (s >= 255) s -= 255;
    return exp[s];
  }
  uint8_t inv(uint8_t a) const { return a ? exp[255 - log[a]] : 0; }
  uint8_t div(uint8_t a, uint8_t b) const {
    if (!a || !b) return 0;
    int s = log[a] - log[b];
    if (s < 0) s += 255;
    return exp[s];
  }
};
const GF8 &gf8();

// Full (k+m) x k generator, identity on top, row-major (isa-l layout).
// Returns false on invalid parameters.
bool gen_matrix_rs_van_isa(std::vector<uint8_t> &a, int k, int m);
bool gen_matrix_cauchy_isa(std::vector<uint8_t> &a, int k, int m);
bool gen_matrix_rs_van_jerasure(std::vector<uint8_t> &a, int k, int m);
bool gen_matrix(int technique, std::vector<uint8_t> &a, int k, int m);

// Gauss-Jordan inverse of a k x k matrix; false if singular.
bool gf_invert(const uint8_t *in, uint8_t *out, int k);

// jerasure Cauchy-original family (w=8): full generator (identity top,
// coding row i = inv(i XOR (m+j))), companion-basis bitmatrix, GF(2)
// survivor inversion and decode-row composition over bit rows. Mirrors
// jerasure cauchy.c / jerasure.c semantics as called from
// ErasureCodeJerasure.cc:499-514,568-574.
bool gen_matrix_cauchy_orig(std::vector<uint8_t> &a, int k, int m);
// jerasure cauchy.c cauchy_n_ones (w=8): ones in the companion bitmatrix
// of e = sum_c popcount(e * 2^c).
int cauchy_n_ones(uint8_t e);
// jerasure cauchy.c cauchy_improve_coding_matrix over m x k coding rows.
void improve_cauchy_matrix(uint8_t *coding, int k, int m);
// cauchy_good general branch: cauchy_original + improve (m == 2 would use
// jerasure's unsourceable cbest tables — callers gate on that).
bool gen_matrix_cauchy_good(std::vector<uint8_t> &a, int k, int m);
// bitmat: (m*w) x (k*w) bits, one byte per bit, row-major
void matrix_to_bitmatrix(const uint8_t *coding_rows, int k, int m, int w,
                         std::vector<uint8_t> &bitmat);
// rows out: (n_erased*w) x (k*w) bit rows over the k survivors
bool compose_bit_decode_rows(const std::vector<uint8_t> &bitmat, int k,
                             int m, int w, uint64_t present_mask,
                             std::vector<int> &survivors,
                             std::vector<int> &erased,
                             std::vector<uint8_t> &rows);

// Decode-row composition for a given erasure pattern (mirrors
// ErasureCodeIsa.cc:510-567): picks the first k present chunks in id order
// as survivors (ErasureCode.cc:154-170), inverts the survivor submatrix,
// and emits one coefficient row per erased chunk over those survivors.
// Outputs: survivors[k] (chunk ids), erased[] (chunk ids, ascending),
// rows (n_erased x k coefficients). Returns false if undecodable.
bool compose_decode_rows(const std::vector<uint8_t> &gen, int k, int m,
                         uint64_t present_mask,
                         std::vector<int> &survivors,
                         std::vector<int> &erased,
                         std::vector<uint8_t> &rows);

// GF(2^16) (w=16 jerasure reed_sol_van): gf-complete default w=16 field
// (poly 0x1100B) + the same big-Vandermonde construction with u16
// coefficients/symbols (galois_w16_region_multiply semantics,
// ErasureCodeJerasure.cc:316-319).
struct GF16 {
  std::vector<uint16_t> log, exp;
  GF16();
  uint16_t mul(uint16_t a, uint16_t b) const {
    if (!a || !b) return 0;
    unsigned s = log[a] + log[b];
    if (s >= 65535) s -= 65535;
    return exp[s];
  }
  uint16_t inv(uint16_t a) const { return a ? exp[65535 - log[a]] : 0; }
  uint16_t div(uint16_t a, uint16_t b) const {
    if (!a || !b) return 0;
    int s = (int)log[a] - (int)log[b];
    if (s < 0) s += 65535;
    return exp[s];
  }
};
const GF16 &gf16();
// full (k+m) x k generator, identity top
bool gen_matrix_rs_van_jerasure_w16(std::vector<uint16_t> &a, int k, int m);
bool gf16_invert(const uint16_t *in, uint16_t *out, int k);
bool compose_decode_rows16(const std::vector<uint16_t> &gen, int k, int m,
                           uint64_t present_mask,
                           std::vector<int> &survivors,
                           std::vector<int> &erased,
                           std::vector<uint16_t> &rows);

// SHEC (shingled EC) matrix + decode search, restated from the reference's
// OWN in-tree implementation (src/erasure-code/shec/ErasureCodeShec.cc —
// unlike the GF submodules this algorithm is fully present):
//  - shec_matrix: shec_reedsolomon_coding_matrix (:700-768): RS-van
//    jerasure matrix with shingle ranges zeroed; `single` selects the
//    SINGLE vs MULTIPLE technique split (m1/c1 search via
//    shec_calc_recovery_efficiency1, :660-697).
//  - shec_decode_plan: shec_make_decoding_matrix (:770-...): minimal
//    parity-subset search with determinant test, inversion, and the
//    minimum set. Returns false when the pattern is unrecoverable.
struct ShecPlan {
  // phase 1: recover erased data chunks
  std::vector<int> src_ids;   // chunk ids (data survivors + chosen parity)
  std::vector<int> out_ids;   // erased data chunk ids to recover
  std::vector<uint8_t> rows;  // out x src coefficients
  // phase 2: re-encode erased wanted parity from (recovered) data
  std::vector<int> parity_out;          // chunk ids >= k
  std::vector<uint8_t> parity_rows;     // parity_out x k coefficients
  std::vector<int> minimum;             // chunk ids to fetch
};
bool shec_matrix(std::vector<uint8_t> &coding /* m x k */, int k, int m,
                 int c, bool single);
bool shec_decode_plan(const std::vector<uint8_t> &coding, int k, int m,
                      uint64_t want_mask, uint64_t avail_mask,
                      ShecPlan &plan);

}  // namespace ecx
