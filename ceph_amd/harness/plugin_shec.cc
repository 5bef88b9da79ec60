// plugin_shec.cc — libec_shec.so: shingled erasure codes on the MI355X
// core. Unlike the GF libraries, SHEC's algorithms live in the reference
// tree itself (src/erasure-code/shec/ErasureCodeShec.cc) and are restated
// in ceph_amd/csrc/gf.cpp (shec_matrix / shec_decode_plan). The byte work
// is plain GF(2^8) matrix arithmetic, so encode reuses the standard
// mi355x matmul kernel with the shingled matrix (ecx_set_matrix) and
// decode drives the same kernel with the plan's composed rows
// (ecx_matmul_chunks_host == the jerasure_matrix_dotprod role,
// ErasureCodeShec.cc:1030-1046). No CPU fallback.
#include <cerrno>
#include <cstring>
#include <memory>
#include <ostream>
#include <vector>

#include "../../include/ec_mi355x.h"
#include "../csrc/gf.h"
#include "erasure_code_plugin.h"

using namespace ecx;

namespace {

// Shared by the plugin's _minimum_to_decode and the CPU test probe below:
// the reference's locality-aware minimum (ErasureCodeShec.cc:130-178 with
// the construction at :943-962). Takes the FULL want mask — the
// want&&avail terms of the construction add wanted available chunks.
int shec_minimum_mask(const std::vector<uint8_t> &coding, int k, int m,
                      uint64_t want, uint64_t avail, uint64_t *out_mask) {
  if ((want & avail) == want) {
    *out_mask = want;
    return 0;
  }
  ShecPlan plan;
  if (!ecx::shec_decode_plan(coding, k, m, want, avail, plan)) return -EIO;
  uint64_t mask = 0;
  for (int id : plan.minimum) mask |= 1ull << id;
  *out_mask = mask;
  return 0;
}

}  // namespace

// CPU-only probe for tests (no GPU context needed): builds the shingled
// matrix for (k,m,c) and runs the exact minimum computation the plugin's
// _minimum_to_decode uses. dlopen-able from pytest on a GPU-less box.
extern "C" int ecx_shec_minimum_probe(int k, int m, int c, int single,
                                      uint64_t want_mask, uint64_t avail_mask,
                                      uint64_t *minimum_mask) {
  std::vector<uint8_t> coding;
  if (!ecx::shec_matrix(coding, k, m, c, single != 0)) return -EINVAL;
  return shec_minimum_mask(coding, k, m, want_mask, avail_mask, minimum_mask);
}

namespace {

class ErasureCodeShec final : public ErasureCode {
  ecx_ctx *ctx_ = nullptr;
  int k_ = 0, m_ = 0, c_ = 0, w_ = 8, device_ = 0, streams_ = 2;
  bool single_;
  std::vector<uint8_t> coding_;  // m x k shingled matrix

 public:
  explicit ErasureCodeShec(bool single) : single_(single) {}
  ~ErasureCodeShec() override {
    if (ctx_) ecx_destroy(ctx_);
  }

  unsigned int get_chunk_count() const override { return k_ + m_; }
  unsigned int get_data_chunk_count() const override { return k_; }
  size_t get_minimum_granularity() override { return 16; }

  plugin_flags get_supported_optimizations() const override {
    // ErasureCodeShec.h:65-70 minus CRC (not implemented here) and minus
    // OPTIMIZED (the reference shec does not claim it either)
    return FLAG_EC_PLUGIN_PARTIAL_READ_OPTIMIZATION |
           FLAG_EC_PLUGIN_PARTIAL_WRITE_OPTIMIZATION |
           FLAG_EC_PLUGIN_ZERO_INPUT_ZERO_OUTPUT_OPTIMIZATION |
           FLAG_EC_PLUGIN_PARITY_DELTA_OPTIMIZATION;
  }

  int init(ErasureCodeProfile &profile, std::ostream *ss) override {
    int err = ErasureCode::parse(profile, ss);
    // parameter rules of ErasureCodeShec::parse (:519-577): defaults
    // (4,3,2); c<=m, m<=k, k<=12, k+m<=20
    err |= to_int("k", profile, &k_, "4", ss);
    err |= to_int("m", profile, &m_, "3", ss);
    err |= to_int("c", profile, &c_, "2", ss);
    err |= to_int("w", profile, &w_, "8", ss);
    err |= to_int("mi355x-device", profile, &device_, "0", ss);
    err |= to_int("mi355x-streams", profile, &streams_, "2", ss);
    if (k_ <= 0 || m_ <= 0 || c_ <= 0 || m_ < c_ || k_ > 12 ||
        k_ + m_ > 20 || k_ < m_) {
      if (ss)
        *ss << "shec: invalid (k,m,c)=(" << k_ << "," << m_ << "," << c_
            << ")\n";
      return -EINVAL;
    }
    if (w_ != 8) {
      if (ss) *ss << "shec: w=" << w_ << " must be 8 here\n";
      return -EINVAL;
    }
    if (err) return err;
    profile["technique"] = single_ ? "single" : "multiple";
    if (!ecx::shec_matrix(coding_, k_, m_, c_, single_)) return -EINVAL;
    int r = ecx_create2(k_, m_, ECX_T_RS_VAN_JERASURE, 8, 2048, device_,
                        streams_, &ctx_);
    if (r != ECX_OK) {
      if (ss)
        *ss << "shec: ecx_create failed (" << r
            << (r == ECX_ERR_NO_GPU ? ": no GPU — no CPU fallback" : "")
            << ")\n";
      return r;
    }
    r = ecx_set_matrix(ctx_, coding_.data());
    if (r != ECX_OK) return r;
    return ErasureCode::init(profile, ss);
  }

  unsigned int get_chunk_size(unsigned int stripe_width) const override {
    // ErasureCodeShec.cc:63-72 with get_alignment() = k*w*sizeof(int)
    unsigned align = (unsigned)k_ * w_ * 4u;
    unsigned tail = stripe_width % align;
    unsigned padded = stripe_width + (tail ? align - tail : 0);
    return padded / k_;
  }

  int encode_chunks(const shard_id_map<buffer> &in,
                    shard_id_map<buffer> &out) override {
    size_t size = 0;
    const uint8_t *data[64] = {};
    uint8_t *parity[64] = {};
    for (auto &&[shard, b] : in) {
      if (!size) size = b.length();
      else if (size != b.length()) return -EINVAL;
      if ((int)shard < k_) data[(int)shard] = b.c_str();
    }
    for (auto &&[shard, b] : out) {
      if (!size) size = b.length();
      else if (size != b.length()) return -EINVAL;
      if ((int)shard < k_)
        data[(int)shard] = b.c_str();
      else
        parity[(int)shard - k_] = b.c_str();
    }
    if (!size) return 0;
    return ecx_encode_chunks_host(ctx_, data, parity, size);
  }

  int decode_chunks(const shard_id_set &want_to_read,
                    shard_id_map<buffer> &in,
                    shard_id_map<buffer> &out) override {
    size_t size = 0;
    uint8_t *chunks[64] = {};
    uint64_t avail = 0, want = 0;
    for (auto &&[shard, b] : in) {
      size = b.length();
      chunks[(int)shard] = b.c_str();
      avail |= 1ull << (int)shard;
    }
    for (auto &&[shard, b] : out) {
      size = b.length();
      chunks[(int)shard] = b.c_str();
      want |= 1ull << (int)shard;
    }
    for (auto &&s : want_to_read) want |= 1ull << (int)s;
    want &= ~avail;
    if (!size || !want) return 0;

    ShecPlan plan;
    if (!ecx::shec_decode_plan(coding_, k_, m_, want, avail, plan))
      return -EIO;
    // phase 1: recover erased data (ErasureCodeShec.cc:1030-1038)
    if (!plan.out_ids.empty()) {
      const uint8_t *srcs[64];
      uint8_t *outs[64];
      for (size_t i = 0; i < plan.src_ids.size(); i++)
        srcs[i] = chunks[plan.src_ids[i]];
      for (size_t i = 0; i < plan.out_ids.size(); i++) {
        outs[i] = chunks[plan.out_ids[i]];
        if (!outs[i]) return -EINVAL;
      }
      int r = ecx_matmul_chunks_host(ctx_, srcs, (int)plan.src_ids.size(),
                                     outs, (int)plan.out_ids.size(),
                                     plan.rows.data(), size);
      if (r != ECX_OK) return r;
    }
    // phase 2: re-encode wanted lost parity from (recovered) data (:1040-46)
    if (!plan.parity_out.empty()) {
      const uint8_t *srcs[64];
      uint8_t *outs[64];
      for (int i = 0; i < k_; i++) srcs[i] = chunks[i];  // NULL => zeros
      for (size_t i = 0; i < plan.parity_out.size(); i++) {
        outs[i] = chunks[plan.parity_out[i]];
        if (!outs[i]) return -EINVAL;
      }
      int r = ecx_matmul_chunks_host(ctx_, srcs, k_, outs,
                                     (int)plan.parity_out.size(),
                                     plan.parity_rows.data(), size);
      if (r != ECX_OK) return r;
    }
    return 0;
  }

  // SHEC's locality-aware minimum (ErasureCodeShec::_minimum_to_decode
  // delegates to the decoding-matrix search)
  int _minimum_to_decode(const shard_id_set &want_to_read,
                         const shard_id_set &available,
                         shard_id_set *minimum) override {
    uint64_t want = 0, avail = available.low_mask();
    for (auto &&s : want_to_read) want |= 1ull << (int)s;
    // FULL want mask (not want & ~avail): the reference's minimum
    // construction (ErasureCodeShec.cc:943-962, mirrored in gf.cpp) adds
    // wanted-but-available chunks via its want[i]&&avails[i] terms; the
    // plan search itself only uses want&&!avail terms, so this is safe.
    uint64_t min_mask = 0;
    int r = shec_minimum_mask(coding_, k_, m_, want, avail, &min_mask);
    if (r) return r;
    for (int i = 0; i < k_ + m_; i++)
      if (min_mask & (1ull << i)) minimum->insert(i);
    return 0;
  }

  void encode_delta(const buffer &old_data, const buffer &new_data,
                    buffer *delta) override {
    ecx_encode_delta_host(ctx_, old_data.c_str(), new_data.c_str(),
                          delta->c_str(), delta->length());
  }

  void apply_delta(const shard_id_map<buffer> &in,
                   shard_id_map<buffer> &out) override {
    // ErasureCodeShec apply_delta (:470-505): per (data, coding) pair,
    // coefficient from the shingled matrix (zero => no-op)
    for (auto &&[datashard, databuf] : in) {
      if ((int)datashard >= k_) continue;
      for (auto &&[codingshard, codingbuf] : out) {
        if ((int)codingshard < k_) continue;
        ecx_apply_delta_host(ctx_, databuf.c_str(), (int)datashard,
                             (int)codingshard,
                             const_cast<uint8_t *>(codingbuf.c_str()),
                             codingbuf.length());
      }
    }
  }
};

class ErasureCodePluginShec final : public ErasureCodePlugin {
 public:
  int factory(const std::string &, ErasureCodeProfile &profile,
              ErasureCodeInterfaceRef *erasure_code,
              std::ostream *ss) override {
    // technique dispatch mirrors ErasureCodePluginShec.cc:40-57
    if (profile.find("technique") == profile.end())
      profile["technique"] = "multiple";
    std::string t = profile["technique"];
    if (t != "single" && t != "multiple") {
      if (ss)
        *ss << "technique=" << t
            << " is not a valid coding technique. Choose one of: single, "
               "multiple\n";
      return -ENOENT;
    }
    auto interface = std::make_shared<ErasureCodeShec>(t == "single");
    int r = interface->init(profile, ss);
    if (r) return r;
    *erasure_code = interface;
    return 0;
  }
};

}  // namespace

extern "C" {
const char *__erasure_code_version() { return ECX_HARNESS_VERSION; }

int __erasure_code_init(const char *plugin_name, const char *) {
  auto &instance = ErasureCodePluginRegistry::instance();
  auto plugin = std::make_unique<ErasureCodePluginShec>();
  int r = instance.add(plugin_name, plugin.get());
  if (r == 0) plugin.release();
  return r;
}
}
