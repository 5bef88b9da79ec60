// plugin_oracle.cc — libec_oracle.so: TEST FIXTURE ONLY.
//
// A CPU plugin over the oracle restatement (oracle/ec_ref.c), in the same
// role as the reference's fixture plugins (src/test/erasure-code/
// ErasureCodePluginExample.cc): it exercises the registry/interface on a
// GPU-less box and serves as the CPU leg of BASELINE config 1 (the
// plumbing check) in the ec_benchmark CLI. It is NOT the product path —
// the product plugin is libec_mi355x.so and refuses to run without a GPU.
#include <cerrno>
#include <cstring>
#include <memory>
#include <ostream>
#include <vector>

#include "../../oracle/ec_ref.h"
#include "erasure_code_plugin.h"

using namespace ecx;

namespace {

int technique_id(const std::string &t) {
  if (t == "reed_sol_van") return ECREF_T_RS_VAN_ISA;
  if (t == "cauchy") return ECREF_T_CAUCHY_ISA;
  if (t == "jerasure_reed_sol_van") return ECREF_T_RS_VAN_JERASURE;
  if (t == "cauchy_orig") return 3;  // bitmatrix/packet layout
  if (t == "cauchy_good") return 5;  // bitmatrix (orig + improve pass)
  if (t == "reed_sol_r6_op") return ECREF_T_RS_VAN_ISA;  // RAID6 == isa m=2
  return -1;
}

class ErasureCodeOracle final : public ErasureCode {
  int k_ = 0, m_ = 0, packetsize_ = 2048, w_ = 8;
  std::vector<uint16_t> gen16_;
  std::string technique_;
  std::vector<uint8_t> gen_;
  std::vector<uint8_t> bitmat_;
  bool is_bitmatrix() const {
    return technique_ == "cauchy_orig" || technique_ == "cauchy_good";
  }

 public:
  explicit ErasureCodeOracle(std::string t) : technique_(std::move(t)) {}

  unsigned int get_chunk_count() const override { return k_ + m_; }
  unsigned int get_data_chunk_count() const override { return k_; }

  plugin_flags get_supported_optimizations() const override {
    return FLAG_EC_PLUGIN_PARTIAL_READ_OPTIMIZATION |
           FLAG_EC_PLUGIN_PARTIAL_WRITE_OPTIMIZATION |
           FLAG_EC_PLUGIN_ZERO_INPUT_ZERO_OUTPUT_OPTIMIZATION |
           FLAG_EC_PLUGIN_PARITY_DELTA_OPTIMIZATION |
           FLAG_EC_PLUGIN_OPTIMIZED_SUPPORTED;
  }

  int init(ErasureCodeProfile &profile, std::ostream *ss) override {
    int err = ErasureCode::parse(profile, ss);
    err |= to_int("k", profile, &k_, "8", ss);
    err |= to_int("m", profile, &m_, "3", ss);
    err |= to_int("w", profile, &w_, "8", ss);
    err |= sanity_check_k_m(k_, m_, ss);
    if (w_ == 16 && technique_ == "jerasure_reed_sol_van") {
      if (err) return err;
      profile["technique"] = technique_;
      ecref_gf16_init();
      gen16_.resize((size_t)(k_ + m_) * k_);
      if (ecref_matrix_rs_vandermonde_jerasure_w16(gen16_.data(), k_, m_))
        return -EINVAL;
      return ErasureCode::init(profile, ss);
    }
    if (w_ != 8) {
      if (ss) *ss << "oracle: w=" << w_ << " unsupported here\n";
      return -EINVAL;
    }
    int t = technique_id(technique_);
    if (t < 0) {
      if (ss) *ss << "oracle: unknown technique " << technique_ << "\n";
      err = -EINVAL;
    }
    if (technique_ == "cauchy_good" && m_ == 2) {
      if (ss) *ss << "oracle: cauchy_good m=2 needs jerasure's cbest "
                     "tables (unsourceable) — refused\n";
      return -EINVAL;
    }
    if (technique_ == "reed_sol_r6_op" && m_ != 2) {
      if (ss) *ss << "reed_sol_r6_op: m must be 2\n";
      err = -EINVAL;
    }
    if (is_bitmatrix())
      err |= to_int("packetsize", profile, &packetsize_, "2048", ss);
    if (err) return err;
    profile["technique"] = technique_;
    gen_.resize((size_t)(k_ + m_) * k_);
    if (is_bitmatrix()) {
      std::vector<uint8_t> coding((size_t)m_ * k_);
      if ((technique_ == "cauchy_good"
               ? ecref_matrix_cauchy_good_jerasure(coding.data(), k_, m_)
               : ecref_matrix_cauchy_orig_jerasure(coding.data(), k_, m_))
          != 0)
        return -EINVAL;
      for (int i = 0; i < k_; i++) {
        std::memset(&gen_[(size_t)i * k_], 0, k_);
        gen_[(size_t)i * k_ + i] = 1;
      }
      std::memcpy(&gen_[(size_t)k_ * k_], coding.data(), (size_t)m_ * k_);
      bitmat_.resize((size_t)m_ * w_ * k_ * w_);
      ecref_matrix_to_bitmatrix(coding.data(), k_, m_, w_, bitmat_.data());
    } else if (ecref_matrix(t, gen_.data(), k_, m_) != 0) {
      return -EINVAL;
    }
    return ErasureCode::init(profile, ss);
  }

  bool is_w16() const { return !gen16_.empty(); }

  unsigned int get_chunk_size(unsigned int stripe_width) const override {
    if (is_w16()) return ecref_chunk_size_jerasure(k_, 16, stripe_width);
    if (is_bitmatrix()) {
      // ErasureCodeJerasureCauchy::get_alignment rule
      unsigned align = (unsigned)k_ * w_ * packetsize_ * 4u;
      unsigned tail = stripe_width % align;
      unsigned padded = stripe_width + (tail ? align - tail : 0);
      return padded / k_;
    }
    if (technique_ == "jerasure_reed_sol_van")
      return ecref_chunk_size_jerasure(k_, 8, stripe_width);
    return ecref_chunk_size_isa(k_, stripe_width);
  }

  int encode_chunks(const shard_id_map<buffer> &in,
                    shard_id_map<buffer> &out) override {
    size_t size = 0;
    const uint8_t *data[128] = {};
    uint8_t *parity[128] = {};
    for (auto &&[shard, b] : in) {
      if (!size) size = b.length();
      data[(int)shard] = b.c_str();
    }
    std::vector<buffer> scratch;
    for (int j = 0; j < m_; j++) parity[j] = nullptr;
    for (auto &&[shard, b] : out) {
      if (!size) size = b.length();
      if ((int)shard < k_) {
        // data position via the out map == source (LRC local-layer case)
        data[(int)shard] = b.c_str();
        continue;
      }
      parity[(int)shard - k_] = b.c_str();
    }
    // parity chunks not requested still need computing space
    for (int j = 0; j < m_; j++) {
      if (!parity[j]) {
        scratch.push_back(buffer::create_aligned(size));
        parity[j] = scratch.back().c_str();
      }
    }
    if (!size) return 0;
    if (is_w16()) {
      ecref_encode16(k_, m_, gen16_.data() + (size_t)k_ * k_, data, parity,
                     size);
      return 0;
    }
    if (is_bitmatrix())
      return ecref_bitmatrix_encode(k_, m_, w_, bitmat_.data(), data, parity,
                                    size, packetsize_) == 0 ? 0 : -EINVAL;
    ecref_encode(k_, m_, gen_.data() + (size_t)k_ * k_, data, parity, size);
    return 0;
  }

  int decode_chunks(const shard_id_set &, shard_id_map<buffer> &in,
                    shard_id_map<buffer> &out) override {
    size_t size = 0;
    uint8_t *chunks[128] = {};
    uint8_t present[128];
    std::memset(present, 0, sizeof(present));
    std::vector<buffer> scratch;
    for (auto &&[shard, b] : in) {
      size = b.length();
      chunks[(int)shard] = b.c_str();
      present[(int)shard] = 1;
    }
    for (auto &&[shard, b] : out) {
      size = b.length();
      chunks[(int)shard] = b.c_str();
      present[(int)shard] = 0;
    }
    if (!size) return 0;
    for (int i = 0; i < k_ + m_; i++) {
      if (present[i] && !chunks[i]) chunks[i] = nullptr;  // zeros
      if (!present[i] && !chunks[i]) {
        scratch.push_back(buffer::create_aligned(size));
        chunks[i] = scratch.back().c_str();
      }
    }
    if (is_w16())
      return ecref_decode16(chunks, present, k_, m_, size) == 0 ? 0 : -EIO;
    if (is_bitmatrix())
      return ecref_bitmatrix_decode(k_, m_, w_, bitmat_.data(), chunks,
                                    present, size, packetsize_) == 0
                 ? 0
                 : -EIO;
    return ecref_decode(technique_id(technique_), k_, m_, chunks, present,
                        size) == 0
               ? 0
               : -EIO;
  }

  void encode_delta(const buffer &old_data, const buffer &new_data,
                    buffer *delta) override {
    ecref_xor_region(old_data.c_str(), new_data.c_str(), delta->c_str(),
                     delta->length());
  }

  void apply_delta(const shard_id_map<buffer> &in,
                   shard_id_map<buffer> &out) override {
    for (auto &&[datashard, databuf] : in) {
      if ((int)datashard >= k_) continue;
      for (auto &&[codingshard, codingbuf] : out) {
        if ((int)codingshard < k_) continue;
        if (is_bitmatrix()) {
          // schedule-delta semantics (schedule_apply_delta filtered to
          // one (s, d) pair, ErasureCodeJerasure.cc:348-377): the pair's
          // w x w bitmatrix block applied per superword, XOR into parity
          const int i = (int)codingshard - k_, j = (int)datashard;
          const int W = k_ * w_;
          const size_t sw = (size_t)w_ * packetsize_;
          uint8_t *cp = const_cast<uint8_t *>(codingbuf.c_str());
          const uint8_t *dp = databuf.c_str();
          for (size_t off = 0; off + sw <= codingbuf.length(); off += sw)
            for (int r = 0; r < w_; r++)
              for (int c = 0; c < w_; c++)
                if (bitmat_[(size_t)(i * w_ + r) * W + j * w_ + c])
                  ecref_xor_region(cp + off + (size_t)r * packetsize_,
                                   dp + off + (size_t)c * packetsize_,
                                   cp + off + (size_t)r * packetsize_,
                                   packetsize_);
          continue;
        }
        uint8_t c = gen_[(size_t)(int)codingshard * k_ + (int)datashard];
        ecref_region_mul_xor(c, databuf.c_str(),
                             const_cast<uint8_t *>(codingbuf.c_str()),
                             codingbuf.length());
      }
    }
  }
};

class ErasureCodePluginOracle final : public ErasureCodePlugin {
 public:
  int factory(const std::string &, ErasureCodeProfile &profile,
              ErasureCodeInterfaceRef *erasure_code,
              std::ostream *ss) override {
    std::string technique = "reed_sol_van";
    if (auto it = profile.find("technique"); it != profile.end())
      technique = it->second;
    auto interface = std::make_shared<ErasureCodeOracle>(technique);
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
  ecref_gf_init();
  auto &instance = ErasureCodePluginRegistry::instance();
  auto plugin = std::make_unique<ErasureCodePluginOracle>();
  int r = instance.add(plugin_name, plugin.get());
  if (r == 0) plugin.release();
  return r;
}
}
