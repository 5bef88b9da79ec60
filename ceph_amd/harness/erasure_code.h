// erasure_code.h — standalone-harness mirror of the reference's EC plugin
// contract: ceph::ErasureCodeInterface (src/erasure-code/
// ErasureCodeInterface.h:183-732, modern shard_id forms only — the
// [[deprecated]] std::set<int> forms and the CRUSH create_rule hook are
// intentionally omitted in the standalone harness; a real-Ceph build of the
// plugin shim implements them against Ceph's own headers, see
// INTEGRATION.md) and the ErasureCode base class defaults
// (src/erasure-code/ErasureCode.{h,cc}).
#pragma once

#include <iosfwd>
#include <memory>
#include <sstream>
#include <vector>

#include "ec_types.h"

namespace ecx {

class ErasureCodeInterface {
 public:
  virtual ~ErasureCodeInterface() = default;

  // ErasureCodeInterface.h:201 — 0 on success, -errno on error; profile is
  // echoed back (with filled defaults) through get_profile().
  virtual int init(ErasureCodeProfile &profile, std::ostream *ss) = 0;
  virtual const ErasureCodeProfile &get_profile() const = 0;

  virtual unsigned int get_chunk_count() const = 0;        // :240
  virtual unsigned int get_data_chunk_count() const = 0;   // :250
  virtual unsigned int get_coding_chunk_count() const = 0; // :262
  virtual int get_sub_chunk_count() = 0;                   // :272
  virtual unsigned int get_chunk_size(unsigned stripe_width) const = 0;  // :291
  virtual size_t get_minimum_granularity() = 0;            // :361

  // :310 — minimum chunk set (+ sub-chunk offsets for array codes)
  virtual int minimum_to_decode(
      const shard_id_set &want_to_read, const shard_id_set &available,
      shard_id_set &minimum_set,
      shard_id_map<std::vector<std::pair<int, int>>> *minimum_sub_chunks) = 0;
  // :345
  virtual int minimum_to_decode_with_cost(const shard_id_set &want_to_read,
                                          const shard_id_map<int> &available,
                                          shard_id_set *minimum) = 0;

  // :402 — split+pad in, emit all chunks (bufferlist simplified to buffer)
  virtual int encode(const shard_id_set &want_to_encode, const buffer &in,
                     shard_id_map<buffer> *encoded) = 0;
  // :448 — in immutable data chunks (absent => zeros), out caller-allocated
  virtual int encode_chunks(const shard_id_map<buffer> &in,
                            shard_id_map<buffer> &out) = 0;
  // :470
  virtual void encode_delta(const buffer &old_data, const buffer &new_data,
                            buffer *delta_maybe_in_place) = 0;
  // :498
  virtual void apply_delta(const shard_id_map<buffer> &in,
                           shard_id_map<buffer> &out) = 0;
  // :538
  virtual int decode(const shard_id_set &want_to_read,
                     const shard_id_map<buffer> &chunks,
                     shard_id_map<buffer> *decoded, int chunk_size) = 0;
  // :570
  virtual int decode_chunks(const shard_id_set &want_to_read,
                            shard_id_map<buffer> &in,
                            shard_id_map<buffer> &out) = 0;
  // :612
  virtual const std::vector<shard_id_t> &get_chunk_mapping() const = 0;

  // :636-693 optimization flags
  using plugin_flags = uint64_t;
  enum {
    FLAG_EC_PLUGIN_PARTIAL_READ_OPTIMIZATION = 1 << 0,
    FLAG_EC_PLUGIN_PARTIAL_WRITE_OPTIMIZATION = 1 << 1,
    FLAG_EC_PLUGIN_ZERO_INPUT_ZERO_OUTPUT_OPTIMIZATION = 1 << 2,
    FLAG_EC_PLUGIN_ZERO_PADDING_OPTIMIZATION = 1 << 3,
    FLAG_EC_PLUGIN_PARITY_DELTA_OPTIMIZATION = 1 << 4,
    FLAG_EC_PLUGIN_REQUIRE_SUB_CHUNKS = 1 << 5,
    FLAG_EC_PLUGIN_OPTIMIZED_SUPPORTED = 1 << 6,
    FLAG_EC_PLUGIN_CRC_ENCODE_DECODE_SUPPORT = 1 << 7,
    FLAG_EC_PLUGIN_DIRECT_READS = 1 << 8,
  };
  virtual plugin_flags get_supported_optimizations() const = 0;
};

using ErasureCodeInterfaceRef = std::shared_ptr<ErasureCodeInterface>;

// Base class with the shared defaults (ErasureCode.{h,cc}).
class ErasureCode : public ErasureCodeInterface {
 public:
  static const unsigned SIMD_ALIGN;  // = 64 (ErasureCode.cc:43)

  std::vector<shard_id_t> chunk_mapping;
  ErasureCodeProfile _profile;

  int init(ErasureCodeProfile &profile, std::ostream *ss) override;
  const ErasureCodeProfile &get_profile() const override { return _profile; }

  unsigned int get_coding_chunk_count() const override {
    return get_chunk_count() - get_data_chunk_count();
  }
  int get_sub_chunk_count() override { return 1; }
  size_t get_minimum_granularity() override { return 1; }

  int sanity_check_k_m(int k, int m, std::ostream *ss);

  virtual int _minimum_to_decode(const shard_id_set &want_to_read,
                                 const shard_id_set &available_chunks,
                                 shard_id_set *minimum);
  int minimum_to_decode(
      const shard_id_set &want_to_read, const shard_id_set &available,
      shard_id_set &minimum_set,
      shard_id_map<std::vector<std::pair<int, int>>> *minimum_sub_chunks)
      override;
  int minimum_to_decode_with_cost(const shard_id_set &want_to_read,
                                  const shard_id_map<int> &available,
                                  shard_id_set *minimum) override;

  // ErasureCode.cc:277-312: split input into k chunks, pad, alloc parity
  int encode_prepare(const buffer &raw, shard_id_map<buffer> &encoded) const;
  int encode(const shard_id_set &want_to_encode, const buffer &in,
             shard_id_map<buffer> *encoded) override;
  int decode(const shard_id_set &want_to_read,
             const shard_id_map<buffer> &chunks,
             shard_id_map<buffer> *decoded, int chunk_size) override;
  virtual int _decode(const shard_id_set &want_to_read,
                      const shard_id_map<buffer> &chunks,
                      shard_id_map<buffer> *decoded);

  const std::vector<shard_id_t> &get_chunk_mapping() const override {
    return chunk_mapping;
  }

  void encode_delta(const buffer &, const buffer &, buffer *) override;
  void apply_delta(const shard_id_map<buffer> &,
                   shard_id_map<buffer> &) override;

  // profile helpers (ErasureCode.cc:512-560): missing/empty keys get the
  // default WRITTEN BACK into the profile (factory() equality gate relies
  // on this, ErasureCodePlugin.cc:126-130)
  static int to_int(const std::string &name, ErasureCodeProfile &profile,
                    int *value, const std::string &default_value,
                    std::ostream *ss);
  static int to_bool(const std::string &name, ErasureCodeProfile &profile,
                     bool *value, const std::string &default_value,
                     std::ostream *ss);
  static int to_string(const std::string &name, ErasureCodeProfile &profile,
                       std::string *value, const std::string &default_value,
                       std::ostream *ss);

 protected:
  int parse(const ErasureCodeProfile &profile, std::ostream *ss);
  shard_id_t chunk_index(int raw_shard) const;
};

}  // namespace ecx
