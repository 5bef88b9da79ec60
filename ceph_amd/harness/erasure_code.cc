// erasure_code.cc — base-class defaults mirroring src/erasure-code/
// ErasureCode.cc (the reference's shared plugin scaffolding).
#include "erasure_code.h"

#include <cerrno>
#include <cstring>
#include <ostream>

namespace ecx {

const unsigned ErasureCode::SIMD_ALIGN = 64;  // ErasureCode.cc:43

int ErasureCode::init(ErasureCodeProfile &profile, std::ostream *ss) {
  // the CRUSH rule keys are accepted and echoed (cluster mgmt itself is out
  // of scope for the standalone harness — SURVEY §2b)
  std::string s;
  to_string("crush-root", profile, &s, "default", ss);
  to_string("crush-failure-domain", profile, &s, "host", ss);
  _profile = profile;
  return 0;
}

int ErasureCode::sanity_check_k_m(int k, int m, std::ostream *ss) {
  // ErasureCode.cc:105-121
  if (k < 2) {
    if (ss) *ss << "k=" << k << " must be >= 2\n";
    return -EINVAL;
  }
  if (m < 1) {
    if (ss) *ss << "m=" << m << " must be >= 1\n";
    return -EINVAL;
  }
  if (k + m > 127) {
    if (ss) *ss << "(k+m)=" << (k + m) << " must be <= 127\n";
    return -EINVAL;
  }
  return 0;
}

shard_id_t ErasureCode::chunk_index(int raw_shard) const {
  // ErasureCode.cc:123-126
  return chunk_mapping.size() > (size_t)raw_shard
             ? chunk_mapping[raw_shard]
             : shard_id_t((int8_t)raw_shard);
}

int ErasureCode::_minimum_to_decode(const shard_id_set &want_to_read,
                                    const shard_id_set &available_chunks,
                                    shard_id_set *minimum) {
  // ErasureCode.cc:154-170: want if all available, else first k available
  if (available_chunks.includes(want_to_read)) {
    *minimum = want_to_read;
  } else {
    unsigned k = get_data_chunk_count();
    if (available_chunks.size() < k) return -EIO;
    unsigned j = 0;
    for (auto i = available_chunks.begin();
         j < k && i != available_chunks.end(); ++i, ++j)
      minimum->insert(*i);
  }
  return 0;
}

int ErasureCode::minimum_to_decode(
    const shard_id_set &want_to_read, const shard_id_set &available,
    shard_id_set &minimum_set,
    shard_id_map<std::vector<std::pair<int, int>>> *minimum_sub_chunks) {
  int r = _minimum_to_decode(want_to_read, available, &minimum_set);
  if (minimum_sub_chunks == nullptr || r != 0) return r;
  std::vector<std::pair<int, int>> defaults;
  defaults.emplace_back(0, get_sub_chunk_count());
  for (auto &&id : minimum_set) (*minimum_sub_chunks)[id] = defaults;
  return 0;
}

int ErasureCode::minimum_to_decode_with_cost(
    const shard_id_set &want_to_read, const shard_id_map<int> &available,
    shard_id_set *minimum) {
  // ErasureCode.cc:225-235 (cost ignored by the default implementation)
  shard_id_set available_chunks;
  for (auto &&[shard, cost] : available) {
    (void)cost;
    available_chunks.insert(shard);
  }
  return _minimum_to_decode(want_to_read, available_chunks, minimum);
}

int ErasureCode::encode_prepare(const buffer &raw,
                                shard_id_map<buffer> &encoded) const {
  // ErasureCode.cc:277-312
  unsigned k = get_data_chunk_count();
  unsigned m = get_chunk_count() - k;
  unsigned blocksize = get_chunk_size(raw.length());
  unsigned padded_chunks = k - raw.length() / blocksize;

  for (unsigned i = 0; i < k - padded_chunks; i++)
    encoded[chunk_index(i)] =
        buffer::copy(raw.c_str() + (size_t)i * blocksize, blocksize,
                     SIMD_ALIGN);
  if (padded_chunks) {
    unsigned remainder = raw.length() - (k - padded_chunks) * blocksize;
    buffer buf = buffer::create_aligned(blocksize, SIMD_ALIGN);
    std::memcpy(buf.c_str(),
                raw.c_str() + (size_t)(k - padded_chunks) * blocksize,
                remainder);
    std::memset(buf.c_str() + remainder, 0, blocksize - remainder);
    encoded[chunk_index(k - padded_chunks)] = buf;
    for (unsigned i = k - padded_chunks + 1; i < k; i++) {
      buffer z = buffer::create_aligned(blocksize, SIMD_ALIGN);
      z.zero();
      encoded[chunk_index(i)] = z;
    }
  }
  for (unsigned i = k; i < k + m; i++)
    encoded[chunk_index(i)] = buffer::create_aligned(blocksize, SIMD_ALIGN);
  return 0;
}

int ErasureCode::encode(const shard_id_set &want_to_encode, const buffer &in,
                        shard_id_map<buffer> *encoded) {
  // ErasureCode.cc:335-369
  unsigned k = get_data_chunk_count();
  unsigned m = get_chunk_count() - k;
  if (!encoded || !encoded->empty()) return -EINVAL;
  int err = encode_prepare(in, *encoded);
  if (err) return err;

  shard_id_map<buffer> in_shards(get_chunk_count());
  shard_id_map<buffer> out_shards(get_chunk_count());
  for (unsigned raw = 0; raw < k + m; raw++) {
    shard_id_t shard = chunk_index(raw);
    if (!encoded->contains(shard)) continue;
    if (raw < k)
      in_shards[shard] = encoded->at(shard);
    else
      out_shards[shard] = encoded->at(shard);
  }
  err = encode_chunks(in_shards, out_shards);
  if (err) return err;
  for (int i = 0; i < (int)(k + m); i++)
    if (!want_to_encode.contains(i)) encoded->erase(i);
  return 0;
}

int ErasureCode::_decode(const shard_id_set &want_to_read,
                         const shard_id_map<buffer> &chunks,
                         shard_id_map<buffer> *decoded) {
  // ErasureCode.cc:412-464
  if (!decoded || !decoded->empty()) return -EINVAL;
  if (!want_to_read.empty() && chunks.empty()) return -1;

  shard_id_set have;
  for (auto &&[shard, b] : chunks) {
    (void)b;
    have.insert(shard);
  }
  if (have.includes(want_to_read)) {
    for (auto &&shard : want_to_read) (*decoded)[shard] = chunks.at(shard);
    return 0;
  }
  unsigned k = get_data_chunk_count();
  unsigned m = get_chunk_count() - k;
  unsigned blocksize = (*chunks.begin()).second.length();
  shard_id_set erasures;
  for (int i = 0; i < (int)(k + m); i++) {
    shard_id_t s((int8_t)i);
    if (!chunks.contains(s)) {
      buffer b = buffer::create_aligned(blocksize, SIMD_ALIGN);
      (*decoded)[s] = b;
      erasures.insert(s);
    } else {
      (*decoded)[s] = chunks.at(s);
    }
  }
  shard_id_map<buffer> in(get_chunk_count());
  shard_id_map<buffer> out(get_chunk_count());
  for (auto &&[shard, b] : *decoded) {
    if (erasures.contains(shard))
      out[shard] = b;
    else
      in[shard] = b;
  }
  return decode_chunks(want_to_read, in, out);
}

int ErasureCode::decode(const shard_id_set &want_to_read,
                        const shard_id_map<buffer> &chunks,
                        shard_id_map<buffer> *decoded, int) {
  return _decode(want_to_read, chunks, decoded);
}

void ErasureCode::encode_delta(const buffer &, const buffer &, buffer *) {
  throw std::runtime_error("encode_delta not supported by this plugin");
}
void ErasureCode::apply_delta(const shard_id_map<buffer> &,
                              shard_id_map<buffer> &) {
  throw std::runtime_error("apply_delta not supported by this plugin");
}

int ErasureCode::parse(const ErasureCodeProfile &profile, std::ostream *ss) {
  // mapping string (ErasureCode.cc:491-510): 'D' = data position
  auto it = profile.find("mapping");
  if (it != profile.end()) {
    int position = 0;
    std::vector<shard_id_t> coding;
    for (char c : it->second) {
      if (c == 'D')
        chunk_mapping.push_back(shard_id_t((int8_t)position));
      else
        coding.push_back(shard_id_t((int8_t)position));
      position++;
    }
    chunk_mapping.insert(chunk_mapping.end(), coding.begin(), coding.end());
  }
  (void)ss;
  return 0;
}

int ErasureCode::to_int(const std::string &name, ErasureCodeProfile &profile,
                        int *value, const std::string &default_value,
                        std::ostream *ss) {
  // ErasureCode.cc:512-533: write default back on missing/empty/bad
  if (profile.find(name) == profile.end() || profile[name].empty())
    profile[name] = default_value;
  try {
    *value = std::stoi(profile[name]);
  } catch (const std::exception &) {
    if (ss)
      *ss << "could not convert " << name << "=" << profile[name]
          << " to int, set to default " << default_value << "\n";
    *value = std::stoi(default_value);
    profile[name] = default_value;
    return -EINVAL;
  }
  return 0;
}

int ErasureCode::to_bool(const std::string &name, ErasureCodeProfile &profile,
                         bool *value, const std::string &default_value,
                         std::ostream *) {
  if (profile.find(name) == profile.end() || profile[name].empty())
    profile[name] = default_value;
  const std::string &p = profile[name];
  *value = (p == "yes" || p == "true");
  return 0;
}

int ErasureCode::to_string(const std::string &name,
                           ErasureCodeProfile &profile, std::string *value,
                           const std::string &default_value, std::ostream *) {
  if (profile.find(name) == profile.end() || profile[name].empty())
    profile[name] = default_value;
  *value = profile[name];
  return 0;
}

}  // namespace ecx
