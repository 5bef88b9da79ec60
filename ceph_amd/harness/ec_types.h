// ec_types.h — standalone-harness equivalents of the reference's shard-id
// and buffer types: shard_id_t (src/include/types.h:494-537),
// shard_id_set (bitset_set, src/common/bitset_set.h), shard_id_map
// (mini_flat_map, src/common/mini_flat_map.h:16-34 — a vector of optionals
// with O(1) lookup), and a bufferptr-lite aligned refcounted buffer
// (src/include/buffer.h:148-150 create_aligned). Re-created, not copied:
// only the semantics the EC path needs.
#pragma once

#include <cstdint>
#include <cstdlib>
#include <cstring>
#include <map>
#include <memory>
#include <optional>
#include <stdexcept>
#include <string>
#include <vector>

namespace ecx {

struct shard_id_t {
  int8_t id = 0;
  shard_id_t() = default;
  explicit constexpr shard_id_t(int8_t i) : id(i) {}
  explicit constexpr operator int() const { return id; }
  shard_id_t &operator++() { ++id; return *this; }
  bool operator==(const shard_id_t &o) const { return id == o.id; }
  bool operator<(const shard_id_t &o) const { return id < o.id; }
};

// dense bitset over shard ids 0..127 (bitset_set<128, shard_id_t> analogue)
class shard_id_set {
  uint64_t bits_[2] = {0, 0};

 public:
  void insert(shard_id_t s) { bits_[(uint8_t)s.id >> 6] |= 1ull << (s.id & 63); }
  void insert(int s) { insert(shard_id_t((int8_t)s)); }
  void erase(shard_id_t s) { bits_[(uint8_t)s.id >> 6] &= ~(1ull << (s.id & 63)); }
  void erase(int s) { erase(shard_id_t((int8_t)s)); }
  bool contains(shard_id_t s) const {
    return bits_[(uint8_t)s.id >> 6] >> (s.id & 63) & 1;
  }
  bool contains(int s) const { return contains(shard_id_t((int8_t)s)); }
  size_t size() const {
    return __builtin_popcountll(bits_[0]) + __builtin_popcountll(bits_[1]);
  }
  bool empty() const { return !bits_[0] && !bits_[1]; }
  bool includes(const shard_id_set &o) const {
    return (o.bits_[0] & ~bits_[0]) == 0 && (o.bits_[1] & ~bits_[1]) == 0;
  }
  uint64_t low_mask() const { return bits_[0]; }  // ids < 64 (k+m <= 64 here)

  class const_iterator {
    const shard_id_set *s_;
    int i_;
    void advance() {
      while (i_ < 128 && !s_->contains(shard_id_t((int8_t)i_))) ++i_;
    }

   public:
    const_iterator(const shard_id_set *s, int i) : s_(s), i_(i) { advance(); }
    shard_id_t operator*() const { return shard_id_t((int8_t)i_); }
    const_iterator &operator++() { ++i_; advance(); return *this; }
    bool operator!=(const const_iterator &o) const { return i_ != o.i_; }
  };
  const_iterator begin() const { return {this, 0}; }
  const_iterator end() const { return {this, 128}; }
};

// mini_flat_map analogue: vector of optionals indexed by shard id
template <typename T>
class shard_id_map {
  std::vector<std::optional<T>> v_;
  size_t n_ = 0;

 public:
  shard_id_map() : v_(0) {}
  explicit shard_id_map(int max_size) : v_(max_size) {}
  bool contains(shard_id_t s) const {
    return (size_t)(uint8_t)s.id < v_.size() && v_[s.id].has_value();
  }
  bool contains(int s) const { return contains(shard_id_t((int8_t)s)); }
  T &operator[](shard_id_t s) {
    if (!v_[s.id]) { v_[s.id].emplace(); ++n_; }
    return *v_[s.id];
  }
  T &operator[](int s) { return (*this)[shard_id_t((int8_t)s)]; }
  T &at(shard_id_t s) { return *v_[s.id]; }
  const T &at(shard_id_t s) const { return *v_[s.id]; }
  const T &at(int s) const { return *v_[(int8_t)s]; }
  void erase(shard_id_t s) {
    if (contains(s)) { v_[s.id].reset(); --n_; }
  }
  void erase(int s) { erase(shard_id_t((int8_t)s)); }
  size_t size() const { return n_; }
  bool empty() const { return n_ == 0; }
  int max_size() const { return (int)v_.size(); }

  template <typename Self>
  struct iter {
    Self *m;
    int i;
    void advance() {
      while (i < m->max_size() && !m->contains(shard_id_t((int8_t)i))) ++i;
    }
    iter(Self *mm, int ii) : m(mm), i(ii) { advance(); }
    auto operator*() const {
      return std::pair<shard_id_t, decltype((m->at(shard_id_t((int8_t)i))))>(
          shard_id_t((int8_t)i), m->at(shard_id_t((int8_t)i)));
    }
    iter &operator++() { ++i; advance(); return *this; }
    bool operator!=(const iter &o) const { return i != o.i; }
  };
  auto begin() { return iter<shard_id_map>{this, 0}; }
  auto end() { return iter<shard_id_map>{this, max_size()}; }
  auto begin() const { return iter<const shard_id_map>{this, 0}; }
  auto end() const { return iter<const shard_id_map>{this, max_size()}; }
};

// bufferptr-lite: refcounted aligned contiguous bytes
class buffer {
  std::shared_ptr<uint8_t[]> raw_;
  uint8_t *p_ = nullptr;
  size_t len_ = 0;

 public:
  buffer() = default;
  static buffer create_aligned(size_t len, size_t align = 64) {
    buffer b;
    void *p = nullptr;
    if (posix_memalign(&p, align, len ? len : align))
      throw std::bad_alloc();
    b.raw_ = std::shared_ptr<uint8_t[]>((uint8_t *)p,
                                        [](uint8_t *q) { free(q); });
    b.p_ = (uint8_t *)p;
    b.len_ = len;
    return b;
  }
  static buffer copy(const void *src, size_t len, size_t align = 64) {
    buffer b = create_aligned(len, align);
    std::memcpy(b.p_, src, len);
    return b;
  }
  // sub-view sharing the refcount (substr_of analogue)
  buffer substr(size_t off, size_t len) const {
    buffer b = *this;
    b.p_ = p_ + off;
    b.len_ = len;
    return b;
  }
  uint8_t *c_str() { return p_; }
  const uint8_t *c_str() const { return p_; }
  size_t length() const { return len_; }
  void zero() { std::memset(p_, 0, len_); }
};

using ErasureCodeProfile = std::map<std::string, std::string>;

}  // namespace ecx
