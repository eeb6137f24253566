// brpc_amd: open-addressing hash map with linear probing.
// Parity: reference butil/containers/flat_map.h (used for method maps,
// HTTP headers). Keys and values stored inline; no per-node allocation.
#pragma once

#include <functional>
#include <utility>
#include <vector>

namespace bam {

template <typename K, typename V, typename Hash = std::hash<K>, typename Eq = std::equal_to<K>>
class FlatMap {
 public:
  struct Slot {
    K key;
    V value;
    uint8_t state = 0;  // 0 empty, 1 used, 2 tombstone
  };

  FlatMap() : slots_(16), used_(0) {}

  V* seek(const K& key) {
    size_t mask = slots_.size() - 1;
    size_t i = Hash()(key) & mask;
    for (size_t probes = 0; probes <= mask; ++probes, i = (i + 1) & mask) {
      Slot& s = slots_[i];
      if (s.state == 0) return nullptr;
      if (s.state == 1 && Eq()(s.key, key)) return &s.value;
    }
    return nullptr;
  }
  const V* seek(const K& key) const { return const_cast<FlatMap*>(this)->seek(key); }

  V& operator[](const K& key) {
    if (used_ * 4 >= slots_.size() * 3) rehash(slots_.size() * 2);
    size_t mask = slots_.size() - 1;
    size_t i = Hash()(key) & mask;
    size_t first_tomb = (size_t)-1;
    for (;; i = (i + 1) & mask) {
      Slot& s = slots_[i];
      if (s.state == 1 && Eq()(s.key, key)) return s.value;
      if (s.state == 2 && first_tomb == (size_t)-1) first_tomb = i;
      if (s.state == 0) {
        size_t at = (first_tomb != (size_t)-1) ? first_tomb : i;
        Slot& t = slots_[at];
        t.key = key;
        t.value = V();
        t.state = 1;
        ++used_;
        return t.value;
      }
    }
  }

  bool insert(const K& key, const V& value) {
    V& v = (*this)[key];
    v = value;
    return true;
  }

  size_t erase(const K& key) {
    size_t mask = slots_.size() - 1;
    size_t i = Hash()(key) & mask;
    for (size_t probes = 0; probes <= mask; ++probes, i = (i + 1) & mask) {
      Slot& s = slots_[i];
      if (s.state == 0) return 0;
      if (s.state == 1 && Eq()(s.key, key)) {
        s.state = 2;
        s.value = V();
        --used_;
        return 1;
      }
    }
    return 0;
  }

  size_t size() const { return used_; }
  bool empty() const { return used_ == 0; }
  void clear() {
    slots_.assign(16, Slot());
    used_ = 0;
  }

  template <typename Fn>
  void for_each(Fn&& fn) const {
    for (const Slot& s : slots_)
      if (s.state == 1) fn(s.key, s.value);
  }
  template <typename Fn>
  void for_each_mutable(Fn&& fn) {
    for (Slot& s : slots_)
      if (s.state == 1) fn(s.key, s.value);
  }

 private:
  void rehash(size_t new_cap) {
    std::vector<Slot> old;
    old.swap(slots_);
    slots_.assign(new_cap, Slot());
    used_ = 0;
    for (Slot& s : old) {
      if (s.state == 1) (*this)[s.key] = std::move(s.value);
    }
  }

  std::vector<Slot> slots_;
  size_t used_;
};

// Case-insensitive string helpers for HTTP headers (parity:
// butil/containers/case_ignored_flat_map.h).
struct CaseIgnoredHash {
  size_t operator()(const std::string& s) const {
    size_t h = 1315423911u;
    for (char c : s) h = (h << 5) ^ (size_t)(c | 0x20) ^ (h >> 2);
    return h;
  }
};
struct CaseIgnoredEq {
  bool operator()(const std::string& a, const std::string& b) const {
    if (a.size() != b.size()) return false;
    for (size_t i = 0; i < a.size(); ++i)
      if ((a[i] | 0x20) != (b[i] | 0x20)) return false;
    return true;
  }
};
template <typename V>
using CaseIgnoredFlatMap = FlatMap<std::string, V, CaseIgnoredHash, CaseIgnoredEq>;

}  // namespace bam
