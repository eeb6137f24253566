// brpc_amd: lock-free-ish object pool (TLS freelist + global overflow).
// Parity: reference butil/object_pool.h — get_object/return_object with
// thread-local caches so the hot path is pointer pops with no atomics.
#pragma once

#include <mutex>
#include <new>
#include <vector>

namespace bam {

template <typename T>
class ObjectPool {
 public:
  static const size_t kLocalCap = 256;     // max objects cached per thread
  static const size_t kTransferBatch = 64; // moved between TLS and global

  static T* get() {
    Local& lc = local();
    if (lc.items.empty()) refill(lc);
    if (!lc.items.empty()) {
      T* obj = lc.items.back();
      lc.items.pop_back();
      return obj;
    }
    return new (std::nothrow) T;
  }

  static void put(T* obj) {
    if (obj == nullptr) return;
    Local& lc = local();
    lc.items.push_back(obj);
    if (lc.items.size() >= kLocalCap) spill(lc);
  }

  // Objects alive in global freelist (diagnostics only).
  static size_t free_count() {
    Global& g = global();
    std::lock_guard<std::mutex> lk(g.mu);
    return g.items.size();
  }

 private:
  struct Global {
    std::mutex mu;
    std::vector<T*> items;
  };
  struct Local {
    std::vector<T*> items;
    ~Local() {
      // Return everything to the global pool on thread exit.
      Global& g = global();
      std::lock_guard<std::mutex> lk(g.mu);
      g.items.insert(g.items.end(), items.begin(), items.end());
      items.clear();
    }
  };

  static Global& global() {
    // Leaked on purpose: thread-exit Local destructors and daemon threads
    // return objects during static destruction (see resource_pool.h).
    static Global* g = new Global;
    return *g;
  }
  static Local& local() {
    static thread_local Local lc;
    return lc;
  }

  static void refill(Local& lc) {
    Global& g = global();
    std::lock_guard<std::mutex> lk(g.mu);
    size_t n = g.items.size() < kTransferBatch ? g.items.size() : kTransferBatch;
    for (size_t i = 0; i < n; ++i) {
      lc.items.push_back(g.items.back());
      g.items.pop_back();
    }
  }

  static void spill(Local& lc) {
    Global& g = global();
    std::lock_guard<std::mutex> lk(g.mu);
    for (size_t i = 0; i < kTransferBatch && !lc.items.empty(); ++i) {
      g.items.push_back(lc.items.back());
      lc.items.pop_back();
    }
  }
};

template <typename T>
inline T* get_object() {
  return ObjectPool<T>::get();
}
template <typename T>
inline void return_object(T* obj) {
  ObjectPool<T>::put(obj);
}

}  // namespace bam
