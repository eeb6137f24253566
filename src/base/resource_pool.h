// brpc_amd: id-addressed resource pool — O(1) 32-bit id -> T* and never
// frees memory, so a stale id dereference is safe (reads an old object).
// Parity: reference butil/resource_pool.h; foundation of the versioned-id
// idiom used by Socket / fiber ids (see rpc/versioned_ref.h).
#pragma once

#include <atomic>
#include <mutex>
#include <vector>

#include "base/logging.h"

namespace bam {

typedef uint32_t ResourceId;

template <typename T>
class ResourcePool {
 public:
  static const size_t kBlockSize = 256;  // items per block
  static const size_t kMaxBlocks = 1u << 16;

  // Gets a free item; *id receives its dense id.
  static T* get_resource(ResourceId* id) {
    Singleton& s = singleton();
    {
      std::lock_guard<std::mutex> lk(s.free_mu);
      if (!s.free_ids.empty()) {
        ResourceId rid = s.free_ids.back();
        s.free_ids.pop_back();
        *id = rid;
        return address_resource(rid);
      }
    }
    ResourceId rid = s.next_id.fetch_add(1, std::memory_order_relaxed);
    size_t block_idx = rid / kBlockSize;
    CHECK_LT(block_idx, kMaxBlocks) << "ResourcePool exhausted";
    Block* b = s.blocks[block_idx].load(std::memory_order_acquire);
    if (b == nullptr) {
      Block* nb = new Block;
      if (!s.blocks[block_idx].compare_exchange_strong(b, nb, std::memory_order_acq_rel)) {
        delete nb;  // raced; b now holds winner
      } else {
        b = nb;
      }
    }
    *id = rid;
    return &b->items[rid % kBlockSize];
  }

  static T* address_resource(ResourceId id) {
    Singleton& s = singleton();
    Block* b = s.blocks[id / kBlockSize].load(std::memory_order_acquire);
    if (b == nullptr) return nullptr;
    return &b->items[id % kBlockSize];
  }

  static void return_resource(ResourceId id) {
    Singleton& s = singleton();
    std::lock_guard<std::mutex> lk(s.free_mu);
    s.free_ids.push_back(id);
  }

 private:
  struct Block {
    T items[kBlockSize];
  };
  struct Singleton {
    std::atomic<ResourceId> next_id{0};
    std::mutex free_mu;
    std::vector<ResourceId> free_ids;
    std::atomic<Block*> blocks[kMaxBlocks];
    Singleton() {
      for (size_t i = 0; i < kMaxBlocks; ++i) blocks[i].store(nullptr, std::memory_order_relaxed);
    }
  };
  static Singleton& singleton() {
    // Heap-allocated and deliberately leaked: daemon threads (timer
    // thread, fiber workers) keep returning resources during static
    // destruction at process exit — destroying this vector/mutex then is
    // a use-after-free (caught by ASan in the test suite).
    static Singleton* s = new Singleton;
    return *s;
  }
};

template <typename T>
inline T* get_resource(ResourceId* id) {
  return ResourcePool<T>::get_resource(id);
}
template <typename T>
inline T* address_resource(ResourceId id) {
  return ResourcePool<T>::address_resource(id);
}
template <typename T>
inline void return_resource(ResourceId id) {
  ResourcePool<T>::return_resource(id);
}

}  // namespace bam
