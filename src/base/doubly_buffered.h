// brpc_amd: DoublyBufferedData — read-mostly data with near-lock-free reads.
// Parity: reference butil/containers/doubly_buffered_data.h. Readers grab a
// per-thread mutex and read the foreground copy; Modify() mutates the
// background copy, flips the index, then serially acquires every reader
// mutex to ensure no reader still sees the old foreground. Backbone of
// every load balancer's server list.
#pragma once

#include <atomic>
#include <functional>
#include <memory>
#include <mutex>
#include <unordered_map>
#include <vector>

namespace bam {

template <typename T>
class DoublyBufferedData {
 public:
  class ScopedPtr {
   public:
    ScopedPtr() : data_(nullptr), lock_(nullptr) {}
    ~ScopedPtr() {
      if (lock_) lock_->unlock();
    }
    ScopedPtr(const ScopedPtr&) = delete;
    ScopedPtr& operator=(const ScopedPtr&) = delete;
    const T* get() const { return data_; }
    const T* operator->() const { return data_; }
    const T& operator*() const { return *data_; }

   private:
    friend class DoublyBufferedData;
    const T* data_;
    std::mutex* lock_;
  };

  DoublyBufferedData() : index_(0) {
    static std::atomic<uint64_t> next_id{1};
    id_ = next_id.fetch_add(1, std::memory_order_relaxed);
  }

  // Returns 0 on success. Holds the calling thread's wrapper lock until
  // `ptr` goes out of scope.
  int Read(ScopedPtr* ptr) {
    Wrapper* w = local_wrapper();
    w->mu.lock();
    ptr->data_ = &data_[index_.load(std::memory_order_acquire)];
    ptr->lock_ = &w->mu;
    return 0;
  }

  // fn(background_copy) -> bool; if true, flips and applies to the other
  // copy too so both stay in sync.
  template <typename Fn>
  size_t Modify(Fn&& fn) {
    std::lock_guard<std::mutex> lk(modify_mu_);
    int bg = !index_.load(std::memory_order_relaxed);
    if (!fn(data_[bg])) return 0;
    index_.store(bg, std::memory_order_release);
    // Wait until every reader has left the old foreground.
    {
      std::lock_guard<std::mutex> wlk(wrappers_mu_);
      for (auto& w : wrappers_) {
        w->mu.lock();
        w->mu.unlock();
      }
    }
    fn(data_[!bg]);
    return 1;
  }

 private:
  struct Wrapper {
    std::mutex mu;
  };

  Wrapper* local_wrapper() {
    // Keyed by a process-unique instance id (NOT `this`): a destroyed
    // instance's address can be reused, and handing back the old wrapper
    // would lock a destroyed mutex. The TLS map holds shared_ptrs so stale
    // entries stay valid memory until the thread exits.
    static thread_local std::unordered_map<uint64_t, std::shared_ptr<Wrapper>> tls_map;
    auto it = tls_map.find(id_);
    if (it != tls_map.end()) return it->second.get();
    auto w = std::make_shared<Wrapper>();
    {
      std::lock_guard<std::mutex> lk(wrappers_mu_);
      wrappers_.push_back(w);
    }
    tls_map[id_] = w;
    return w.get();
  }

  T data_[2];
  uint64_t id_;
  std::atomic<int> index_;
  std::mutex modify_mu_;
  std::mutex wrappers_mu_;
  std::vector<std::shared_ptr<Wrapper>> wrappers_;
};

}  // namespace bam
