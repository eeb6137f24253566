// brpc_amd: Chase-Lev work-stealing deque (fixed capacity).
// Parity: reference bthread/work_stealing_queue.h. Owner pushes/pops at the
// bottom without contention; thieves steal from the top with CAS.
#pragma once

#include <atomic>
#include <cstdint>

namespace bam {

template <typename T>
class WorkStealingQueue {
 public:
  explicit WorkStealingQueue(size_t cap = 8192)
      : bottom_(0), top_(0), cap_(cap), mask_(cap - 1), buf_(new T[cap]) {
    // cap must be a power of two
  }
  ~WorkStealingQueue() { delete[] buf_; }

  // Owner-only. Returns false when full.
  bool push(const T& v) {
    uint64_t b = bottom_.load(std::memory_order_relaxed);
    uint64_t t = top_.load(std::memory_order_acquire);
    if (b - t >= cap_) return false;
    buf_[b & mask_] = v;
    bottom_.store(b + 1, std::memory_order_release);
    return true;
  }

  // Owner-only. LIFO pop.
  bool pop(T* out) {
    uint64_t b = bottom_.load(std::memory_order_relaxed);
    uint64_t t = top_.load(std::memory_order_relaxed);
    if (t >= b) return false;
    --b;
    bottom_.store(b, std::memory_order_relaxed);
    std::atomic_thread_fence(std::memory_order_seq_cst);
    t = top_.load(std::memory_order_relaxed);
    if (t > b) {
      bottom_.store(b + 1, std::memory_order_relaxed);
      return false;
    }
    *out = buf_[b & mask_];
    if (t != b) return true;
    // Last element: race with thieves.
    bool won = top_.compare_exchange_strong(t, t + 1, std::memory_order_seq_cst,
                                            std::memory_order_relaxed);
    bottom_.store(b + 1, std::memory_order_relaxed);
    return won;
  }

  // Any thread. FIFO steal.
  bool steal(T* out) {
    uint64_t t = top_.load(std::memory_order_acquire);
    std::atomic_thread_fence(std::memory_order_seq_cst);
    uint64_t b = bottom_.load(std::memory_order_acquire);
    if (t >= b) return false;
    T v = buf_[t & mask_];
    if (!top_.compare_exchange_strong(t, t + 1, std::memory_order_seq_cst,
                                      std::memory_order_relaxed)) {
      return false;
    }
    *out = v;
    return true;
  }

  size_t volatile_size() const {
    uint64_t b = bottom_.load(std::memory_order_relaxed);
    uint64_t t = top_.load(std::memory_order_relaxed);
    return b > t ? (size_t)(b - t) : 0;
  }

  size_t capacity() const { return cap_; }

 private:
  std::atomic<uint64_t> bottom_;
  char pad_[64 - sizeof(std::atomic<uint64_t>)];
  std::atomic<uint64_t> top_;
  char pad2_[64 - sizeof(std::atomic<uint64_t>)];
  size_t cap_;
  size_t mask_;
  T* buf_;
};

}  // namespace bam
