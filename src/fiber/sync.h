// brpc_amd: fiber-aware synchronization primitives built on butex.
// Parity: reference bthread/mutex.h, condition_variable.h,
// countdown_event.h, rwlock.h — blocking parks the fiber (or falls back to
// futex for plain pthreads), never blocks the worker.
#pragma once

#include <cerrno>

#include "base/time.h"
#include "fiber/butex.h"

namespace bam {

class FiberMutex {
 public:
  FiberMutex() : word_(butex_create()) { word_->store(0, std::memory_order_relaxed); }
  ~FiberMutex() { butex_destroy(word_); }
  FiberMutex(const FiberMutex&) = delete;
  FiberMutex& operator=(const FiberMutex&) = delete;

  void lock() {
    // 0 free, 1 locked, 2 locked with (possible) waiters.
    int expected = 0;
    if (word_->compare_exchange_strong(expected, 1, std::memory_order_acquire)) return;
    do {
      if (expected == 2 ||
          word_->compare_exchange_strong(expected, 2, std::memory_order_acquire)) {
        butex_wait(word_, 2, nullptr);
      }
      expected = 0;
    } while (!word_->compare_exchange_strong(expected, 2, std::memory_order_acquire));
  }

  bool try_lock() {
    int expected = 0;
    return word_->compare_exchange_strong(expected, 1, std::memory_order_acquire);
  }

  void unlock() {
    int prev = word_->exchange(0, std::memory_order_release);
    if (prev == 2) butex_wake(word_);
  }

  std::atomic<int>* butex_word() { return word_; }

 private:
  std::atomic<int>* word_;
};

class FiberMutexGuard {
 public:
  explicit FiberMutexGuard(FiberMutex& m) : m_(m) { m_.lock(); }
  ~FiberMutexGuard() { m_.unlock(); }

 private:
  FiberMutex& m_;
};

class FiberCond {
 public:
  FiberCond() : seq_(butex_create()) { seq_->store(0, std::memory_order_relaxed); }
  ~FiberCond() { butex_destroy(seq_); }

  // mutex must be held; atomically releases it while waiting.
  void wait(FiberMutex& mu) {
    int expected = seq_->load(std::memory_order_acquire);
    mu.unlock();
    butex_wait(seq_, expected, nullptr);
    mu.lock();
  }

  // Returns false on timeout.
  bool wait_until(FiberMutex& mu, int64_t abstime_us) {
    int expected = seq_->load(std::memory_order_acquire);
    mu.unlock();
    int rc = butex_wait(seq_, expected, &abstime_us);
    mu.lock();
    return !(rc == -1 && errno == ETIMEDOUT);
  }

  void notify_one() {
    seq_->fetch_add(1, std::memory_order_release);
    butex_wake(seq_);
  }
  void notify_all() {
    seq_->fetch_add(1, std::memory_order_release);
    butex_wake_all(seq_);
  }

 private:
  std::atomic<int>* seq_;
};

class CountdownEvent {
 public:
  explicit CountdownEvent(int initial = 1) : word_(butex_create()) {
    word_->store(initial, std::memory_order_relaxed);
  }
  ~CountdownEvent() { butex_destroy(word_); }

  void signal(int n = 1) {
    int prev = word_->fetch_sub(n, std::memory_order_acq_rel);
    if (prev - n <= 0) butex_wake_all(word_);
  }

  void add_count(int n = 1) { word_->fetch_add(n, std::memory_order_release); }

  void wait() {
    for (;;) {
      int v = word_->load(std::memory_order_acquire);
      if (v <= 0) return;
      butex_wait(word_, v, nullptr);
    }
  }

  bool timed_wait(int64_t abstime_us) {
    for (;;) {
      int v = word_->load(std::memory_order_acquire);
      if (v <= 0) return true;
      if (butex_wait(word_, v, &abstime_us) == -1 && errno == ETIMEDOUT) return false;
    }
  }

 private:
  std::atomic<int>* word_;
};

// Counting semaphore (parity: reference bthread/semaphore — butex-backed;
// blocking acquire parks the fiber, never the worker pthread).
class FiberSemaphore {
 public:
  explicit FiberSemaphore(int initial = 0) : count_(butex_create()) {
    count_->store(initial, std::memory_order_relaxed);
  }
  ~FiberSemaphore() { butex_destroy(count_); }

  void release(int n = 1) {
    count_->fetch_add(n, std::memory_order_release);
    if (n == 1) butex_wake(count_);
    else butex_wake_all(count_);
  }

  void acquire() {
    for (;;) {
      int v = count_->load(std::memory_order_acquire);
      if (v > 0 && count_->compare_exchange_weak(v, v - 1, std::memory_order_acq_rel)) return;
      if (v <= 0) butex_wait(count_, v, nullptr);
    }
  }

  bool try_acquire() {
    int v = count_->load(std::memory_order_acquire);
    return v > 0 && count_->compare_exchange_strong(v, v - 1, std::memory_order_acq_rel);
  }

 private:
  std::atomic<int>* count_;
};

// Reader-writer lock (parity: reference bthread/rwlock). Writer-preferring:
// state = -1 writer held, 0 free, >0 reader count; writers_waiting_ blocks
// new readers so writers cannot starve.
class FiberRWLock {
 public:
  FiberRWLock() : state_(butex_create()), writers_waiting_(0) {
    state_->store(0, std::memory_order_relaxed);
  }
  ~FiberRWLock() { butex_destroy(state_); }

  void rdlock() {
    for (;;) {
      int v = state_->load(std::memory_order_acquire);
      if (v >= 0 && writers_waiting_.load(std::memory_order_acquire) == 0) {
        if (state_->compare_exchange_weak(v, v + 1, std::memory_order_acq_rel)) return;
        continue;
      }
      butex_wait(state_, v, nullptr);
    }
  }

  void wrlock() {
    writers_waiting_.fetch_add(1, std::memory_order_acq_rel);
    for (;;) {
      int v = state_->load(std::memory_order_acquire);
      if (v == 0 && state_->compare_exchange_weak(v, -1, std::memory_order_acq_rel)) {
        writers_waiting_.fetch_sub(1, std::memory_order_acq_rel);
        return;
      }
      if (v != 0) butex_wait(state_, v, nullptr);
    }
  }

  void unlock() {
    int v = state_->load(std::memory_order_acquire);
    if (v == -1) {
      state_->store(0, std::memory_order_release);
    } else {
      if (state_->fetch_sub(1, std::memory_order_acq_rel) != 1) {
        // readers remain; only the last reader wakes writers
        return;
      }
    }
    butex_wake_all(state_);
  }

 private:
  std::atomic<int>* state_;
  std::atomic<int> writers_waiting_;
};

}  // namespace bam
