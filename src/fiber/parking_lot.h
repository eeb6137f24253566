// brpc_amd: ParkingLot — futex-based sleep/wake for idle fiber workers.
// Parity: reference bthread/parking_lot.h.
#pragma once

#include <linux/futex.h>
#include <sys/syscall.h>
#include <unistd.h>

#include <atomic>

namespace bam {

inline long sys_futex(void* addr, int op, int val, const struct timespec* timeout) {
  return syscall(SYS_futex, addr, op | FUTEX_PRIVATE_FLAG, val, timeout, nullptr, 0);
}

class ParkingLot {
 public:
  class State {
   public:
    State() : val_(0) {}
    bool stopped() const { return val_ & 1; }

   private:
    friend class ParkingLot;
    explicit State(int v) : val_(v) {}
    int val_;
  };

  ParkingLot() : pending_signal_(0), waiters_(0) {}

  // Wake up at most `num_task` waiters; returns #waiters woken.
  // Fast path: under load every worker is spinning (not futex-parked), so
  // skip the FUTEX_WAKE syscall when nobody waits. A waiter that loses the
  // race (increments waiters_ after we read 0) still cannot sleep through
  // this signal: its futex_wait sees pending_signal_ != expected and
  // returns immediately.
  int signal(int num_task) {
    pending_signal_.fetch_add(num_task << 1, std::memory_order_release);
    if (waiters_.load(std::memory_order_acquire) == 0) return 0;
    return (int)sys_futex(&pending_signal_, FUTEX_WAKE, num_task, nullptr);
  }

  State get_state() { return State(pending_signal_.load(std::memory_order_acquire)); }

  // Sleep until signal() changes the state observed by get_state().
  void wait(const State& expected) {
    waiters_.fetch_add(1, std::memory_order_acq_rel);
    sys_futex(&pending_signal_, FUTEX_WAIT, expected.val_, nullptr);
    waiters_.fetch_sub(1, std::memory_order_acq_rel);
  }

  void stop() {
    pending_signal_.fetch_or(1, std::memory_order_release);
    sys_futex(&pending_signal_, FUTEX_WAKE, 10000, nullptr);
  }

 private:
  std::atomic<int> pending_signal_;
  std::atomic<int> waiters_;
};

}  // namespace bam
