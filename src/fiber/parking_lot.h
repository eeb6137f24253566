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

  ParkingLot() : pending_signal_(0) {}

  // Wake up at most `num_task` waiters; returns #waiters woken.
  int signal(int num_task) {
    pending_signal_.fetch_add(num_task << 1, std::memory_order_release);
    return (int)sys_futex(&pending_signal_, FUTEX_WAKE, num_task, nullptr);
  }

  State get_state() { return State(pending_signal_.load(std::memory_order_acquire)); }

  // Sleep until signal() changes the state observed by get_state().
  void wait(const State& expected) {
    sys_futex(&pending_signal_, FUTEX_WAIT, expected.val_, nullptr);
  }

  void stop() {
    pending_signal_.fetch_or(1, std::memory_order_release);
    sys_futex(&pending_signal_, FUTEX_WAKE, 10000, nullptr);
  }

 private:
  std::atomic<int> pending_signal_;
};

}  // namespace bam
