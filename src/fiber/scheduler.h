// brpc_amd: fiber scheduler internals — TaskControl (global, owns worker
// pthreads + stealing) and TaskGroup (per-worker run context).
// Parity: reference bthread/task_control.h + task_group.h, clean-room.
#pragma once

#include <atomic>

// ASan fiber support (parity: reference bthread/stack_inl.h ASan hooks):
// every context switch notifies the sanitizer of the destination stack so
// fake-stack bookkeeping and use-after-return detection stay correct
// across fibers. No-ops without -fsanitize=address.
#if defined(__SANITIZE_ADDRESS__)
#define BAM_ASAN_ENABLED 1
#elif defined(__has_feature)
#if __has_feature(address_sanitizer)
#define BAM_ASAN_ENABLED 1
#endif
#endif
#ifdef BAM_ASAN_ENABLED
#include <sanitizer/common_interface_defs.h>
#define BAM_ASAN_START_SWITCH(fss, bottom, size) \
  __sanitizer_start_switch_fiber(fss, bottom, size)
#define BAM_ASAN_FINISH_SWITCH(fss, bottom_old, size_old) \
  __sanitizer_finish_switch_fiber(fss, bottom_old, size_old)
#else
#define BAM_ASAN_START_SWITCH(fss, bottom, size) ((void)0)
#define BAM_ASAN_FINISH_SWITCH(fss, bottom_old, size_old) ((void)0)
#endif
#include <deque>
#include <mutex>
#include <vector>

#include "fiber/context.h"
#include "fiber/fiber.h"
#include "fiber/parking_lot.h"
#include "fiber/work_stealing_queue.h"

namespace bam {

typedef void (*RemainedFn)(void*);

struct FiberMeta {
  void* ctx_sp = nullptr;
#ifdef BAM_ASAN_ENABLED
  void* asan_fake_stack = nullptr;
  const void* asan_stack_bottom = nullptr;
  size_t asan_stack_size = 0;
#endif
  void (*fn)(void*) = nullptr;
  void* arg = nullptr;
  char* stack_base = nullptr;
  size_t stack_size = 0;
  uint32_t index = 0;                  // ResourcePool id
  std::atomic<uint32_t> version{1};    // bumped at exit; fiber_t carries it
  std::atomic<int>* version_butex = nullptr;  // mirrors version; joiners wait here
  void* keytable = nullptr;                   // fiber-local storage (fiber/key.h)
  // Sleep/interrupt state (≙ reference bthread_interrupt/bthread_stop,
  // bthread/task_group.cpp interrupt paths): sleep_butex is lazily
  // created once per meta SLOT and never destroyed (metas are pooled),
  // so wakers can touch it without lifetime games.
  std::atomic<std::atomic<int>*> sleep_butex{nullptr};
  std::atomic<bool> interrupted{false};
  std::atomic<bool> stop_requested{false};
  bool is_main = false;
};

// Carried through bam_jump_context: tells the destination context what to
// do on behalf of the suspended source before anyone can wake it.
struct Transfer {
  RemainedFn remained = nullptr;
  void* remained_arg = nullptr;
};

class TaskControl;

class TaskGroup {
 public:
  explicit TaskGroup(TaskControl* c, int index);

  // Runs on the worker pthread; returns when the control stops.
  void run_main_loop();

  // --- called with this == tls_task_group ---
  // Suspend the current fiber; after the switch, run `remained(arg)` on the
  // next context (this is what publishes the suspended fiber: pushes it to
  // a queue / waiter list — never before the switch).
  void sched(RemainedFn remained, void* remained_arg);
  // Switch directly to `next` (urgent start fast path).
  void sched_to(FiberMeta* next, RemainedFn remained, void* remained_arg);

  FiberMeta* cur() const { return cur_; }
  TaskControl* control() const { return control_; }
  int index() const { return index_; }

  bool push_local(fiber_t t) { return rq_.push(t); }
  bool pop_local(fiber_t* t) { return rq_.pop(t); }
  bool steal_local(fiber_t* t) { return rq_.steal(t); }

  void push_remote(fiber_t t) {
    std::lock_guard<std::mutex> lk(remote_mu_);
    remote_.push_back(t);
  }
  bool pop_remote(fiber_t* t) {
    std::lock_guard<std::mutex> lk(remote_mu_);
    if (remote_.empty()) return false;
    *t = remote_.front();
    remote_.pop_front();
    return true;
  }

 private:
  friend class TaskControl;
  friend void fiber_entry_fn(void*);

  bool wait_task(fiber_t* t);  // pop/steal/park
  static void main_loop_resumed(Transfer* tr);

  TaskControl* control_;
  int index_;
  FiberMeta main_meta_;    // represents the worker pthread's own context
  FiberMeta* cur_;
  WorkStealingQueue<fiber_t> rq_;
  std::mutex remote_mu_;
  std::deque<fiber_t> remote_;
  uint64_t steal_seed_;
};

class TaskControl {
 public:
  static TaskControl* singleton();  // lazily starts workers

  int concurrency() const { return (int)groups_.size(); }
  TaskGroup* group(int i) { return groups_[i]; }

  // Queue a ready fiber from any thread. If the caller is a worker, prefers
  // its local queue; otherwise round-robins remote queues.
  void ready_to_run(fiber_t t, bool from_worker_local);
  void signal_workers(int n) { parking_lot_.signal(n); }

  bool stopped() const { return stopped_.load(std::memory_order_acquire); }

  ParkingLot& parking_lot() { return parking_lot_; }

  std::atomic<int64_t> nfibers_created{0};
  std::atomic<int64_t> nfibers_active{0};

  static void set_concurrency_hint(int n);

 private:
  TaskControl();
  void start_workers(int n);

  std::vector<TaskGroup*> groups_;
  std::atomic<bool> stopped_{false};
  ParkingLot parking_lot_;
  std::atomic<uint32_t> rr_{0};
};

extern thread_local TaskGroup* tls_task_group;

// ALWAYS use this accessor instead of reading tls_task_group directly in
// code that can run on a fiber: the compiler may cache the TLS address in a
// callee-saved register across a context switch, and a migrated fiber would
// then read the OLD worker pthread's slot. The noinline call forces a fresh
// TLS lookup on the current pthread.
__attribute__((noinline)) TaskGroup* current_task_group();

// --- internal helpers shared by butex.cc / fiber.cc ---
FiberMeta* fiber_meta_of(fiber_t t);             // nullptr if version mismatch
fiber_t fiber_id_of(FiberMeta* m);
FiberMeta* create_fiber_meta(void (*fn)(void*), void* arg, uint32_t stack_size);
void run_remained(Transfer* tr);

}  // namespace bam
