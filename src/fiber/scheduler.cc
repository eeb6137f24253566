#include "fiber/scheduler.h"

#include <stdlib.h>
#include <string.h>
#include <sys/mman.h>
#include <unistd.h>

#include <thread>

#include "base/fast_rand.h"
#include "base/logging.h"
#include "base/resource_pool.h"
#include "fiber/butex.h"
#include "fiber/key.h"

namespace bam {

thread_local TaskGroup* tls_task_group = nullptr;

__attribute__((noinline)) TaskGroup* current_task_group() {
  asm volatile("" ::: "memory");  // defeat TLS-address caching across switches
  return tls_task_group;
}

static const uint32_t kDefaultStackSize = 256 * 1024;
static const size_t kGuardSize = 4096;

// ---------------- stack pool ----------------
namespace {

struct StackPool {
  std::mutex mu;
  // Pooled size classes (parity: reference bthread SMALL/NORMAL/LARGE
  // stack attrs, bthread/stack_inl.h:31): 32 KiB / 256 KiB / 2 MiB.
  // High-fiber-count workloads request FIBER_ATTR_SMALL-equivalent sizes
  // instead of overpaying 256 KiB each.
  std::vector<char*> free_stacks[3];
  static const size_t kMaxPooled = 64;
};
const size_t kStackClasses[3] = {32 * 1024, 256 * 1024, 2 * 1024 * 1024};
int stack_class_of(size_t size) {
  for (int i = 0; i < 3; ++i)
    if (size == kStackClasses[i]) return i;
  return -1;
}
StackPool& stack_pool() {
  static StackPool* p = new StackPool;
  return *p;
}

char* alloc_stack(size_t size) {
  const int cls = stack_class_of(size);
  if (cls >= 0) {
    StackPool& p = stack_pool();
    std::lock_guard<std::mutex> lk(p.mu);
    if (!p.free_stacks[cls].empty()) {
      char* s = p.free_stacks[cls].back();
      p.free_stacks[cls].pop_back();
      return s;
    }
  }
  char* mem = (char*)mmap(nullptr, size + kGuardSize, PROT_READ | PROT_WRITE,
                          MAP_PRIVATE | MAP_ANONYMOUS | MAP_STACK, -1, 0);
  if (mem == MAP_FAILED) return nullptr;
  mprotect(mem, kGuardSize, PROT_NONE);  // guard page at the low end
  return mem + kGuardSize;
}

void free_stack(char* base, size_t size) {
  const int cls = stack_class_of(size);
  if (cls >= 0) {
    StackPool& p = stack_pool();
    std::lock_guard<std::mutex> lk(p.mu);
    if (p.free_stacks[cls].size() < StackPool::kMaxPooled) {
      p.free_stacks[cls].push_back(base);
      return;
    }
  }
  munmap(base - kGuardSize, size + kGuardSize);
}

}  // namespace

// ---------------- fiber meta ----------------

void fiber_entry_fn(void* raw);

FiberMeta* fiber_meta_of(fiber_t t) {
  if (t == 0) return nullptr;
  ResourceId rid = (uint32_t)(t & 0xffffffffu) - 1;
  FiberMeta* m = address_resource<FiberMeta>(rid);
  if (m == nullptr) return nullptr;
  if (m->version.load(std::memory_order_acquire) != (uint32_t)(t >> 32)) return nullptr;
  return m;
}

fiber_t fiber_id_of(FiberMeta* m) {
  return ((uint64_t)m->version.load(std::memory_order_relaxed) << 32) | (m->index + 1);
}

FiberMeta* create_fiber_meta(void (*fn)(void*), void* arg, uint32_t stack_size) {
  if (stack_size == 0) stack_size = kDefaultStackSize;
  // Round odd requests up to a pooled class so stacks recycle.
  for (size_t c : kStackClasses) {
    if (stack_size <= c) {
      stack_size = (uint32_t)c;
      break;
    }
  }
  ResourceId rid;
  FiberMeta* m = get_resource<FiberMeta>(&rid);
  CHECK(m != nullptr);
  m->index = rid;
  if (m->version_butex == nullptr) {
    m->version_butex = butex_create();
    m->version_butex->store((int)m->version.load(std::memory_order_relaxed),
                            std::memory_order_relaxed);
  }
  m->fn = fn;
  m->arg = arg;
  // Meta slots are pooled: a previous fiber's interrupt/stop marks must
  // not leak into this one (they poisoned later sleeps before this reset).
  m->interrupted.store(false, std::memory_order_relaxed);
  m->stop_requested.store(false, std::memory_order_relaxed);
  m->stack_size = stack_size;
  m->stack_base = alloc_stack(stack_size);
  CHECK(m->stack_base != nullptr) << "fiber stack allocation failed";
  m->ctx_sp = make_context(m->stack_base, stack_size, fiber_entry_fn);
  m->is_main = false;
  return m;
}

static void release_fiber_meta(void* raw) {
  FiberMeta* m = (FiberMeta*)raw;
  free_stack(m->stack_base, m->stack_size);
  m->stack_base = nullptr;
  m->ctx_sp = nullptr;
  TaskControl::singleton()->nfibers_active.fetch_sub(1, std::memory_order_relaxed);
  return_resource<FiberMeta>(m->index);
}

void run_remained(Transfer* tr) {
  if (tr != nullptr && tr->remained != nullptr) {
    RemainedFn fn = tr->remained;
    void* arg = tr->remained_arg;
    tr->remained = nullptr;
    fn(arg);
  }
}

// Entry point of every fresh fiber context. Never returns.
void fiber_entry_fn(void* raw) {
  BAM_ASAN_FINISH_SWITCH(nullptr, nullptr, nullptr);
  run_remained((Transfer*)raw);
  for (;;) {
    TaskGroup* g = current_task_group();
    FiberMeta* m = g->cur();
    m->fn(m->arg);
    destroy_current_keytable();  // run fiber-local destructors
    // Fiber finished: invalidate the id, wake joiners, then free resources
    // from the next context (we cannot free the stack we stand on).
    g = current_task_group();  // may have migrated
    m = g->cur();
    uint32_t nv = m->version.fetch_add(1, std::memory_order_acq_rel) + 1;
    m->version_butex->store((int)nv, std::memory_order_release);
    butex_wake_all(m->version_butex);
    g->sched(release_fiber_meta, m);
    // A released meta's context is never resumed; not reached.
    LOG(FATAL) << "resumed a dead fiber context";
  }
}

// ---------------- TaskGroup ----------------

TaskGroup::TaskGroup(TaskControl* c, int index)
    : control_(c), index_(index), cur_(nullptr), rq_(8192), steal_seed_(fast_rand()) {
  main_meta_.is_main = true;
}

void TaskGroup::sched_to(FiberMeta* next, RemainedFn remained, void* remained_arg) {
  FiberMeta* cur = cur_;
  cur_ = next;
  Transfer tr{remained, remained_arg};
#ifdef BAM_ASAN_ENABLED
  const void* next_bottom = next->is_main ? nullptr : next->stack_base;
  size_t next_size = next->is_main ? 0 : next->stack_size;
  BAM_ASAN_START_SWITCH(&cur->asan_fake_stack, next_bottom, next_size);
#endif
  Transfer* ret = (Transfer*)bam_jump_context(&cur->ctx_sp, next->ctx_sp, &tr);
#ifdef BAM_ASAN_ENABLED
  BAM_ASAN_FINISH_SWITCH(cur->asan_fake_stack, &cur->asan_stack_bottom,
                         &cur->asan_stack_size);
#endif
  run_remained(ret);
  // NOTE: `this` may be stale here if the fiber migrated; callers must
  // re-read tls_task_group after sched()/sched_to() returns.
}

void TaskGroup::sched(RemainedFn remained, void* remained_arg) {
  fiber_t t;
  FiberMeta* next = nullptr;
  if (pop_local(&t)) next = fiber_meta_of(t);
  if (next == nullptr) next = &main_meta_;
  sched_to(next, remained, remained_arg);
}

bool TaskGroup::wait_task(fiber_t* t) {
  int idle_spins = 0;
  while (!control_->stopped()) {
    if (pop_local(t)) return true;
    if (pop_remote(t)) return true;
    const int n = control_->concurrency();
    for (int round = 0; round < 2; ++round) {
      uint64_t off = fast_rand();
      for (int i = 0; i < n; ++i) {
        TaskGroup* g = control_->group((off + i) % n);
        if (g == this) continue;
        if (g->steal_local(t)) return true;
        if (g->pop_remote(t)) return true;
      }
    }
    // Bounded spin before parking: a futex sleep/wake costs ~5-15 µs per
    // hop, which dominates single-flight RPC latency. Spin ~40 µs first
    // (pause-loop), then park for real.
    if (++idle_spins < 200) {
      for (int k = 0; k < 64; ++k) __builtin_ia32_pause();
      continue;
    }
    idle_spins = 0;
    ParkingLot::State st = control_->parking_lot().get_state();
    if (st.stopped()) return false;
    if (pop_local(t) || pop_remote(t)) return true;  // re-check before parking
    control_->parking_lot().wait(st);
  }
  return false;
}

void TaskGroup::run_main_loop() {
  tls_task_group = this;
  cur_ = &main_meta_;
  fiber_t t;
  while (wait_task(&t)) {
    FiberMeta* m = fiber_meta_of(t);
    if (m == nullptr) continue;  // stale (should not happen for queued fibers)
    Transfer tr{nullptr, nullptr};
    cur_ = m;
#ifdef BAM_ASAN_ENABLED
    BAM_ASAN_START_SWITCH(&main_meta_.asan_fake_stack, m->stack_base, m->stack_size);
#endif
    Transfer* ret = (Transfer*)bam_jump_context(&main_meta_.ctx_sp, m->ctx_sp, &tr);
#ifdef BAM_ASAN_ENABLED
    BAM_ASAN_FINISH_SWITCH(main_meta_.asan_fake_stack, nullptr, nullptr);
#endif
    run_remained(ret);
    cur_ = &main_meta_;
  }
}

// ---------------- TaskControl ----------------

static std::atomic<int> g_concurrency_hint{0};

void TaskControl::set_concurrency_hint(int n) {
  g_concurrency_hint.store(n, std::memory_order_relaxed);
}

TaskControl::TaskControl() {
  int n = g_concurrency_hint.load(std::memory_order_relaxed);
  bool chosen = n > 0;
  if (!chosen) {
    const char* env = getenv("BAM_FIBER_WORKERS");
    if (env != nullptr) {
      n = atoi(env);
      chosen = n > 0;
    }
  }
  if (n <= 0) n = (int)std::thread::hardware_concurrency();
  if (n <= 0) n = 4;
  // Fewer workers win on big machines: the steal scan is O(workers) and
  // the spin-before-park burn multiplies. Same-box measurements (MI355X
  // host, echo bench): cap 8 vs 16 -> HBM p99 ~380 vs ~700-1850 us at
  // equal QPS, host path +12% QPS. The reference's default is 8+1
  // workers too. An explicit BAM_FIBER_WORKERS / concurrency hint is
  // taken as-is (no cap).
  if (!chosen && n > 8) n = 8;
  start_workers(n);
}

TaskControl* TaskControl::singleton() {
  static TaskControl* c = new TaskControl;  // leaked; workers run for process life
  return c;
}

void TaskControl::start_workers(int n) {
  groups_.reserve(n);
  for (int i = 0; i < n; ++i) groups_.push_back(new TaskGroup(this, i));
  for (int i = 0; i < n; ++i) {
    TaskGroup* g = groups_[i];
    std::thread([g] { g->run_main_loop(); }).detach();
  }
}

void TaskControl::ready_to_run(fiber_t t, bool prefer_local) {
  TaskGroup* g = current_task_group();
  if (prefer_local && g != nullptr && g->control() == this) {
    if (!g->push_local(t)) g->push_remote(t);
  } else {
    TaskGroup* target =
        groups_[rr_.fetch_add(1, std::memory_order_relaxed) % groups_.size()];
    target->push_remote(t);
  }
  parking_lot_.signal(1);
}

}  // namespace bam
