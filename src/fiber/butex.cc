#include "fiber/butex.h"

#include <errno.h>
#include <execinfo.h>

#include <mutex>

#include "base/logging.h"
#include "base/time.h"
#include "fiber/parking_lot.h"  // sys_futex
#include "fiber/scheduler.h"
#include "fiber/timer_thread.h"

namespace bam {

namespace {

struct Butex;

struct ButexWaiter {
  ButexWaiter* next = nullptr;
  ButexWaiter* prev = nullptr;
  enum Type { FIBER, PTHREAD } type;
  FiberMeta* meta = nullptr;   // FIBER
  std::atomic<int> sig{0};     // PTHREAD: futex word
  int expected = 0;
  Butex* owner = nullptr;
  std::atomic<int> state{0};   // 0 waiting, 1 woken, 2 timedout
  TimerId timer_id = 0;
  int64_t abstime_us = 0;
  bool has_abstime = false;
};

// value must stay the first member: butex_create returns &value and the
// other entry points recover the Butex via a cast.
struct Butex {
  std::atomic<int> value;
  std::mutex mu;
  ButexWaiter head;  // circular sentinel

  Butex() {
    head.next = &head;
    head.prev = &head;
  }
};

inline Butex* container_of_value(std::atomic<int>* v) { return (Butex*)v; }

inline void list_insert(ButexWaiter* head, ButexWaiter* w) {
  w->prev = head->prev;
  w->next = head;
  head->prev->next = w;
  head->prev = w;
}

inline void list_remove(ButexWaiter* w) {
  w->prev->next = w->next;
  w->next->prev = w->prev;
  w->next = w->prev = nullptr;
}

inline bool in_list(ButexWaiter* w) { return w->next != nullptr; }

void wake_one_waiter(ButexWaiter* w, int new_state) {
  w->state.store(new_state, std::memory_order_release);
  if (w->type == ButexWaiter::FIBER) {
    TaskControl::singleton()->ready_to_run(fiber_id_of(w->meta), /*prefer_local=*/true);
  } else {
    w->sig.store(1, std::memory_order_release);
    sys_futex(&w->sig, FUTEX_WAKE, 1, nullptr);
  }
}

// Timer callback: a = Butex*, b = ButexWaiter*. Only dereferences the
// waiter if it is still linked in the butex's list (checked by pointer
// identity under the lock), so a concurrently-woken-and-freed waiter is
// never touched.
void butex_timeout_cb(void* a, void* b) {
  Butex* bt = (Butex*)a;
  ButexWaiter* target = (ButexWaiter*)b;
  ButexWaiter* found = nullptr;
  {
    std::lock_guard<std::mutex> lk(bt->mu);
    for (ButexWaiter* w = bt->head.next; w != &bt->head; w = w->next) {
      if (w == target) {
        found = w;
        list_remove(w);
        break;
      }
    }
  }
  if (found != nullptr) wake_one_waiter(found, 2 /*timedout*/);
}

// Remained closure: runs on the next context after the waiter's fiber has
// fully switched away — this is what makes parking race-free with wake.
void remained_add_waiter(void* raw) {
  ButexWaiter* w = (ButexWaiter*)raw;
  Butex* b = w->owner;
  b->mu.lock();
  if (b->value.load(std::memory_order_acquire) != w->expected) {
    // Value changed while switching: don't park, requeue the fiber.
    b->mu.unlock();
    w->state.store(1, std::memory_order_release);
    TaskControl::singleton()->ready_to_run(fiber_id_of(w->meta), true);
    return;
  }
  list_insert(&b->head, w);
  if (w->has_abstime) {
    w->timer_id = timer_add(w->abstime_us, butex_timeout_cb, b, w);
  }
  b->mu.unlock();
}

}  // namespace

std::atomic<int>* butex_create() {
  Butex* b = new Butex;
  return &b->value;
}

void butex_destroy(std::atomic<int>* v) {
  if (v == nullptr) return;
  Butex* b = container_of_value(v);
  CHECK(b->head.next == &b->head) << "destroying butex with waiters";
  delete b;
}

// Contention accounting (parity: the reference's contention profiler —
// instrumented bthread_mutex feeding /hotspots/contention). We keep
// process-global park counts + parked time, exposed as vars
// fiber_butex_{waits,wait_us} (registered in server.cc).
static std::atomic<int64_t> g_butex_waits{0};
static std::atomic<int64_t> g_butex_wait_us{0};
int64_t butex_total_waits() { return g_butex_waits.load(std::memory_order_relaxed); }
int64_t butex_total_wait_us() { return g_butex_wait_us.load(std::memory_order_relaxed); }

// Sampled contention sites: cheap ring, 1/64 parks pay a backtrace().
namespace {
constexpr int kContentionRing = 4096;
std::mutex g_cont_mu;
ContentionSample g_cont_ring[kContentionRing];
std::atomic<uint64_t> g_cont_n{0};

void record_contention(int64_t wait_us) {
  static thread_local uint32_t tl_counter = 0;
  if ((tl_counter++ & 63) != 0) return;
  ContentionSample s;
  s.nframes = backtrace(s.frames, 4);
  s.wait_us = wait_us;
  uint64_t idx = g_cont_n.fetch_add(1, std::memory_order_relaxed) % kContentionRing;
  std::lock_guard<std::mutex> lk(g_cont_mu);
  g_cont_ring[idx] = s;
}
}  // namespace

size_t butex_contention_samples(ContentionSample* out, size_t max) {
  uint64_t n = g_cont_n.load(std::memory_order_relaxed);
  size_t have = n < (uint64_t)kContentionRing ? (size_t)n : (size_t)kContentionRing;
  if (have > max) have = max;
  std::lock_guard<std::mutex> lk(g_cont_mu);
  for (size_t i = 0; i < have; ++i) out[i] = g_cont_ring[i];
  return have;
}

int butex_wait(std::atomic<int>* v, int expected, const int64_t* abstime_us) {
  Butex* b = container_of_value(v);
  if (b->value.load(std::memory_order_acquire) != expected) {
    errno = EWOULDBLOCK;
    return -1;
  }
  if (abstime_us != nullptr && *abstime_us <= monotonic_time_us()) {
    errno = ETIMEDOUT;
    return -1;
  }
  TaskGroup* g = current_task_group();
  if (g != nullptr && g->cur() != nullptr && !g->cur()->is_main) {
    // Fiber path: park on the waiter list via the remained closure.
    ButexWaiter w;
    w.type = ButexWaiter::FIBER;
    w.meta = g->cur();
    w.expected = expected;
    w.owner = b;
    if (abstime_us != nullptr) {
      w.has_abstime = true;
      w.abstime_us = *abstime_us;
    }
    g_butex_waits.fetch_add(1, std::memory_order_relaxed);
    const int64_t park_t0 = monotonic_time_us();
    g->sched(remained_add_waiter, &w);
    // Resumed (possibly on another worker).
    const int64_t parked = monotonic_time_us() - park_t0;
    g_butex_wait_us.fetch_add(parked, std::memory_order_relaxed);
    record_contention(parked);
    if (w.timer_id != 0) timer_delete(w.timer_id);
    if (w.state.load(std::memory_order_acquire) == 2) {
      errno = ETIMEDOUT;
      return -1;
    }
    return 0;
  }
  // Pthread path: futex on a private per-waiter word.
  g_butex_waits.fetch_add(1, std::memory_order_relaxed);
  const int64_t park_t0 = monotonic_time_us();
  ButexWaiter w;
  w.type = ButexWaiter::PTHREAD;
  w.expected = expected;
  w.owner = b;
  {
    std::lock_guard<std::mutex> lk(b->mu);
    if (b->value.load(std::memory_order_acquire) != expected) {
      errno = EWOULDBLOCK;
      return -1;
    }
    list_insert(&b->head, &w);
  }
  for (;;) {
    if (w.sig.load(std::memory_order_acquire) != 0) break;
    struct timespec rel;
    struct timespec* prel = nullptr;
    if (abstime_us != nullptr) {
      int64_t now = monotonic_time_us();
      int64_t left = *abstime_us - now;
      if (left <= 0) {
        // Timed out: remove ourselves unless a waker got there first.
        bool removed = false;
        {
          std::lock_guard<std::mutex> lk(b->mu);
          if (in_list(&w)) {
            list_remove(&w);
            removed = true;
          }
        }
        if (removed) {
          g_butex_wait_us.fetch_add(monotonic_time_us() - park_t0, std::memory_order_relaxed);
          errno = ETIMEDOUT;
          return -1;
        }
        // Woken concurrently; wait for the sig store to land.
        while (w.sig.load(std::memory_order_acquire) == 0) sched_yield();
        break;
      }
      rel.tv_sec = left / 1000000;
      rel.tv_nsec = (left % 1000000) * 1000;
      prel = &rel;
    }
    sys_futex(&w.sig, FUTEX_WAIT, 0, prel);
  }
  g_butex_wait_us.fetch_add(monotonic_time_us() - park_t0, std::memory_order_relaxed);
  if (w.state.load(std::memory_order_acquire) == 2) {
    errno = ETIMEDOUT;
    return -1;
  }
  return 0;
}

static int wake_some(std::atomic<int>* v, int limit) {
  Butex* b = container_of_value(v);
  ButexWaiter* local[64];
  int n = 0;
  {
    std::lock_guard<std::mutex> lk(b->mu);
    while (n < limit && n < 64 && b->head.next != &b->head) {
      ButexWaiter* w = b->head.next;
      list_remove(w);
      local[n++] = w;
    }
  }
  for (int i = 0; i < n; ++i) {
    if (local[i]->timer_id != 0) timer_delete(local[i]->timer_id);
    wake_one_waiter(local[i], 1 /*woken*/);
  }
  return n;
}

int butex_wake(std::atomic<int>* v) { return wake_some(v, 1); }

int butex_wake_all(std::atomic<int>* v) {
  int total = 0;
  for (;;) {
    int n = wake_some(v, 64);
    total += n;
    if (n < 64) break;
  }
  return total;
}

}  // namespace bam
