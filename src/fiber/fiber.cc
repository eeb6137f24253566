#include "fiber/fiber.h"

#include <errno.h>
#include <poll.h>
#include <sys/epoll.h>
#include <vector>
#include <thread>
#include <mutex>
#include <unistd.h>

#include "base/time.h"
#include "fiber/butex.h"
#include "fiber/scheduler.h"
#include "fiber/timer_thread.h"

namespace bam {

namespace {

struct StartUrgentArgs {
  FiberMeta* caller;
};

// Remained for urgent start: requeue the caller (it was preempted).
void remained_requeue_caller(void* raw) {
  FiberMeta* caller = (FiberMeta*)raw;
  TaskControl::singleton()->ready_to_run(fiber_id_of(caller), true);
}

}  // namespace

static int start_fiber(fiber_t* tid, void (*fn)(void*), void* arg, const FiberAttr* attr,
                       bool urgent) {
  TaskControl* c = TaskControl::singleton();
  uint32_t ss = attr != nullptr ? attr->stack_size : 0;
  FiberMeta* m = create_fiber_meta(fn, arg, ss);
  if (m == nullptr) return ENOMEM;
  fiber_t t = fiber_id_of(m);
  if (tid != nullptr) *tid = t;
  c->nfibers_created.fetch_add(1, std::memory_order_relaxed);
  c->nfibers_active.fetch_add(1, std::memory_order_relaxed);
  TaskGroup* g = current_task_group();
  if (urgent && g != nullptr && g->cur() != nullptr && !g->cur()->is_main) {
    // Run the new fiber immediately; requeue the caller.
    g->sched_to(m, remained_requeue_caller, g->cur());
    return 0;
  }
  c->ready_to_run(t, /*prefer_local=*/g != nullptr);
  return 0;
}

int fiber_start_urgent(fiber_t* tid, void (*fn)(void*), void* arg, const FiberAttr* attr) {
  return start_fiber(tid, fn, arg, attr, true);
}

int fiber_start_background(fiber_t* tid, void (*fn)(void*), void* arg, const FiberAttr* attr) {
  return start_fiber(tid, fn, arg, attr, false);
}

int fiber_join(fiber_t tid) {
  if (tid == 0) return 0;
  const uint32_t expected_version = (uint32_t)(tid >> 32);
  for (;;) {
    FiberMeta* m = fiber_meta_of(tid);
    if (m == nullptr) return 0;  // already ended (or id recycled)
    std::atomic<int>* vb = m->version_butex;
    if (vb == nullptr) return 0;
    if ((uint32_t)vb->load(std::memory_order_acquire) != expected_version) return 0;
    butex_wait(vb, (int)expected_version, nullptr);
    // loop: re-check (handles spurious wake / EWOULDBLOCK)
  }
}

bool fiber_exists(fiber_t tid) { return fiber_meta_of(tid) != nullptr; }

int fiber_yield() {
  TaskGroup* g = current_task_group();
  if (g == nullptr || g->cur() == nullptr || g->cur()->is_main) {
    sched_yield();
    return 0;
  }
  // Requeue self and switch to the next ready fiber (or main).
  g->sched(remained_requeue_caller, g->cur());
  return 0;
}

namespace {
void sleep_wake_cb(void* a, void* /*b*/) {
  std::atomic<int>* word = (std::atomic<int>*)a;
  word->store(1, std::memory_order_release);
  butex_wake_all(word);
  word->store(2, std::memory_order_release);  // destroy-safe marker (see below)
}
}  // namespace

namespace {
// Bumps the sequence word and wakes the sleeper. The word is the meta's
// persistent sleep butex, so no destroy hand-off is needed.
void seq_sleep_wake_cb(void* a, void*) {
  std::atomic<int>* word = (std::atomic<int>*)a;
  word->fetch_add(1, std::memory_order_release);
  butex_wake_all(word);
}
}  // namespace

int fiber_usleep(uint64_t us) {
  TaskGroup* g = current_task_group();
  if (g == nullptr || g->cur() == nullptr || g->cur()->is_main) {
    usleep(us);
    return 0;
  }
  FiberMeta* m = g->cur();
  if (m->stop_requested.load(std::memory_order_acquire)) {
    errno = ESTOP;
    return -1;
  }
  std::atomic<int>* word = m->sleep_butex.load(std::memory_order_acquire);
  if (word == nullptr) {
    word = butex_create();
    m->sleep_butex.store(word, std::memory_order_release);
  }
  const int start = word->load(std::memory_order_acquire);
  TimerId tid = timer_add(monotonic_time_us() + (int64_t)us, seq_sleep_wake_cb,
                          word, nullptr);
  while (word->load(std::memory_order_acquire) == start) {
    if (m->interrupted.load(std::memory_order_acquire) ||
        m->stop_requested.load(std::memory_order_acquire)) {
      break;
    }
    butex_wait(word, start, nullptr);
  }
  const bool intr = m->interrupted.exchange(false, std::memory_order_acq_rel);
  const bool stop = m->stop_requested.load(std::memory_order_acquire);
  if (word->load(std::memory_order_acquire) == start) {
    // Timer has not fired: cancel it; if it is already in flight, wait for
    // the bump so it cannot shorten this slot's NEXT sleep.
    if (timer_delete(tid) != 0) {
      while (word->load(std::memory_order_acquire) == start) sched_yield();
    }
  }
  if (stop) {
    errno = ESTOP;
    return -1;
  }
  if (intr) {
    errno = EINTR;
    return -1;
  }
  return 0;
}

namespace {
// Resolves a live meta for tid (version must match). nullptr when ended.
FiberMeta* live_meta_of(fiber_t tid) {
  if (tid == 0) return nullptr;
  FiberMeta* m = fiber_meta_of(tid);
  if (m == nullptr) return nullptr;
  std::atomic<int>* vb = m->version_butex;
  if (vb == nullptr) return nullptr;
  if ((uint32_t)vb->load(std::memory_order_acquire) != (uint32_t)(tid >> 32))
    return nullptr;
  return m;
}
}  // namespace

int fiber_interrupt(fiber_t tid) {
  FiberMeta* m = live_meta_of(tid);
  if (m == nullptr) return EINVAL;
  m->interrupted.store(true, std::memory_order_release);
  std::atomic<int>* word = m->sleep_butex.load(std::memory_order_acquire);
  if (word != nullptr) butex_wake_all(word);
  return 0;
}

int fiber_stop(fiber_t tid) {
  FiberMeta* m = live_meta_of(tid);
  if (m == nullptr) return EINVAL;
  m->stop_requested.store(true, std::memory_order_release);
  std::atomic<int>* word = m->sleep_butex.load(std::memory_order_acquire);
  if (word != nullptr) butex_wake_all(word);
  return 0;
}

bool fiber_stop_requested() {
  TaskGroup* g = current_task_group();
  if (g == nullptr || g->cur() == nullptr || g->cur()->is_main) return false;
  return g->cur()->stop_requested.load(std::memory_order_acquire);
}

fiber_t fiber_self() {
  TaskGroup* g = current_task_group();
  if (g == nullptr || g->cur() == nullptr || g->cur()->is_main) return 0;
  return fiber_id_of(g->cur());
}

bool is_running_on_fiber() { return fiber_self() != 0; }

void fiber_set_concurrency(int n) { TaskControl::set_concurrency_hint(n); }
int fiber_get_concurrency() { return TaskControl::singleton()->concurrency(); }
int64_t fiber_count_created() {
  return TaskControl::singleton()->nfibers_created.load(std::memory_order_relaxed);
}
int64_t fiber_count_active() {
  return TaskControl::singleton()->nfibers_active.load(std::memory_order_relaxed);
}



// Epoll-integrated user-fd wait (parity: reference bthread_fd_wait,
// bthread/fd.cpp — bthread owns its own epoll, independent of the RPC
// dispatcher). A waiter parks on a pooled butex; the fd-wait epoll thread
// wakes it when the fd turns ready. Versioned slots make a late epoll
// event harmless after the waiter left.
namespace {

struct FdWaitSlot {
  std::atomic<int>* butex = nullptr;
  std::atomic<uint32_t> ver{0};
};

constexpr int kFdWaitSlots = 1024;

struct FdWaitLoop {
  int epfd = -1;
  FdWaitSlot slots[kFdWaitSlots];
  std::mutex free_mu;
  std::vector<int> freelist;

  FdWaitLoop() {
    epfd = epoll_create1(EPOLL_CLOEXEC);
    freelist.reserve(kFdWaitSlots);
    for (int i = kFdWaitSlots - 1; i >= 0; --i) {
      slots[i].butex = butex_create();
      freelist.push_back(i);
    }
    std::thread([this] { run(); }).detach();
  }

  void run() {
    struct epoll_event evs[64];
    for (;;) {
      int n = epoll_wait(epfd, evs, 64, -1);
      if (n < 0) {
        if (errno == EINTR) continue;
        return;
      }
      for (int i = 0; i < n; ++i) {
        const uint64_t tag = evs[i].data.u64;
        const int idx = (int)(tag & 0xffffffffu);
        const uint32_t ver = (uint32_t)(tag >> 32);
        if (idx < 0 || idx >= kFdWaitSlots) continue;
        FdWaitSlot& s = slots[idx];
        if (s.ver.load(std::memory_order_acquire) != ver) continue;  // stale
        s.butex->fetch_add(1, std::memory_order_release);
        butex_wake_all(s.butex);
      }
    }
  }

  int acquire() {
    std::lock_guard<std::mutex> lk(free_mu);
    if (freelist.empty()) return -1;
    int idx = freelist.back();
    freelist.pop_back();
    return idx;
  }

  void release(int idx) {
    std::lock_guard<std::mutex> lk(free_mu);
    freelist.push_back(idx);
  }
};

FdWaitLoop& fd_wait_loop() {
  static FdWaitLoop* l = new FdWaitLoop;
  return *l;
}

}  // namespace

int fiber_fd_wait(int fd, short events, int timeout_ms) {
  const int64_t deadline =
      timeout_ms >= 0 ? monotonic_time_us() + (int64_t)timeout_ms * 1000 : 0;
  // Fast path: already ready.
  struct pollfd pfd;
  pfd.fd = fd;
  pfd.events = events;
  pfd.revents = 0;
  int rc = ::poll(&pfd, 1, 0);
  if (rc > 0) return (pfd.revents & POLLNVAL) ? -1 : 0;
  if (rc < 0 && errno != EINTR) return -1;

  FdWaitLoop& loop = fd_wait_loop();
  const int idx = loop.acquire();
  if (idx < 0) {
    // All slots busy (pathological): degrade to coarse polling.
    for (;;) {
      pfd.revents = 0;
      rc = ::poll(&pfd, 1, 20);
      if (rc > 0) return (pfd.revents & POLLNVAL) ? -1 : 0;
      if (rc < 0 && errno != EINTR) return -1;
      if (timeout_ms >= 0 && monotonic_time_us() >= deadline) {
        errno = ETIMEDOUT;
        return -1;
      }
    }
  }
  FdWaitSlot& slot = loop.slots[idx];
  const uint32_t ver = slot.ver.fetch_add(1, std::memory_order_acq_rel) + 1;
  struct epoll_event ev;
  ev.events = EPOLLONESHOT | (events & POLLIN ? EPOLLIN : 0u) |
              (events & POLLOUT ? EPOLLOUT : 0u) | EPOLLRDHUP;
  ev.data.u64 = ((uint64_t)ver << 32) | (uint32_t)idx;
  bool armed = epoll_ctl(loop.epfd, EPOLL_CTL_ADD, fd, &ev) == 0;
  if (!armed && errno == EEXIST) armed = epoll_ctl(loop.epfd, EPOLL_CTL_MOD, fd, &ev) == 0;
  int result = -1;
  if (!armed) {
    errno = EINVAL;
  } else {
    for (;;) {
      const int v = slot.butex->load(std::memory_order_acquire);
      pfd.revents = 0;
      rc = ::poll(&pfd, 1, 0);
      if (rc > 0) {
        result = (pfd.revents & POLLNVAL) ? -1 : 0;
        break;
      }
      if (rc < 0 && errno != EINTR) break;
      if (timeout_ms >= 0 && monotonic_time_us() >= deadline) {
        errno = ETIMEDOUT;
        break;
      }
      int64_t abst = timeout_ms >= 0 ? deadline : monotonic_time_us() + 2000000;
      butex_wait(slot.butex, v, &abst);
      // Spurious wake / re-arm after ONESHOT delivery for the next loop.
      epoll_ctl(loop.epfd, EPOLL_CTL_MOD, fd, &ev);
    }
    epoll_ctl(loop.epfd, EPOLL_CTL_DEL, fd, nullptr);
  }
  slot.ver.fetch_add(1, std::memory_order_release);  // invalidate late events
  loop.release(idx);
  return result;
}

}  // namespace bam
