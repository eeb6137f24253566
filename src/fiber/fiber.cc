#include "fiber/fiber.h"

#include <errno.h>
#include <poll.h>
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

int fiber_usleep(uint64_t us) {
  TaskGroup* g = current_task_group();
  if (g == nullptr || g->cur() == nullptr || g->cur()->is_main) {
    usleep(us);
    return 0;
  }
  std::atomic<int>* word = butex_create();
  word->store(0, std::memory_order_relaxed);
  timer_add(monotonic_time_us() + (int64_t)us, sleep_wake_cb, word, nullptr);
  int v;
  while ((v = word->load(std::memory_order_acquire)) < 1) {
    butex_wait(word, v, nullptr);
  }
  // The callback may still be inside butex_wake_all; it stores 2 when it is
  // completely done touching the butex — only then is destroy safe.
  while (word->load(std::memory_order_acquire) != 2) sched_yield();
  butex_destroy(word);
  return 0;
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



int fiber_fd_wait(int fd, short events, int timeout_ms) {
  const int64_t deadline = monotonic_time_us() + (int64_t)timeout_ms * 1000;
  struct pollfd pfd;
  for (;;) {
    pfd.fd = fd;
    pfd.events = events;
    pfd.revents = 0;
    int rc = ::poll(&pfd, 1, 0);
    if (rc > 0) return (pfd.revents & (POLLERR | POLLNVAL)) ? -1 : 0;
    if (rc < 0 && errno != EINTR) return -1;
    if (timeout_ms >= 0 && monotonic_time_us() >= deadline) {
      errno = ETIMEDOUT;
      return -1;
    }
    fiber_usleep(500);
  }
}

}  // namespace bam
