// C++-side fiber test scenarios, driven from pytest through the bindings.
// (Fibers never execute Python code — the GIL and stack switching do not
// mix — so the scenario bodies live here.)
#include <errno.h>

#include <atomic>
#include <mutex>
#include <string>
#include <vector>

#include "base/fast_rand.h"
#include "base/time.h"
#include "fiber/butex.h"
#include "fiber/execution_queue.h"
#include "fiber/fiber.h"
#include "fiber/sync.h"
#include "fiber/timer_thread.h"

namespace bam {
namespace selftest {

// ---- start/join ----
namespace {
struct AddArg {
  std::atomic<int64_t>* counter;
  int iters;
};
void add_fn(void* raw) {
  AddArg* a = (AddArg*)raw;
  for (int i = 0; i < a->iters; ++i) {
    a->counter->fetch_add(1, std::memory_order_relaxed);
    if ((i & 63) == 0) fiber_yield();
  }
}
}  // namespace

int64_t start_join_test(int nfibers, int iters) {
  std::atomic<int64_t> counter{0};
  AddArg arg{&counter, iters};
  std::vector<fiber_t> tids(nfibers);
  for (int i = 0; i < nfibers; ++i) {
    if (fiber_start_background(&tids[i], add_fn, &arg) != 0) return -1;
  }
  for (int i = 0; i < nfibers; ++i) fiber_join(tids[i]);
  return counter.load();
}

// ---- urgent start preempts ----
namespace {
std::atomic<int> g_order_token{0};
void urgent_child(void*) { g_order_token.store(1, std::memory_order_release); }
struct UrgentArg {
  bool child_ran_first;
};
void urgent_parent(void* raw) {
  UrgentArg* a = (UrgentArg*)raw;
  g_order_token.store(0, std::memory_order_release);
  fiber_t t;
  fiber_start_urgent(&t, urgent_child, nullptr);
  // With urgent start from a worker, the child runs before we resume.
  a->child_ran_first = g_order_token.load(std::memory_order_acquire) == 1;
  fiber_join(t);
}
}  // namespace

bool urgent_test() {
  UrgentArg a{false};
  fiber_t t;
  fiber_start_background(&t, urgent_parent, &a);
  fiber_join(t);
  return a.child_ran_first;
}

// ---- semaphore + rwlock ----
namespace {
struct SemArg {
  FiberSemaphore* sem;
  std::atomic<int>* acquired;
};
void sem_fn(void* raw) {
  SemArg* a = (SemArg*)raw;
  a->sem->acquire();
  a->acquired->fetch_add(1);
}
struct RwArg {
  FiberRWLock* rw;
  std::atomic<int>* readers_in;
  std::atomic<int>* max_readers;
  std::atomic<long long>* counter;
  int iters;
  bool writer;
};
void rw_fn(void* raw) {
  RwArg* a = (RwArg*)raw;
  for (int i = 0; i < a->iters; ++i) {
    if (a->writer) {
      a->rw->wrlock();
      long long v = a->counter->load(std::memory_order_relaxed);
      a->counter->store(v + 1, std::memory_order_relaxed);
      if (a->readers_in->load() != 0) a->max_readers->store(-1);  // reader inside write CS!
      a->rw->unlock();
    } else {
      a->rw->rdlock();
      int in = a->readers_in->fetch_add(1) + 1;
      int m = a->max_readers->load();
      while (m >= 0 && in > m && !a->max_readers->compare_exchange_weak(m, in)) {
      }
      a->readers_in->fetch_sub(1);
      a->rw->unlock();
    }
  }
}
}  // namespace

bool semaphore_test() {
  FiberSemaphore sem(0);
  std::atomic<int> acquired{0};
  fiber_t t[4];
  SemArg a{&sem, &acquired};
  for (int i = 0; i < 4; ++i) fiber_start_background(&t[i], sem_fn, &a);
  fiber_usleep(20000);
  if (acquired.load() != 0) return false;  // nothing released yet
  sem.release(2);
  fiber_usleep(50000);
  if (acquired.load() != 2) return false;
  sem.release(2);
  for (int i = 0; i < 4; ++i) fiber_join(t[i]);
  return acquired.load() == 4 && !sem.try_acquire();
}

// Returns 0 = ok, 1 = correctness held but no reader overlap observed
// (benign on a starved box), 2 = CORRUPTION (writer not exclusive /
// reader inside a write section).
int rwlock_test(int nreaders, int nwriters, int iters) {
  FiberRWLock rw;
  std::atomic<int> readers_in{0}, max_readers{0};
  std::atomic<long long> counter{0};
  std::vector<fiber_t> tids;
  std::vector<RwArg> args(nreaders + nwriters);
  for (int i = 0; i < nreaders + nwriters; ++i) {
    args[i] = RwArg{&rw, &readers_in, &max_readers, &counter, iters, i < nwriters};
    fiber_t t;
    fiber_start_background(&t, rw_fn, &args[i]);
    tids.push_back(t);
  }
  for (fiber_t t : tids) fiber_join(t);
  // writers mutually exclusive => counter exact (max_readers == -1 flags
  // a reader observed inside a write section)
  if (counter.load() != (long long)nwriters * iters || max_readers.load() < 0) return 2;
  return max_readers.load() > 1 ? 0 : 1;
}

// ---- usleep accuracy ----
namespace {
struct SleepArg {
  int64_t us;
  int64_t measured;
};
void sleep_fn(void* raw) {
  SleepArg* a = (SleepArg*)raw;
  int64_t t0 = monotonic_time_us();
  fiber_usleep(a->us);
  a->measured = monotonic_time_us() - t0;
}
}  // namespace

int64_t usleep_test(int64_t us) {
  SleepArg a{us, 0};
  fiber_t t;
  fiber_start_background(&t, sleep_fn, &a);
  fiber_join(t);
  return a.measured;
}

// ---- butex wake/wait + timeout ----
namespace {
struct ButexArg {
  std::atomic<int>* b;
  int rc;
  int saved_errno;
  int64_t waited_us;
  int64_t timeout_us;  // 0 = none
};
void butex_wait_fn(void* raw) {
  ButexArg* a = (ButexArg*)raw;
  int64_t t0 = monotonic_time_us();
  if (a->timeout_us > 0) {
    int64_t abst = t0 + a->timeout_us;
    a->rc = butex_wait(a->b, 0, &abst);
  } else {
    a->rc = butex_wait(a->b, 0, nullptr);
  }
  a->saved_errno = errno;
  a->waited_us = monotonic_time_us() - t0;
}
}  // namespace

bool butex_wake_test() {
  std::atomic<int>* b = butex_create();
  b->store(0);
  ButexArg a{b, -99, 0, 0, 0};
  fiber_t t;
  fiber_start_background(&t, butex_wait_fn, &a);
  usleep(30000);  // let it park
  b->store(1, std::memory_order_release);
  butex_wake_all(b);
  fiber_join(t);
  butex_destroy(b);
  return a.rc == 0 && a.waited_us >= 20000;
}

bool butex_timeout_test() {
  std::atomic<int>* b = butex_create();
  b->store(0);
  ButexArg a{b, -99, 0, 0, 50000 /*50ms*/};
  fiber_t t;
  fiber_start_background(&t, butex_wait_fn, &a);
  fiber_join(t);
  // Upper bound generous: the claim is "timed out (not hung)"; a loaded
  // box (or the ASan build) can oversleep a 50 ms wait by a lot.
  bool ok = a.rc == -1 && a.saved_errno == ETIMEDOUT && a.waited_us >= 45000 &&
            a.waited_us < 5000000;
  butex_destroy(b);
  return ok;
}

// ---- mutex + condition stress ----
namespace {
struct MutexArg {
  FiberMutex* mu;
  int64_t* shared;  // unsynchronized; mutex must protect it
  int iters;
};
void mutex_fn(void* raw) {
  MutexArg* a = (MutexArg*)raw;
  for (int i = 0; i < a->iters; ++i) {
    FiberMutexGuard g(*a->mu);
    int64_t v = *a->shared;
    if ((i & 15) == 0) fiber_yield();  // force contention across workers
    *a->shared = v + 1;
  }
}
}  // namespace

int64_t mutex_test(int nfibers, int iters) {
  FiberMutex mu;
  int64_t shared = 0;
  MutexArg a{&mu, &shared, iters};
  std::vector<fiber_t> tids(nfibers);
  for (int i = 0; i < nfibers; ++i) fiber_start_background(&tids[i], mutex_fn, &a);
  for (int i = 0; i < nfibers; ++i) fiber_join(tids[i]);
  return shared;
}

// ---- countdown event ----
namespace {
struct CdArg {
  CountdownEvent* ev;
};
void cd_fn(void* raw) {
  CdArg* a = (CdArg*)raw;
  fiber_usleep(1000 + fast_rand_less_than(5000));
  a->ev->signal();
}
}  // namespace

bool countdown_test(int n) {
  CountdownEvent ev(n);
  CdArg a{&ev};
  for (int i = 0; i < n; ++i) {
    fiber_t t;
    fiber_start_background(&t, cd_fn, &a);
  }
  return ev.timed_wait(monotonic_time_us() + 2000000);
}

// ---- timer thread ----
namespace {
void timer_note(void* a, void*) { ((std::atomic<int>*)a)->fetch_add(1); }
}  // namespace

bool timer_test() {
  std::atomic<int> fired{0};
  int64_t now = monotonic_time_us();
  TimerId t1 = timer_add(now + 20000, timer_note, &fired, nullptr);
  TimerId t2 = timer_add(now + 30000, timer_note, &fired, nullptr);
  (void)t1;
  if (timer_delete(t2) != 0) return false;  // cancel before run
  usleep(100000);
  if (fired.load() != 1) return false;
  if (timer_delete(t1) != -1) return false;  // already ran
  return true;
}

}  // namespace selftest
}  // namespace bam

// ---- fiber-local storage ----

#include "fiber/key.h"

namespace bam {
namespace selftest {

namespace {
std::atomic<int> g_dtor_runs{0};
void key_dtor(void* p) {
  g_dtor_runs.fetch_add(1);
  delete (int*)p;
}
struct KeyArg {
  fiber_key_t key;
  bool ok;
};
void key_fiber(void* raw) {
  KeyArg* a = (KeyArg*)raw;
  a->ok = fiber_getspecific(a->key) == nullptr;  // fresh fiber: empty
  int* v = new int(42);
  fiber_setspecific(a->key, v);
  fiber_yield();
  a->ok = a->ok && fiber_getspecific(a->key) == v;  // survives reschedule
}
}  // namespace

bool fiber_key_test() {
  fiber_key_t key;
  if (fiber_key_create(&key, key_dtor) != 0) return false;
  g_dtor_runs.store(0);
  KeyArg args[8];
  fiber_t tids[8];
  for (int i = 0; i < 8; ++i) {
    args[i].key = key;
    args[i].ok = false;
    fiber_start_background(&tids[i], key_fiber, &args[i]);
  }
  for (int i = 0; i < 8; ++i) fiber_join(tids[i]);
  bool all_ok = true;
  for (int i = 0; i < 8; ++i) all_ok = all_ok && args[i].ok;
  // destructors ran at each fiber's exit
  all_ok = all_ok && g_dtor_runs.load() == 8;
  fiber_key_delete(key);
  return all_ok;
}

}  // namespace selftest
}  // namespace bam

// ---- GPU-wait park/wake (fiber/gpu_wait.h), GPU-free ----
// Proves the north-star scheduler property: a fiber blocked on a GPU
// ticket PARKS (releases its worker) and a wake callback — in production
// fired by a hipLaunchHostFunc marker — resumes it. Here the "GPU" is a
// plain thread flipping the pinned-flag stand-in.

#include <thread>

#include "fiber/gpu_wait.h"

namespace bam {
namespace selftest {

namespace {
constexpr int kFakeKind = 7;  // unused by the real HIP lib (kinds 0-6)
std::atomic<unsigned long long> g_fake_flag{0};
std::atomic<int> g_fake_requests{0};
int fake_request_wake(int /*dev*/, int /*kind*/) {
  g_fake_requests.fetch_add(1);
  return 0;  // a wake marker is "enqueued"; the test thread fires it
}
struct GpuWaitArg {
  std::atomic<int>* done;
};
void gpu_waiter_fiber(void* raw) {
  GpuWaitArg* a = (GpuWaitArg*)raw;
  gpu_fiber_wait_u64((const volatile unsigned long long*)&g_fake_flag, 1, 0, kFakeKind);
  a->done->fetch_add(1);
}
void control_fiber(void* raw) { ((std::atomic<int>*)raw)->store(1); }
}  // namespace

bool gpu_wait_selftest() {
  auto* saved = gpu_wait_get_request_fn();
  gpu_wait_set_request_fn(fake_request_wake);
  g_fake_flag.store(0);
  g_fake_requests.store(0);
  const int64_t parks_before = gpu_wait_parks();

  // More waiters than workers: if waits consumed worker pthreads, the
  // control fiber below could never run.
  const int nwaiters = fiber_get_concurrency() * 2 + 4;
  std::vector<fiber_t> tids(nwaiters);
  std::atomic<int> ndone{0};
  GpuWaitArg arg{&ndone};
  for (int i = 0; i < nwaiters; ++i)
    fiber_start_background(&tids[i], gpu_waiter_fiber, &arg);
  usleep(20000);  // let every waiter reach the park
  bool ok = ndone.load() == 0;  // nobody released yet

  std::atomic<int> control{0};
  fiber_t ct;
  fiber_start_background(&ct, control_fiber, &control);
  const int64_t t0 = monotonic_time_us();
  fiber_join(ct);
  // Parked waiters must have yielded their workers: the control fiber runs
  // promptly even though every worker had a "blocked" fiber.
  ok = ok && (monotonic_time_us() - t0) < 1000000 && control.load() == 1;

  // Fire the "GPU completion": publish the ticket, run the wake callback
  // exactly as the HIP host-func thread would.
  g_fake_flag.store(1, std::memory_order_release);
  std::thread([] { gpu_fiber_wake(0, kFakeKind); }).join();
  for (int i = 0; i < nwaiters; ++i) fiber_join(tids[i]);
  ok = ok && ndone.load() == nwaiters;
  ok = ok && gpu_wait_parks() > parks_before;  // they parked, not spun
  ok = ok && g_fake_requests.load() > 0;       // wake markers were requested

  gpu_wait_set_request_fn(saved);
  return ok;
}

}  // namespace selftest
}  // namespace bam

// ---- epoll-integrated fiber_fd_wait ----

#include <poll.h>
#include <unistd.h>

namespace bam {
namespace selftest {

// Returns wake latency in µs (write happens 20 ms after the wait starts),
// or -1 on failure. Epoll integration should wake within ~a scheduler hop,
// not the old 500 µs poll granularity.
int64_t fd_wait_selftest() {
  int fds[2];
  if (pipe(fds) != 0) return -1;
  std::atomic<int64_t> woke_at{0};
  std::atomic<int> rc{-2};
  struct Arg {
    int fd;
    std::atomic<int64_t>* woke_at;
    std::atomic<int>* rc;
  } arg{fds[0], &woke_at, &rc};
  fiber_t th;
  fiber_start_background(&th, [](void* raw) {
    Arg* a = (Arg*)raw;
    a->rc->store(fiber_fd_wait(a->fd, POLLIN, 2000));
    a->woke_at->store(monotonic_time_us());
  }, &arg);
  usleep(20000);
  const int64_t wrote_at = monotonic_time_us();
  char c = 'x';
  if (write(fds[1], &c, 1) != 1) return -1;
  fiber_join(th);
  close(fds[0]);
  close(fds[1]);
  if (rc.load() != 0) return -1;
  // timeout path too
  int fds2[2];
  if (pipe(fds2) != 0) return -1;
  int64_t t0 = monotonic_time_us();
  int trc = fiber_fd_wait(fds2[0], POLLIN, 50);
  int64_t waited = monotonic_time_us() - t0;
  close(fds2[0]);
  close(fds2[1]);
  if (trc != -1 || waited < 40000) return -1;
  return woke_at.load() - wrote_at;
}

}  // namespace selftest
}  // namespace bam

// ---- stack size classes ----

namespace bam {
namespace selftest {

namespace {
void deep_user(void* raw) {
  // touch ~20 KiB of stack: fits SMALL (32 KiB) but proves isolation
  volatile char pad[20 * 1024];
  pad[0] = 1;
  pad[sizeof(pad) - 1] = 2;
  *(std::atomic<int>*)raw += pad[0] + pad[sizeof(pad) - 1];
}
}  // namespace

bool stack_class_selftest() {
  std::atomic<int> acc{0};
  const FiberAttr attrs[] = {FIBER_ATTR_SMALL, FIBER_ATTR_NORMAL, FIBER_ATTR_LARGE, {}};
  std::vector<fiber_t> tids;
  for (const FiberAttr& a : attrs) {
    for (int i = 0; i < 8; ++i) {
      fiber_t t;
      if (fiber_start_background(&t, deep_user, &acc, &a) != 0) return false;
      tids.push_back(t);
    }
  }
  for (fiber_t t : tids) fiber_join(t);
  return acc.load() == (int)(4 * 8 * 3);
}


// fiber_interrupt / fiber_stop semantics (≙ reference
// bthread_interrupt/bthread_stop unittests in test/bthread_unittest.cpp):
// (1) interrupt cuts a long sleep short with EINTR; (2) stop makes the
// current AND subsequent sleeps return ESTOP immediately; (3) a full
// sleep after an uninterrupted one still times out normally (the
// persistent per-slot sleep butex must not leak wakes across sleeps).
bool fiber_interrupt_test(std::string* err) {
  struct State {
    std::atomic<int> phase{0};
    std::atomic<int64_t> t_interrupted{0};
    std::atomic<int> rc1{0}, e1{0}, rc2{0}, e2{0}, rc3{0}, e3{0};
  } st;
  auto body = [](void* raw) {
    State* s = (State*)raw;
    int64_t t0 = monotonic_time_us();
    int rc = fiber_usleep(5 * 1000 * 1000);  // interrupted early
    s->rc1.store(rc);
    s->e1.store(rc == -1 ? errno : 0);
    s->t_interrupted.store(monotonic_time_us() - t0);
    s->phase.store(1);
    rc = fiber_usleep(20 * 1000);  // normal full sleep, must NOT be cut
    s->rc2.store(rc);
    s->e2.store(rc == -1 ? errno : 0);
    s->phase.store(2);
    while (s->phase.load() != 3) fiber_yield();  // wait for stop request
    rc = fiber_usleep(5 * 1000 * 1000);  // stopped: immediate ESTOP
    s->rc3.store(rc);
    s->e3.store(rc == -1 ? errno : 0);
  };
  fiber_t th;
  if (fiber_start_background(&th, body, &st) != 0) {
    *err = "start failed";
    return false;
  }
  usleep(50 * 1000);  // let it enter the long sleep
  if (fiber_interrupt(th) != 0) {
    *err = "interrupt failed";
    return false;
  }
  int64_t give_up = monotonic_time_us() + 3 * 1000 * 1000;
  while (st.phase.load() < 2 && monotonic_time_us() < give_up) usleep(1000);
  if (st.phase.load() < 2) {
    *err = "interrupted sleep did not finish";
    return false;
  }
  if (st.rc1.load() != -1 || st.e1.load() != EINTR) {
    *err = "sleep 1: want -1/EINTR, got " + std::to_string(st.rc1.load()) + "/" +
           std::to_string(st.e1.load());
    return false;
  }
  if (st.t_interrupted.load() > 2 * 1000 * 1000) {
    *err = "interrupt took too long";
    return false;
  }
  if (st.rc2.load() != 0) {
    *err = "sleep 2 (uninterrupted) failed: errno " + std::to_string(st.e2.load());
    return false;
  }
  if (fiber_stop(th) != 0) {
    *err = "stop failed";
    return false;
  }
  st.phase.store(3);
  if (fiber_join(th) != 0) {
    *err = "join failed";
    return false;
  }
  if (st.rc3.load() != -1 || st.e3.load() != ESTOP) {
    *err = "sleep 3: want -1/ESTOP, got " + std::to_string(st.rc3.load()) + "/" +
           std::to_string(st.e3.load());
    return false;
  }
  return true;
}


// ExecutionQueue urgent lane (≙ reference TASK_OPTIONS_URGENT): urgent
// tasks submitted while the consumer is busy run before earlier normal
// tasks; order within each lane is preserved.
bool execution_queue_urgent_test(std::string* err) {
  ExecutionQueue<int> q;
  std::vector<int> order;
  std::mutex mu;
  std::atomic<bool> gate{false};
  q.start([&](std::vector<int>& batch) {
    // first batch blocks until everything is queued so later submissions
    // land in the lanes, not in this batch
    if (!gate.exchange(true)) {
      while (gate.load()) {
        if (gate.load(std::memory_order_acquire) && order.empty()) {
          // wait for release marker (order gets -1 pushed by the test)
          std::lock_guard<std::mutex> lk(mu);
          if (!order.empty()) break;
        }
        fiber_yield();
        std::lock_guard<std::mutex> lk(mu);
        if (!order.empty()) break;
      }
    }
    std::lock_guard<std::mutex> lk(mu);
    for (int v : batch) order.push_back(v);
  });
  q.execute(100);  // wakes the consumer; blocks in the gate
  usleep(20 * 1000);
  q.execute(1);
  q.execute(2);
  q.execute_urgent(91);
  q.execute_urgent(92);
  {
    std::lock_guard<std::mutex> lk(mu);
    order.push_back(-1);  // release the gate
  }
  q.stop();
  q.join();
  std::lock_guard<std::mutex> lk(mu);
  // expected: -1 (marker), 100, then urgent 91,92 BEFORE normal 1,2
  std::vector<int> want = {-1, 100, 91, 92, 1, 2};
  if (order != want) {
    std::string got;
    for (int v : order) got += std::to_string(v) + ",";
    *err = "order " + got;
    return false;
  }
  return true;
}

}  // namespace selftest
}  // namespace bam
