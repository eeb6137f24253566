// brpc_amd: public fiber API — M:N user-space threads.
// Capability parity with reference bthread/bthread.h (start_urgent /
// start_background / join / yield / usleep / self), rebuilt for the MI355X
// runtime: the same scheduler that runs RPC fibers also integrates HIP
// stream/event waits (a fiber blocking on a GPU op parks on a butex that a
// hipEvent poller wakes — see hip/gpu_api.h + fiber/gpu_wait.h).
#pragma once

#include <stdint.h>

namespace bam {

typedef uint64_t fiber_t;  // versioned id; 0 = invalid

struct FiberAttr {
  uint32_t stack_size = 0;  // 0 = default (256 KiB)
};

// Stack size classes (parity: reference BTHREAD_ATTR_SMALL/NORMAL/LARGE).
// Requests are rounded up to a pooled class {32K, 256K, 2M}.
constexpr FiberAttr FIBER_ATTR_SMALL{32 * 1024};
constexpr FiberAttr FIBER_ATTR_NORMAL{256 * 1024};
constexpr FiberAttr FIBER_ATTR_LARGE{2 * 1024 * 1024};

// Starts a fiber. *tid receives its id. "urgent" runs the new fiber
// immediately on the calling worker (the caller is re-queued) — the
// latency trick used on the RPC dispatch path; "background" enqueues.
int fiber_start_urgent(fiber_t* tid, void (*fn)(void*), void* arg,
                       const FiberAttr* attr = nullptr);
int fiber_start_background(fiber_t* tid, void (*fn)(void*), void* arg,
                           const FiberAttr* attr = nullptr);

// Waits for fiber termination. Returns 0; joining an ended/invalid id
// returns 0 immediately.
// errno value a stopped fiber's sleeps return (≙ reference bthread ESTOP).
enum { ESTOP = 2017 };

int fiber_join(fiber_t tid);
// Wakes `tid` out of a fiber_usleep early: the sleep returns -1 with
// errno=EINTR (≙ reference bthread_interrupt). No-op on finished ids.
int fiber_interrupt(fiber_t tid);
// Marks `tid` stopped and interrupts it: current and FUTURE fiber_usleep
// calls return -1/ESTOP immediately (≙ bthread_stop). The fiber observes
// it via the sleep result (or fiber_stop_requested()) and exits itself.
int fiber_stop(fiber_t tid);
// True if fiber_stop was called on the CALLING fiber.
bool fiber_stop_requested();
// True if the fiber still runs (diagnostics).
bool fiber_exists(fiber_t tid);

int fiber_yield();
int fiber_usleep(uint64_t us);

// Waits for readability(POLLIN)/writability(POLLOUT) of fd without
// blocking the worker pthread (parity: reference bthread_fd_wait).
// Polls non-blockingly and parks the fiber between checks (0.5 ms
// granularity — the RPC runtime's own fds use edge-triggered epoll; this
// API is for user fds). Returns 0 when ready, -1 on timeout/error.
int fiber_fd_wait(int fd, short events, int timeout_ms);
fiber_t fiber_self();
bool is_running_on_fiber();

// Worker pool control.
void fiber_set_concurrency(int n);  // only effective before first start
int fiber_get_concurrency();
// #fibers created / active (diagnostics, /status page).
int64_t fiber_count_created();
int64_t fiber_count_active();

}  // namespace bam
