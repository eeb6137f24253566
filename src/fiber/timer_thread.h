// brpc_amd: TimerThread — one dedicated pthread + min-heap of timers.
// Parity: reference bthread/timer_thread.h. Backs fiber_usleep, butex
// timed waits, RPC deadlines and backup-request triggers.
#pragma once

#include <stdint.h>

namespace bam {

typedef uint64_t TimerId;  // 0 = invalid

// Schedules fn(a, b) at abstime_us (CLOCK_MONOTONIC microseconds).
// Callbacks run on the timer pthread with no locks held — they must be
// short (typically: wake a butex / queue a fiber).
TimerId timer_add(int64_t abstime_us, void (*fn)(void*, void*), void* a, void* b);

// Returns 0 if the timer was cancelled before running, -1 if it already ran
// or is running.
int timer_delete(TimerId id);

}  // namespace bam
