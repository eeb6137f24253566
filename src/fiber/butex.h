// brpc_amd: butex — the fiber-aware futex.
// Parity: reference bthread/butex.h. A 32-bit word; butex_wait parks the
// calling fiber (or pthread) while the word equals `expected`; butex_wake
// requeues a waiter onto a worker run queue. Foundation for fiber mutex /
// condition / join / RPC completion, and for GPU-event waits (the hipEvent
// poller wakes a butex when a stream op completes).
#pragma once

#include <time.h>

#include <atomic>

namespace bam {

// Returns a pointer to the butex's 32-bit value word.
std::atomic<int>* butex_create();
void butex_destroy(std::atomic<int>* b);

// Returns 0 on wake; -1 with errno = EWOULDBLOCK if *b != expected at entry,
// ETIMEDOUT if abstime (CLOCK_MONOTONIC us) passed.
int butex_wait(std::atomic<int>* b, int expected, const int64_t* abstime_us = nullptr);

// Contention totals (fiber parks + parked microseconds).
int64_t butex_total_waits();
int64_t butex_total_wait_us();

// Sampled contention SITES (parity: the reference contention profiler fed
// by instrumented bthread_mutex): 1 of every 64 parks records the caller
// backtrace + parked time. Renders at /contention.
struct ContentionSample {
  void* frames[4];
  int nframes;
  int64_t wait_us;
};
// Copies up to `max` recent samples into out; returns the count copied.
size_t butex_contention_samples(ContentionSample* out, size_t max);

int butex_wake(std::atomic<int>* b);      // wake one; returns #woken
int butex_wake_all(std::atomic<int>* b);  // returns #woken

}  // namespace bam
