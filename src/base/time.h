// brpc_amd: time helpers (parity with reference butil/time.h).
#pragma once

#include <stdint.h>
#include <time.h>

namespace bam {

inline int64_t monotonic_time_ns() {
  struct timespec ts;
  clock_gettime(CLOCK_MONOTONIC, &ts);
  return (int64_t)ts.tv_sec * 1000000000LL + ts.tv_nsec;
}

inline int64_t monotonic_time_us() { return monotonic_time_ns() / 1000; }
inline int64_t monotonic_time_ms() { return monotonic_time_ns() / 1000000; }

inline int64_t gettimeofday_us() {
  struct timespec ts;
  clock_gettime(CLOCK_REALTIME, &ts);
  return (int64_t)ts.tv_sec * 1000000LL + ts.tv_nsec / 1000;
}

// A stopwatch for benchmark sections.
class Timer {
 public:
  Timer() : start_(0), stop_(0) {}
  void start() { start_ = monotonic_time_ns(); }
  void stop() { stop_ = monotonic_time_ns(); }
  int64_t n_elapsed() const { return stop_ - start_; }
  int64_t u_elapsed() const { return n_elapsed() / 1000; }
  int64_t m_elapsed() const { return n_elapsed() / 1000000; }

 private:
  int64_t start_;
  int64_t stop_;
};

}  // namespace bam
