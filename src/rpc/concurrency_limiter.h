// brpc_amd: adaptive concurrency limiting.
// Parity: reference brpc/policy/auto_concurrency_limiter.{h,cpp},
// policy/timeout_concurrency_limiter.cpp, adaptive_max_concurrency.h
// (clean-room; the algorithm family is documented publicly in
// docs/en/auto_concurrency_limiter.md):
//  * "constant": plain max_concurrency gate (ServerOptions::max_concurrency)
//  * "auto": gradient limiter — tracks no-load latency (min_latency) by
//    periodically shrinking the limit to drain queues, and peak QPS; sets
//    limit ≈ peak_qps * min_latency * (1 + alpha)
//  * "timeout:<ms>": rejects a request when the estimated queueing delay
//    (concurrency * avg_latency) would exceed the latency budget
// Configure via ServerOptions::adaptive_max_concurrency = "" | "auto" |
// "timeout:20" | "<number>".
#pragma once

#include <stdint.h>

#include <atomic>
#include <mutex>
#include <string>

namespace bam {

class ConcurrencyLimiter {
 public:
  virtual ~ConcurrencyLimiter() = default;
  // Called before dispatch with the would-be concurrency (inflight+1).
  // false => reject with ELIMIT.
  virtual bool OnRequest(int32_t current_concurrency) = 0;
  // Called as each response completes.
  virtual void OnResponse(int error_code, int64_t latency_us) = 0;
  virtual int32_t MaxConcurrency() const = 0;

  // Parses an adaptive_max_concurrency spec; nullptr for "" / "constant"
  // (use the plain max_concurrency gate).
  static ConcurrencyLimiter* Create(const std::string& spec);
};

class AutoConcurrencyLimiter : public ConcurrencyLimiter {
 public:
  AutoConcurrencyLimiter();
  bool OnRequest(int32_t current_concurrency) override;
  void OnResponse(int error_code, int64_t latency_us) override;
  int32_t MaxConcurrency() const override { return max_concurrency_.load(); }

 private:
  void reset_window_locked(int64_t now_us);

  std::atomic<int32_t> max_concurrency_;
  std::mutex mu_;
  // sampling window
  int64_t window_start_us_ = 0;
  int64_t total_latency_us_ = 0;
  int64_t succ_count_ = 0;
  int64_t fail_count_ = 0;
  // learned state
  double min_latency_us_ = -1;  // EMA of no-load latency
  double peak_qps_ = 0;
  int64_t remeasure_after_us_ = 0;  // next min-latency probe
  bool in_probe_ = false;
  int32_t saved_limit_ = 0;
};

class TimeoutConcurrencyLimiter : public ConcurrencyLimiter {
 public:
  explicit TimeoutConcurrencyLimiter(int64_t budget_ms);
  bool OnRequest(int32_t current_concurrency) override;
  void OnResponse(int error_code, int64_t latency_us) override;
  int32_t MaxConcurrency() const override;

 private:
  const int64_t budget_us_;
  std::atomic<int64_t> avg_latency_us_;  // EMA
};

}  // namespace bam
