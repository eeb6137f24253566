// brpc_amd: adaptive concurrency limiters (see concurrency_limiter.h).
#include "rpc/concurrency_limiter.h"

#include <stdlib.h>

#include <algorithm>

#include "base/time.h"

namespace bam {

namespace {
constexpr int32_t kInitialLimit = 40;
constexpr int32_t kMinLimit = 4;
constexpr int64_t kWindowUs = 100 * 1000;        // sample window: 100 ms
constexpr int64_t kRemeasureEveryUs = 2 * 1000 * 1000;  // probe min latency every 2 s
constexpr double kAlpha = 0.3;                    // headroom over qps*latency
constexpr double kEma = 0.7;                      // smoothing for learned state
}  // namespace

ConcurrencyLimiter* ConcurrencyLimiter::Create(const std::string& spec) {
  if (spec.empty() || spec == "constant") return nullptr;
  if (spec == "auto") return new AutoConcurrencyLimiter;
  if (spec.rfind("timeout", 0) == 0) {
    int64_t ms = 50;
    size_t colon = spec.find(':');
    if (colon != std::string::npos) ms = atoll(spec.c_str() + colon + 1);
    return new TimeoutConcurrencyLimiter(ms > 0 ? ms : 50);
  }
  // plain number = constant limiter expressed adaptively
  int32_t n = atoi(spec.c_str());
  if (n > 0) {
    class Constant : public ConcurrencyLimiter {
     public:
      explicit Constant(int32_t n) : n_(n) {}
      bool OnRequest(int32_t c) override { return c <= n_; }
      void OnResponse(int, int64_t) override {}
      int32_t MaxConcurrency() const override { return n_; }

     private:
      int32_t n_;
    };
    return new Constant(n);
  }
  return nullptr;
}

AutoConcurrencyLimiter::AutoConcurrencyLimiter() : max_concurrency_(kInitialLimit) {}

bool AutoConcurrencyLimiter::OnRequest(int32_t current) {
  return current <= max_concurrency_.load(std::memory_order_relaxed);
}

void AutoConcurrencyLimiter::reset_window_locked(int64_t now_us) {
  window_start_us_ = now_us;
  total_latency_us_ = 0;
  succ_count_ = 0;
  fail_count_ = 0;
}

void AutoConcurrencyLimiter::OnResponse(int error_code, int64_t latency_us) {
  std::lock_guard<std::mutex> lk(mu_);
  const int64_t now = monotonic_time_us();
  if (window_start_us_ == 0) {
    reset_window_locked(now);
    remeasure_after_us_ = now + kRemeasureEveryUs;
  }
  if (error_code == 0 && latency_us > 0) {
    total_latency_us_ += latency_us;
    ++succ_count_;
  } else {
    ++fail_count_;
  }
  const int64_t elapsed = now - window_start_us_;
  if (elapsed < kWindowUs || succ_count_ < 8) return;

  const double avg_latency = (double)total_latency_us_ / (double)succ_count_;
  const double qps = (double)succ_count_ * 1e6 / (double)elapsed;
  if (min_latency_us_ < 0) {
    min_latency_us_ = avg_latency;
  } else if (in_probe_ || avg_latency < min_latency_us_) {
    // During a probe the queue is drained: trust the sampled latency.
    min_latency_us_ = kEma * min_latency_us_ + (1 - kEma) * avg_latency;
  }
  peak_qps_ = std::max(peak_qps_ * 0.98, qps);  // decaying peak

  int32_t next;
  if (in_probe_) {
    in_probe_ = false;
    next = saved_limit_;  // restore, then recompute below
  } else {
    next = max_concurrency_.load(std::memory_order_relaxed);
  }
  // gradient target: enough concurrency to sustain peak qps at no-load
  // latency, plus alpha headroom
  const double target = peak_qps_ * (min_latency_us_ / 1e6) * (1.0 + kAlpha) + 1;
  next = (int32_t)std::min<double>(std::max<double>(target, kMinLimit), 10000);

  if (now >= remeasure_after_us_) {
    // shrink to drain queues and re-observe the no-load latency
    saved_limit_ = next;
    next = std::max(kMinLimit, next / 2);
    in_probe_ = true;
    remeasure_after_us_ = now + kRemeasureEveryUs;
  }
  max_concurrency_.store(next, std::memory_order_relaxed);
  reset_window_locked(now);
}

TimeoutConcurrencyLimiter::TimeoutConcurrencyLimiter(int64_t budget_ms)
    : budget_us_(budget_ms * 1000), avg_latency_us_(0) {}

bool TimeoutConcurrencyLimiter::OnRequest(int32_t current) {
  if (current <= 1) return true;  // liveness: a lone request always runs
  const int64_t lat = avg_latency_us_.load(std::memory_order_relaxed);
  if (lat <= 0) return true;  // no signal yet
  // estimated time-to-serve for the NEWEST request if admitted
  return (int64_t)current * lat <= budget_us_;
}

void TimeoutConcurrencyLimiter::OnResponse(int error_code, int64_t latency_us) {
  if (error_code != 0 || latency_us <= 0) return;
  int64_t prev = avg_latency_us_.load(std::memory_order_relaxed);
  int64_t next = prev <= 0 ? latency_us : (int64_t)(kEma * prev + (1 - kEma) * latency_us);
  avg_latency_us_.store(next, std::memory_order_relaxed);
}

int32_t TimeoutConcurrencyLimiter::MaxConcurrency() const {
  const int64_t lat = avg_latency_us_.load(std::memory_order_relaxed);
  return lat > 0 ? (int32_t)std::max<int64_t>(1, budget_us_ / lat) : INT32_MAX;
}

}  // namespace bam
