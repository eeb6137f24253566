#include "var/variable.h"

#include <unistd.h>

#include <algorithm>
#include <condition_variable>
#include <set>
#include <thread>

#include "base/time.h"

namespace bam {
namespace var {

namespace detail {
uint64_t next_combiner_id() {
  static std::atomic<uint64_t> id{1};
  return id.fetch_add(1, std::memory_order_relaxed);
}
}  // namespace detail

// ---------------- registry ----------------

namespace {
struct Registry {
  std::mutex mu;
  std::map<std::string, Variable*> vars;
};
Registry& registry() {
  static Registry* r = new Registry;
  return *r;
}
}  // namespace

Variable::~Variable() {}

int Variable::expose(const std::string& name) {
  hide();
  Registry& r = registry();
  std::lock_guard<std::mutex> lk(r.mu);
  name_ = name;
  r.vars[name] = this;
  return 0;
}

void Variable::hide() {
  if (name_.empty()) return;
  Registry& r = registry();
  std::lock_guard<std::mutex> lk(r.mu);
  auto it = r.vars.find(name_);
  if (it != r.vars.end() && it->second == this) r.vars.erase(it);
  name_.clear();
}

void Variable::dump_exposed(std::ostream& os, const std::string& filter) {
  Registry& r = registry();
  std::lock_guard<std::mutex> lk(r.mu);
  for (const auto& kv : r.vars) {
    if (!filter.empty() && kv.first.find(filter) == std::string::npos) continue;
    os << kv.first << " : ";
    kv.second->describe(os);
    os << "\n";
  }
}

Variable* Variable::find_exposed(const std::string& name) {
  Registry& r = registry();
  std::lock_guard<std::mutex> lk(r.mu);
  auto it = r.vars.find(name);
  return it == r.vars.end() ? nullptr : it->second;
}

size_t Variable::count_exposed() {
  Registry& r = registry();
  std::lock_guard<std::mutex> lk(r.mu);
  return r.vars.size();
}

// ---------------- sampler thread ----------------

namespace {
struct Sampler {
  std::mutex mu;
  std::set<WindowedInt*> targets;
  std::set<SamplerTick*> ticks;
  bool started = false;

  void ensure_started() {
    if (started) return;
    started = true;
    std::thread([this] { run(); }).detach();
  }

  void run() {
    for (;;) {
      usleep(1000000);
      std::vector<WindowedInt*> snapshot;
      {
        std::lock_guard<std::mutex> lk(mu);
        snapshot.assign(targets.begin(), targets.end());
      }
      for (WindowedInt* w : snapshot) w->take_sample();
      std::vector<SamplerTick*> tick_snapshot;
      {
        std::lock_guard<std::mutex> lk(mu);
        tick_snapshot.assign(ticks.begin(), ticks.end());
      }
      for (SamplerTick* t : tick_snapshot) t->run();
    }
  }
};
Sampler& sampler() {
  static Sampler* s = new Sampler;
  return *s;
}
}  // namespace

void register_sampler(WindowedInt* w) {
  Sampler& s = sampler();
  std::lock_guard<std::mutex> lk(s.mu);
  s.targets.insert(w);
  s.ensure_started();
}

void unregister_sampler(WindowedInt* w) {
  Sampler& s = sampler();
  std::lock_guard<std::mutex> lk(s.mu);
  s.targets.erase(w);
}

SamplerTick::SamplerTick(std::function<void()> fn) : fn_(std::move(fn)) {
  Sampler& s = sampler();
  std::lock_guard<std::mutex> lk(s.mu);
  s.ticks.insert(this);
  s.ensure_started();
}

SamplerTick::~SamplerTick() {
  Sampler& s = sampler();
  std::lock_guard<std::mutex> lk(s.mu);
  s.ticks.erase(this);
}

// ---------------- WindowedInt ----------------

WindowedInt::WindowedInt(SourceFn src, int window_sec, bool per_second)
    : src_(std::move(src)), window_(window_sec), per_second_(per_second), latest_(0),
      nsamples_(0) {
  for (auto& v : ring_) v = 0;
  register_sampler(this);
}

WindowedInt::~WindowedInt() { unregister_sampler(this); }

void WindowedInt::take_sample() {
  int64_t v = src_();
  std::lock_guard<std::mutex> lk(mu_);
  ring_[nsamples_ % 64] = v;
  latest_ = v;
  ++nsamples_;
}

int64_t WindowedInt::get_value() const {
  std::lock_guard<std::mutex> lk(mu_);
  if (nsamples_ == 0) return 0;
  int w = std::min(window_, nsamples_ - 1);
  if (w <= 0) {
    return per_second_ ? latest_ : latest_;  // best effort before 2 samples
  }
  int64_t old = ring_[(nsamples_ - 1 - w) % 64];
  int64_t diff = latest_ - old;
  return per_second_ ? diff / w : diff;
}

// ---------------- LatencyRecorder ----------------

LatencyRecorder::LatencyRecorder() {
  qps_window_.reset(new WindowedInt([this] { return count_.get_value(); }, 1, true));
  tick_.reset(new SamplerTick([this] { take_hist_snapshot(); }));
}

LatencyRecorder::LatencyRecorder(const std::string& prefix) : LatencyRecorder() {
  expose(prefix);
}

LatencyRecorder::~LatencyRecorder() {
  tick_.reset();  // stop snapshots before members die
  for (Variable* v : exposed_) delete v;
}

void LatencyRecorder::take_hist_snapshot() {
  std::vector<uint32_t> snap(detail::LatencyHistogram::kBuckets);
  hist_.merge(snap.data());
  std::lock_guard<std::mutex> lk(snap_mu_);
  snaps_.push_back(std::move(snap));
  while ((int)snaps_.size() > kWindowSec) snaps_.pop_front();
  // windowed max decays with the same cadence
  window_max_.store(0, std::memory_order_relaxed);
}

LatencyRecorder& LatencyRecorder::operator<<(int64_t latency_us) {
  count_ << 1;
  sum_us_ << latency_us;
  int64_t cur = window_max_.load(std::memory_order_relaxed);
  while (latency_us > cur &&
         !window_max_.compare_exchange_weak(cur, latency_us, std::memory_order_relaxed)) {
  }
  hist_.add(latency_us > 0 ? (uint64_t)latency_us : 0);
  return *this;
}

int64_t LatencyRecorder::qps() const { return qps_window_->get_value(); }

int64_t LatencyRecorder::latency_avg() const {
  int64_t c = count_.get_value();
  return c > 0 ? sum_us_.get_value() / c : 0;
}

int64_t LatencyRecorder::latency_percentile(double p) const {
  constexpr int kB = detail::LatencyHistogram::kBuckets;
  std::vector<uint32_t> now(kB);
  hist_.merge(now.data());
  {
    // Subtract the kWindowSec-old snapshot -> counts for the last ~10 s
    // only. A younger recorder (deque not yet full) uses its lifetime.
    std::lock_guard<std::mutex> lk(snap_mu_);
    if ((int)snaps_.size() >= kWindowSec) {
      const std::vector<uint32_t>& old = snaps_.front();
      for (int i = 0; i < kB; ++i) now[i] -= std::min(now[i], old[i]);
    }
  }
  uint64_t total = 0;
  for (int i = 0; i < kB; ++i) total += now[i];
  if (total == 0) return 0;
  uint64_t rank = (uint64_t)(p * (total - 1));
  uint64_t acc = 0;
  for (int i = 0; i < kB; ++i) {
    acc += now[i];
    if (acc > rank) return detail::LatencyHistogram::value_of(i);
  }
  return detail::LatencyHistogram::value_of(kB - 1);
}

void LatencyRecorder::expose(const std::string& prefix) {
  prefix_ = prefix;
  exposed_.push_back(new PassiveStatus(
      prefix + "_latency", [this] { return std::to_string(latency_avg()); }));
  exposed_.push_back(new PassiveStatus(
      prefix + "_latency_9999", [this] { return std::to_string(latency_percentile(0.9999)); }));
  exposed_.push_back(new PassiveStatus(
      prefix + "_latency_99", [this] { return std::to_string(latency_percentile(0.99)); }));
  exposed_.push_back(new PassiveStatus(
      prefix + "_latency_50", [this] { return std::to_string(latency_percentile(0.50)); }));
  exposed_.push_back(
      new PassiveStatus(prefix + "_qps", [this] { return std::to_string(qps()); }));
  exposed_.push_back(
      new PassiveStatus(prefix + "_count", [this] { return std::to_string(count()); }));
  for (size_t i = 0; i < exposed_.size(); ++i) {
    // names set in PassiveStatus ctor via expose? No: expose explicitly
  }
  // PassiveStatus above were constructed without names; expose them now.
  const char* suffixes[] = {"_latency", "_latency_9999", "_latency_99",
                            "_latency_50", "_qps", "_count"};
  for (size_t i = 0; i < exposed_.size() && i < 6; ++i) {
    exposed_[i]->expose(prefix + suffixes[i]);
  }
}

}  // namespace var
}  // namespace bam
