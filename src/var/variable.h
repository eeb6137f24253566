// brpc_amd: var — lock-cheap metrics (≙ reference bvar L2).
// Variable base + global name registry; Adder/Maxer/Miner reducers with
// per-thread cells (writes are a relaxed atomic op on a thread-local cell,
// no contention); Window/PerSecond via a 1 Hz sampler thread;
// LatencyRecorder = qps + avg + max + percentiles. Parity: bvar/variable.h,
// bvar/reducer.h, bvar/window.h, bvar/latency_recorder.h.
#pragma once

#include <stdint.h>

#include <atomic>
#include <deque>
#include <functional>
#include <limits>
#include <map>
#include <memory>
#include <mutex>
#include <sstream>
#include <string>
#include <type_traits>
#include <vector>

namespace bam {
namespace var {

class Variable {
 public:
  virtual ~Variable();
  virtual void describe(std::ostream& os) const = 0;
  std::string get_description() const {
    std::ostringstream os;
    describe(os);
    return os.str();
  }
  // Registers under `name` (replaces previous owner of the name).
  int expose(const std::string& name);
  void hide();
  const std::string& name() const { return name_; }

  static void dump_exposed(std::ostream& os, const std::string& filter = "");
  static Variable* find_exposed(const std::string& name);
  static size_t count_exposed();

 protected:
  std::string name_;
};

// ---------------- reducers ----------------

namespace detail {

// Unique id per combiner instance: thread-local cell caches are keyed by
// (address, id) so a new combiner reusing a destroyed one's address never
// sees a stale Cell* (ABA across recorder lifetimes).
uint64_t next_combiner_id();

// Per-thread cell list; read = combine over all cells (+ sum of cells from
// dead threads). Writes touch only the caller's cell.
template <typename T, typename Op>
class AgentCombiner {
 public:
  struct Cell {
    std::atomic<T> value{T()};
  };

  T combine(T identity) const {
    T acc = identity;
    std::lock_guard<std::mutex> lk(mu_);
    acc = Op()(acc, terminated_);
    for (const auto& c : cells_) acc = Op()(acc, c->value.load(std::memory_order_relaxed));
    return acc;
  }

  void reset_all(T identity) {
    std::lock_guard<std::mutex> lk(mu_);
    terminated_ = identity;
    for (auto& c : cells_) c->value.store(identity, std::memory_order_relaxed);
  }

  Cell* local_cell() {
    // tls map: combiner address -> (instance id, cell)
    static thread_local std::map<const void*, std::pair<uint64_t, Cell*>> tls;
    auto it = tls.find(this);
    if (it != tls.end() && it->second.first == id_) return it->second.second;
    auto cell = std::make_shared<Cell>();
    {
      std::lock_guard<std::mutex> lk(mu_);
      cells_.push_back(cell);
    }
    tls[this] = {id_, cell.get()};
    return cell.get();
  }

 private:
  mutable std::mutex mu_;
  std::vector<std::shared_ptr<Cell>> cells_;
  T terminated_ = T();
  const uint64_t id_ = next_combiner_id();
};

// Per-thread log-bucketed latency histogram, merged on read. Clean-room
// equivalent of the reference's per-thread percentile sample intervals
// (bvar/detail/percentile.h): writes are ONE relaxed increment in a
// thread-local cell (no contention, no sample-dropping bias — the round-1
// last-8192-samples ring under-weighted bursts); reads merge every cell.
// Buckets: exact for v < 64, then 64 sub-buckets per octave up to 2^40 µs
// (≈1.5 % worst-case value error).
class LatencyHistogram {
 public:
  static constexpr int kSub = 64;
  static constexpr int kMaxExp = 40;
  static constexpr int kBuckets = kSub + (kMaxExp - 6) * kSub;  // 2240

  static int index_of(uint64_t v) {
    if (v < (uint64_t)kSub) return (int)v;
    int hb = 63 - __builtin_clzll(v);  // highest set bit, >= 6
    if (hb >= kMaxExp) {
      hb = kMaxExp - 1;
      v = (1ull << kMaxExp) - 1;
    }
    return kSub + (hb - 6) * kSub + (int)((v >> (hb - 6)) & (kSub - 1));
  }

  static int64_t value_of(int idx) {
    if (idx < kSub) return idx;
    int rel = idx - kSub;
    int hb = 6 + rel / kSub;
    int sub = rel % kSub;
    uint64_t base = (1ull << hb) + ((uint64_t)sub << (hb - 6));
    uint64_t width = 1ull << (hb - 6);
    return (int64_t)(base + width / 2);
  }

  struct Cell {
    std::atomic<uint32_t> counts[kBuckets];
    Cell() {
      for (auto& c : counts) c.store(0, std::memory_order_relaxed);
    }
  };

  void add(uint64_t v) {
    local_cell()->counts[index_of(v)].fetch_add(1, std::memory_order_relaxed);
  }

  // Sums every thread's cell into out[kBuckets].
  void merge(uint32_t* out) const {
    for (int i = 0; i < kBuckets; ++i) out[i] = 0;
    std::lock_guard<std::mutex> lk(mu_);
    for (const auto& c : cells_)
      for (int i = 0; i < kBuckets; ++i)
        out[i] += c->counts[i].load(std::memory_order_relaxed);
  }

 private:
  Cell* local_cell() {
    static thread_local std::map<const void*, std::pair<uint64_t, Cell*>> tls;
    auto it = tls.find(this);
    if (it != tls.end() && it->second.first == id_) return it->second.second;
    auto cell = std::make_shared<Cell>();
    {
      std::lock_guard<std::mutex> lk(mu_);
      cells_.push_back(cell);
    }
    tls[this] = {id_, cell.get()};
    return cell.get();
  }

  mutable std::mutex mu_;
  std::vector<std::shared_ptr<Cell>> cells_;
  const uint64_t id_ = next_combiner_id();
};

struct AddOp {
  template <typename T>
  T operator()(T a, T b) const {
    return a + b;
  }
};
struct MaxOp {
  template <typename T>
  T operator()(T a, T b) const {
    return a > b ? a : b;
  }
};
struct MinOp {
  template <typename T>
  T operator()(T a, T b) const {
    return a < b ? a : b;
  }
};

}  // namespace detail

template <typename T>
class Adder : public Variable {
 public:
  Adder() {}
  explicit Adder(const std::string& name) { expose(name); }
  ~Adder() override { hide(); }

  Adder& operator<<(T v) {
    auto* cell = combiner_.local_cell();
    // fetch_add for integral; CAS loop otherwise
    atomic_add(cell->value, v);
    return *this;
  }
  T get_value() const { return combiner_.combine(T()); }
  void reset() { combiner_.reset_all(T()); }
  void describe(std::ostream& os) const override { os << get_value(); }

 private:
  static void atomic_add(std::atomic<T>& a, T v) {
    if constexpr (std::is_integral<T>::value) {
      a.fetch_add(v, std::memory_order_relaxed);
    } else {
      T cur = a.load(std::memory_order_relaxed);
      while (!a.compare_exchange_weak(cur, cur + v, std::memory_order_relaxed)) {
      }
    }
  }
  detail::AgentCombiner<T, detail::AddOp> combiner_;
};

template <typename T>
class Maxer : public Variable {
 public:
  Maxer() {}
  explicit Maxer(const std::string& name) { expose(name); }
  ~Maxer() override { hide(); }
  Maxer& operator<<(T v) {
    auto* cell = combiner_.local_cell();
    T cur = cell->value.load(std::memory_order_relaxed);
    while (v > cur &&
           !cell->value.compare_exchange_weak(cur, v, std::memory_order_relaxed)) {
    }
    return *this;
  }
  T get_value() const { return combiner_.combine(std::numeric_limits<T>::lowest()); }
  void reset() { combiner_.reset_all(std::numeric_limits<T>::lowest()); }
  void describe(std::ostream& os) const override { os << get_value(); }

 private:
  detail::AgentCombiner<T, detail::MaxOp> combiner_;
};

template <typename T>
class Miner : public Variable {
 public:
  Miner() {}
  explicit Miner(const std::string& name) { expose(name); }
  ~Miner() override { hide(); }
  Miner& operator<<(T v) {
    auto* cell = combiner_.local_cell();
    T cur = cell->value.load(std::memory_order_relaxed);
    while (v < cur &&
           !cell->value.compare_exchange_weak(cur, v, std::memory_order_relaxed)) {
    }
    return *this;
  }
  T get_value() const { return combiner_.combine(std::numeric_limits<T>::max()); }
  void describe(std::ostream& os) const override { os << get_value(); }

 private:
  detail::AgentCombiner<T, detail::MinOp> combiner_;
};

// ---------------- status ----------------

template <typename T>
class Status : public Variable {
 public:
  explicit Status(T v = T()) : value_(v) {}
  Status(const std::string& name, T v) : value_(v) { expose(name); }
  ~Status() override { hide(); }
  void set_value(T v) {
    std::lock_guard<std::mutex> lk(mu_);
    value_ = v;
  }
  T get_value() const {
    std::lock_guard<std::mutex> lk(mu_);
    return value_;
  }
  void describe(std::ostream& os) const override { os << get_value(); }

 private:
  mutable std::mutex mu_;
  T value_;
};

class PassiveStatus : public Variable {
 public:
  typedef std::function<std::string()> Fn;
  explicit PassiveStatus(Fn fn) : fn_(std::move(fn)) {}
  PassiveStatus(const std::string& name, Fn fn) : fn_(std::move(fn)) { expose(name); }
  ~PassiveStatus() override { hide(); }
  void describe(std::ostream& os) const override { os << fn_(); }

 private:
  Fn fn_;
};

// ---------------- multi-dimension (labeled) vars ----------------

// Parity: reference bvar MultiDimension (mvar): one logical metric with
// per-label-value child reducers, dumped as name{label="v",...}.
template <typename VarType>
class MultiDimension : public Variable {
 public:
  MultiDimension(const std::string& name, std::vector<std::string> label_names)
      : labels_(std::move(label_names)) {
    expose(name);
  }
  ~MultiDimension() override {
    hide();
    for (auto& kv : children_) delete kv.second;
  }

  VarType* get_stats(const std::vector<std::string>& label_values) {
    std::string key = join(label_values);
    std::lock_guard<std::mutex> lk(mu_);
    auto it = children_.find(key);
    if (it != children_.end()) return it->second;
    VarType* child = new VarType;
    children_[key] = child;
    return child;
  }

  size_t count_stats() const {
    std::lock_guard<std::mutex> lk(mu_);
    return children_.size();
  }

  void describe(std::ostream& os) const override {
    std::lock_guard<std::mutex> lk(mu_);
    bool first = true;
    for (const auto& kv : children_) {
      if (!first) os << " ";
      first = false;
      os << "{" << kv.first << "}=" << kv.second->get_value();
    }
  }

 private:
  static std::string join(const std::vector<std::string>& vals) {
    std::string k;
    for (size_t i = 0; i < vals.size(); ++i) {
      if (i) k += ",";
      k += vals[i];
    }
    return k;
  }
  std::vector<std::string> labels_;
  mutable std::mutex mu_;
  std::map<std::string, VarType*> children_;
};

// ---------------- windowed values (1 Hz sampler) ----------------

// Samples an int64 source once per second into a 61-slot ring; value(w) =
// latest - sample[w seconds ago]; per_second(w) = value(w)/w.
class WindowedInt : public Variable {
 public:
  typedef std::function<int64_t()> SourceFn;
  WindowedInt(SourceFn src, int window_sec, bool per_second);
  ~WindowedInt() override;
  int64_t get_value() const;
  void describe(std::ostream& os) const override { os << get_value(); }
  void take_sample();  // called by the sampler thread

 private:
  SourceFn src_;
  int window_;
  bool per_second_;
  mutable std::mutex mu_;
  int64_t ring_[64];
  int64_t latest_;
  int nsamples_;
};

// Registers fn to run at 1 Hz on the global sampler thread.
void register_sampler(WindowedInt* w);
void unregister_sampler(WindowedInt* w);

// ---------------- latency recorder ----------------

// qps + avg + max + p50/p90/p99/p999. Percentiles come from per-thread
// log-bucket histograms (detail::LatencyHistogram) merged on read, over a
// sliding ~10 s window maintained by 1 Hz snapshots on the sampler thread
// — trustworthy under bursty load (parity: reference
// bvar/detail/percentile.h interval merge), unlike a last-N-samples ring.
class LatencyRecorder {
 public:
  LatencyRecorder();
  explicit LatencyRecorder(const std::string& prefix);
  ~LatencyRecorder();

  LatencyRecorder& operator<<(int64_t latency_us);

  int64_t count() const { return count_.get_value(); }
  int64_t qps() const;  // over the last 1s window
  int64_t latency_avg() const;
  int64_t latency_max() const { return window_max_.load(std::memory_order_relaxed); }
  int64_t latency_percentile(double p) const;

  void expose(const std::string& prefix);

 private:
  friend class LatencyDumper;
  void take_hist_snapshot();  // 1 Hz, sampler thread

  Adder<int64_t> count_;
  Adder<int64_t> sum_us_;
  std::atomic<int64_t> window_max_{0};
  detail::LatencyHistogram hist_;
  static constexpr int kWindowSec = 10;
  mutable std::mutex snap_mu_;
  std::deque<std::vector<uint32_t>> snaps_;  // oldest..newest cumulative counts
  std::unique_ptr<WindowedInt> qps_window_;
  std::unique_ptr<class SamplerTick> tick_;
  std::vector<Variable*> exposed_;
  std::string prefix_;
};

// Runs fn at ~1 Hz on the global sampler thread until destroyed.
class SamplerTick {
 public:
  explicit SamplerTick(std::function<void()> fn);
  ~SamplerTick();
  void run() { fn_(); }

 private:
  std::function<void()> fn_;
};

}  // namespace var
}  // namespace bam
