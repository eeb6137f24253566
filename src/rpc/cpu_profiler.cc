// brpc_amd: built-in CPU hotspot sampler.
// Parity: reference builtin/hotspots_service.cpp (gperftools
// ProfilerStart, weak-linked). Self-contained redesign — no gperftools in
// the image: an ITIMER_PROF / SIGPROF handler captures backtraces into a
// preallocated lock-free slot array for N seconds; the report aggregates
// by symbolized frame (dladdr) with self/cumulative counts. Exposed at
// /hotspots/cpu?seconds=N (builtin_services.cc).
#include <cxxabi.h>
#include <dlfcn.h>
#include <execinfo.h>
#include <signal.h>
#include <string.h>
#include <sys/time.h>

#include <algorithm>
#include <atomic>
#include <map>
#include <mutex>
#include <sstream>
#include <string>
#include <vector>

#include "fiber/butex.h"
#include "fiber/fiber.h"

namespace bam {

namespace {

constexpr int kMaxDepth = 24;
constexpr int kMaxSamples = 40000;

struct Sample {
  void* frames[kMaxDepth];
  int depth;
};

Sample* g_samples = nullptr;          // preallocated outside the handler
std::atomic<int> g_sample_idx{0};
std::atomic<bool> g_active{false};

void sigprof_handler(int, siginfo_t*, void*) {
  if (!g_active.load(std::memory_order_relaxed)) return;
  int i = g_sample_idx.fetch_add(1, std::memory_order_relaxed);
  if (i >= kMaxSamples) return;
  // backtrace() is not strictly async-signal-safe but is the standard
  // practice for SIGPROF samplers (gperftools does the equivalent via
  // libunwind); frames land in preallocated memory, no malloc here after
  // the first call (primed in CpuProfile below).
  g_samples[i].depth = backtrace(g_samples[i].frames, kMaxDepth);
}

std::string frame_name(void* ip) {
  Dl_info info;
  if (dladdr(ip, &info) != 0 && info.dli_sname != nullptr) {
    int status = 0;
    char* dem = abi::__cxa_demangle(info.dli_sname, nullptr, nullptr, &status);
    if (status == 0 && dem != nullptr) {
      std::string out(dem);
      free(dem);
      return out;
    }
    return info.dli_sname;
  }
  char buf[32];
  snprintf(buf, sizeof(buf), "%p", ip);
  return buf;
}

}  // namespace

// Renders the sampled butex contention sites (parity: reference
// /hotspots/contention fed by instrumented bthread_mutex): top callsites
// by total parked time among the last 4096 sampled parks (1/64 sampling).
std::string ContentionProfile() {
  std::vector<ContentionSample> samples(4096);
  size_t n = butex_contention_samples(samples.data(), samples.size());
  struct Site {
    int64_t total_us = 0;
    int64_t count = 0;
    void* frames[4];
    int nframes = 0;
  };
  std::map<std::string, Site> sites;  // keyed by caller frame chain addrs
  for (size_t i = 0; i < n; ++i) {
    const ContentionSample& s = samples[i];
    char key[80];
    // frames[0] is butex internals; aggregate by the 3 caller frames.
    snprintf(key, sizeof(key), "%p|%p|%p", s.nframes > 1 ? s.frames[1] : nullptr,
             s.nframes > 2 ? s.frames[2] : nullptr, s.nframes > 3 ? s.frames[3] : nullptr);
    Site& site = sites[key];
    site.total_us += s.wait_us;
    site.count += 1;
    if (site.nframes == 0) {
      site.nframes = s.nframes;
      memcpy(site.frames, s.frames, sizeof(s.frames));
    }
  }
  std::vector<const Site*> order;
  for (const auto& kv : sites) order.push_back(&kv.second);
  std::sort(order.begin(), order.end(),
            [](const Site* a, const Site* b) { return a->total_us > b->total_us; });
  std::ostringstream os;
  os << "contention profile: " << n << " sampled parks (1/64 sampling), "
     << sites.size() << " sites; totals are sampled-us (x64 ~= real)\n";
  os << "  total_us   count  callsite\n";
  int shown = 0;
  for (const Site* site : order) {
    if (++shown > 30) break;
    os << "  " << site->total_us << "  " << site->count << "  ";
    for (int d = 1; d < site->nframes && d < 4; ++d) {
      if (d > 1) os << " <- ";
      os << frame_name(site->frames[d]);
    }
    os << "\n";
  }
  return os.str();
}

// Shared sampling pass: arms ITIMER_PROF for `seconds` at `hz` and fills
// g_samples. Returns the sample count. Callers hold profile_mutex().
static std::mutex& profile_mutex() {
  static std::mutex mu;
  return mu;
}

static int collect_cpu_samples(int seconds, int hz) {
  if (g_samples == nullptr) g_samples = new Sample[kMaxSamples];
  {
    // prime backtrace()'s lazy libgcc initialization outside the handler
    void* prime[4];
    backtrace(prime, 4);
  }
  g_sample_idx.store(0, std::memory_order_relaxed);

  struct sigaction sa, old_sa;
  memset(&sa, 0, sizeof(sa));
  sa.sa_sigaction = sigprof_handler;
  sa.sa_flags = SA_SIGINFO | SA_RESTART;
  sigaction(SIGPROF, &sa, &old_sa);
  struct itimerval it, old_it;
  it.it_interval.tv_sec = 0;
  it.it_interval.tv_usec = 1000000 / hz;
  it.it_value = it.it_interval;
  g_active.store(true, std::memory_order_release);
  setitimer(ITIMER_PROF, &it, &old_it);

  fiber_usleep((uint64_t)seconds * 1000000);

  g_active.store(false, std::memory_order_release);
  memset(&it, 0, sizeof(it));
  setitimer(ITIMER_PROF, &it, nullptr);
  sigaction(SIGPROF, &old_sa, nullptr);
  return std::min(g_sample_idx.load(std::memory_order_relaxed), kMaxSamples);
}

// Samples the process for `seconds` (clamped to [1,30]) at `hz` and
// returns a text report. Serializes concurrent profile requests.
std::string CpuProfile(int seconds, int hz) {
  std::lock_guard<std::mutex> lk(profile_mutex());
  if (seconds < 1) seconds = 1;
  if (seconds > 30) seconds = 30;
  if (hz < 10) hz = 10;
  if (hz > 1000) hz = 1000;
  const int n = collect_cpu_samples(seconds, hz);
  std::map<std::string, std::pair<int, int>> agg;  // name -> {self, cumulative}
  for (int i = 0; i < n; ++i) {
    const Sample& s = g_samples[i];
    // frames[0..1] are the handler itself; self = first frame past them
    bool first = true;
    std::map<std::string, bool> seen;  // cumulative counts once per sample
    for (int d = 2; d < s.depth; ++d) {
      std::string name = frame_name(s.frames[d]);
      if (first) {
        ++agg[name].first;
        first = false;
      }
      if (!seen[name]) {
        ++agg[name].second;
        seen[name] = true;
      }
    }
  }
  std::vector<std::pair<std::string, std::pair<int, int>>> rows(agg.begin(), agg.end());
  std::sort(rows.begin(), rows.end(),
            [](const auto& a, const auto& b) { return a.second.first > b.second.first; });
  std::ostringstream os;
  os << "cpu profile: " << n << " samples @ " << hz << " Hz over " << seconds << " s\n";
  os << "  self   cum  symbol\n";
  int emitted = 0;
  for (const auto& r : rows) {
    if (r.second.first == 0 && emitted > 60) break;
    os << "  " << r.second.first << "  " << r.second.second << "  " << r.first << "\n";
    if (++emitted >= 100) break;
  }
  if (n == 0) os << "(no samples: process mostly idle or blocked — ITIMER_PROF counts CPU time)\n";
  return os.str();
}


// Legacy gperftools CPU-profile binary (what the pprof tool downloads from
// /pprof/profile; parity: reference builtin/pprof_service.cpp serving
// gperftools ProfilerStart output). Format: 8-byte LE words —
// header [0, 3, 0, period_us, 0], samples [count, num_pcs, pcs...],
// trailer [0, 1, 0].
std::string CpuProfileBinary(int seconds, int hz) {
  std::lock_guard<std::mutex> lk(profile_mutex());
  if (seconds < 1) seconds = 1;
  if (seconds > 30) seconds = 30;
  if (hz < 10) hz = 10;
  if (hz > 1000) hz = 1000;
  const int n = collect_cpu_samples(seconds, hz);
  std::string out;
  auto put = [&out](uint64_t w) { out.append((const char*)&w, 8); };
  put(0);
  put(3);
  put(0);
  put((uint64_t)(1000000 / hz));
  put(0);
  for (int i = 0; i < n; ++i) {
    const Sample& s = g_samples[i];
    int depth = s.depth > 2 ? s.depth - 2 : 0;  // frames[0..1] = handler
    if (depth == 0) continue;
    put(1);
    put((uint64_t)depth);
    for (int d = 2; d < s.depth; ++d) put((uint64_t)(uintptr_t)s.frames[d]);
  }
  put(0);
  put(1);
  put(0);
  return out;
}

// /pprof/symbol POST body: "0xaddr+0xaddr+..." -> "0xaddr\tname" lines.
std::string SymbolizeAddresses(const std::string& body) {
  std::ostringstream os;
  size_t pos = 0;
  while (pos < body.size()) {
    size_t plus = body.find('+', pos);
    std::string tok = body.substr(pos, plus == std::string::npos
                                           ? std::string::npos : plus - pos);
    pos = plus == std::string::npos ? body.size() : plus + 1;
    while (!tok.empty() && (tok.back() == '\n' || tok.back() == '\r')) tok.pop_back();
    if (tok.empty()) continue;
    uintptr_t addr = (uintptr_t)strtoull(tok.c_str(), nullptr, 16);
    os << tok << "\t" << frame_name((void*)addr) << "\n";
  }
  return os.str();
}

}  // namespace bam
