#include "rpc/rpcz.h"

#include <atomic>
#include <mutex>
#include <sstream>
#include <vector>

#include "base/flags.h"
#include "base/time.h"

namespace bam {

BAM_DEFINE_bool(enable_rpcz, true, "record per-RPC spans for /rpcz");
BAM_DEFINE_int64(rpcz_sample_mod, 16,
                 "record 1 of every N spans per thread (1 = all; parity: the "
                 "reference samples spans through a budgeted bvar collector)");
BAM_DEFINE_int64(rpcz_max_spans, 2048, "max spans kept in the rpcz ring");

namespace rpcz {

namespace {
struct Ring {
  std::mutex mu;
  std::vector<Span> spans;
  size_t next = 0;
  std::atomic<int64_t> total{0};
};
Ring& ring() {
  static Ring* r = new Ring;
  return *r;
}
}  // namespace

bool enabled() { return FLAG_enable_rpcz; }
void set_enabled(bool on) { FLAG_enable_rpcz = on; }

void RecordSpan(const Span& span) {
  if (!FLAG_enable_rpcz) return;
  // Per-thread sampling keeps the global ring mutex off the hot path
  // (2 spans/call x >100k QPS would serialize on it).
  static thread_local uint64_t tl_counter = 0;
  int64_t mod = FLAG_rpcz_sample_mod;
  if (mod > 1 && (tl_counter++ % (uint64_t)mod) != 0) return;
  Ring& r = ring();
  std::lock_guard<std::mutex> lk(r.mu);
  size_t cap = (size_t)FLAG_rpcz_max_spans;
  if (r.spans.size() < cap) {
    r.spans.push_back(span);
  } else {
    if (r.next >= r.spans.size()) r.next = 0;
    r.spans[r.next] = span;
  }
  ++r.next;
  r.total.fetch_add(1, std::memory_order_relaxed);
}

int64_t span_count() { return ring().total.load(std::memory_order_relaxed); }

void DumpRecentSpans(IOBuf* out, bool verbose) {
  Ring& r = ring();
  std::ostringstream os;
  std::lock_guard<std::mutex> lk(r.mu);
  os << "total_spans: " << r.total.load() << " (showing last " << r.spans.size() << ")\n";
  os << "time_us | side | method | remote | latency_us | error | req_bytes | resp_bytes\n";
  for (const Span& s : r.spans) {
    os << s.start_us << " | " << (s.server_side ? "S" : "C") << " | " << s.full_method
       << " | " << endpoint2str(s.remote) << " | " << (s.end_us - s.start_us) << " | "
       << s.error_code << " | " << s.request_size << " | " << s.response_size;
    if (verbose && s.log_id != 0) os << " | log_id=" << s.log_id;
    os << "\n";
  }
  out->append(os.str());
}

}  // namespace rpcz
}  // namespace bam
