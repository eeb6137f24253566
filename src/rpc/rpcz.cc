#include "rpc/rpcz.h"

#include <mutex>

#include "fiber/key.h"

#include <atomic>
#include <mutex>
#include <sstream>
#include <vector>

#include "base/flags.h"
#include "base/recordio.h"
#include "base/time.h"

namespace bam {

BAM_DEFINE_bool(enable_rpcz, true, "record per-RPC spans for /rpcz");
BAM_DEFINE_int64(rpcz_sample_mod, 16,
                 "record 1 of every N spans per thread (1 = all; parity: the "
                 "reference samples spans through a budgeted bvar collector)");
BAM_DEFINE_int64(rpcz_max_spans, 2048, "max spans kept in the rpcz ring");
BAM_DEFINE_string(rpcz_db_path, "",
                  "when set, sampled spans also persist to this recordio "
                  "file (parity: the reference's leveldb-backed SpanDB; "
                  "queried back via /rpcz?db=N)");

namespace rpcz {

namespace {

// ---- persistent span store (recordio-backed SpanDB) ----
// Spans serialize as one line per record: tab-separated fields. Writes
// are buffered under a mutex and flushed opportunistically.
struct SpanDb {
  std::mutex mu;
  RecordWriter* writer = nullptr;
  std::string path;
  int64_t written = 0;

  void maybe_open() {
    if (writer != nullptr && path == FLAG_rpcz_db_path) return;
    delete writer;
    writer = nullptr;
    path = FLAG_rpcz_db_path;
    if (!path.empty()) {
      writer = new RecordWriter(path);
      if (!writer->ok()) {
        delete writer;
        writer = nullptr;
      }
    }
  }
};
SpanDb& span_db() {
  static SpanDb* db = new SpanDb;
  return *db;
}

std::string span_to_record(const Span& s) {
  std::ostringstream os;
  os << s.start_us << '\t' << s.end_us << '\t' << (s.server_side ? 'S' : 'C') << '\t'
     << s.full_method << '\t' << endpoint2str(s.remote) << '\t' << s.error_code << '\t'
     << s.log_id << '\t' << s.request_size << '\t' << s.response_size << '\t'
     << s.trace_id << '\t' << s.span_id << '\t' << s.parent_span_id;
  return os.str();
}

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

namespace {
fiber_key_t g_trace_key;
std::once_flag g_trace_key_once;

void trace_dtor(void* p) { delete (TraceContext*)p; }

fiber_key_t trace_key() {
  std::call_once(g_trace_key_once, [] { fiber_key_create(&g_trace_key, trace_dtor); });
  return g_trace_key;
}
}  // namespace

TraceContext current_trace() {
  TraceContext* c = (TraceContext*)fiber_getspecific(trace_key());
  return c != nullptr ? *c : TraceContext{};
}

void set_current_trace(uint64_t trace_id, uint64_t span_id) {
  TraceContext* c = (TraceContext*)fiber_getspecific(trace_key());
  if (c == nullptr) {
    c = new TraceContext;
    fiber_setspecific(trace_key(), c);
  }
  c->trace_id = trace_id;
  c->span_id = span_id;
}

void clear_current_trace() {
  TraceContext* c = (TraceContext*)fiber_getspecific(trace_key());
  if (c != nullptr) c->trace_id = c->span_id = 0;
}

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
  if (!FLAG_rpcz_db_path.empty()) {
    SpanDb& db = span_db();
    std::lock_guard<std::mutex> dlk(db.mu);
    db.maybe_open();
    if (db.writer != nullptr) {
      db.writer->Write(span_to_record(span));
      if ((++db.written & 63) == 0) db.writer->Flush();
    }
  }
}

// Reads back up to `max` most recent persisted spans (whole-file scan —
// recordio is append-only; /rpcz?db=N is a diagnostics query, not a hot
// path). Returns lines.
std::vector<std::string> ReadPersistedSpans(int max) {
  std::vector<std::string> out;
  std::string path;
  {
    SpanDb& db = span_db();
    std::lock_guard<std::mutex> lk(db.mu);
    if (db.writer != nullptr) db.writer->Flush();
    path = db.path.empty() ? FLAG_rpcz_db_path : db.path;
  }
  if (path.empty()) return out;
  RecordReader reader(path);
  if (!reader.ok()) return out;
  std::string rec;
  while (reader.Next(&rec)) {
    out.push_back(rec);
    if ((int)out.size() > max * 4) out.erase(out.begin(), out.begin() + max);
  }
  if ((int)out.size() > max) out.erase(out.begin(), out.end() - max);
  return out;
}

int64_t span_count() { return ring().total.load(std::memory_order_relaxed); }

void DumpPersistedSpans(IOBuf* out, int max) {
  std::vector<std::string> recs = ReadPersistedSpans(max > 0 ? max : 100);
  std::ostringstream os;
  os << "persisted_spans (last " << recs.size() << " from " << FLAG_rpcz_db_path << ")\n";
  os << "start_us\tend_us\tside\tmethod\tremote\terror\tlog_id\treq\tresp"
        "\ttrace_id\tspan_id\tparent_span_id\n";
  for (const std::string& r : recs) os << r << "\n";
  out->append(os.str());
}

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
    if (verbose && s.trace_id != 0) {
      os << " | trace=" << std::hex << s.trace_id << " span=" << s.span_id;
      if (s.parent_span_id != 0) os << " parent=" << s.parent_span_id;
      os << std::dec;
    }
    os << "\n";
  }
  out->append(os.str());
}

}  // namespace rpcz
}  // namespace bam
