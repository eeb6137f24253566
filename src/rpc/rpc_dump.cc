#include "rpc/rpc_dump.h"

#include <atomic>
#include <memory>
#include <mutex>

#include "base/fast_rand.h"
#include "base/flags.h"
#include "base/recordio.h"
#include "rpc/wire.h"

namespace bam {

BAM_DEFINE_bool(rpc_dump, false, "sample server requests into -rpc_dump_file");
BAM_DEFINE_int64(rpc_dump_ratio, 100, "sample 1/N of requests when -rpc_dump is on");
BAM_DEFINE_string(rpc_dump_file, "rpc_dump.recordio", "recordio output for sampled requests");

namespace rpc_dump {

namespace {
std::mutex g_mu;
std::unique_ptr<RecordWriter> g_writer;
std::string g_writer_path;
std::atomic<int64_t> g_count{0};
}  // namespace

void EncodeSample(const std::string& service, const std::string& method, const IOBuf& body,
                  std::string* out) {
  wire::put_str_field(out, 1, service);
  wire::put_str_field(out, 2, method);
  wire::put_str_field(out, 3, body.to_string());
}

bool DecodeSample(const std::string& rec, std::string* service, std::string* method,
                  std::string* body) {
  wire::Reader r(rec.data(), rec.size());
  int wt;
  for (int f; (f = r.read_tag(&wt)) != 0;) {
    if (f == 1) *service = r.read_string();
    else if (f == 2) *method = r.read_string();
    else if (f == 3) *body = r.read_string();
    else r.skip(wt);
    if (!r.ok()) return false;
  }
  return true;
}

void SampleRequest(const std::string& service, const std::string& method, const IOBuf& body) {
  if (!FLAG_rpc_dump) return;
  int64_t ratio = FLAG_rpc_dump_ratio;
  if (ratio > 1 && fast_rand_less_than((uint64_t)ratio) != 0) return;
  std::string rec;
  EncodeSample(service, method, body, &rec);
  std::lock_guard<std::mutex> lk(g_mu);
  if (g_writer == nullptr || g_writer_path != FLAG_rpc_dump_file) {
    g_writer.reset(new RecordWriter(FLAG_rpc_dump_file));
    g_writer_path = FLAG_rpc_dump_file;
  }
  if (g_writer->ok()) {
    g_writer->Write(rec);
    g_writer->Flush();
    g_count.fetch_add(1, std::memory_order_relaxed);
  }
}

int64_t sampled_count() { return g_count.load(std::memory_order_relaxed); }

}  // namespace rpc_dump
}  // namespace bam
