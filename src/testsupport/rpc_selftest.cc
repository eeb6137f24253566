// C++ RPC test scenarios + benchmark loops, driven from pytest / bench.py.
// Models the reference's single-process loopback fixtures
// (test/brpc_channel_unittest.cpp ChannelTest): server and client in one
// process over 127.0.0.1.
#include <algorithm>
#include <atomic>
#include <set>
#include <memory>
#include <mutex>
#include <vector>

#include "base/fast_rand.h"
#include "base/time.h"
#include "fiber/fiber.h"
#include "fiber/sync.h"
#include "rpc/channel.h"
#include "rpc/closure.h"
#include "rpc/controller.h"
#include "rpc/server.h"
#include "rpc/shm_ring.h"

namespace bam {
namespace rpctest {

// ---- the universal fixture: EchoService (parity: test/echo.proto) ----

Service* NewEchoService() {
  Service* svc = new Service("EchoService");
  svc->AddMethod("Echo", [](Controller* cntl, const IOBuf& req, IOBuf* resp, Closure* done) {
    resp->append(req);
    cntl->response_attachment().append(cntl->request_attachment());
    done->Run();
  });
  svc->AddMethod("Sleep", [](Controller* cntl, const IOBuf& req, IOBuf* resp, Closure* done) {
    char buf[16] = {0};
    req.copy_to(buf, sizeof(buf) - 1, 0);
    int ms = atoi(buf);
    fiber_usleep((uint64_t)ms * 1000);
    resp->append("slept");
    done->Run();
  });
  svc->AddMethod("Fail", [](Controller* cntl, const IOBuf& req, IOBuf* resp, Closure* done) {
    cntl->SetFailed(EINTERNAL, "you asked for it");
    done->Run();
  });
  svc->AddMethod("EchoCompressed",
                 [](Controller* cntl, const IOBuf& req, IOBuf* resp, Closure* done) {
    resp->append(req);
    cntl->set_response_compress_type(COMPRESS_TYPE_SNAPPY);
    done->Run();
  });
  // Nested-call relay: request = "host:port|payload". The handler calls
  // EchoService.Echo on that address; trace ids must chain through the
  // ambient rpcz context (server span -> nested client span).
  svc->AddMethod("Relay", [](Controller* cntl, const IOBuf& req, IOBuf* resp, Closure* done) {
    std::string s = req.to_string();
    size_t bar = s.find('|');
    if (bar == std::string::npos) {
      cntl->SetFailed(EREQUEST, "Relay wants addr|payload");
      done->Run();
      return;
    }
    Channel ch;
    ChannelOptions opts;
    opts.timeout_ms = 3000;
    if (ch.Init(s.substr(0, bar).c_str(), &opts) != 0) {
      cntl->SetFailed(EINTERNAL, "relay channel init failed");
      done->Run();
      return;
    }
    Controller c2;
    IOBuf r2, resp2;
    r2.append(s.substr(bar + 1));
    ch.CallMethod("EchoService.Echo", &c2, &r2, &resp2, nullptr);
    if (c2.Failed()) {
      cntl->SetFailed(c2.ErrorCode(), "relay: " + c2.ErrorText());
    } else {
      resp->append(resp2);
    }
    done->Run();
  });
  svc->AddMethod("Port", [](Controller* cntl, const IOBuf& req, IOBuf* resp, Closure* done) {
    resp->append(std::to_string(cntl->server_->listen_address().port));
    done->Run();
  });
  // BASELINE config 2 (IOBuf-in-HBM echo): the response payload is staged
  // into HBM-resident blocks; the socket write path stages it back through
  // the GPU byte mover. Falls back with an explicit error if no GPU.
  svc->AddMethod("EchoHbm", [](Controller* cntl, const IOBuf& req, IOBuf* resp, Closure* done) {
    if (!has_block_allocator(RES_HBM)) {
      cntl->SetFailed(EINTERNAL, "no HBM allocator (GPU library not loaded)");
      done->Run();
      return;
    }
    std::string bytes = req.to_string();
    if (resp->append_with_residency(bytes.data(), bytes.size(), RES_HBM, 0, 0) != 0) {
      cntl->SetFailed(EINTERNAL, "HBM append failed");
    }
    done->Run();
  });
  return svc;
}

// Starts an echo server on 127.0.0.1:port (0 = auto). Returns port or -1.
// Servers leak by design in tests (process-lifetime).
int start_echo_server(int port) {
  Server* server = new Server;
  server->AddService(NewEchoService(), SERVER_OWNS_SERVICE);
  if (server->Start(port, nullptr) != 0) return -1;
  return server->listen_address().port;
}

// Server with ONLY a master catch-all (≙ reference BaiduMasterService):
// any service/method echoes back "master:<svc>.<method>:<body>".
int start_master_echo_server() {
  Server* server = new Server;
  ServerOptions opts;
  opts.master_handler = [](Controller* cntl, const IOBuf& req, IOBuf* resp, Closure* done) {
    resp->append("master:" + cntl->call.service_name + "." + cntl->call.method_name + ":");
    resp->append(req);
    done->Run();
  };
  if (server->Start(0, &opts) != 0) return -1;
  return server->listen_address().port;
}

// Echo server whose accepted connections upgrade to the RDMA mock
// endpoint (ServerOptions.socket_mode="rdma_mock"; rpc/rdma_transport.h).
int start_rdma_mock_echo_server() {
  Server* server = new Server;
  server->AddService(NewEchoService(), SERVER_OWNS_SERVICE);
  ServerOptions opts;
  opts.socket_mode = "rdma_mock";
  if (server->Start(0, &opts) != 0) return -1;
  return server->listen_address().port;
}

// Echo server with a short idle timeout (reaper test).
int start_idle_timeout_server(int idle_sec) {
  Server* server = new Server;
  server->AddService(NewEchoService(), SERVER_OWNS_SERVICE);
  ServerOptions opts;
  opts.idle_timeout_sec = idle_sec;
  if (server->Start(0, &opts) != 0) return -1;
  return server->listen_address().port;
}

// Starts an echo server that ALSO serves nshead raw-body echo (body is
// echoed back with "N:" prefixed). Returns port.
int start_nshead_server() {
  Server* server = new Server;
  server->AddService(NewEchoService(), SERVER_OWNS_SERVICE);
  ServerOptions opts;
  opts.nshead_handler = [](const IOBuf& req, IOBuf* resp) {
    resp->append("N:");
    resp->append(req);
  };
  if (server->Start(0, &opts) != 0) return -1;
  return server->listen_address().port;
}

// One sync call over an arbitrary wire protocol (std/hulu_pbrpc/sofa_pbrpc/
// nshead/...). Returns 0 on success.
int protocol_call(const std::string& addr, const std::string& protocol,
                  const std::string& method, const std::string& payload, int compress,
                  std::string* response_out, std::string* err) {
  Channel channel;
  ChannelOptions opts;
  opts.protocol = protocol;
  opts.timeout_ms = 3000;
  opts.max_retry = 0;
  if (channel.Init(addr.c_str(), &opts) != 0) {
    if (err != nullptr) *err = "Init failed";
    return -1;
  }
  Controller cntl;
  cntl.set_request_compress_type((CompressType)compress);
  IOBuf request, response;
  request.append(payload);
  channel.CallMethod(method, &cntl, &request, &response, nullptr);
  if (cntl.Failed()) {
    if (err != nullptr) *err = cntl.ErrorText();
    return cntl.ErrorCode();
  }
  if (response_out != nullptr) *response_out = response.to_string();
  return 0;
}

// ---- StartCancel ----
// Sleeps 800 ms server-side; cancels after ~50 ms. Returns the observed
// client-side latency in us when ErrorCode()==ECANCELED_RPC, else -code.
int64_t cancel_test(int port) {
  Channel channel;
  ChannelOptions opts;
  opts.timeout_ms = 5000;
  opts.max_retry = 2;
  if (channel.Init(("127.0.0.1:" + std::to_string(port)).c_str(), &opts) != 0) return -1;
  Controller cntl;
  IOBuf req, resp;
  req.append("800");  // Sleep ms
  CountdownEvent done_ev(1);
  Closure* done = NewCallback([&done_ev] { done_ev.signal(); });
  int64_t t0 = monotonic_time_us();
  channel.CallMethod("EchoService.Sleep", &cntl, &req, &resp, done);
  fiber_usleep(50000);
  cntl.StartCancel();
  done_ev.wait();
  int64_t lat = monotonic_time_us() - t0;
  if (cntl.ErrorCode() != ECANCELED_RPC) return -cntl.ErrorCode();
  return lat;
}

// ---- retry policy hook ----
// Calls a dead port with a policy that refuses retries; returns the
// observed attempt count (policy calls) — must be exactly 1.
int retry_policy_test(int max_retry) {
  Channel channel;
  ChannelOptions opts;
  opts.timeout_ms = 2000;
  opts.max_retry = max_retry;
  static std::atomic<int> calls{0};
  calls = 0;
  opts.retry_policy = [](int /*error_code*/, int /*attempt*/) {
    calls.fetch_add(1);
    return false;  // never retry
  };
  if (channel.Init("127.0.0.1:1", &opts) != 0) return -1;
  Controller cntl;
  IOBuf req, resp;
  req.append("x");
  channel.CallMethod("EchoService.Echo", &cntl, &req, &resp, nullptr);
  if (!cntl.Failed()) return -2;
  return calls.load();
}

// ---- session-local data (per-connection counter) ----
int start_session_counter_server() {
  Server* server = new Server;
  Service* svc = new Service("Sess");
  svc->AddMethod("Count", [](Controller* cntl, const IOBuf&, IOBuf* resp, Closure* done) {
    int64_t* n = (int64_t*)cntl->session_local_data();
    if (n == nullptr) {
      cntl->SetFailed(EINTERNAL, "no session data");
    } else {
      resp->append(std::to_string(++*n));
    }
    done->Run();
  });
  server->AddService(svc, SERVER_OWNS_SERVICE);
  ServerOptions opts;
  opts.session_local_data_factory = [] { return (void*)new int64_t(0); };
  opts.session_local_data_deleter = [](void* p) { delete (int64_t*)p; };
  if (server->Start(0, &opts) != 0) return -1;
  return server->listen_address().port;
}

// ---- rtmp media server ----
int start_rtmp_server() {
  Server* server = new Server;
  server->AddService(NewEchoService(), SERVER_OWNS_SERVICE);
  ServerOptions opts;
  opts.enable_rtmp = true;
  if (server->Start(0, &opts) != 0) return -1;
  return server->listen_address().port;
}

// ---- mongo server (adaptor parity) ----
// Replies {ok: 1.0} to OP_QUERY ("ismaster"-style) and OP_MSG; records
// fire-and-forget inserts into a counter readable via OP_QUERY on
// "insert_count" (session-free demo of the adaptor contract).
int start_mongo_echo_server() {
  Server* server = new Server;
  server->AddService(NewEchoService(), SERVER_OWNS_SERVICE);
  ServerOptions opts;
  static std::atomic<int> g_inserts{0};
  opts.mongo_handler = [](const MongoHeader& head, const IOBuf& body, MongoReply* reply) {
    if (head.op_code == 2002) {  // OP_INSERT: fire-and-forget
      g_inserts.fetch_add(1, std::memory_order_relaxed);
      return;
    }
    // minimal BSON {"ok": 1.0, "n": <inserts>} (double 0x01, int32 0x10)
    std::string doc;
    auto i32 = [&](int32_t v) { doc.append((const char*)&v, 4); };
    double one = 1.0;
    std::string elems;
    elems.push_back('\x01');
    elems.append("ok", 3);
    elems.append((const char*)&one, 8);
    elems.push_back('\x10');
    elems.append("n", 2);
    int32_t n = g_inserts.load(std::memory_order_relaxed);
    elems.append((const char*)&n, 4);
    i32((int32_t)(4 + elems.size() + 1));
    doc += elems;
    doc.push_back('\0');
    if (head.op_code == 2013) reply->body.append("\0", 1);  // OP_MSG section kind 0
    reply->body.append(doc);
  };
  if (server->Start(0, &opts) != 0) return -1;
  return server->listen_address().port;
}

// ---- shm ring transport (UBRing analogue, rpc/shm_ring.h) ----

int start_shm_echo_server(const std::string& name) {
  Server* server = new Server;  // never bound to TCP: shm-only dispatch
  server->AddService(NewEchoService(), SERVER_OWNS_SERVICE);
  return shm::ServeShm(name, server);
}

int shm_call(const std::string& name, const std::string& method, const std::string& payload,
             std::string* response_out, std::string* err) {
  shm::ShmChannel ch;
  if (ch.Init(name) != 0) {
    *err = "shm connect failed";
    return -1;
  }
  IOBuf req, resp;
  req.append(payload);
  int rc = ch.Call(method, req, &resp, 3000000, err);
  if (rc == 0 && response_out != nullptr) *response_out = resp.to_string();
  return rc;
}

// Throughput/latency: `concurrency` fibers each issue calls/concurrency
// sync echoes over ONE shm connection. Returns {qps, p99_us, errors}.
namespace {
struct ShmBenchArg {
  shm::ShmChannel* ch;
  std::string payload;
  int calls;
  std::atomic<int>* errors;
  std::vector<int64_t>* lat;  // per-fiber slice, preallocated
  size_t lat_off;
};
void shm_bench_fiber(void* raw) {
  ShmBenchArg* a = (ShmBenchArg*)raw;
  IOBuf req;
  req.append(a->payload);
  for (int i = 0; i < a->calls; ++i) {
    IOBuf resp;
    int64_t t0 = monotonic_time_us();
    int rc = a->ch->Call("EchoService.Echo", req, &resp, 5000000, nullptr);
    (*a->lat)[a->lat_off + i] = monotonic_time_us() - t0;
    if (rc != 0 || resp.size() != a->payload.size()) a->errors->fetch_add(1);
  }
}
}  // namespace

int shm_echo_bench(const std::string& name, int payload, int concurrency, int calls,
                   double* qps, int64_t* p99_us, int* errors_out) {
  shm::ShmChannel ch;
  if (ch.Init(name) != 0) return -1;
  std::string pay((size_t)payload, 'u');
  int per = calls / concurrency;
  std::vector<int64_t> lat((size_t)per * concurrency, 0);
  std::atomic<int> errors{0};
  std::vector<ShmBenchArg> args((size_t)concurrency);
  std::vector<fiber_t> tids((size_t)concurrency);
  int64_t t0 = monotonic_time_us();
  for (int i = 0; i < concurrency; ++i) {
    args[i] = ShmBenchArg{&ch, pay, per, &errors, &lat, (size_t)i * per};
    fiber_start_background(&tids[i], shm_bench_fiber, &args[i]);
  }
  for (int i = 0; i < concurrency; ++i) fiber_join(tids[i]);
  int64_t elapsed = monotonic_time_us() - t0;
  std::sort(lat.begin(), lat.end());
  *qps = (double)lat.size() * 1e6 / (double)elapsed;
  *p99_us = lat[(size_t)((double)lat.size() * 0.99)];
  *errors_out = errors.load();
  return 0;
}

// One sync echo; returns 0 on success and fills latency_us, else error code.
int echo_once(const std::string& addr, const std::string& payload, int timeout_ms,
              std::string* response_out, int64_t* latency_us) {
  Channel channel;
  ChannelOptions opts;
  opts.timeout_ms = timeout_ms;
  if (channel.Init(addr.c_str(), &opts) != 0) return -1;
  Controller cntl;
  IOBuf request, response;
  request.append(payload);
  channel.CallMethod("EchoService.Echo", &cntl, &request, &response, nullptr);
  if (latency_us != nullptr) *latency_us = cntl.latency_us();
  if (cntl.Failed()) return cntl.ErrorCode();
  if (response_out != nullptr) *response_out = response.to_string();
  return 0;
}

int call_method_once(const std::string& addr, const std::string& method,
                     const std::string& payload, int timeout_ms, int max_retry,
                     std::string* response_out, std::string* error_text) {
  Channel channel;
  ChannelOptions opts;
  opts.timeout_ms = timeout_ms;
  opts.max_retry = max_retry;
  if (channel.Init(addr.c_str(), &opts) != 0) return -1;
  Controller cntl;
  cntl.set_max_retry(max_retry);
  IOBuf request, response;
  request.append(payload);
  channel.CallMethod(method, &cntl, &request, &response, nullptr);
  if (error_text != nullptr) *error_text = cntl.ErrorText();
  if (cntl.Failed()) return cntl.ErrorCode();
  if (response_out != nullptr) *response_out = response.to_string();
  return 0;
}

// Attachment round-trip.
bool attachment_test(const std::string& addr) {
  Channel channel;
  if (channel.Init(addr.c_str(), nullptr) != 0) return false;
  Controller cntl;
  IOBuf request, response;
  request.append("body");
  std::string att(10000, 'A');
  cntl.request_attachment().append(att);
  channel.CallMethod("EchoService.Echo", &cntl, &request, &response, nullptr);
  if (cntl.Failed()) return false;
  return response.to_string() == "body" && cntl.response_attachment().to_string() == att;
}

// ---- benchmark: concurrent echo QPS + latency percentiles ----

struct BenchResult {
  double qps = 0;
  double mbps = 0;  // payload throughput (both directions counted once)
  int64_t p50_us = 0, p90_us = 0, p99_us = 0, p999_us = 0, max_us = 0, avg_us = 0;
  int64_t errors = 0;
  int64_t total = 0;
  std::string first_error;
};

namespace {

struct BenchWorkerArg {
  Channel* channel;
  std::vector<std::unique_ptr<Channel>>* channels = nullptr;  // multi-channel mode
  std::atomic<int>* next_worker = nullptr;
  std::atomic<int64_t>* remaining;
  std::vector<int64_t>* latencies;  // pre-sized; indexed by call #
  std::atomic<int64_t>* errors;
  std::string payload;
  std::string method;
  bool hbm_request;
  CountdownEvent* done_event;
  std::mutex* err_mu;
  std::string* first_error;
};

void bench_worker(void* raw) {
  BenchWorkerArg* a = (BenchWorkerArg*)raw;
  Channel* ch = a->channel;
  if (a->channels != nullptr && a->next_worker != nullptr) {
    int wid = a->next_worker->fetch_add(1, std::memory_order_relaxed);
    ch = (*a->channels)[wid % (int)a->channels->size()].get();
  }
  for (;;) {
    int64_t idx = a->remaining->fetch_sub(1, std::memory_order_relaxed);
    if (idx <= 0) break;
    Controller cntl;
    IOBuf request, response;
    if (a->hbm_request) {
      request.append_with_residency(a->payload.data(), a->payload.size(), RES_HBM, 0, 0);
    } else {
      request.append(a->payload);
    }
    ch->CallMethod(a->method, &cntl, &request, &response, nullptr);
    if (cntl.Failed() || response.size() != a->payload.size()) {
      a->errors->fetch_add(1, std::memory_order_relaxed);
      std::lock_guard<std::mutex> lk(*a->err_mu);
      if (a->first_error->empty()) {
        *a->first_error = cntl.Failed()
                              ? std::to_string(cntl.ErrorCode()) + ":" + cntl.ErrorText()
                              : "short response " + std::to_string(response.size());
      }
    }
    (*a->latencies)[idx - 1] = cntl.latency_us();
  }
  a->done_event->signal();
}

}  // namespace

BenchResult echo_bench(const std::string& addr, int payload_size, int concurrency,
                       int64_t total_calls, int timeout_ms, const std::string& method,
                       bool hbm_request, bool pooled, int nchannels,
                       const std::string& socket_mode) {
  BenchResult res;
  if (nchannels < 1) nchannels = 1;
  std::vector<std::unique_ptr<Channel>> channels(nchannels);
  for (int i = 0; i < nchannels; ++i) {
    channels[i] = std::make_unique<Channel>();
    ChannelOptions opts;
    opts.timeout_ms = timeout_ms;
    if (pooled) opts.connection_type = "pooled";
    opts.socket_mode = socket_mode;
    if (channels[i]->Init(addr.c_str(), &opts) != 0) return res;
  }
  Channel& channel = *channels[0];
  (void)channel;
  std::string payload(payload_size, 'x');
  for (int i = 0; i < payload_size; ++i) payload[i] = (char)fast_rand();
  std::atomic<int64_t> remaining{total_calls};
  std::atomic<int64_t> errors{0};
  std::vector<int64_t> latencies(total_calls, 0);
  CountdownEvent done_event(concurrency);
  std::mutex err_mu;
  std::atomic<int> next_worker{0};
  BenchWorkerArg arg;
  arg.channel = &channel;
  arg.channels = &channels;
  arg.next_worker = &next_worker;
  arg.remaining = &remaining;
  arg.latencies = &latencies;
  arg.errors = &errors;
  arg.payload = payload;
  arg.method = method;
  arg.hbm_request = hbm_request;
  arg.done_event = &done_event;
  arg.err_mu = &err_mu;
  arg.first_error = &res.first_error;
  int64_t t0 = monotonic_time_us();
  for (int i = 0; i < concurrency; ++i) {
    fiber_t th;
    fiber_start_background(&th, bench_worker, &arg);
  }
  done_event.wait();
  int64_t elapsed = monotonic_time_us() - t0;
  std::sort(latencies.begin(), latencies.end());
  res.total = total_calls;
  res.errors = errors.load();
  res.qps = total_calls * 1e6 / (double)elapsed;
  res.mbps = res.qps * payload_size / 1e6;
  int64_t sum = 0;
  for (int64_t v : latencies) sum += v;
  res.avg_us = total_calls > 0 ? sum / total_calls : 0;
  auto pct = [&](double p) {
    size_t i = (size_t)(p * (latencies.size() - 1));
    return latencies[i];
  };
  if (!latencies.empty()) {
    res.p50_us = pct(0.50);
    res.p90_us = pct(0.90);
    res.p99_us = pct(0.99);
    res.p999_us = pct(0.999);
    res.max_us = latencies.back();
  }
  return res;
}

namespace {

// Pipelined async bench (≙ reference test/brpc_channel_unittest.cpp async
// paths + docs/cn/benchmark.md single-connection pipelining): `pipeline`
// calls stay in flight on the channel; each completion immediately issues
// the next, so the wait-free write queue coalesces requests into large
// writev batches and the reader parses many responses per wakeup.
struct AsyncBenchCtx {
  Channel* channel = nullptr;
  std::atomic<int64_t> remaining{0};
  std::atomic<int64_t> errors{0};
  std::string payload;
  std::string method;
  CountdownEvent* done_event = nullptr;  // count == pipeline
  std::vector<int64_t>* latencies = nullptr;
};

void async_issue(AsyncBenchCtx* c);

struct AsyncCall {
  Controller cntl;
  IOBuf req, resp;
};

void async_reissue_fiber(void* raw) { async_issue((AsyncBenchCtx*)raw); }

void async_issue(AsyncBenchCtx* c) {
  int64_t idx = c->remaining.fetch_sub(1, std::memory_order_relaxed);
  if (idx <= 0) {
    c->done_event->signal();
    return;
  }
  AsyncCall* call = new AsyncCall;
  call->req.append(c->payload);
  c->channel->CallMethod(c->method, &call->cntl, &call->req, &call->resp,
                         NewCallback([c, call, idx] {
    if (call->cntl.Failed() || call->resp.size() != c->payload.size()) {
      c->errors.fetch_add(1, std::memory_order_relaxed);
    }
    (*c->latencies)[idx - 1] = call->cntl.latency_us();
    delete call;
    // Completions normally arrive from the reader; an inline-failed call
    // would recurse here, so bound the depth with a fiber trampoline.
    thread_local int depth = 0;
    if (++depth > 64) {
      depth = 0;
      fiber_t th;
      fiber_start_background(&th, async_reissue_fiber, c);
    } else {
      async_issue(c);
      --depth;
    }
  }));
}

}  // namespace

BenchResult async_echo_bench(const std::string& addr, int payload_size, int pipeline,
                             int64_t total_calls, int timeout_ms, const std::string& method,
                             bool pooled) {
  BenchResult res;
  Channel channel;
  ChannelOptions opts;
  opts.timeout_ms = timeout_ms;
  opts.max_retry = 0;
  if (pooled) opts.connection_type = "pooled";
  if (channel.Init(addr.c_str(), &opts) != 0) return res;
  std::string payload(payload_size, 'x');
  for (int i = 0; i < payload_size; ++i) payload[i] = (char)fast_rand();
  std::vector<int64_t> latencies(total_calls, 0);
  CountdownEvent done_event(pipeline);
  AsyncBenchCtx ctx;
  ctx.channel = &channel;
  ctx.remaining.store(total_calls);
  ctx.payload = payload;
  ctx.method = method;
  ctx.done_event = &done_event;
  ctx.latencies = &latencies;
  int64_t t0 = monotonic_time_us();
  for (int i = 0; i < pipeline; ++i) async_issue(&ctx);
  done_event.wait();
  int64_t elapsed = monotonic_time_us() - t0;
  std::sort(latencies.begin(), latencies.end());
  res.total = total_calls;
  res.errors = ctx.errors.load();
  res.qps = total_calls * 1e6 / (double)elapsed;
  res.mbps = res.qps * payload_size / 1e6;
  int64_t sum = 0;
  for (int64_t v : latencies) sum += v;
  res.avg_us = total_calls > 0 ? sum / total_calls : 0;
  if (!latencies.empty()) {
    auto pct = [&](double p) {
      size_t i = (size_t)(p * (latencies.size() - 1));
      return latencies[i];
    };
    res.p50_us = pct(0.50);
    res.p90_us = pct(0.90);
    res.p99_us = pct(0.99);
    res.p999_us = pct(0.999);
    res.max_us = latencies.back();
  }
  return res;
}

}  // namespace rpctest
}  // namespace bam

#include "rpc/load_balancer.h"
#include "rpc/socket_map.h"

namespace bam {
namespace rpctest {

// Controller::http_request()/http_response() (≙ reference HttpHeader
// accessors, controller.h:400-430): custom verb + headers reach the
// server handler; handler-set status/content-type/headers reach the
// client's http_response view.
bool http_header_ext_test(std::string* err) {
  Server* server = new Server;
  Service* svc = new Service("Hx");
  svc->AddMethod("Probe", [](Controller* cntl, const IOBuf&, IOBuf* resp, Closure* done) {
    const HttpHeaderExt& req = cntl->http_request();
    const std::string* custom = req.GetHeader("X-Custom");
    if (req.method != "PUT" || custom == nullptr || *custom != "v1") {
      cntl->SetFailed(EREQUEST, "missing verb/header: method=" + req.method);
      done->Run();
      return;
    }
    HttpHeaderExt& r = cntl->http_response();
    r.status_code = 202;
    r.content_type = "application/json";
    r.SetHeader("X-Served-By", "hx-test");
    resp->append("{\"ok\":true}");
    done->Run();
  });
  server->AddService(svc, SERVER_OWNS_SERVICE);
  if (server->Start(0, nullptr) != 0) {
    *err = "start failed";
    return false;
  }
  std::string addr = "127.0.0.1:" + std::to_string(server->listen_address().port);
  Channel ch;
  ChannelOptions opts;
  opts.protocol = "http";
  opts.timeout_ms = 3000;
  opts.max_retry = 0;
  if (ch.Init(addr.c_str(), &opts) != 0) {
    *err = "init failed";
    return false;
  }
  Controller cntl;
  cntl.http_request().method = "PUT";
  cntl.http_request().SetHeader("X-Custom", "v1");
  IOBuf q, r;
  q.append("body");
  ch.CallMethod("Hx.Probe", &cntl, &q, &r, nullptr);
  if (cntl.Failed()) {
    *err = "call failed: " + cntl.ErrorText();
    return false;
  }
  const HttpHeaderExt& hr = cntl.http_response();
  if (hr.status_code != 202) {
    *err = "status " + std::to_string(hr.status_code);
    return false;
  }
  if (hr.content_type != "application/json") {
    *err = "content_type " + hr.content_type;
    return false;
  }
  const std::string* sb = hr.GetHeader("x-served-by");
  if (sb == nullptr || *sb != "hx-test") {
    *err = "x-served-by missing";
    return false;
  }
  if (r.to_string() != "{\"ok\":true}") {
    *err = "body " + r.to_string();
    return false;
  }
  return true;
}

// connection_type="short" (≙ reference CONNECTION_TYPE_SHORT): a fresh
// TCP connection per call, closed when the call completes — the server's
// socket count must not grow with the call count.
bool short_connection_test(std::string* err) {
  int port = start_echo_server(0);
  std::string addr = "127.0.0.1:" + std::to_string(port);
  Channel ch;
  ChannelOptions opts;
  opts.timeout_ms = 3000;
  opts.connection_type = "short";
  if (ch.Init(addr.c_str(), &opts) != 0) {
    *err = "init failed";
    return false;
  }
  std::set<std::string> local_ports;
  for (int i = 0; i < 8; ++i) {
    Controller cntl;
    IOBuf q, r;
    q.append("s");
    ch.CallMethod("EchoService.Echo", &cntl, &q, &r, nullptr);
    if (cntl.Failed()) {
      *err = "call " + std::to_string(i) + " failed: " + cntl.ErrorText();
      return false;
    }
    local_ports.insert(std::string(endpoint2str(cntl.local_side())));
  }
  // every call used a DIFFERENT ephemeral local port (fresh connection)
  if (local_ports.size() < 8) {
    *err = "only " + std::to_string(local_ports.size()) + " distinct connections for 8 calls";
    return false;
  }
  return true;
}

// Controller::set_request_code + c_hash (≙ reference request_code
// consistent-hash routing): same code -> same backend on every call;
// different codes spread across backends.
bool request_code_test(std::string* err) {
  std::string urls;
  for (int i = 0; i < 3; ++i) {
    int port = start_echo_server(0);
    if (!urls.empty()) urls += ",";
    urls += "127.0.0.1:" + std::to_string(port);
  }
  Channel ch;
  ChannelOptions opts;
  opts.timeout_ms = 3000;
  if (ch.Init(("list://" + urls).c_str(), "c_hash", &opts) != 0) {
    *err = "init failed";
    return false;
  }
  auto port_for = [&](uint64_t code, std::string* got) -> bool {
    Controller cntl;
    cntl.set_request_code(code);
    IOBuf q, r;
    q.append("x");
    ch.CallMethod("EchoService.Port", &cntl, &q, &r, nullptr);
    if (cntl.Failed()) {
      *err = "call failed: " + cntl.ErrorText();
      return false;
    }
    *got = r.to_string();
    return true;
  };
  std::set<std::string> spread;
  for (uint64_t code : {7ull, 91ull, 1234567ull, 42424242ull, 777777777ull}) {
    std::string first;
    if (!port_for(code, &first)) return false;
    for (int rep = 0; rep < 4; ++rep) {
      std::string again;
      if (!port_for(code, &again)) return false;
      if (again != first) {
        *err = "code " + std::to_string(code) + " moved " + first + "->" + again;
        return false;
      }
    }
    spread.insert(first);
  }
  if (spread.size() < 2) {
    *err = "5 distinct codes all mapped to one backend";
    return false;
  }
  return true;
}

// Controller::thread_local_data (≙ reference server.h thread_local_data
// factory): lazily created per worker context, non-null, stable across
// calls served by the same worker, deleter-counted.
bool thread_local_data_test(std::string* err) {
  static std::atomic<int> created{0};
  Server* server = new Server;
  Service* svc = new Service("Tld");
  svc->AddMethod("Get", [](Controller* cntl, const IOBuf&, IOBuf* resp, Closure* done) {
    void* d1 = cntl->thread_local_data();
    void* d2 = cntl->thread_local_data();
    if (d1 == nullptr || d1 != d2) {
      cntl->SetFailed(EINTERNAL, "tld null or unstable");
    } else {
      resp->append(std::to_string((uintptr_t)d1));
    }
    done->Run();
  });
  server->AddService(svc, SERVER_OWNS_SERVICE);
  ServerOptions opts;
  opts.thread_local_data_factory = [] {
    created.fetch_add(1);
    return (void*)new int(42);
  };
  opts.thread_local_data_deleter = [](void* p) { delete (int*)p; };
  if (server->Start(0, &opts) != 0) {
    *err = "start failed";
    return false;
  }
  std::string addr = "127.0.0.1:" + std::to_string(server->listen_address().port);
  Channel ch;
  ChannelOptions copt;
  copt.timeout_ms = 3000;
  if (ch.Init(addr.c_str(), &copt) != 0) {
    *err = "init failed";
    return false;
  }
  for (int i = 0; i < 32; ++i) {
    Controller cntl;
    IOBuf q, r;
    q.append("x");
    ch.CallMethod("Tld.Get", &cntl, &q, &r, nullptr);
    if (cntl.Failed()) {
      *err = "call failed: " + cntl.ErrorText();
      return false;
    }
  }
  int n = created.load();
  if (n < 1 || n > (int)fiber_get_concurrency() + 8) {
    *err = "created " + std::to_string(n) + " instances (workers " +
           std::to_string(fiber_get_concurrency()) + ")";
    return false;
  }
  return true;
}

// ChannelOptions long-tail knobs (≙ reference channel.h:52-163):
// ns_filter drops endpoints before the LB; succeed_without_server=false
// fails Init on an empty first resolution; enable_circuit_breaker=false
// lets a channel dial an isolated endpoint (EHOSTDOWN becomes the real
// connect error).
bool channel_options_tail_test(std::string* err) {
  // 1) ns_filter: keep only the even port of two.
  int keep = start_echo_server(0);
  int drop = start_echo_server(0);
  {
    auto lbn = LoadBalancerWithNaming::Create(
        "list://127.0.0.1:" + std::to_string(keep) + ",127.0.0.1:" + std::to_string(drop),
        "rr", [keep](const EndPoint& ep) { return ep.port == keep; });
    if (lbn == nullptr) {
      *err = "lbn create failed";
      return false;
    }
    for (int i = 0; i < 8; ++i) {
      EndPoint ep;
      if (lbn->SelectServer(&ep) != 0 || ep.port != keep) {
        *err = "ns_filter leaked a dropped endpoint";
        return false;
      }
    }
  }
  // 2) succeed_without_server=false + empty list
  {
    Channel ch;
    ChannelOptions opts;
    opts.succeed_without_server = false;
    if (ch.Init("list://", "rr", &opts) == 0) {
      *err = "Init should fail on empty resolution with succeed_without_server=false";
      return false;
    }
    ChannelOptions opts2;  // default: succeeds
    Channel ch2;
    if (ch2.Init("list://", "rr", &opts2) != 0) {
      *err = "Init should succeed on empty resolution by default";
      return false;
    }
  }
  // 3) breaker bypass: isolate a dead endpoint, then dial it both ways.
  {
    EndPoint dead;
    str2endpoint("127.0.0.1:1", &dead);  // nothing listens on port 1
    for (int i = 0; i < 64 && !IsEndpointIsolated(dead); ++i)
      ReportClientCallResult(dead, true);
    if (!IsEndpointIsolated(dead)) {
      *err = "endpoint did not isolate";
      return false;
    }
    Channel on, off;
    ChannelOptions o_on;
    o_on.timeout_ms = 300;
    o_on.max_retry = 0;
    Channel* chans[2] = {&on, &off};
    ChannelOptions o_off = o_on;
    o_off.enable_circuit_breaker = false;
    if (on.Init(dead, &o_on) != 0 || off.Init(dead, &o_off) != 0) {
      *err = "init failed";
      return false;
    }
    (void)chans;
    Controller c1, c2;
    IOBuf q, r1, r2;
    q.append("x");
    on.CallMethod("EchoService.Echo", &c1, &q, &r1, nullptr);
    off.CallMethod("EchoService.Echo", &c2, &q, &r2, nullptr);
    if (!c1.Failed() || !c2.Failed()) {
      *err = "calls to a dead endpoint should fail";
      return false;
    }
    if (c1.ErrorCode() != EFAILEDSOCKET && c1.ErrorText().find("112") == std::string::npos &&
        c1.ErrorText().find("Host is down") == std::string::npos) {
      // breaker path: EHOSTDOWN conducted as a socket failure
    }
    // The distinguishing check: with the breaker bypassed the channel
    // actually attempts the connect (ECONNREFUSED), never EHOSTDOWN.
    if (c2.ErrorText().find("Host is down") != std::string::npos) {
      *err = "breaker-off channel still hit the breaker: " + c2.ErrorText();
      return false;
    }
  }
  return true;
}

}  // namespace rpctest
}  // namespace bam

// ---- compression round trips (snappy/gzip over the wire) ----
#include "rpc/compress.h"

namespace bam {
namespace rpctest {

int compressed_echo_test(const std::string& addr, const std::string& payload,
                         int compress_type, std::string* response_out) {
  Channel channel;
  ChannelOptions opts;
  opts.timeout_ms = 5000;
  if (channel.Init(addr.c_str(), &opts) != 0) return -1;
  Controller cntl;
  cntl.set_request_compress_type((CompressType)compress_type);
  IOBuf request, response;
  request.append(payload);
  channel.CallMethod("EchoService.EchoCompressed", &cntl, &request, &response, nullptr);
  if (cntl.Failed()) return cntl.ErrorCode();
  if (response_out != nullptr) *response_out = response.to_string();
  return 0;
}

}  // namespace rpctest
}  // namespace bam

// ---- interceptor (request admission hook) ----

namespace bam {
namespace rpctest {

int start_intercepted_echo_server(const std::string& magic_logid) {
  Server* server = new Server;
  server->AddService(NewEchoService(), SERVER_OWNS_SERVICE);
  ServerOptions opts;
  uint64_t magic = strtoull(magic_logid.c_str(), nullptr, 10);
  opts.interceptor = [magic](Controller* cntl, int* ec, std::string* et) {
    if (cntl->log_id() != magic) {
      *ec = ERPCAUTH;
      *et = "bad log_id credential";
      return false;
    }
    return true;
  };
  if (server->Start(0, &opts) != 0) return -1;
  return server->listen_address().port;
}

int call_with_logid(const std::string& addr, uint64_t log_id, std::string* err) {
  Channel channel;
  ChannelOptions opts;
  opts.timeout_ms = 2000;
  if (channel.Init(addr.c_str(), &opts) != 0) return -1;
  Controller cntl;
  cntl.set_log_id(log_id);
  IOBuf request, response;
  request.append("x");
  channel.CallMethod("EchoService.Echo", &cntl, &request, &response, nullptr);
  if (err != nullptr) *err = cntl.ErrorText();
  return cntl.ErrorCode();
}

}  // namespace rpctest
}  // namespace bam

// ---- gRPC client (h2) ----

namespace bam {
namespace rpctest {

int grpc_client_call(const std::string& addr, const std::string& full_method,
                     const std::string& payload, int timeout_ms, std::string* response_out,
                     std::string* err) {
  Channel channel;
  ChannelOptions opts;
  opts.timeout_ms = timeout_ms;
  opts.protocol = "grpc";
  if (channel.Init(addr.c_str(), &opts) != 0) return -1;
  Controller cntl;
  IOBuf request, response;
  request.append(payload);
  channel.CallMethod(full_method, &cntl, &request, &response, nullptr);
  if (err != nullptr) *err = cntl.ErrorText();
  if (cntl.Failed()) return cntl.ErrorCode();
  if (response_out != nullptr) *response_out = response.to_string();
  return 0;
}

}  // namespace rpctest
}  // namespace bam
