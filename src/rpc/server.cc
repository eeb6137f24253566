#include "rpc/server.h"

#include <signal.h>
#include <string.h>

#include <algorithm>
#include <mutex>

#include "base/time.h"
#include "fiber/fiber.h"
#include "rpc/ssl_util.h"
#include "fiber/butex.h"
#include "fiber/gpu_wait.h"
#include "rpc/concurrency_limiter.h"

#include <sys/socket.h>
#include <unistd.h>

#include "base/logging.h"
#include "rpc/event_dispatcher.h"
#include "rpc/rdma_transport.h"
#include "rpc/policy/std_protocol.h"
#include "rpc/policy/http_protocol.h"
#include "rpc/redis.h"

namespace bam {

Server::Server() : messenger_(/*server_side=*/true) {}

Server::~Server() {
  Stop(0);
  for (auto& kv : ownership_) {
    if (kv.second == SERVER_OWNS_SERVICE) delete kv.first;
  }
}

int Server::AddService(Service* service, ServiceOwnership ownership) {
  if (service == nullptr || IsRunning()) return -1;
  if (services_.count(service->name()) != 0) return -1;
  services_[service->name()] = service;
  ownership_[service] = ownership;
  return 0;
}

int Server::AddService(Service* service, ServiceOwnership ownership,
                       const std::string& restful_mappings) {
  if (AddService(service, ownership) != 0) return -1;
  return AddServiceRestfulOnly(service, restful_mappings);
}

int Server::AddServiceRestfulOnly(Service* service, const std::string& restful_mappings) {
  // "PATH => Method [, PATH => Method]..."
  size_t pos = 0;
  while (pos < restful_mappings.size()) {
    size_t comma = restful_mappings.find(',', pos);
    std::string item = restful_mappings.substr(
        pos, comma == std::string::npos ? std::string::npos : comma - pos);
    pos = comma == std::string::npos ? restful_mappings.size() : comma + 1;
    size_t arrow = item.find("=>");
    if (arrow == std::string::npos) return -1;
    auto trim = [](std::string x) {
      size_t a = x.find_first_not_of(" \t");
      size_t b = x.find_last_not_of(" \t");
      return a == std::string::npos ? std::string() : x.substr(a, b - a + 1);
    };
    std::string path = trim(item.substr(0, arrow));
    std::string method = trim(item.substr(arrow + 2));
    if (path.empty() || method.empty() || service->FindMethod(method) == nullptr) return -1;
    if (path.size() >= 2 && path.compare(path.size() - 2, 2, "/*") == 0) {
      restful_prefix_.emplace_back(path.substr(0, path.size() - 1), service->name(), method);
    } else {
      restful_exact_[path] = {service->name(), method};
    }
  }
  return 0;
}

bool Server::MapRestfulPath(const std::string& path, std::string* service,
                            std::string* method) const {
  auto it = restful_exact_.find(path);
  if (it != restful_exact_.end()) {
    *service = it->second.first;
    *method = it->second.second;
    return true;
  }
  for (const auto& t : restful_prefix_) {
    const std::string& prefix = std::get<0>(t);
    if (path.compare(0, prefix.size(), prefix) == 0) {
      *service = std::get<1>(t);
      *method = std::get<2>(t);
      return true;
    }
  }
  return false;
}

void Server::SetMethodMaxConcurrency(const std::string& full_method, int32_t limit) {
  std::lock_guard<std::mutex> lk(gates_mu_);
  MethodGate*& g = method_gates_[full_method];
  if (g == nullptr) g = new MethodGate;
  g->max = limit;
  gate_count_.store((int)method_gates_.size(), std::memory_order_release);
}

bool Server::BeginMethod(const std::string& service, const std::string& method) {
  if (!has_method_gates()) return true;  // common case: no per-method caps
  MethodGate* g;
  {
    std::lock_guard<std::mutex> lk(gates_mu_);
    if (method_gates_.empty()) return true;
    auto it = method_gates_.find(service + "." + method);
    if (it == method_gates_.end()) return true;
    g = it->second;
  }
  if (g->max <= 0) return true;
  if (g->current.fetch_add(1, std::memory_order_relaxed) >= g->max) {
    g->current.fetch_sub(1, std::memory_order_relaxed);
    return false;
  }
  return true;
}

void Server::EndMethod(const std::string& service, const std::string& method) {
  if (!has_method_gates()) return;
  MethodGate* g;
  {
    std::lock_guard<std::mutex> lk(gates_mu_);
    if (method_gates_.empty()) return;
    auto it = method_gates_.find(service + "." + method);
    if (it == method_gates_.end()) return;
    g = it->second;
  }
  if (g->max > 0) g->current.fetch_sub(1, std::memory_order_relaxed);
}

const MethodFn* Server::FindMethod(const std::string& service, const std::string& method,
                                   Service** svc_out) const {
  auto it = services_.find(service);
  if (it == services_.end()) {
    // Single-service convenience: empty service name matches the only one.
    if (service.empty() && services_.size() == 1) it = services_.begin();
    else return nullptr;
  }
  if (svc_out != nullptr) *svc_out = it->second;
  return it->second->FindMethod(method);
}

void Server::OnNewConnections(Socket* listen_socket) {
  Server* server = (Server*)listen_socket->user();
  for (;;) {
    int fd = accept4(listen_socket->fd(), nullptr, nullptr, SOCK_NONBLOCK | SOCK_CLOEXEC);
    if (fd < 0) {
      if (errno == EAGAIN || errno == EWOULDBLOCK) return;
      if (errno == EINTR) continue;
      if (errno == EMFILE || errno == ENFILE) {
        LOG(WARNING) << "accept failed: out of fds";
        return;
      }
      return;
    }
    InputMessenger* messenger = server->messenger();
    SocketOptions opts;
    opts.fd = fd;
    opts.user = server;
    opts.on_edge_triggered_events = [messenger](Socket* s) { messenger->OnNewMessages(s); };
    SocketId sid;
    if (Socket::Create(opts, &sid) != 0) {
      ::close(fd);
      continue;
    }
    if (!server->options_.socket_mode.empty()) {
      SocketUniquePtr sp;
      if (Socket::Address(sid, &sp) == 0 &&
          server->options_.socket_mode == "rdma_mock") {
        std::string terr;
        Transport* t = rdma::CreateRdmaTransport(sp.get(), rdma::mock_provider(), 16,
                                                 64 << 10, &terr);
        if (t != nullptr) {
          sp->set_transport(t);
          // The peer may have paired and delivered before the transport
          // was installed — kick the input path once to drain any early data.
          sp->on_input_event();
        } else {
          sp->SetFailed(ECONNRESET, ("rdma transport: " + terr).c_str());
        }
      }
    }
    if (server->ssl_ctx_ != nullptr) {
      SocketUniquePtr sp;
      if (Socket::Address(sid, &sp) == 0) {
        void* ssl = ssl::NewSsl(server->ssl_ctx_, fd, /*client=*/false);
        if (ssl != nullptr) {
          sp->set_ssl(ssl);
        } else {
          sp->SetFailed(ECONNRESET, "SSL_new failed");
        }
      }
    }
  }
}

int Server::Start(int port, const ServerOptions* opt) {
  EndPoint ep;
  hostname2endpoint("0.0.0.0", port, &ep);
  return Start(ep, opt);
}

int Server::Start(const EndPoint& ep, const ServerOptions* opt) {
  if (IsRunning()) return -1;
  if (opt != nullptr) options_ = *opt;
  // process-wide contention vars (≙ contention profiler surface)
  static var::PassiveStatus* g_waits = new var::PassiveStatus(
      "fiber_butex_waits", [] { return std::to_string(butex_total_waits()); });
  static var::PassiveStatus* g_wait_us = new var::PassiveStatus(
      "fiber_butex_wait_us", [] { return std::to_string(butex_total_wait_us()); });
  static var::PassiveStatus* g_gpu_parks = new var::PassiveStatus(
      "gpu_wait_parks", [] { return std::to_string(gpu_wait_parks()); });
  static var::PassiveStatus* g_gpu_wakes = new var::PassiveStatus(
      "gpu_wait_wake_requests", [] { return std::to_string(gpu_wait_wake_requests()); });
  (void)g_waits;
  (void)g_wait_us;
  (void)g_gpu_parks;
  (void)g_gpu_wakes;
  if (limiter_ == nullptr) {
    limiter_ = ConcurrencyLimiter::Create(options_.adaptive_max_concurrency);
  }
  // Idle-connection reaper (≙ reference ServerOptions.idle_timeout_sec,
  // server.h:62): a background fiber closes server-side connections with
  // no reads/writes for idle_timeout_sec. The listening socket never has
  // payload traffic but also never matches (user() != this is skipped;
  // the listen socket is reaped only by Stop()).
  if (options_.idle_timeout_sec > 0) {
    struct ReaperArg {
      Server* server;
      int idle_sec;
    };
    auto* ra = new ReaperArg{this, options_.idle_timeout_sec};
    fiber_t th;
    fiber_start_background(&th, [](void* raw) {
      ReaperArg* a = (ReaperArg*)raw;
      const int64_t idle_us = (int64_t)a->idle_sec * 1000000;
      // sleep FIRST: this fiber starts while Start() is still working and
      // running_ flips true only at its end.
      do {
        fiber_usleep(std::min<int64_t>(idle_us / 2 + 1000, 1000000));
        std::vector<SocketId> ids;
        ListSockets(&ids);
        const int64_t now = monotonic_time_us();
        for (SocketId id : ids) {
          SocketUniquePtr sk;
          if (Socket::Address(id, &sk) != 0) continue;
          if (sk->user() != (void*)a->server) continue;  // not ours / listener path
          if (id == a->server->listen_socket_id()) continue;
          int64_t last = sk->last_active_us.load(std::memory_order_relaxed);
          if (last != 0 && now - last > idle_us) {
            sk->SetFailed(ETIMEDOUT, "idle connection reaped");
          }
        }
      } while (a->server->IsRunning());
      delete a;
    }, ra);
  }
  // gRPC health checking (≙ reference grpc_health_check, brpc/grpc.cpp +
  // builtin registration server.cpp:501): grpc clients probe
  // /grpc.health.v1.Health/Check; reply HealthCheckResponse{status:
  // SERVING} (field 1 varint 1).
  if (options_.has_builtin_services &&
      services_.find("grpc.health.v1.Health") == services_.end()) {
    Service* health = new Service("grpc.health.v1.Health");
    health->AddMethod("Check",
                      [](Controller*, const IOBuf&, IOBuf* resp, Closure* done) {
      resp->append("\x08\x01", 2);
      done->Run();
    });
    health->AddMethod("Watch",
                      [](Controller*, const IOBuf&, IOBuf* resp, Closure* done) {
      resp->append("\x08\x01", 2);
      done->Run();
    });
    AddService(health, SERVER_OWNS_SERVICE);
  }
  if (!options_.ssl_cert.empty() || !options_.ssl_key.empty()) {
    ssl_ctx_ = ssl::NewServerCtx(options_.ssl_cert, options_.ssl_key);
    if (ssl_ctx_ == nullptr) {
      LOG(ERROR) << "server TLS context failed: " << ssl::LastError();
      return -1;
    }
  }
  policy::RegisterStdProtocol();
  policy::RegisterH2Protocol();
  policy::RegisterThriftProtocol();
  policy::RegisterHuluProtocol();
  policy::RegisterSofaProtocol();
  if (options_.nshead_handler) policy::RegisterNsheadProtocol();
  if (options_.mongo_handler) policy::RegisterMongoProtocol();
  if (options_.enable_rtmp) policy::RegisterRtmpProtocol();
  if (options_.redis_service != nullptr) policy::RegisterRedisProtocol();
  int listen_fd = tcp_listen(ep);
  if (listen_fd < 0) {
    LOG(ERROR) << "tcp_listen on " << endpoint2str(ep) << " failed";
    return -1;
  }
  get_local_side(listen_fd, &listen_ep_);
  if (listen_ep_.ip.s_addr == 0) hostname2endpoint("127.0.0.1", listen_ep_.port, &listen_ep_);
  SocketOptions opts;
  opts.fd = listen_fd;
  opts.user = this;
  opts.on_edge_triggered_events = OnNewConnections;
  if (Socket::Create(opts, &listen_socket_) != 0) {
    ::close(listen_fd);
    return -1;
  }
  running_.store(true, std::memory_order_release);
  return 0;
}

int Server::Stop(int /*wait_ms*/) {
  if (!running_.exchange(false, std::memory_order_acq_rel)) return 0;
  SocketUniquePtr s;
  if (Socket::Address(listen_socket_, &s) == 0) {
    s->SetFailed(ELOGOFF, "server stopped");
  }
  return 0;
}

var::LatencyRecorder* Server::method_status(const std::string& service,
                                            const std::string& method) {
  std::string key = service + "." + method;
  std::lock_guard<std::mutex> lk(status_mu_);
  auto it = method_status_.find(key);
  if (it != method_status_.end()) return it->second;
  var::LatencyRecorder* rec = new var::LatencyRecorder;
  rec->expose("rpc_server_" + std::to_string(listen_ep_.port) + "_" + key);
  method_status_[key] = rec;
  return rec;
}

int Server::Join() {
  while (concurrency.load(std::memory_order_acquire) > 0) usleep(1000);
  return 0;
}

void* Server::BorrowTld() {
  {
    std::lock_guard<std::mutex> lk(tld_mu_);
    if (!tld_pool_.empty()) {
      void* d = tld_pool_.back();
      tld_pool_.pop_back();
      return d;
    }
  }
  return options().thread_local_data_factory ? options().thread_local_data_factory()
                                             : nullptr;
}

void Server::ReturnTld(void* data) {
  if (data == nullptr) return;
  std::lock_guard<std::mutex> lk(tld_mu_);
  tld_pool_.push_back(data);
}

static std::atomic<bool> g_asked_to_quit{false};
static void quit_handler(int) { g_asked_to_quit.store(true, std::memory_order_release); }

bool Server::IsAskedToQuit() { return g_asked_to_quit.load(std::memory_order_acquire); }

void Server::RunUntilAskedToQuit() {
  static std::once_flag once;
  std::call_once(once, [] {
    struct sigaction sa;
    memset(&sa, 0, sizeof(sa));
    sa.sa_handler = quit_handler;
    sigaction(SIGINT, &sa, nullptr);
    sigaction(SIGTERM, &sa, nullptr);
  });
  while (!IsAskedToQuit()) usleep(100 * 1000);
  Stop(0);
  Join();
}

}  // namespace bam
