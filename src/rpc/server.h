// brpc_amd: Server — accepts connections, routes parsed requests to
// registered service methods running in fibers.
// API parity: reference brpc/server.h (AddService, Start(port, opts),
// Stop/Join, ServerOptions{num_threads, max_concurrency, idle_timeout}).
#pragma once

#include <atomic>
#include <functional>
#include <map>
#include <memory>
#include <string>
#include <tuple>
#include <vector>

#include "base/endpoint.h"
#include "rpc/controller.h"
#include "rpc/input_messenger.h"
#include "var/variable.h"

namespace bam {

// A service method: fills *response (and/or cntl error state) then MUST
// call done->Run() exactly once (may be after returning, for async).
typedef std::function<void(Controller* cntl, const IOBuf& request, IOBuf* response,
                           Closure* done)>
    MethodFn;

class Service {
 public:
  explicit Service(std::string name) : name_(std::move(name)) {}
  virtual ~Service() {}
  const std::string& name() const { return name_; }
  void AddMethod(const std::string& method, MethodFn fn) { methods_[method] = std::move(fn); }
  const MethodFn* FindMethod(const std::string& method) const {
    auto it = methods_.find(method);
    return it == methods_.end() ? nullptr : &it->second;
  }
  const std::map<std::string, MethodFn>& methods() const { return methods_; }

 private:
  std::string name_;
  std::map<std::string, MethodFn> methods_;
};

enum ServiceOwnership { SERVER_OWNS_SERVICE, SERVER_DOESNT_OWN_SERVICE };

class RedisService;

// Mongo wire support (parity: reference mongo_service_adaptor.h — the
// framework frames the 16-byte mongo header and hands the RAW body to the
// user handler; BSON interpretation is the user's business).
struct MongoHeader {
  int32_t message_length = 0;
  int32_t request_id = 0;
  int32_t response_to = 0;
  int32_t op_code = 0;
};
struct MongoReply {
  IOBuf body;               // reply documents / OP_MSG sections (raw bytes)
  int32_t response_flags = 0;
  int64_t cursor_id = 0;
  int32_t number_returned = 1;
  bool always_reply = false;  // reply even with an empty body
};
typedef std::function<void(const MongoHeader& head, const IOBuf& body, MongoReply* reply)>
    MongoHandlerFn;

// Runs before every request handler (parity: reference brpc/interceptor.h).
// Return false to reject; fill *error_code/*error_text.
typedef std::function<bool(Controller* cntl, int* error_code, std::string* error_text)>
    InterceptorFn;

struct ServerOptions {
  int idle_timeout_sec = -1;
  // "" = TCP; "rdma_mock" upgrades every accepted connection to the RDMA
  // endpoint (mock provider; parity: reference ServerOptions.socket_mode).
  std::string socket_mode;
  int max_concurrency = 0;          // 0 = unlimited
  bool has_builtin_services = true;
  RedisService* redis_service = nullptr;  // serve RESP on the same port
  InterceptorFn interceptor;              // request admission hook
  // Verifies RpcMeta.authentication_data once per connection
  // (rpc/authenticator.h); not owned. nullptr = no auth.
  const class Authenticator* auth = nullptr;
  // TLS: PEM content (starting with "-----BEGIN") or file paths. Both
  // set => every accepted connection speaks TLS (rpc/ssl_util.h).
  std::string ssl_cert;
  std::string ssl_key;
  // nshead raw-body service (parity: reference ServerOptions::nshead_service):
  // called with the request body; fills the response body.
  std::function<void(const IOBuf& req_body, IOBuf* resp_body)> nshead_handler;
  // mongo service adaptor (see MongoHandlerFn above).
  MongoHandlerFn mongo_handler;
  // RTMP media endpoint (policy/rtmp_protocol.cc): handshake + chunk
  // streams + connect/createStream/publish/play and a built-in
  // publish->play relay hub.
  bool enable_rtmp = false;
  // Adaptive concurrency (rpc/concurrency_limiter.h): "" | "auto" |
  // "timeout:<ms>" | "<n>". Applies on top of max_concurrency.
  std::string adaptive_max_concurrency;
  // Per-connection user state (parity: reference
  // ServerOptions::session_local_data_factory): created lazily on first
  // Controller::session_local_data() of a connection, destroyed when the
  // connection recycles.
  std::function<void*()> session_local_data_factory;
  std::function<void(void*)> session_local_data_deleter;
  // Per-worker user state (parity: reference
  // ServerOptions::thread_local_data_factory + reserved count): created
  // lazily on first Controller::thread_local_data() in each fiber/thread
  // that runs this server's handlers; destroyed at fiber exit.
  std::function<void*()> thread_local_data_factory;
  std::function<void(void*)> thread_local_data_deleter;
  // Generic/proxy catch-all (parity: reference BaiduMasterService,
  // brpc/baidu_master_service.h:36-73): when a baidu_std request names a
  // service/method this server does not register, the master handler —
  // if set — receives the raw serialized request instead of ENOMETHOD.
  // cntl->call.service_name/method_name carry the original names.
  MethodFn master_handler;
};

class MethodStatusRecorder;  // var/latency recorder per method (var layer)

class Server {
 public:
  Server();
  ~Server();

  int AddService(Service* service, ServiceOwnership ownership);
  // Restful URL mapping (parity: reference AddService(..., restful_mappings)
  // "/v1/echo => Echo, /store/* => Put"): maps exact paths or trailing-*
  // prefixes onto this service's methods for the HTTP/h2 protocols.
  int AddService(Service* service, ServiceOwnership ownership,
                 const std::string& restful_mappings);
  // Adds restful mappings for a service already registered via AddService.
  int AddServiceRestfulOnly(Service* service, const std::string& restful_mappings);
  // Resolves a restful path; false if unmapped.
  bool MapRestfulPath(const std::string& path, std::string* service,
                      std::string* method) const;
  // Per-method concurrency cap (parity: reference method_max_concurrency /
  // MethodStatus gate). full_method = "Service.Method".
  void SetMethodMaxConcurrency(const std::string& full_method, int32_t limit);
  // Admission for one method; returns false (ELIMIT) when capped. Paired
  // with EndMethod from the response path.
  bool BeginMethod(const std::string& service, const std::string& method);
  void EndMethod(const std::string& service, const std::string& method);
  bool has_method_gates() const { return gate_count_.load(std::memory_order_acquire) > 0; }
  int Start(int port, const ServerOptions* opt);  // port 0 = pick free port
  int Start(const EndPoint& ep, const ServerOptions* opt);
  int Stop(int wait_ms = 0);
  int Join();
  // thread_local_data plumbing (≙ reference keytable pool, server.cpp:934:
  // a finished worker context returns its data for the next request to
  // reuse — data is NOT reset between borrowings, same as bthread_local).
  void* BorrowTld();
  void ReturnTld(void* data);
  // Blocks until SIGINT/SIGTERM (≙ reference server.cpp:1895
  // RunUntilAskedToQuit + IsAskedToQuit): installs the quit handler on
  // first use, sleeps in 100 ms ticks, then Stop()+Join().
  void RunUntilAskedToQuit();
  static bool IsAskedToQuit();

  bool IsRunning() const { return running_.load(std::memory_order_acquire); }
  EndPoint listen_address() const { return listen_ep_; }
  SocketId listen_socket_id() const { return listen_socket_; }

  // Request routing (called by protocol ProcessRequest).
  const MethodFn* FindMethod(const std::string& service, const std::string& method,
                             Service** svc_out = nullptr) const;

  // Per-method latency recorder (parity: reference details/method_status.h).
  var::LatencyRecorder* method_status(const std::string& service, const std::string& method);

  // stats
  std::atomic<int64_t> nprocessed{0};
  std::atomic<int32_t> concurrency{0};
  int max_concurrency() const { return options_.max_concurrency; }
  const ServerOptions& options() const { return options_; }
  class ConcurrencyLimiter* limiter() const { return limiter_; }
  InputMessenger* messenger() { return &messenger_; }
  RedisService* redis_service() const { return options_.redis_service; }

  const std::map<std::string, Service*>& services() const { return services_; }

 private:
  static void OnNewConnections(Socket* listen_socket);

  std::map<std::string, Service*> services_;
  std::map<Service*, ServiceOwnership> ownership_;
  std::mutex status_mu_;
  std::map<std::string, var::LatencyRecorder*> method_status_;
  std::map<std::string, std::pair<std::string, std::string>> restful_exact_;
  std::vector<std::tuple<std::string, std::string, std::string>> restful_prefix_;
  struct MethodGate {
    std::atomic<int32_t> current{0};
    int32_t max = 0;
  };
  mutable std::mutex gates_mu_;
  std::map<std::string, MethodGate*> method_gates_;
  std::atomic<int> gate_count_{0};
  ServerOptions options_;
  EndPoint listen_ep_;
  SocketId listen_socket_ = 0;
  std::atomic<bool> running_{false};
  std::mutex tld_mu_;
  std::vector<void*> tld_pool_;
  InputMessenger messenger_;
  void* ssl_ctx_ = nullptr;  // SSL_CTX* when TLS enabled (never freed: sockets may outlive Stop)
  class ConcurrencyLimiter* limiter_ = nullptr;  // never freed (sockets may outlive Stop)
};

}  // namespace bam
