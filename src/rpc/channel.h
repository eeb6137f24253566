// brpc_amd: Channel — the client-side call endpoint.
// API parity: reference brpc/channel.h (Init with "ip:port" or naming URL +
// load-balancer name, ChannelOptions{timeout_ms, connect_timeout_ms,
// max_retry, protocol}, CallMethod). The combo channels (Parallel /
// Selective / Partition) layer on the same CallMethod contract.
#pragma once

#include <memory>
#include <string>

#include "rpc/controller.h"

namespace bam {

class LoadBalancerWithNaming;

struct ChannelOptions {
  int32_t connect_timeout_ms = 200;
  int32_t timeout_ms = 500;
  int32_t backup_request_ms = -1;
  int max_retry = 3;
  std::string protocol = "std";
  // "single" (default) or "pooled" (N connections round-robined; the
  // reference's multi-connection mode — its highest-throughput config).
  std::string connection_type = "single";
  int connection_pool_size = 8;
  std::string connection_group;
  // Byte transport: "" = TCP (default), "rdma_mock" = the RDMA endpoint
  // machinery over the in-process mock provider (rpc/rdma_transport.h);
  // a verbs provider slots in the same way on RDMA-capable hosts.
  std::string socket_mode;
  // Client credential source (rpc/authenticator.h); not owned. When set,
  // every request carries RpcMeta.authentication_data.
  const class Authenticator* auth = nullptr;
  // TLS to the server (certificate verification off, like the reference's
  // default ChannelSSLOptions).
  bool ssl = false;
  // Custom retry decision (parity: reference brpc/retry_policy.h
  // RetryPolicy::DoRetry): called with the failing error_code and the
  // attempt index (0-based); return false to stop retrying. nullptr =
  // default policy (retry everything but deadline errors up to max_retry).
  std::function<bool(int error_code, int attempt)> retry_policy;
  // Consult the per-endpoint circuit breaker before picking a socket
  // (parity: reference ChannelOptions.enable_circuit_breaker). false =
  // this channel keeps dialing isolated endpoints (probes, admin tools).
  bool enable_circuit_breaker = true;
  // Naming-service Init(naming_url) succeeds even when the first
  // resolution returns no servers (parity: reference
  // ChannelOptions.succeed_without_server, default on) — calls fail with
  // EHOSTDOWN until servers appear. false = Init fails instead.
  bool succeed_without_server = true;
  // Filters resolved endpoints before they reach the load balancer
  // (parity: reference ns_filter). Return false to drop a server.
  std::function<bool(const EndPoint&)> ns_filter;
};

class ChannelBase {
 public:
  virtual ~ChannelBase() {}
  // full_method = "ServiceName.MethodName" (or "ServiceName/Method").
  virtual void CallMethod(const std::string& full_method, Controller* cntl,
                          const IOBuf* request, IOBuf* response, Closure* done) = 0;
};

class Channel : public ChannelBase {
 public:
  Channel() {}
  ~Channel() override;

  // "ip:port" / "host:port" single server, or "list://h1:p1,h2:p2" /
  // "file://path" with a load balancer name ("rr", "random", "c_hash",
  // "la", "p2c").
  int Init(const char* server_addr, const ChannelOptions* options);
  int Init(const char* naming_url, const char* lb_name, const ChannelOptions* options);
  int Init(EndPoint ep, const ChannelOptions* options);

  void CallMethod(const std::string& full_method, Controller* cntl, const IOBuf* request,
                  IOBuf* response, Closure* done) override;

  const ChannelOptions& options() const { return options_; }

 private:
  ChannelOptions options_;
  EndPoint server_ep_;
  bool single_server_ = false;
  int protocol_index_ = -1;  // resolved from options_.protocol
  std::atomic<uint64_t> cached_socket_{0};  // single-server fast path
  std::shared_ptr<LoadBalancerWithNaming> lb_;
  friend void IssueRPC(Controller*);
};

// Starts the RPC described by cntl->call (already filled). Used by Channel
// and by the retry path in HandleSessionError.
void IssueRPC(Controller* cntl);

}  // namespace bam
