#include "rpc/channel.h"

#include "rpc/authenticator.h"

#include <string.h>

#include "base/fast_rand.h"
#include "base/logging.h"
#include "base/time.h"
#include "rpc/load_balancer.h"
#include "rpc/policy/http_protocol.h"
#include "rpc/policy/std_protocol.h"
#include "rpc/redis.h"
#include "rpc/rpcz.h"
#include "rpc/socket_map.h"

namespace bam {

Channel::~Channel() {}

int Channel::Init(EndPoint ep, const ChannelOptions* options) {
  if (options != nullptr) options_ = *options;
  server_ep_ = ep;
  single_server_ = true;
  policy::RegisterStdProtocol();
  if (options_.protocol == "redis") policy::RegisterRedisProtocol();
  if (options_.protocol == "thrift") policy::RegisterThriftProtocol();
  if (options_.protocol == "grpc" || options_.protocol == "h2") policy::RegisterGrpcClientProtocol();
  if (options_.protocol == "http") policy::RegisterHttpProtocol();
  if (options_.protocol == "hulu_pbrpc") policy::RegisterHuluProtocol();
  if (options_.protocol == "sofa_pbrpc") policy::RegisterSofaProtocol();
  if (options_.protocol == "nshead") policy::RegisterNsheadProtocol();
  if (options_.protocol == "esp") policy::RegisterEspProtocol();
  if (options_.protocol == "nova_pbrpc") policy::RegisterNovaProtocol();
  if (options_.protocol == "ubrpc") policy::RegisterUbrpcProtocol();
  if (options_.protocol == "public_pbrpc") policy::RegisterPublicPbrpcProtocol();
  protocol_index_ = FindClientProtocolIndex(options_.protocol.empty() ? "std" : options_.protocol);
  if (protocol_index_ < 0) return -1;
  return 0;
}

int Channel::Init(const char* server_addr, const ChannelOptions* options) {
  EndPoint ep;
  if (str2endpoint(server_addr, &ep) != 0) {
    // maybe a naming URL used with the 2-arg Init by mistake
    return -1;
  }
  return Init(ep, options);
}

int Channel::Init(const char* naming_url, const char* lb_name, const ChannelOptions* options) {
  if (options != nullptr) options_ = *options;
  single_server_ = false;
  policy::RegisterStdProtocol();
  protocol_index_ = FindClientProtocolIndex(options_.protocol.empty() ? "std" : options_.protocol);
  if (protocol_index_ < 0) return -1;
  lb_ = LoadBalancerWithNaming::Create(naming_url, lb_name, options_.ns_filter);
  if (lb_ == nullptr) return -1;
  if (!options_.succeed_without_server) {
    EndPoint probe;
    if (lb_->SelectServer(&probe) != 0) {
      LOG(ERROR) << "no server resolved from " << naming_url
                 << " and succeed_without_server=false";
      lb_.reset();
      return -1;
    }
  }
  return 0;
}

// ---------------- call machinery ----------------

static void split_full_method(const std::string& full, std::string* service,
                              std::string* method) {
  // Absolute paths ("/v1/x/y.txt", http-ish protocols) stay whole —
  // splitting on the last '.'/'/' would mangle dotted file components.
  if (!full.empty() && full[0] == '/') {
    *service = full;
    method->clear();
    return;
  }
  size_t pos = full.find_last_of("./");
  if (pos == std::string::npos) {
    *service = "";
    *method = full;
  } else {
    *service = full.substr(0, pos);
    *method = full.substr(pos + 1);
  }
}

// Completes the RPC: cancels the timeout timer, destroys the (locked)
// session and runs done.
void EndRPC(Controller* cntl, SessionId locked_id) {
  cntl->end_us_ = monotonic_time_us();
  if (cntl->call.short_socket != 0) {
    SocketUniquePtr s;
    if (Socket::Address(cntl->call.short_socket, &s) == 0) {
      s->SetFailed(0, "short connection done");
    }
    cntl->call.short_socket = 0;
  }
  if (rpcz::enabled()) {
    rpcz::Span span;
    span.start_us = cntl->start_us_;
    span.end_us = cntl->end_us_;
    span.full_method = cntl->call.service_name + "." + cntl->call.method_name;
    span.remote = cntl->remote_side_;
    span.error_code = cntl->error_code_;
    span.log_id = cntl->log_id_;
    span.trace_id = cntl->trace_id_;
    span.span_id = cntl->span_id_;
    span.parent_span_id = cntl->parent_span_id_;
    span.server_side = false;
    span.request_size = cntl->call.request_buf.size();
    rpcz::RecordSpan(span);
  }
  if (cntl->call.timeout_timer != 0) {
    timer_delete(cntl->call.timeout_timer);
    cntl->call.timeout_timer = 0;
  }
  if (cntl->call.lb != nullptr) {
    cntl->call.lb->Feedback(cntl->call.server_ep, cntl->error_code_,
                            cntl->end_us_ - cntl->start_us_);
  }
  if (cntl->remote_side_.port != 0) {
    const int ec = cntl->error_code_;
    const bool network_error = ec != 0 && (ec < 1000 || ec == EFAILEDSOCKET);
    ReportClientCallResult(cntl->remote_side_, network_error);
  }
  Closure* done = cntl->call.done;
  session_unlock_and_destroy(locked_id);
  if (done != nullptr) done->Run();
}

// on_error handler: runs LOCKED; must unlock or destroy.
static int HandleSessionError(SessionId id, void* data, int error_code) {
  Controller* cntl = (Controller*)data;
  if (error_code == EBACKUPREQUEST) {
    // Backup-request trigger: launch a second attempt; the earlier one
    // stays pending and whichever response arrives first wins.
    if (cntl->retry_count_ < cntl->max_retry_) {
      ++cntl->retry_count_;
      session_bump_slot(id);
      IssueRPC(cntl);
    }
    session_unlock(id);
    return 0;
  }
  if (!session_is_current(id)) {
    // failure of a stale attempt (e.g. old socket died after retry)
    session_unlock(id);
    return 0;
  }
  // Cancellation is unconditionally terminal (parity: reference StartCancel
  // semantics) — a user retry_policy must not resurrect a call the caller
  // just canceled, so it is short-circuited before the policy hook.
  const bool policy_says_retry =
      error_code != ECANCELED_RPC &&
      (cntl->call.retry_policy
           ? cntl->call.retry_policy(error_code, cntl->retry_count_)
           : error_code != ERPCTIMEDOUT);
  if (policy_says_retry && cntl->retry_count_ < cntl->max_retry_) {
    ++cntl->retry_count_;
    session_bump_slot(id);
    if (cntl->call.lb != nullptr && cntl->call.server_ep.port != 0) {
      cntl->call.lb->Feedback(cntl->call.server_ep, error_code, 0);
    }
    IssueRPC(cntl);  // synchronous failures enqueue on the session (no deadlock)
    session_unlock(id);
    return 0;
  }
  cntl->SetFailed(error_code, error_code == ERPCTIMEDOUT
                                  ? "RPC deadline exceeded"
                                  : std::string("socket error: ") + strerror(error_code));
  EndRPC(cntl, id);
  return 0;
}

static void TimeoutCb(void* a, void* /*b*/) {
  SessionId cid = (SessionId)(uintptr_t)a;
  session_error(cid, ERPCTIMEDOUT);
}

static void BackupRequestCb(void* a, void* /*b*/) {
  SessionId cid = (SessionId)(uintptr_t)a;
  session_error(cid, EBACKUPREQUEST);
}

void IssueRPC(Controller* cntl) {
  // Select a server.
  EndPoint ep = cntl->call.server_ep;
  if (cntl->call.lb != nullptr) {
    int sel_rc = cntl->has_request_code()
                     ? cntl->call.lb->SelectServerByCode(cntl->request_code(), &ep)
                     : cntl->call.lb->SelectServer(&ep);
    if (sel_rc != 0) {
      session_error(session_current_id(cntl->call.cid), EFAILEDSOCKET);
      return;
    }
    cntl->call.server_ep = ep;
  }
  SocketUniquePtr sock;
  // Fast path: the channel's cached socket (skips the global map mutex).
  std::atomic<uint64_t>* cache = cntl->call.short_conn ? nullptr : cntl->call.socket_cache;
  if (cache != nullptr) {
    uint64_t sid = cache->load(std::memory_order_acquire);
    if (sid != 0 && (Socket::Address(sid, &sock) != 0 || sock->Failed())) {
      sock.reset(nullptr);
    }
  }
  if (!sock) {
    if (GetClientSocket(ep, &sock, cntl->call.connection_shard, cntl->call.use_breaker,
                        cntl->call.ssl, cntl->call.protocol_index,
                        cntl->call.socket_mode, cntl->call.short_conn) != 0) {
      // Conduct the failure through the session so retry/ending logic runs.
      session_error(session_current_id(cntl->call.cid), EFAILEDSOCKET);
      return;
    }
    if (cntl->call.short_conn) {
      // A retry replaces the socket: close the previous attempt's now.
      if (cntl->call.short_socket != 0) {
        SocketUniquePtr prev;
        if (Socket::Address(cntl->call.short_socket, &prev) == 0) {
          prev->SetFailed(0, "short connection superseded by retry");
        }
      }
      cntl->call.short_socket = sock->id();
    } else if (cache != nullptr) {
      cache->store(sock->id(), std::memory_order_release);
    }
  }
  cntl->remote_side_ = ep;
  cntl->local_side_ = sock->local_side();
  cntl->call.auth_data.clear();
  if (cntl->call.auth != nullptr) {
    if (cntl->call.auth->GenerateCredential(&cntl->call.auth_data) != 0) {
      session_error(session_current_id(cntl->call.cid), ERPCAUTH);
      return;
    }
  }
  SessionId current = session_current_id(cntl->call.cid);
  const Protocol* proto = GetProtocol(cntl->call.protocol_index);
  if (proto != nullptr && proto->issue_request != nullptr) {
    sock->add_pending_session(current);
    cntl->call.pending_socket = sock->id();
    if (proto->issue_request(sock.get(), cntl, current) != 0) {
      session_error(current, EFAILEDSOCKET);
    }
    return;
  }
  IOBuf packet;
  if (proto != nullptr && proto->pack_request != nullptr) {
    proto->pack_request(&packet, cntl, current);
  } else {
    policy::PackStdRequest(&packet, cntl, current);
  }
  sock->add_pending_session(current);
  if (proto != nullptr && proto->client_pipelined) sock->push_pipeline(current);
  cntl->call.pending_socket = sock->id();
  Socket::WriteOptions wo;
  wo.id_wait = current;
  sock->Write(&packet, &wo);  // failure conducts through the session
}

void Channel::CallMethod(const std::string& full_method, Controller* cntl,
                         const IOBuf* request, IOBuf* response, Closure* done) {
  cntl->start_us_ = monotonic_time_us();
  if (rpcz::enabled()) {
    // Inherit the ambient trace (server handler context) or start a new
    // one; every client call gets a fresh span id (≙ reference
    // channel.cpp:524 CreateClientSpan).
    if (cntl->trace_id_ == 0) {
      rpcz::TraceContext tc = rpcz::current_trace();
      if (tc.trace_id != 0) {
        cntl->trace_id_ = tc.trace_id;
        cntl->parent_span_id_ = tc.span_id;
      } else {
        cntl->trace_id_ = fast_rand() | 1;
      }
    }
    if (cntl->span_id_ == 0) cntl->span_id_ = fast_rand() | 1;
  }
  if (cntl->timeout_ms_ == -1) cntl->timeout_ms_ = options_.timeout_ms;
  if (cntl->max_retry_ == 3 /*default*/) cntl->max_retry_ = options_.max_retry;
  split_full_method(full_method, &cntl->call.service_name, &cntl->call.method_name);
  if (request != nullptr) cntl->call.request_buf = *request;  // zero-copy ref share
  cntl->call.response = response;
  cntl->call.done = done;
  cntl->call.protocol_index = protocol_index_;
  cntl->call.auth = options_.auth;
  cntl->call.use_breaker = options_.enable_circuit_breaker;
  cntl->call.ssl = options_.ssl;
  cntl->call.socket_mode = options_.socket_mode.empty() ? nullptr : options_.socket_mode.c_str();
  cntl->call.retry_policy = options_.retry_policy;
  cntl->call.short_conn = options_.connection_type == "short";
  if (options_.connection_type == "pooled") {
    static std::atomic<uint32_t> rr{0};
    cntl->call.connection_shard =
        1 + (int)(rr.fetch_add(1, std::memory_order_relaxed) %
                  (uint32_t)(options_.connection_pool_size > 0 ? options_.connection_pool_size
                                                               : 8));
  }
  if (single_server_) {
    cntl->call.server_ep = server_ep_;
    cntl->call.lb = nullptr;
    if (cntl->call.connection_shard == 0) cntl->call.socket_cache = &cached_socket_;
  } else {
    cntl->call.lb = lb_.get();
  }

  SessionId cid;
  int rc = session_create(&cid, cntl, HandleSessionError, cntl->max_retry_ + 1);
  if (rc != 0) {
    cntl->SetFailed(EINTERNAL, "session_create failed");
    if (done) done->Run();
    return;
  }
  cntl->cid_ = cid;
  cntl->call.cid = cid;

  if (cntl->backup_request_ms_ == -1) cntl->backup_request_ms_ = options_.backup_request_ms;
  if (cntl->timeout_ms_ > 0) {
    cntl->call.timeout_timer = timer_add(cntl->start_us_ + cntl->timeout_ms_ * 1000,
                                         TimeoutCb, (void*)(uintptr_t)cid, nullptr);
  }
  if (cntl->backup_request_ms_ > 0 &&
      (cntl->timeout_ms_ <= 0 || cntl->backup_request_ms_ < cntl->timeout_ms_)) {
    timer_add(cntl->start_us_ + cntl->backup_request_ms_ * 1000, BackupRequestCb,
              (void*)(uintptr_t)cid, nullptr);
  }
  IssueRPC(cntl);
  if (done == nullptr) {
    session_join(cid);
  }
}

}  // namespace bam
