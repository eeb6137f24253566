#include "rpc/socket_map.h"

#include <string.h>

#include "rpc/rdma_transport.h"

#include <errno.h>
#include <poll.h>
#include <unistd.h>

#include <map>
#include <mutex>

#include "base/flags.h"
#include "base/time.h"
#include "fiber/fiber.h"
#include "rpc/input_messenger.h"
#include "rpc/ssl_util.h"

namespace bam {

BAM_DEFINE_bool(enable_circuit_breaker, true,
                "isolate endpoints after repeated connection failures");
BAM_DEFINE_int64(circuit_breaker_max_failures, 3,
                 "consecutive network failures before isolating an endpoint");
BAM_DEFINE_int64(health_check_interval_ms, 500,
                 "re-probe interval for isolated endpoints");

namespace {

InputMessenger* client_messenger() {
  static InputMessenger* m = new InputMessenger(/*server_side=*/false);
  return m;
}

struct ClientSocketMap {
  std::mutex mu;
  std::map<std::pair<EndPoint, int>, SocketId> sockets;
};

ClientSocketMap& the_map() {
  static ClientSocketMap* m = new ClientSocketMap;
  return *m;
}

struct EndpointHealth {
  std::atomic<int> consecutive_errors{0};
  std::atomic<bool> isolated{false};
};

struct HealthRegistry {
  std::mutex mu;
  std::map<EndPoint, EndpointHealth*> map;
};
HealthRegistry& health_registry() {
  static HealthRegistry* r = new HealthRegistry;
  return *r;
}

std::atomic<bool> g_any_errors{false};  // no lookup on success until an error exists

EndpointHealth* health_of(const EndPoint& ep, bool create) {
  HealthRegistry& r = health_registry();
  std::lock_guard<std::mutex> lk(r.mu);
  auto it = r.map.find(ep);
  if (it != r.map.end()) return it->second;
  if (!create) return nullptr;
  EndpointHealth* h = new EndpointHealth;
  r.map[ep] = h;
  return h;
}

struct HealthCheckArg {
  EndPoint ep;
  EndpointHealth* h;
};

// Probes the endpoint until a connect succeeds, then revives it.
void health_check_fiber(void* raw) {
  HealthCheckArg* a = (HealthCheckArg*)raw;
  for (;;) {
    fiber_usleep((uint64_t)FLAG_health_check_interval_ms * 1000);
    bool in_progress = false;
    int fd = tcp_connect(a->ep, &in_progress);
    if (fd >= 0) {
      bool ok = !in_progress;
      if (in_progress) {
        struct pollfd pfd{fd, POLLOUT, 0};
        if (::poll(&pfd, 1, 300) > 0 && (pfd.revents & POLLOUT)) {
          int err = 0;
          socklen_t len = sizeof(err);
          getsockopt(fd, SOL_SOCKET, SO_ERROR, &err, &len);
          ok = err == 0;
        }
      }
      ::close(fd);
      if (ok) {
        a->h->consecutive_errors.store(0, std::memory_order_relaxed);
        a->h->isolated.store(false, std::memory_order_release);
        delete a;
        return;
      }
    }
  }
}

}  // namespace

void ReportClientCallResult(const EndPoint& ep, bool network_error) {
  if (!FLAG_enable_circuit_breaker) return;
  if (!network_error) {
    // Success: skip the registry mutex entirely unless some endpoint has
    // ever errored (the overwhelmingly common case at high QPS).
    if (!g_any_errors.load(std::memory_order_acquire)) return;
    EndpointHealth* h = health_of(ep, false);
    if (h != nullptr) h->consecutive_errors.store(0, std::memory_order_relaxed);
    return;
  }
  g_any_errors.store(true, std::memory_order_release);
  EndpointHealth* h = health_of(ep, true);
  int n = h->consecutive_errors.fetch_add(1, std::memory_order_relaxed) + 1;
  if (n >= FLAG_circuit_breaker_max_failures &&
      !h->isolated.exchange(true, std::memory_order_acq_rel)) {
    HealthCheckArg* a = new HealthCheckArg{ep, h};
    fiber_t th;
    if (fiber_start_background(&th, health_check_fiber, a) != 0) {
      h->isolated.store(false, std::memory_order_release);
      delete a;
    }
  }
}

bool IsEndpointIsolated(const EndPoint& ep) {
  EndpointHealth* h = health_of(ep, false);
  return h != nullptr && h->isolated.load(std::memory_order_acquire);
}

int GetClientSocket(const EndPoint& ep, SocketUniquePtr* out, int shard, bool use_breaker,
                    bool ssl, int protocol_index, const char* socket_mode,
                    bool short_conn) {
  if (use_breaker && IsEndpointIsolated(ep)) {
    errno = EHOSTDOWN;
    return -1;
  }
  if (short_conn) {
    // "short" connection type (≙ reference CONNECTION_TYPE_SHORT): a
    // fresh socket per call, never entered into the map; the caller
    // closes it when the call completes.
    InputMessenger* messenger = client_messenger();
    SocketOptions opts;
    opts.remote_side = ep;
    opts.connect_on_create = true;
    opts.on_edge_triggered_events = [messenger](Socket* s) { messenger->OnNewMessages(s); };
    SocketId sid;
    if (Socket::Create(opts, &sid) != 0) return -1;
    if (Socket::Address(sid, out) != 0) return -1;
    (*out)->client_protocol_hint = protocol_index;
    return 0;
  }
  ClientSocketMap& m = the_map();
  // Key = (endpoint, shard, ssl, protocol) folded into one int (parity:
  // the reference keys client sockets by ChannelSignature). Distinct wire
  // protocols must not share a connection — a response smaller than
  // another protocol's minimum header would starve behind the socket's
  // preferred-protocol parse. Shards are small ints: bits 20-27 carry the
  // protocol index, bit 30 carries TLS.
  const int proto_bits = protocol_index >= 0 ? ((protocol_index + 1) & 0xff) << 20 : 0;
  const bool rdma = socket_mode != nullptr && strcmp(socket_mode, "rdma_mock") == 0;
  const auto key = std::make_pair(
      ep, shard | proto_bits | (ssl ? (1 << 30) : 0) | (rdma ? (1 << 29) : 0));
  {
    std::lock_guard<std::mutex> lk(m.mu);
    auto it = m.sockets.find(key);
    if (it != m.sockets.end()) {
      if (Socket::Address(it->second, out) == 0 && !(*out)->Failed()) return 0;
      m.sockets.erase(it);
      out->reset(nullptr);
    }
  }
  // Create outside the lock (connect may take time), then publish.
  InputMessenger* messenger = client_messenger();
  SocketOptions opts;
  opts.remote_side = ep;
  opts.connect_on_create = true;
  opts.on_edge_triggered_events = [messenger](Socket* s) { messenger->OnNewMessages(s); };
  SocketId sid;
  if (Socket::Create(opts, &sid) != 0) return -1;
  if (Socket::Address(sid, out) != 0) return -1;
  (*out)->client_protocol_hint = protocol_index;
  if (socket_mode != nullptr && strcmp(socket_mode, "rdma_mock") == 0) {
    std::string terr;
    Transport* t = rdma::CreateRdmaTransport((*out).get(), rdma::mock_provider(), 16,
                                             64 << 10, &terr);
    if (t == nullptr) {
      (*out)->SetFailed(ECONNRESET, ("rdma transport: " + terr).c_str());
      out->reset(nullptr);
      return -1;
    }
    (*out)->set_transport(t);
  }
  if (ssl) {
    static void* g_client_ctx = ssl::NewClientCtx();  // process-lifetime
    void* h = g_client_ctx != nullptr
                  ? ssl::NewSsl(g_client_ctx, (*out)->fd(), /*client=*/true)
                  : nullptr;
    if (h == nullptr) {
      (*out)->SetFailed(ECONNRESET, "client SSL_new failed");
      out->reset(nullptr);
      return -1;
    }
    (*out)->set_ssl(h);
  }
  {
    std::lock_guard<std::mutex> lk(m.mu);
    auto it = m.sockets.find(key);
    if (it != m.sockets.end()) {
      // Raced with another creator: prefer the existing healthy one.
      SocketUniquePtr existing;
      if (Socket::Address(it->second, &existing) == 0 && !existing->Failed()) {
        (*out)->SetFailed(ECANCELED, "duplicate connection");
        out->reset(existing.release());
        return 0;
      }
      it->second = sid;
      return 0;
    }
    m.sockets[key] = sid;
  }
  return 0;
}

void RemoveClientSocket(const EndPoint& ep, SocketId expected) {
  ClientSocketMap& m = the_map();
  std::lock_guard<std::mutex> lk(m.mu);
  for (auto it = m.sockets.begin(); it != m.sockets.end(); ++it) {
    if (it->first.first == ep && it->second == expected) {
      m.sockets.erase(it);
      return;
    }
  }
}

}  // namespace bam
