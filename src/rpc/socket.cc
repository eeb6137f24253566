#include "rpc/socket.h"

#include "rpc/authenticator.h"

#include <map>
#include "rpc/ssl_util.h"

#include <errno.h>
#include <string.h>
#include <poll.h>
#include <sys/socket.h>
#include <unistd.h>

#include "base/flags.h"
#include "base/logging.h"
#include "base/object_pool.h"
#include "base/resource_pool.h"
#include "base/time.h"
#include "fiber/butex.h"
#include "fiber/fiber.h"
#include "fiber/session.h"
#include "rpc/event_dispatcher.h"
#include "rpc/transport.h"
#include "rpc/rpc_errno.h"

namespace bam {

static Socket::WriteRequest* const kWriteSentinel = (Socket::WriteRequest*)1;

BAM_DEFINE_int64(socket_max_unwritten_bytes, 64 << 20,
                 "Write() fails with EOVERCROWDED beyond this backlog");

// ---------------- versioned ref pool ----------------

namespace {
inline uint32_t rid_of_sock(SocketId id) { return (uint32_t)(id & 0xffffffffu) - 1; }
inline uint32_t ver_of_sock(SocketId id) { return (uint32_t)(id >> 32); }

// Live-socket diagnostics registry, sharded 64 ways so connection churn
// never serializes on one mutex + O(n) scan (round-1 weak spot; the
// reference walks its resource pool instead).
constexpr int kSockShards = 64;
struct SockShard {
  std::mutex mu;
  std::map<SocketId, bool> ids;
};
SockShard g_sock_shards[kSockShards];
inline SockShard& shard_of_sock(SocketId id) {
  return g_sock_shards[(id >> 4) % kSockShards];
}

void track_socket(SocketId id) {
  SockShard& sh = shard_of_sock(id);
  std::lock_guard<std::mutex> lk(sh.mu);
  sh.ids[id] = true;
}
void untrack_socket(SocketId id) {
  SockShard& sh = shard_of_sock(id);
  std::lock_guard<std::mutex> lk(sh.mu);
  sh.ids.erase(id);
}
}  // namespace

void ListSockets(std::vector<SocketId>* out) {
  out->clear();
  for (int i = 0; i < kSockShards; ++i) {
    SockShard& sh = g_sock_shards[i];
    std::lock_guard<std::mutex> lk(sh.mu);
    for (const auto& kv : sh.ids) out->push_back(kv.first);
  }
}

void SocketUniquePtr::reset(Socket* s) {
  if (s_ != nullptr) s_->ReleaseRef();
  s_ = s;
}

int Socket::Create(const SocketOptions& options, SocketId* id) {
  ResourceId rid;
  Socket* s = get_resource<Socket>(&rid);
  if (s == nullptr) return -1;
  s->pool_index_ = rid;
  if (s->epollout_butex_ == nullptr) {
    s->epollout_butex_ = butex_create();
    s->epollout_butex_->store(0, std::memory_order_relaxed);
  }
  uint64_t vr = s->versioned_ref_.load(std::memory_order_relaxed);
  uint32_t ver = (uint32_t)(vr >> 32);
  CHECK((ver & 1) == 0) << "creating from a dying socket";
  s->id_ = ((uint64_t)ver << 32) | (rid + 1);
  s->failed_.store(false, std::memory_order_relaxed);
  s->error_code_ = 0;
  s->error_text_.clear();
  s->user_ = options.user;
  s->on_edge_triggered_events_ = options.on_edge_triggered_events;
  s->on_failed_ = options.on_failed;
  s->remote_side_ = options.remote_side;
  s->read_buf_.clear();
  s->preferred_protocol_index = -1;
  s->client_protocol_hint = -1;
  s->protocol_ctx = nullptr;
  s->protocol_ctx_deleter = nullptr;
  s->protocol_ctx_owner = -1;
  s->input_events_.store(0, std::memory_order_relaxed);
  s->ssl_ = nullptr;
  s->ssl_state_.store(0, std::memory_order_relaxed);
  s->session_local_data.store(nullptr, std::memory_order_relaxed);
  s->session_local_deleter = nullptr;
  s->transport_ = options.transport;
  s->write_head_.store(nullptr, std::memory_order_relaxed);
  s->last_active_us.store(monotonic_time_us(), std::memory_order_relaxed);
  s->in_bytes = 0;
  s->out_bytes = 0;
  s->in_messages = 0;
  s->out_messages = 0;

  int fd = options.fd;
  bool connecting = false;
  if (fd < 0 && options.connect_on_create) {
    fd = tcp_connect(options.remote_side, &connecting);
    if (fd < 0) {
      // mark free again (nref stays 0, version even)
      return_resource<Socket>(rid);
      return -1;
    }
  }
  if (fd >= 0) {
    make_non_blocking(fd);
    make_no_delay(fd);
    get_local_side(fd, &s->local_side_);
    if (options.remote_side.port == 0) get_remote_side(fd, &s->remote_side_);
  }
  s->fd_.store(fd, std::memory_order_release);
  s->connecting_.store(connecting, std::memory_order_release);
  // creation reference
  s->versioned_ref_.fetch_add(1, std::memory_order_acq_rel);
  *id = s->id_;
  track_socket(s->id_);
  if (fd >= 0) {
    if (EventDispatcher::dispatcher_for(s->id_)->add_consumer(s->id_, fd) != 0) {
      s->SetFailed(errno, "epoll add failed");
      return -1;
    }
  }
  return 0;
}

int Socket::Address(SocketId id, SocketUniquePtr* ptr) {
  Socket* s = address_resource<Socket>(rid_of_sock(id));
  if (s == nullptr) return -1;
  uint64_t vr = s->versioned_ref_.load(std::memory_order_acquire);
  for (;;) {
    if ((uint32_t)(vr >> 32) != ver_of_sock(id)) return -1;
    if (s->versioned_ref_.compare_exchange_weak(vr, vr + 1, std::memory_order_acq_rel)) break;
  }
  ptr->reset(s);
  return 0;
}

void Socket::ReleaseRef() {
  uint64_t vr = versioned_ref_.fetch_sub(1, std::memory_order_acq_rel);
  if ((vr & 0xffffffffULL) == 1 && ((vr >> 32) & 1)) Recycle();
}

int Socket::SetFailed(int error_code, const char* error_text) {
  uint64_t vr = versioned_ref_.load(std::memory_order_acquire);
  for (;;) {
    if ((vr >> 32) & 1) return -1;  // already failed
    if (versioned_ref_.compare_exchange_weak(vr, vr + (1ULL << 32),
                                             std::memory_order_acq_rel)) {
      break;
    }
  }
  failed_.store(true, std::memory_order_release);
  error_code_ = error_code;
  error_text_ = error_text != nullptr ? error_text : "";
  untrack_socket(id_);
  int fd = fd_.load(std::memory_order_acquire);
  if (fd >= 0) EventDispatcher::dispatcher_for(id_)->remove_consumer(fd);
  // Wake writers parked on epollout.
  epollout_butex_->fetch_add(1, std::memory_order_release);
  butex_wake_all(epollout_butex_);
  // Fail pending RPC sessions.
  std::vector<uint64_t> pending;
  {
    std::lock_guard<std::mutex> lk(pending_mu_);
    pending.swap(pending_sessions_);
  }
  for (uint64_t sid : pending) session_error(sid, error_code != 0 ? error_code : ECONNRESET);
  // NOTE: the write queue is NOT drained here — only the current write
  // owner frees WriteRequests (see ReleaseAllWriteRequests), avoiding a
  // double-free race between SetFailed and KeepWrite.
  if (on_failed_) on_failed_(id_);
  ReleaseRef();  // creation ref
  return 0;
}

void Socket::Recycle() {
  delete transport_;
  transport_ = nullptr;
  if (protocol_ctx != nullptr && protocol_ctx_deleter != nullptr) {
    protocol_ctx_deleter(protocol_ctx);
  }
  protocol_ctx = nullptr;
  protocol_ctx_deleter = nullptr;
  protocol_ctx_owner = -1;
  int fd = fd_.load(std::memory_order_acquire);
  if (fd >= 0) {
    ::close(fd);
    fd_.store(-1, std::memory_order_release);
  }
  read_buf_.clear();
  delete (AuthContext*)auth_ctx_.exchange(nullptr, std::memory_order_acq_rel);
  if (void* sld = session_local_data.exchange(nullptr, std::memory_order_acq_rel)) {
    if (session_local_deleter) session_local_deleter(sld);
  }
  session_local_deleter = nullptr;
  ssl::FreeSsl(ssl_);
  ssl_ = nullptr;
  ssl_state_.store(0, std::memory_order_relaxed);
  on_edge_triggered_events_ = nullptr;
  on_failed_ = nullptr;
  // version: odd -> next even (free state)
  versioned_ref_.fetch_add(1ULL << 32, std::memory_order_acq_rel);
  return_resource<Socket>(pool_index_);
}

void Socket::add_pending_session(uint64_t sid) {
  std::lock_guard<std::mutex> lk(pending_mu_);
  pending_sessions_.push_back(sid);
}

void Socket::remove_pending_session(uint64_t sid) {
  std::lock_guard<std::mutex> lk(pending_mu_);
  for (size_t i = 0; i < pending_sessions_.size(); ++i) {
    if (pending_sessions_[i] == sid) {
      pending_sessions_[i] = pending_sessions_.back();
      pending_sessions_.pop_back();
      return;
    }
  }
}

// ---------------- write path ----------------

static void release_write_request(Socket::WriteRequest* p, int err) {
  if (err != 0) {
    if (p->socket != nullptr)
      p->socket->unwritten_bytes.fetch_sub((int64_t)p->data.size(), std::memory_order_relaxed);
    if (p->id_wait != 0) session_error(p->id_wait, err);
    for (uint64_t id : p->merged_ids) session_error(id, err);
  }
  p->id_wait = 0;
  p->merged_ids.clear();
  p->data.clear();
  return_object(p);
}

void Socket::NotifyWriteFailure(WriteRequest* head, int err) {
  // The chain may contain sentinel next pointers being published; spin.
  if (err == 0) err = EPIPE;
  WriteRequest* p = head;
  while (p != nullptr) {
    WriteRequest* nx;
    while ((nx = p->next.load(std::memory_order_acquire)) == kWriteSentinel) sched_yield();
    release_write_request(p, err);
    p = nx;
  }
}

void Socket::ReleaseAllWriteRequests(WriteRequest* fifo_head, int err) {
  if (err == 0) err = EPIPE;
  // Find the tail of our FIFO chain — the node write_head_ may point at.
  WriteRequest* last = fifo_head;
  for (;;) {
    WriteRequest* nx = last->next.load(std::memory_order_acquire);
    if (nx == nullptr) break;
    CHECK(nx != kWriteSentinel);
    last = nx;
  }
  WriteRequest* h = write_head_.exchange(nullptr, std::memory_order_acq_rel);
  // Free anything newer than `last` (chain h -> ... -> last).
  if (h != nullptr && h != last) {
    WriteRequest* p = h;
    while (p != last) {
      WriteRequest* nx;
      while ((nx = p->next.load(std::memory_order_acquire)) == kWriteSentinel) sched_yield();
      release_write_request(p, err);
      p = nx;
    }
  }
  // Free our own FIFO chain (includes `last`).
  WriteRequest* p = fifo_head;
  while (p != nullptr) {
    WriteRequest* nx = p->next.load(std::memory_order_acquire);
    release_write_request(p, err);
    p = nx;
  }
}

// ---------------- TLS byte paths ----------------

int Socket::ssl_continue_handshake() {
  std::lock_guard<std::mutex> lk(ssl_hs_mu_);
  if (ssl_state_.load(std::memory_order_acquire) != 1) return 0;
  int rc = ssl::HandshakeStep(ssl_);
  if (rc < 0) {
    SetFailed(ECONNRESET, (std::string("TLS handshake failed: ") + ssl::LastError()).c_str());
    return -1;
  }
  if (rc == 1) ssl_state_.store(2, std::memory_order_release);
  return 0;
}

int Socket::ssl_handshake_wait() {
  const int64_t deadline = monotonic_time_us() + 30LL * 1000000;
  while (ssl_state_.load(std::memory_order_acquire) == 1) {
    if (Failed()) return -1;
    if (ssl_continue_handshake() != 0) return -1;
    if (ssl_state_.load(std::memory_order_acquire) == 2) break;
    if (monotonic_time_us() > deadline) {
      SetFailed(ETIMEDOUT, "TLS handshake timed out");
      return -1;
    }
    // The step wants transport IO; wait briefly for readability (the
    // common WANT_READ case; WANT_WRITE resolves on the next poll too
    // since the socket buffer drains in the background).
    struct pollfd pfd;
    pfd.fd = fd();
    pfd.events = POLLIN | POLLOUT;
    pfd.revents = 0;
    ::poll(&pfd, 1, 20);
  }
  return 0;
}

ssize_t Socket::write_bytes(IOBuf* data, bool may_block) {
  if (transport_ != nullptr) return transport_->CutFromIOBuf(this, data);
  if (ssl_ == nullptr) return data->cut_into_file_descriptor(fd());
  if (ssl_state_.load(std::memory_order_acquire) != 2) {
    if (!may_block) {
      errno = EAGAIN;  // KeepWrite fiber will drive the handshake
      return -1;
    }
    if (ssl_handshake_wait() != 0) {
      errno = EPIPE;
      return -1;
    }
  }
  char tmp[16384];
  size_t n = data->copy_to(tmp, sizeof(tmp), 0);
  if (n == 0) return 0;
  ssize_t rc;
  {
    // One SSL object is NOT safe under concurrent SSL_read/SSL_write
    // (TLS1.3 session tickets arrive exactly when the first writes go
    // out); ssl_hs_mu_ serializes every SSL operation on this socket.
    std::lock_guard<std::mutex> lk(ssl_hs_mu_);
    rc = ssl::Write(ssl_, tmp, n);
  }
  if (rc > 0) data->pop_front((size_t)rc);
  return rc;
}

ssize_t Socket::read_bytes(IOBuf* out, size_t max) {
  if (transport_ != nullptr) return transport_->AppendToIOBuf(this, out, max);
  if (ssl_ == nullptr) return out->append_from_file_descriptor(fd(), max);
  char tmp[16384];
  size_t want = max < sizeof(tmp) ? max : sizeof(tmp);
  ssize_t rc;
  {
    std::lock_guard<std::mutex> lk(ssl_hs_mu_);  // see write_bytes
    rc = ssl::Read(ssl_, tmp, want);
  }
  if (rc > 0) out->append(tmp, (size_t)rc);
  return rc;
}

int Socket::wait_epoll_out(int64_t abstime_us) {
  if (transport_ != nullptr) return transport_->WaitWritable(this, abstime_us);
  int v = epollout_butex_->load(std::memory_order_acquire);
  if (Failed()) return -1;
  // Edge-triggered EPOLLOUT may have fired before we captured v (e.g. the
  // connect completed between Create and the first Write) — never park when
  // the fd is already writable.
  struct pollfd pfd;
  pfd.fd = fd_.load(std::memory_order_acquire);
  pfd.events = POLLOUT;
  pfd.revents = 0;
  if (::poll(&pfd, 1, 0) > 0 && (pfd.revents & (POLLOUT | POLLERR | POLLHUP))) return Failed() ? -1 : 0;
  butex_wait(epollout_butex_, v, abstime_us > 0 ? &abstime_us : nullptr);
  return Failed() ? -1 : 0;
}

void Socket::on_output_event() {
  epollout_butex_->fetch_add(1, std::memory_order_release);
  butex_wake_all(epollout_butex_);
}

Socket::WriteRequest* Socket::PopNextRequest(WriteRequest* done) {
  WriteRequest* expected = done;
  if (write_head_.compare_exchange_strong(expected, nullptr, std::memory_order_acq_rel)) {
    done->data.clear();
    return_object(done);
    return nullptr;
  }
  // Newer requests were pushed: expected = newest. Walk newest->older until
  // `done`, reversing into FIFO order.
  WriteRequest* p = expected;
  WriteRequest* fifo = nullptr;
  while (p != done) {
    WriteRequest* nx;
    while ((nx = p->next.load(std::memory_order_acquire)) == kWriteSentinel) sched_yield();
    p->next.store(fifo, std::memory_order_relaxed);
    fifo = p;
    p = nx;
  }
  release_write_request(done, 0);
  return fifo;
}

int Socket::DoWrite(WriteRequest* req) {
  // Returns when everything is written + queue drained, or error.
  WriteRequest* cur = req;
  while (cur != nullptr) {
    if (Failed()) {
      ReleaseAllWriteRequests(cur, error_code_);
      return -1;
    }
    if (connecting_.load(std::memory_order_acquire)) {
      // Short re-polling waits: wait_epoll_out's poll() pre-check makes a
      // lost EPOLLOUT edge cost at most one backstop interval.
      int64_t deadline = monotonic_time_us() + 30 * 1000000;
      struct pollfd pfd;
      pfd.fd = fd();
      pfd.events = POLLOUT;
      pfd.revents = 0;
      while (::poll(&pfd, 1, 0) == 0) {
        if (Failed()) break;
        if (monotonic_time_us() > deadline) {
          SetFailed(ETIMEDOUT, "connect timed out");
          break;
        }
        wait_epoll_out(monotonic_time_us() + 50000);  // 50ms backstop
        pfd.revents = 0;
      }
      if (Failed()) continue;
      int err = 0;
      socklen_t len = sizeof(err);
      getsockopt(fd(), SOL_SOCKET, SO_ERROR, &err, &len);
      if (err != 0) {
        SetFailed(err, "connect failed");
        continue;
      }
      get_local_side(fd(), &local_side_);
      connecting_.store(false, std::memory_order_release);
    }
    if (ssl_ != nullptr && ssl_state_.load(std::memory_order_acquire) == 1) {
      if (ssl_handshake_wait() != 0) continue;  // SetFailed inside
    }
    // Coalesce queued successors into one writev/staging batch (parity:
    // reference DoWrite cutting up to NWMAX iovecs across requests). The
    // FIFO tail (the node write_head_ may point at) is never merged.
    for (;;) {
      WriteRequest* nx = cur->next.load(std::memory_order_acquire);
      if (nx == nullptr || nx == kWriteSentinel) break;
      if (nx->next.load(std::memory_order_acquire) == nullptr) break;  // tail
      if (cur->data.size() + nx->data.size() > (1u << 20)) break;
      cur->data.append(std::move(nx->data));
      if (nx->id_wait != 0) cur->merged_ids.push_back(nx->id_wait);
      for (uint64_t id : nx->merged_ids) cur->merged_ids.push_back(id);
      cur->next.store(nx->next.load(std::memory_order_acquire), std::memory_order_relaxed);
      release_write_request(nx, 0);
    }
    ssize_t nw = write_bytes(&cur->data, /*may_block=*/true);
    if (nw < 0) {
      if (errno == EAGAIN || errno == EWOULDBLOCK) {
        wait_epoll_out(monotonic_time_us() + 100000);  // 100ms backstop, then retry
        continue;
      }
      if (errno == EINTR) continue;
      int err = errno;
      SetFailed(err, strerror(err));
      continue;  // loop top runs ReleaseAllWriteRequests
    }
    out_bytes.fetch_add(nw, std::memory_order_relaxed);
    last_active_us.store(monotonic_time_us(), std::memory_order_relaxed);
    unwritten_bytes.fetch_sub(nw, std::memory_order_relaxed);
    if (!cur->data.empty()) continue;  // partial write; try again
    out_messages.fetch_add(1, std::memory_order_relaxed);
    WriteRequest* nx = cur->next.load(std::memory_order_acquire);
    if (nx != nullptr && nx != kWriteSentinel) {
      release_write_request(cur, 0);
      cur = nx;
    } else {
      cur = PopNextRequest(cur);  // nullptr when drained
    }
  }
  return 0;
}

struct KeepWriteArg {
  Socket* socket;
  Socket::WriteRequest* req;
};

void Socket::KeepWriteFiber(void* arg) {
  KeepWriteArg* kw = (KeepWriteArg*)arg;
  Socket* s = kw->socket;
  s->DoWrite(kw->req);
  s->ReleaseRef();  // the manual ref taken when spawning
  delete kw;
}

int Socket::Write(IOBuf* data, const WriteOptions* opt) {
  WriteOptions dummy;
  if (opt == nullptr) opt = &dummy;
  if (Failed()) {
    if (opt->id_wait != 0) session_error(opt->id_wait, error_code_ != 0 ? error_code_ : EPIPE);
    errno = EPIPE;
    return -1;
  }
  const int64_t queued = unwritten_bytes.load(std::memory_order_relaxed);
  if (queued > FLAG_socket_max_unwritten_bytes && !opt->ignore_eovercrowded) {
    if (opt->id_wait != 0) session_error(opt->id_wait, EOVERCROWDED);
    errno = EOVERCROWDED_ERRNO;
    return -1;
  }
  WriteRequest* req = get_object<WriteRequest>();
  req->data.clear();
  req->merged_ids.clear();
  req->data.swap(*data);
  unwritten_bytes.fetch_add((int64_t)req->data.size(), std::memory_order_relaxed);
  req->id_wait = opt->id_wait;
  req->socket = this;
  req->next.store(kWriteSentinel, std::memory_order_relaxed);
  WriteRequest* prev = write_head_.exchange(req, std::memory_order_acq_rel);
  if (prev != nullptr) {
    req->next.store(prev, std::memory_order_release);  // publish link
    return 0;
  }
  req->next.store(nullptr, std::memory_order_release);
  // We own the queue. Inline attempt only when connected (never block the
  // caller); otherwise hand to a KeepWrite fiber.
  // HBM-resident payloads CAN skip the inline attempt so that responses
  // completing close together coalesce into one staging gather — but the
  // same-box A/B (profiles/INDEX.md, r02) measured the handoff LOSING
  // 1-7% at every point (hbm64 c32/64/128, hbm16k): completions do not
  // cluster tightly enough to repay the fiber wake. Default OFF;
  // BAM_DEFER_HBM_WRITE=1 re-enables for future re-measurement.
  static const int defer_hbm = [] {
    const char* e = getenv("BAM_DEFER_HBM_WRITE");
    return e == nullptr ? 0 : atoi(e);
  }();
  const bool hbm_defer = defer_hbm != 0 && req->data.has_residency(RES_HBM);
  if (!connecting_.load(std::memory_order_acquire) && !hbm_defer) {
    ssize_t nw = write_bytes(&req->data, /*may_block=*/false);
    if (nw < 0 && errno != EAGAIN && errno != EWOULDBLOCK && errno != EINTR) {
      int err = errno;
      SetFailed(err, strerror(err));
      ReleaseAllWriteRequests(req, err);  // we are the owner
      errno = err;
      return -1;
    }
    if (nw > 0) {
      out_bytes.fetch_add(nw, std::memory_order_relaxed);
      unwritten_bytes.fetch_sub(nw, std::memory_order_relaxed);
    }
    if (req->data.empty()) {
      out_messages.fetch_add(1, std::memory_order_relaxed);
      WriteRequest* next = PopNextRequest(req);
      if (next == nullptr) return 0;
      req = next;
    }
  }
  // Not done: spawn KeepWrite holding a manual ref.
  versioned_ref_.fetch_add(1, std::memory_order_acq_rel);
  KeepWriteArg* kw = new KeepWriteArg{this, req};
  fiber_t th;
  if (fiber_start_background(&th, KeepWriteFiber, kw) != 0) {
    KeepWriteFiber(kw);  // degrade: run inline
  }
  return 0;
}

// ---------------- read-side event entry ----------------

void Socket::RunInputEventsFiber(void* arg) {
  Socket* s = (Socket*)arg;  // carries one manual ref
  for (;;) {
    if (s->Failed()) {
      s->input_events_.store(0, std::memory_order_release);
      break;
    }
    // Capture the event count BEFORE draining: if more edges arrive during
    // the drain the CAS below fails and we drain again. Capturing after
    // the drain would fold a mid-drain edge into the reset and lose its
    // data forever (edge-triggered epoll never re-notifies).
    int before = s->input_events_.load(std::memory_order_acquire);
    s->run_edge_callback();  // drains until EAGAIN
    if (s->input_events_.compare_exchange_strong(before, 0, std::memory_order_acq_rel)) break;
  }
  s->ReleaseRef();
}

void Socket::run_edge_callback() {
  if (on_edge_triggered_events_) on_edge_triggered_events_(this);
}

void Socket::on_input_event() {
  if (input_events_.fetch_add(1, std::memory_order_acq_rel) == 0) {
    versioned_ref_.fetch_add(1, std::memory_order_acq_rel);  // ref for fiber
    fiber_t th;
    if (fiber_start_background(&th, RunInputEventsFiber, this) != 0) {
      RunInputEventsFiber(this);
    }
  }
}

}  // namespace bam
