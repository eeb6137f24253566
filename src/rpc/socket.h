// brpc_amd: Socket — the central connection object.
// Parity: reference brpc/socket.h redesigned for this runtime:
//  * addressed by versioned 64-bit SocketId through a resource pool (ABA-safe)
//  * wait-free multi-producer Write(): producers atomically push WriteRequests;
//    the winner writes inline once, then a KeepWrite fiber drains
//  * read side: InputMessenger drains via edge-triggered epoll + fibers
//  * residency-aware: outgoing IOBufs may reference HBM blocks; the write
//    path stages them through the pinned ring (base/iobuf.cc + hip/)
#pragma once

#include <atomic>
#include <deque>
#include <functional>
#include <mutex>
#include <vector>

#include "base/endpoint.h"
#include "base/iobuf.h"
#include "rpc/protocol.h"

namespace bam {

class Socket;

// RAII reference to a socket; releasing decrements the versioned refcount.
class SocketUniquePtr {
 public:
  SocketUniquePtr() : s_(nullptr) {}
  explicit SocketUniquePtr(Socket* s) : s_(s) {}
  ~SocketUniquePtr() { reset(nullptr); }
  SocketUniquePtr(const SocketUniquePtr&) = delete;
  SocketUniquePtr& operator=(const SocketUniquePtr&) = delete;
  SocketUniquePtr(SocketUniquePtr&& o) noexcept : s_(o.s_) { o.s_ = nullptr; }
  SocketUniquePtr& operator=(SocketUniquePtr&& o) noexcept {
    if (this != &o) {
      reset(o.s_);
      o.s_ = nullptr;
    }
    return *this;
  }
  Socket* get() const { return s_; }
  Socket* operator->() const { return s_; }
  Socket& operator*() const { return *s_; }
  explicit operator bool() const { return s_ != nullptr; }
  void reset(Socket* s);
  Socket* release() {
    Socket* s = s_;
    s_ = nullptr;
    return s;
  }

 private:
  Socket* s_;
};

struct SocketOptions {
  int fd = -1;                    // already-established fd (server side)
  EndPoint remote_side;
  bool connect_on_create = false; // client side: connect to remote_side
  void* user = nullptr;           // owner cookie (InputMessenger, Acceptor)
  std::function<void(Socket*)> on_edge_triggered_events;  // readable callback
  std::function<void(SocketId)> on_failed;                // teardown hook
  // Byte transport under this socket (rpc/transport.h). nullptr = the
  // inlined TCP path (parity: reference socket.cpp:1752 `_conn ?
  // CutMessageIntoFileDescriptor : _transport->CutFromIOBuf`). Owned by
  // the socket; deleted at recycle.
  class Transport* transport = nullptr;
};

class Socket {
 public:
  struct WriteRequest {
    IOBuf data;
    std::atomic<WriteRequest*> next{nullptr};
    uint64_t id_wait = 0;  // session to error on write failure
    Socket* socket = nullptr;
    // id_waits of requests coalesced into this one by KeepWrite (failure
    // conduction for merged writes).
    std::vector<uint64_t> merged_ids;
  };

  struct WriteOptions {
    uint64_t id_wait = 0;        // correlation session notified on failure
    bool ignore_eovercrowded = false;
  };

  // Creates a socket, returns 0 and its id.
  static int Create(const SocketOptions& options, SocketId* id);
  // Re-addresses an id; returns 0 and a referenced ptr, or -1 if recycled.
  static int Address(SocketId id, SocketUniquePtr* ptr);

  // Wait-free write: ownership of *data is taken (moved-from on return).
  // Returns 0 on success (queued or written), -1 with errno otherwise.
  int Write(IOBuf* data, const WriteOptions* opt = nullptr);

  // Marks failed: new Address() fail, pending writes error out, epoll
  // deregistered, fd closed when the last ref drops.
  int SetFailed(int error_code, const char* error_text);
  bool Failed() const { return failed_.load(std::memory_order_acquire); }

  int fd() const { return fd_.load(std::memory_order_acquire); }
  SocketId id() const { return id_; }
  const EndPoint& remote_side() const { return remote_side_; }
  const EndPoint& local_side() const { return local_side_; }
  void* user() const { return user_; }

  // Read buffer for InputMessenger.
  IOBuf& read_buf() { return read_buf_; }
  int preferred_protocol_index = -1;
  // For client sockets: the protocol this connection was created for
  // (socket map key). Lets magic-less client protocols (esp) gate their
  // parse to their own connections.
  int client_protocol_hint = -1;
  // Per-connection protocol state (e.g. the h2 session); freed by the
  // deleter when the socket is recycled. protocol_ctx_owner is the
  // registry index of the protocol that installed it: every protocol MUST
  // check the owner before reinterpreting the pointer (an RTMP context
  // reinterpreted as an h2 session locks garbage as a mutex — found by
  // the ASan harness).
  void* protocol_ctx = nullptr;
  void (*protocol_ctx_deleter)(void*) = nullptr;
  int protocol_ctx_owner = -1;

  // Server-side auth result, set on the first verified request of the
  // connection (rpc/authenticator.h); owned by the socket, freed at
  // recycle. nullptr = not yet authenticated.
  class AuthContext* auth_context() const {
    return (class AuthContext*)auth_ctx_.load(std::memory_order_acquire);
  }
  // Returns false if another thread won the race (ctx not installed).
  bool set_auth_context(class AuthContext* ctx) {
    void* expected = nullptr;
    return auth_ctx_.compare_exchange_strong(expected, ctx, std::memory_order_acq_rel);
  }

  // Correlation sessions waiting for responses on this socket; failed
  // when the connection breaks. (Parity: reference conducts errors to
  // ids; we keep an explicit registry.)
  void add_pending_session(uint64_t sid);
  void remove_pending_session(uint64_t sid);

  // Called by the event dispatcher.
  void on_input_event();    // edge-triggered readable
  void on_output_event();   // edge-triggered writable (wakes epollout waiters)
  void run_edge_callback();

  // Blocks current fiber until the fd is writable (or failed/timeout).
  int wait_epoll_out(int64_t abstime_us);

  // --- TLS (rpc/ssl_util.h; handles are opaque so this header stays
  // OpenSSL-free). The handshake is driven lazily: from write paths via
  // write_bytes() and from input edges via ssl_continue_handshake().
  void set_ssl(void* ssl) {
    ssl_ = ssl;
    ssl_state_.store(1, std::memory_order_release);
  }
  bool ssl_enabled() const { return ssl_ != nullptr; }
  int ssl_state() const { return ssl_state_.load(std::memory_order_acquire); }
  // One non-blocking handshake step; -1 = failed (socket SetFailed).
  int ssl_continue_handshake();
  // Poll-loop handshake (client side / write path); 0 done, -1 failed.
  int ssl_handshake_wait();
  // TLS-aware byte movement; plain fd readv/writev (with HBM staging)
  // when TLS is off. may_block=false never waits for the handshake
  // (returns EAGAIN so the caller hands off to KeepWrite).
  ssize_t write_bytes(IOBuf* data, bool may_block);
  ssize_t read_bytes(IOBuf* out, size_t max);
  class Transport* transport() const { return transport_; }
  void set_transport(class Transport* t) { transport_ = t; }  // takes ownership

  // Unwritten bytes queued on this socket; Write fails with EOVERCROWDED
  // above -socket_max_unwritten_bytes (parity: reference socket.cpp:1640).
  std::atomic<int64_t> unwritten_bytes{0};

  // Per-connection stats (builtin /connections page).
  std::atomic<int64_t> in_bytes{0};
  // monotonic µs of the last read or write (idle-timeout reaping).
  std::atomic<int64_t> last_active_us{0};
  std::atomic<int64_t> out_bytes{0};
  std::atomic<int64_t> in_messages{0};
  std::atomic<int64_t> out_messages{0};

  // internal: versioned-ref bookkeeping
  void ReleaseRef();

  Socket() {}  // public for ResourcePool; use Create()

 private:
  friend class SocketUniquePtr;

  static void KeepWriteFiber(void* arg);
  static void RunInputEventsFiber(void* arg);
  // Writes req->data (+ successors) until EAGAIN/empty. Returns 0, or -1.
  int DoWrite(WriteRequest* req);
  // After finishing `done`, returns the next request (FIFO) or nullptr if
  // the queue drained (CAS head -> nullptr succeeded).
  WriteRequest* PopNextRequest(WriteRequest* done);
  void NotifyWriteFailure(WriteRequest* head_chain, int err);
  // Owner-side cleanup on failure: frees its own FIFO chain plus anything
  // newer in the queue (SetFailed never touches the queue; the owner does).
  void ReleaseAllWriteRequests(WriteRequest* fifo_head, int err);
  void Recycle();  // last ref of a failed socket dropped

  SocketId id_ = 0;
  uint32_t pool_index_ = 0;
  // packs (version << 32) | nref
  std::atomic<uint64_t> versioned_ref_{0};
  std::atomic<int> fd_{-1};
  std::atomic<bool> failed_{false};
  std::atomic<bool> connecting_{false};
  int error_code_ = 0;
  std::string error_text_;
  EndPoint remote_side_;
  EndPoint local_side_;
  void* user_ = nullptr;
  std::atomic<void*> auth_ctx_{nullptr};

 public:
  // Per-connection user data (≙ reference session_local_data): installed
  // by Controller::session_local_data(), freed at recycle via the deleter.
  std::atomic<void*> session_local_data{nullptr};
  std::function<void(void*)> session_local_deleter;

 private:
  class Transport* transport_ = nullptr;  // owned; null = inline TCP
  void* ssl_ = nullptr;                 // SSL* when TLS is enabled
  std::atomic<int> ssl_state_{0};       // 0 off, 1 handshaking, 2 ready
  std::mutex ssl_hs_mu_;                // serializes handshake stepping
  std::function<void(Socket*)> on_edge_triggered_events_;
  std::function<void(SocketId)> on_failed_;
  IOBuf read_buf_;

 public:
  std::atomic<int> input_events_{0};

 private:

  std::atomic<WriteRequest*> write_head_{nullptr};
  std::atomic<int>* epollout_butex_ = nullptr;

  std::mutex pending_mu_;
  std::vector<uint64_t> pending_sessions_;

 public:
  // FIFO correlation queue for pipelined protocols (redis/memcache):
  // requests push, responses pop in order.
  void push_pipeline(uint64_t sid) {
    std::lock_guard<std::mutex> lk(pending_mu_);
    pipeline_q_.push_back(sid);
  }
  uint64_t pop_pipeline() {
    std::lock_guard<std::mutex> lk(pending_mu_);
    if (pipeline_q_.empty()) return 0;
    uint64_t sid = pipeline_q_.front();
    pipeline_q_.pop_front();
    return sid;
  }
  uint64_t peek_pipeline() {
    std::lock_guard<std::mutex> lk(pending_mu_);
    return pipeline_q_.empty() ? 0 : pipeline_q_.front();
  }

 private:
  std::deque<uint64_t> pipeline_q_;

  friend class EventDispatcher;
};

// Iterates live sockets (builtin /connections).
void ListSockets(std::vector<SocketId>* out);

}  // namespace bam
