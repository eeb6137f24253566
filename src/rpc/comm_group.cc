#include "rpc/comm_group.h"

#include <arpa/inet.h>
#include <errno.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <string.h>
#include <sys/socket.h>
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <condition_variable>
#include <deque>
#include <mutex>
#include <memory>
#include <thread>
#include <vector>

#include "base/gpu_loader.h"
#include "base/logging.h"
#include "base/time.h"

namespace bam {

// ---------------- TCP full mesh ----------------
// Rank i listens on base_port+i. At init, rank i connects to every HIGHER
// rank's listener (sending its own rank as a 4-byte hello) and accepts one
// connection from every LOWER rank — a deterministic full mesh with no
// duplicate links. Blocking sockets; every logical message is framed
// <u64 len><bytes> and per-peer send/recv are each serialized by a mutex.

// Every frame is tagged with a logical CHANNEL so independent consumers on
// one connection (a collective round's Broadcast recv, a user p2p Recv, a
// HostBroadcast) never steal each other's messages: a reader that pulls a
// frame for another channel parks it in that channel's pending queue.
enum MeshChan { kChanP2P = 0, kChanHost = 1, kChanColl = 2, kChanCount = 3 };

struct CommGroup::Mesh {
  int nranks = 0;
  int rank = 0;
  int listen_fd = -1;
  struct Conn {
    int fd = -1;
    std::mutex send_mu;
    std::mutex recv_mu;              // at most one thread reads the fd
    std::mutex pend_mu;
    std::condition_variable pend_cv;
    std::deque<std::string> pending[kChanCount];
    bool dead = false;
  };
  std::unique_ptr<Conn[]> conns;  // index = peer rank
  std::thread acceptor;
  std::mutex mu;
  std::condition_variable cv;
  int accepted = 0;
  bool accept_fail = false;

  ~Mesh() {
    if (listen_fd >= 0) ::close(listen_fd);
    if (acceptor.joinable()) acceptor.join();
    for (int i = 0; i < nranks; ++i)
      if (conns[i].fd >= 0) ::close(conns[i].fd);
  }

  static bool write_all(int fd, const void* p, size_t n) {
    const char* b = (const char*)p;
    while (n > 0) {
      ssize_t w = ::send(fd, b, n, MSG_NOSIGNAL);
      if (w < 0) {
        if (errno == EINTR) continue;
        return false;
      }
      b += w;
      n -= (size_t)w;
    }
    return true;
  }

  static bool read_all(int fd, void* p, size_t n) {
    char* b = (char*)p;
    while (n > 0) {
      ssize_t r = ::recv(fd, b, n, 0);
      if (r <= 0) {
        if (r < 0 && errno == EINTR) continue;
        return false;
      }
      b += r;
      n -= (size_t)r;
    }
    return true;
  }

  bool send_msg(int peer, int chan, const void* data, size_t n) {
    Conn& c = conns[peer];
    std::lock_guard<std::mutex> lk(c.send_mu);
    uint64_t hdr[2] = {(uint64_t)chan, n};
    return write_all(c.fd, hdr, 16) && (n == 0 || write_all(c.fd, data, n));
  }

  // Pulls the next frame for `chan` from `peer`. A frame for a different
  // channel is parked in its pending queue and its waiters are woken.
  // Returns the payload in *out (or straight into dst when dst != null and
  // the size matches exactly).
  bool recv_chan(int peer, int chan, std::string* out, void* dst, size_t dst_n) {
    Conn& c = conns[peer];
    for (;;) {
      {
        std::unique_lock<std::mutex> lk(c.pend_mu);
        if (!c.pending[chan].empty()) {
          std::string f = std::move(c.pending[chan].front());
          c.pending[chan].pop_front();
          lk.unlock();
          if (dst != nullptr) {
            if (f.size() != dst_n) return false;
            memcpy(dst, f.data(), f.size());
          } else {
            *out = std::move(f);
          }
          return true;
        }
        if (c.dead) return false;
      }
      if (c.recv_mu.try_lock()) {
        std::lock_guard<std::mutex> lk(c.recv_mu, std::adopt_lock);
        // Re-check pending under the read lock: another reader may have
        // parked our frame between the check above and the lock.
        {
          std::lock_guard<std::mutex> plk(c.pend_mu);
          if (!c.pending[chan].empty()) continue;
        }
        uint64_t hdr[2];
        if (!read_all(c.fd, hdr, 16) || hdr[0] >= kChanCount || hdr[1] > (1ull << 33)) {
          std::lock_guard<std::mutex> plk(c.pend_mu);
          c.dead = true;
          c.pend_cv.notify_all();
          return false;
        }
        const int fchan = (int)hdr[0];
        const size_t len = (size_t)hdr[1];
        if (fchan == chan && dst != nullptr) {
          if (len != dst_n) {
            std::lock_guard<std::mutex> plk(c.pend_mu);
            c.dead = true;
            c.pend_cv.notify_all();
            return false;
          }
          return len == 0 || read_all(c.fd, dst, len);
        }
        std::string f(len, 0);
        if (len != 0 && !read_all(c.fd, &f[0], len)) {
          std::lock_guard<std::mutex> plk(c.pend_mu);
          c.dead = true;
          c.pend_cv.notify_all();
          return false;
        }
        if (fchan == chan) {
          *out = std::move(f);
          return true;
        }
        std::lock_guard<std::mutex> plk(c.pend_mu);
        c.pending[fchan].push_back(std::move(f));
        c.pend_cv.notify_all();
      } else {
        // Another thread owns the fd; wait for it to park our frame (or
        // release the lock).
        std::unique_lock<std::mutex> lk(c.pend_mu);
        c.pend_cv.wait_for(lk, std::chrono::milliseconds(5));
      }
    }
  }

  bool recv_msg(int peer, int chan, std::string* out) {
    return recv_chan(peer, chan, out, nullptr, 0);
  }

  bool recv_into(int peer, int chan, void* dst, size_t n) {
    std::string tmp;
    return recv_chan(peer, chan, &tmp, dst, n);
  }
};

namespace {

int connect_with_retry(const std::string& host, int port, int64_t deadline_us) {
  for (;;) {
    int fd = ::socket(AF_INET, SOCK_STREAM, 0);
    if (fd < 0) return -1;
    sockaddr_in sa;
    memset(&sa, 0, sizeof(sa));
    sa.sin_family = AF_INET;
    sa.sin_port = htons((uint16_t)port);
    inet_pton(AF_INET, host.c_str(), &sa.sin_addr);
    if (::connect(fd, (sockaddr*)&sa, sizeof(sa)) == 0) {
      int one = 1;
      setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
      return fd;
    }
    ::close(fd);
    if (monotonic_time_us() > deadline_us) return -1;
    usleep(20000);  // peer's listener may not be up yet
  }
}

}  // namespace

int CommGroup::init(const Options& opt, std::string* err) {
  opt_ = opt;
  if (opt.nranks < 1 || opt.rank < 0 || opt.rank >= opt.nranks || opt.base_port <= 0) {
    *err = "bad CommGroup options";
    return -1;
  }
  mesh_ = new Mesh;
  mesh_->nranks = opt.nranks;
  mesh_->rank = opt.rank;
  mesh_->conns.reset(new Mesh::Conn[opt.nranks]);
  const int64_t deadline = monotonic_time_us() + (int64_t)opt.connect_timeout_ms * 1000;

  if (opt.nranks > 1) {
    // Listener for lower ranks.
    int lfd = ::socket(AF_INET, SOCK_STREAM, 0);
    int one = 1;
    setsockopt(lfd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in sa;
    memset(&sa, 0, sizeof(sa));
    sa.sin_family = AF_INET;
    sa.sin_port = htons((uint16_t)(opt.base_port + opt.rank));
    inet_pton(AF_INET, opt.host.c_str(), &sa.sin_addr);
    if (::bind(lfd, (sockaddr*)&sa, sizeof(sa)) != 0 || ::listen(lfd, opt.nranks) != 0) {
      *err = std::string("bind/listen rank port failed: ") + strerror(errno);
      ::close(lfd);
      return -1;
    }
    mesh_->listen_fd = lfd;
    const int expect_in = opt.rank;  // every lower rank dials us
    Mesh* m = mesh_;
    mesh_->acceptor = std::thread([m, expect_in] {
      for (int i = 0; i < expect_in; ++i) {
        int fd = ::accept(m->listen_fd, nullptr, nullptr);
        if (fd < 0) {
          std::lock_guard<std::mutex> lk(m->mu);
          m->accept_fail = true;
          m->cv.notify_all();
          return;
        }
        int one = 1;
        setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
        uint32_t peer = 0;
        if (!Mesh::read_all(fd, &peer, 4) || peer >= (uint32_t)m->nranks) {
          ::close(fd);
          std::lock_guard<std::mutex> lk(m->mu);
          m->accept_fail = true;
          m->cv.notify_all();
          return;
        }
        m->conns[peer].fd = fd;
        std::lock_guard<std::mutex> lk(m->mu);
        ++m->accepted;
        m->cv.notify_all();
      }
    });

    // Dial every higher rank.
    for (int peer = opt.rank + 1; peer < opt.nranks; ++peer) {
      int fd = connect_with_retry(opt.host, opt.base_port + peer, deadline);
      if (fd < 0) {
        *err = "connect to rank " + std::to_string(peer) + " timed out";
        return -1;
      }
      uint32_t me = (uint32_t)opt.rank;
      if (!Mesh::write_all(fd, &me, 4)) {
        *err = "hello to rank " + std::to_string(peer) + " failed";
        ::close(fd);
        return -1;
      }
      mesh_->conns[peer].fd = fd;
    }
    // Wait for all lower ranks.
    {
      std::unique_lock<std::mutex> lk(mesh_->mu);
      if (!mesh_->cv.wait_for(lk, std::chrono::milliseconds(opt.connect_timeout_ms), [&] {
            return mesh_->accepted >= opt.rank || mesh_->accept_fail;
          }) ||
          mesh_->accept_fail) {
        *err = "accept from lower ranks failed/timed out";
        return -1;
      }
    }
  }

  if (opt.backend == "rccl") {
    const gpu::GpuApi* api = gpu::api();
    if (api == nullptr || api->comm_create == nullptr) {
      *err = "rccl backend unavailable: HIP lib not loaded or has no comm surface";
      return -1;
    }
    char uid[128];
    std::string blob;
    if (opt.rank == 0) {
      if (api->comm_uid(uid) != 0) {
        *err = std::string("ncclGetUniqueId: ") +
               (api->comm_last_error ? api->comm_last_error() : "?");
        return -1;
      }
      blob.assign(uid, 128);
    }
    if (HostBroadcast(&blob, 0) != 0 || blob.size() != 128) {
      *err = "uid rendezvous failed";
      return -1;
    }
    memcpy(uid, blob.data(), 128);
    rccl_ = api->comm_create(opt.nranks, opt.rank, uid, opt.dev);
    if (rccl_ == nullptr) {
      *err = std::string("ncclCommInitRank: ") +
             (api->comm_last_error ? api->comm_last_error() : "?");
      return -1;
    }
  } else if (opt.backend != "tcp") {
    *err = "unknown backend " + opt.backend;
    return -1;
  }
  return 0;
}

CommGroup* CommGroup::Create(const Options& opt, std::string* err) {
  std::string dummy;
  if (err == nullptr) err = &dummy;
  CommGroup* g = new CommGroup;
  if (g->init(opt, err) != 0) {
    delete g;
    return nullptr;
  }
  return g;
}

CommGroup::~CommGroup() {
  if (rccl_ != nullptr) {
    const gpu::GpuApi* api = gpu::api();
    if (api != nullptr && api->comm_destroy != nullptr) api->comm_destroy(rccl_);
  }
  delete mesh_;
}

// ---------------- host control plane ----------------

int CommGroup::HostSend(int peer, const void* data, size_t n) {
  if (peer < 0 || peer >= opt_.nranks || peer == opt_.rank) return -1;
  return mesh_->send_msg(peer, kChanHost, data, n) ? 0 : -1;
}

int CommGroup::HostRecv(int peer, std::string* out) {
  if (peer < 0 || peer >= opt_.nranks || peer == opt_.rank) return -1;
  return mesh_->recv_msg(peer, kChanHost, out) ? 0 : -1;
}

int CommGroup::HostBroadcast(std::string* blob, int root) {
  if (opt_.nranks == 1) return 0;
  if (opt_.rank == root) {
    for (int p = 0; p < opt_.nranks; ++p) {
      if (p == opt_.rank) continue;
      if (!mesh_->send_msg(p, kChanHost, blob->data(), blob->size())) return -1;
    }
    return 0;
  }
  return mesh_->recv_msg(root, kChanHost, blob) ? 0 : -1;
}

int CommGroup::Barrier() {
  if (opt_.nranks == 1) return 0;
  // All-to-root-to-all with empty payloads.
  char z = 0;
  if (opt_.rank == 0) {
    std::string tmp;
    for (int p = 1; p < opt_.nranks; ++p)
      if (!mesh_->recv_msg(p, kChanHost, &tmp)) return -1;
    for (int p = 1; p < opt_.nranks; ++p)
      if (!mesh_->send_msg(p, kChanHost, &z, 1)) return -1;
    return 0;
  }
  if (!mesh_->send_msg(0, kChanHost, &z, 1)) return -1;
  std::string tmp;
  return mesh_->recv_msg(0, kChanHost, &tmp) ? 0 : -1;
}

// ---------------- data plane ----------------

int CommGroup::Broadcast(void* buf, size_t n, int root) {
  if (opt_.nranks == 1) return 0;
  if (rccl_ != nullptr) {
    const gpu::GpuApi* api = gpu::api();
    return api->comm_broadcast(rccl_, buf, n, root);
  }
  // tcp backend (host buffers): root pushes to every peer.
  if (opt_.rank == root) {
    for (int p = 0; p < opt_.nranks; ++p) {
      if (p == opt_.rank) continue;
      if (!mesh_->send_msg(p, kChanColl, buf, n)) return -1;
    }
    return 0;
  }
  return mesh_->recv_into(root, kChanColl, buf, n) ? 0 : -1;
}

int CommGroup::AllGather(const void* send, void* recv, size_t per_rank) {
  if (opt_.nranks == 1) {
    memcpy(recv, send, per_rank);
    return 0;
  }
  if (rccl_ != nullptr) {
    const gpu::GpuApi* api = gpu::api();
    return api->comm_allgather(rccl_, send, recv, per_rank);
  }
  // tcp: one sender thread per peer (sends to distinct sockets cannot
  // chain-block each other), main thread receives every peer's shard.
  char* out = (char*)recv;
  memcpy(out + (size_t)opt_.rank * per_rank, send, per_rank);
  std::atomic<bool> ok{true};
  std::vector<std::thread> senders;
  senders.reserve(opt_.nranks - 1);
  for (int p = 0; p < opt_.nranks; ++p) {
    if (p == opt_.rank) continue;
    senders.emplace_back([this, p, send, per_rank, &ok] {
      if (!mesh_->send_msg(p, kChanColl, send, per_rank)) ok.store(false);
    });
  }
  for (int p = 0; p < opt_.nranks; ++p) {
    if (p == opt_.rank) continue;
    if (!mesh_->recv_into(p, kChanColl, out + (size_t)p * per_rank, per_rank)) ok.store(false);
  }
  for (auto& t : senders) t.join();
  return ok.load() ? 0 : -1;
}

int CommGroup::Send(const void* buf, size_t n, int peer) {
  if (rccl_ != nullptr) return gpu::api()->comm_send(rccl_, buf, n, peer);
  return mesh_->send_msg(peer, kChanP2P, buf, n) ? 0 : -1;
}

int CommGroup::Recv(void* buf, size_t n, int peer) {
  if (rccl_ != nullptr) return gpu::api()->comm_recv(rccl_, buf, n, peer);
  return mesh_->recv_into(peer, kChanP2P, buf, n) ? 0 : -1;
}

int CommGroup::SendRecv(const void* sbuf, size_t sn, int speer, void* rbuf, size_t rn,
                        int rpeer) {
  if (rccl_ != nullptr) return gpu::api()->comm_sendrecv(rccl_, sbuf, sn, speer, rbuf, rn, rpeer);
  // tcp: overlap directions with a sender thread.
  std::atomic<bool> ok{true};
  std::thread t([this, sbuf, sn, speer, &ok] {
    if (sn > 0 && !mesh_->send_msg(speer, kChanP2P, sbuf, sn)) ok.store(false);
  });
  if (rn > 0 && !mesh_->recv_into(rpeer, kChanP2P, rbuf, rn)) ok.store(false);
  t.join();
  return ok.load() ? 0 : -1;
}

}  // namespace bam
