// brpc_amd: shared-memory ring RPC (see shm_ring.h).
#include "rpc/shm_ring.h"

#include <fcntl.h>
#include <string.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

#include <atomic>
#include <map>
#include <memory>
#include <mutex>
#include <thread>

#include "base/logging.h"
#include "base/time.h"
#include "fiber/fiber.h"
#include "fiber/sync.h"
#include "rpc/rpc_errno.h"
#include "rpc/server.h"

namespace bam {
namespace shm {

namespace {

constexpr uint32_t kMagic = 0xba5ee75u;

struct alignas(64) Cursor {
  std::atomic<uint64_t> v{0};
  char pad[64 - sizeof(std::atomic<uint64_t>)];
};

struct SegHdr {
  uint32_t magic;
  uint32_t ring_bytes;            // per direction
  std::atomic<uint32_t> closed;   // either side sets on teardown
  uint32_t reserved;
  Cursor req_head;   // producer: client
  Cursor req_tail;   // consumer: server
  Cursor resp_head;  // producer: server
  Cursor resp_tail;  // consumer: client
  // memory: [req ring][resp ring]
};

// Record header (both directions), 16 bytes, 8-aligned total size.
// Requests:  aux = method_len, body = method + payload.
// Responses: aux = (uint32_t)error_code, body = error_text | payload.
struct RecHdr {
  uint32_t body_len;
  uint32_t aux;
  uint64_t cid;
};

struct Ring {
  char* base;          // ring storage
  uint32_t bytes;      // power of two NOT required (we use %)
  Cursor* head;
  Cursor* tail;
};

inline void ring_copy_in(Ring& r, uint64_t pos, const void* src, size_t n) {
  size_t off = (size_t)(pos % r.bytes);
  size_t first = r.bytes - off < n ? r.bytes - off : n;
  memcpy(r.base + off, src, first);
  if (first < n) memcpy(r.base, (const char*)src + first, n - first);
}

inline void ring_copy_out(Ring& r, uint64_t pos, void* dst, size_t n) {
  size_t off = (size_t)(pos % r.bytes);
  size_t first = r.bytes - off < n ? r.bytes - off : n;
  memcpy(dst, r.base + off, first);
  if (first < n) memcpy((char*)dst + first, r.base, n - first);
}

inline size_t rec_size(size_t body) { return (sizeof(RecHdr) + body + 7) & ~(size_t)7; }

// Producer: writes one record; false if the ring lacks space.
bool ring_push(Ring& r, uint64_t cid, uint32_t aux, const void* a, size_t alen,
               const IOBuf* b) {
  size_t body = alen + (b != nullptr ? b->size() : 0);
  size_t need = rec_size(body);
  if (need > r.bytes / 2) return false;  // refuse giant records
  uint64_t head = r.head->v.load(std::memory_order_relaxed);
  uint64_t tail = r.tail->v.load(std::memory_order_acquire);
  if (head - tail + need > r.bytes) return false;  // full
  RecHdr h{(uint32_t)body, aux, cid};
  ring_copy_in(r, head, &h, sizeof(h));
  uint64_t pos = head + sizeof(h);
  if (alen != 0) {
    ring_copy_in(r, pos, a, alen);
    pos += alen;
  }
  if (b != nullptr && !b->empty()) {
    std::string flat = b->to_string();  // records are small; one copy is fine
    ring_copy_in(r, pos, flat.data(), flat.size());
  }
  r.head->v.store(head + need, std::memory_order_release);
  return true;
}

// Consumer: pops one record. Returns 1 = got one, 0 = empty, -1 = the
// segment is corrupt (body_len from shared memory fails validation — any
// same-user process can scribble on the mapping; never trust it).
int ring_pop(Ring& r, RecHdr* h, std::string* body) {
  uint64_t tail = r.tail->v.load(std::memory_order_relaxed);
  uint64_t head = r.head->v.load(std::memory_order_acquire);
  if (head == tail) return 0;
  if (head - tail < sizeof(*h)) return -1;
  ring_copy_out(r, tail, h, sizeof(*h));
  // A valid producer never writes a record larger than half the ring
  // (ring_push refuses), and the record must fit in the published span.
  if (h->body_len > r.bytes / 2 || rec_size(h->body_len) > head - tail) return -1;
  body->resize(h->body_len);
  if (h->body_len != 0) ring_copy_out(r, tail + sizeof(*h), &(*body)[0], h->body_len);
  r.tail->v.store(tail + rec_size(h->body_len), std::memory_order_release);
  return 1;
}

// Adaptive wait: spin with pause, then nap. Returns false on stop().
// `closed` is SERVER-owned (set by StopShm): clients fail fast when the
// server goes away; a departing client must NOT touch it — other/later
// channels keep using the segment.
// Pollers run on DEDICATED pthreads (not fibers): a microsecond-latency
// transport wants a hot spin loop, and parking it inside the fiber worker
// pool starves RPC fibers (measured: 8-worker pool + spinning pollers
// collapsed concurrent shm QPS 4x). Spin, then nap via nanosleep.
template <typename HasWorkFn, typename StopFn>
bool poll_wait(HasWorkFn has_work, StopFn stop) {
  for (int spin = 0; spin < 4000; ++spin) {
    if (has_work()) return true;
    if (stop()) return false;
#if defined(__x86_64__)
    __builtin_ia32_pause();
#endif
  }
  // 20 µs naps while traffic is plausible, backing off to 1 ms after
  // ~50 ms idle so an idle (or leaked) connection's poller costs ~nothing.
  long nap_ns = 20000;
  int idle = 0;
  for (;;) {
    if (has_work()) return true;
    if (stop()) return false;
    struct timespec nap {0, nap_ns};
    nanosleep(&nap, nullptr);
    if (++idle > 2500 && nap_ns < 1000000) nap_ns = 1000000;
  }
}

struct Segment {
  SegHdr* hdr = nullptr;
  size_t map_len = 0;
  Ring req, resp;
  std::string shm_name;

  ~Segment() {
    if (hdr != nullptr) munmap(hdr, map_len);
  }

  bool map(const std::string& name, uint32_t ring_bytes, bool create) {
    shm_name = "/bam_shm_" + name;
    int fd;
    if (create) {
      shm_unlink(shm_name.c_str());
      fd = shm_open(shm_name.c_str(), O_CREAT | O_EXCL | O_RDWR, 0600);
    } else {
      fd = shm_open(shm_name.c_str(), O_RDWR, 0600);
    }
    if (fd < 0) return false;
    size_t len = sizeof(SegHdr) + 2ull * (create ? ring_bytes : 0);
    if (!create) {
      struct stat st;
      if (fstat(fd, &st) != 0 || (size_t)st.st_size < sizeof(SegHdr)) {
        close(fd);
        return false;
      }
      len = (size_t)st.st_size;
    } else if (ftruncate(fd, (off_t)len) != 0) {
      close(fd);
      return false;
    }
    void* m = mmap(nullptr, len, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
    close(fd);
    if (m == MAP_FAILED) return false;
    hdr = (SegHdr*)m;
    map_len = len;
    if (create) {
      memset(hdr, 0, sizeof(SegHdr));
      hdr->ring_bytes = ring_bytes;
      hdr->magic = kMagic;  // last: publishes the segment
    } else if (hdr->magic != kMagic) {
      return false;
    }
    char* store = (char*)hdr + sizeof(SegHdr);
    req = Ring{store, hdr->ring_bytes, &hdr->req_head, &hdr->req_tail};
    resp = Ring{store + hdr->ring_bytes, hdr->ring_bytes, &hdr->resp_head, &hdr->resp_tail};
    return true;
  }
};

// ---------------- server side ----------------

struct ShmServer {
  std::shared_ptr<Segment> seg;
  Server* server;
  std::mutex resp_mu;  // handlers (possibly async fibers) produce responses
  std::atomic<bool> stopping{false};
};

std::mutex g_servers_mu;
std::map<std::string, std::shared_ptr<ShmServer>> g_servers;

void shm_server_poll(void* raw) {
  std::shared_ptr<ShmServer>* sp = (std::shared_ptr<ShmServer>*)raw;
  std::shared_ptr<ShmServer> srv = *sp;
  delete sp;
  Segment& seg = *srv->seg;
  RecHdr h;
  std::string body;
  while (!srv->stopping.load(std::memory_order_acquire)) {
    if (!poll_wait(
            [&] {
              return seg.hdr->req_head.v.load(std::memory_order_acquire) !=
                     seg.hdr->req_tail.v.load(std::memory_order_relaxed);
            },
            [&] { return srv->stopping.load(std::memory_order_acquire); })) {
      break;
    }
    int got;
    while ((got = ring_pop(seg.req, &h, &body)) != 0) {
      if (got < 0) {
        LOG(ERROR) << "shm segment corrupt (bad record header); closing server ring";
        seg.hdr->closed.store(1, std::memory_order_release);
        srv->stopping.store(true, std::memory_order_release);
        break;
      }
      // body = method + payload
      size_t mlen = h.aux <= body.size() ? h.aux : body.size();
      std::string full_method = body.substr(0, mlen);
      size_t dot = full_method.find_last_of("./");
      std::string svc = dot == std::string::npos ? "" : full_method.substr(0, dot);
      std::string method = dot == std::string::npos ? full_method : full_method.substr(dot + 1);
      IOBuf request;
      request.append(body.data() + mlen, body.size() - mlen);
      const MethodFn* fn = srv->server->FindMethod(svc, method);
      uint64_t cid = h.cid;
      if (fn == nullptr) {
        std::string etext = "unknown method " + full_method;
        std::lock_guard<std::mutex> lk(srv->resp_mu);
        ring_push(seg.resp, cid, (uint32_t)ENOMETHOD, etext.data(), etext.size(), nullptr);
        continue;
      }
      // Inline execution (the reference's process-in-place mode): a ring
      // transport is for microsecond handlers; async handlers still work —
      // done may run later from another fiber (resp_mu serializes).
      Controller* cntl = new Controller;
      cntl->server_ = srv->server;
      IOBuf* resp = new IOBuf;
      std::shared_ptr<ShmServer> srv_ref = srv;
      Closure* done = NewCallback([srv_ref, cntl, resp, cid] {
        std::lock_guard<std::mutex> lk(srv_ref->resp_mu);
        Segment& sg = *srv_ref->seg;
        if (cntl->Failed()) {
          ring_push(sg.resp, cid, (uint32_t)cntl->ErrorCode(), cntl->ErrorText().data(),
                    cntl->ErrorText().size(), nullptr);
        } else {
          ring_push(sg.resp, cid, 0, nullptr, 0, resp);
        }
        srv_ref->server->nprocessed.fetch_add(1, std::memory_order_relaxed);
        delete resp;
        delete cntl;
      });
      (*fn)(cntl, request, resp, done);
    }
  }
  std::lock_guard<std::mutex> lk(g_servers_mu);
  // leave map entry removal to StopShm
}

}  // namespace

int ServeShm(const std::string& name, Server* server, uint32_t ring_bytes) {
  auto srv = std::make_shared<ShmServer>();
  srv->seg = std::make_shared<Segment>();
  srv->server = server;
  if (!srv->seg->map(name, ring_bytes, /*create=*/true)) return -1;
  {
    std::lock_guard<std::mutex> lk(g_servers_mu);
    if (g_servers.count(name) != 0) return -1;
    g_servers[name] = srv;
  }
  auto* arg = new std::shared_ptr<ShmServer>(srv);
  std::thread(shm_server_poll, arg).detach();
  return 0;
}

void StopShm(const std::string& name) {
  std::shared_ptr<ShmServer> srv;
  {
    std::lock_guard<std::mutex> lk(g_servers_mu);
    auto it = g_servers.find(name);
    if (it == g_servers.end()) return;
    srv = it->second;
    g_servers.erase(it);
  }
  srv->stopping.store(true, std::memory_order_release);
  srv->seg->hdr->closed.store(1, std::memory_order_release);
  shm_unlink(srv->seg->shm_name.c_str());
}

// ---------------- client side ----------------

struct PendingCall {
  std::atomic<bool> ready{false};  // spin target (set before ev.signal)
  CountdownEvent ev{1};
  IOBuf response;
  int error_code = 0;
  std::string error_text;
};

struct ShmChannel::Impl {
  Segment seg;
  std::mutex req_mu;  // concurrent Call()s produce requests
  static constexpr int kPendShards = 16;
  struct PendShard {
    std::mutex mu;
    std::map<uint64_t, std::shared_ptr<PendingCall>> map;
  };
  PendShard pend[kPendShards];
  PendShard& shard_of(uint64_t cid) { return pend[cid % kPendShards]; }
  std::atomic<uint64_t> next_cid{1};
  std::atomic<bool> stopping{false};
  std::atomic<bool> corrupt{false};  // poller saw an invalid record header
  CountdownEvent poller_exited{1};
};

namespace {
void shm_client_poll(void* raw) {
  ShmChannel::Impl* impl = (ShmChannel::Impl*)raw;
  RecHdr h;
  std::string body;
  while (!impl->stopping.load(std::memory_order_acquire)) {
    if (!poll_wait(
            [&] {
              return impl->seg.hdr->resp_head.v.load(std::memory_order_acquire) !=
                     impl->seg.hdr->resp_tail.v.load(std::memory_order_relaxed);
            },
            [&] {
              return impl->stopping.load(std::memory_order_acquire) ||
                     impl->seg.hdr->closed.load(std::memory_order_acquire) != 0;
            })) {
      break;
    }
    int got;
    while ((got = ring_pop(impl->seg.resp, &h, &body)) != 0) {
      if (got < 0) {
        LOG(ERROR) << "shm segment corrupt (bad record header); abandoning client ring";
        impl->corrupt.store(true, std::memory_order_release);
        impl->stopping.store(true, std::memory_order_release);
        break;
      }
      std::shared_ptr<PendingCall> pc;
      {
        auto& sh = impl->shard_of(h.cid);
        std::lock_guard<std::mutex> lk(sh.mu);
        auto it = sh.map.find(h.cid);
        if (it != sh.map.end()) {
          pc = it->second;
          sh.map.erase(it);
        }
      }
      if (!pc) continue;  // timed out and abandoned
      if (h.aux != 0) {
        pc->error_code = (int)h.aux;
        pc->error_text = body;
      } else {
        pc->response.append(body);
      }
      pc->ready.store(true, std::memory_order_release);
      pc->ev.signal();
    }
  }
  impl->poller_exited.signal();
}
}  // namespace

ShmChannel::~ShmChannel() {
  if (impl_ != nullptr) {
    impl_->stopping.store(true, std::memory_order_release);
    if (impl_->seg.hdr != nullptr) {
      impl_->poller_exited.timed_wait(monotonic_time_us() + 2000000);
    }
    delete impl_;
  }
}

int ShmChannel::Init(const std::string& name) {
  impl_ = new Impl;
  if (!impl_->seg.map(name, 0, /*create=*/false)) {
    delete impl_;
    impl_ = nullptr;
    return -1;
  }
  std::thread(shm_client_poll, impl_).detach();
  return 0;
}

int ShmChannel::Call(const std::string& full_method, const IOBuf& request, IOBuf* response,
                     int64_t timeout_us, std::string* error_text) {
  if (impl_ == nullptr) return EINTERNAL;
  uint64_t cid = impl_->next_cid.fetch_add(1, std::memory_order_relaxed);
  auto pc = std::make_shared<PendingCall>();
  {
    auto& sh = impl_->shard_of(cid);
    std::lock_guard<std::mutex> lk(sh.mu);
    sh.map[cid] = pc;
  }
  const int64_t deadline = monotonic_time_us() + timeout_us;
  bool pushed = false;
  for (;;) {
    {
      std::lock_guard<std::mutex> lk(impl_->req_mu);
      pushed = ring_push(impl_->seg.req, cid, (uint32_t)full_method.size(),
                         full_method.data(), full_method.size(), &request);
    }
    if (pushed) break;
    if (monotonic_time_us() > deadline ||
        impl_->corrupt.load(std::memory_order_acquire) ||
        impl_->seg.hdr->closed.load(std::memory_order_acquire) != 0) {
      auto& sh = impl_->shard_of(cid);
      std::lock_guard<std::mutex> lk(sh.mu);
      sh.map.erase(cid);
      if (error_text != nullptr) *error_text = "shm ring full / closed";
      return EOVERCROWDED;
    }
    fiber_usleep(5);  // backpressure: ring full
  }
  // Spin briefly before parking: responses on the shm path often land in
  // single-digit microseconds, far below a butex park/wake round-trip.
  // The pollers run on their own pthreads, so this spin only trades the
  // CALLER's worker slot.
  bool got = false;
  for (int spin = 0; spin < 2000 && !got; ++spin) {
    got = pc->ready.load(std::memory_order_acquire);
#if defined(__x86_64__)
    __builtin_ia32_pause();
#endif
  }
  if (!got && !pc->ev.timed_wait(deadline)) {
    auto& sh = impl_->shard_of(cid);
    std::lock_guard<std::mutex> lk(sh.mu);
    sh.map.erase(cid);
    if (error_text != nullptr) *error_text = "shm call timed out";
    return ERPCTIMEDOUT;
  }
  if (pc->error_code != 0) {
    if (error_text != nullptr) *error_text = pc->error_text;
    return pc->error_code;
  }
  if (response != nullptr) response->swap(pc->response);
  return 0;
}

}  // namespace shm
}  // namespace bam
