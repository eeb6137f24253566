#include "rpc/stream.h"

#include <deque>
#include <mutex>

#include <string.h>

#include "base/logging.h"
#include "base/resource_pool.h"
#include "fiber/butex.h"
#include "fiber/fiber.h"
#include "rpc/comm_group.h"
#include "rpc/controller.h"
#include "rpc/socket.h"
#include "rpc/wire.h"

namespace bam {

namespace {

enum FrameType { FRAME_DATA = 0, FRAME_CLOSE = 1, FRAME_FEEDBACK = 2, FRAME_GPU_DATA = 3 };
const char kStreamMagic[4] = {'S', 'T', 'R', 'M'};
const size_t kFrameHeaderLen = 28;

struct StreamMeta {
  std::mutex mu;
  std::mutex write_mu;  // serializes writers (frames must stay ordered)
  std::atomic<uint32_t> version{1};
  bool closed = true;
  bool connected = false;
  uint64_t remote_sid = 0;
  SocketId socket_id = 0;
  StreamOptions opt;
  // send window
  int64_t written_bytes = 0;
  std::atomic<int64_t> acked_bytes{0};
  std::atomic<int>* window_butex = nullptr;  // bumped on feedback / close
  // recv
  std::deque<IOBuf> pending;
  bool consumer_running = false;
  int64_t consumed_total = 0;
  int64_t feedback_sent_at = 0;
  std::atomic<int>* closed_butex = nullptr;  // bumped on close
};

inline ResourceId rid_of(StreamId id) { return (uint32_t)(id & 0xffffffffu) - 1; }
inline uint32_t ver_of(StreamId id) { return (uint32_t)(id >> 32); }

StreamMeta* meta_of(StreamId id) {
  if (id == 0) return nullptr;
  StreamMeta* m = address_resource<StreamMeta>(rid_of(id));
  if (m == nullptr || m->version.load(std::memory_order_acquire) != ver_of(id)) return nullptr;
  return m;
}

StreamId alloc_stream(const StreamOptions& opt) {
  ResourceId rid;
  StreamMeta* m = get_resource<StreamMeta>(&rid);
  if (m == nullptr) return 0;
  if (m->window_butex == nullptr) {
    m->window_butex = butex_create();
    m->closed_butex = butex_create();
  }
  std::lock_guard<std::mutex> lk(m->mu);
  m->closed = false;
  m->connected = false;
  m->remote_sid = 0;
  m->socket_id = 0;
  m->opt = opt;
  m->written_bytes = 0;
  m->acked_bytes.store(0, std::memory_order_relaxed);
  m->pending.clear();
  m->consumer_running = false;
  m->consumed_total = 0;
  m->feedback_sent_at = 0;
  return ((uint64_t)m->version.load(std::memory_order_relaxed) << 32) | (rid + 1);
}

void pack_frame(IOBuf* out, int type, uint64_t dst_sid, uint64_t aux, IOBuf* payload) {
  char h[kFrameHeaderLen];
  memcpy(h, kStreamMagic, 4);
  h[4] = (char)type;
  h[5] = h[6] = h[7] = 0;
  uint32_t len = payload != nullptr ? (uint32_t)payload->size() : 0;
  wire::put_u32_be(h + 8, len);
  wire::put_u32_be(h + 12, (uint32_t)(dst_sid >> 32));
  wire::put_u32_be(h + 16, (uint32_t)dst_sid);
  wire::put_u32_be(h + 20, (uint32_t)(aux >> 32));
  wire::put_u32_be(h + 24, (uint32_t)aux);
  out->append(h, kFrameHeaderLen);
  if (payload != nullptr) out->append(std::move(*payload));
}

int send_frame(SocketId sock_id, int type, uint64_t dst_sid, uint64_t aux, IOBuf* payload) {
  SocketUniquePtr sock;
  if (Socket::Address(sock_id, &sock) != 0 || sock->Failed()) return EPIPE;
  IOBuf frame;
  pack_frame(&frame, type, dst_sid, aux, payload);
  return sock->Write(&frame) == 0 ? 0 : EPIPE;
}

// closes locally: wakes writers/waiters, runs on_closed. mu must NOT be held.
void close_local(StreamId sid, bool notify_peer) {
  StreamMeta* m = meta_of(sid);
  if (m == nullptr) return;
  std::function<void(StreamId)> on_closed;
  uint64_t remote = 0;
  SocketId sock = 0;
  {
    std::lock_guard<std::mutex> lk(m->mu);
    if (m->closed) return;
    m->closed = true;
    on_closed = m->opt.on_closed;
    remote = m->remote_sid;
    sock = m->socket_id;
  }
  if (notify_peer && remote != 0) send_frame(sock, FRAME_CLOSE, remote, 0, nullptr);
  m->window_butex->fetch_add(1, std::memory_order_release);
  butex_wake_all(m->window_butex);
  m->closed_butex->fetch_add(1, std::memory_order_release);
  butex_wake_all(m->closed_butex);
  if (on_closed) on_closed(sid);
}

// ---- consumer fiber ----

void consumer_fiber(void* arg) {
  StreamId sid = (StreamId)(uintptr_t)arg;
  for (;;) {
    StreamMeta* m = meta_of(sid);  // re-validate: stream may be recycled
    if (m == nullptr) return;
    IOBuf msg;
    bool has = false;
    std::function<void(StreamId, IOBuf*)> on_received;
    {
      std::lock_guard<std::mutex> lk(m->mu);
      if (!m->pending.empty()) {
        msg.swap(m->pending.front());
        m->pending.pop_front();
        has = true;
        on_received = m->opt.on_received;
      } else {
        m->consumer_running = false;
        return;
      }
    }
    size_t n = msg.size();
    if (on_received) on_received(sid, &msg);
    // feedback accounting
    bool send_fb = false;
    uint64_t remote = 0;
    SocketId sock = 0;
    int64_t total = 0;
    {
      std::lock_guard<std::mutex> lk(m->mu);
      m->consumed_total += (int64_t)n;
      total = m->consumed_total;
      if (total - m->feedback_sent_at >= (int64_t)m->opt.max_buf_size / 2) {
        m->feedback_sent_at = total;
        send_fb = true;
        remote = m->remote_sid;
        sock = m->socket_id;
      }
    }
    if (send_fb && remote != 0) send_frame(sock, FRAME_FEEDBACK, remote, (uint64_t)total, nullptr);
  }
}

}  // namespace

// ---------------- public API ----------------

int StreamCreate(StreamId* sid, Controller* cntl, const StreamOptions& opt) {
  StreamId s = alloc_stream(opt);
  if (s == 0) return ENOMEM;
  *sid = s;
  cntl->call.stream_id = s;
  return 0;
}

int StreamAccept(StreamId* sid, Controller* cntl, const StreamOptions& opt) {
  if (cntl->remote_stream_id_ == 0) return EINVAL;  // request carried no stream
  StreamId s = alloc_stream(opt);
  if (s == 0) return ENOMEM;
  StreamMeta* m = meta_of(s);
  {
    std::lock_guard<std::mutex> lk(m->mu);
    m->remote_sid = cntl->remote_stream_id_;
    m->socket_id = cntl->server_socket_;
    m->connected = true;
  }
  cntl->response_stream_id_ = s;
  *sid = s;
  return 0;
}

namespace stream_internal {

int ConnectLocalStream(StreamId local, uint64_t remote_sid, uint64_t socket_id) {
  StreamMeta* m = meta_of(local);
  if (m == nullptr) return EINVAL;
  std::lock_guard<std::mutex> lk(m->mu);
  m->remote_sid = remote_sid;
  m->socket_id = socket_id;
  m->connected = true;
  return 0;
}

void OnStreamFrame(uint64_t dst_sid, int type, uint64_t aux, IOBuf* payload,
                   uint64_t socket_id) {
  StreamMeta* m = meta_of(dst_sid);
  if (m == nullptr) return;
  switch (type) {
    case FRAME_DATA: {
      bool start = false;
      {
        std::lock_guard<std::mutex> lk(m->mu);
        if (m->closed) return;
        m->pending.push_back(std::move(*payload));
        if (!m->consumer_running) {
          m->consumer_running = true;
          start = true;
        }
      }
      if (start) {
        fiber_t th;
        if (fiber_start_background(&th, consumer_fiber, (void*)(uintptr_t)dst_sid) != 0) {
          consumer_fiber((void*)(uintptr_t)dst_sid);
        }
      }
      break;
    }
    case FRAME_GPU_DATA: {
      // aux = payload bytes, in flight as an RCCL p2p send from the peer.
      // Land it straight into a fresh HBM block (zero host staging), then
      // deliver like any DATA message. The Recv PARKS this fiber until the
      // xGMI transfer completes — dedicate the socket to streaming when
      // using the GPU leg (head-of-line applies, as with any byte stream).
      CommGroup* grp = nullptr;
      int peer = -1;
      {
        std::lock_guard<std::mutex> lk(m->mu);
        if (m->closed) return;
        grp = m->opt.gpu_group;
        peer = m->opt.gpu_peer;
      }
      if (grp == nullptr || peer < 0) {
        LOG(ERROR) << "GPU stream frame on a stream without a gpu_group";
        close_local(dst_sid, true);
        return;
      }
      IOBuf buf;
      void* dst_ptr = nullptr;
      const Residency res = grp->backend() == "rccl" ? RES_HBM : RES_HOST;
      if (buf.append_writable_block((size_t)aux, res, 0, &dst_ptr) != 0 ||
          grp->Recv(dst_ptr, (size_t)aux, peer) != 0) {
        close_local(dst_sid, true);
        return;
      }
      bool start = false;
      {
        std::lock_guard<std::mutex> lk(m->mu);
        if (m->closed) return;
        m->pending.push_back(std::move(buf));
        if (!m->consumer_running) {
          m->consumer_running = true;
          start = true;
        }
      }
      if (start) {
        fiber_t th;
        if (fiber_start_background(&th, consumer_fiber, (void*)(uintptr_t)dst_sid) != 0) {
          consumer_fiber((void*)(uintptr_t)dst_sid);
        }
      }
      break;
    }
    case FRAME_FEEDBACK: {
      int64_t prev = m->acked_bytes.load(std::memory_order_relaxed);
      if ((int64_t)aux > prev) m->acked_bytes.store((int64_t)aux, std::memory_order_release);
      m->window_butex->fetch_add(1, std::memory_order_release);
      butex_wake_all(m->window_butex);
      break;
    }
    case FRAME_CLOSE:
      close_local(dst_sid, /*notify_peer=*/false);
      break;
    default:
      break;
  }
}

}  // namespace stream_internal

int StreamWrite(StreamId sid, IOBuf* data) {
  StreamMeta* m = meta_of(sid);
  if (m == nullptr) return EINVAL;
  // Serialize writers: window reservation and the socket write must stay
  // in the same order or frames would interleave.
  std::lock_guard<std::mutex> wlk(m->write_mu);
  if (meta_of(sid) != m) return EINVAL;  // recycled while acquiring
  size_t n = data->size();
  uint64_t remote = 0;
  SocketId sock = 0;
  for (;;) {
    int v = m->window_butex->load(std::memory_order_acquire);
    {
      std::lock_guard<std::mutex> lk(m->mu);
      if (m->closed || !m->connected) return EINVAL;
      if (m->written_bytes - m->acked_bytes.load(std::memory_order_acquire) + (int64_t)n <=
          (int64_t)m->opt.max_buf_size) {
        m->written_bytes += (int64_t)n;
        remote = m->remote_sid;
        sock = m->socket_id;
        break;
      }
    }
    butex_wait(m->window_butex, v, nullptr);  // woken by feedback or close
  }
  // xGMI leg: a single HBM-resident span travels as RCCL p2p; only the
  // size descriptor rides TCP. The descriptor write and the RCCL send are
  // ordered together (per-group mutex) so the receiver's posting order —
  // descriptor arrival order — matches RCCL's send matching order even
  // with several streams sharing one group.
  if (m->opt.gpu_group != nullptr && m->opt.gpu_peer >= 0 && data->backing_block_num() == 1) {
    IOBuf::Span sp = data->span_at(0);
    // rccl groups move HBM spans only; tcp groups (tests / no-GPU
    // fallback) take any single span through the same leg.
    if (sp.res == RES_HBM || m->opt.gpu_group->backend() == "tcp") {
      static std::mutex g_gpu_send_mu;
      std::lock_guard<std::mutex> glk(g_gpu_send_mu);
      IOBuf frame;
      pack_frame(&frame, FRAME_GPU_DATA, remote, (uint64_t)n, nullptr);
      SocketUniquePtr s;
      if (Socket::Address(sock, &s) != 0 || s->Failed() || s->Write(&frame) != 0 ||
          m->opt.gpu_group->Send(sp.data, n, m->opt.gpu_peer) != 0) {
        close_local(sid, false);
        return EPIPE;
      }
      data->clear();
      return 0;
    }
  }
  IOBuf frame;
  pack_frame(&frame, FRAME_DATA, remote, 0, data);
  SocketUniquePtr s;
  if (Socket::Address(sock, &s) != 0 || s->Failed() || s->Write(&frame) != 0) {
    close_local(sid, false);
    return EPIPE;
  }
  return 0;
}

int StreamClose(StreamId sid) {
  close_local(sid, /*notify_peer=*/true);
  StreamMeta* m = meta_of(sid);
  if (m != nullptr) {
    // Invalidate the id (streams are single-use).
    m->version.fetch_add(1, std::memory_order_acq_rel);
    return_resource<StreamMeta>(rid_of(sid));
  }
  return 0;
}

int StreamWait(StreamId sid) {
  for (;;) {
    StreamMeta* m = meta_of(sid);
    if (m == nullptr) return 0;
    int v;
    {
      std::lock_guard<std::mutex> lk(m->mu);
      if (m->closed) return 0;
      v = m->closed_butex->load(std::memory_order_acquire);
    }
    butex_wait(m->closed_butex, v, nullptr);
  }
}

bool StreamExists(StreamId sid) { return meta_of(sid) != nullptr; }


// ---------------- wire protocol ("STRM" frames) ----------------

namespace stream_internal {

namespace {

struct StreamFrameMessage : public InputMessageBase {
  int type = 0;
  uint64_t dst_sid = 0;
  uint64_t aux = 0;
  IOBuf payload;
};

ParseResult ParseStreamFrame(IOBuf* source, Socket* /*sock*/, bool /*eof*/) {
  char auxbuf[kFrameHeaderLen];
  if (source->size() < kFrameHeaderLen)
    return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  const char* h = (const char*)source->fetch(auxbuf, kFrameHeaderLen);
  if (h == nullptr || memcmp(h, kStreamMagic, 4) != 0)
    return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  uint32_t len = wire::get_u32_be(h + 8);
  if (len > (256u << 20)) return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
  if (source->size() < kFrameHeaderLen + len)
    return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  StreamFrameMessage* msg = new StreamFrameMessage;
  msg->type = h[4];
  msg->dst_sid = ((uint64_t)wire::get_u32_be(h + 12) << 32) | wire::get_u32_be(h + 16);
  msg->aux = ((uint64_t)wire::get_u32_be(h + 20) << 32) | wire::get_u32_be(h + 24);
  source->pop_front(kFrameHeaderLen);
  source->cutn(&msg->payload, len);
  return ParseResult::make_ok(msg);
}

void ProcessStreamFrame(InputMessageBase* msg_base) {
  StreamFrameMessage* msg = (StreamFrameMessage*)msg_base;
  OnStreamFrame(msg->dst_sid, msg->type, msg->aux, &msg->payload, msg->socket_id);
  delete msg;
}

}  // namespace

void RegisterStreamProtocol() {
  static std::once_flag flag;
  std::call_once(flag, [] {
    Protocol p;
    p.parse = ParseStreamFrame;
    p.process_request = ProcessStreamFrame;   // frames flow both directions
    p.process_response = ProcessStreamFrame;
    p.support_server = true;
    p.support_client = true;
    p.name = "strm";
    RegisterProtocol(p);
  });
}

}  // namespace stream_internal

}  // namespace bam
