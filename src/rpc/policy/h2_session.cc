#include "rpc/policy/h2_session.h"

#include <string.h>

#include "base/logging.h"

namespace bam {
namespace policy {

namespace {

constexpr char kPreface[] = "PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n";
constexpr size_t kPrefaceLen = 24;

enum FrameType : uint8_t {
  F_DATA = 0,
  F_HEADERS = 1,
  F_PRIORITY = 2,
  F_RST_STREAM = 3,
  F_SETTINGS = 4,
  F_PUSH_PROMISE = 5,
  F_PING = 6,
  F_GOAWAY = 7,
  F_WINDOW_UPDATE = 8,
  F_CONTINUATION = 9,
};

enum Flags : uint8_t {
  FL_END_STREAM = 0x1,  // also SETTINGS/PING ACK
  FL_END_HEADERS = 0x4,
  FL_PADDED = 0x8,
  FL_PRIORITY = 0x20,
};

uint32_t rd_u32(const char* p) {
  return ((uint32_t)(uint8_t)p[0] << 24) | ((uint32_t)(uint8_t)p[1] << 16) |
         ((uint32_t)(uint8_t)p[2] << 8) | (uint8_t)p[3];
}

}  // namespace

H2Session::H2Session(bool server, Callbacks cbs)
    : server_(server), cbs_(std::move(cbs)), preface_done_(false),
      next_stream_id_(server ? 2 : 1) {}

void H2Session::frame_header(std::string* out, size_t len, uint8_t type, uint8_t flags,
                             int32_t sid) {
  out->push_back((char)(len >> 16));
  out->push_back((char)(len >> 8));
  out->push_back((char)len);
  out->push_back((char)type);
  out->push_back((char)flags);
  out->push_back((char)((sid >> 24) & 0x7f));
  out->push_back((char)(sid >> 16));
  out->push_back((char)(sid >> 8));
  out->push_back((char)sid);
}

void H2Session::ensure_preface() {
  if (sent_settings_) return;
  sent_settings_ = true;
  if (!server_) out_.append(kPreface, kPrefaceLen);
  // Our SETTINGS: a generous max concurrent streams + default windows.
  std::string payload;
  auto put_setting = [&](uint16_t id, uint32_t v) {
    payload.push_back((char)(id >> 8));
    payload.push_back((char)id);
    payload.push_back((char)(v >> 24));
    payload.push_back((char)(v >> 16));
    payload.push_back((char)(v >> 8));
    payload.push_back((char)v);
  };
  put_setting(3 /*MAX_CONCURRENT_STREAMS*/, 1024);
  put_setting(4 /*INITIAL_WINDOW_SIZE*/, 1u << 20);
  frame_header(&out_, payload.size(), F_SETTINGS, 0, 0);
  out_.append(payload);
  // Grow the connection receive window so bulk uploads never stall on us.
  send_window_update(0, (1u << 30) - 65535);
}

void H2Session::send_settings_ack() {
  frame_header(&out_, 0, F_SETTINGS, FL_END_STREAM /*ACK*/, 0);
}

void H2Session::send_window_update(int32_t sid, uint32_t increment) {
  frame_header(&out_, 4, F_WINDOW_UPDATE, 0, sid);
  out_.push_back((char)(increment >> 24));
  out_.push_back((char)(increment >> 16));
  out_.push_back((char)(increment >> 8));
  out_.push_back((char)increment);
}

ssize_t H2Session::Consume(const char* data, size_t n) {
  ensure_preface();
  const char* p = data;
  const char* end = data + n;
  if (server_ && !preface_done_) {
    if ((size_t)(end - p) < kPrefaceLen) return p - data;
    if (memcmp(p, kPreface, kPrefaceLen) != 0) return -1;
    p += kPrefaceLen;
    preface_done_ = true;
  }
  for (;;) {
    if ((size_t)(end - p) < 9) break;
    const size_t len = ((size_t)(uint8_t)p[0] << 16) | ((size_t)(uint8_t)p[1] << 8) |
                       (uint8_t)p[2];
    if (len > (16u << 20)) return -1;  // sanity
    if ((size_t)(end - p) < 9 + len) break;
    const uint8_t type = (uint8_t)p[3];
    const uint8_t flags = (uint8_t)p[4];
    const int32_t sid = (int32_t)(rd_u32(p + 5) & 0x7fffffff);
    if (!handle_frame(type, flags, sid, p + 9, len)) return -1;
    p += 9 + len;
  }
  return p - data;
}

bool H2Session::handle_frame(uint8_t type, uint8_t flags, int32_t sid, const char* p,
                             size_t len) {
  if (in_headers_ && type != F_CONTINUATION) return false;  // protocol error
  switch (type) {
    case F_DATA: {
      const char* body = p;
      size_t blen = len;
      if (flags & FL_PADDED) {
        if (blen < 1) return false;
        uint8_t pad = (uint8_t)p[0];
        if ((size_t)pad + 1 > blen) return false;
        body += 1;
        blen -= 1 + pad;
      }
      if (blen > 0 && cbs_.on_data) cbs_.on_data(sid, body, blen);
      // Replenish both windows immediately (we buffer upstream).
      if (len > 0) {
        send_window_update(0, (uint32_t)len);
        send_window_update(sid, (uint32_t)len);
      }
      if (flags & FL_END_STREAM) {
        if (cbs_.on_end_stream) cbs_.on_end_stream(sid);
      }
      return true;
    }
    case F_HEADERS: {
      const char* block = p;
      size_t blen = len;
      if (flags & FL_PADDED) {
        if (blen < 1) return false;
        uint8_t pad = (uint8_t)block[0];
        block += 1;
        if ((size_t)pad + 1 > len) return false;
        blen -= 1 + pad;
      }
      if (flags & FL_PRIORITY) {
        if (blen < 5) return false;
        block += 5;
        blen -= 5;
      }
      hdr_sid_ = sid;
      hdr_block_.assign(block, blen);
      hdr_end_stream_ = (flags & FL_END_STREAM) != 0;
      if (flags & FL_END_HEADERS) {
        std::vector<hpack::Header> hs;
        if (!hdec_.Decode(hdr_block_.data(), hdr_block_.size(), &hs)) return false;
        if (cbs_.on_header)
          for (auto& h : hs) cbs_.on_header(sid, h.first, h.second);
        if (hdr_end_stream_ && cbs_.on_end_stream) cbs_.on_end_stream(sid);
      } else {
        in_headers_ = true;
      }
      return true;
    }
    case F_CONTINUATION: {
      if (!in_headers_ || sid != hdr_sid_) return false;
      hdr_block_.append(p, len);
      if (flags & FL_END_HEADERS) {
        in_headers_ = false;
        std::vector<hpack::Header> hs;
        if (!hdec_.Decode(hdr_block_.data(), hdr_block_.size(), &hs)) return false;
        if (cbs_.on_header)
          for (auto& h : hs) cbs_.on_header(sid, h.first, h.second);
        if (hdr_end_stream_ && cbs_.on_end_stream) cbs_.on_end_stream(sid);
      }
      return true;
    }
    case F_SETTINGS: {
      if (flags & FL_END_STREAM) return true;  // their ACK of ours
      if (len % 6 != 0) return false;
      for (size_t off = 0; off + 6 <= len; off += 6) {
        uint16_t id = ((uint16_t)(uint8_t)p[off] << 8) | (uint8_t)p[off + 1];
        uint32_t v = rd_u32(p + off + 2);
        if (id == 4 /*INITIAL_WINDOW_SIZE*/) {
          int64_t delta = (int64_t)v - peer_initial_window_;
          peer_initial_window_ = (int32_t)v;
          for (auto& kv : send_streams_) kv.second.window += delta;
        } else if (id == 5 /*MAX_FRAME_SIZE*/) {
          if (v >= 16384 && v <= (16u << 20)) peer_max_frame_ = v;
        }
      }
      send_settings_ack();
      pump_all();
      return true;
    }
    case F_PING: {
      if (!(flags & FL_END_STREAM) && len == 8) {
        frame_header(&out_, 8, F_PING, FL_END_STREAM, 0);
        out_.append(p, 8);
      }
      return true;
    }
    case F_WINDOW_UPDATE: {
      if (len != 4) return false;
      uint32_t inc = rd_u32(p) & 0x7fffffff;
      if (sid == 0) {
        peer_conn_window_ += inc;
      } else {
        auto it = send_streams_.find(sid);
        if (it != send_streams_.end()) it->second.window += inc;
      }
      pump_all();
      return true;
    }
    case F_RST_STREAM: {
      if (len != 4) return false;
      send_streams_.erase(sid);
      if (cbs_.on_rst) cbs_.on_rst(sid, rd_u32(p));
      return true;
    }
    case F_GOAWAY: {
      if (cbs_.on_goaway) cbs_.on_goaway(len >= 8 ? rd_u32(p + 4) : 0);
      return true;
    }
    case F_PRIORITY:
    case F_PUSH_PROMISE:
    default:
      return true;  // ignore
  }
}

int32_t H2Session::SubmitRequest(const std::vector<hpack::Header>& headers,
                                 const std::string& body, bool end_stream) {
  ensure_preface();
  int32_t sid = next_stream_id_;
  next_stream_id_ += 2;
  std::string block;
  henc_.Encode(headers, &block);
  const bool has_body = !body.empty();
  frame_header(&out_, block.size(), F_HEADERS,
               FL_END_HEADERS | (has_body || !end_stream ? 0 : FL_END_STREAM), sid);
  out_.append(block);
  SendStream& ss = send_streams_[sid];
  ss.window = peer_initial_window_;
  ss.headers_sent = true;
  if (has_body) {
    ss.body = body;
    ss.end_stream_after_body = end_stream;
    pump_order_.push_back(sid);
    pump_stream(sid, ss);
  } else if (!end_stream) {
    // caller streams data later (not used yet)
  }
  return sid;
}

void H2Session::SubmitResponse(int32_t sid, const std::vector<hpack::Header>& headers,
                               const std::string& body,
                               const std::vector<hpack::Header>& trailers,
                               bool send_trailers) {
  ensure_preface();
  std::string block;
  henc_.Encode(headers, &block);
  const bool more = !body.empty() || send_trailers;
  frame_header(&out_, block.size(), F_HEADERS, FL_END_HEADERS | (more ? 0 : FL_END_STREAM),
               sid);
  out_.append(block);
  SendStream& ss = send_streams_[sid];
  if (ss.window == 0 && !ss.headers_sent) ss.window = peer_initial_window_;
  ss.headers_sent = true;
  ss.body = body;
  ss.off = 0;
  ss.end_stream_after_body = !send_trailers;
  if (send_trailers) {
    ss.has_trailers = true;
    henc_.Encode(trailers, &ss.trailer_block);
  }
  if (more) {
    pump_order_.push_back(sid);
    pump_stream(sid, ss);
  } else {
    send_streams_.erase(sid);
  }
}

void H2Session::SubmitRstStream(int32_t sid, uint32_t error) {
  frame_header(&out_, 4, F_RST_STREAM, 0, sid);
  out_.push_back((char)(error >> 24));
  out_.push_back((char)(error >> 16));
  out_.push_back((char)(error >> 8));
  out_.push_back((char)error);
  send_streams_.erase(sid);
}

void H2Session::SubmitGoaway(uint32_t error) {
  frame_header(&out_, 8, F_GOAWAY, 0, 0);
  int32_t last = next_stream_id_;
  out_.push_back((char)(last >> 24));
  out_.push_back((char)(last >> 16));
  out_.push_back((char)(last >> 8));
  out_.push_back((char)last);
  out_.push_back((char)(error >> 24));
  out_.push_back((char)(error >> 16));
  out_.push_back((char)(error >> 8));
  out_.push_back((char)error);
}

void H2Session::pump_stream(int32_t sid, SendStream& ss) {
  while (ss.off < ss.body.size()) {
    size_t left = ss.body.size() - ss.off;
    int64_t allowed = peer_conn_window_ < ss.window ? peer_conn_window_ : ss.window;
    if (allowed <= 0) return;  // flow controlled; resume on WINDOW_UPDATE
    size_t chunk = left;
    if ((int64_t)chunk > allowed) chunk = (size_t)allowed;
    if (chunk > peer_max_frame_) chunk = peer_max_frame_;
    const bool last = ss.off + chunk == ss.body.size();
    const bool end_stream = last && ss.end_stream_after_body;
    frame_header(&out_, chunk, F_DATA, end_stream ? FL_END_STREAM : 0, sid);
    out_.append(ss.body.data() + ss.off, chunk);
    ss.off += chunk;
    peer_conn_window_ -= (int64_t)chunk;
    ss.window -= (int64_t)chunk;
  }
  if (ss.off >= ss.body.size()) {
    if (ss.has_trailers) {
      frame_header(&out_, ss.trailer_block.size(), F_HEADERS,
                   FL_END_HEADERS | FL_END_STREAM, sid);
      out_.append(ss.trailer_block);
      ss.has_trailers = false;
    }
    send_streams_.erase(sid);
  }
}

void H2Session::pump_all() {
  for (size_t i = 0; i < pump_order_.size();) {
    int32_t sid = pump_order_[i];
    auto it = send_streams_.find(sid);
    if (it == send_streams_.end()) {
      pump_order_.erase(pump_order_.begin() + i);
      continue;
    }
    pump_stream(sid, it->second);
    if (send_streams_.find(sid) == send_streams_.end()) {
      pump_order_.erase(pump_order_.begin() + i);
    } else {
      ++i;  // still blocked on flow control
    }
  }
}

void H2Session::TakeOutput(std::string* out) {
  ensure_preface();
  out->append(out_);
  out_.clear();
}

}  // namespace policy
}  // namespace bam
