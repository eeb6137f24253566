// brpc_amd: RTMP chunk-stream reader/writer shared by the server protocol
// (policy/rtmp_protocol.cc) and the client (rtmp_client.cc).
// Parity: reference policy/rtmp_protocol.cpp chunk handling (clean-room):
// basic header (fmt 2b + csid 6b, 1-3 bytes), message headers fmt0(11)/
// fmt1(7)/fmt2(3)/fmt3(0), extended timestamp, per-csid reassembly,
// SetChunkSize(1) updates the reader inline.
#pragma once

#include <stdint.h>
#include <string.h>

#include <map>
#include <string>

namespace bam {
namespace rtmp {

constexpr uint8_t kMsgSetChunkSize = 1;
constexpr uint8_t kMsgAck = 3;
constexpr uint8_t kMsgUserControl = 4;
constexpr uint8_t kMsgWindowAckSize = 5;
constexpr uint8_t kMsgSetPeerBandwidth = 6;
constexpr uint8_t kMsgAudio = 8;
constexpr uint8_t kMsgVideo = 9;
constexpr uint8_t kMsgDataAmf0 = 18;
constexpr uint8_t kMsgCommandAmf0 = 20;

struct Message {
  uint8_t type = 0;
  uint32_t timestamp = 0;
  uint32_t stream_id = 0;  // message stream id
  std::string payload;
};

class ChunkReader {
 public:
  uint32_t in_chunk_size = 128;

  void feed(const char* data, size_t n) { buf_.append(data, n); }
  size_t buffered() const { return buf_.size(); }

  // Extracts one complete message. Returns 1 = message, 0 = need more
  // data, -1 = protocol error. Handles SetChunkSize internally AND
  // returns it (callers may ignore).
  int next(Message* out) {
    for (;;) {
      size_t pos = 0;
      const char* p = buf_.data();
      const size_t n = buf_.size();
      if (n < 1) return 0;
      uint8_t b0 = (uint8_t)p[pos++];
      uint8_t fmt = b0 >> 6;
      uint32_t csid = b0 & 0x3f;
      if (csid == 0) {
        if (n < pos + 1) return 0;
        csid = 64 + (uint8_t)p[pos++];
      } else if (csid == 1) {
        if (n < pos + 2) return 0;
        csid = 64 + (uint8_t)p[pos] + 256u * (uint8_t)p[pos + 1];
        pos += 2;
      }
      Cs& cs = cs_[csid];
      uint32_t ts = cs.ts;
      if (fmt == 0) {
        if (n < pos + 11) return 0;
        ts = u24(p + pos);
        cs.len = u24(p + pos + 3);
        cs.type = (uint8_t)p[pos + 6];
        memcpy(&cs.msid, p + pos + 7, 4);  // little-endian on the wire
        pos += 11;
        cs.ts_delta = 0;
      } else if (fmt == 1) {
        if (n < pos + 7) return 0;
        cs.ts_delta = u24(p + pos);
        cs.len = u24(p + pos + 3);
        cs.type = (uint8_t)p[pos + 6];
        pos += 7;
        ts = cs.ts + cs.ts_delta;
      } else if (fmt == 2) {
        if (n < pos + 3) return 0;
        cs.ts_delta = u24(p + pos);
        pos += 3;
        ts = cs.ts + cs.ts_delta;
      } else {  // fmt 3: continuation or repeat
        if (cs.got == 0) ts = cs.ts + cs.ts_delta;
      }
      // Latch the message timestamp on its FIRST chunk; continuation
      // chunks (fmt3 mid-message) must not recompute it.
      if (cs.got == 0) cs.cur_ts = ts;
      bool ext_ts = false;
      if ((fmt == 0 && ts == 0xffffff) ||
          ((fmt == 1 || fmt == 2) && cs.ts_delta == 0xffffff)) {
        if (n < pos + 4) return 0;
        ts = ((uint32_t)(uint8_t)p[pos] << 24) | ((uint32_t)(uint8_t)p[pos + 1] << 16) |
             ((uint32_t)(uint8_t)p[pos + 2] << 8) | (uint8_t)p[pos + 3];
        pos += 4;
        ext_ts = true;
        if (cs.got == 0) cs.cur_ts = ts;
      }
      (void)ext_ts;
      if (cs.len > (64u << 20)) return -1;
      uint32_t remain = cs.len - cs.got;
      uint32_t take = remain < in_chunk_size ? remain : in_chunk_size;
      if (n < pos + take) return 0;
      cs.partial.append(p + pos, take);
      cs.got += take;
      pos += take;
      buf_.erase(0, pos);
      if (cs.got < cs.len) continue;  // more chunks of this message
      cs.ts = cs.cur_ts;
      out->type = (uint8_t)cs.type;
      out->timestamp = cs.cur_ts;
      out->stream_id = cs.msid;
      out->payload.swap(cs.partial);
      cs.partial.clear();
      cs.got = 0;
      if (out->type == kMsgSetChunkSize && out->payload.size() >= 4) {
        in_chunk_size = ((uint32_t)(uint8_t)out->payload[0] << 24) |
                        ((uint32_t)(uint8_t)out->payload[1] << 16) |
                        ((uint32_t)(uint8_t)out->payload[2] << 8) |
                        (uint8_t)out->payload[3];
        if (in_chunk_size == 0 || in_chunk_size > (16u << 20)) return -1;
      }
      return 1;
    }
  }

 private:
  struct Cs {
    uint32_t ts = 0, ts_delta = 0, len = 0, type = 0, msid = 0, got = 0, cur_ts = 0;
    std::string partial;
  };
  static uint32_t u24(const char* p) {
    return ((uint32_t)(uint8_t)p[0] << 16) | ((uint32_t)(uint8_t)p[1] << 8) | (uint8_t)p[2];
  }
  std::string buf_;
  std::map<uint32_t, Cs> cs_;
};

// Serializes one message as fmt0 + fmt3 continuation chunks.
inline void write_message(std::string* out, uint32_t csid, uint8_t type, uint32_t msid,
                          uint32_t ts, const std::string& payload, uint32_t chunk_size) {
  size_t off = 0;
  bool first = true;
  do {
    if (first) {
      out->push_back((char)(0x00 | (csid & 0x3f)));  // fmt0 (csid < 64 assumed)
      uint32_t wts = ts >= 0xffffff ? 0xffffff : ts;
      out->push_back((char)(wts >> 16));
      out->push_back((char)(wts >> 8));
      out->push_back((char)wts);
      out->push_back((char)(payload.size() >> 16));
      out->push_back((char)(payload.size() >> 8));
      out->push_back((char)payload.size());
      out->push_back((char)type);
      out->append((const char*)&msid, 4);  // LE
      if (ts >= 0xffffff) {
        out->push_back((char)(ts >> 24));
        out->push_back((char)(ts >> 16));
        out->push_back((char)(ts >> 8));
        out->push_back((char)ts);
      }
      first = false;
    } else {
      out->push_back((char)(0xc0 | (csid & 0x3f)));  // fmt3
    }
    size_t take = payload.size() - off < chunk_size ? payload.size() - off : chunk_size;
    out->append(payload.data() + off, take);
    off += take;
  } while (off < payload.size());
}

inline std::string u32_be(uint32_t v) {
  std::string s;
  s.push_back((char)(v >> 24));
  s.push_back((char)(v >> 16));
  s.push_back((char)(v >> 8));
  s.push_back((char)v);
  return s;
}

}  // namespace rtmp
}  // namespace bam
