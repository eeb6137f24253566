// brpc_amd: RTMP client implementation (see rtmp_client.h).
#include "rpc/rtmp_client.h"

#include <errno.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <poll.h>
#include <string.h>
#include <sys/socket.h>
#include <unistd.h>

#include <vector>

#include "base/endpoint.h"
#include "base/fast_rand.h"
#include "base/time.h"
#include "rpc/amf.h"

namespace bam {

namespace {
constexpr size_t kHsSize = 1536;
}

RtmpClient::~RtmpClient() { Close(); }

void RtmpClient::Close() {
  if (fd_ >= 0) {
    ::close(fd_);
    fd_ = -1;
  }
}

int RtmpClient::send_bytes(const std::string& bytes) {
  size_t off = 0;
  while (off < bytes.size()) {
    ssize_t w = ::send(fd_, bytes.data() + off, bytes.size() - off, MSG_NOSIGNAL);
    if (w <= 0) {
      if (errno == EINTR || errno == EAGAIN) continue;
      return -1;
    }
    off += (size_t)w;
  }
  return 0;
}

int RtmpClient::read_more(int timeout_ms) {
  struct pollfd pfd{fd_, POLLIN, 0};
  int pr = ::poll(&pfd, 1, timeout_ms);
  if (pr <= 0) return -1;
  char buf[65536];
  ssize_t r = ::recv(fd_, buf, sizeof(buf), 0);
  if (r <= 0) return -1;
  reader_.feed(buf, (size_t)r);
  return 0;
}

int RtmpClient::send_command(const std::string& payload, uint32_t msid) {
  std::string bytes;
  rtmp::write_message(&bytes, 3, rtmp::kMsgCommandAmf0, msid, 0, payload, out_chunk_);
  return send_bytes(bytes);
}

int RtmpClient::wait_command_reply(const std::string& expect_cmd, int timeout_ms) {
  const int64_t deadline = monotonic_time_us() + (int64_t)timeout_ms * 1000;
  for (;;) {
    rtmp::Message m;
    int rc = reader_.next(&m);
    if (rc < 0) return -1;
    if (rc == 1) {
      if (m.type == rtmp::kMsgCommandAmf0) {
        std::vector<amf::Value> vals;
        if (amf::DecodeAll(m.payload.data(), m.payload.size(), &vals) && !vals.empty() &&
            vals[0].type == amf::Value::STRING) {
          if (vals[0].str == expect_cmd) return 0;
          if (vals[0].str == "_error") return -1;
          if (expect_cmd == "onStatus" && vals[0].str == "onStatus") return 0;
        }
      }
      continue;  // control messages etc.
    }
    int64_t left_us = deadline - monotonic_time_us();
    if (left_us <= 0) return -1;
    if (read_more((int)(left_us / 1000) + 1) != 0) return -1;
  }
}

int RtmpClient::Connect(const std::string& host, int port, const std::string& app,
                        int timeout_ms) {
  Close();
  timeout_ms_ = timeout_ms;
  EndPoint ep;
  if (hostname2endpoint(host.c_str(), port, &ep) != 0) return -1;
  fd_ = ::socket(AF_INET, SOCK_STREAM | SOCK_CLOEXEC, 0);
  if (fd_ < 0) return -1;
  struct sockaddr_in sa;
  memset(&sa, 0, sizeof(sa));
  sa.sin_family = AF_INET;
  sa.sin_addr = ep.ip;
  sa.sin_port = htons((uint16_t)ep.port);
  if (::connect(fd_, (struct sockaddr*)&sa, sizeof(sa)) != 0) {
    Close();
    return -1;
  }
  int one = 1;
  setsockopt(fd_, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
  // C0 + C1
  std::string c0c1(1, '\x03');
  std::string c1(kHsSize, '\0');
  for (size_t i = 8; i < kHsSize; ++i) c1[i] = (char)fast_rand();
  c0c1 += c1;
  if (send_bytes(c0c1) != 0) {
    Close();
    return -1;
  }
  // S0 + S1 + S2
  std::string srv;
  const int64_t deadline = monotonic_time_us() + (int64_t)timeout_ms * 1000;
  while (srv.size() < 1 + 2 * kHsSize) {
    struct pollfd pfd{fd_, POLLIN, 0};
    int64_t left = (deadline - monotonic_time_us()) / 1000;
    if (left <= 0 || ::poll(&pfd, 1, (int)left) <= 0) {
      Close();
      return -1;
    }
    char buf[4096];
    ssize_t r = ::recv(fd_, buf, sizeof(buf), 0);
    if (r <= 0) {
      Close();
      return -1;
    }
    srv.append(buf, (size_t)r);
  }
  if (srv[0] != 0x03) {
    Close();
    return -1;
  }
  // C2 = echo of S1
  if (send_bytes(srv.substr(1, kHsSize)) != 0) {
    Close();
    return -1;
  }
  // leftover bytes past the handshake belong to the chunk stream
  if (srv.size() > 1 + 2 * kHsSize) {
    reader_.feed(srv.data() + 1 + 2 * kHsSize, srv.size() - 1 - 2 * kHsSize);
  }
  // connect(app)
  std::string payload;
  amf::Encode(amf::Value::Str("connect"), &payload);
  amf::Encode(amf::Value::Number(next_txn_++), &payload);
  amf::Value obj = amf::Value::Object();
  obj.obj["app"] = amf::Value::Str(app);
  obj.obj["flashVer"] = amf::Value::Str("BAM/1.0");
  obj.obj["tcUrl"] = amf::Value::Str("rtmp://" + host + ":" + std::to_string(port) + "/" + app);
  amf::Encode(obj, &payload);
  if (send_command(payload) != 0) {
    Close();
    return -1;
  }
  if (wait_command_reply("_result", timeout_ms) != 0) {
    Close();
    return -1;
  }
  return 0;
}

int RtmpClient::Publish(const std::string& stream_name) {
  std::string payload;
  amf::Encode(amf::Value::Str("createStream"), &payload);
  amf::Encode(amf::Value::Number(next_txn_++), &payload);
  amf::Encode(amf::Value::Null(), &payload);
  if (send_command(payload) != 0 || wait_command_reply("_result", timeout_ms_) != 0)
    return -1;
  payload.clear();
  amf::Encode(amf::Value::Str("publish"), &payload);
  amf::Encode(amf::Value::Number(next_txn_++), &payload);
  amf::Encode(amf::Value::Null(), &payload);
  amf::Encode(amf::Value::Str(stream_name), &payload);
  amf::Encode(amf::Value::Str("live"), &payload);
  if (send_command(payload, 1) != 0 || wait_command_reply("onStatus", timeout_ms_) != 0)
    return -1;
  return 0;
}

int RtmpClient::Play(const std::string& stream_name) {
  std::string payload;
  amf::Encode(amf::Value::Str("createStream"), &payload);
  amf::Encode(amf::Value::Number(next_txn_++), &payload);
  amf::Encode(amf::Value::Null(), &payload);
  if (send_command(payload) != 0 || wait_command_reply("_result", timeout_ms_) != 0)
    return -1;
  payload.clear();
  amf::Encode(amf::Value::Str("play"), &payload);
  amf::Encode(amf::Value::Number(next_txn_++), &payload);
  amf::Encode(amf::Value::Null(), &payload);
  amf::Encode(amf::Value::Str(stream_name), &payload);
  if (send_command(payload, 1) != 0 || wait_command_reply("onStatus", timeout_ms_) != 0)
    return -1;
  return 0;
}

int RtmpClient::PushFrame(uint8_t type, uint32_t timestamp, const std::string& payload) {
  if (fd_ < 0) return -1;
  std::string bytes;
  rtmp::write_message(&bytes, type == rtmp::kMsgAudio ? 4 : 5, type, 1, timestamp, payload,
                      out_chunk_);
  return send_bytes(bytes);
}

int RtmpClient::PollFrame(rtmp::Message* out, int timeout_ms) {
  const int64_t deadline = monotonic_time_us() + (int64_t)timeout_ms * 1000;
  for (;;) {
    rtmp::Message m;
    int rc = reader_.next(&m);
    if (rc < 0) return -1;
    if (rc == 1) {
      if (m.type == rtmp::kMsgAudio || m.type == rtmp::kMsgVideo ||
          m.type == rtmp::kMsgDataAmf0) {
        *out = std::move(m);
        return 0;
      }
      continue;  // control / command noise
    }
    int64_t left_us = deadline - monotonic_time_us();
    if (left_us <= 0) return -1;
    if (read_more((int)(left_us / 1000) + 1) != 0) return -1;
  }
}

}  // namespace bam
