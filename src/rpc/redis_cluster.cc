#include "rpc/redis_cluster.h"

#include "base/logging.h"
#include "rpc/channel.h"
#include "rpc/controller.h"

namespace bam {

namespace {

// CRC16-CCITT (XMODEM: poly 0x1021, init 0) — the redis cluster key hash.
uint16_t crc16_ccitt(const char* data, size_t n) {
  uint16_t crc = 0;
  for (size_t i = 0; i < n; ++i) {
    crc ^= (uint16_t)((uint8_t)data[i]) << 8;
    for (int b = 0; b < 8; ++b)
      crc = (crc & 0x8000) ? (uint16_t)((crc << 1) ^ 0x1021) : (uint16_t)(crc << 1);
  }
  return crc;
}

}  // namespace

uint16_t RedisClusterClient::SlotOf(const std::string& key) {
  // {hash tag}: only the braces' content hashes (multi-key colocation).
  size_t open = key.find('{');
  if (open != std::string::npos) {
    size_t close = key.find('}', open + 1);
    if (close != std::string::npos && close > open + 1) {
      return crc16_ccitt(key.data() + open + 1, close - open - 1) % 16384;
    }
  }
  return crc16_ccitt(key.data(), key.size()) % 16384;
}

int RedisClusterClient::CallNode(const std::string& addr,
                                 const std::vector<std::string>& args, RedisReply* out,
                                 bool asking) {
  policy::RegisterRedisProtocol();
  ChannelOptions opts;
  opts.timeout_ms = timeout_ms_;
  opts.protocol = "redis";
  opts.max_retry = 0;
  Channel ch;
  if (ch.Init(addr.c_str(), &opts) != 0) {
    err_ = "channel init failed for " + addr;
    return -1;
  }
  std::string cmd;
  if (asking) EncodeRedisCommand({"ASKING"}, &cmd);
  std::string main_cmd;
  EncodeRedisCommand(args, &main_cmd);
  cmd += main_cmd;
  Controller cntl;
  IOBuf req, resp;
  req.append(cmd);
  ch.CallMethod("redis.command", &cntl, &req, &resp, nullptr);
  if (cntl.Failed()) {
    err_ = cntl.ErrorText();
    return cntl.ErrorCode();
  }
  std::string raw = resp.to_string();
  size_t off = 0;
  if (asking) {
    // skip the +OK of ASKING
    RedisReply ok;
    ssize_t c = ParseRedisValue(raw.data(), raw.size(), &ok);
    if (c <= 0) {
      err_ = "bad ASKING reply";
      return -1;
    }
    off = (size_t)c;
  }
  ssize_t c = ParseRedisValue(raw.data() + off, raw.size() - off, out);
  if (c <= 0) {
    err_ = "bad redis reply";
    return -1;
  }
  return 0;
}

int RedisClusterClient::RefreshSlots(const std::string& via) {
  RedisReply r;
  if (CallNode(via, {"CLUSTER", "SLOTS"}, &r, false) != 0) return -1;
  if (r.type != RedisReply::ARRAY) {
    err_ = "CLUSTER SLOTS: not an array";
    return -1;
  }
  std::vector<Range> ranges;
  for (const RedisReply& row : r.elements) {
    if (row.type != RedisReply::ARRAY || row.elements.size() < 3) continue;
    const RedisReply& master = row.elements[2];
    if (master.type != RedisReply::ARRAY || master.elements.size() < 2) continue;
    Range rg;
    rg.start = (int)row.elements[0].integer;
    rg.end = (int)row.elements[1].integer;
    rg.addr = master.elements[0].str + ":" + std::to_string(master.elements[1].integer);
    ranges.push_back(std::move(rg));
  }
  if (ranges.empty()) {
    err_ = "CLUSTER SLOTS: empty";
    return -1;
  }
  ranges_.swap(ranges);
  return 0;
}

int RedisClusterClient::Init(const std::string& seed, int timeout_ms) {
  seed_ = seed;
  timeout_ms_ = timeout_ms;
  return RefreshSlots(seed);
}

int RedisClusterClient::nslots_mapped() const {
  int n = 0;
  for (const auto& r : ranges_) n += r.end - r.start + 1;
  return n;
}

int RedisClusterClient::Command(const std::vector<std::string>& args, RedisReply* out) {
  if (args.empty()) return -1;
  std::string addr = seed_;
  if (args.size() >= 2 && !ranges_.empty()) {
    uint16_t slot = SlotOf(args[1]);
    for (const auto& r : ranges_) {
      if ((int)slot >= r.start && (int)slot <= r.end) {
        addr = r.addr;
        break;
      }
    }
  }
  bool asking = false;
  for (int attempt = 0; attempt < 3; ++attempt) {
    int rc = CallNode(addr, args, out, asking);
    asking = false;
    // Redirects surface either as an ERROR reply or as a failed call
    // whose error text is the server's "-MOVED/-ASK ..." line (the redis
    // client protocol conducts error replies through the controller).
    std::string e;
    if (rc == 0 && out->type == RedisReply::ERROR) e = out->str;
    else if (rc != 0) e = err_;
    if (!e.empty()) {
      bool moved = e.rfind("MOVED ", 0) == 0;
      bool ask = e.rfind("ASK ", 0) == 0;
      if (moved || ask) {
        size_t sp = e.find(' ', moved ? 6 : 4);
        if (sp != std::string::npos) {
          addr = e.substr(sp + 1);
          // trim trailing junk after host:port
          size_t end = addr.find_first_of(" \r\n");
          if (end != std::string::npos) addr = addr.substr(0, end);
          if (moved) RefreshSlots(addr);  // topology changed: remap
          asking = ask;
          continue;
        }
      }
    }
    if (rc != 0) return rc;
    return 0;
  }
  err_ = "redirect loop";
  return -1;
}

}  // namespace bam
