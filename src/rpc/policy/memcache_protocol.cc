// brpc_amd: memcached binary protocol — client side.
// Parity: reference brpc/memcache.cpp + policy/memcache_binary_protocol.cpp.
// 24-byte header: magic(0x80 req / 0x81 resp), opcode, key_len u16,
// extras_len u8, data_type u8, vbucket/status u16, total_body u32,
// opaque u32, cas u64 (big-endian). Responses correlate FIFO on the
// connection (same pipelined-queue seam as redis).
#include <mutex>

#include "base/logging.h"
#include "fiber/session.h"
#include "rpc/channel.h"
#include "rpc/controller.h"
#include "rpc/memcache.h"
#include "rpc/wire.h"

namespace bam {

void EndRPC(Controller* cntl, SessionId locked_id);  // channel.cc

namespace {

const size_t kHeaderLen = 24;

void put_u16_be(char* p, uint16_t v) {
  p[0] = (char)(v >> 8);
  p[1] = (char)v;
}
uint16_t get_u16_be(const char* p) {
  return (uint16_t)(((uint8_t)p[0] << 8) | (uint8_t)p[1]);
}
void put_u64_be(char* p, uint64_t v) {
  for (int i = 7; i >= 0; --i) {
    p[i] = (char)v;
    v >>= 8;
  }
}
uint64_t get_u64_be(const char* p) {
  uint64_t v = 0;
  for (int i = 0; i < 8; ++i) v = (v << 8) | (uint8_t)p[i];
  return v;
}

}  // namespace

void PackMemcacheRequest(IOBuf* out, uint8_t opcode, const std::string& key,
                         const std::string& value, const std::string& extras, uint64_t cas,
                         uint16_t vbucket) {
  char h[kHeaderLen];
  memset(h, 0, sizeof(h));
  h[0] = (char)0x80;
  h[1] = (char)opcode;
  put_u16_be(h + 2, (uint16_t)key.size());
  h[4] = (char)extras.size();
  put_u16_be(h + 6, vbucket);  // request header: vbucket id (couchbase)
  uint32_t body = (uint32_t)(extras.size() + key.size() + value.size());
  wire::put_u32_be(h + 8, body);
  put_u64_be(h + 16, cas);
  out->append(h, kHeaderLen);
  out->append(extras);
  out->append(key);
  out->append(value);
}

namespace policy {

namespace {

struct MemcacheMessage : public InputMessageBase {
  MemcacheResponse resp;
};

ParseResult ParseMemcache(IOBuf* source, Socket* sock, bool /*eof*/) {
  char aux[kHeaderLen];
  if (source->size() < kHeaderLen) return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  const char* h = (const char*)source->fetch(aux, kHeaderLen);
  const bool server_side = sock->user() != nullptr;
  if (server_side || (uint8_t)h[0] != 0x81)
    return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  uint32_t body = wire::get_u32_be(h + 8);
  uint16_t key_len = get_u16_be(h + 2);
  uint8_t extras_len = (uint8_t)h[4];
  if (body > (64u << 20) || key_len + extras_len > body)
    return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
  if (source->size() < kHeaderLen + body)
    return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  MemcacheMessage* msg = new MemcacheMessage;
  msg->resp.opcode = (uint8_t)h[1];
  msg->resp.status = get_u16_be(h + 6);
  msg->resp.cas = get_u64_be(h + 16);
  source->pop_front(kHeaderLen);
  std::string extras, key, value;
  source->cutn(&extras, extras_len);
  source->cutn(&key, key_len);
  source->cutn(&value, body - extras_len - key_len);
  msg->resp.extras = extras;
  msg->resp.key = key;
  msg->resp.value = value;
  return ParseResult::make_ok(msg);
}

void ProcessMemcacheResponse(InputMessageBase* msg_base) {
  MemcacheMessage* msg = (MemcacheMessage*)msg_base;
  SocketUniquePtr sock;
  uint64_t cid = 0;
  if (Socket::Address(msg->socket_id, &sock) == 0) cid = sock->pop_pipeline();
  if (cid == 0) {
    delete msg;
    return;
  }
  void* data = nullptr;
  if (session_lock(cid, &data) != 0) {
    delete msg;
    return;
  }
  Controller* cntl = (Controller*)data;
  if (sock) sock->remove_pending_session(cid);
  // Deliver the raw response fields re-encoded (status u16 | cas u64 |
  // extras | value) for the client wrapper to unpack.
  if (cntl->call.response != nullptr) {
    cntl->call.response->clear();
    char h[14];
    put_u16_be(h, msg->resp.status);
    put_u64_be(h + 2, msg->resp.cas);
    wire::put_u32_be(h + 10, (uint32_t)msg->resp.extras.size());
    cntl->call.response->append(h, sizeof(h));
    cntl->call.response->append(msg->resp.extras);
    cntl->call.response->append(msg->resp.value);
  }
  delete msg;
  EndRPC(cntl, cid);
}

void PackMemcacheFromController(IOBuf* out, Controller* cntl, uint64_t /*cid*/) {
  out->append(cntl->call.request_buf);  // pre-packed by MemcacheClient
}

}  // namespace

void RegisterMemcacheProtocol() {
  static std::once_flag flag;
  std::call_once(flag, [] {
    Protocol p;
    p.parse = ParseMemcache;
    p.process_request = nullptr;
    p.process_response = ProcessMemcacheResponse;
    p.pack_request = PackMemcacheFromController;
    p.client_pipelined = true;
    p.support_server = false;
    p.support_client = true;
    p.name = "memcache";
    RegisterProtocol(p);
  });
}

}  // namespace policy

// ---------------- client wrapper ----------------

MemcacheClient::MemcacheClient(const std::string& addr, int timeout_ms) {
  ChannelOptions opts;
  opts.timeout_ms = timeout_ms;
  opts.protocol = "memcache";
  policy::RegisterMemcacheProtocol();
  init_ok_ = channel_.Init(addr.c_str(), &opts) == 0;
}

int MemcacheClient::SaslAuthPlain(const std::string& user, const std::string& password) {
  std::string token;
  token.push_back('\0');
  token += user;
  token.push_back('\0');
  token += password;
  MemcacheResponse resp;
  int rc = RawCall(0x21, "PLAIN", token, "", &resp);
  if (rc != 0) return rc;
  return resp.status == 0 ? 0 : 10000 + resp.status;
}

int MemcacheClient::RawCall(uint8_t opcode, const std::string& key, const std::string& value,
                            const std::string& extras, MemcacheResponse* out,
                            uint16_t vbucket) {
  if (!init_ok_) return -1;
  Controller cntl;
  IOBuf request, response;
  PackMemcacheRequest(&request, opcode, key, value, extras, 0, vbucket);
  channel_.CallMethod("memcache.op", &cntl, &request, &response, nullptr);
  if (cntl.Failed()) return cntl.ErrorCode();
  std::string raw = response.to_string();
  if (raw.size() < 14) return -2;
  out->opcode = opcode;
  out->status = get_u16_be(raw.data());
  out->cas = get_u64_be(raw.data() + 2);
  uint32_t extras_len = wire::get_u32_be(raw.data() + 10);
  out->extras = raw.substr(14, extras_len);
  out->value = raw.substr(14 + extras_len);
  return 0;
}

int MemcacheClient::Set(const std::string& key, const std::string& value, uint32_t flags,
                        uint32_t exptime) {
  char extras[8];
  wire::put_u32_be(extras, flags);
  wire::put_u32_be(extras + 4, exptime);
  MemcacheResponse resp;
  int rc = RawCall(0x01, key, value, std::string(extras, 8), &resp);
  if (rc != 0) return rc;
  return resp.status == 0 ? 0 : (int)resp.status + 10000;
}

int MemcacheClient::Get(const std::string& key, std::string* value) {
  MemcacheResponse resp;
  int rc = RawCall(0x00, key, "", "", &resp);
  if (rc != 0) return rc;
  if (resp.status != 0) return (int)resp.status + 10000;  // 1 = key not found
  *value = resp.value;
  return 0;
}

int MemcacheClient::Delete(const std::string& key) {
  MemcacheResponse resp;
  int rc = RawCall(0x04, key, "", "", &resp);
  if (rc != 0) return rc;
  return resp.status == 0 ? 0 : (int)resp.status + 10000;
}

int MemcacheClient::Version(std::string* version) {
  MemcacheResponse resp;
  int rc = RawCall(0x0b, "", "", "", &resp);
  if (rc != 0) return rc;
  *version = resp.value;
  return 0;
}

}  // namespace bam
