#include "rpc/couchbase.h"

#include <zlib.h>

#include "base/json.h"
#include "base/logging.h"
#include "rpc/channel.h"
#include "rpc/controller.h"

namespace bam {

namespace {
constexpr uint16_t kNotMyVBucket = 0x0007;
}

uint16_t CouchbaseClient::VBucketOf(const std::string& key, size_t nvb) {
  if (nvb == 0) return 0;
  uint32_t crc = (uint32_t)crc32(0L, (const Bytef*)key.data(), (uInt)key.size());
  return (uint16_t)((crc >> 16) & (nvb - 1));
}

int CouchbaseClient::FetchConfig() {
  Channel ch;
  ChannelOptions copt;
  copt.protocol = "http";
  copt.timeout_ms = 3000;
  copt.max_retry = 0;
  if (ch.Init(config_addr_.c_str(), &copt) != 0) {
    err_ = "config channel init failed";
    return -1;
  }
  Controller cntl;
  IOBuf req, resp;
  ch.CallMethod("/pools/default/b/" + bucket_, &cntl, &req, &resp, nullptr);
  if (cntl.Failed()) {
    err_ = "config fetch: " + cntl.ErrorText();
    return -1;
  }
  json::Value root;
  if (!json::Parse(resp.to_string(), &root) || root.type != json::Value::OBJECT) {
    err_ = "config: malformed JSON";
    return -1;
  }
  auto vm = root.obj->find("vBucketServerMap");
  if (vm == root.obj->end() || vm->second.type != json::Value::OBJECT) {
    err_ = "config: no vBucketServerMap";
    return -1;
  }
  const json::Object& m = *vm->second.obj;
  auto sl = m.find("serverList");
  auto vb = m.find("vBucketMap");
  if (sl == m.end() || sl->second.type != json::Value::ARRAY || vb == m.end() ||
      vb->second.type != json::Value::ARRAY) {
    err_ = "config: incomplete vBucketServerMap";
    return -1;
  }
  std::vector<std::string> servers;
  for (const json::Value& s : *sl->second.arr)
    if (s.type == json::Value::STRING) servers.push_back(s.str);
  std::vector<int> primaries;
  for (const json::Value& row : *vb->second.arr) {
    if (row.type != json::Value::ARRAY || row.arr->empty()) {
      err_ = "config: bad vBucketMap row";
      return -1;
    }
    primaries.push_back((int)(*row.arr)[0].num);
  }
  if (servers.empty() || primaries.empty() ||
      (primaries.size() & (primaries.size() - 1)) != 0) {
    err_ = "config: vbucket count must be a power of two";
    return -1;
  }
  servers_.swap(servers);
  vbucket_primary_.swap(primaries);
  conns_.clear();  // topology changed: reconnect lazily
  return 0;
}

int CouchbaseClient::Init(const std::string& config_addr, const std::string& bucket,
                          const std::string& user, const std::string& password) {
  config_addr_ = config_addr;
  bucket_ = bucket;
  user_ = user;
  password_ = password;
  return FetchConfig();
}

MemcacheClient* CouchbaseClient::conn_for(int idx) {
  auto it = conns_.find(idx);
  if (it != conns_.end()) return it->second.get();
  if (idx < 0 || idx >= (int)servers_.size()) return nullptr;
  auto c = std::make_unique<MemcacheClient>(servers_[idx], 3000);
  if (!c->ok()) return nullptr;
  if (!user_.empty() && c->SaslAuthPlain(user_, password_) != 0) {
    err_ = "SASL auth failed on " + servers_[idx];
    return nullptr;
  }
  MemcacheClient* p = c.get();
  conns_[idx] = std::move(c);
  return p;
}

int CouchbaseClient::Op(uint8_t opcode, const std::string& key, const std::string& value,
                        const std::string& extras, MemcacheResponse* out) {
  for (int attempt = 0; attempt < 2; ++attempt) {
    if (vbucket_primary_.empty()) return -1;
    uint16_t vb = VBucketOf(key, vbucket_primary_.size());
    int srv = vbucket_primary_[vb];
    MemcacheClient* c = conn_for(srv);
    if (c == nullptr) return -1;
    int rc = c->RawCall(opcode, key, value, extras, out, vb);
    if (rc != 0) return rc;
    if (out->status == kNotMyVBucket) {
      // Stale map (rebalance): refetch the config and retry once
      // (parity: reference couchbase retry policy on NOT_MY_VBUCKET).
      if (FetchConfig() != 0) return -1;
      continue;
    }
    return out->status == 0 ? 0 : 10000 + out->status;
  }
  err_ = "NOT_MY_VBUCKET persisted after config refresh";
  return -1;
}

int CouchbaseClient::Set(const std::string& key, const std::string& value) {
  MemcacheResponse r;
  std::string extras(8, '\0');  // flags + exptime
  return Op(0x01, key, value, extras, &r);
}

int CouchbaseClient::Get(const std::string& key, std::string* value) {
  MemcacheResponse r;
  int rc = Op(0x00, key, "", "", &r);
  if (rc == 0 && value != nullptr) *value = r.value;
  return rc;
}

int CouchbaseClient::Delete(const std::string& key) {
  MemcacheResponse r;
  return Op(0x04, key, "", "", &r);
}

}  // namespace bam
