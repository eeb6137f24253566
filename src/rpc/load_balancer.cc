#include "rpc/load_balancer.h"

#include "rpc/channel.h"
#include "rpc/controller.h"

#include <netdb.h>
#include <netinet/in.h>
#include <stdio.h>
#include <string.h>

#include <algorithm>
#include <atomic>
#include <mutex>

#include "base/fast_rand.h"
#include "base/json.h"
#include "base/logging.h"

namespace bam {

namespace {

struct ServerList {
  std::vector<EndPoint> servers;
};

// ---- round robin ----
class RoundRobinLB : public LoadBalancer {
 public:
  int SelectServer(EndPoint* out) override {
    DoublyBufferedData<ServerList>::ScopedPtr ptr;
    if (data_.Read(&ptr) != 0 || ptr->servers.empty()) return ENODATA;
    uint32_t i = index_.fetch_add(1, std::memory_order_relaxed);
    *out = ptr->servers[i % ptr->servers.size()];
    return 0;
  }
  void SetServers(const std::vector<EndPoint>& servers) override {
    data_.Modify([&](ServerList& sl) {
      sl.servers = servers;
      return true;
    });
  }
  const char* name() const override { return "rr"; }

 private:
  DoublyBufferedData<ServerList> data_;
  std::atomic<uint32_t> index_{0};
};

// ---- random ----
class RandomLB : public LoadBalancer {
 public:
  int SelectServer(EndPoint* out) override {
    DoublyBufferedData<ServerList>::ScopedPtr ptr;
    if (data_.Read(&ptr) != 0 || ptr->servers.empty()) return ENODATA;
    *out = ptr->servers[fast_rand_less_than(ptr->servers.size())];
    return 0;
  }
  void SetServers(const std::vector<EndPoint>& servers) override {
    data_.Modify([&](ServerList& sl) {
      sl.servers = servers;
      return true;
    });
  }
  const char* name() const override { return "random"; }

 private:
  DoublyBufferedData<ServerList> data_;
};

}  // namespace

LoadBalancer* LoadBalancer::CreateByName(const std::string& name) {
  if (name == "rr" || name.empty()) return new RoundRobinLB;
  if (name == "random") return new RandomLB;
  LoadBalancer* extended = CreateExtendedLoadBalancer(name);
  if (extended != nullptr) return extended;
  LOG(ERROR) << "unknown load balancer: " << name;
  return nullptr;
}

// ---------------- naming resolution ----------------

int ResolveNamingUrl(const std::string& url, std::vector<EndPoint>* out) {
  out->clear();
  auto parse_csv = [&](const std::string& csv) {
    size_t pos = 0;
    while (pos < csv.size()) {
      size_t comma = csv.find_first_of(",\n ", pos);
      std::string item = csv.substr(pos, comma == std::string::npos ? comma : comma - pos);
      if (!item.empty() && item[0] != '#') {
        EndPoint ep;
        if (str2endpoint(item.c_str(), &ep) == 0) out->push_back(ep);
      }
      if (comma == std::string::npos) break;
      pos = comma + 1;
    }
  };
  if (url.rfind("list://", 0) == 0) {
    parse_csv(url.substr(7));
    return 0;
  }
  if (url.rfind("file://", 0) == 0) {
    FILE* f = fopen(url.substr(7).c_str(), "r");
    if (f == nullptr) return -1;
    std::string content;
    char buf[4096];
    size_t n;
    while ((n = fread(buf, 1, sizeof(buf), f)) > 0) content.append(buf, n);
    fclose(f);
    parse_csv(content);
    return 0;
  }
  if (url.rfind("dns://", 0) == 0) {
    // dns://host:port — ALL A records of host, each at :port (parity:
    // reference policy/domain_naming_service.cpp). Re-resolved by the
    // refresher fiber like every naming scheme here.
    std::string rest = url.substr(6);
    size_t colon = rest.find(':');
    if (colon == std::string::npos) return -1;
    std::string host = rest.substr(0, colon);
    int port = atoi(rest.c_str() + colon + 1);
    struct addrinfo hints, *res = nullptr;
    memset(&hints, 0, sizeof(hints));
    hints.ai_family = AF_INET;
    hints.ai_socktype = SOCK_STREAM;
    if (getaddrinfo(host.c_str(), nullptr, &hints, &res) != 0) return -1;
    for (struct addrinfo* p = res; p != nullptr; p = p->ai_next) {
      EndPoint ep;
      ep.ip = ((struct sockaddr_in*)p->ai_addr)->sin_addr;
      ep.port = port;
      out->push_back(ep);
    }
    freeaddrinfo(res);
    return out->empty() ? -1 : 0;
  }
  if (url.rfind("consul://", 0) == 0) {
    // consul://host:port/service — GET /v1/health/service/<svc>?passing=1
    // over our own HTTP client; entries come from Service.Address/Port
    // (parity: reference policy/consul_naming_service.cpp).
    std::string rest = url.substr(9);
    size_t slash = rest.find('/');
    if (slash == std::string::npos) return -1;
    std::string agent = rest.substr(0, slash);
    std::string service = rest.substr(slash + 1);
    Channel ch;
    ChannelOptions copt;
    copt.protocol = "http";
    copt.timeout_ms = 3000;
    copt.max_retry = 0;
    if (ch.Init(agent.c_str(), &copt) != 0) return -1;
    Controller cntl;
    IOBuf req, resp;
    ch.CallMethod("/v1/health/service/" + service + "?passing=1", &cntl, &req, &resp,
                  nullptr);
    if (cntl.Failed()) return -1;
    json::Value root;
    if (!json::Parse(resp.to_string(), &root) || root.type != json::Value::ARRAY) return -1;
    for (const json::Value& entry : *root.arr) {
      if (entry.type != json::Value::OBJECT) continue;
      auto sit = entry.obj->find("Service");
      if (sit == entry.obj->end() || sit->second.type != json::Value::OBJECT) continue;
      const json::Object& svc = *sit->second.obj;
      auto ait = svc.find("Address");
      auto pit = svc.find("Port");
      if (ait == svc.end() || pit == svc.end()) continue;
      EndPoint ep;
      std::string hp = ait->second.str + ":" + std::to_string((int)pit->second.num);
      if (str2endpoint(hp.c_str(), &ep) == 0) out->push_back(ep);
    }
    return 0;
  }
  if (url.rfind("nacos://", 0) == 0) {
    // nacos://host:port/serviceName[?namespaceId=..&groupName=..] —
    // GET /nacos/v1/ns/instance/list?serviceName=<svc>&healthyOnly=true
    // (parity: reference policy/nacos_naming_service.cpp); entries come
    // from hosts[].{ip,port}, healthy/enabled only.
    std::string rest = url.substr(8);
    size_t slash = rest.find('/');
    if (slash == std::string::npos) return -1;
    std::string agent = rest.substr(0, slash);
    std::string svc_and_params = rest.substr(slash + 1);
    std::string extra;
    size_t q = svc_and_params.find('?');
    if (q != std::string::npos) {
      extra = "&" + svc_and_params.substr(q + 1);
      svc_and_params = svc_and_params.substr(0, q);
    }
    Channel ch;
    ChannelOptions copt;
    copt.protocol = "http";
    copt.timeout_ms = 3000;
    copt.max_retry = 0;
    if (ch.Init(agent.c_str(), &copt) != 0) return -1;
    Controller cntl;
    IOBuf req, resp;
    ch.CallMethod("/nacos/v1/ns/instance/list?serviceName=" + svc_and_params +
                      "&healthyOnly=true" + extra,
                  &cntl, &req, &resp, nullptr);
    if (cntl.Failed()) return -1;
    json::Value root;
    if (!json::Parse(resp.to_string(), &root) || root.type != json::Value::OBJECT) return -1;
    auto hit = root.obj->find("hosts");
    if (hit == root.obj->end() || hit->second.type != json::Value::ARRAY) return -1;
    for (const json::Value& host : *hit->second.arr) {
      if (host.type != json::Value::OBJECT) continue;
      const json::Object& ho = *host.obj;
      auto en = ho.find("enabled");
      if (en != ho.end() && en->second.type == json::Value::BOOL && !en->second.b) continue;
      auto he = ho.find("healthy");
      if (he != ho.end() && he->second.type == json::Value::BOOL && !he->second.b) continue;
      auto iit = ho.find("ip");
      auto pit = ho.find("port");
      if (iit == ho.end() || pit == ho.end()) continue;
      EndPoint ep;
      std::string hp = iit->second.str + ":" + std::to_string((int)pit->second.num);
      if (str2endpoint(hp.c_str(), &ep) == 0) out->push_back(ep);
    }
    return 0;
  }
  if (url.rfind("dlist://", 0) == 0) {
    // dlist://name1:port1,name2:port2 — like list:// but entries may be
    // DNS names, re-resolved on every refresh (parity: reference
    // DomainListNamingService, policy/list_naming_service.cpp:106).
    std::string rest = url.substr(8);
    size_t pos = 0;
    while (pos < rest.size()) {
      size_t comma = rest.find(',', pos);
      std::string item = rest.substr(pos, comma == std::string::npos
                                              ? std::string::npos : comma - pos);
      pos = comma == std::string::npos ? rest.size() : comma + 1;
      if (item.empty()) continue;
      EndPoint ep;
      if (str2endpoint(item.c_str(), &ep) == 0) {
        out->push_back(ep);
        continue;
      }
      size_t colon = item.rfind(':');
      if (colon == std::string::npos) continue;
      std::string host = item.substr(0, colon);
      int port = atoi(item.c_str() + colon + 1);
      struct addrinfo hints, *res = nullptr;
      memset(&hints, 0, sizeof(hints));
      hints.ai_family = AF_INET;
      hints.ai_socktype = SOCK_STREAM;
      if (getaddrinfo(host.c_str(), nullptr, &hints, &res) != 0) continue;
      for (struct addrinfo* pp = res; pp != nullptr; pp = pp->ai_next) {
        EndPoint e2;
        e2.ip = ((struct sockaddr_in*)pp->ai_addr)->sin_addr;
        e2.port = port;
        out->push_back(e2);
      }
      freeaddrinfo(res);
    }
    return out->empty() ? -1 : 0;
  }
  if (url.rfind("discovery://", 0) == 0) {
    // discovery://host:port/appid[?env=..&status=..] — bilibili discovery:
    // GET /discovery/fetchs?appid=<appid>&env=<env>&status=<status>; JSON
    // data.<appid>.instances[].addrs carries "scheme://ip:port" strings
    // (parity: reference policy/discovery_naming_service.cpp:362-430).
    std::string rest = url.substr(12);
    size_t slash = rest.find('/');
    if (slash == std::string::npos) return -1;
    std::string agent = rest.substr(0, slash);
    std::string appid = rest.substr(slash + 1);
    std::string env = "prod", status = "1";
    size_t q = appid.find('?');
    if (q != std::string::npos) {
      std::string params = appid.substr(q + 1);
      appid = appid.substr(0, q);
      size_t p2 = 0;
      while (p2 < params.size()) {
        size_t amp = params.find('&', p2);
        std::string kv = params.substr(p2, amp == std::string::npos
                                               ? std::string::npos : amp - p2);
        p2 = amp == std::string::npos ? params.size() : amp + 1;
        size_t eq = kv.find('=');
        if (eq == std::string::npos) continue;
        std::string k = kv.substr(0, eq), v = kv.substr(eq + 1);
        if (k == "env") env = v;
        else if (k == "status") status = v;
      }
    }
    Channel ch;
    ChannelOptions copt;
    copt.protocol = "http";
    copt.timeout_ms = 3000;
    copt.max_retry = 0;
    if (ch.Init(agent.c_str(), &copt) != 0) return -1;
    Controller cntl;
    IOBuf req, resp;
    ch.CallMethod("/discovery/fetchs?appid=" + appid + "&env=" + env +
                      "&status=" + status,
                  &cntl, &req, &resp, nullptr);
    if (cntl.Failed()) return -1;
    json::Value root;
    if (!json::Parse(resp.to_string(), &root) || root.type != json::Value::OBJECT)
      return -1;
    auto dit = root.obj->find("data");
    if (dit == root.obj->end() || dit->second.type != json::Value::OBJECT) return -1;
    auto ait = dit->second.obj->find(appid);
    if (ait == dit->second.obj->end() || ait->second.type != json::Value::OBJECT)
      return -1;
    auto iit = ait->second.obj->find("instances");
    if (iit == ait->second.obj->end() || iit->second.type != json::Value::ARRAY)
      return -1;
    for (const json::Value& inst : *iit->second.arr) {
      if (inst.type != json::Value::OBJECT) continue;
      auto adit = inst.obj->find("addrs");
      if (adit == inst.obj->end() || adit->second.type != json::Value::ARRAY) continue;
      for (const json::Value& a : *adit->second.arr) {
        if (a.type != json::Value::STRING) continue;
        std::string addr = a.str;
        size_t sep = addr.find("://");
        if (sep != std::string::npos) addr = addr.substr(sep + 3);
        EndPoint ep;
        if (str2endpoint(addr.c_str(), &ep) == 0) out->push_back(ep);
      }
    }
    return 0;
  }
  if (url.rfind("remotefile://", 0) == 0) {
    // remotefile://host:port/path — fetch a server-list file over HTTP
    // (parity: reference policy/remote_file_naming_service.cpp); same
    // line format as file://.
    std::string rest = url.substr(13);
    size_t slash = rest.find('/');
    if (slash == std::string::npos) return -1;
    std::string agent = rest.substr(0, slash);
    std::string path = rest.substr(slash);
    Channel ch;
    ChannelOptions copt;
    copt.protocol = "http";
    copt.timeout_ms = 3000;
    copt.max_retry = 0;
    if (ch.Init(agent.c_str(), &copt) != 0) return -1;
    Controller cntl;
    IOBuf req, resp;
    ch.CallMethod(path, &cntl, &req, &resp, nullptr);
    if (cntl.Failed()) return -1;
    parse_csv(resp.to_string());
    return 0;
  }
  // bare "host:port" treated as a single-entry list
  EndPoint ep;
  if (str2endpoint(url.c_str(), &ep) == 0) {
    out->push_back(ep);
    return 0;
  }
  return -1;
}

// ---------------- LB with naming refresher ----------------

std::shared_ptr<LoadBalancerWithNaming> LoadBalancerWithNaming::Create(
    const std::string& naming_url, const std::string& lb_name,
    std::function<bool(const EndPoint&)> ns_filter) {
  LoadBalancer* lb = LoadBalancer::CreateByName(lb_name);
  if (lb == nullptr) return nullptr;
  auto lbn = std::shared_ptr<LoadBalancerWithNaming>(new LoadBalancerWithNaming);
  lbn->url_ = naming_url;
  lbn->lb_.reset(lb);
  lbn->ns_filter_ = std::move(ns_filter);
  if (lbn->Refresh() != 0) {
    LOG(WARNING) << "initial naming resolution failed for " << naming_url;
  }
  return lbn;
}

LoadBalancerWithNaming::~LoadBalancerWithNaming() {}

int LoadBalancerWithNaming::Refresh() {
  std::vector<EndPoint> servers;
  if (ResolveNamingUrl(url_, &servers) != 0) return -1;
  if (ns_filter_) {
    std::vector<EndPoint> kept;
    for (const EndPoint& ep : servers)
      if (ns_filter_(ep)) kept.push_back(ep);
    servers.swap(kept);
  }
  lb_->SetServers(servers);
  return 0;
}

}  // namespace bam
