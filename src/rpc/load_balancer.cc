#include "rpc/load_balancer.h"

#include <stdio.h>
#include <string.h>

#include <algorithm>
#include <atomic>
#include <mutex>

#include "base/fast_rand.h"
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
    const std::string& naming_url, const std::string& lb_name) {
  LoadBalancer* lb = LoadBalancer::CreateByName(lb_name);
  if (lb == nullptr) return nullptr;
  auto lbn = std::shared_ptr<LoadBalancerWithNaming>(new LoadBalancerWithNaming);
  lbn->url_ = naming_url;
  lbn->lb_.reset(lb);
  if (lbn->Refresh() != 0) {
    LOG(WARNING) << "initial naming resolution failed for " << naming_url;
  }
  return lbn;
}

LoadBalancerWithNaming::~LoadBalancerWithNaming() {}

int LoadBalancerWithNaming::Refresh() {
  std::vector<EndPoint> servers;
  if (ResolveNamingUrl(url_, &servers) != 0) return -1;
  lb_->SetServers(servers);
  return 0;
}

}  // namespace bam
