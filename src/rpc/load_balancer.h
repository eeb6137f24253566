// brpc_amd: LoadBalancer interface + naming integration.
// Parity: reference brpc/load_balancer.h (SelectServer / Feedback /
// AddServer over DoublyBufferedData) and details/load_balancer_with_naming.
#pragma once

#include <functional>
#include <memory>
#include <string>
#include <vector>

#include "base/doubly_buffered.h"
#include "base/endpoint.h"

namespace bam {

class LoadBalancer {
 public:
  virtual ~LoadBalancer() {}
  // Returns 0 and a server on success, nonzero if none available.
  virtual int SelectServer(EndPoint* out) = 0;
  // Keyed selection (consistent hashing; ≙ reference request_code
  // routing). Default: ignore the code.
  virtual int SelectServerByCode(uint64_t /*code*/, EndPoint* out) {
    return SelectServer(out);
  }
  // Latency/error feedback after a call (la / p2c use it).
  virtual void Feedback(const EndPoint& server, int error_code, int64_t latency_us) {}
  virtual void SetServers(const std::vector<EndPoint>& servers) = 0;
  virtual const char* name() const = 0;

  // Factory: "rr", "random", "c_hash", "la", "p2c". nullptr if unknown.
  static LoadBalancer* CreateByName(const std::string& name);
};

// Periodically refreshes the server list from a naming service URL
// ("list://h:p,h:p", "file://path") into the wrapped LB.
// Parity: reference details/naming_service_thread + load_balancer_with_naming.
class LoadBalancerWithNaming : public LoadBalancer {
 public:
  static std::shared_ptr<LoadBalancerWithNaming> Create(
      const std::string& naming_url, const std::string& lb_name,
      std::function<bool(const EndPoint&)> ns_filter = nullptr);
  ~LoadBalancerWithNaming() override;

  int SelectServer(EndPoint* out) override { return lb_->SelectServer(out); }
  int SelectServerByCode(uint64_t code, EndPoint* out) override {
    return lb_->SelectServerByCode(code, out);
  }
  void Feedback(const EndPoint& server, int error_code, int64_t latency_us) override {
    lb_->Feedback(server, error_code, latency_us);
  }
  void SetServers(const std::vector<EndPoint>& servers) override { lb_->SetServers(servers); }
  const char* name() const override { return lb_->name(); }

  // Re-resolves the naming url now (tests; the refresher fiber also calls).
  int Refresh();
  const std::string& url() const { return url_; }

 private:
  LoadBalancerWithNaming() {}
  std::string url_;
  std::unique_ptr<LoadBalancer> lb_;
  std::function<bool(const EndPoint&)> ns_filter_;
  bool stop_refresher_ = false;
};

// Resolves a naming URL into endpoints. Supports list:// and file://.
// Returns 0 on success.
int ResolveNamingUrl(const std::string& url, std::vector<EndPoint>* out);

// Implemented in policy/extended_lb.cc (c_hash / la / p2c / wrr); a weak
// fallback returning nullptr exists until that TU is linked.
LoadBalancer* CreateExtendedLoadBalancer(const std::string& name);

}  // namespace bam
