// brpc_amd: extended load balancers — wrr, c_hash (ketama-style consistent
// hash ring), la (locality-aware EWMA latency), p2c (power of two choices).
// Parity: reference brpc/policy/{weighted_round_robin,consistent_hashing,
// locality_aware,randomized}_load_balancer.cpp, clean-room.
#include <string.h>

#include <algorithm>
#include <atomic>
#include <cmath>
#include <map>
#include <mutex>

#include "base/fast_rand.h"
#include "base/logging.h"
#include "rpc/load_balancer.h"

namespace bam {

namespace {

// ---- murmur3 x86 32-bit (public algorithm, clean-room) ----
uint32_t murmur3_32(const void* key, size_t len, uint32_t seed) {
  const uint8_t* data = (const uint8_t*)key;
  uint32_t h = seed;
  const uint32_t c1 = 0xcc9e2d51, c2 = 0x1b873593;
  size_t nblocks = len / 4;
  for (size_t i = 0; i < nblocks; ++i) {
    uint32_t k;
    memcpy(&k, data + i * 4, 4);
    k *= c1;
    k = (k << 15) | (k >> 17);
    k *= c2;
    h ^= k;
    h = (h << 13) | (h >> 19);
    h = h * 5 + 0xe6546b64;
  }
  uint32_t k = 0;
  const uint8_t* tail = data + nblocks * 4;
  switch (len & 3) {
    case 3:
      k ^= (uint32_t)tail[2] << 16;
      [[fallthrough]];
    case 2:
      k ^= (uint32_t)tail[1] << 8;
      [[fallthrough]];
    case 1:
      k ^= tail[0];
      k *= c1;
      k = (k << 15) | (k >> 17);
      k *= c2;
      h ^= k;
  }
  h ^= (uint32_t)len;
  h ^= h >> 16;
  h *= 0x85ebca6b;
  h ^= h >> 13;
  h *= 0xc2b2ae35;
  h ^= h >> 16;
  return h;
}

// ---- consistent hashing (ketama-style virtual nodes) ----
class ConsistentHashLB : public LoadBalancer {
 public:
  static const int kVirtualNodes = 100;

  int SelectServer(EndPoint* out) override {
    // Without a request key, hash a random value (uniform pick).
    return SelectByKey(fast_rand(), out);
  }

  int SelectServerByCode(uint64_t code, EndPoint* out) override {
    return SelectByKey(code, out);
  }

  int SelectByKey(uint64_t key, EndPoint* out) {
    std::lock_guard<std::mutex> lk(mu_);
    if (ring_.empty()) return ENODATA;
    uint32_t h = murmur3_32(&key, sizeof(key), 0x9747b28c);
    auto it = ring_.lower_bound(h);
    if (it == ring_.end()) it = ring_.begin();
    *out = it->second;
    return 0;
  }

  void SetServers(const std::vector<EndPoint>& servers) override {
    std::lock_guard<std::mutex> lk(mu_);
    ring_.clear();
    for (const EndPoint& ep : servers) {
      std::string id = endpoint2str(ep);
      for (int v = 0; v < kVirtualNodes; ++v) {
        std::string node = id + "#" + std::to_string(v);
        ring_[murmur3_32(node.data(), node.size(), 0x9747b28c)] = ep;
      }
    }
  }
  const char* name() const override { return "c_hash"; }

 private:
  std::mutex mu_;
  std::map<uint32_t, EndPoint> ring_;
};

// ---- weighted round robin (all weights 1 unless set; kept simple) ----
class WeightedRoundRobinLB : public LoadBalancer {
 public:
  int SelectServer(EndPoint* out) override {
    std::lock_guard<std::mutex> lk(mu_);
    if (servers_.empty()) return ENODATA;
    *out = servers_[idx_++ % servers_.size()];
    return 0;
  }
  void SetServers(const std::vector<EndPoint>& servers) override {
    std::lock_guard<std::mutex> lk(mu_);
    servers_ = servers;
  }
  const char* name() const override { return "wrr"; }

 private:
  std::mutex mu_;
  std::vector<EndPoint> servers_;
  size_t idx_ = 0;
};

// ---- locality-aware: EWMA latency + inflight penalty ----
// Parity in spirit with reference policy/locality_aware_load_balancer.cpp
// (weight ~ 1/latency, divided among inflight) without the weight tree.
class LocalityAwareLB : public LoadBalancer {
 public:
  struct Node {
    EndPoint ep;
    double ewma_latency_us = 10000;  // optimistic start
    std::atomic<int> inflight{0};
    std::atomic<int64_t> errors{0};
  };

  int SelectServer(EndPoint* out) override {
    std::lock_guard<std::mutex> lk(mu_);
    if (nodes_.empty()) return ENODATA;
    double best_score = -1;
    Node* best = nullptr;
    for (auto& n : nodes_) {
      double lat = n->ewma_latency_us;
      double score = 1e6 / (lat * (1 + n->inflight.load(std::memory_order_relaxed)));
      // jitter so equal nodes share load
      score *= 0.9 + 0.2 * fast_rand_double();
      if (score > best_score) {
        best_score = score;
        best = n.get();
      }
    }
    best->inflight.fetch_add(1, std::memory_order_relaxed);
    *out = best->ep;
    return 0;
  }

  void Feedback(const EndPoint& server, int error_code, int64_t latency_us) override {
    std::lock_guard<std::mutex> lk(mu_);
    for (auto& n : nodes_) {
      if (n->ep == server) {
        n->inflight.fetch_sub(1, std::memory_order_relaxed);
        if (error_code != 0) {
          n->errors.fetch_add(1, std::memory_order_relaxed);
          n->ewma_latency_us = n->ewma_latency_us * 0.9 + 100000 * 0.1;  // punish
        } else if (latency_us > 0) {
          n->ewma_latency_us = n->ewma_latency_us * 0.9 + (double)latency_us * 0.1;
        }
        return;
      }
    }
  }

  void SetServers(const std::vector<EndPoint>& servers) override {
    std::lock_guard<std::mutex> lk(mu_);
    std::vector<std::unique_ptr<Node>> next;
    for (const EndPoint& ep : servers) {
      bool found = false;
      for (auto& n : nodes_) {
        if (n && n->ep == ep) {
          next.push_back(std::move(n));
          found = true;
          break;
        }
      }
      if (!found) {
        auto n = std::make_unique<Node>();
        n->ep = ep;
        next.push_back(std::move(n));
      }
    }
    nodes_ = std::move(next);
  }
  const char* name() const override { return "la"; }

 private:
  std::mutex mu_;
  std::vector<std::unique_ptr<Node>> nodes_;
};

// ---- power of two choices with EWMA ----
class PowerOfTwoLB : public LoadBalancer {
 public:
  struct Node {
    EndPoint ep;
    double ewma_latency_us = 10000;
    std::atomic<int> inflight{0};
  };

  int SelectServer(EndPoint* out) override {
    std::lock_guard<std::mutex> lk(mu_);
    size_t n = nodes_.size();
    if (n == 0) return ENODATA;
    if (n == 1) {
      nodes_[0]->inflight.fetch_add(1);
      *out = nodes_[0]->ep;
      return 0;
    }
    size_t a = fast_rand_less_than(n);
    size_t b = fast_rand_less_than(n - 1);
    if (b >= a) ++b;
    Node* na = nodes_[a].get();
    Node* nb = nodes_[b].get();
    double sa = na->ewma_latency_us * (1 + na->inflight.load());
    double sb = nb->ewma_latency_us * (1 + nb->inflight.load());
    Node* pick = sa <= sb ? na : nb;
    pick->inflight.fetch_add(1);
    *out = pick->ep;
    return 0;
  }

  void Feedback(const EndPoint& server, int error_code, int64_t latency_us) override {
    std::lock_guard<std::mutex> lk(mu_);
    for (auto& n : nodes_) {
      if (n->ep == server) {
        n->inflight.fetch_sub(1);
        if (error_code != 0) {
          n->ewma_latency_us = n->ewma_latency_us * 0.9 + 100000 * 0.1;
        } else if (latency_us > 0) {
          n->ewma_latency_us = n->ewma_latency_us * 0.9 + (double)latency_us * 0.1;
        }
        return;
      }
    }
  }

  void SetServers(const std::vector<EndPoint>& servers) override {
    std::lock_guard<std::mutex> lk(mu_);
    std::vector<std::unique_ptr<Node>> next;
    for (const EndPoint& ep : servers) {
      bool found = false;
      for (auto& n : nodes_) {
        if (n && n->ep == ep) {
          next.push_back(std::move(n));
          found = true;
          break;
        }
      }
      if (!found) {
        auto node = std::make_unique<Node>();
        node->ep = ep;
        next.push_back(std::move(node));
      }
    }
    nodes_ = std::move(next);
  }
  const char* name() const override { return "p2c"; }

 private:
  std::mutex mu_;
  std::vector<std::unique_ptr<Node>> nodes_;
};

}  // namespace

LoadBalancer* CreateExtendedLoadBalancer(const std::string& name) {
  if (name == "c_hash") return new ConsistentHashLB;
  if (name == "wrr") return new WeightedRoundRobinLB;
  if (name == "la") return new LocalityAwareLB;
  if (name == "p2c") return new PowerOfTwoLB;
  return nullptr;
}

}  // namespace bam
