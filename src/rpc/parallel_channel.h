// brpc_amd: combo channels.
// ParallelChannel — fan one call out to N sub-channels, merge responses
// (parity: brpc/parallel_channel.h: CallMapper / ResponseMerger /
// fail_limit). On an 8-GPU node the sub-channels are GPU-resident echo
// servers and the fan-out is BASELINE config 4; the GPU-payload variant
// rides RCCL broadcast/all-gather via the torch-side harness (bench).
// SelectiveChannel — picks ONE healthy sub-channel per call with failover.
// PartitionChannel — fan-out across partitions of a sharded service
// resolved from one naming URL with "i/N " tags.
#pragma once

#include <memory>
#include <vector>

#include "rpc/channel.h"

namespace bam {

// Maps the main call onto sub-channel i. Returning false skips the sub.
typedef std::function<bool(int sub_index, const IOBuf& request, IOBuf* sub_request)>
    CallMapper;
// Merges a successful sub-response into the main response (called in
// sub-channel order). Nonzero return fails the whole call.
typedef std::function<int(IOBuf* main_response, const IOBuf& sub_response)> ResponseMerger;

struct ParallelChannelOptions {
  // <0: all subs must succeed; otherwise the call fails once more than
  // fail_limit subs failed (and succeeds when nsubs - fail_limit succeeded).
  int fail_limit = -1;
  int32_t timeout_ms = 500;
};

class ParallelChannel : public ChannelBase {
 public:
  ParallelChannel() {}
  ~ParallelChannel() override;

  int Init(const ParallelChannelOptions* options);
  // Takes ownership when `owned`.
  int AddChannel(ChannelBase* sub, bool owned = false, CallMapper mapper = nullptr,
                 ResponseMerger merger = nullptr);
  size_t channel_count() const { return subs_.size(); }

  void CallMethod(const std::string& full_method, Controller* cntl, const IOBuf* request,
                  IOBuf* response, Closure* done) override;

 private:
  struct Sub {
    ChannelBase* channel;
    bool owned;
    CallMapper mapper;
    ResponseMerger merger;
  };
  ParallelChannelOptions options_;
  std::vector<Sub> subs_;
};

struct SelectiveChannelOptions {
  int32_t timeout_ms = 500;
  int max_retry = 3;  // failover attempts across sub-channels
};

class SelectiveChannel : public ChannelBase {
 public:
  SelectiveChannel() {}
  ~SelectiveChannel() override;

  int Init(const char* lb_name /*"rr"|"random"|...*/, const SelectiveChannelOptions* opt);
  // Returns a handle in *handle (for RemoveAndDestroyChannel).
  int AddChannel(ChannelBase* sub, size_t* handle = nullptr);

  void CallMethod(const std::string& full_method, Controller* cntl, const IOBuf* request,
                  IOBuf* response, Closure* done) override;

 private:
  struct Sub {
    ChannelBase* channel;
    std::atomic<int> consecutive_failures{0};
    std::atomic<int64_t> isolated_until_us{0};
  };
  SelectiveChannelOptions options_;
  std::vector<std::unique_ptr<Sub>> subs_;
  std::atomic<uint32_t> rr_{0};
};

struct PartitionChannelOptions {
  ParallelChannelOptions parallel;
  std::string lb_name = "rr";
  ChannelOptions sub_options;
};

// Naming entries may carry "i/N " partition tags, e.g.
// "list://0/2 127.0.0.1:8000,1/2 127.0.0.1:8001". Untagged entries join
// every partition.
class PartitionChannel : public ChannelBase {
 public:
  int Init(int num_partitions, const char* naming_url, const PartitionChannelOptions* opt);
  size_t partition_count() const { return parallel_.channel_count(); }

  void CallMethod(const std::string& full_method, Controller* cntl, const IOBuf* request,
                  IOBuf* response, Closure* done) override;

 private:
  ParallelChannel parallel_;
};

// Parses "i/N host:port" partition tags from a naming URL.
int ResolvePartitionedNaming(const std::string& url, int num_partitions,
                             std::vector<std::vector<EndPoint>>* partitions);

// Discovers servers partitioned under DIFFERENT schemes from one naming
// URL ("0/2 a,1/2 b,0/3 c,1/3 d,2/3 e" = a 2-partition group and a
// 3-partition group), builds a PartitionChannel per scheme and splits
// traffic between schemes by capacity (= number of servers in the
// scheme). Purpose (parity: reference brpc/partition_channel.h
// DynamicPartitionChannel): migrate a sharded service from M to N
// partitions without touching client code. Delta: partition groups are
// resolved at Init from the naming URL (static list naming), not
// re-discovered live.
class DynamicPartitionChannel : public ChannelBase {
 public:
  ~DynamicPartitionChannel();
  int Init(const char* naming_url, const PartitionChannelOptions* opt);
  size_t scheme_count() const { return schemes_.size(); }
  int scheme_capacity(size_t i) const { return schemes_[i].capacity; }

  void CallMethod(const std::string& full_method, Controller* cntl, const IOBuf* request,
                  IOBuf* response, Closure* done) override;

 private:
  struct Scheme {
    int num_partitions = 0;
    int capacity = 0;  // number of servers carrying this scheme
    PartitionChannel* chan = nullptr;
  };
  std::vector<Scheme> schemes_;
  int total_capacity_ = 0;
};

}  // namespace bam
