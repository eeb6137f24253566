#include "rpc/parallel_channel.h"

#include <map>

#include "base/fast_rand.h"

#include <atomic>

#include "base/logging.h"
#include "base/time.h"
#include "rpc/load_balancer.h"

namespace bam {

void EndRPC(Controller* cntl, SessionId locked_id);  // channel.cc

// ---------------- ParallelChannel ----------------

ParallelChannel::~ParallelChannel() {
  for (Sub& s : subs_) {
    if (s.owned) delete s.channel;
  }
}

int ParallelChannel::Init(const ParallelChannelOptions* options) {
  if (options != nullptr) options_ = *options;
  return 0;
}

int ParallelChannel::AddChannel(ChannelBase* sub, bool owned, CallMapper mapper,
                                ResponseMerger merger) {
  subs_.push_back(Sub{sub, owned, std::move(mapper), std::move(merger)});
  return 0;
}

namespace {

// Aggregation state shared by the N sub-calls.
// Parity: reference ParallelChannelDone (parallel_channel.cpp:90-186).
struct ParallelDone {
  std::atomic<int> pending;      // sub-calls not yet completed
  std::atomic<int> nfailed{0};
  std::atomic<bool> finalized{false};
  std::atomic<int> destroy_refs;  // when 0: delete this
  Controller* main_cntl = nullptr;
  SessionId main_cid = 0;
  IOBuf* main_response = nullptr;
  int fail_limit = 0;
  int nsubs = 0;
  struct SubResult {
    Controller cntl;
    IOBuf response;
    ResponseMerger merger;
    bool issued = false;
    std::atomic<bool> completed{false};  // set before OnSubDone accounting
  };
  std::vector<SubResult> subs;

  explicit ParallelDone(int n) : pending(n), destroy_refs(n), nsubs(n), subs(n) {}

  void OnSubDone(int idx) {
    subs[idx].completed.store(true, std::memory_order_release);
    if (subs[idx].issued && subs[idx].cntl.Failed())
      nfailed.fetch_add(1, std::memory_order_relaxed);
    int left = pending.fetch_sub(1, std::memory_order_acq_rel) - 1;
    int failed = nfailed.load(std::memory_order_relaxed);
    bool over_limit = failed > fail_limit;
    if (left == 0 || (over_limit && !finalized.load(std::memory_order_acquire))) {
      Finalize();
    }
    if (destroy_refs.fetch_sub(1, std::memory_order_acq_rel) == 1) delete this;
  }

  void Finalize() {
    if (finalized.exchange(true, std::memory_order_acq_rel)) return;
    // Lock the main session; if destroyed (main timeout already fired),
    // nothing to deliver.
    void* data = nullptr;
    if (session_lock(main_cid, &data) != 0) return;
    Controller* cntl = main_cntl;
    int failed = nfailed.load(std::memory_order_relaxed);
    if (failed > fail_limit) {
      // propagate the first sub error
      for (auto& s : subs) {
        if (s.issued && s.completed.load(std::memory_order_acquire) && s.cntl.Failed()) {
          cntl->SetFailed(s.cntl.ErrorCode() == ERPCTIMEDOUT ? ERPCTIMEDOUT : ETOOMANYFAILS,
                          "sub-call failed: " + s.cntl.ErrorText());
          break;
        }
      }
      if (!cntl->Failed()) cntl->SetFailed(ETOOMANYFAILS, "too many sub-channel failures");
    } else {
      for (auto& s : subs) {
        if (!s.issued || !s.completed.load(std::memory_order_acquire) || s.cntl.Failed())
          continue;
        if (s.merger != nullptr) {
          if (s.merger(main_response, s.response) != 0) {
            cntl->SetFailed(ERESPONSE, "response merger failed");
            break;
          }
        } else if (main_response != nullptr) {
          main_response->append(s.response);
        }
      }
    }
    EndRPC(cntl, main_cid);
  }
};

// Main-session error handler (timeout): finalize with the error.
int ParallelMainOnError(SessionId id, void* data, int error_code) {
  Controller* cntl = (Controller*)data;
  cntl->SetFailed(error_code,
                  error_code == ERPCTIMEDOUT ? "RPC deadline exceeded" : "parallel call error");
  EndRPC(cntl, id);
  return 0;
}

void ParallelTimeoutCb(void* a, void* /*b*/) {
  session_error((SessionId)(uintptr_t)a, ERPCTIMEDOUT);
}

}  // namespace

void ParallelChannel::CallMethod(const std::string& full_method, Controller* cntl,
                                 const IOBuf* request, IOBuf* response, Closure* done) {
  cntl->start_us_ = monotonic_time_us();
  if (cntl->timeout_ms_ == -1) cntl->timeout_ms_ = options_.timeout_ms;
  const int n = (int)subs_.size();
  if (n == 0) {
    cntl->SetFailed(EINTERNAL, "ParallelChannel has no sub channels");
    if (done) done->Run();
    return;
  }
  SessionId cid;
  session_create(&cid, cntl, ParallelMainOnError, 1);
  cntl->cid_ = cid;
  cntl->call.cid = cid;
  cntl->call.done = done;
  cntl->call.response = response;

  ParallelDone* pd = new ParallelDone(n);
  pd->main_cntl = cntl;
  pd->main_cid = cid;
  pd->main_response = response;
  // fail_limit counts tolerated sub-call failures; default (-1) = none.
  pd->fail_limit = options_.fail_limit < 0 ? 0 : options_.fail_limit;

  if (cntl->timeout_ms_ > 0) {
    cntl->call.timeout_timer = timer_add(cntl->start_us_ + cntl->timeout_ms_ * 1000,
                                         ParallelTimeoutCb, (void*)(uintptr_t)cid, nullptr);
  }

  const int64_t sub_timeout = cntl->timeout_ms_;  // cntl may be freed by an
                                                  // early async completion
  for (int i = 0; i < n; ++i) {
    ParallelDone::SubResult& sr = pd->subs[i];
    IOBuf sub_request;
    bool use = true;
    if (subs_[i].mapper != nullptr) {
      use = subs_[i].mapper(i, request != nullptr ? *request : IOBuf(), &sub_request);
    } else if (request != nullptr) {
      sub_request = *request;  // zero-copy ref share
    }
    sr.merger = subs_[i].merger;
    if (!use) {
      // skipped sub counts as instant success with empty response
      sr.issued = false;
      pd->OnSubDone(i);
      continue;
    }
    sr.issued = true;
    sr.cntl.set_timeout_ms(sub_timeout);
    Closure* sub_done = NewCallback([pd, i] { pd->OnSubDone(i); });
    subs_[i].channel->CallMethod(full_method, &sr.cntl, &sub_request, &sr.response, sub_done);
  }
  if (done == nullptr) session_join(cid);
}

// ---------------- SelectiveChannel ----------------

SelectiveChannel::~SelectiveChannel() {
  for (auto& s : subs_) delete s->channel;
}

int SelectiveChannel::Init(const char* /*lb_name*/, const SelectiveChannelOptions* opt) {
  if (opt != nullptr) options_ = *opt;
  return 0;
}

int SelectiveChannel::AddChannel(ChannelBase* sub, size_t* handle) {
  auto s = std::make_unique<Sub>();
  s->channel = sub;
  subs_.push_back(std::move(s));
  if (handle != nullptr) *handle = subs_.size() - 1;
  return 0;
}

void SelectiveChannel::CallMethod(const std::string& full_method, Controller* cntl,
                                  const IOBuf* request, IOBuf* response, Closure* done) {
  if (subs_.empty()) {
    cntl->SetFailed(EINTERNAL, "SelectiveChannel has no sub channels");
    if (done) done->Run();
    return;
  }
  if (cntl->timeout_ms_ == -1) cntl->timeout_ms_ = options_.timeout_ms;
  // Failover loop (synchronous semantics; async done is invoked at the end).
  int attempts = std::min<int>(options_.max_retry + 1, (int)subs_.size());
  int64_t now = monotonic_time_us();
  for (int a = 0; a < attempts; ++a) {
    // pick next healthy sub
    Sub* pick = nullptr;
    for (size_t k = 0; k < subs_.size(); ++k) {
      Sub* s = subs_[rr_.fetch_add(1, std::memory_order_relaxed) % subs_.size()].get();
      if (s->isolated_until_us.load(std::memory_order_relaxed) <= now) {
        pick = s;
        break;
      }
    }
    if (pick == nullptr) pick = subs_[0].get();  // all isolated: try anyway
    Controller sub_cntl;
    sub_cntl.set_timeout_ms(cntl->timeout_ms_);
    IOBuf sub_resp;
    pick->channel->CallMethod(full_method, &sub_cntl, request, &sub_resp, nullptr);
    if (!sub_cntl.Failed()) {
      pick->consecutive_failures.store(0, std::memory_order_relaxed);
      if (response != nullptr) response->swap(sub_resp);
      cntl->error_code_ = 0;
      if (done) done->Run();
      return;
    }
    int fails = pick->consecutive_failures.fetch_add(1, std::memory_order_relaxed) + 1;
    if (fails >= 3) {
      pick->isolated_until_us.store(now + 3000000, std::memory_order_relaxed);  // 3s isolation
    }
    cntl->SetFailed(sub_cntl.ErrorCode(), sub_cntl.ErrorText());
    if (sub_cntl.ErrorCode() == ERPCTIMEDOUT) break;  // deadline spent
  }
  if (done) done->Run();
}

// ---------------- PartitionChannel ----------------

int ResolvePartitionedNaming(const std::string& url, int num_partitions,
                             std::vector<std::vector<EndPoint>>* partitions) {
  partitions->assign(num_partitions, {});
  std::string body = url;
  auto scheme = url.find("://");
  std::string prefix;
  if (scheme != std::string::npos) {
    prefix = url.substr(0, scheme + 3);
    body = url.substr(scheme + 3);
  }
  size_t pos = 0;
  while (pos < body.size()) {
    size_t comma = body.find(',', pos);
    std::string item = body.substr(pos, comma == std::string::npos ? comma : comma - pos);
    pos = comma == std::string::npos ? body.size() : comma + 1;
    if (item.empty()) continue;
    int part = -1;
    std::string addr = item;
    size_t slash = item.find('/');
    size_t space = item.find(' ');
    if (slash != std::string::npos && space != std::string::npos && slash < space) {
      part = atoi(item.substr(0, slash).c_str());
      addr = item.substr(space + 1);
    }
    EndPoint ep;
    if (str2endpoint(addr.c_str(), &ep) != 0) continue;
    if (part >= 0 && part < num_partitions) {
      (*partitions)[part].push_back(ep);
    } else if (part < 0) {
      for (auto& p : *partitions) p.push_back(ep);
    }
  }
  return 0;
}

int PartitionChannel::Init(int num_partitions, const char* naming_url,
                           const PartitionChannelOptions* opt) {
  PartitionChannelOptions o;
  if (opt != nullptr) o = *opt;
  std::vector<std::vector<EndPoint>> parts;
  if (ResolvePartitionedNaming(naming_url, num_partitions, &parts) != 0) return -1;
  parallel_.Init(&o.parallel);
  for (int i = 0; i < num_partitions; ++i) {
    if (parts[i].empty()) return -1;  // a partition without servers is fatal
    std::string list = "list://";
    for (size_t k = 0; k < parts[i].size(); ++k) {
      if (k) list += ",";
      list += endpoint2str(parts[i][k]);
    }
    Channel* ch = new Channel;
    if (ch->Init(list.c_str(), o.lb_name.c_str(), &o.sub_options) != 0) {
      delete ch;
      return -1;
    }
    parallel_.AddChannel(ch, /*owned=*/true);
  }
  return 0;
}

// ---------------- DynamicPartitionChannel ----------------

DynamicPartitionChannel::~DynamicPartitionChannel() {
  for (Scheme& sc : schemes_) delete sc.chan;
}

int DynamicPartitionChannel::Init(const char* naming_url, const PartitionChannelOptions* opt) {
  // Group "i/N addr" entries by N; untagged entries are rejected here
  // (they belong to every scheme in PartitionChannel, which is ambiguous
  // across schemes).
  std::string url = naming_url;
  std::string body = url;
  std::string prefix = "list://";
  auto scheme_sep = url.find("://");
  if (scheme_sep != std::string::npos) body = url.substr(scheme_sep + 3);
  std::map<int, std::vector<std::string>> groups;  // N -> ["i/N addr", ...]
  size_t pos = 0;
  while (pos < body.size()) {
    size_t comma = body.find(',', pos);
    std::string item =
        body.substr(pos, comma == std::string::npos ? comma : comma - pos);
    pos = comma == std::string::npos ? body.size() : comma + 1;
    if (item.empty()) continue;
    size_t slash = item.find('/');
    size_t space = item.find(' ');
    if (slash == std::string::npos || space == std::string::npos || slash >= space) continue;
    int nparts = atoi(item.substr(slash + 1, space - slash - 1).c_str());
    if (nparts <= 0) continue;
    groups[nparts].push_back(item);
  }
  if (groups.empty()) return -1;
  for (auto& kv : groups) {
    std::string sub_url = prefix;
    for (size_t i = 0; i < kv.second.size(); ++i) {
      if (i) sub_url += ",";
      sub_url += kv.second[i];
    }
    PartitionChannel* pc = new PartitionChannel;
    if (pc->Init(kv.first, sub_url.c_str(), opt) != 0) {
      delete pc;  // scheme with an empty partition: skip it entirely
      continue;
    }
    Scheme sc;
    sc.num_partitions = kv.first;
    sc.capacity = (int)kv.second.size();
    sc.chan = pc;
    schemes_.push_back(sc);
    total_capacity_ += sc.capacity;
  }
  return schemes_.empty() ? -1 : 0;
}

void DynamicPartitionChannel::CallMethod(const std::string& full_method, Controller* cntl,
                                         const IOBuf* request, IOBuf* response, Closure* done) {
  // Capacity-weighted scheme pick (traffic splits by server counts, so a
  // growing N-partition deployment takes over proportionally).
  uint64_t r = fast_rand() % (uint64_t)total_capacity_;
  for (Scheme& sc : schemes_) {
    if (r < (uint64_t)sc.capacity) {
      sc.chan->CallMethod(full_method, cntl, request, response, done);
      return;
    }
    r -= sc.capacity;
  }
  schemes_.back().chan->CallMethod(full_method, cntl, request, response, done);
}

void PartitionChannel::CallMethod(const std::string& full_method, Controller* cntl,
                                  const IOBuf* request, IOBuf* response, Closure* done) {
  parallel_.CallMethod(full_method, cntl, request, response, done);
}

}  // namespace bam
