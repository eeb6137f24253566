#include "rpc/collective_channel.h"

#include <string.h>

#include <map>
#include <mutex>

#include "base/gpu_loader.h"
#include "base/logging.h"
#include "base/snappy.h"
#include "fiber/sync.h"
#include "rpc/channel.h"
#include "rpc/rpc_errno.h"
#include "rpc/server.h"

namespace bam {

namespace {

constexpr uint32_t kMagic = 0xC011EC70u;

#pragma pack(push, 1)
struct ControlHdr {
  uint32_t magic;
  uint32_t round;
  uint64_t req_len;
  uint64_t resp_cap;
  uint32_t root;
  uint16_t method_len;
};
#pragma pack(pop)

// ---- residency-aware scratch ----

void* coll_alloc(bool device, size_t n) {
  if (device) return gpu::api()->alloc_hbm((uint32_t)n, 0);
  return malloc(n);
}

void coll_free(bool device, void* p, size_t n) {
  if (p == nullptr) return;
  if (device) gpu::api()->free_hbm(p, (uint32_t)n, 0);
  else free(p);
}

void coll_copy(bool device, void* dst, const void* src, size_t n) {
  if (device) gpu::api()->memcpy_res(dst, 2, 0, src, 2, 0, n);
  else memcpy(dst, src, n);
}

// A growable buffer bound to one residency.
struct Scratch {
  void* p = nullptr;
  size_t cap = 0;
  bool device = false;
  ~Scratch() { coll_free(device, p, cap); }
  void* ensure(bool dev, size_t n) {
    if (p != nullptr && (device != dev || cap < n)) {
      coll_free(device, p, cap);
      p = nullptr;
    }
    if (p == nullptr) {
      device = dev;
      cap = n;
      p = coll_alloc(dev, n);
    }
    return p;
  }
};

// ---- method registry ----

std::mutex g_reg_mu;
std::map<std::string, CollectiveMethodFn>* g_registry = nullptr;

int echo_fn(const void* req, size_t len, void* resp, size_t resp_cap, size_t* resp_len,
            int dev) {
  if (len > resp_cap) return EINVAL;
  coll_copy(dev >= 0, resp, req, len);
  *resp_len = len;
  return 0;
}

// The 16KB+snappy server work: decompress the broadcast payload, then
// recompress the response — both on the GPU for device groups
// (hip/snappy.hip kernels), host codec otherwise.
// Stage-coded errors so GPU failures are diagnosable from test output.
thread_local std::string g_snappy_echo_err;

int snappy_echo_fn(const void* req, size_t len, void* resp, size_t resp_cap,
                   size_t* resp_len, int dev) {
  if (dev >= 0) {
    const gpu::GpuApi* api = gpu::api();
    // The snappy stream's preamble is the uncompressed length as a varint
    // — read it from the device copy to size the scratch exactly.
    unsigned char head[8] = {0};
    api->memcpy_res(head, 0, 0, req, 2, 0, len < 8 ? len : 8);
    uint64_t plain_len_hdr = 0;
    int shift = 0;
    for (int i = 0; i < 8; ++i) {
      plain_len_hdr |= (uint64_t)(head[i] & 0x7f) << shift;
      if ((head[i] & 0x80) == 0) break;
      shift += 7;
    }
    Scratch plain;  // decompressed intermediate
    size_t plain_cap = plain_len_hdr + 4096;
    void* pp = plain.ensure(true, plain_cap);
    if (pp == nullptr) return ENOMEM;
    size_t plain_len = 0;
    if (api->snappy_decompress(req, len, pp, plain_cap, &plain_len, dev) != 0) {
      g_snappy_echo_err = std::string("gpu decompress failed (len=") +
                          std::to_string(len) + ", cap=" + std::to_string(plain_cap) +
                          "): " + api->last_error();
      return EINVAL;
    }
    // The device compressor needs worst-case output room (like host
    // MaxCompressedLength); compress into scratch, then move the actual
    // bytes into the (tightly sized) response slot.
    Scratch comp;
    size_t comp_cap = plain_len + plain_len / 3 + 128;
    void* cp = comp.ensure(true, comp_cap);
    if (cp == nullptr) return ENOMEM;
    size_t comp_len = 0;
    if (api->snappy_compress(pp, plain_len, cp, comp_cap, &comp_len, dev) != 0) {
      g_snappy_echo_err = std::string("gpu compress failed (plain=") +
                          std::to_string(plain_len) + ", cap=" + std::to_string(comp_cap) +
                          "): " + api->last_error();
      return EINVAL;
    }
    if (comp_len > resp_cap) {
      g_snappy_echo_err = "compressed " + std::to_string(comp_len) + " > resp_cap " +
                          std::to_string(resp_cap);
      return EINVAL;
    }
    api->memcpy_res(resp, 2, 0, cp, 2, 0, comp_len);
    *resp_len = comp_len;
    return 0;
  }
  std::string in((const char*)req, len), plain, out;
  if (!snappy::Uncompress(in.data(), in.size(), &plain)) return EINVAL;
  snappy::Compress(plain.data(), plain.size(), &out);
  if (out.size() > resp_cap) return EINVAL;
  memcpy(resp, out.data(), out.size());
  *resp_len = out.size();
  return 0;
}

std::map<std::string, CollectiveMethodFn>& registry() {
  if (g_registry == nullptr) {
    g_registry = new std::map<std::string, CollectiveMethodFn>;
    (*g_registry)["echo"] = echo_fn;
    (*g_registry)["snappy_echo"] = snappy_echo_fn;
    // Test method: echoes unless BAM_COLL_FAIL=1 in THIS process — drives
    // the error-marker slot path without desyncing the group.
    (*g_registry)["maybe_fail_echo"] = [](const void* req, size_t req_len, void* resp,
                                          size_t resp_cap, size_t* resp_len, int dev) {
      const char* e = getenv("BAM_COLL_FAIL");
      if (e != nullptr && e[0] == '1') return 77;
      return echo_fn(req, req_len, resp, resp_cap, resp_len, dev);
    };
  }
  return *g_registry;
}

CollectiveMethodFn find_method(const std::string& name) {
  std::lock_guard<std::mutex> lk(g_reg_mu);
  auto& reg = registry();
  auto it = reg.find(name);
  return it == reg.end() ? CollectiveMethodFn() : it->second;
}

// ---- one participant round (shared by server handler and caller) ----
// Returns 0 or errno. `local_slot`/`gathered` are managed scratches.
int run_round(CommGroup* group, const std::string& method, const void* req_ext,
              size_t req_len, size_t resp_cap, int root, Scratch* req_scr,
              Scratch* slot_scr, void* gathered_ext, Scratch* gathered_scr,
              std::string* err) {
  const bool device = group->backend() == "rccl";
  const int dev = device ? 0 : -1;
  const size_t slot = CollectiveChannel::slot_size(resp_cap);
  CollectiveMethodFn fn = find_method(method);
  if (!fn) {
    *err = "unknown collective method " + method;
    return ENOMETHOD;
  }
  // Request buffer: the caller passes its own (req_ext); participants
  // receive into scratch.
  void* req = (void*)req_ext;
  if (req == nullptr) {
    req = req_scr->ensure(device, req_len > 0 ? req_len : 1);
    if (req == nullptr) return ENOMEM;
  }
  if (group->Broadcast(req, req_len, root) != 0) {
    *err = "broadcast failed";
    return EINTERNAL;
  }
  void* myslot = slot_scr->ensure(device, slot);
  if (myslot == nullptr) return ENOMEM;
  size_t resp_len = 0;
  int rc = fn(req, req_len, (char*)myslot + 8, resp_cap, &resp_len, dev);
  if (rc != 0) {
    *err = "method " + method + " failed rc=" + std::to_string(rc);
    if (!g_snappy_echo_err.empty()) {
      *err += ": " + g_snappy_echo_err;
      g_snappy_echo_err.clear();
    }
    // DO NOT bail before the all-gather: the peers are already committed
    // to the collective and a missing participant hangs the group. Ship
    // an error marker (len = kErrorSlot) so every rank sees a per-rank
    // failure instead, then report the local error after the gather.
  }
  uint64_t hdr = rc == 0 ? resp_len : CollectiveChannel::kErrorSlot;
  if (device) {
    gpu::api()->memcpy_res(myslot, 2, 0, &hdr, 0, 0, 8);
  } else {
    memcpy(myslot, &hdr, 8);
  }
  void* gathered = gathered_ext;
  if (gathered == nullptr) {
    gathered = gathered_scr->ensure(device, slot * group->nranks());
    if (gathered == nullptr) return ENOMEM;
  }
  if (group->AllGather(myslot, gathered, slot) != 0) {
    *err = "allgather failed";
    return EINTERNAL;
  }
  return rc;
}

struct ServerGroupState {
  CommGroup* group;
  Scratch req, slot, gathered;
  std::mutex mu;  // one round at a time per group
};

}  // namespace

void RegisterCollectiveMethod(const std::string& name, CollectiveMethodFn fn) {
  std::lock_guard<std::mutex> lk(g_reg_mu);
  registry()[name] = std::move(fn);
}

int RegisterCollectiveService(Server* server, CommGroup* group) {
  if (server == nullptr || group == nullptr) return -1;
  auto* st = new ServerGroupState;  // lives as long as the process
  st->group = group;
  Service* svc = new Service("Collective");
  svc->AddMethod("Round", [st](Controller* cntl, const IOBuf& request, IOBuf* response,
                               Closure* done) {
    std::string blob = request.to_string();
    ControlHdr h;
    if (blob.size() < sizeof(h)) {
      cntl->SetFailed(EREQUEST, "short collective control");
      done->Run();
      return;
    }
    memcpy(&h, blob.data(), sizeof(h));
    if (h.magic != kMagic || blob.size() < sizeof(h) + h.method_len) {
      cntl->SetFailed(EREQUEST, "bad collective control");
      done->Run();
      return;
    }
    std::string method(blob.data() + sizeof(h), h.method_len);
    std::string err;
    std::lock_guard<std::mutex> lk(st->mu);
    int rc = run_round(st->group, method, nullptr, h.req_len, h.resp_cap, (int)h.root,
                       &st->req, &st->slot, nullptr, &st->gathered, &err);
    if (rc != 0) {
      cntl->SetFailed(rc, err);
    } else {
      response->append("ok", 2);
    }
    done->Run();
  });
  return server->AddService(svc, SERVER_OWNS_SERVICE);
}

int CollectiveChannel::Init(CommGroup* group, const std::vector<std::string>& server_addrs,
                            const Options* opt) {
  if (group == nullptr || (int)server_addrs.size() != group->nranks()) return -1;
  group_ = group;
  if (opt != nullptr) opt_ = *opt;
  controls_.clear();
  controls_.resize(group->nranks());
  for (int r = 0; r < group->nranks(); ++r) {
    if (r == group->rank()) continue;
    if (server_addrs[r].empty()) {
      err_ = "missing server addr for rank " + std::to_string(r);
      return -1;
    }
    ChannelOptions copt;
    copt.timeout_ms = opt_.timeout_ms;
    copt.max_retry = 0;  // a collective round must not be replayed
    auto* ch = new Channel;
    if (ch->Init(server_addrs[r].c_str(), &copt) != 0) {
      delete ch;
      err_ = "control channel to rank " + std::to_string(r) + " failed";
      return -1;
    }
    controls_[r].reset(ch);
  }
  return 0;
}

int CollectiveChannel::Call(const std::string& method, const void* req, size_t req_len,
                            void* gathered, size_t resp_cap) {
  if (group_ == nullptr) return EINTERNAL;
  static std::mutex call_mu;  // rounds on one process are serialized
  std::lock_guard<std::mutex> lk(call_mu);
  ++round_;
  // 1. control fan-out (metadata only — the payload rides RCCL).
  ControlHdr h{kMagic, round_, req_len, resp_cap, (uint32_t)group_->rank(),
               (uint16_t)method.size()};
  IOBuf control;
  control.append(&h, sizeof(h));
  control.append(method.data(), method.size());
  const int n = group_->nranks();
  struct Pending {
    Controller cntl;
    IOBuf resp;
  };
  std::vector<std::unique_ptr<Pending>> pend(n);
  CountdownEvent all_acked(0);
  int launched = 0;
  for (int r = 0; r < n; ++r) {
    if (r == group_->rank()) continue;
    pend[r].reset(new Pending);
    all_acked.add_count(1);
    ++launched;
    IOBuf creq = control;  // block-ref copy, cheap
    controls_[r]->CallMethod("Collective.Round", &pend[r]->cntl, &creq, &pend[r]->resp,
                             NewCallback([&all_acked] { all_acked.signal(); }));
  }
  // 2-4. participate (broadcast own payload as root, run local method,
  // all-gather into the caller-provided buffer).
  Scratch req_scr, slot_scr, gathered_scr;
  std::string err;
  int rc = run_round(group_, method, req, req_len, resp_cap, group_->rank(), &req_scr,
                     &slot_scr, gathered, &gathered_scr, &err);
  // 5. control acks.
  all_acked.wait();
  for (int r = 0; r < n; ++r) {
    if (pend[r] != nullptr && pend[r]->cntl.Failed() && rc == 0) {
      rc = pend[r]->cntl.ErrorCode();
      err = "rank " + std::to_string(r) + ": " + pend[r]->cntl.ErrorText();
    }
  }
  if (rc != 0) err_ = err;
  (void)launched;
  return rc;
}

}  // namespace bam
