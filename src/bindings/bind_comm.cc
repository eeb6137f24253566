// Python surface for CommGroup (in-framework RCCL/xGMI collectives +
// TCP-mesh control plane). Handles are process-local ints.
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstring>
#include <map>
#include <mutex>
#include <string>

#include <algorithm>
#include <vector>

#include "base/gpu_loader.h"
#include "base/time.h"
#include "bindings/bind.h"
#include "rpc/collective_channel.h"
#include "rpc/comm_group.h"
#include "rpc/server.h"
#include "rpc/stream.h"

namespace {

std::mutex g_mu;
std::map<int, bam::CommGroup*> g_groups;
int g_next = 1;

bam::CommGroup* get(int h) {
  std::lock_guard<std::mutex> lk(g_mu);
  auto it = g_groups.find(h);
  return it == g_groups.end() ? nullptr : it->second;
}

// ---- GPU-side selftests / micro-benches (no torch in the data path) ----

// Broadcast + AllGather on HBM buffers, verified against host patterns.
bool gpu_collective_roundtrip(int h, size_t n) {
  bam::CommGroup* g = get(h);
  const bam::gpu::GpuApi* api = bam::gpu::api();
  if (g == nullptr || api == nullptr) return false;
  const int rank = g->rank(), nranks = g->nranks(), dev = 0;
  char* buf = (char*)api->alloc_hbm((uint32_t)n, dev);
  char* gathered = (char*)api->alloc_hbm((uint32_t)(n * nranks), dev);
  if (buf == nullptr || gathered == nullptr) return false;
  std::string host(n, 0);
  bool ok = true;
  // Broadcast: root fills 0xB7-pattern, everyone must read it back.
  if (rank == 0) {
    for (size_t i = 0; i < n; ++i) host[i] = (char)(0xB7 ^ (i & 0xff));
    api->memcpy_res(buf, 2, dev, host.data(), 0, 0, n);
  }
  ok = ok && g->Broadcast(buf, n, 0) == 0;
  std::string back(n, 1);
  api->memcpy_res(&back[0], 0, 0, buf, 2, dev, n);
  for (size_t i = 0; ok && i < n; ++i)
    ok = back[i] == (char)(0xB7 ^ (i & 0xff));
  // AllGather: every rank contributes its rank byte.
  std::string mine(n, (char)(0x40 + rank));
  api->memcpy_res(buf, 2, dev, mine.data(), 0, 0, n);
  ok = ok && g->AllGather(buf, gathered, n) == 0;
  std::string all(n * nranks, 0);
  api->memcpy_res(&all[0], 0, 0, gathered, 2, dev, n * nranks);
  for (int r = 0; ok && r < nranks; ++r)
    for (size_t i = 0; ok && i < n; ++i)
      ok = all[(size_t)r * n + i] == (char)(0x40 + r);
  api->free_hbm(buf, (uint32_t)n, dev);
  api->free_hbm(gathered, (uint32_t)(n * nranks), dev);
  return ok;
}

// p2p frame exchange throughput (config-3 shape): each iteration moves one
// `frame` of HBM bytes to `peer` and receives one back (full duplex over
// xGMI). Returns GB/s of payload moved out of this rank.
double gpu_p2p_gbps(int h, int peer, size_t frame, int iters) {
  bam::CommGroup* g = get(h);
  const bam::gpu::GpuApi* api = bam::gpu::api();
  if (g == nullptr || api == nullptr || iters <= 0) return -1;
  const int dev = 0;
  char* sbuf = (char*)api->alloc_hbm((uint32_t)frame, dev);
  char* rbuf = (char*)api->alloc_hbm((uint32_t)frame, dev);
  if (sbuf == nullptr || rbuf == nullptr) return -1;
  api->fill(sbuf, frame, 0x5a5a5a5a5a5a5a5aULL, dev);
  // warm
  if (g->SendRecv(sbuf, frame, peer, rbuf, frame, peer) != 0) return -1;
  int64_t t0 = bam::monotonic_time_us();
  for (int i = 0; i < iters; ++i) {
    if (g->SendRecv(sbuf, frame, peer, rbuf, frame, peer) != 0) return -1;
  }
  int64_t us = bam::monotonic_time_us() - t0;
  api->free_hbm(sbuf, (uint32_t)frame, dev);
  api->free_hbm(rbuf, (uint32_t)frame, dev);
  return (double)frame * iters / (us * 1e-6) / 1e9;
}

// ---- collective fan-out (BASELINE config 4) ----

std::mutex g_srv_mu;
std::vector<bam::Server*> g_coll_servers;

// Starts a Server on `port` (0 = ephemeral) with the collective service
// bound to group `h`. Returns the bound port.
int fanout_serve(int h, int port) {
  bam::CommGroup* g = get(h);
  if (g == nullptr) throw std::runtime_error("bad group handle");
  auto* srv = new bam::Server;
  if (bam::RegisterCollectiveService(srv, g) != 0)
    throw std::runtime_error("RegisterCollectiveService failed");
  if (srv->Start(port, nullptr) != 0) throw std::runtime_error("server start failed");
  std::lock_guard<std::mutex> lk(g_srv_mu);
  g_coll_servers.push_back(srv);
  return srv->listen_address().port;
}

// Runs `rounds` collective fan-out calls from this (caller) rank and
// reports qps + latency percentiles. Payload is uploaded to HBM for
// "rccl" groups. verify: check every gathered slot echoes len bytes.
struct FanoutResult {
  int rc = 0;
  std::string error;
  int rounds = 0;
  double qps = 0, p50_us = 0, p99_us = 0;
  bool data_ok = false;
};

FanoutResult fanout_call_impl(int h, const std::vector<std::string>& addrs,
                              const std::string& method, const std::string& payload,
                              size_t resp_cap, int rounds, bool verify) {
  bam::CommGroup* g = get(h);
  if (g == nullptr) throw std::runtime_error("bad group handle");
  const bool device = g->backend() == "rccl";
  bam::CollectiveChannel ch;
  if (ch.Init(g, addrs) != 0) throw std::runtime_error("CollectiveChannel: " + ch.last_error());
  const size_t slot = bam::CollectiveChannel::slot_size(resp_cap);
  const size_t gathered_bytes = slot * g->nranks();
  void* req = nullptr;
  void* gathered = nullptr;
  const bam::gpu::GpuApi* api = bam::gpu::api();
  if (device) {
    req = api->alloc_hbm((uint32_t)payload.size(), 0);
    gathered = api->alloc_hbm((uint32_t)gathered_bytes, 0);
    api->memcpy_res(req, 2, 0, payload.data(), 0, 0, payload.size());
  } else {
    req = malloc(payload.size());
    gathered = malloc(gathered_bytes);
    memcpy(req, payload.data(), payload.size());
  }
  std::vector<double> lat_us;
  lat_us.reserve(rounds);
  int rc = 0;
  std::string first_err;
  int64_t t0 = bam::monotonic_time_us();
  for (int i = 0; i < rounds && rc == 0; ++i) {
    int64_t s = bam::monotonic_time_us();
    rc = ch.Call(method, req, payload.size(), gathered, resp_cap);
    if (rc != 0) first_err = ch.last_error();
    lat_us.push_back((double)(bam::monotonic_time_us() - s));
  }
  int64_t total_us = bam::monotonic_time_us() - t0;
  bool data_ok = true;
  if (rc == 0 && verify) {
    std::string host(gathered_bytes, 0);
    if (device) {
      api->memcpy_res(&host[0], 0, 0, gathered, 2, 0, gathered_bytes);
    } else {
      memcpy(&host[0], gathered, gathered_bytes);
    }
    for (int r = 0; r < g->nranks() && data_ok; ++r) {
      uint64_t len = 0;
      memcpy(&len, host.data() + (size_t)r * slot, 8);
      if (method == "echo") {
        data_ok = len == payload.size() &&
                  memcmp(host.data() + (size_t)r * slot + 8, payload.data(), len) == 0;
      } else {
        data_ok = len > 0 && len <= resp_cap;
      }
    }
  }
  if (device) {
    api->free_hbm(req, (uint32_t)payload.size(), 0);
    api->free_hbm(gathered, (uint32_t)gathered_bytes, 0);
  } else {
    free(req);
    free(gathered);
  }
  std::sort(lat_us.begin(), lat_us.end());
  auto pct = [&](double p) {
    if (lat_us.empty()) return 0.0;
    size_t i = (size_t)(p * (lat_us.size() - 1));
    return lat_us[i];
  };
  FanoutResult out;
  out.rc = rc;
  out.error = first_err;
  out.rounds = rounds;
  out.qps = total_us > 0 ? rounds * 1e6 / total_us : 0.0;
  out.p50_us = pct(0.5);
  out.p99_us = pct(0.99);
  out.data_ok = data_ok;
  return out;
}

// py::dict must only be touched WITH the GIL; the blocking work runs
// without it.
py::dict fanout_call(int h, const std::vector<std::string>& addrs, const std::string& method,
                     const std::string& payload, size_t resp_cap, int rounds, bool verify) {
  FanoutResult r;
  {
    py::gil_scoped_release rel;
    r = fanout_call_impl(h, addrs, method, payload, resp_cap, rounds, verify);
  }
  py::dict out;
  out["rc"] = r.rc;
  out["error"] = r.error;
  out["rounds"] = r.rounds;
  out["qps"] = r.qps;
  out["p50_us"] = r.p50_us;
  out["p99_us"] = r.p99_us;
  out["data_ok"] = r.data_ok;
  return out;
}

// ---- streaming over the comm data plane (BASELINE config 3) ----

// Receiver: a Server whose "StreamService.OpenComm" accepts streams bound
// to group `h` with data arriving from group-rank `peer` (RCCL p2p over
// xGMI for "rccl" groups). Frames land in HBM blocks; the sink counts
// bytes and releases them.
int stream_comm_serve(int h, int peer, int port) {
  bam::CommGroup* g = get(h);
  if (g == nullptr) throw std::runtime_error("bad group handle");
  auto* srv = new bam::Server;
  auto* svc = new bam::Service("StreamService");
  svc->AddMethod("OpenComm", [g, peer](bam::Controller* cntl, const bam::IOBuf& req,
                                       bam::IOBuf* resp, bam::Closure* done) {
    bam::StreamOptions sopt;
    sopt.max_buf_size = 256u << 20;
    sopt.gpu_group = g;
    sopt.gpu_peer = peer;
    sopt.on_received = [](bam::StreamId, bam::IOBuf* msg) { msg->clear(); };
    sopt.on_closed = [](bam::StreamId sid) { bam::StreamClose(sid); };
    bam::StreamId sid;
    if (bam::StreamAccept(&sid, cntl, sopt) != 0) cntl->SetFailed(1003, "no stream");
    resp->append("ok");
    done->Run();
  });
  srv->AddService(svc, bam::SERVER_OWNS_SERVICE);
  if (srv->Start(port, nullptr) != 0) throw std::runtime_error("server start failed");
  std::lock_guard<std::mutex> lk(g_srv_mu);
  g_coll_servers.push_back(srv);
  return srv->listen_address().port;
}

// Sender: streams `frames` frames of `frame_size` bytes to the server at
// `addr`, payload moving over the group's data plane to `peer`. Returns
// GB/s of payload.
double stream_comm_send(int h, const std::string& addr, int peer, int frames,
                        size_t frame_size) {
  bam::CommGroup* g = get(h);
  if (g == nullptr) throw std::runtime_error("bad group handle");
  const bool device = g->backend() == "rccl";
  bam::Channel channel;
  bam::ChannelOptions copt;
  copt.timeout_ms = 20000;
  if (channel.Init(addr.c_str(), &copt) != 0) return -1;
  bam::StreamOptions sopt;
  sopt.max_buf_size = 256u << 20;
  sopt.gpu_group = g;
  sopt.gpu_peer = peer;
  bam::Controller cntl;
  bam::StreamId sid;
  bam::StreamCreate(&sid, &cntl, sopt);
  bam::IOBuf request, response;
  request.append("open");
  channel.CallMethod("StreamService.OpenComm", &cntl, &request, &response, nullptr);
  if (cntl.Failed()) {
    bam::StreamClose(sid);
    return -2;
  }
  const bam::gpu::GpuApi* api = bam::gpu::api();
  std::string host_frame(frame_size, 'X');
  int64_t t0 = bam::monotonic_time_us();
  for (int i = 0; i < frames; ++i) {
    bam::IOBuf data;
    void* p = nullptr;
    if (data.append_writable_block(frame_size, device ? bam::RES_HBM : bam::RES_HOST, 0,
                                   &p) != 0) {
      bam::StreamClose(sid);
      return -4;
    }
    if (device) {
      if (i == 0) api->fill(p, frame_size, 0x5858585858585858ULL, 0);
    } else {
      memcpy(p, host_frame.data(), frame_size);
    }
    if (bam::StreamWrite(sid, &data) != 0) {
      bam::StreamClose(sid);
      return -3;
    }
  }
  int64_t us = bam::monotonic_time_us() - t0;
  bam::StreamClose(sid);
  return (double)frames * frame_size / (us * 1e-6) / 1e9;
}

}  // namespace

void bind_comm(py::module_& m) {
  auto c = m.def_submodule("comm");
  c.def(
      "create",
      [](int nranks, int rank, const std::string& backend, const std::string& host,
         int base_port, int dev, int timeout_ms) {
        bam::CommGroup::Options o;
        o.nranks = nranks;
        o.rank = rank;
        o.backend = backend;
        o.host = host;
        o.base_port = base_port;
        o.dev = dev;
        o.connect_timeout_ms = timeout_ms;
        std::string err;
        bam::CommGroup* g = bam::CommGroup::Create(o, &err);
        if (g == nullptr) throw std::runtime_error("CommGroup: " + err);
        std::lock_guard<std::mutex> lk(g_mu);
        int h = g_next++;
        g_groups[h] = g;
        return h;
      },
      py::arg("nranks"), py::arg("rank"), py::arg("backend") = "tcp",
      py::arg("host") = "127.0.0.1", py::arg("base_port") = 0, py::arg("dev") = 0,
      py::arg("timeout_ms") = 30000, py::call_guard<py::gil_scoped_release>());
  c.def("destroy", [](int h) {
    bam::CommGroup* g = nullptr;
    {
      std::lock_guard<std::mutex> lk(g_mu);
      auto it = g_groups.find(h);
      if (it != g_groups.end()) {
        g = it->second;
        g_groups.erase(it);
      }
    }
    delete g;
  });
  c.def("rank", [](int h) { return get(h)->rank(); });
  c.def("nranks", [](int h) { return get(h)->nranks(); });
  c.def(
      "barrier", [](int h) { return get(h)->Barrier(); },
      py::call_guard<py::gil_scoped_release>());
  // Host-buffer collectives (backend "tcp"; used by the CPU tests).
  c.def(
      "broadcast",
      [](int h, py::bytes data, size_t n, int root) {
        bam::CommGroup* g = get(h);
        std::string buf(data);
        buf.resize(n);
        int rc;
        {
          py::gil_scoped_release rel;
          rc = g->Broadcast(&buf[0], n, root);
        }
        if (rc != 0) throw std::runtime_error("broadcast failed");
        return py::bytes(buf);
      },
      py::arg("h"), py::arg("data"), py::arg("n"), py::arg("root"));
  c.def(
      "allgather",
      [](int h, py::bytes data) {
        bam::CommGroup* g = get(h);
        std::string mine(data);
        std::string all(mine.size() * g->nranks(), 0);
        int rc;
        {
          py::gil_scoped_release rel;
          rc = g->AllGather(mine.data(), &all[0], mine.size());
        }
        if (rc != 0) throw std::runtime_error("allgather failed");
        return py::bytes(all);
      },
      py::arg("h"), py::arg("data"));
  c.def(
      "send",
      [](int h, int peer, py::bytes data) {
        std::string buf(data);
        py::gil_scoped_release rel;
        if (get(h)->Send(buf.data(), buf.size(), peer) != 0)
          throw std::runtime_error("send failed");
      },
      py::arg("h"), py::arg("peer"), py::arg("data"));
  c.def(
      "recv",
      [](int h, int peer, size_t n) {
        std::string buf(n, 0);
        {
          py::gil_scoped_release rel;
          if (get(h)->Recv(&buf[0], n, peer) != 0) throw std::runtime_error("recv failed");
        }
        return py::bytes(buf);
      },
      py::arg("h"), py::arg("peer"), py::arg("n"));
  c.def(
      "host_broadcast",
      [](int h, py::bytes data, int root) {
        std::string blob(data);
        {
          py::gil_scoped_release rel;
          if (get(h)->HostBroadcast(&blob, root) != 0)
            throw std::runtime_error("host_broadcast failed");
        }
        return py::bytes(blob);
      },
      py::arg("h"), py::arg("data"), py::arg("root"));
  // GPU data-plane checks (backend "rccl").
  c.def("gpu_collective_roundtrip", &gpu_collective_roundtrip, py::arg("h"), py::arg("n"),
        py::call_guard<py::gil_scoped_release>());
  c.def("gpu_p2p_gbps", &gpu_p2p_gbps, py::arg("h"), py::arg("peer"), py::arg("frame"),
        py::arg("iters") = 20, py::call_guard<py::gil_scoped_release>());
  // Collective fan-out (CollectiveChannel, BASELINE config 4).
  c.def("fanout_serve", &fanout_serve, py::arg("h"), py::arg("port") = 0,
        py::call_guard<py::gil_scoped_release>());
  c.def("fanout_call", &fanout_call, py::arg("h"), py::arg("addrs"), py::arg("method"),
        py::arg("payload"), py::arg("resp_cap"), py::arg("rounds") = 1,
        py::arg("verify") = true);
  // Streaming over the comm data plane (CommGroup p2p; config 3).
  c.def("stream_comm_serve", &stream_comm_serve, py::arg("h"), py::arg("peer"),
        py::arg("port") = 0, py::call_guard<py::gil_scoped_release>());
  c.def("stream_comm_send", &stream_comm_send, py::arg("h"), py::arg("addr"),
        py::arg("peer"), py::arg("frames"), py::arg("frame_size"),
        py::call_guard<py::gil_scoped_release>());
}
