// Combo-channel + LB test scenarios (pytest drives via bindings).
// Models reference test/brpc_parallel_channel_unittest.cpp + LB unittests:
// multiple in-process servers on loopback ports.
#include <set>
#include <string>
#include <vector>

#include "rpc/load_balancer.h"
#include "rpc/parallel_channel.h"
#include "rpc/server.h"

namespace bam {
namespace rpctest {

Service* NewEchoService();
int start_echo_server(int port);

// Fan one echo out to N servers; expect the merged response = N copies.
int parallel_echo_test(const std::vector<int>& ports, const std::string& payload,
                       int fail_limit, std::string* merged, std::string* err) {
  ParallelChannel pchan;
  ParallelChannelOptions popt;
  popt.fail_limit = fail_limit;
  popt.timeout_ms = 3000;
  pchan.Init(&popt);
  for (int port : ports) {
    Channel* sub = new Channel;
    ChannelOptions copt;
    copt.timeout_ms = 2000;
    if (sub->Init(("127.0.0.1:" + std::to_string(port)).c_str(), &copt) != 0) {
      delete sub;
      return -1;
    }
    pchan.AddChannel(sub, /*owned=*/true);
  }
  Controller cntl;
  IOBuf request, response;
  request.append(payload);
  pchan.CallMethod("EchoService.Echo", &cntl, &request, &response, nullptr);
  if (err != nullptr) *err = cntl.ErrorText();
  if (cntl.Failed()) return cntl.ErrorCode();
  if (merged != nullptr) *merged = response.to_string();
  return 0;
}

// SelectiveChannel failover: first sub is a dead port.
int selective_test(int dead_port, int live_port, std::string* resp_out) {
  SelectiveChannel schan;
  SelectiveChannelOptions sopt;
  sopt.timeout_ms = 1000;
  sopt.max_retry = 3;
  schan.Init("rr", &sopt);
  for (int port : {dead_port, live_port}) {
    Channel* sub = new Channel;
    ChannelOptions copt;
    copt.timeout_ms = 500;
    copt.max_retry = 0;
    sub->Init(("127.0.0.1:" + std::to_string(port)).c_str(), &copt);
    schan.AddChannel(sub);
  }
  Controller cntl;
  IOBuf request, response;
  request.append("sel");
  schan.CallMethod("EchoService.Echo", &cntl, &request, &response, nullptr);
  if (cntl.Failed()) return cntl.ErrorCode();
  if (resp_out != nullptr) *resp_out = response.to_string();
  return 0;
}

// PartitionChannel: "i/N addr" tagged naming.
int partition_test(const std::vector<int>& ports, std::string* merged) {
  std::string url = "list://";
  for (size_t i = 0; i < ports.size(); ++i) {
    if (i) url += ",";
    url += std::to_string(i) + "/" + std::to_string(ports.size()) + " 127.0.0.1:" +
           std::to_string(ports[i]);
  }
  PartitionChannel pc;
  PartitionChannelOptions opt;
  opt.parallel.timeout_ms = 3000;
  if (pc.Init((int)ports.size(), url.c_str(), &opt) != 0) return -1;
  Controller cntl;
  IOBuf request, response;
  request.append("P");
  pc.CallMethod("EchoService.Echo", &cntl, &request, &response, nullptr);
  if (cntl.Failed()) return cntl.ErrorCode();
  if (merged != nullptr) *merged = response.to_string();
  return 0;
}

// DynamicPartitionChannel: two co-existing schemes (ports2 = 2-partition
// group, ports3 = 3-partition group). Runs `calls` echoes; returns the
// number that were served by the 3-partition scheme (its merged response
// has 3 fragments vs 2), so the caller can assert the 3/5-2/5 capacity
// split. -1 on setup error, -2 on call failure.
int dynpart_test(const std::vector<int>& ports2, const std::vector<int>& ports3, int calls) {
  std::string url = "list://";
  for (size_t i = 0; i < ports2.size(); ++i) {
    url += std::to_string(i) + "/" + std::to_string(ports2.size()) + " 127.0.0.1:" +
           std::to_string(ports2[i]) + ",";
  }
  for (size_t i = 0; i < ports3.size(); ++i) {
    if (i) url += ",";
    url += std::to_string(i) + "/" + std::to_string(ports3.size()) + " 127.0.0.1:" +
           std::to_string(ports3[i]);
  }
  DynamicPartitionChannel dc;
  PartitionChannelOptions opt;
  opt.parallel.timeout_ms = 3000;
  if (dc.Init(url.c_str(), &opt) != 0) return -1;
  if (dc.scheme_count() != 2) return -1;
  int hits3 = 0;
  for (int i = 0; i < calls; ++i) {
    Controller cntl;
    IOBuf request, response;
    request.append("D");
    dc.CallMethod("EchoService.Echo", &cntl, &request, &response, nullptr);
    if (cntl.Failed()) return -2;
    if (response.size() == 3) ++hits3;  // each partition echoes 1 byte
  }
  return hits3;
}

// LB distribution check: run n calls via naming URL + lb; count distinct
// servers hit (server returns its port via a Port method).
int lb_spread_test(const std::string& lb_name, const std::vector<int>& ports, int ncalls) {
  std::string url = "list://";
  for (size_t i = 0; i < ports.size(); ++i) {
    if (i) url += ",";
    url += "127.0.0.1:" + std::to_string(ports[i]);
  }
  Channel chan;
  ChannelOptions copt;
  copt.timeout_ms = 2000;
  if (chan.Init(url.c_str(), lb_name.c_str(), &copt) != 0) return -1;
  std::set<std::string> seen;
  for (int i = 0; i < ncalls; ++i) {
    Controller cntl;
    IOBuf request, response;
    request.append("x");
    chan.CallMethod("EchoService.Port", &cntl, &request, &response, nullptr);
    if (cntl.Failed()) return -cntl.ErrorCode();
    seen.insert(response.to_string());
  }
  return (int)seen.size();
}

}  // namespace rpctest
}  // namespace bam

namespace bam {
namespace rpctest {

// Backup request: first attempt lands on a slow server; the backup fires
// after backup_ms and (via rr) reaches the fast server. Returns max
// latency over `calls` sync calls.
int64_t backup_request_test(int slow_port, int fast_port, int backup_ms, int calls) {
  std::string url = "list://127.0.0.1:" + std::to_string(slow_port) + ",127.0.0.1:" +
                    std::to_string(fast_port);
  Channel chan;
  ChannelOptions copt;
  copt.timeout_ms = 5000;
  copt.backup_request_ms = backup_ms;
  if (chan.Init(url.c_str(), "rr", &copt) != 0) return -1;
  int64_t max_lat = 0;
  for (int i = 0; i < calls; ++i) {
    Controller cntl;
    IOBuf request, response;
    request.append("1000");  // slow server sleeps 1000 ms
    chan.CallMethod("EchoService.Sleep", &cntl, &request, &response, nullptr);
    if (cntl.Failed()) return -cntl.ErrorCode();
    if (cntl.latency_us() > max_lat) max_lat = cntl.latency_us();
  }
  return max_lat;
}

}  // namespace rpctest
}  // namespace bam
