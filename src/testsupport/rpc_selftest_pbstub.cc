// Generated-stub integration test: a tools/bamproto.py-generated typed
// service (examples/gen/echo.bam.h) registered on a real Server and
// called through the typed Stub over a real Channel — the reference's
// protoc-stub workflow (brpc/channel.h:189-228, server.cpp:844-875)
// rebuilt on the self-contained proto runtime.
#include <atomic>

#include "base/time.h"
#include "fiber/sync.h"
#include "examples/gen/echo.bam.h"

namespace bam {
namespace rpctest {

namespace {

class TypedEchoImpl : public example::EchoServiceBase {
 public:
  void TypedEcho(Controller* cntl, const example::EchoRequest* request,
                 example::EchoResponse* response, Closure* done) override {
    response->message = "echo:" + request->message;
    response->seq = request->seq;
    response->stats["ntags"] = (int64_t)request->tags.size();
    int64_t sum = 0;
    for (int32_t t : request->tags) sum += t;
    response->stats["sum"] = sum;
    done->Run();
  }
};

}  // namespace

bool pb_stub_test(std::string* err) {
  Server* server = new Server;  // leaked: sockets may outlive the test
  auto* impl = new TypedEchoImpl;
  if (impl->RegisterTo(server) != 0) {
    *err = "RegisterTo failed";
    return false;
  }
  if (server->Start(0, nullptr) != 0) {
    *err = "server start failed";
    return false;
  }
  int port = server->listen_address().port;

  Channel channel;
  ChannelOptions copt;
  copt.timeout_ms = 5000;
  if (channel.Init(("127.0.0.1:" + std::to_string(port)).c_str(), &copt) != 0) {
    *err = "channel init failed";
    return false;
  }
  example::EchoService_Stub stub(&channel);

  // sync call
  {
    example::EchoRequest req;
    req.message = "typed-hello";
    req.seq = 42;
    req.tags = {1, 2, 3, 500000};
    example::EchoResponse resp;
    Controller cntl;
    stub.TypedEcho(&cntl, &req, &resp, nullptr);
    if (cntl.Failed()) {
      *err = "sync call failed: " + cntl.ErrorText();
      return false;
    }
    if (resp.message != "echo:typed-hello" || resp.seq != 42 ||
        resp.stats.count("ntags") == 0 || resp.stats["ntags"] != 4 ||
        resp.stats.count("sum") == 0 || resp.stats["sum"] != 500006) {
      *err = "sync response mismatch: message='" + resp.message +
             "' seq=" + std::to_string(resp.seq) +
             " stats.size=" + std::to_string(resp.stats.size());
      for (const auto& kv : resp.stats)
        *err += " [" + kv.first + "=" + std::to_string(kv.second) + "]";
      return false;
    }
  }
  // async call
  {
    auto* req = new example::EchoRequest;
    req->message = "async";
    req->seq = 7;
    auto* resp = new example::EchoResponse;
    auto* cntl = new Controller;
    cntl->set_timeout_ms(5000);
    CountdownEvent ev(1);
    bool ok = false;
    stub.TypedEcho(cntl, req, resp, NewCallback([&ev, &ok, cntl, resp, req] {
      ok = !cntl->Failed() && resp->message == "echo:async" && resp->seq == 7;
      ev.signal();
    }));
    ev.wait();
    delete req;
    delete resp;
    delete cntl;
    if (!ok) {
      *err = "async call failed";
      return false;
    }
  }
  // mcpack converters on the generated structs (protoc-gen-mcpack role)
  {
    example::EchoRequest req;
    req.message = "mc";
    req.seq = -5;
    req.tags = {7, 8};
    std::string mc;
    if (!req.SerializeAsMcpack(&mc)) {
      *err = "mcpack serialize failed";
      return false;
    }
    example::EchoRequest back;
    if (!back.ParseFromMcpack(mc) || back.message != "mc" || back.seq != -5 ||
        back.tags != std::vector<int32_t>({7, 8})) {
      *err = "mcpack round trip mismatch";
      return false;
    }
  }
  return true;
}

}  // namespace rpctest
}  // namespace bam
