// Streaming RPC test scenarios + throughput bench (BASELINE config 3's
// host-path harness; the GPU/xGMI variant lives in bench tooling).
// Models reference test/brpc_streaming_rpc_unittest.cpp.
#include <atomic>
#include <string>

#include "base/fast_rand.h"
#include "base/iobuf.h"
#include "base/time.h"
#include "fiber/sync.h"
#include "rpc/channel.h"
#include "rpc/controller.h"
#include "rpc/server.h"
#include "rpc/stream.h"

namespace bam {
namespace rpctest {

// Server with a stream-accepting method: echoes every frame back.
int start_stream_echo_server() {
  Server* server = new Server;
  Service* svc = new Service("StreamService");
  svc->AddMethod("Open", [](Controller* cntl, const IOBuf& req, IOBuf* resp, Closure* done) {
    StreamOptions sopt;
    sopt.max_buf_size = 4u << 20;
    // echo frames back on the same stream
    sopt.on_received = [](StreamId sid, IOBuf* msg) {
      IOBuf copy = *msg;
      StreamWrite(sid, &copy);
    };
    sopt.on_closed = [](StreamId sid) { StreamClose(sid); };
    StreamId sid;
    if (StreamAccept(&sid, cntl, sopt) != 0) {
      cntl->SetFailed(EREQUEST, "no stream in request");
    }
    resp->append("accepted");
    done->Run();
  });
  // sink that re-uploads every frame into HBM (GPU staging, both legs)
  svc->AddMethod("OpenSinkHbm", [](Controller* cntl, const IOBuf& req, IOBuf* resp, Closure* done) {
    StreamOptions sopt;
    sopt.max_buf_size = 64u << 20;
    sopt.on_received = [](StreamId, IOBuf* msg) {
      if (!has_block_allocator(RES_HBM)) return;
      std::string bytes = msg->to_string();
      IOBuf dev;
      dev.append_with_residency(bytes.data(), bytes.size(), RES_HBM, 0, 0);
    };
    sopt.on_closed = [](StreamId sid) { StreamClose(sid); };
    StreamId sid;
    if (StreamAccept(&sid, cntl, sopt) != 0) cntl->SetFailed(EREQUEST, "no stream");
    resp->append("ok");
    done->Run();
  });
  // sink: counts bytes, no echo (throughput direction test)
  svc->AddMethod("OpenSink", [](Controller* cntl, const IOBuf& req, IOBuf* resp, Closure* done) {
    StreamOptions sopt;
    sopt.max_buf_size = 8u << 20;
    sopt.on_received = [](StreamId, IOBuf*) {};
    sopt.on_closed = [](StreamId sid) { StreamClose(sid); };
    StreamId sid;
    if (StreamAccept(&sid, cntl, sopt) != 0) cntl->SetFailed(EREQUEST, "no stream");
    resp->append("ok");
    done->Run();
  });
  server->AddService(svc, SERVER_OWNS_SERVICE);
  if (server->Start(0, nullptr) != 0) return -1;
  return server->listen_address().port;
}

// Round trip: write nframes of frame_size; expect all echoed back.
int stream_echo_test(int port, int nframes, int frame_size, std::string* err) {
  Channel channel;
  ChannelOptions copt;
  copt.timeout_ms = 5000;
  if (channel.Init(("127.0.0.1:" + std::to_string(port)).c_str(), &copt) != 0) return -1;

  std::atomic<int64_t> received_bytes{0};
  std::atomic<int> received_frames{0};
  CountdownEvent all_received(1);
  const int64_t expect_bytes = (int64_t)nframes * frame_size;

  StreamOptions sopt;
  sopt.max_buf_size = 4u << 20;
  sopt.on_received = [&](StreamId, IOBuf* msg) {
    received_bytes.fetch_add((int64_t)msg->size(), std::memory_order_relaxed);
    if (received_frames.fetch_add(1, std::memory_order_relaxed) + 1 == nframes) {
      all_received.signal();
    }
  };

  Controller cntl;
  StreamId sid;
  if (StreamCreate(&sid, &cntl, sopt) != 0) return -2;
  IOBuf request, response;
  request.append("open");
  channel.CallMethod("StreamService.Open", &cntl, &request, &response, nullptr);
  if (cntl.Failed()) {
    *err = cntl.ErrorText();
    StreamClose(sid);
    return cntl.ErrorCode();
  }
  std::string frame(frame_size, 'S');
  for (int i = 0; i < nframes; ++i) {
    IOBuf data;
    data.append(frame);
    int rc = StreamWrite(sid, &data);
    if (rc != 0) {
      *err = "StreamWrite failed rc=" + std::to_string(rc);
      StreamClose(sid);
      return rc;
    }
  }
  if (!all_received.timed_wait(monotonic_time_us() + 10 * 1000000)) {
    *err = "timed out: got " + std::to_string(received_bytes.load()) + "/" +
           std::to_string(expect_bytes) + " bytes";
    StreamClose(sid);
    return ETIMEDOUT;
  }
  StreamClose(sid);
  return received_bytes.load() == expect_bytes ? 0 : -3;
}

// One-direction throughput: returns MB/s.
double stream_throughput_test(int port, int nframes, int frame_size) {
  Channel channel;
  ChannelOptions copt;
  copt.timeout_ms = 5000;
  if (channel.Init(("127.0.0.1:" + std::to_string(port)).c_str(), &copt) != 0) return -1;
  StreamOptions sopt;
  sopt.max_buf_size = 8u << 20;
  Controller cntl;
  StreamId sid;
  StreamCreate(&sid, &cntl, sopt);
  IOBuf request, response;
  request.append("open");
  channel.CallMethod("StreamService.OpenSink", &cntl, &request, &response, nullptr);
  if (cntl.Failed()) {
    StreamClose(sid);
    return -2;
  }
  std::string frame(frame_size, 'T');
  int64_t t0 = monotonic_time_us();
  for (int i = 0; i < nframes; ++i) {
    IOBuf data;
    data.append(frame);
    if (StreamWrite(sid, &data) != 0) {
      StreamClose(sid);
      return -3;
    }
  }
  int64_t elapsed = monotonic_time_us() - t0;
  StreamClose(sid);
  return (double)nframes * frame_size / (double)elapsed;  // bytes/us == MB/s
}

// BASELINE config 3 (single-GPU analogue): streaming with 1 MB frames that
// LIVE IN HBM — every frame is staged into HBM-resident IOBuf blocks and
// the socket write path stages it back out through the gfx950 gather leg
// (hip/gpu_runtime.hip). sink_to_hbm additionally re-uploads every
// received frame into HBM on the server (both staging directions).
// Returns MB/s of application payload, or <0 on error.
double stream_throughput_hbm_test(int port, int nframes, int frame_size, bool sink_to_hbm) {
  if (!has_block_allocator(RES_HBM)) return -10;  // no GPU runtime loaded
  Channel channel;
  ChannelOptions copt;
  copt.timeout_ms = 20000;
  if (channel.Init(("127.0.0.1:" + std::to_string(port)).c_str(), &copt) != 0) return -1;
  StreamOptions sopt;
  sopt.max_buf_size = 64u << 20;
  Controller cntl;
  StreamId sid;
  StreamCreate(&sid, &cntl, sopt);
  IOBuf request, response;
  request.append("open");
  channel.CallMethod(sink_to_hbm ? "StreamService.OpenSinkHbm" : "StreamService.OpenSink",
                     &cntl, &request, &response, nullptr);
  if (cntl.Failed()) {
    StreamClose(sid);
    return -2;
  }
  std::string frame((size_t)frame_size, 'H');
  int64_t t0 = monotonic_time_us();
  for (int i = 0; i < nframes; ++i) {
    IOBuf data;
    if (data.append_with_residency(frame.data(), frame.size(), RES_HBM, 0, 0) != 0) {
      StreamClose(sid);
      return -4;
    }
    int rc = StreamWrite(sid, &data);
    if (rc != 0) {
      StreamClose(sid);
      return -3;
    }
  }
  int64_t elapsed = monotonic_time_us() - t0;
  StreamClose(sid);
  return (double)nframes * frame_size / (double)elapsed;  // bytes/us == MB/s
}

}  // namespace rpctest
}  // namespace bam
