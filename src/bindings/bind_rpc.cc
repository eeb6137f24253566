#include "rpc/load_balancer.h"
#include "rpc/mysql_client.h"
#include "rpc/flv.h"
#include "rpc/ts.h"
#include "rpc/rtmp_client.h"
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "bindings/bind.h"
#include "rpc/rdma_transport.h"

namespace bam {
namespace rpctest {
struct BenchResult {
  double qps;
  double mbps;
  int64_t p50_us, p90_us, p99_us, p999_us, max_us, avg_us;
  int64_t errors;
  int64_t total;
  std::string first_error;
};
int start_echo_server(int port);
int start_nshead_server();
int start_master_echo_server();
int start_rdma_mock_echo_server();
int start_idle_timeout_server(int idle_sec);
int64_t cancel_test(int port);
int retry_policy_test(int max_retry);
// naming resolution (rpc/load_balancer.h)
int start_session_counter_server();
int start_rtmp_server();
int start_mongo_echo_server();
// (mysql client bound directly below via rpc/mysql_client.h)
int start_shm_echo_server(const std::string& name);
int shm_call(const std::string& name, const std::string& method, const std::string& payload,
             std::string* response_out, std::string* err);
int shm_echo_bench(const std::string& name, int payload, int concurrency, int calls,
                   double* qps, int64_t* p99_us, int* errors_out);
int protocol_call(const std::string& addr, const std::string& protocol,
                  const std::string& method, const std::string& payload, int compress,
                  std::string* response_out, std::string* err);
int echo_once(const std::string& addr, const std::string& payload, int timeout_ms,
              std::string* response_out, int64_t* latency_us);
int call_method_once(const std::string& addr, const std::string& method,
                     const std::string& payload, int timeout_ms, int max_retry,
                     std::string* response_out, std::string* error_text);
bool attachment_test(const std::string& addr);
bool pb_stub_test(std::string* err);
bool channel_options_tail_test(std::string* err);
bool thread_local_data_test(std::string* err);
bool request_code_test(std::string* err);
bool short_connection_test(std::string* err);
bool http_header_ext_test(std::string* err);
BenchResult echo_bench(const std::string& addr, int payload_size, int concurrency,
                       int64_t total_calls, int timeout_ms, const std::string& method,
                       bool hbm_request, bool pooled, int nchannels,
                       const std::string& socket_mode = "");
BenchResult async_echo_bench(const std::string& addr, int payload_size, int pipeline,
                             int64_t total_calls, int timeout_ms, const std::string& method,
                             bool pooled);
}  // namespace rpctest
}  // namespace bam

void bind_rpc(py::module_& m) {
  auto r = m.def_submodule("rpc");
  r.def("rdma_live_recv_blocks", &bam::rdma::live_recv_blocks);
  // RTMP play session remuxed into a standard FLV document (parity:
  // reference rtmp.cpp FLV writer): collects up to ntags media messages.
  r.def("rtmp_play_to_flv",
        [](const std::string& host, int port, const std::string& app,
           const std::string& stream, int ntags, int timeout_ms) {
          std::string flv_doc;
          {
            py::gil_scoped_release rel;
            bam::RtmpClient c;
            if (c.Connect(host, port, app, timeout_ms) != 0)
              throw std::runtime_error("rtmp connect failed");
            if (c.Play(stream) != 0) throw std::runtime_error("rtmp play failed");
            bam::flv::AppendHeader(&flv_doc);
            for (int i = 0; i < ntags; ++i) {
              bam::rtmp::Message m;
              if (c.PollFrame(&m, timeout_ms) != 0) break;
              if (m.type == 8 || m.type == 9 || m.type == 18)
                bam::flv::AppendTag(&flv_doc, m.type, m.timestamp, m.payload);
              else
                --i;  // control message: not an FLV tag
            }
            c.Close();
          }
          return py::bytes(flv_doc);
        },
        py::arg("host"), py::arg("port"), py::arg("app"), py::arg("stream"),
        py::arg("ntags"), py::arg("timeout_ms") = 5000);
  // HLS leg (≙ reference brpc/ts.cpp): FLV tags -> MPEG-TS + m3u8.
  r.def("flv_to_ts", [](py::bytes flv_doc) {
    std::vector<bam::flv::Tag> tags;
    if (!bam::flv::Parse(std::string(flv_doc), &tags))
      throw std::runtime_error("bad FLV document");
    std::string out;
    if (!bam::ts::FlvToTs(tags, &out))
      throw std::runtime_error("no muxable a/v tags");
    return py::bytes(out);
  });
  r.def("ts_mux_tags", [](py::list tags) {
    bam::ts::TsMuxer mux;
    std::string out;
    mux.WriteTables(&out);
    for (auto item : tags) {
      py::tuple t = item.cast<py::tuple>();
      bam::flv::Tag tag;
      tag.type = t[0].cast<int>();
      tag.timestamp = t[1].cast<uint32_t>();
      tag.payload = t[2].cast<std::string>();
      mux.Write(tag, &out);
    }
    return py::bytes(out);
  });
  r.def("hls_playlist", [](py::list segs, int target_s, int seq, bool ended) {
    std::vector<bam::ts::HlsSegment> v;
    for (auto item : segs) {
      py::tuple t = item.cast<py::tuple>();
      v.push_back({t[0].cast<std::string>(), t[1].cast<double>()});
    }
    return bam::ts::MakeHlsPlaylist(v, target_s, seq, ended);
  }, py::arg("segments"), py::arg("target_duration_s") = 10,
     py::arg("media_sequence") = 0, py::arg("ended") = true);
  r.def("flv_parse", [](py::bytes doc) {
    std::vector<bam::flv::Tag> tags;
    bool ha = false, hv = false;
    if (!bam::flv::Parse(std::string(doc), &tags, &ha, &hv))
      throw std::runtime_error("malformed FLV");
    py::list out;
    for (auto& t : tags)
      out.append(py::make_tuple((int)t.type, t.timestamp, py::bytes(t.payload)));
    return py::make_tuple(ha, hv, out);
  });
  r.def("flv_build", [](py::list tags) {
    std::string doc;
    bam::flv::AppendHeader(&doc);
    for (auto t : tags) {
      py::tuple tt = t.cast<py::tuple>();
      bam::flv::AppendTag(&doc, (uint8_t)tt[0].cast<int>(), tt[1].cast<uint32_t>(),
                          tt[2].cast<std::string>());
    }
    return py::bytes(doc);
  });
  r.def("http_header_ext_test", []() {
    std::string err;
    bool ok;
    {
      py::gil_scoped_release rel;
      ok = bam::rpctest::http_header_ext_test(&err);
    }
    return py::make_tuple(ok, err);
  });
  r.def("short_connection_test", []() {
    std::string err;
    bool ok;
    {
      py::gil_scoped_release rel;
      ok = bam::rpctest::short_connection_test(&err);
    }
    return py::make_tuple(ok, err);
  });
  r.def("request_code_test", []() {
    std::string err;
    bool ok;
    {
      py::gil_scoped_release rel;
      ok = bam::rpctest::request_code_test(&err);
    }
    return py::make_tuple(ok, err);
  });
  r.def("thread_local_data_test", []() {
    std::string err;
    bool ok;
    {
      py::gil_scoped_release rel;
      ok = bam::rpctest::thread_local_data_test(&err);
    }
    return py::make_tuple(ok, err);
  });
  r.def("channel_options_tail_test", []() {
    std::string err;
    bool ok;
    {
      py::gil_scoped_release rel;
      ok = bam::rpctest::channel_options_tail_test(&err);
    }
    return py::make_tuple(ok, err);
  });
  r.def("pb_stub_test", []() {
    std::string err;
    bool ok;
    {
      py::gil_scoped_release rel;
      ok = bam::rpctest::pb_stub_test(&err);
    }
    return py::make_tuple(ok, err);
  });
  py::class_<bam::MysqlResult>(r, "MysqlResult")
      .def_readonly("ok", &bam::MysqlResult::ok)
      .def_readonly("affected_rows", &bam::MysqlResult::affected_rows)
      .def_readonly("last_insert_id", &bam::MysqlResult::last_insert_id)
      .def_readonly("error_code", &bam::MysqlResult::error_code)
      .def_readonly("error_message", &bam::MysqlResult::error_message)
      .def_readonly("columns", &bam::MysqlResult::columns)
      .def_readonly("rows", &bam::MysqlResult::rows);
  py::class_<bam::MysqlClient>(r, "MysqlClient")
      .def(py::init<>())
      .def("connect", &bam::MysqlClient::Connect, py::arg("host"), py::arg("port"),
           py::arg("user"), py::arg("password"), py::arg("db") = "",
           py::arg("timeout_ms") = 3000, py::call_guard<py::gil_scoped_release>())
      .def("query",
           [](bam::MysqlClient& c, const std::string& sql) {
             bam::MysqlResult res;
             {
               py::gil_scoped_release rel;
               c.Query(sql, &res);
             }
             return res;
           })
      .def("ping", &bam::MysqlClient::Ping, py::call_guard<py::gil_scoped_release>())
      .def("prepare",
           [](bam::MysqlClient& c, const std::string& sql) {
             int nparams = 0;
             int64_t sid;
             {
               py::gil_scoped_release rel;
               sid = c.Prepare(sql, &nparams);
             }
             return py::make_tuple(sid, nparams);
           })
      .def("execute_prepared",
           [](bam::MysqlClient& c, int64_t sid, const std::vector<std::string>& params) {
             bam::MysqlResult res;
             {
               py::gil_scoped_release rel;
               c.ExecutePrepared(sid, params, &res);
             }
             return res;
           })
      .def("close_statement", &bam::MysqlClient::CloseStatement,
           py::call_guard<py::gil_scoped_release>())
      .def("close", &bam::MysqlClient::Close)
      .def("connected", &bam::MysqlClient::connected)
      .def("server_version", &bam::MysqlClient::server_version);
  py::class_<bam::RtmpClient>(r, "RtmpClient")
      .def(py::init<>())
      .def("connect", &bam::RtmpClient::Connect, py::arg("host"), py::arg("port"),
           py::arg("app"), py::arg("timeout_ms") = 3000,
           py::call_guard<py::gil_scoped_release>())
      .def("publish", &bam::RtmpClient::Publish, py::call_guard<py::gil_scoped_release>())
      .def("play", &bam::RtmpClient::Play, py::call_guard<py::gil_scoped_release>())
      .def("push_frame",
           [](bam::RtmpClient& c, int type, uint32_t ts, py::bytes payload) {
             std::string p = payload.cast<std::string>();
             py::gil_scoped_release rel;
             return c.PushFrame((uint8_t)type, ts, p);
           })
      .def("poll_frame",
           [](bam::RtmpClient& c, int timeout_ms) -> py::object {
             bam::rtmp::Message m;
             int rc;
             {
               py::gil_scoped_release rel;
               rc = c.PollFrame(&m, timeout_ms);
             }
             if (rc != 0) return py::none();
             return py::make_tuple((int)m.type, m.timestamp, py::bytes(m.payload));
           },
           py::arg("timeout_ms") = 3000)
      .def("close", &bam::RtmpClient::Close);
  r.def("start_echo_server", &bam::rpctest::start_echo_server, py::arg("port") = 0,
        py::call_guard<py::gil_scoped_release>());
  r.def("start_idle_timeout_server", &bam::rpctest::start_idle_timeout_server,
        py::call_guard<py::gil_scoped_release>());
  r.def("start_rdma_mock_server", &bam::rpctest::start_rdma_mock_echo_server,
        py::call_guard<py::gil_scoped_release>());
  r.def("start_master_echo_server", &bam::rpctest::start_master_echo_server,
        py::call_guard<py::gil_scoped_release>());
  r.def("start_nshead_server", &bam::rpctest::start_nshead_server,
        py::call_guard<py::gil_scoped_release>());
  r.def("cancel_test", &bam::rpctest::cancel_test,
        py::call_guard<py::gil_scoped_release>());
  r.def("resolve_naming", [](const std::string& url) {
    std::vector<std::string> out;
    std::vector<bam::EndPoint> eps;
    {
      py::gil_scoped_release rel;
      if (bam::ResolveNamingUrl(url, &eps) != 0) return out;
    }
    for (const auto& ep : eps) out.push_back(bam::endpoint2str(ep));
    return out;
  });
  r.def("retry_policy_test", &bam::rpctest::retry_policy_test,
        py::call_guard<py::gil_scoped_release>());
  r.def("start_session_counter_server", &bam::rpctest::start_session_counter_server,
        py::call_guard<py::gil_scoped_release>());
  r.def("start_rtmp_server", &bam::rpctest::start_rtmp_server,
        py::call_guard<py::gil_scoped_release>());
  r.def("start_mongo_server", &bam::rpctest::start_mongo_echo_server,
        py::call_guard<py::gil_scoped_release>());
  r.def("start_shm_server", &bam::rpctest::start_shm_echo_server,
        py::call_guard<py::gil_scoped_release>());
  r.def("shm_call",
        [](const std::string& name, const std::string& method, const std::string& payload) {
          std::string resp, err;
          int rc;
          {
            py::gil_scoped_release rel;
            rc = bam::rpctest::shm_call(name, method, payload, &resp, &err);
          }
          return py::make_tuple(rc, py::bytes(resp), err);
        });
  r.def("shm_echo_bench",
        [](const std::string& name, int payload, int concurrency, int calls) {
          double qps = 0;
          int64_t p99 = 0;
          int errors = 0;
          int rc;
          {
            py::gil_scoped_release rel;
            rc = bam::rpctest::shm_echo_bench(name, payload, concurrency, calls, &qps, &p99,
                                              &errors);
          }
          py::dict d;
          d["rc"] = rc;
          d["qps"] = qps;
          d["p99_us"] = p99;
          d["errors"] = errors;
          return d;
        });
  r.def("protocol_call",
        [](const std::string& addr, const std::string& protocol, const std::string& method,
           const std::string& payload, int compress) {
          std::string resp, err;
          int rc;
          {
            py::gil_scoped_release rel;
            rc = bam::rpctest::protocol_call(addr, protocol, method, payload, compress, &resp,
                                             &err);
          }
          return py::make_tuple(rc, py::bytes(resp), err);
        },
        py::arg("addr"), py::arg("protocol"), py::arg("method"), py::arg("payload"),
        py::arg("compress") = 0);
  r.def("echo_once",
        [](const std::string& addr, py::bytes payload, int timeout_ms) {
          char* ptr;
          Py_ssize_t len;
          PyBytes_AsStringAndSize(payload.ptr(), &ptr, &len);
          std::string p(ptr, len), resp;
          int64_t lat = 0;
          int rc;
          {
            py::gil_scoped_release rel;
            rc = bam::rpctest::echo_once(addr, p, timeout_ms, &resp, &lat);
          }
          return py::make_tuple(rc, py::bytes(resp), lat);
        },
        py::arg("addr"), py::arg("payload"), py::arg("timeout_ms") = 1000);
  r.def("call_method_once",
        [](const std::string& addr, const std::string& method, py::bytes payload,
           int timeout_ms, int max_retry) {
          char* ptr;
          Py_ssize_t len;
          PyBytes_AsStringAndSize(payload.ptr(), &ptr, &len);
          std::string p(ptr, len), resp, err;
          int rc;
          {
            py::gil_scoped_release rel;
            rc = bam::rpctest::call_method_once(addr, method, p, timeout_ms, max_retry,
                                                &resp, &err);
          }
          return py::make_tuple(rc, py::bytes(resp), err);
        },
        py::arg("addr"), py::arg("method"), py::arg("payload"), py::arg("timeout_ms") = 1000,
        py::arg("max_retry") = 3);
  r.def("attachment_test", &bam::rpctest::attachment_test,
        py::call_guard<py::gil_scoped_release>());
  r.def("echo_bench",
        [](const std::string& addr, int payload_size, int concurrency, int64_t total,
           int timeout_ms, const std::string& method, bool hbm_request, bool pooled,
           int nchannels) {
          bam::rpctest::BenchResult b;
          {
            py::gil_scoped_release rel;
            b = bam::rpctest::echo_bench(addr, payload_size, concurrency, total, timeout_ms,
                                         method, hbm_request, pooled, nchannels);
          }
          py::dict d;
          d["qps"] = b.qps;
          d["mbps"] = b.mbps;
          d["p50_us"] = b.p50_us;
          d["p90_us"] = b.p90_us;
          d["p99_us"] = b.p99_us;
          d["p999_us"] = b.p999_us;
          d["max_us"] = b.max_us;
          d["avg_us"] = b.avg_us;
          d["errors"] = b.errors;
          d["first_error"] = b.first_error;
          d["total"] = b.total;
          return d;
        },
        py::arg("addr"), py::arg("payload_size") = 64, py::arg("concurrency") = 8,
        py::arg("total") = 10000, py::arg("timeout_ms") = 5000,
        py::arg("method") = "EchoService.Echo", py::arg("hbm_request") = false,
        py::arg("pooled") = false, py::arg("nchannels") = 1);
  r.def("echo_bench_mode",
        [](const std::string& addr, int payload_size, int concurrency, int64_t total,
           int timeout_ms, const std::string& method, const std::string& socket_mode) {
          bam::rpctest::BenchResult b;
          {
            py::gil_scoped_release rel;
            b = bam::rpctest::echo_bench(addr, payload_size, concurrency, total,
                                         timeout_ms, method, false, false, 1,
                                         socket_mode);
          }
          py::dict d;
          d["qps"] = b.qps;
          d["p99_us"] = b.p99_us;
          d["errors"] = b.errors;
          d["first_error"] = b.first_error;
          d["total"] = b.total;
          return d;
        },
        py::arg("addr"), py::arg("payload_size"), py::arg("concurrency"),
        py::arg("total"), py::arg("timeout_ms"), py::arg("method"),
        py::arg("socket_mode"));
  r.def("async_echo_bench",
        [](const std::string& addr, int payload_size, int pipeline, int64_t total,
           int timeout_ms, const std::string& method, bool pooled) {
          bam::rpctest::BenchResult b;
          {
            py::gil_scoped_release rel;
            b = bam::rpctest::async_echo_bench(addr, payload_size, pipeline, total,
                                               timeout_ms, method, pooled);
          }
          py::dict d;
          d["qps"] = b.qps;
          d["mbps"] = b.mbps;
          d["p50_us"] = b.p50_us;
          d["p99_us"] = b.p99_us;
          d["p999_us"] = b.p999_us;
          d["avg_us"] = b.avg_us;
          d["errors"] = b.errors;
          d["first_error"] = b.first_error;
          d["total"] = b.total;
          return d;
        },
        py::arg("addr"), py::arg("payload_size") = 64, py::arg("pipeline") = 64,
        py::arg("total") = 10000, py::arg("timeout_ms") = 5000,
        py::arg("method") = "EchoService.Echo", py::arg("pooled") = false);
}

// ---- combo channels & LBs ----
namespace bam {
namespace rpctest {
int parallel_echo_test(const std::vector<int>& ports, const std::string& payload,
                       int fail_limit, std::string* merged, std::string* err);
int selective_test(int dead_port, int live_port, std::string* resp_out);
int partition_test(const std::vector<int>& ports, std::string* merged);
int dynpart_test(const std::vector<int>& ports2, const std::vector<int>& ports3, int calls);
int lb_spread_test(const std::string& lb_name, const std::vector<int>& ports, int ncalls);
int compressed_echo_test(const std::string& addr, const std::string& payload,
                         int compress_type, std::string* response_out);
int64_t backup_request_test(int slow_port, int fast_port, int backup_ms, int calls);
int start_intercepted_echo_server(const std::string& magic_logid);
int call_with_logid(const std::string& addr, uint64_t log_id, std::string* err);
int grpc_client_call(const std::string& addr, const std::string& full_method,
                     const std::string& payload, int timeout_ms, std::string* response_out,
                     std::string* err);
}  // namespace rpctest
}  // namespace bam

void bind_rpc_combo(py::module_& m) {
  auto r = m.def_submodule("combo");
  r.def("parallel_echo",
        [](const std::vector<int>& ports, py::bytes payload, int fail_limit) {
          char* ptr;
          Py_ssize_t len;
          PyBytes_AsStringAndSize(payload.ptr(), &ptr, &len);
          std::string p(ptr, len), merged, err;
          int rc;
          {
            py::gil_scoped_release rel;
            rc = bam::rpctest::parallel_echo_test(ports, p, fail_limit, &merged, &err);
          }
          return py::make_tuple(rc, py::bytes(merged), err);
        },
        py::arg("ports"), py::arg("payload"), py::arg("fail_limit") = -1);
  r.def("selective",
        [](int dead_port, int live_port) {
          std::string resp;
          int rc;
          {
            py::gil_scoped_release rel;
            rc = bam::rpctest::selective_test(dead_port, live_port, &resp);
          }
          return py::make_tuple(rc, py::bytes(resp));
        });
  r.def("partition",
        [](const std::vector<int>& ports) {
          std::string merged;
          int rc;
          {
            py::gil_scoped_release rel;
            rc = bam::rpctest::partition_test(ports, &merged);
          }
          return py::make_tuple(rc, py::bytes(merged));
        });
  r.def("dynamic_partition", &bam::rpctest::dynpart_test,
        py::call_guard<py::gil_scoped_release>());
  r.def("lb_spread", &bam::rpctest::lb_spread_test,
        py::call_guard<py::gil_scoped_release>());
  r.def("backup_request", &bam::rpctest::backup_request_test,
        py::call_guard<py::gil_scoped_release>());
  r.def("start_intercepted_server", &bam::rpctest::start_intercepted_echo_server,
        py::call_guard<py::gil_scoped_release>());
  r.def("grpc_call", [](const std::string& addr, const std::string& method,
                        const std::string& payload, int timeout_ms) {
    std::string resp, err;
    int rc;
    {
      py::gil_scoped_release rel;
      rc = bam::rpctest::grpc_client_call(addr, method, payload, timeout_ms, &resp, &err);
    }
    return py::make_tuple(rc, py::bytes(resp), err);
  }, py::arg("addr"), py::arg("method"), py::arg("payload"), py::arg("timeout_ms") = 3000);
  r.def("call_with_logid", [](const std::string& addr, uint64_t log_id) {
    std::string err;
    int rc;
    {
      py::gil_scoped_release rel;
      rc = bam::rpctest::call_with_logid(addr, log_id, &err);
    }
    return py::make_tuple(rc, err);
  });
  r.def("compressed_echo",
        [](const std::string& addr, py::bytes payload, int ctype) {
          char* ptr;
          Py_ssize_t len;
          PyBytes_AsStringAndSize(payload.ptr(), &ptr, &len);
          std::string p(ptr, len), resp;
          int rc;
          {
            py::gil_scoped_release rel;
            rc = bam::rpctest::compressed_echo_test(addr, p, ctype, &resp);
          }
          return py::make_tuple(rc, py::bytes(resp));
        });
}

// ---- streaming ----
namespace bam {
namespace rpctest {
int start_stream_echo_server();
int stream_echo_test(int port, int nframes, int frame_size, std::string* err);
double stream_throughput_test(int port, int nframes, int frame_size);
double stream_throughput_hbm_test(int port, int nframes, int frame_size, bool sink_to_hbm);
}  // namespace rpctest
}  // namespace bam

void bind_rpc_stream(py::module_& m) {
  auto s = m.def_submodule("stream");
  s.def("start_server", &bam::rpctest::start_stream_echo_server,
        py::call_guard<py::gil_scoped_release>());
  s.def("echo_test",
        [](int port, int nframes, int frame_size) {
          std::string err;
          int rc;
          {
            py::gil_scoped_release rel;
            rc = bam::rpctest::stream_echo_test(port, nframes, frame_size, &err);
          }
          return py::make_tuple(rc, err);
        });
  s.def("throughput_hbm", &bam::rpctest::stream_throughput_hbm_test,
        py::call_guard<py::gil_scoped_release>());
  s.def("throughput", &bam::rpctest::stream_throughput_test,
        py::call_guard<py::gil_scoped_release>());
}
