// Public Python API: Server with Python-defined services (handlers run on
// the usercode pthread pool with the GIL — never on fiber workers) and a
// Channel for client calls. This is the user-facing surface mirroring the
// reference's Channel/Server/Controller API in Python.
#include <pybind11/functional.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <memory>

#include "rpc/redis_cluster.h"
#include "rpc/couchbase.h"
#include "bindings/bind.h"
#include "rpc/authenticator.h"
#include "rpc/ssl_util.h"
#include "rpc/channel.h"
#include "rpc/controller.h"
#include "rpc/parallel_channel.h"
#include "rpc/server.h"
#include "fiber/fiber.h"
#include "rpc/policy/http_protocol.h"
#include "rpc/usercode_pool.h"

namespace {

using namespace bam;

struct PyRpcError : public std::runtime_error {
  int code;
  PyRpcError(int c, const std::string& msg) : std::runtime_error(msg), code(c) {}
};

// Wraps a Python callable as a MethodFn: dispatched to the usercode pool,
// runs under the GIL there, then done->Run() sends the response.
MethodFn wrap_py_handler(py::function fn) {
  // keep the callable alive via shared_ptr captured in the lambda
  auto holder = std::make_shared<py::object>(std::move(fn));
  return [holder](Controller* cntl, const IOBuf& request, IOBuf* response, Closure* done) {
    std::string req_bytes = request.to_string();
    std::string att_bytes = cntl->request_attachment().to_string();
    SubmitUserCode([holder, cntl, req_bytes, att_bytes, response, done] {
      py::gil_scoped_acquire gil;
      try {
        py::object result =
            (*holder)(py::bytes(req_bytes), py::bytes(att_bytes));
        if (py::isinstance<py::tuple>(result)) {
          py::tuple t = result.cast<py::tuple>();
          std::string body = t[0].cast<std::string>();
          std::string att = t[1].cast<std::string>();
          response->append(body);
          cntl->response_attachment().append(att);
        } else if (!result.is_none()) {
          std::string body = result.cast<std::string>();
          response->append(body);
        }
      } catch (const std::exception& e) {
        cntl->SetFailed(EINTERNAL, std::string("python handler: ") + e.what());
      }
      {
        py::gil_scoped_release rel;
        done->Run();
      }
    });
  };
}

class PyServer {
 public:
  PyServer() : server_(new Server) {}

  void set_method_max_concurrency(const std::string& full_method, int limit) {
    server_->SetMethodMaxConcurrency(full_method, limit);
  }

  void add_restful_mapping(const std::string& service, const std::string& mappings) {
    auto it = services_.find(service);
    if (it == services_.end()) throw std::runtime_error("unknown service " + service);
    // re-register the mappings on the existing service object
    if (server_->AddServiceRestfulOnly(it->second, mappings) != 0)
      throw std::runtime_error("bad restful mapping: " + mappings);
  }

  void add_method(const std::string& service, const std::string& method, py::function fn) {
    Service* svc;
    auto it = services_.find(service);
    if (it == services_.end()) {
      svc = new Service(service);
      services_[service] = svc;
      server_->AddService(svc, SERVER_OWNS_SERVICE);
    } else {
      svc = it->second;
    }
    svc->AddMethod(method, wrap_py_handler(std::move(fn)));
  }

  int start(int port, int max_concurrency, const std::string& auth_user,
            const std::string& auth_password, const std::string& ssl_cert,
            const std::string& ssl_key, const std::string& adaptive_max_concurrency,
            const std::string& socket_mode) {
    ServerOptions opts;
    opts.socket_mode = socket_mode;
    opts.max_concurrency = max_concurrency;
    opts.adaptive_max_concurrency = adaptive_max_concurrency;
    if (!auth_user.empty()) {
      auth_.reset(new PasswordAuthenticator(auth_user, auth_password));
      opts.auth = auth_.get();
    }
    opts.ssl_cert = ssl_cert;
    opts.ssl_key = ssl_key;
    if (server_->Start(port, &opts) != 0) throw std::runtime_error("Server.start failed");
    return server_->listen_address().port;
  }

  void stop() { server_->Stop(0); }
  bool running() const { return server_->IsRunning(); }
  int port() const { return server_->listen_address().port; }
  int64_t processed() const { return server_->nprocessed.load(); }

 private:
  Server* server_;  // leaked deliberately: sockets may still reference it
  std::map<std::string, Service*> services_;
  std::unique_ptr<PasswordAuthenticator> auth_;
};

class PyChannel {
 public:
  PyChannel(const std::string& addr, const std::string& lb, int timeout_ms, int max_retry,
            int backup_request_ms, int compress, const std::string& auth_user,
            const std::string& auth_password, bool ssl, const std::string& socket_mode,
            const std::string& protocol) {
    ChannelOptions opts;
    opts.ssl = ssl;
    opts.socket_mode = socket_mode;
    if (!protocol.empty()) opts.protocol = protocol;
    opts.timeout_ms = timeout_ms;
    opts.max_retry = max_retry;
    opts.backup_request_ms = backup_request_ms;
    if (!auth_user.empty()) {
      auth_.reset(new PasswordAuthenticator(auth_user, auth_password));
      opts.auth = auth_.get();
    }
    compress_ = (CompressType)compress;
    int rc = lb.empty() ? channel_.Init(addr.c_str(), &opts)
                        : channel_.Init(addr.c_str(), lb.c_str(), &opts);
    if (rc != 0) throw std::runtime_error("Channel.Init failed for " + addr);
  }

  py::tuple call(const std::string& full_method, const std::string& request,
                 const std::string& attachment, int timeout_ms, uint64_t log_id,
                 uint64_t request_code, bool has_code) {
    Controller cntl;
    if (timeout_ms > 0) cntl.set_timeout_ms(timeout_ms);
    if (log_id != 0) cntl.set_log_id(log_id);
    if (has_code) cntl.set_request_code(request_code);
    cntl.set_request_compress_type(compress_);
    IOBuf req, resp;
    req.append(request);
    if (!attachment.empty()) cntl.request_attachment().append(attachment);
    {
      py::gil_scoped_release rel;
      channel_.CallMethod(full_method, &cntl, &req, &resp, nullptr);
    }
    if (cntl.Failed()) throw PyRpcError(cntl.ErrorCode(), cntl.ErrorText());
    return py::make_tuple(py::bytes(resp.to_string()),
                          py::bytes(cntl.response_attachment().to_string()),
                          cntl.latency_us());
  }

  // HTTP-flavored call: custom verb + headers in, (status, headers, body)
  // out — the python face of Controller::http_request()/http_response().
  py::tuple http_call(const std::string& path, py::bytes body,
                      const std::string& verb, py::dict headers, int timeout_ms) {
    Controller cntl;
    if (timeout_ms > 0) cntl.set_timeout_ms(timeout_ms);
    if (!verb.empty()) cntl.http_request().method = verb;
    for (auto item : headers) {
      cntl.http_request().SetHeader(py::cast<std::string>(item.first),
                                    py::cast<std::string>(item.second));
    }
    IOBuf req, resp;
    req.append(std::string(body));
    int ec = 0;
    std::string etext;
    {
      py::gil_scoped_release rel;
      channel_.CallMethod(path, &cntl, &req, &resp, nullptr);
      if (cntl.Failed()) {
        ec = cntl.ErrorCode();
        etext = cntl.ErrorText();
      }
    }
    py::dict rh;
    int status = 0;
    if (cntl.has_http_response()) {
      const HttpHeaderExt& hr = cntl.http_response();
      status = hr.status_code;
      for (const auto& kv : hr.headers) rh[py::str(kv.first)] = kv.second;
    }
    if (ec != 0 && status == 0) throw PyRpcError(ec, etext);
    // on HTTP error statuses the page body rides the attachment
    std::string body_out = resp.empty() && status / 100 != 2
                               ? cntl.response_attachment().to_string()
                               : resp.to_string();
    return py::make_tuple(status, rh, py::bytes(body_out));
  }

 private:
  Channel channel_;
  std::unique_ptr<PasswordAuthenticator> auth_;  // outlives channel_ (declared first... kept until PyChannel dies)
  CompressType compress_ = COMPRESS_TYPE_NONE;
};

}  // namespace

void bind_api(py::module_& m) {
  static py::exception<PyRpcError> exc(m, "RpcError");
  py::register_exception_translator([](std::exception_ptr p) {
    try {
      if (p) std::rethrow_exception(p);
    } catch (const PyRpcError& e) {
      // RpcError(code, message): args = (code, message)
      PyErr_SetObject(exc.ptr(), py::make_tuple(e.code, e.what()).ptr());
    }
  });

  py::class_<PyServer>(m, "Server")
      .def(py::init<>())
      .def("add_method", &PyServer::add_method, py::arg("service"), py::arg("method"),
           py::arg("handler"))
      .def("set_method_max_concurrency", &PyServer::set_method_max_concurrency)
      .def("add_restful_mapping", &PyServer::add_restful_mapping)
      .def("start", &PyServer::start, py::arg("port") = 0, py::arg("max_concurrency") = 0,
           py::arg("auth_user") = "", py::arg("auth_password") = "",
           py::arg("ssl_cert") = "", py::arg("ssl_key") = "",
           py::arg("adaptive_max_concurrency") = "", py::arg("socket_mode") = "")
      .def("stop", &PyServer::stop)
      .def("running", &PyServer::running)
      .def("port", &PyServer::port)
      .def("processed", &PyServer::processed);

  py::class_<PyChannel>(m, "Channel")
      .def(py::init<const std::string&, const std::string&, int, int, int, int,
                    const std::string&, const std::string&, bool, const std::string&,
                    const std::string&>(),
           py::arg("addr"), py::arg("lb") = "", py::arg("timeout_ms") = 500,
           py::arg("max_retry") = 3, py::arg("backup_request_ms") = -1,
           py::arg("compress") = 0, py::arg("auth_user") = "", py::arg("auth_password") = "",
           py::arg("ssl") = false, py::arg("socket_mode") = "",
           py::arg("protocol") = "")
      .def("call", &PyChannel::call, py::arg("method"), py::arg("request"),
           py::arg("attachment") = std::string(), py::arg("timeout_ms") = 0,
           py::arg("log_id") = 0, py::arg("request_code") = 0,
           py::arg("has_request_code") = false)
      .def("http_call", &PyChannel::http_call, py::arg("path"), py::arg("body") = py::bytes(),
           py::arg("method") = "", py::arg("headers") = py::dict(),
           py::arg("timeout_ms") = 0);

  m.def("http_call_progressive",
        [](const std::string& addr, const std::string& method, py::bytes request,
           int timeout_ms) {
          bam::ChannelOptions opts;
          opts.protocol = "http";
          opts.timeout_ms = timeout_ms;
          opts.max_retry = 0;
          bam::Channel ch;
          if (ch.Init(addr.c_str(), &opts) != 0)
            throw std::runtime_error("channel init failed");
          bam::Controller cntl;
          cntl.set_timeout_ms(timeout_ms);
          std::vector<std::string> chunks;
          bool saw_done = false;
          cntl.response_read_progressively(
              [&chunks, &saw_done](const bam::IOBuf& c, bool done) {
                if (!c.empty()) chunks.push_back(c.to_string());
                if (done) saw_done = true;
              });
          bam::IOBuf req, resp;
          req.append(std::string(request));
          {
            py::gil_scoped_release rel;
            ch.CallMethod(method, &cntl, &req, &resp, nullptr);
          }
          if (cntl.Failed()) throw PyRpcError(cntl.ErrorCode(), cntl.ErrorText());
          py::list out;
          for (auto& c : chunks) out.append(py::bytes(c));
          return py::make_tuple(out, saw_done, py::bytes(resp.to_string()));
        },
        py::arg("addr"), py::arg("method"), py::arg("request") = py::bytes(""),
        py::arg("timeout_ms") = 5000);

  py::class_<bam::CouchbaseClient>(m, "CouchbaseClient")
      .def(py::init<>())
      .def("init",
           [](bam::CouchbaseClient& c, const std::string& addr, const std::string& bucket,
              const std::string& user, const std::string& password) {
             py::gil_scoped_release rel;
             return c.Init(addr, bucket, user, password);
           },
           py::arg("config_addr"), py::arg("bucket"), py::arg("user") = "",
           py::arg("password") = "")
      .def("set",
           [](bam::CouchbaseClient& c, const std::string& k, py::bytes v) {
             std::string val(v);
             py::gil_scoped_release rel;
             return c.Set(k, val);
           })
      .def("get",
           [](bam::CouchbaseClient& c, const std::string& k) -> py::object {
             std::string v;
             int rc;
             {
               py::gil_scoped_release rel;
               rc = c.Get(k, &v);
             }
             if (rc != 0) return py::none();
             return py::bytes(v);
           })
      .def("delete_key",
           [](bam::CouchbaseClient& c, const std::string& k) {
             py::gil_scoped_release rel;
             return c.Delete(k);
           })
      .def("nvbuckets", &bam::CouchbaseClient::nvbuckets)
      .def("nservers", &bam::CouchbaseClient::nservers)
      .def("last_error", &bam::CouchbaseClient::last_error)
      .def_static("vbucket_of", [](py::bytes key, size_t nvb) {
        return bam::CouchbaseClient::VBucketOf(std::string(key), nvb);
      });

  m.def("gen_self_signed_cert", [](const std::string& cn) {
    std::string cert, key;
    if (ssl::GenerateSelfSignedCert(&cert, &key, cn) != 0) {
      throw std::runtime_error(std::string("cert generation failed: ") + ssl::LastError());
    }
    return py::make_tuple(cert, key);
  }, py::arg("cn") = "localhost");
}

// ---- redis bindings ----
#include "rpc/redis.h"

namespace {

py::object reply_to_py(const bam::RedisReply& r) {
  using bam::RedisReply;
  switch (r.type) {
    case RedisReply::NIL:
      return py::none();
    case RedisReply::STATUS:
      return py::str(r.str);
    case RedisReply::ERROR:
      throw PyRpcError(2002, r.str);
    case RedisReply::INTEGER:
      return py::int_(r.integer);
    case RedisReply::STRING:
      return py::bytes(r.str);
    case RedisReply::ARRAY: {
      py::list lst;
      for (const auto& e : r.elements) lst.append(reply_to_py(e));
      return lst;
    }
  }
  return py::none();
}

bam::RedisReply py_to_reply(py::handle obj) {
  using bam::RedisReply;
  if (obj.is_none()) return RedisReply::Nil();
  if (py::isinstance<py::bool_>(obj)) return RedisReply::Integer(obj.cast<bool>() ? 1 : 0);
  if (py::isinstance<py::int_>(obj)) return RedisReply::Integer(obj.cast<int64_t>());
  if (py::isinstance<py::bytes>(obj)) return RedisReply::Bulk(obj.cast<std::string>());
  if (py::isinstance<py::str>(obj)) {
    std::string v = obj.cast<std::string>();
    // redis convention: a leading '-' marks an error reply ("-ERR ...")
    if (!v.empty() && v[0] == '-') return RedisReply::Error(v.substr(1));
    return RedisReply::Status(v);
  }
  if (py::isinstance<py::list>(obj) || py::isinstance<py::tuple>(obj)) {
    RedisReply arr;
    arr.type = RedisReply::ARRAY;
    for (py::handle e : obj) arr.elements.push_back(py_to_reply(e));
    return arr;
  }
  return RedisReply::Error("ERR unconvertible python reply");
}

class PyRedisServer {
 public:
  PyRedisServer() : server_(new Server), service_(new RedisService) {}

  void add_handler(const std::string& command, py::function fn) {
    auto holder = std::make_shared<py::object>(std::move(fn));
    service_->AddCommandHandler(
        command, [holder](const std::vector<std::string>& args) -> RedisReply {
          // Handlers run in fibers for redis (fast path) — but Python needs
          // the GIL and must not run on a fiber: hop to the usercode pool
          // and wait.
          RedisReply out;
          std::atomic<bool> done{false};
          SubmitUserCode([&] {
            py::gil_scoped_acquire gil;
            try {
              py::list pyargs;
              for (const auto& a : args) pyargs.append(py::bytes(a));
              out = py_to_reply((*holder)(pyargs));
            } catch (const std::exception& e) {
              out = RedisReply::Error(std::string("ERR ") + e.what());
            }
            done.store(true, std::memory_order_release);
          });
          while (!done.load(std::memory_order_acquire)) fiber_yield();
          return out;
        });
  }

  int start(int port) {
    ServerOptions opts;
    opts.redis_service = service_;
    if (server_->Start(port, &opts) != 0) throw std::runtime_error("redis server start failed");
    return server_->listen_address().port;
  }

 private:
  Server* server_;
  RedisService* service_;
};

}  // namespace

void bind_redis(py::module_& m) {
  py::class_<PyRedisServer>(m, "RedisServer")
      .def(py::init<>())
      .def("add_handler", &PyRedisServer::add_handler)
      .def("start", &PyRedisServer::start, py::arg("port") = 0);

  py::class_<bam::RedisClusterClient>(m, "RedisClusterClient")
      .def(py::init<>())
      .def("init",
           [](bam::RedisClusterClient& c, const std::string& seed, int timeout_ms) {
             py::gil_scoped_release rel;
             return c.Init(seed, timeout_ms);
           },
           py::arg("seed"), py::arg("timeout_ms") = 1000)
      .def("command",
           [](bam::RedisClusterClient& c, const std::vector<std::string>& args) {
             bam::RedisReply r;
             int rc;
             {
               py::gil_scoped_release rel;
               rc = c.Command(args, &r);
             }
             if (rc != 0) throw std::runtime_error("cluster command: " + c.last_error());
             return reply_to_py(r);
           })
      .def("nslots_mapped", &bam::RedisClusterClient::nslots_mapped)
      .def_static("slot_of", [](py::bytes key) {
        return bam::RedisClusterClient::SlotOf(std::string(key));
      });

  m.def("redis_call", [](const std::string& addr, const std::vector<std::string>& args,
                         int timeout_ms) {
    bam::ChannelOptions opts;
    opts.timeout_ms = timeout_ms;
    opts.protocol = "redis";
    bam::policy::RegisterRedisProtocol();
    bam::Channel ch;
    if (ch.Init(addr.c_str(), &opts) != 0) throw std::runtime_error("redis channel init");
    std::string cmd;
    bam::EncodeRedisCommand(args, &cmd);
    bam::Controller cntl;
    bam::IOBuf req, resp;
    req.append(cmd);
    {
      py::gil_scoped_release rel;
      ch.CallMethod("redis.command", &cntl, &req, &resp, nullptr);
    }
    if (cntl.Failed()) throw PyRpcError(cntl.ErrorCode(), cntl.ErrorText());
    std::string raw = resp.to_string();
    bam::RedisReply reply;
    ssize_t c = bam::ParseRedisValue(raw.data(), raw.size(), &reply);
    if (c <= 0) throw std::runtime_error("bad redis reply");
    return reply_to_py(reply);
  }, py::arg("addr"), py::arg("args"), py::arg("timeout_ms") = 1000);
}

// ---- memcache client ----
#include "rpc/memcache.h"

void bind_memcache(py::module_& m) {
  py::class_<bam::MemcacheClient>(m, "MemcacheClient")
      .def(py::init<const std::string&, int>(), py::arg("addr"), py::arg("timeout_ms") = 1000)
      .def("ok", &bam::MemcacheClient::ok)
      .def("sasl_auth_plain", &bam::MemcacheClient::SaslAuthPlain,
           py::call_guard<py::gil_scoped_release>())
      .def("set", [](bam::MemcacheClient& c, const std::string& key, const std::string& value,
                     uint32_t flags, uint32_t exptime) {
        return c.Set(key, value, flags, exptime);
      }, py::arg("key"), py::arg("value"), py::arg("flags") = 0, py::arg("exptime") = 0,
         py::call_guard<py::gil_scoped_release>())
      .def("get", [](bam::MemcacheClient& c, const std::string& key) -> py::object {
        std::string value;
        int rc;
        {
          py::gil_scoped_release rel;
          rc = c.Get(key, &value);
        }
        if (rc == 10001) return py::none();  // key not found
        if (rc != 0) throw std::runtime_error("memcache get rc=" + std::to_string(rc));
        return py::bytes(value);
      })
      .def("delete", &bam::MemcacheClient::Delete, py::call_guard<py::gil_scoped_release>())
      .def("version", [](bam::MemcacheClient& c) {
        std::string v;
        int rc;
        {
          py::gil_scoped_release rel;
          rc = c.Version(&v);
        }
        if (rc != 0) throw std::runtime_error("memcache version rc=" + std::to_string(rc));
        return v;
      });
}

// ---- json2pb ----
#include "rpc/json2pb.h"
#include "rpc/thrift_codec.h"

namespace {

bam::json2pb::Schema schema_from_py(py::dict d);

bam::json2pb::FieldDesc field_from_py(py::handle spec) {
  using bam::json2pb::FieldDesc;
  py::tuple t = spec.cast<py::tuple>();
  FieldDesc f;
  f.number = t[0].cast<int>();
  std::string ty = t[1].cast<std::string>();
  if (ty.rfind("repeated ", 0) == 0) {
    f.repeated = true;
    ty = ty.substr(9);
  }
  if (ty == "int64") f.type = FieldDesc::INT64;
  else if (ty == "int32") f.type = FieldDesc::INT32;
  else if (ty == "uint64") f.type = FieldDesc::UINT64;
  else if (ty == "uint32") f.type = FieldDesc::UINT32;
  else if (ty == "bool") f.type = FieldDesc::BOOL;
  else if (ty == "double") f.type = FieldDesc::DOUBLE;
  else if (ty == "float") f.type = FieldDesc::FLOAT;
  else if (ty == "string") f.type = FieldDesc::STRING;
  else if (ty == "bytes") f.type = FieldDesc::BYTES;
  else if (ty == "message") {
    f.type = FieldDesc::MESSAGE;
    f.message_fields = std::make_shared<bam::json2pb::Schema>(schema_from_py(t[2].cast<py::dict>()));
  } else {
    throw std::runtime_error("unknown field type: " + ty);
  }
  return f;
}

bam::json2pb::Schema schema_from_py(py::dict d) {
  bam::json2pb::Schema s;
  for (auto item : d) {
    s[item.first.cast<std::string>()] = field_from_py(item.second);
  }
  return s;
}

}  // namespace

void bind_json2pb(py::module_& m) {
  auto j = m.def_submodule("json2pb");
  j.def("json_to_pb", [](py::dict schema, const std::string& json_text) {
    auto s = schema_from_py(schema);
    std::string wire, err;
    if (!bam::json2pb::JsonToPb(s, json_text, &wire, &err))
      throw std::runtime_error("json_to_pb: " + err);
    return py::bytes(wire);
  });
  j.def("pb_to_json", [](py::dict schema, py::bytes wire) {
    auto s = schema_from_py(schema);
    std::string text, err;
    if (!bam::json2pb::PbToJson(s, wire.cast<std::string>(), &text, &err))
      throw std::runtime_error("pb_to_json: " + err);
    return text;
  });
  // Descriptor-driven (any runtime-parsed .proto; no hand schema).
  j.def("json_to_pb_proto", [](const std::string& proto_src, const std::string& message,
                               const std::string& json_text) {
    bam::proto::DescriptorPool pool;
    std::string err;
    if (pool.ParseProtoText(proto_src, &err) != 0)
      throw std::runtime_error("proto parse: " + err);
    std::string wire;
    if (!bam::json2pb::JsonToPbByDescriptor(pool, message, json_text, &wire, &err))
      throw std::runtime_error("json_to_pb: " + err);
    return py::bytes(wire);
  });
  j.def("pb_to_json_proto", [](const std::string& proto_src, const std::string& message,
                               py::bytes wire) {
    bam::proto::DescriptorPool pool;
    std::string err;
    if (pool.ParseProtoText(proto_src, &err) != 0)
      throw std::runtime_error("proto parse: " + err);
    std::string text;
    if (!bam::json2pb::PbToJsonByDescriptor(pool, message, wire.cast<std::string>(), &text,
                                            &err))
      throw std::runtime_error("pb_to_json: " + err);
    return text;
  });
}

// ---- thrift struct codec (rpc/thrift_codec.h) ----
// JSON-described structs: keys are "id:type" with type in {bool, byte,
// i16, i32, i64, double, str, struct, list:<t>, set:<t>, map:<kt>:<vt>}.
namespace {

bam::thrift::TType ttype_of(const std::string& t) {
  using namespace bam::thrift;
  if (t == "bool") return T_BOOL;
  if (t == "byte") return T_BYTE;
  if (t == "i16") return T_I16;
  if (t == "i32") return T_I32;
  if (t == "i64") return T_I64;
  if (t == "double") return T_DOUBLE;
  if (t == "str") return T_STRING;
  if (t == "struct") return T_STRUCT;
  throw std::runtime_error("bad thrift type " + t);
}

bam::thrift::TValue tvalue_from_py(const std::string& type_desc, py::handle v);

void tstruct_from_py(py::dict d, bam::thrift::TStruct* out) {
  for (auto item : d) {
    std::string key = item.first.cast<std::string>();
    size_t colon = key.find(':');
    if (colon == std::string::npos) throw std::runtime_error("key needs id:type");
    int16_t id = (int16_t)atoi(key.substr(0, colon).c_str());
    out->emplace_back(id, tvalue_from_py(key.substr(colon + 1), item.second));
  }
}

bam::thrift::TValue tvalue_from_py(const std::string& td, py::handle v) {
  using namespace bam::thrift;
  if (td == "bool") return TValue::Bool(v.cast<bool>());
  if (td == "byte") return TValue::Byte((int8_t)v.cast<int64_t>());
  if (td == "i16") return TValue::I16((int16_t)v.cast<int64_t>());
  if (td == "i32") return TValue::I32((int32_t)v.cast<int64_t>());
  if (td == "i64") return TValue::I64(v.cast<int64_t>());
  if (td == "double") return TValue::Double(v.cast<double>());
  if (td == "str") return TValue::Str(v.cast<std::string>());
  if (td == "struct") {
    TValue x = TValue::Struct();
    tstruct_from_py(v.cast<py::dict>(), x.st.get());
    return x;
  }
  if (td.rfind("list:", 0) == 0 || td.rfind("set:", 0) == 0) {
    const bool is_set = td[0] == 's';
    std::string et = td.substr(td.find(':') + 1);
    TValue x = is_set ? TValue::Set(ttype_of(et)) : TValue::List(ttype_of(et));
    for (auto e : v.cast<py::list>()) x.list->push_back(tvalue_from_py(et, e));
    return x;
  }
  if (td.rfind("map:", 0) == 0) {
    size_t c1 = td.find(':'), c2 = td.find(':', c1 + 1);
    std::string kt = td.substr(c1 + 1, c2 - c1 - 1), vt = td.substr(c2 + 1);
    TValue x = TValue::Map(ttype_of(kt), ttype_of(vt));
    for (auto e : v.cast<py::dict>())
      x.map->emplace_back(tvalue_from_py(kt, e.first), tvalue_from_py(vt, e.second));
    return x;
  }
  throw std::runtime_error("bad thrift type " + td);
}

py::object tvalue_to_py(const bam::thrift::TValue& v);

py::dict tstruct_to_py(const bam::thrift::TStruct& st) {
  using namespace bam::thrift;
  py::dict d;
  for (const auto& kv : st) {
    const TValue& v = kv.second;
    std::string t;
    switch (v.type) {
      case T_BOOL: t = "bool"; break;
      case T_BYTE: t = "byte"; break;
      case T_I16: t = "i16"; break;
      case T_I32: t = "i32"; break;
      case T_I64: t = "i64"; break;
      case T_DOUBLE: t = "double"; break;
      case T_STRING: t = "str"; break;
      case T_STRUCT: t = "struct"; break;
      case T_LIST: t = "list:?"; break;
      case T_SET: t = "set:?"; break;
      case T_MAP: t = "map:?:?"; break;
      default: t = "?";
    }
    d[py::str(std::to_string(kv.first) + ":" + t)] = tvalue_to_py(v);
  }
  return d;
}

py::object tvalue_to_py(const bam::thrift::TValue& v) {
  using namespace bam::thrift;
  switch (v.type) {
    case T_BOOL: return py::bool_(v.i != 0);
    case T_BYTE:
    case T_I16:
    case T_I32:
    case T_I64: return py::int_(v.i);
    case T_DOUBLE: return py::float_(v.d);
    case T_STRING: return py::bytes(v.s);
    case T_STRUCT: return tstruct_to_py(v.st != nullptr ? *v.st : TStruct());
    case T_LIST:
    case T_SET: {
      py::list l;
      if (v.list != nullptr)
        for (const auto& e : *v.list) l.append(tvalue_to_py(e));
      return std::move(l);
    }
    case T_MAP: {
      py::dict d;
      if (v.map != nullptr)
        for (const auto& e : *v.map) d[tvalue_to_py(e.first)] = tvalue_to_py(e.second);
      return std::move(d);
    }
    default: return py::none();
  }
}

}  // namespace

// ---- thrift passthrough client ----
void bind_thrift(py::module_& m) {
  m.def("thrift_struct_encode", [](py::dict d) {
    bam::thrift::TStruct st;
    tstruct_from_py(d, &st);
    std::string out;
    bam::thrift::WriteStruct(st, &out);
    return py::bytes(out);
  });
  m.def("thrift_struct_decode", [](py::bytes data) {
    std::string in(data);
    bam::thrift::TStruct st;
    if (!bam::thrift::ReadStruct(in.data(), in.size(), &st))
      throw std::runtime_error("malformed thrift struct");
    return tstruct_to_py(st);
  });
  m.def("thrift_call", [](const std::string& addr, const std::string& method,
                          const std::string& payload, int timeout_ms) {
    bam::ChannelOptions opts;
    opts.timeout_ms = timeout_ms;
    opts.protocol = "thrift";
    bam::policy::RegisterThriftProtocol();
    bam::Channel ch;
    if (ch.Init(addr.c_str(), &opts) != 0) throw std::runtime_error("thrift channel init");
    bam::Controller cntl;
    bam::IOBuf req, resp;
    req.append(payload);
    cntl.call.method_name = method;
    {
      py::gil_scoped_release rel;
      ch.CallMethod("thrift." + method, &cntl, &req, &resp, nullptr);
    }
    if (cntl.Failed()) throw PyRpcError(cntl.ErrorCode(), cntl.ErrorText());
    return py::bytes(resp.to_string());
  }, py::arg("addr"), py::arg("method"), py::arg("payload"), py::arg("timeout_ms") = 1000);
}
