#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "base/crc32c.h"
#include "base/endpoint.h"
#include "base/fast_rand.h"
#include "base/iobuf.h"
#include "bindings/bind.h"

namespace {

using bam::IOBuf;

py::bytes iobuf_to_bytes(const IOBuf& b) {
  std::string s = b.to_string();
  return py::bytes(s);
}

}  // namespace

void bind_base(py::module_& m) {
  py::enum_<bam::Residency>(m, "Residency")
      .value("HOST", bam::RES_HOST)
      .value("PINNED", bam::RES_PINNED)
      .value("HBM", bam::RES_HBM);

  py::class_<IOBuf>(m, "IOBuf")
      .def(py::init<>())
      .def("append", [](IOBuf& b, py::bytes data) {
        char* ptr;
        Py_ssize_t len;
        PyBytes_AsStringAndSize(data.ptr(), &ptr, &len);
        b.append(ptr, (size_t)len);
      })
      .def("append_iobuf", [](IOBuf& b, const IOBuf& other) { b.append(other); })
      .def("append_with_residency",
           [](IOBuf& b, py::bytes data, bam::Residency res, int dev, uint32_t block_payload) {
             char* ptr;
             Py_ssize_t len;
             PyBytes_AsStringAndSize(data.ptr(), &ptr, &len);
             return b.append_with_residency(ptr, (size_t)len, res, dev, block_payload);
           },
           py::arg("data"), py::arg("res"), py::arg("dev") = 0, py::arg("block_payload") = 0)
      .def("cutn_to_iobuf", [](IOBuf& b, IOBuf& out, size_t n) { return b.cutn(&out, n); })
      .def("cutn", [](IOBuf& b, size_t n) {
        std::string s;
        b.cutn(&s, n);
        return py::bytes(s);
      })
      .def("pop_front", &IOBuf::pop_front)
      .def("cut_until", [](IOBuf& b, py::bytes delim) -> py::object {
        IOBuf out;
        if (b.cut_until(&out, delim.cast<std::string>()) != 0) return py::none();
        return py::bytes(out.to_string());
      })
      .def("pop_back", &IOBuf::pop_back)
      .def("copy_to",
           [](const IOBuf& b, size_t n, size_t pos) {
             std::string s;
             b.copy_to(&s, n, pos);
             return py::bytes(s);
           },
           py::arg("n") = (size_t)-1L, py::arg("pos") = 0)
      .def("to_bytes", &iobuf_to_bytes)
      .def("size", &IOBuf::size)
      .def("empty", &IOBuf::empty)
      .def("clear", &IOBuf::clear)
      .def("backing_block_num", &IOBuf::backing_block_num)
      .def("cpu_addressable", &IOBuf::cpu_addressable)
      .def("hbm_bytes", &IOBuf::hbm_bytes)
      .def("__len__", &IOBuf::size);

  m.def("iobuf_block_count", &IOBuf::block_count);
  m.def("iobuf_block_memory", &IOBuf::block_memory);
  m.def("iobuf_flush_tls_cache", &bam::iobuf_flush_tls_cache);

  m.def("crc32c", [](py::bytes data, uint32_t init) {
    char* ptr;
    Py_ssize_t len;
    PyBytes_AsStringAndSize(data.ptr(), &ptr, &len);
    return bam::crc32c::Extend(init, ptr, (size_t)len);
  }, py::arg("data"), py::arg("init") = 0);
  m.def("crc32c_combine", &bam::crc32c::Combine);
  m.def("crc32c_hw", &bam::crc32c::IsFastCrc32Supported);

  m.def("fast_rand", []() { return bam::fast_rand(); });

  py::class_<bam::EndPoint>(m, "EndPoint")
      .def(py::init<>())
      .def_property_readonly("port", [](const bam::EndPoint& ep) { return ep.port; })
      .def("__str__", [](const bam::EndPoint& ep) { return bam::endpoint2str(ep); });
  m.def("str2endpoint", [](const std::string& s) {
    bam::EndPoint ep;
    if (bam::str2endpoint(s.c_str(), &ep) != 0) throw std::runtime_error("bad endpoint: " + s);
    return ep;
  });
}

// ---- GPU submodule (wired to libbrpc_hip.so via base/gpu_loader) ----
#include "base/gpu_loader.h"

namespace bam {
namespace gputest {
bool hbm_iobuf_roundtrip(size_t n, int dev);
bool crc_matches(size_t n, int dev);
bool crc_extend_matches(size_t n1, size_t n2, int dev);
bool gather_matches(size_t total, uint32_t block, int dev);
bool pinned_roundtrip(size_t n);
double crc_gbps(size_t n, int iters, int dev);
double gather_gbps(size_t total, uint32_t block, int iters, int dev);
double d2h_gbps(size_t n, int iters, int dev);
bool snappy_cross_check(size_t n, int mode, int dev);
double snappy_compress_gbps(size_t n, int iters, int dev);
}  // namespace gputest
}  // namespace bam

void bind_gpu(py::module_& m) {
  auto g = m.def_submodule("gpu");
  g.def("initialize", &bam::gpu::initialize, py::call_guard<py::gil_scoped_release>());
  g.def("device_count", &bam::gpu::device_count, py::call_guard<py::gil_scoped_release>());
  g.def("loaded", &bam::gpu::loaded);
  g.def("load_error", &bam::gpu::load_error);
  g.def("hbm_iobuf_roundtrip", &bam::gputest::hbm_iobuf_roundtrip, py::arg("n"),
        py::arg("dev") = 0, py::call_guard<py::gil_scoped_release>());
  g.def("crc_matches", &bam::gputest::crc_matches, py::arg("n"), py::arg("dev") = 0,
        py::call_guard<py::gil_scoped_release>());
  g.def("crc_extend_matches", &bam::gputest::crc_extend_matches, py::arg("n1"), py::arg("n2"),
        py::arg("dev") = 0, py::call_guard<py::gil_scoped_release>());
  g.def("gather_matches", &bam::gputest::gather_matches, py::arg("total"), py::arg("block"),
        py::arg("dev") = 0, py::call_guard<py::gil_scoped_release>());
  g.def("pinned_roundtrip", &bam::gputest::pinned_roundtrip, py::arg("n"),
        py::call_guard<py::gil_scoped_release>());
  g.def("crc_gbps", &bam::gputest::crc_gbps, py::arg("n"), py::arg("iters") = 10,
        py::arg("dev") = 0, py::call_guard<py::gil_scoped_release>());
  g.def("gather_gbps", &bam::gputest::gather_gbps, py::arg("total"), py::arg("block"),
        py::arg("iters") = 10, py::arg("dev") = 0, py::call_guard<py::gil_scoped_release>());
  g.def("d2h_gbps", &bam::gputest::d2h_gbps, py::arg("n"), py::arg("iters") = 10,
        py::arg("dev") = 0, py::call_guard<py::gil_scoped_release>());
  g.def("snappy_cross_check", &bam::gputest::snappy_cross_check, py::arg("n"),
        py::arg("mode") = 0, py::arg("dev") = 0, py::call_guard<py::gil_scoped_release>());
  g.def("snappy_compress_gbps", &bam::gputest::snappy_compress_gbps, py::arg("n"),
        py::arg("iters") = 5, py::arg("dev") = 0, py::call_guard<py::gil_scoped_release>());
}

// ---- snappy host codec bindings (oracle for the gfx950 kernel) ----
#include "base/snappy.h"

void bind_snappy(py::module_& m) {
  auto sn = m.def_submodule("snappy");
  sn.def("compress", [](py::bytes data) {
    char* ptr;
    Py_ssize_t len;
    PyBytes_AsStringAndSize(data.ptr(), &ptr, &len);
    std::string out;
    bam::snappy::Compress(ptr, (size_t)len, &out);
    return py::bytes(out);
  });
  sn.def("uncompress", [](py::bytes data) -> py::object {
    char* ptr;
    Py_ssize_t len;
    PyBytes_AsStringAndSize(data.ptr(), &ptr, &len);
    std::string out;
    if (!bam::snappy::Uncompress(ptr, (size_t)len, &out)) return py::none();
    return py::bytes(out);
  });
}

// ---- recordio + rpc_dump + flags bindings ----
#include "base/flags.h"
#include "base/recordio.h"
#include "rpc/rpc_dump.h"

void bind_util(py::module_& m) {
  auto u = m.def_submodule("util");
  py::class_<bam::RecordWriter>(u, "RecordWriter")
      .def(py::init<const std::string&>())
      .def("ok", &bam::RecordWriter::ok)
      .def("write", [](bam::RecordWriter& w, py::bytes data) {
        return w.Write(data.cast<std::string>());
      })
      .def("flush", &bam::RecordWriter::Flush);
  py::class_<bam::RecordReader>(u, "RecordReader")
      .def(py::init<const std::string&>())
      .def("ok", &bam::RecordReader::ok)
      .def("next", [](bam::RecordReader& r) -> py::object {
        std::string payload;
        if (!r.Next(&payload)) return py::none();
        return py::bytes(payload);
      })
      .def("last_error", &bam::RecordReader::last_error);
  u.def("set_flag", &bam::flags::SetFlagValue);
  u.def("get_flag", &bam::flags::GetFlagValue);
  u.def("rpc_dump_count", &bam::rpc_dump::sampled_count);
  u.def("decode_dump_record", [](py::bytes rec) -> py::object {
    std::string service, method, body;
    if (!bam::rpc_dump::DecodeSample(rec.cast<std::string>(), &service, &method, &body))
      return py::none();
    return py::make_tuple(service, method, py::bytes(body));
  });
}

// ---- codecs + containers selftests ----
#include "base/codecs.h"
#include "base/containers.h"
#include "var/variable.h"
#include "base/mcpack.h"

void bind_codecs(py::module_& m) {
  auto c = m.def_submodule("codecs");
  c.def("base64_encode", [](py::bytes data) {
    std::string out;
    bam::Base64Encode(data.cast<std::string>(), &out);
    return out;
  });
  c.def("base64_decode", [](const std::string& data) -> py::object {
    std::string out;
    if (!bam::Base64Decode(data, &out)) return py::none();
    return py::bytes(out);
  });
  c.def("sha1_hex", [](py::bytes data) { return bam::SHA1HexDigest(data.cast<std::string>()); });

  // mcpack v2 codec (base/mcpack.h): python structures <-> mcpack bytes.
  struct McpackConv {
    static bam::mcpack::Value from_py(py::handle h) {
      namespace mc = bam::mcpack;
      if (h.is_none()) return mc::Value();
      if (py::isinstance<py::bool_>(h)) return mc::Value::Bool(h.cast<bool>());
      if (py::isinstance<py::int_>(h)) {
        long long v = h.cast<long long>();
        return mc::Value::Int(v);
      }
      if (py::isinstance<py::float_>(h)) return mc::Value::Double(h.cast<double>());
      if (py::isinstance<py::bytes>(h)) return mc::Value::Bin(h.cast<std::string>());
      if (py::isinstance<py::str>(h)) return mc::Value::Str(h.cast<std::string>());
      if (py::isinstance<py::dict>(h)) {
        mc::Value v = mc::Value::Object();
        for (auto kv : h.cast<py::dict>()) {
          v.obj[kv.first.cast<std::string>()] = from_py(kv.second);
        }
        return v;
      }
      if (py::isinstance<py::list>(h) || py::isinstance<py::tuple>(h)) {
        mc::Value v = mc::Value::Array();
        for (auto it : h.cast<py::sequence>()) v.arr.push_back(from_py(it));
        return v;
      }
      throw std::runtime_error("mcpack: unsupported python type");
    }
    static py::object to_py(const bam::mcpack::Value& v) {
      namespace mc = bam::mcpack;
      switch (v.type) {
        case mc::Value::NIL: return py::none();
        case mc::Value::BOOL: return py::bool_(v.b);
        case mc::Value::INT: return py::int_(v.i);
        case mc::Value::UINT: return py::int_(v.u);
        case mc::Value::DOUBLE: return py::float_(v.d);
        case mc::Value::STRING: return py::str(v.str);
        case mc::Value::BINARY: return py::bytes(v.str);
        case mc::Value::OBJECT: {
          py::dict d;
          for (const auto& kv : v.obj) d[py::str(kv.first)] = to_py(kv.second);
          return d;
        }
        case mc::Value::ARRAY: {
          py::list l;
          for (const auto& it : v.arr) l.append(to_py(it));
          return l;
        }
      }
      return py::none();
    }
  };
  c.def("mcpack_dumps", [](py::dict d) {
    bam::mcpack::Value v = McpackConv::from_py(d);
    std::string out;
    if (!bam::mcpack::Serialize(v, &out)) throw std::runtime_error("mcpack serialize failed");
    return py::bytes(out);
  });
  c.def("mcpack_loads", [](py::bytes data) {
    std::string s = data.cast<std::string>();
    bam::mcpack::Value v;
    std::string err;
    if (!bam::mcpack::Parse(s.data(), s.size(), &v, &err))
      throw std::runtime_error("mcpack parse failed: " + err);
    return McpackConv::to_py(v);
  });
  c.def("mcpack_to_json", [](py::bytes data) {
    std::string s = data.cast<std::string>();
    bam::mcpack::Value v;
    std::string err;
    if (!bam::mcpack::Parse(s.data(), s.size(), &v, &err))
      throw std::runtime_error("mcpack parse failed: " + err);
    std::string out;
    bam::mcpack::ToJson(v, &out);
    return out;
  });
  c.def("murmur3_32", [](py::bytes data, uint32_t seed) {
    std::string s = data.cast<std::string>();
    return bam::MurmurHash3_32(s.data(), s.size(), seed);
  }, py::arg("data"), py::arg("seed") = 0);
  c.def("containers_selftest", [] {
    bam::BoundedQueue<int> q(3);
    if (!q.push(1) || !q.push(2) || !q.push(3) || q.push(4)) return false;
    int v;
    if (!q.pop(&v) || v != 1) return false;
    if (!q.push(4)) return false;
    bam::MPSCQueue<int> mq;
    for (int i = 0; i < 100; ++i) mq.push(i);
    for (int i = 0; i < 100; ++i) {
      int x;
      if (!mq.pop(&x) || x != i) return false;
    }
    bam::MRUCache<std::string, int> cache(2);
    cache.Put("a", 1);
    cache.Put("b", 2);
    cache.Get("a");       // a is now MRU
    cache.Put("c", 3);    // evicts b
    return cache.Get("b") == nullptr && *cache.Get("a") == 1 && *cache.Get("c") == 3;
  });
  c.def("multidim_selftest", [] {
    static bam::var::MultiDimension<bam::var::Adder<int64_t>> mdim(
        "selftest_mdim", {"method"});
    *mdim.get_stats({"echo"}) << 5;
    *mdim.get_stats({"echo"}) << 2;
    *mdim.get_stats({"sleep"}) << 1;
    return mdim.count_stats() == 2 && mdim.get_stats({"echo"})->get_value() == 7;
  });
}
