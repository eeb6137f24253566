// Python surface for the self-contained protobuf runtime (base/proto.h).
// Tests use the installed python google.protobuf as a wire/JSON oracle.
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <memory>

#include "base/proto.h"
#include "rpc/policy/hpack.h"
#include "bindings/bind.h"

namespace {

struct PyPool {
  std::shared_ptr<bam::proto::DescriptorPool> pool =
      std::make_shared<bam::proto::DescriptorPool>();
};

struct PyMsg {
  std::shared_ptr<bam::proto::DescriptorPool> pool;  // keep alive
  std::unique_ptr<bam::proto::DynMessage> msg;
};

}  // namespace

void bind_hpack(py::module_& m) {
  auto h = m.def_submodule("hpack");
  h.def("huffman_encode", [](py::bytes data) {
    std::string out;
    bam::hpack::HuffmanEncode(std::string(data), &out);
    return py::bytes(out);
  });
  h.def("huffman_decode", [](py::bytes data) {
    std::string in(data), out;
    if (!bam::hpack::HuffmanDecode(in.data(), in.size(), &out))
      throw std::runtime_error("huffman decode failed");
    return py::bytes(out);
  });
  h.def("huffman_table", []() {
    std::vector<std::pair<uint32_t, int>> t;
    bam::hpack::HuffmanTable(&t);
    return t;
  });
  h.def("encode_int", [](uint64_t v, int prefix, int flags) {
    std::string out;
    bam::hpack::EncodeInt(&out, v, prefix, (uint8_t)flags);
    return py::bytes(out);
  });
  py::class_<bam::hpack::Encoder>(h, "Encoder")
      .def(py::init<size_t>(), py::arg("max_table_size") = 4096)
      .def("encode", [](bam::hpack::Encoder& e,
                        const std::vector<std::pair<std::string, std::string>>& headers) {
        std::string out;
        e.Encode(headers, &out);
        return py::bytes(out);
      });
  py::class_<bam::hpack::Decoder>(h, "Decoder")
      .def(py::init<size_t>(), py::arg("max_table_size") = 4096)
      .def("decode", [](bam::hpack::Decoder& d, py::bytes block) {
        std::string in(block);
        std::vector<bam::hpack::Header> out;
        if (!d.Decode(in.data(), in.size(), &out))
          throw std::runtime_error("hpack decode failed");
        py::list res;
        for (auto& kv : out) res.append(py::make_tuple(py::bytes(kv.first), py::bytes(kv.second)));
        return res;
      });
}

void bind_proto(py::module_& m) {
  auto p = m.def_submodule("proto");

  py::class_<PyPool>(p, "Pool")
      .def(py::init<>())
      .def("parse",
           [](PyPool& self, const std::string& text) {
             std::string err;
             if (self.pool->ParseProtoText(text, &err) != 0)
               throw std::runtime_error("proto parse: " + err);
           })
      .def("messages", [](PyPool& self) { return self.pool->message_names(); })
      .def("services", [](PyPool& self) { return self.pool->service_names(); })
      .def("service_methods",
           [](PyPool& self, const std::string& name) {
             const bam::proto::ServiceDef* s = self.pool->FindService(name);
             if (s == nullptr) throw std::runtime_error("no service " + name);
             std::vector<std::vector<std::string>> out;
             for (const auto& mth : s->methods)
               out.push_back({mth.name, mth.input_type, mth.output_type});
             return out;
           })
      .def("describe_message",
           [](PyPool& self, const std::string& name) {
             const bam::proto::MessageDef* d = self.pool->FindMessage(name);
             if (d == nullptr) throw std::runtime_error("no message " + name);
             py::list out;
             for (const auto& f : d->fields) {
               py::dict fd;
               fd["name"] = f.name;
               fd["json_name"] = f.json_name;
               fd["number"] = f.number;
               fd["type"] = (int)f.type;
               fd["repeated"] = f.repeated;
               fd["packed"] = f.packed;
               fd["is_map"] = f.is_map;
               fd["oneof_index"] = f.oneof_index;
               fd["type_name"] = f.type_name;
               out.append(fd);
             }
             return out;
           })
      .def("enum_values",
           [](PyPool& self, const std::string& name) {
             const bam::proto::EnumDef* e = self.pool->FindEnum(name);
             if (e == nullptr) throw std::runtime_error("no enum " + name);
             return e->values;
           })
      .def("enums",
           [](PyPool& self) {
             std::vector<std::string> out;
             for (const auto& kv : self.pool->enums_) out.push_back(kv.first);
             return out;
           })
      .def("new_message", [](PyPool& self, const std::string& full_name) {
        const bam::proto::MessageDef* d = self.pool->FindMessage(full_name);
        if (d == nullptr) throw std::runtime_error("no message " + full_name);
        PyMsg out;
        out.pool = self.pool;
        out.msg.reset(new bam::proto::DynMessage(self.pool.get(), d));
        return out;
      });

  py::class_<PyMsg>(p, "Message")
      .def("parse_wire",
           [](PyMsg& self, py::bytes data) {
             std::string s(data);
             self.msg->clear();
             if (!self.msg->ParseWire(s.data(), s.size()))
               throw std::runtime_error("wire parse failed");
           })
      .def("serialize_wire",
           [](PyMsg& self) {
             std::string out;
             self.msg->SerializeWire(&out);
             return py::bytes(out);
           })
      .def("from_json",
           [](PyMsg& self, const std::string& text) {
             std::string err;
             if (!self.msg->FromJson(text, &err))
               throw std::runtime_error("from_json: " + err);
           })
      .def("to_json",
           [](PyMsg& self, bool original_names) {
             std::string out;
             self.msg->ToJson(&out, original_names);
             return out;
           },
           py::arg("original_names") = false)
      .def("get_int", [](PyMsg& self, const std::string& n, size_t i) {
        return self.msg->get_int(n, i);
      }, py::arg("name"), py::arg("idx") = 0)
      .def("get_str", [](PyMsg& self, const std::string& n, size_t i) {
        return py::bytes(self.msg->get_str(n, i));
      }, py::arg("name"), py::arg("idx") = 0)
      .def("get_double", [](PyMsg& self, const std::string& n, size_t i) {
        return self.msg->get_double(n, i);
      }, py::arg("name"), py::arg("idx") = 0)
      .def("count", [](PyMsg& self, const std::string& n) { return self.msg->count(n); })
      .def("set_int", [](PyMsg& self, const std::string& n, int64_t v) {
        self.msg->set_int(n, v);
      })
      .def("set_str", [](PyMsg& self, const std::string& n, py::bytes v) {
        self.msg->set_str(n, std::string(v));
      })
      .def("set_double", [](PyMsg& self, const std::string& n, double v) {
        self.msg->set_double(n, v);
      })
      .def("add_int", [](PyMsg& self, const std::string& n, int64_t v) {
        self.msg->add_int(n, v);
      })
      .def("add_str", [](PyMsg& self, const std::string& n, py::bytes v) {
        self.msg->add_str(n, std::string(v));
      });
}
