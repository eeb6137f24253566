// brpc_amd: self-contained protobuf runtime — .proto parsing, descriptor
// pool, dynamic messages, wire (de)serialization and JSON conversion.
//
// Parity intent: the reference integrates google::protobuf generated stubs
// into Channel/Server (brpc/channel.h:189-228, server.cpp:844-875) and
// ships descriptor-driven json2pb (json2pb/json_to_pb.h:55). This image
// has no C++ libprotobuf, so the capability is rebuilt clean-room: a
// DescriptorPool parses .proto source at runtime (proto2/proto3 common
// subset: scalar types, string/bytes, enums, nested messages, repeated +
// packed, oneof, map<k,v>, services), DynMessage holds field values keyed
// by number with full wire round-tripping (unknown fields preserved), and
// tools/bamproto.py generates C++ structs + Channel/Server stubs from the
// same descriptors (see examples/echo.proto). Wire bytes interoperate
// with any protobuf runtime — tests/test_proto.py cross-checks every
// path against the installed python google.protobuf as an oracle.
#pragma once

#include <stdint.h>

#include <map>
#include <memory>
#include <string>
#include <vector>

namespace bam {
namespace proto {

struct FieldDef {
  enum Type {
    TYPE_DOUBLE, TYPE_FLOAT, TYPE_INT32, TYPE_INT64, TYPE_UINT32, TYPE_UINT64,
    TYPE_SINT32, TYPE_SINT64, TYPE_FIXED32, TYPE_FIXED64, TYPE_SFIXED32,
    TYPE_SFIXED64, TYPE_BOOL, TYPE_STRING, TYPE_BYTES, TYPE_ENUM, TYPE_MESSAGE,
  };
  std::string name;
  std::string json_name;   // lowerCamelCase (proto3 JSON)
  int number = 0;
  Type type = TYPE_INT64;
  bool repeated = false;
  bool packed = false;       // wire packing for repeated scalars
  bool is_map = false;       // synthesized entry message in type_name
  int oneof_index = -1;      // -1 = not in a oneof
  std::string type_name;     // full name for MESSAGE/ENUM
};

struct MessageDef {
  std::string full_name;
  std::vector<FieldDef> fields;
  std::vector<std::string> oneof_names;
  const FieldDef* field_by_number(int n) const;
  const FieldDef* field_by_name(const std::string& n) const;
};

struct EnumDef {
  std::string full_name;
  std::map<std::string, int32_t> values;
  std::map<int32_t, std::string> names;
};

struct MethodDef {
  std::string name;
  std::string input_type;   // full message name
  std::string output_type;
};

struct ServiceDef {
  std::string full_name;
  std::vector<MethodDef> methods;
};

class DescriptorPool {
 public:
  // Parses one .proto source file (syntax proto2/proto3). Imports are
  // resolved against previously-parsed files (parse dependencies first).
  // Returns 0, or -1 with *err.
  int ParseProtoText(const std::string& text, std::string* err);

  const MessageDef* FindMessage(const std::string& full_name) const;
  const EnumDef* FindEnum(const std::string& full_name) const;
  const ServiceDef* FindService(const std::string& full_name) const;
  std::vector<std::string> message_names() const;
  std::vector<std::string> service_names() const;

  // Internal storage (exposed for the file-local parser + codegen tool).
  std::map<std::string, MessageDef> messages_;
  std::map<std::string, EnumDef> enums_;
  std::map<std::string, ServiceDef> services_;
};

// A dynamic message: values keyed by field number.
class DynMessage {
 public:
  DynMessage(const DescriptorPool* pool, const MessageDef* def)
      : pool_(pool), def_(def) {}

  const MessageDef* descriptor() const { return def_; }

  struct Value {
    uint64_t u = 0;      // varint/fixed raw (zigzag already decoded for sint)
    double d = 0;        // double/float
    std::string s;       // string/bytes
    std::shared_ptr<DynMessage> m;
  };

  // Parses standard protobuf wire bytes. Unknown fields are kept verbatim
  // and re-emitted on serialize. Returns false on malformed input.
  bool ParseWire(const char* data, size_t n);
  void SerializeWire(std::string* out) const;

  // JSON (proto3 mapping: camelCase names accepted and emitted, int64 as
  // string, bytes as base64, enums by name).
  bool FromJson(const std::string& json_text, std::string* err);
  void ToJson(std::string* out, bool original_names = false) const;

  // Field access (by field name).
  bool has(const std::string& name) const;
  size_t count(const std::string& name) const;
  int64_t get_int(const std::string& name, size_t idx = 0) const;
  uint64_t get_uint(const std::string& name, size_t idx = 0) const;
  double get_double(const std::string& name, size_t idx = 0) const;
  bool get_bool(const std::string& name, size_t idx = 0) const;
  const std::string& get_str(const std::string& name, size_t idx = 0) const;
  DynMessage* mutable_msg(const std::string& name, size_t idx = 0);
  void set_int(const std::string& name, int64_t v);
  void set_uint(const std::string& name, uint64_t v);
  void set_double(const std::string& name, double v);
  void set_bool(const std::string& name, bool v);
  void set_str(const std::string& name, const std::string& v);
  void add_int(const std::string& name, int64_t v);
  void add_str(const std::string& name, const std::string& v);
  DynMessage* add_msg(const std::string& name);
  void clear() { fields_.clear(); unknown_.clear(); }

  const std::map<int, std::vector<Value>>& raw_fields() const { return fields_; }

 private:
  friend struct WireCodec;
  const DescriptorPool* pool_;
  const MessageDef* def_;
  std::map<int, std::vector<Value>> fields_;
  std::string unknown_;  // unrecognized field bytes, re-emitted verbatim
};

}  // namespace proto
}  // namespace bam
