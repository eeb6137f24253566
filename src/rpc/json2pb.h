// brpc_amd: json2pb — schema-driven JSON ⇄ protobuf-wire conversion.
// Parity: reference src/json2pb (JsonToProtoMessage / ProtoMessageToJson).
// Instead of compiled protobuf descriptors, schemas are runtime trees
// (built from Python dicts via the bindings) describing field numbers and
// types — the wire bytes are standard protobuf and interoperate with any
// protobuf runtime (including the std protocol's payloads).
#pragma once

#include <map>
#include <memory>
#include <string>
#include <vector>

#include "base/proto.h"

namespace bam {
namespace json2pb {

struct FieldDesc {
  enum Type { INT64, UINT64, INT32, UINT32, BOOL, DOUBLE, FLOAT, STRING, BYTES, MESSAGE };
  int number = 0;
  Type type = INT64;
  bool repeated = false;
  std::shared_ptr<std::map<std::string, FieldDesc>> message_fields;  // MESSAGE
};

typedef std::map<std::string, FieldDesc> Schema;

// JSON text -> protobuf wire bytes. false on schema/json mismatch.
bool JsonToPb(const Schema& schema, const std::string& json_text, std::string* wire,
              std::string* error = nullptr);
// protobuf wire bytes -> JSON text.
bool PbToJson(const Schema& schema, const std::string& wire, std::string* json_text,
              std::string* error = nullptr);

// ---- descriptor-driven (round 2; parity: reference json2pb works on any
// pb Message via descriptors — here via base/proto.h's DescriptorPool,
// so ANY .proto parses at runtime, no hand-written schema needed). ----
bool JsonToPbByDescriptor(const ::bam::proto::DescriptorPool& pool,
                          const std::string& message_full_name, const std::string& json_text,
                          std::string* wire, std::string* error = nullptr);
bool PbToJsonByDescriptor(const ::bam::proto::DescriptorPool& pool,
                          const std::string& message_full_name, const std::string& wire,
                          std::string* json_text, std::string* error = nullptr);

}  // namespace json2pb
}  // namespace bam
