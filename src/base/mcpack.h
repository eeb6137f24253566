// brpc_amd: mcpack v2 binary codec (schema-less value tree).
// Parity: reference mcpack2pb/ (field_type.h layouts, serializer.cpp
// head structs) — clean-room. Wire format:
//   FieldFixedHead {type u8, name_size u8} name value      (primitives)
//   FieldShortHead {type|0x80, name_size u8, value_size u8} name value
//                                             (string<=254 / binary<=255)
//   FieldLongHead  {type u8, name_size u8, value_size u32} name value
//   OBJECT/ARRAY value = ItemsHead{item_count u32} + items
//   ISOARRAY value = IsoItemsHead{type u8} + packed primitives
// Names are C strings (name_size includes the '\0'; 0 for array items);
// strings end with '\0' (counted in value_size). Deleted fields
// (type & 0x70 == 0) are skipped. Max depth 128.
// Where the reference generates per-message protobuf converters
// (protoc-gen-mcpack), this codec exposes a dynamic Value tree plus a
// JSON bridge — the schema-less equivalent for a payload-centric runtime.
#pragma once

#include <stdint.h>

#include <map>
#include <string>
#include <vector>

namespace bam {
namespace mcpack {

struct Value {
  enum Type { NIL, BOOL, INT, UINT, DOUBLE, STRING, BINARY, OBJECT, ARRAY };
  Type type = NIL;
  bool b = false;
  int64_t i = 0;
  uint64_t u = 0;
  double d = 0;
  std::string str;                    // STRING/BINARY payload
  std::map<std::string, Value> obj;   // OBJECT fields
  std::vector<Value> arr;             // ARRAY items

  static Value Bool(bool v) { Value x; x.type = BOOL; x.b = v; return x; }
  static Value Int(int64_t v) { Value x; x.type = INT; x.i = v; return x; }
  static Value Uint(uint64_t v) { Value x; x.type = UINT; x.u = v; return x; }
  static Value Double(double v) { Value x; x.type = DOUBLE; x.d = v; return x; }
  static Value Str(std::string v) { Value x; x.type = STRING; x.str = std::move(v); return x; }
  static Value Bin(std::string v) { Value x; x.type = BINARY; x.str = std::move(v); return x; }
  static Value Object() { Value x; x.type = OBJECT; return x; }
  static Value Array() { Value x; x.type = ARRAY; return x; }
};

// Serializes `root` (must be OBJECT) as an unnamed mcpack object.
bool Serialize(const Value& root, std::string* out);

// Parses one unnamed mcpack object. Accepts all primitive widths,
// short/long string heads, ISOARRAY, and skips deleted fields.
bool Parse(const char* data, size_t n, Value* out, std::string* error = nullptr);

// JSON bridge (≙ mcpack2pb's role of making a readable front-end).
void ToJson(const Value& v, std::string* out);

}  // namespace mcpack
}  // namespace bam
