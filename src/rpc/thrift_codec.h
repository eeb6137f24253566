// brpc_amd: Thrift struct codec — TBinaryProtocol value model.
// Parity: reference brpc/thrift_message.h + policy/thrift_protocol.cpp,
// which (de)serialize user thrift structs via the thrift runtime. This
// image has no thrift library, so the capability is clean-room: a dynamic
// value tree (TValue) round-trips any TBinary struct — bool/byte/i16/i32/
// i64/double/string/struct/map/set/list — so handlers and clients can
// BUILD and INSPECT real thrift structs rather than passing opaque bytes
// (the round-1 gap).
#pragma once

#include <stdint.h>

#include <map>
#include <memory>
#include <string>
#include <vector>

namespace bam {
namespace thrift {

// TBinary wire type ids.
enum TType : uint8_t {
  T_STOP = 0,
  T_BOOL = 2,
  T_BYTE = 3,
  T_DOUBLE = 4,
  T_I16 = 6,
  T_I32 = 8,
  T_I64 = 10,
  T_STRING = 11,
  T_STRUCT = 12,
  T_MAP = 13,
  T_SET = 14,
  T_LIST = 15,
};

struct TValue;

// A struct is an ordered list of (field id, value).
typedef std::vector<std::pair<int16_t, TValue>> TStruct;

struct TValue {
  TType type = T_STOP;
  int64_t i = 0;        // bool/byte/i16/i32/i64
  double d = 0;         // double
  std::string s;        // string/binary
  std::shared_ptr<TStruct> st;                       // struct
  std::shared_ptr<std::vector<TValue>> list;         // list/set elements
  std::shared_ptr<std::vector<std::pair<TValue, TValue>>> map;  // map entries
  TType elem_type = T_STOP;   // list/set element type; map value type
  TType key_type = T_STOP;    // map key type

  static TValue Bool(bool v);
  static TValue Byte(int8_t v);
  static TValue I16(int16_t v);
  static TValue I32(int32_t v);
  static TValue I64(int64_t v);
  static TValue Double(double v);
  static TValue Str(std::string v);
  static TValue Struct();
  static TValue List(TType elem);
  static TValue Set(TType elem);
  static TValue Map(TType key, TType value);

  // struct helpers
  TValue& add_field(int16_t id, TValue v);
  const TValue* field(int16_t id) const;
};

// Serializes a struct body (fields + T_STOP) in TBinaryProtocol.
void WriteStruct(const TStruct& st, std::string* out);
// Parses a struct body. Returns false on malformed input.
bool ReadStruct(const char* data, size_t n, TStruct* out, size_t* consumed = nullptr);

}  // namespace thrift
}  // namespace bam
