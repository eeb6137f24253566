// brpc_amd: AMF0 codec for RTMP command/data messages.
// Parity: reference brpc/amf.{h,cpp} (clean-room). Types implemented:
// number(0x00), boolean(0x01), string(0x02), object(0x03), null(0x05),
// undefined(0x06), ecma-array(0x08), object-end(0x09), strict-array(0x0a),
// long-string(0x0c).
#pragma once

#include <stdint.h>

#include <map>
#include <string>
#include <vector>

namespace bam {
namespace amf {

struct Value {
  enum Type { NUMBER, BOOLEAN, STRING, OBJECT, NUL, UNDEFINED, ECMA_ARRAY, STRICT_ARRAY };
  Type type = NUL;
  double num = 0;
  bool b = false;
  std::string str;
  std::map<std::string, Value> obj;  // OBJECT / ECMA_ARRAY
  std::vector<Value> arr;            // STRICT_ARRAY

  static Value Number(double v) { Value x; x.type = NUMBER; x.num = v; return x; }
  static Value Bool(bool v) { Value x; x.type = BOOLEAN; x.b = v; return x; }
  static Value Str(std::string v) { Value x; x.type = STRING; x.str = std::move(v); return x; }
  static Value Object() { Value x; x.type = OBJECT; return x; }
  static Value Null() { return Value(); }
};

void Encode(const Value& v, std::string* out);

// Decodes one value from data+pos; advances *pos. false on error/truncation.
bool Decode(const char* data, size_t n, size_t* pos, Value* out);

// Decodes all values until the buffer ends.
bool DecodeAll(const char* data, size_t n, std::vector<Value>* out);

}  // namespace amf
}  // namespace bam
