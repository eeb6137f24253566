// brpc_amd: AMF0 codec (see amf.h).
#include "rpc/amf.h"

#include <string.h>

namespace bam {
namespace amf {

namespace {

constexpr uint8_t kNumber = 0x00;
constexpr uint8_t kBoolean = 0x01;
constexpr uint8_t kString = 0x02;
constexpr uint8_t kObject = 0x03;
constexpr uint8_t kNull = 0x05;
constexpr uint8_t kUndefined = 0x06;
constexpr uint8_t kEcmaArray = 0x08;
constexpr uint8_t kObjectEnd = 0x09;
constexpr uint8_t kStrictArray = 0x0a;
constexpr uint8_t kLongString = 0x0c;

void put_u16(std::string* out, uint16_t v) {
  out->push_back((char)(v >> 8));
  out->push_back((char)v);
}
void put_u32(std::string* out, uint32_t v) {
  out->push_back((char)(v >> 24));
  out->push_back((char)(v >> 16));
  out->push_back((char)(v >> 8));
  out->push_back((char)v);
}
void put_double(std::string* out, double d) {
  uint64_t bits;
  memcpy(&bits, &d, 8);
  for (int i = 7; i >= 0; --i) out->push_back((char)(bits >> (i * 8)));
}
void put_short_str(std::string* out, const std::string& s) {
  put_u16(out, (uint16_t)s.size());
  out->append(s);
}

bool get_u16(const char* p, size_t n, size_t* pos, uint16_t* v) {
  if (n - *pos < 2) return false;
  *v = ((uint16_t)(uint8_t)p[*pos] << 8) | (uint8_t)p[*pos + 1];
  *pos += 2;
  return true;
}
bool get_u32(const char* p, size_t n, size_t* pos, uint32_t* v) {
  if (n - *pos < 4) return false;
  *v = ((uint32_t)(uint8_t)p[*pos] << 24) | ((uint32_t)(uint8_t)p[*pos + 1] << 16) |
       ((uint32_t)(uint8_t)p[*pos + 2] << 8) | (uint8_t)p[*pos + 3];
  *pos += 4;
  return true;
}

bool decode_value(const char* p, size_t n, size_t* pos, Value* out, int depth);

bool decode_object_body(const char* p, size_t n, size_t* pos, Value* out, int depth) {
  for (;;) {
    uint16_t klen;
    if (!get_u16(p, n, pos, &klen)) return false;
    if (n - *pos < klen) return false;
    std::string key(p + *pos, klen);
    *pos += klen;
    if (klen == 0) {
      if (n - *pos < 1 || (uint8_t)p[*pos] != kObjectEnd) return false;
      *pos += 1;
      return true;
    }
    Value v;
    if (!decode_value(p, n, pos, &v, depth + 1)) return false;
    out->obj[key] = std::move(v);
  }
}

bool decode_value(const char* p, size_t n, size_t* pos, Value* out, int depth) {
  if (depth > 32 || n - *pos < 1) return false;
  uint8_t type = (uint8_t)p[(*pos)++];
  switch (type) {
    case kNumber: {
      if (n - *pos < 8) return false;
      uint64_t bits = 0;
      for (int i = 0; i < 8; ++i) bits = (bits << 8) | (uint8_t)p[*pos + i];
      *pos += 8;
      double d;
      memcpy(&d, &bits, 8);
      *out = Value::Number(d);
      return true;
    }
    case kBoolean:
      if (n - *pos < 1) return false;
      *out = Value::Bool(p[(*pos)++] != 0);
      return true;
    case kString: {
      uint16_t len;
      if (!get_u16(p, n, pos, &len) || n - *pos < len) return false;
      *out = Value::Str(std::string(p + *pos, len));
      *pos += len;
      return true;
    }
    case kLongString: {
      uint32_t len;
      if (!get_u32(p, n, pos, &len) || n - *pos < len) return false;
      *out = Value::Str(std::string(p + *pos, len));
      *pos += len;
      return true;
    }
    case kObject:
      *out = Value::Object();
      return decode_object_body(p, n, pos, out, depth);
    case kEcmaArray: {
      uint32_t count;
      if (!get_u32(p, n, pos, &count)) return false;
      *out = Value::Object();
      out->type = Value::ECMA_ARRAY;
      return decode_object_body(p, n, pos, out, depth);
    }
    case kStrictArray: {
      uint32_t count;
      if (!get_u32(p, n, pos, &count)) return false;
      if (count > 65536) return false;
      out->type = Value::STRICT_ARRAY;
      for (uint32_t i = 0; i < count; ++i) {
        Value v;
        if (!decode_value(p, n, pos, &v, depth + 1)) return false;
        out->arr.push_back(std::move(v));
      }
      return true;
    }
    case kNull:
      *out = Value::Null();
      return true;
    case kUndefined:
      out->type = Value::UNDEFINED;
      return true;
    default:
      return false;
  }
}

}  // namespace

void Encode(const Value& v, std::string* out) {
  switch (v.type) {
    case Value::NUMBER:
      out->push_back((char)kNumber);
      put_double(out, v.num);
      break;
    case Value::BOOLEAN:
      out->push_back((char)kBoolean);
      out->push_back(v.b ? 1 : 0);
      break;
    case Value::STRING:
      if (v.str.size() <= 0xffff) {
        out->push_back((char)kString);
        put_short_str(out, v.str);
      } else {
        out->push_back((char)kLongString);
        put_u32(out, (uint32_t)v.str.size());
        out->append(v.str);
      }
      break;
    case Value::OBJECT:
    case Value::ECMA_ARRAY:
      if (v.type == Value::ECMA_ARRAY) {
        out->push_back((char)kEcmaArray);
        put_u32(out, (uint32_t)v.obj.size());
      } else {
        out->push_back((char)kObject);
      }
      for (const auto& kv : v.obj) {
        put_short_str(out, kv.first);
        Encode(kv.second, out);
      }
      put_u16(out, 0);
      out->push_back((char)kObjectEnd);
      break;
    case Value::STRICT_ARRAY:
      out->push_back((char)kStrictArray);
      put_u32(out, (uint32_t)v.arr.size());
      for (const Value& it : v.arr) Encode(it, out);
      break;
    case Value::UNDEFINED:
      out->push_back((char)kUndefined);
      break;
    case Value::NUL:
      out->push_back((char)kNull);
      break;
  }
}

bool Decode(const char* data, size_t n, size_t* pos, Value* out) {
  return decode_value(data, n, pos, out, 0);
}

bool DecodeAll(const char* data, size_t n, std::vector<Value>* out) {
  size_t pos = 0;
  while (pos < n) {
    Value v;
    if (!decode_value(data, n, &pos, &v, 0)) return false;
    out->push_back(std::move(v));
  }
  return true;
}

}  // namespace amf
}  // namespace bam
