#include "rpc/thrift_codec.h"

#include <string.h>

namespace bam {
namespace thrift {

// ---------------- constructors ----------------

TValue TValue::Bool(bool v) {
  TValue x;
  x.type = T_BOOL;
  x.i = v ? 1 : 0;
  return x;
}
TValue TValue::Byte(int8_t v) {
  TValue x;
  x.type = T_BYTE;
  x.i = v;
  return x;
}
TValue TValue::I16(int16_t v) {
  TValue x;
  x.type = T_I16;
  x.i = v;
  return x;
}
TValue TValue::I32(int32_t v) {
  TValue x;
  x.type = T_I32;
  x.i = v;
  return x;
}
TValue TValue::I64(int64_t v) {
  TValue x;
  x.type = T_I64;
  x.i = v;
  return x;
}
TValue TValue::Double(double v) {
  TValue x;
  x.type = T_DOUBLE;
  x.d = v;
  return x;
}
TValue TValue::Str(std::string v) {
  TValue x;
  x.type = T_STRING;
  x.s = std::move(v);
  return x;
}
TValue TValue::Struct() {
  TValue x;
  x.type = T_STRUCT;
  x.st = std::make_shared<TStruct>();
  return x;
}
TValue TValue::List(TType elem) {
  TValue x;
  x.type = T_LIST;
  x.elem_type = elem;
  x.list = std::make_shared<std::vector<TValue>>();
  return x;
}
TValue TValue::Set(TType elem) {
  TValue x = List(elem);
  x.type = T_SET;
  return x;
}
TValue TValue::Map(TType key, TType value) {
  TValue x;
  x.type = T_MAP;
  x.key_type = key;
  x.elem_type = value;
  x.map = std::make_shared<std::vector<std::pair<TValue, TValue>>>();
  return x;
}

TValue& TValue::add_field(int16_t id, TValue v) {
  st->emplace_back(id, std::move(v));
  return st->back().second;
}

const TValue* TValue::field(int16_t id) const {
  if (st == nullptr) return nullptr;
  for (const auto& kv : *st)
    if (kv.first == id) return &kv.second;
  return nullptr;
}

// ---------------- writer ----------------

namespace {

void wr_u8(std::string* out, uint8_t v) { out->push_back((char)v); }
void wr_i16(std::string* out, int16_t v) {
  out->push_back((char)(v >> 8));
  out->push_back((char)v);
}
void wr_i32(std::string* out, int32_t v) {
  out->push_back((char)(v >> 24));
  out->push_back((char)(v >> 16));
  out->push_back((char)(v >> 8));
  out->push_back((char)v);
}
void wr_i64(std::string* out, int64_t v) {
  for (int i = 7; i >= 0; --i) out->push_back((char)(v >> (8 * i)));
}

void write_value(const TValue& v, std::string* out);

void write_struct_body(const TStruct& st, std::string* out) {
  for (const auto& kv : st) {
    wr_u8(out, kv.second.type);
    wr_i16(out, kv.first);
    write_value(kv.second, out);
  }
  wr_u8(out, T_STOP);
}

void write_value(const TValue& v, std::string* out) {
  switch (v.type) {
    case T_BOOL:
    case T_BYTE:
      wr_u8(out, (uint8_t)v.i);
      break;
    case T_I16:
      wr_i16(out, (int16_t)v.i);
      break;
    case T_I32:
      wr_i32(out, (int32_t)v.i);
      break;
    case T_I64:
      wr_i64(out, v.i);
      break;
    case T_DOUBLE: {
      int64_t bits;
      memcpy(&bits, &v.d, 8);
      wr_i64(out, bits);
      break;
    }
    case T_STRING:
      wr_i32(out, (int32_t)v.s.size());
      out->append(v.s);
      break;
    case T_STRUCT:
      write_struct_body(v.st != nullptr ? *v.st : TStruct(), out);
      break;
    case T_LIST:
    case T_SET: {
      wr_u8(out, v.elem_type);
      wr_i32(out, v.list != nullptr ? (int32_t)v.list->size() : 0);
      if (v.list != nullptr)
        for (const auto& e : *v.list) write_value(e, out);
      break;
    }
    case T_MAP: {
      wr_u8(out, v.key_type);
      wr_u8(out, v.elem_type);
      wr_i32(out, v.map != nullptr ? (int32_t)v.map->size() : 0);
      if (v.map != nullptr) {
        for (const auto& e : *v.map) {
          write_value(e.first, out);
          write_value(e.second, out);
        }
      }
      break;
    }
    default:
      break;
  }
}

}  // namespace

void WriteStruct(const TStruct& st, std::string* out) { write_struct_body(st, out); }

// ---------------- reader ----------------

namespace {

struct Reader {
  const uint8_t* p;
  const uint8_t* end;
  int depth = 0;

  bool u8(uint8_t* v) {
    if (p >= end) return false;
    *v = *p++;
    return true;
  }
  bool i16(int16_t* v) {
    if (end - p < 2) return false;
    *v = (int16_t)(((uint16_t)p[0] << 8) | p[1]);
    p += 2;
    return true;
  }
  bool i32(int32_t* v) {
    if (end - p < 4) return false;
    *v = (int32_t)(((uint32_t)p[0] << 24) | ((uint32_t)p[1] << 16) | ((uint32_t)p[2] << 8) |
                   p[3]);
    p += 4;
    return true;
  }
  bool i64(int64_t* v) {
    if (end - p < 8) return false;
    uint64_t x = 0;
    for (int i = 0; i < 8; ++i) x = (x << 8) | p[i];
    p += 8;
    *v = (int64_t)x;
    return true;
  }

  bool value(uint8_t type, TValue* out);
  bool struct_body(TStruct* out);
};

bool Reader::struct_body(TStruct* out) {
  if (++depth > 64) return false;
  for (;;) {
    uint8_t t;
    if (!u8(&t)) return false;
    if (t == T_STOP) {
      --depth;
      return true;
    }
    int16_t id;
    if (!i16(&id)) return false;
    TValue v;
    if (!value(t, &v)) return false;
    out->emplace_back(id, std::move(v));
  }
}

bool Reader::value(uint8_t type, TValue* out) {
  out->type = (TType)type;
  switch (type) {
    case T_BOOL:
    case T_BYTE: {
      uint8_t b;
      if (!u8(&b)) return false;
      out->i = (int8_t)b;
      return true;
    }
    case T_I16: {
      int16_t v;
      if (!i16(&v)) return false;
      out->i = v;
      return true;
    }
    case T_I32: {
      int32_t v;
      if (!i32(&v)) return false;
      out->i = v;
      return true;
    }
    case T_I64:
      return i64(&out->i);
    case T_DOUBLE: {
      int64_t bits;
      if (!i64(&bits)) return false;
      memcpy(&out->d, &bits, 8);
      return true;
    }
    case T_STRING: {
      int32_t n;
      if (!i32(&n) || n < 0 || end - p < n) return false;
      out->s.assign((const char*)p, (size_t)n);
      p += n;
      return true;
    }
    case T_STRUCT: {
      out->st = std::make_shared<TStruct>();
      return struct_body(out->st.get());
    }
    case T_LIST:
    case T_SET: {
      uint8_t et;
      int32_t n;
      if (!u8(&et) || !i32(&n) || n < 0 || n > (int32_t)(end - p)) return false;
      out->elem_type = (TType)et;
      out->list = std::make_shared<std::vector<TValue>>();
      out->list->reserve((size_t)n);
      for (int32_t i = 0; i < n; ++i) {
        TValue e;
        if (!value(et, &e)) return false;
        out->list->push_back(std::move(e));
      }
      return true;
    }
    case T_MAP: {
      uint8_t kt, vt;
      int32_t n;
      if (!u8(&kt) || !u8(&vt) || !i32(&n) || n < 0 || n > (int32_t)(end - p)) return false;
      out->key_type = (TType)kt;
      out->elem_type = (TType)vt;
      out->map = std::make_shared<std::vector<std::pair<TValue, TValue>>>();
      out->map->reserve((size_t)n);
      for (int32_t i = 0; i < n; ++i) {
        TValue k, v;
        if (!value(kt, &k) || !value(vt, &v)) return false;
        out->map->emplace_back(std::move(k), std::move(v));
      }
      return true;
    }
    default:
      return false;  // unknown type: cannot skip safely in TBinary
  }
}

}  // namespace

bool ReadStruct(const char* data, size_t n, TStruct* out, size_t* consumed) {
  Reader r{(const uint8_t*)data, (const uint8_t*)data + n};
  if (!r.struct_body(out)) return false;
  if (consumed != nullptr) *consumed = (size_t)((const char*)r.p - data);
  return true;
}

}  // namespace thrift
}  // namespace bam
