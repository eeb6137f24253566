// brpc_amd: mcpack v2 codec implementation (see mcpack.h).
#include "base/mcpack.h"

#include <string.h>

namespace bam {
namespace mcpack {

namespace {

// field types (reference mcpack2pb/field_type.h values)
constexpr uint8_t kObject = 0x10;
constexpr uint8_t kArray = 0x20;
constexpr uint8_t kIsoArray = 0x30;
constexpr uint8_t kObjectIsoArray = 0x40;
constexpr uint8_t kString = 0x50;
constexpr uint8_t kBinary = 0x60;
constexpr uint8_t kInt8 = 0x11, kInt16 = 0x12, kInt32 = 0x14, kInt64 = 0x18;
constexpr uint8_t kUint8 = 0x21, kUint16 = 0x22, kUint32 = 0x24, kUint64 = 0x28;
constexpr uint8_t kBool = 0x31;
constexpr uint8_t kFloat = 0x44, kDouble = 0x48;
constexpr uint8_t kNull = 0x61;
constexpr uint8_t kShortMask = 0x80;
constexpr uint8_t kFixedMask = 0x0f;
constexpr uint8_t kNonDeletedMask = 0x70;
constexpr int kMaxDepth = 128;

template <typename T>
void put_pod(std::string* out, T v) {
  out->append((const char*)&v, sizeof(v));
}

void put_name(std::string* out, const std::string& name) {
  if (!name.empty()) {
    out->append(name);
    out->push_back('\0');
  }
}
inline uint8_t name_size(const std::string& name) {
  return name.empty() ? 0 : (uint8_t)(name.size() + 1);
}

bool serialize_field(const Value& v, const std::string& name, std::string* out, int depth);

bool serialize_items(const Value& v, std::string* out, int depth) {
  put_pod<uint32_t>(out, v.type == Value::OBJECT ? (uint32_t)v.obj.size()
                                                 : (uint32_t)v.arr.size());
  if (v.type == Value::OBJECT) {
    for (const auto& kv : v.obj) {
      if (!serialize_field(kv.second, kv.first, out, depth)) return false;
    }
  } else {
    for (const Value& it : v.arr) {
      if (!serialize_field(it, "", out, depth)) return false;
    }
  }
  return true;
}

bool serialize_field(const Value& v, const std::string& name, std::string* out, int depth) {
  if (depth > kMaxDepth || name.size() > 254) return false;
  switch (v.type) {
    case Value::NIL:
      out->push_back((char)kNull);
      out->push_back((char)name_size(name));
      put_name(out, name);
      out->push_back('\0');
      return true;
    case Value::BOOL:
      out->push_back((char)kBool);
      out->push_back((char)name_size(name));
      put_name(out, name);
      out->push_back(v.b ? 1 : 0);
      return true;
    case Value::INT:
      // smallest representation, like the reference's typed fields
      if (v.i >= INT8_MIN && v.i <= INT8_MAX) {
        out->push_back((char)kInt8);
        out->push_back((char)name_size(name));
        put_name(out, name);
        put_pod<int8_t>(out, (int8_t)v.i);
      } else if (v.i >= INT32_MIN && v.i <= INT32_MAX) {
        out->push_back((char)kInt32);
        out->push_back((char)name_size(name));
        put_name(out, name);
        put_pod<int32_t>(out, (int32_t)v.i);
      } else {
        out->push_back((char)kInt64);
        out->push_back((char)name_size(name));
        put_name(out, name);
        put_pod<int64_t>(out, v.i);
      }
      return true;
    case Value::UINT:
      if (v.u <= UINT32_MAX) {
        out->push_back((char)kUint32);
        out->push_back((char)name_size(name));
        put_name(out, name);
        put_pod<uint32_t>(out, (uint32_t)v.u);
      } else {
        out->push_back((char)kUint64);
        out->push_back((char)name_size(name));
        put_name(out, name);
        put_pod<uint64_t>(out, v.u);
      }
      return true;
    case Value::DOUBLE:
      out->push_back((char)kDouble);
      out->push_back((char)name_size(name));
      put_name(out, name);
      put_pod<double>(out, v.d);
      return true;
    case Value::STRING: {
      size_t vsize = v.str.size() + 1;  // trailing '\0' counted
      if (vsize <= 255) {
        out->push_back((char)(kString | kShortMask));
        out->push_back((char)name_size(name));
        out->push_back((char)vsize);
      } else {
        out->push_back((char)kString);
        out->push_back((char)name_size(name));
        put_pod<uint32_t>(out, (uint32_t)vsize);
      }
      put_name(out, name);
      out->append(v.str);
      out->push_back('\0');
      return true;
    }
    case Value::BINARY: {
      size_t vsize = v.str.size();
      if (vsize <= 255) {
        out->push_back((char)(kBinary | kShortMask));
        out->push_back((char)name_size(name));
        out->push_back((char)vsize);
      } else {
        out->push_back((char)kBinary);
        out->push_back((char)name_size(name));
        put_pod<uint32_t>(out, (uint32_t)vsize);
      }
      put_name(out, name);
      out->append(v.str);
      return true;
    }
    case Value::OBJECT:
    case Value::ARRAY: {
      // Uniform INT32-able arrays emit the compact ISOARRAY form
      // (IsoItemsHead{type} + packed values) like the reference writer.
      if (v.type == Value::ARRAY && !v.arr.empty()) {
        bool all_i32 = true;
        for (const Value& it : v.arr) {
          if (it.type != Value::INT || it.i < INT32_MIN || it.i > INT32_MAX) {
            all_i32 = false;
            break;
          }
        }
        if (all_i32) {
          out->push_back((char)kIsoArray);
          out->push_back((char)name_size(name));
          put_pod<uint32_t>(out, (uint32_t)(1 + 4 * v.arr.size()));
          put_name(out, name);
          out->push_back((char)kInt32);
          for (const Value& it : v.arr) put_pod<int32_t>(out, (int32_t)it.i);
          return true;
        }
      }
      out->push_back((char)(v.type == Value::OBJECT ? kObject : kArray));
      out->push_back((char)name_size(name));
      size_t size_pos = out->size();
      put_pod<uint32_t>(out, 0);  // value_size patched below
      put_name(out, name);
      size_t value_start = out->size();
      if (!serialize_items(v, out, depth + 1)) return false;
      uint32_t vsize = (uint32_t)(out->size() - value_start);
      memcpy(&(*out)[size_pos], &vsize, 4);
      return true;
    }
  }
  return false;
}

// ---------------- parser ----------------

struct Cursor {
  const char* p;
  size_t n;
  std::string* err;
  bool fail(const char* what) {
    if (err != nullptr && err->empty()) *err = what;
    return false;
  }
  bool take(void* dst, size_t k) {
    if (n < k) return false;
    memcpy(dst, p, k);
    p += k;
    n -= k;
    return true;
  }
  bool skip(size_t k) {
    if (n < k) return false;
    p += k;
    n -= k;
    return true;
  }
};

bool parse_field(Cursor* c, std::string* name_out, Value* out, int depth);

bool parse_primitive(uint8_t type, Cursor* c, Value* out) {
  size_t vs = type & kFixedMask;
  char buf[8] = {0};
  if (vs > sizeof(buf)) return false;  // e.g. type 0x3a claims 10 bytes (fuzz find)
  if (!c->take(buf, vs)) return false;
  switch (type) {
    case kBool:
      *out = Value::Bool(buf[0] != 0);
      return true;
    case kInt8:
      *out = Value::Int(*(int8_t*)buf);
      return true;
    case kInt16: {
      int16_t v;
      memcpy(&v, buf, 2);
      *out = Value::Int(v);
      return true;
    }
    case kInt32: {
      int32_t v;
      memcpy(&v, buf, 4);
      *out = Value::Int(v);
      return true;
    }
    case kInt64: {
      int64_t v;
      memcpy(&v, buf, 8);
      *out = Value::Int(v);
      return true;
    }
    case kUint8:
      *out = Value::Uint((uint8_t)buf[0]);
      return true;
    case kUint16: {
      uint16_t v;
      memcpy(&v, buf, 2);
      *out = Value::Uint(v);
      return true;
    }
    case kUint32: {
      uint32_t v;
      memcpy(&v, buf, 4);
      *out = Value::Uint(v);
      return true;
    }
    case kUint64: {
      uint64_t v;
      memcpy(&v, buf, 8);
      *out = Value::Uint(v);
      return true;
    }
    case kFloat: {
      float v;
      memcpy(&v, buf, 4);
      *out = Value::Double(v);
      return true;
    }
    case kDouble: {
      double v;
      memcpy(&v, buf, 8);
      *out = Value::Double(v);
      return true;
    }
    default:
      return false;
  }
}

bool parse_items(uint8_t type, Cursor* c, Value* out, int depth) {
  uint32_t count;
  if (!c->take(&count, 4)) return c->fail("truncated ItemsHead");
  *out = type == kObject ? Value::Object() : Value::Array();
  for (uint32_t i = 0; i < count; ++i) {
    std::string nm;
    Value item;
    if (!parse_field(c, &nm, &item, depth + 1)) return false;
    if (type == kObject) {
      out->obj[nm] = std::move(item);
    } else {
      out->arr.push_back(std::move(item));
    }
  }
  return true;
}

bool parse_field(Cursor* c, std::string* name_out, Value* out, int depth) {
  if (depth > kMaxDepth) return c->fail("max depth exceeded");
  uint8_t type, nsize;
  if (!c->take(&type, 1) || !c->take(&nsize, 1)) return c->fail("truncated head");
  // deleted field: skip its full extent
  const bool deleted = (type & kNonDeletedMask) == 0 && type != 0;
  uint8_t base = type & (uint8_t)~kShortMask;
  uint32_t vsize = 0;
  bool have_vsize = false;
  if (type & kShortMask) {
    uint8_t s;
    if (!c->take(&s, 1)) return c->fail("truncated short head");
    vsize = s;
    have_vsize = true;
  } else if (base == kObject || base == kArray || base == kIsoArray ||
             base == kObjectIsoArray || base == kString || base == kBinary) {
    if (!c->take(&vsize, 4)) return c->fail("truncated long head");
    have_vsize = true;
  }
  std::string name;
  if (nsize > 0) {
    if (c->n < nsize) return c->fail("truncated name");
    name.assign(c->p, nsize - 1);  // drop the '\0'
    c->skip(nsize);
  }
  *name_out = std::move(name);
  if (deleted) {
    size_t skip_n = have_vsize ? vsize : (type & kFixedMask);
    if (!c->skip(skip_n)) return c->fail("truncated deleted field");
    out->type = Value::NIL;
    return true;
  }
  if (type == kNull) {
    if (!c->skip(1)) return c->fail("truncated null");
    *out = Value();
    return true;
  }
  if (base == kString) {
    if (!have_vsize || vsize == 0 || c->n < vsize) return c->fail("truncated string");
    *out = Value::Str(std::string(c->p, vsize - 1));
    c->skip(vsize);
    return true;
  }
  if (base == kBinary) {
    if (!have_vsize || c->n < vsize) return c->fail("truncated binary");
    *out = Value::Bin(std::string(c->p, vsize));
    c->skip(vsize);
    return true;
  }
  if (base == kObject || base == kArray) {
    if (!have_vsize || c->n < vsize) return c->fail("truncated object/array");
    Cursor sub{c->p, vsize, c->err};
    if (!parse_items(base, &sub, out, depth)) return false;
    c->skip(vsize);
    return true;
  }
  if (base == kIsoArray) {
    // IsoItemsHead {type u8} + packed primitives
    if (!have_vsize || vsize < 1 || c->n < vsize) return c->fail("truncated isoarray");
    uint8_t item_type = (uint8_t)c->p[0];
    size_t isz = item_type & kFixedMask;
    if (isz == 0 || (vsize - 1) % isz != 0) return c->fail("bad isoarray");
    *out = Value::Array();
    Cursor sub{c->p + 1, vsize - 1, c->err};
    while (sub.n > 0) {
      Value item;
      if (!parse_primitive(item_type, &sub, &item)) return c->fail("bad isoarray item");
      out->arr.push_back(std::move(item));
    }
    c->skip(vsize);
    return true;
  }
  if (base == kObjectIsoArray) {
    // column-major repeated objects; expose as an OBJECT of column arrays
    if (!have_vsize || c->n < vsize) return c->fail("truncated objectisoarray");
    Cursor sub{c->p, vsize, c->err};
    if (!parse_items(kObject, &sub, out, depth)) return false;
    c->skip(vsize);
    return true;
  }
  if ((type & kFixedMask) != 0) {
    if (!parse_primitive(type, c, out)) return c->fail("truncated primitive");
    return true;
  }
  return c->fail("unknown field type");
}

void json_escape(const std::string& s, std::string* out) {
  out->push_back('"');
  for (char ch : s) {
    switch (ch) {
      case '"': out->append("\\\""); break;
      case '\\': out->append("\\\\"); break;
      case '\n': out->append("\\n"); break;
      case '\r': out->append("\\r"); break;
      case '\t': out->append("\\t"); break;
      default:
        if ((unsigned char)ch < 0x20) {
          char b[8];
          snprintf(b, sizeof(b), "\\u%04x", ch);
          out->append(b);
        } else {
          out->push_back(ch);
        }
    }
  }
  out->push_back('"');
}

}  // namespace

bool Serialize(const Value& root, std::string* out) {
  if (root.type != Value::OBJECT) return false;
  return serialize_field(root, "", out, 0);
}

bool Parse(const char* data, size_t n, Value* out, std::string* error) {
  Cursor c{data, n, error};
  std::string name;
  if (!parse_field(&c, &name, out, 0)) return false;
  if (out->type != Value::OBJECT) {
    if (error != nullptr) *error = "root is not an object";
    return false;
  }
  return true;
}

void ToJson(const Value& v, std::string* out) {
  char buf[32];
  switch (v.type) {
    case Value::NIL: out->append("null"); break;
    case Value::BOOL: out->append(v.b ? "true" : "false"); break;
    case Value::INT:
      snprintf(buf, sizeof(buf), "%lld", (long long)v.i);
      out->append(buf);
      break;
    case Value::UINT:
      snprintf(buf, sizeof(buf), "%llu", (unsigned long long)v.u);
      out->append(buf);
      break;
    case Value::DOUBLE:
      snprintf(buf, sizeof(buf), "%.17g", v.d);
      out->append(buf);
      break;
    case Value::STRING: json_escape(v.str, out); break;
    case Value::BINARY: json_escape(v.str, out); break;  // raw bytes as-is
    case Value::OBJECT: {
      out->push_back('{');
      bool first = true;
      for (const auto& kv : v.obj) {
        if (!first) out->push_back(',');
        first = false;
        json_escape(kv.first, out);
        out->push_back(':');
        ToJson(kv.second, out);
      }
      out->push_back('}');
      break;
    }
    case Value::ARRAY: {
      out->push_back('[');
      for (size_t i = 0; i < v.arr.size(); ++i) {
        if (i) out->push_back(',');
        ToJson(v.arr[i], out);
      }
      out->push_back(']');
      break;
    }
  }
}

}  // namespace mcpack
}  // namespace bam
