#include "rpc/json2pb.h"

#include <math.h>
#include <string.h>

#include "base/json.h"
#include "rpc/wire.h"

namespace bam {
namespace json2pb {

namespace {

void encode_field_value(const FieldDesc& f, const json::Value& v, std::string* out);

bool encode_object(const Schema& schema, const json::Value& v, std::string* out,
                   std::string* err) {
  if (v.type != json::Value::OBJECT) {
    if (err) *err = "expected object";
    return false;
  }
  for (const auto& kv : *v.obj) {
    auto it = schema.find(kv.first);
    if (it == schema.end()) continue;  // unknown fields ignored (json2pb behavior)
    const FieldDesc& f = it->second;
    if (f.repeated) {
      if (kv.second.type != json::Value::ARRAY) {
        if (err) *err = "field " + kv.first + " expects array";
        return false;
      }
      for (const json::Value& e : *kv.second.arr) encode_field_value(f, e, out);
    } else {
      encode_field_value(f, kv.second, out);
    }
  }
  return true;
}

void encode_field_value(const FieldDesc& f, const json::Value& v, std::string* out) {
  switch (f.type) {
    case FieldDesc::INT64:
    case FieldDesc::INT32: {
      int64_t i = (int64_t)v.num;
      if (v.type == json::Value::BOOL) i = v.b;
      wire::put_tag(out, f.number, 0);
      wire::put_varint(out, (uint64_t)i);  // two's complement varint
      break;
    }
    case FieldDesc::UINT64:
    case FieldDesc::UINT32:
    case FieldDesc::BOOL: {
      uint64_t u = v.type == json::Value::BOOL ? (uint64_t)v.b : (uint64_t)v.num;
      wire::put_tag(out, f.number, 0);
      wire::put_varint(out, u);
      break;
    }
    case FieldDesc::DOUBLE: {
      wire::put_tag(out, f.number, 1);
      double d = v.num;
      char buf[8];
      memcpy(buf, &d, 8);
      out->append(buf, 8);
      break;
    }
    case FieldDesc::FLOAT: {
      wire::put_tag(out, f.number, 5);
      float fl = (float)v.num;
      char buf[4];
      memcpy(buf, &fl, 4);
      out->append(buf, 4);
      break;
    }
    case FieldDesc::STRING:
    case FieldDesc::BYTES:
      wire::put_str_field(out, f.number, v.str);
      break;
    case FieldDesc::MESSAGE: {
      std::string sub;
      if (f.message_fields != nullptr) {
        encode_object(*f.message_fields, v, &sub, nullptr);
      }
      wire::put_msg_field(out, f.number, sub);
      break;
    }
  }
}

bool decode_object(const Schema& schema, const char* data, size_t n, json::Value* out,
                   std::string* err);

}  // namespace

bool JsonToPb(const Schema& schema, const std::string& json_text, std::string* wire,
              std::string* error) {
  json::Value v;
  if (!json::Parse(json_text, &v, error)) return false;
  wire->clear();
  return encode_object(schema, v, wire, error);
}

bool PbToJson(const Schema& schema, const std::string& data, std::string* json_text,
              std::string* error) {
  // index schema by field number
  std::map<int, std::pair<std::string, const FieldDesc*>> by_num;
  for (const auto& kv : schema) by_num[kv.second.number] = {kv.first, &kv.second};

  json::Value out = json::Value::MakeObject();
  wire::Reader r(data.data(), data.size());
  int wtype;
  for (int fnum; (fnum = r.read_tag(&wtype)) != 0;) {
    auto it = by_num.find(fnum);
    if (it == by_num.end()) {
      r.skip(wtype);
      if (!r.ok()) {
        if (error) *error = "corrupt wire data";
        return false;
      }
      continue;
    }
    const std::string& name = it->second.first;
    const FieldDesc& f = *it->second.second;
    json::Value v;
    switch (f.type) {
      case FieldDesc::INT64:
      case FieldDesc::INT32:
        v = json::Value::Number((double)(int64_t)r.varint());
        break;
      case FieldDesc::UINT64:
      case FieldDesc::UINT32:
        v = json::Value::Number((double)r.varint());
        break;
      case FieldDesc::BOOL:
        v = json::Value::Bool(r.varint() != 0);
        break;
      case FieldDesc::DOUBLE: {
        std::string raw = r.read_fixed(8);
        double d = 0;
        if (raw.size() == 8) memcpy(&d, raw.data(), 8);
        v = json::Value::Number(d);
        break;
      }
      case FieldDesc::FLOAT: {
        std::string raw = r.read_fixed(4);
        float fl = 0;
        if (raw.size() == 4) memcpy(&fl, raw.data(), 4);
        v = json::Value::Number((double)fl);
        break;
      }
      case FieldDesc::STRING:
      case FieldDesc::BYTES:
        v = json::Value::Str(r.read_string());
        break;
      case FieldDesc::MESSAGE: {
        std::string sub = r.read_string();
        if (f.message_fields == nullptr ||
            !decode_object(*f.message_fields, sub.data(), sub.size(), &v, error)) {
          v = json::Value::MakeObject();
        }
        break;
      }
    }
    if (!r.ok()) {
      if (error) *error = "corrupt wire data in field " + name;
      return false;
    }
    if (f.repeated) {
      json::Value& slot = (*out.obj)[name];
      if (slot.type != json::Value::ARRAY) slot = json::Value::MakeArray();
      slot.arr->push_back(std::move(v));
    } else {
      (*out.obj)[name] = std::move(v);
    }
  }
  json_text->clear();
  json::Serialize(out, json_text);
  return true;
}

namespace {
bool decode_object(const Schema& schema, const char* data, size_t n, json::Value* out,
                   std::string* err) {
  std::string text;
  if (!PbToJson(schema, std::string(data, n), &text, err)) return false;
  return json::Parse(text, out, err);
}
}  // namespace

}  // namespace json2pb
}  // namespace bam

namespace bam {
namespace json2pb {

bool JsonToPbByDescriptor(const proto::DescriptorPool& pool,
                          const std::string& message_full_name, const std::string& json_text,
                          std::string* wire, std::string* error) {
  const proto::MessageDef* def = pool.FindMessage(message_full_name);
  if (def == nullptr) {
    if (error != nullptr) *error = "unknown message " + message_full_name;
    return false;
  }
  proto::DynMessage msg(&pool, def);
  std::string err;
  if (!msg.FromJson(json_text, &err)) {
    if (error != nullptr) *error = err;
    return false;
  }
  wire->clear();
  msg.SerializeWire(wire);
  return true;
}

bool PbToJsonByDescriptor(const proto::DescriptorPool& pool,
                          const std::string& message_full_name, const std::string& wire,
                          std::string* json_text, std::string* error) {
  const proto::MessageDef* def = pool.FindMessage(message_full_name);
  if (def == nullptr) {
    if (error != nullptr) *error = "unknown message " + message_full_name;
    return false;
  }
  proto::DynMessage msg(&pool, def);
  if (!msg.ParseWire(wire.data(), wire.size())) {
    if (error != nullptr) *error = "malformed wire bytes";
    return false;
  }
  json_text->clear();
  msg.ToJson(json_text);
  return true;
}

}  // namespace json2pb
}  // namespace bam
