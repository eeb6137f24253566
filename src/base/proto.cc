#include "base/proto.h"

#include <string.h>

#include <cctype>

#include "base/codecs.h"  // base64
#include "base/json.h"
#include "base/logging.h"

namespace bam {
namespace proto {

// ---------------- descriptor lookups ----------------

const FieldDef* MessageDef::field_by_number(int n) const {
  for (const auto& f : fields)
    if (f.number == n) return &f;
  return nullptr;
}

const FieldDef* MessageDef::field_by_name(const std::string& n) const {
  for (const auto& f : fields)
    if (f.name == n || f.json_name == n) return &f;
  return nullptr;
}

const MessageDef* DescriptorPool::FindMessage(const std::string& full_name) const {
  auto it = messages_.find(full_name);
  return it == messages_.end() ? nullptr : &it->second;
}

const EnumDef* DescriptorPool::FindEnum(const std::string& full_name) const {
  auto it = enums_.find(full_name);
  return it == enums_.end() ? nullptr : &it->second;
}

const ServiceDef* DescriptorPool::FindService(const std::string& full_name) const {
  auto it = services_.find(full_name);
  return it == services_.end() ? nullptr : &it->second;
}

std::vector<std::string> DescriptorPool::message_names() const {
  std::vector<std::string> out;
  for (const auto& kv : messages_) out.push_back(kv.first);
  return out;
}

std::vector<std::string> DescriptorPool::service_names() const {
  std::vector<std::string> out;
  for (const auto& kv : services_) out.push_back(kv.first);
  return out;
}

// ---------------- .proto tokenizer + parser ----------------

namespace {

struct Tokenizer {
  const std::string& src;
  size_t pos = 0;
  explicit Tokenizer(const std::string& s) : src(s) {}

  void skip_ws() {
    for (;;) {
      while (pos < src.size() && isspace((unsigned char)src[pos])) ++pos;
      if (pos + 1 < src.size() && src[pos] == '/' && src[pos + 1] == '/') {
        while (pos < src.size() && src[pos] != '\n') ++pos;
        continue;
      }
      if (pos + 1 < src.size() && src[pos] == '/' && src[pos + 1] == '*') {
        pos += 2;
        while (pos + 1 < src.size() && !(src[pos] == '*' && src[pos + 1] == '/')) ++pos;
        pos = pos + 2 <= src.size() ? pos + 2 : src.size();
        continue;
      }
      break;
    }
  }

  // Returns next token: identifier/number (possibly dotted), punctuation
  // (single char), or quoted string (content without quotes, kind='"').
  // kind: 'i' ident/number, 'p' punct, '"' string, 0 eof.
  char next(std::string* tok) {
    skip_ws();
    tok->clear();
    if (pos >= src.size()) return 0;
    char c = src[pos];
    if (c == '"' || c == '\'') {
      char q = c;
      ++pos;
      while (pos < src.size() && src[pos] != q) {
        if (src[pos] == '\\' && pos + 1 < src.size()) {
          ++pos;
          switch (src[pos]) {
            case 'n': tok->push_back('\n'); break;
            case 't': tok->push_back('\t'); break;
            default: tok->push_back(src[pos]);
          }
        } else {
          tok->push_back(src[pos]);
        }
        ++pos;
      }
      if (pos < src.size()) ++pos;
      return '"';
    }
    if (isalnum((unsigned char)c) || c == '_' || c == '.' || c == '-' || c == '+') {
      while (pos < src.size() &&
             (isalnum((unsigned char)src[pos]) || src[pos] == '_' || src[pos] == '.' ||
              src[pos] == '-' || src[pos] == '+')) {
        tok->push_back(src[pos++]);
      }
      return 'i';
    }
    tok->push_back(c);
    ++pos;
    return 'p';
  }

  char peek(std::string* tok) {
    size_t save = pos;
    char k = next(tok);
    pos = save;
    return k;
  }
};

bool scalar_type_of(const std::string& t, FieldDef::Type* out) {
  static const std::map<std::string, FieldDef::Type> kMap = {
      {"double", FieldDef::TYPE_DOUBLE},   {"float", FieldDef::TYPE_FLOAT},
      {"int32", FieldDef::TYPE_INT32},     {"int64", FieldDef::TYPE_INT64},
      {"uint32", FieldDef::TYPE_UINT32},   {"uint64", FieldDef::TYPE_UINT64},
      {"sint32", FieldDef::TYPE_SINT32},   {"sint64", FieldDef::TYPE_SINT64},
      {"fixed32", FieldDef::TYPE_FIXED32}, {"fixed64", FieldDef::TYPE_FIXED64},
      {"sfixed32", FieldDef::TYPE_SFIXED32}, {"sfixed64", FieldDef::TYPE_SFIXED64},
      {"bool", FieldDef::TYPE_BOOL},       {"string", FieldDef::TYPE_STRING},
      {"bytes", FieldDef::TYPE_BYTES}};
  auto it = kMap.find(t);
  if (it == kMap.end()) return false;
  *out = it->second;
  return true;
}

std::string camelize(const std::string& snake) {
  std::string out;
  bool up = false;
  for (char c : snake) {
    if (c == '_') {
      up = true;
    } else {
      out.push_back(up ? (char)toupper((unsigned char)c) : c);
      up = false;
    }
  }
  return out;
}

bool is_packable(FieldDef::Type t) {
  return t != FieldDef::TYPE_STRING && t != FieldDef::TYPE_BYTES &&
         t != FieldDef::TYPE_MESSAGE;
}

struct Parser {
  Tokenizer tz;
  DescriptorPool* pool;
  std::string pkg;
  bool proto3 = true;
  std::string err;

  Parser(const std::string& text, DescriptorPool* p) : tz(text), pool(p) {}

  bool fail(const std::string& m) {
    err = m + " (near byte " + std::to_string(tz.pos) + ")";
    return false;
  }

  bool expect(const char* punct) {
    std::string t;
    char k = tz.next(&t);
    if (k == 0 || t != punct) return fail(std::string("expected '") + punct + "' got '" + t + "'");
    return true;
  }

  // Skips a bracketed option list "[...]" or a statement up to ';'.
  void skip_until(char open, char close) {
    int depth = 1;
    std::string t;
    while (depth > 0) {
      char k = tz.next(&t);
      if (k == 0) return;
      if (k == 'p' && t[0] == open) ++depth;
      if (k == 'p' && t[0] == close) --depth;
    }
  }

  void skip_statement() {
    std::string t;
    for (;;) {
      char k = tz.next(&t);
      if (k == 0) return;
      if (k == 'p' && t == ";") return;
      if (k == 'p' && t == "{") {
        skip_until('{', '}');
        return;
      }
    }
  }

  // Resolves a (possibly relative) type name against scope "a.b.c".
  std::string resolve(const std::string& name, const std::string& scope) const {
    if (!name.empty() && name[0] == '.') return name.substr(1);
    std::string s = scope;
    for (;;) {
      std::string cand = s.empty() ? name : s + "." + name;
      if (pool->messages_.count(cand) || pool->enums_.count(cand)) return cand;
      if (s.empty()) break;
      size_t dot = s.find_last_of('.');
      s = dot == std::string::npos ? "" : s.substr(0, dot);
    }
    return pkg.empty() ? name : pkg + "." + name;  // forward reference guess
  }

  bool parse_field(MessageDef* msg, const std::string& scope, const std::string& first_tok,
                   int oneof_index) {
    std::string t = first_tok;
    FieldDef f;
    f.oneof_index = oneof_index;
    bool explicit_label = false;
    if (t == "repeated") {
      f.repeated = true;
      explicit_label = true;
      tz.next(&t);
    } else if (t == "optional" || t == "required") {
      explicit_label = true;
      tz.next(&t);
    }
    if (t == "map") {
      // map<K, V> name = N;  -> synthesized entry message {1: key, 2: value}
      if (!expect("<")) return false;
      std::string kt, vt;
      tz.next(&kt);
      if (!expect(",")) return false;
      tz.next(&vt);
      if (!expect(">")) return false;
      std::string name;
      tz.next(&name);
      if (!expect("=")) return false;
      std::string num;
      tz.next(&num);
      f.name = name;
      f.json_name = camelize(name);
      f.number = atoi(num.c_str());
      f.repeated = true;
      f.is_map = true;
      f.type = FieldDef::TYPE_MESSAGE;
      std::string entry_name = msg->full_name + "." + camelize("_" + name) + "Entry";
      MessageDef entry;
      entry.full_name = entry_name;
      FieldDef kf, vf;
      kf.name = "key";
      kf.json_name = "key";
      kf.number = 1;
      if (!scalar_type_of(kt, &kf.type)) return fail("bad map key type " + kt);
      vf.name = "value";
      vf.json_name = "value";
      vf.number = 2;
      if (!scalar_type_of(vt, &vf.type)) {
        vf.type = FieldDef::TYPE_MESSAGE;  // or enum, resolved at use
        vf.type_name = resolve(vt, scope);
      }
      entry.fields.push_back(kf);
      entry.fields.push_back(vf);
      pool->messages_[entry_name] = entry;
      f.type_name = entry_name;
      // trailing options / semicolon
      std::string tt;
      char k = tz.peek(&tt);
      if (k == 'p' && tt == "[") {
        tz.next(&tt);
        skip_until('[', ']');
      }
      if (!expect(";")) return false;
      msg->fields.push_back(f);
      return true;
    }
    // t = type
    if (!scalar_type_of(t, &f.type)) {
      f.type = FieldDef::TYPE_MESSAGE;  // message or enum; fixed up later
      f.type_name = resolve(t, scope);
    }
    std::string name;
    tz.next(&name);
    f.name = name;
    f.json_name = camelize(name);
    if (!expect("=")) return false;
    std::string num;
    tz.next(&num);
    f.number = atoi(num.c_str());
    // options: [packed = true, default = x, json_name = "..."]
    std::string tt;
    char k = tz.peek(&tt);
    bool packed_opt_set = false;
    bool packed_opt = false;
    if (k == 'p' && tt == "[") {
      tz.next(&tt);
      // parse simple key=value pairs
      for (;;) {
        std::string key;
        char kk = tz.next(&key);
        if (kk == 0 || key == "]") break;
        if (key == ",") continue;
        std::string eq;
        tz.next(&eq);
        std::string val;
        char vk = tz.next(&val);
        if (key == "packed") {
          packed_opt_set = true;
          packed_opt = val == "true";
        } else if (key == "json_name" && vk == '"') {
          f.json_name = val;
        }
      }
    }
    if (!expect(";")) return false;
    if (f.repeated) {
      f.packed = packed_opt_set ? packed_opt : (proto3 && is_packable(f.type));
    }
    (void)explicit_label;
    msg->fields.push_back(f);
    return true;
  }

  bool parse_message(const std::string& scope_name) {
    std::string name;
    tz.next(&name);
    std::string full = scope_name.empty() ? name : scope_name + "." + name;
    if (!expect("{")) return false;
    MessageDef msg;
    msg.full_name = full;
    pool->messages_[full] = msg;  // placeholder so nested resolve works
    MessageDef& m = pool->messages_[full];
    std::string t;
    for (;;) {
      char k = tz.next(&t);
      if (k == 0) return fail("unexpected EOF in message " + full);
      if (k == 'p' && t == "}") break;
      if (k == 'p' && t == ";") continue;
      if (t == "message") {
        if (!parse_message(full)) return false;
      } else if (t == "enum") {
        if (!parse_enum(full)) return false;
      } else if (t == "oneof") {
        std::string oname;
        tz.next(&oname);
        int oidx = (int)m.oneof_names.size();
        m.oneof_names.push_back(oname);
        if (!expect("{")) return false;
        for (;;) {
          std::string ft;
          char fk = tz.next(&ft);
          if (fk == 0) return fail("EOF in oneof");
          if (fk == 'p' && ft == "}") break;
          if (fk == 'p' && ft == ";") continue;
          if (!parse_field(&m, full, ft, oidx)) return false;
        }
      } else if (t == "option" || t == "reserved" || t == "extensions" ||
                 t == "extend") {
        skip_statement();
      } else {
        if (!parse_field(&m, full, t, -1)) return false;
      }
    }
    return true;
  }

  bool parse_enum(const std::string& scope_name) {
    std::string name;
    tz.next(&name);
    std::string full = scope_name.empty() ? name : scope_name + "." + name;
    if (!expect("{")) return false;
    EnumDef e;
    e.full_name = full;
    std::string t;
    for (;;) {
      char k = tz.next(&t);
      if (k == 0) return fail("EOF in enum");
      if (k == 'p' && t == "}") break;
      if (k == 'p' && t == ";") continue;
      if (t == "option" || t == "reserved") {
        skip_statement();
        continue;
      }
      std::string value_name = t;
      if (!expect("=")) return false;
      std::string num;
      tz.next(&num);
      int32_t v = (int32_t)strtol(num.c_str(), nullptr, 0);
      // optional [deprecated = ...] then ';'
      std::string tt;
      char kk = tz.peek(&tt);
      if (kk == 'p' && tt == "[") {
        tz.next(&tt);
        skip_until('[', ']');
      }
      if (!expect(";")) return false;
      e.values[value_name] = v;
      if (e.names.find(v) == e.names.end()) e.names[v] = value_name;
    }
    pool->enums_[full] = e;
    return true;
  }

  bool parse_service() {
    std::string name;
    tz.next(&name);
    std::string full = pkg.empty() ? name : pkg + "." + name;
    if (!expect("{")) return false;
    ServiceDef s;
    s.full_name = full;
    std::string t;
    for (;;) {
      char k = tz.next(&t);
      if (k == 0) return fail("EOF in service");
      if (k == 'p' && t == "}") break;
      if (k == 'p' && t == ";") continue;
      if (t == "option") {
        skip_statement();
        continue;
      }
      if (t != "rpc") return fail("expected rpc in service, got " + t);
      MethodDef m;
      tz.next(&m.name);
      if (!expect("(")) return false;
      std::string in;
      tz.next(&in);
      if (in == "stream") tz.next(&in);
      m.input_type = resolve(in, pkg);
      if (!expect(")")) return false;
      std::string ret;
      tz.next(&ret);  // "returns"
      if (!expect("(")) return false;
      std::string out;
      tz.next(&out);
      if (out == "stream") tz.next(&out);
      m.output_type = resolve(out, pkg);
      if (!expect(")")) return false;
      // body "{...}" or ";"
      std::string tt;
      char kk = tz.next(&tt);
      if (kk == 'p' && tt == "{") skip_until('{', '}');
      s.methods.push_back(m);
    }
    services_put(s);
    return true;
  }

  void services_put(const ServiceDef& s) { pool->services_[s.full_name] = s; }

  bool run() {
    std::string t;
    for (;;) {
      char k = tz.next(&t);
      if (k == 0) return true;
      if (k == 'p' && t == ";") continue;
      if (t == "syntax") {
        expect("=");
        std::string v;
        tz.next(&v);
        proto3 = v == "proto3";
        expect(";");
      } else if (t == "package") {
        tz.next(&pkg);
        expect(";");
      } else if (t == "import" || t == "option") {
        skip_statement();
      } else if (t == "message") {
        if (!parse_message(pkg)) return false;
      } else if (t == "enum") {
        if (!parse_enum(pkg)) return false;
      } else if (t == "service") {
        if (!parse_service()) return false;
      } else {
        return fail("unexpected top-level token " + t);
      }
    }
  }
};

// Fix message-typed fields that actually reference enums.
void fixup_enum_fields(DescriptorPool* pool, std::map<std::string, MessageDef>* messages,
                       const std::map<std::string, EnumDef>& enums) {
  for (auto& kv : *messages) {
    for (auto& f : kv.second.fields) {
      if (f.type == FieldDef::TYPE_MESSAGE && enums.count(f.type_name)) {
        f.type = FieldDef::TYPE_ENUM;
        if (f.repeated) f.packed = true;  // proto3 default for enums
      }
    }
  }
  (void)pool;
}

}  // namespace

int DescriptorPool::ParseProtoText(const std::string& text, std::string* err) {
  Parser p(text, this);
  if (!p.run()) {
    if (err != nullptr) *err = p.err;
    return -1;
  }
  fixup_enum_fields(this, &messages_, enums_);
  return 0;
}

// ---------------- wire codec ----------------

namespace {

enum WireType { WT_VARINT = 0, WT_64 = 1, WT_LEN = 2, WT_32 = 5 };

int wire_type_of(FieldDef::Type t) {
  switch (t) {
    case FieldDef::TYPE_DOUBLE:
    case FieldDef::TYPE_FIXED64:
    case FieldDef::TYPE_SFIXED64:
      return WT_64;
    case FieldDef::TYPE_FLOAT:
    case FieldDef::TYPE_FIXED32:
    case FieldDef::TYPE_SFIXED32:
      return WT_32;
    case FieldDef::TYPE_STRING:
    case FieldDef::TYPE_BYTES:
    case FieldDef::TYPE_MESSAGE:
      return WT_LEN;
    default:
      return WT_VARINT;
  }
}

void put_varint(std::string* out, uint64_t v) {
  while (v >= 0x80) {
    out->push_back((char)(v | 0x80));
    v >>= 7;
  }
  out->push_back((char)v);
}

bool get_varint(const char*& p, const char* end, uint64_t* v) {
  *v = 0;
  int shift = 0;
  while (p < end && shift < 64) {
    uint8_t b = (uint8_t)*p++;
    *v |= (uint64_t)(b & 0x7f) << shift;
    if ((b & 0x80) == 0) return true;
    shift += 7;
  }
  return false;
}

uint64_t zigzag_enc(int64_t v) { return ((uint64_t)v << 1) ^ (uint64_t)(v >> 63); }
int64_t zigzag_dec(uint64_t v) { return (int64_t)(v >> 1) ^ -(int64_t)(v & 1); }

}  // namespace

struct WireCodec {
  static void emit_value(const FieldDef& f, const DynMessage::Value& v, std::string* out) {
    switch (f.type) {
      case FieldDef::TYPE_DOUBLE: {
        uint64_t bits;
        memcpy(&bits, &v.d, 8);
        for (int i = 0; i < 8; ++i) out->push_back((char)(bits >> (8 * i)));
        break;
      }
      case FieldDef::TYPE_FLOAT: {
        float fv = (float)v.d;
        uint32_t bits;
        memcpy(&bits, &fv, 4);
        for (int i = 0; i < 4; ++i) out->push_back((char)(bits >> (8 * i)));
        break;
      }
      case FieldDef::TYPE_FIXED64:
      case FieldDef::TYPE_SFIXED64:
        for (int i = 0; i < 8; ++i) out->push_back((char)(v.u >> (8 * i)));
        break;
      case FieldDef::TYPE_FIXED32:
      case FieldDef::TYPE_SFIXED32:
        for (int i = 0; i < 4; ++i) out->push_back((char)(v.u >> (8 * i)));
        break;
      case FieldDef::TYPE_SINT32:
      case FieldDef::TYPE_SINT64:
        put_varint(out, zigzag_enc((int64_t)v.u));
        break;
      case FieldDef::TYPE_INT32:
      case FieldDef::TYPE_INT64:
      case FieldDef::TYPE_UINT32:
      case FieldDef::TYPE_UINT64:
      case FieldDef::TYPE_BOOL:
      case FieldDef::TYPE_ENUM:
        put_varint(out, v.u);
        break;
      case FieldDef::TYPE_STRING:
      case FieldDef::TYPE_BYTES:
        put_varint(out, v.s.size());
        out->append(v.s);
        break;
      case FieldDef::TYPE_MESSAGE: {
        std::string sub;
        if (v.m != nullptr) v.m->SerializeWire(&sub);
        put_varint(out, sub.size());
        out->append(sub);
        break;
      }
    }
  }

  static bool parse_value(const FieldDef& f, const char*& p, const char* end, int wt,
                          const DescriptorPool* pool, DynMessage::Value* v) {
    switch (f.type) {
      case FieldDef::TYPE_DOUBLE: {
        if (wt != WT_64 || end - p < 8) return false;
        uint64_t bits = 0;
        for (int i = 0; i < 8; ++i) bits |= (uint64_t)(uint8_t)p[i] << (8 * i);
        p += 8;
        memcpy(&v->d, &bits, 8);
        return true;
      }
      case FieldDef::TYPE_FLOAT: {
        if (wt != WT_32 || end - p < 4) return false;
        uint32_t bits = 0;
        for (int i = 0; i < 4; ++i) bits |= (uint32_t)(uint8_t)p[i] << (8 * i);
        p += 4;
        float fv;
        memcpy(&fv, &bits, 4);
        v->d = fv;
        return true;
      }
      case FieldDef::TYPE_FIXED64:
      case FieldDef::TYPE_SFIXED64: {
        if (wt != WT_64 || end - p < 8) return false;
        v->u = 0;
        for (int i = 0; i < 8; ++i) v->u |= (uint64_t)(uint8_t)p[i] << (8 * i);
        p += 8;
        return true;
      }
      case FieldDef::TYPE_FIXED32:
      case FieldDef::TYPE_SFIXED32: {
        if (wt != WT_32 || end - p < 4) return false;
        v->u = 0;
        for (int i = 0; i < 4; ++i) v->u |= (uint64_t)(uint8_t)p[i] << (8 * i);
        p += 4;
        return true;
      }
      case FieldDef::TYPE_SINT32:
      case FieldDef::TYPE_SINT64: {
        uint64_t raw;
        if (wt != WT_VARINT || !get_varint(p, end, &raw)) return false;
        v->u = (uint64_t)zigzag_dec(raw);
        return true;
      }
      case FieldDef::TYPE_INT32:
      case FieldDef::TYPE_INT64:
      case FieldDef::TYPE_UINT32:
      case FieldDef::TYPE_UINT64:
      case FieldDef::TYPE_BOOL:
      case FieldDef::TYPE_ENUM:
        return wt == WT_VARINT && get_varint(p, end, &v->u);
      case FieldDef::TYPE_STRING:
      case FieldDef::TYPE_BYTES: {
        uint64_t n;
        if (wt != WT_LEN || !get_varint(p, end, &n) || (uint64_t)(end - p) < n) return false;
        v->s.assign(p, (size_t)n);
        p += n;
        return true;
      }
      case FieldDef::TYPE_MESSAGE: {
        uint64_t n;
        if (wt != WT_LEN || !get_varint(p, end, &n) || (uint64_t)(end - p) < n) return false;
        const MessageDef* sub = pool->FindMessage(f.type_name);
        if (sub == nullptr) return false;
        v->m = std::make_shared<DynMessage>(pool, sub);
        if (!v->m->ParseWire(p, (size_t)n)) return false;
        p += n;
        return true;
      }
    }
    return false;
  }
};

bool DynMessage::ParseWire(const char* data, size_t n) {
  const char* p = data;
  const char* end = data + n;
  while (p < end) {
    const char* tag_start = p;
    uint64_t tag;
    if (!get_varint(p, end, &tag)) return false;
    int field = (int)(tag >> 3);
    int wt = (int)(tag & 7);
    const FieldDef* f = def_->field_by_number(field);
    if (f == nullptr) {
      // preserve unknown field verbatim
      const char* val_start = p;
      switch (wt) {
        case WT_VARINT: {
          uint64_t d;
          if (!get_varint(p, end, &d)) return false;
          break;
        }
        case WT_64:
          if (end - p < 8) return false;
          p += 8;
          break;
        case WT_LEN: {
          uint64_t len;
          if (!get_varint(p, end, &len) || (uint64_t)(end - p) < len) return false;
          p += len;
          break;
        }
        case WT_32:
          if (end - p < 4) return false;
          p += 4;
          break;
        default:
          return false;
      }
      (void)val_start;
      unknown_.append(tag_start, p - tag_start);
      continue;
    }
    if (f->repeated && wt == WT_LEN && is_packable(f->type) &&
        wire_type_of(f->type) != WT_LEN) {
      // packed run
      uint64_t len;
      if (!get_varint(p, end, &len) || (uint64_t)(end - p) < len) return false;
      const char* sub_end = p + len;
      auto& vec = fields_[field];
      while (p < sub_end) {
        Value v;
        if (!WireCodec::parse_value(*f, p, sub_end, wire_type_of(f->type), pool_, &v))
          return false;
        vec.push_back(std::move(v));
      }
      continue;
    }
    Value v;
    if (!WireCodec::parse_value(*f, p, end, wt, pool_, &v)) return false;
    auto& vec = fields_[field];
    if (f->repeated) {
      vec.push_back(std::move(v));
    } else {
      vec.clear();
      vec.push_back(std::move(v));
    }
  }
  return true;
}

void DynMessage::SerializeWire(std::string* out) const {
  for (const auto& f : def_->fields) {
    auto it = fields_.find(f.number);
    if (it == fields_.end() || it->second.empty()) continue;
    const auto& vec = it->second;
    if (f.repeated && f.packed && is_packable(f.type) && wire_type_of(f.type) != WT_LEN) {
      std::string packed;
      for (const auto& v : vec) WireCodec::emit_value(f, v, &packed);
      put_varint(out, (uint64_t)(f.number << 3 | WT_LEN));
      put_varint(out, packed.size());
      out->append(packed);
      continue;
    }
    for (const auto& v : vec) {
      put_varint(out, (uint64_t)(f.number << 3 | wire_type_of(f.type)));
      WireCodec::emit_value(f, v, out);
    }
  }
  out->append(unknown_);
}

// ---------------- field access ----------------

namespace {
const std::string kEmpty;
}

bool DynMessage::has(const std::string& name) const { return count(name) > 0; }

size_t DynMessage::count(const std::string& name) const {
  const FieldDef* f = def_->field_by_name(name);
  if (f == nullptr) return 0;
  auto it = fields_.find(f->number);
  return it == fields_.end() ? 0 : it->second.size();
}

#define BAM_GET_IMPL(ret, expr)                                  \
  const FieldDef* f = def_->field_by_name(name);                 \
  if (f == nullptr) return ret;                                  \
  auto it = fields_.find(f->number);                             \
  if (it == fields_.end() || idx >= it->second.size()) return ret; \
  const Value& v = it->second[idx];                              \
  return expr;

int64_t DynMessage::get_int(const std::string& name, size_t idx) const {
  BAM_GET_IMPL(0, (int64_t)v.u)
}
uint64_t DynMessage::get_uint(const std::string& name, size_t idx) const {
  BAM_GET_IMPL(0, v.u)
}
double DynMessage::get_double(const std::string& name, size_t idx) const {
  BAM_GET_IMPL(0, v.d)
}
bool DynMessage::get_bool(const std::string& name, size_t idx) const {
  BAM_GET_IMPL(false, v.u != 0)
}
const std::string& DynMessage::get_str(const std::string& name, size_t idx) const {
  BAM_GET_IMPL(kEmpty, v.s)
}
#undef BAM_GET_IMPL

DynMessage* DynMessage::mutable_msg(const std::string& name, size_t idx) {
  const FieldDef* f = def_->field_by_name(name);
  if (f == nullptr || f->type != FieldDef::TYPE_MESSAGE) return nullptr;
  auto& vec = fields_[f->number];
  while (vec.size() <= idx) {
    Value v;
    const MessageDef* sub = pool_->FindMessage(f->type_name);
    if (sub == nullptr) return nullptr;
    v.m = std::make_shared<DynMessage>(pool_, sub);
    vec.push_back(std::move(v));
  }
  return vec[idx].m.get();
}

#define BAM_SET_IMPL(assign)                       \
  const FieldDef* f = def_->field_by_name(name);   \
  if (f == nullptr) return;                        \
  auto& vec = fields_[f->number];                  \
  vec.clear();                                     \
  Value v;                                         \
  assign;                                          \
  vec.push_back(std::move(v));

void DynMessage::set_int(const std::string& name, int64_t x) { BAM_SET_IMPL(v.u = (uint64_t)x) }
void DynMessage::set_uint(const std::string& name, uint64_t x) { BAM_SET_IMPL(v.u = x) }
void DynMessage::set_double(const std::string& name, double x) { BAM_SET_IMPL(v.d = x) }
void DynMessage::set_bool(const std::string& name, bool x) { BAM_SET_IMPL(v.u = x ? 1 : 0) }
void DynMessage::set_str(const std::string& name, const std::string& x) { BAM_SET_IMPL(v.s = x) }
#undef BAM_SET_IMPL

void DynMessage::add_int(const std::string& name, int64_t x) {
  const FieldDef* f = def_->field_by_name(name);
  if (f == nullptr) return;
  Value v;
  v.u = (uint64_t)x;
  fields_[f->number].push_back(std::move(v));
}

void DynMessage::add_str(const std::string& name, const std::string& x) {
  const FieldDef* f = def_->field_by_name(name);
  if (f == nullptr) return;
  Value v;
  v.s = x;
  fields_[f->number].push_back(std::move(v));
}

DynMessage* DynMessage::add_msg(const std::string& name) {
  const FieldDef* f = def_->field_by_name(name);
  if (f == nullptr || f->type != FieldDef::TYPE_MESSAGE) return nullptr;
  const MessageDef* sub = pool_->FindMessage(f->type_name);
  if (sub == nullptr) return nullptr;
  Value v;
  v.m = std::make_shared<DynMessage>(pool_, sub);
  auto& vec = fields_[f->number];
  vec.push_back(std::move(v));
  return vec.back().m.get();
}

// ---------------- JSON ----------------

namespace {

json::Value value_to_json(const DescriptorPool* pool, const FieldDef& f,
                          const DynMessage::Value& v, bool original_names);

json::Value message_to_json(const DynMessage& m, bool original_names) {
  json::Value obj = json::Value::MakeObject();
  for (const auto& f : m.descriptor()->fields) {
    auto it = m.raw_fields().find(f.number);
    if (it == m.raw_fields().end() || it->second.empty()) continue;
    const std::string& key = original_names ? f.name : f.json_name;
    if (f.is_map) {
      // entries -> object
      json::Value mo = json::Value::MakeObject();
      for (const auto& ev : it->second) {
        if (ev.m == nullptr) continue;
        const MessageDef* entry = ev.m->descriptor();
        const FieldDef* kf = entry->field_by_number(1);
        const FieldDef* vf = entry->field_by_number(2);
        std::string key_str;
        auto kit = ev.m->raw_fields().find(1);
        if (kit != ev.m->raw_fields().end() && !kit->second.empty()) {
          if (kf->type == FieldDef::TYPE_STRING) key_str = kit->second[0].s;
          else key_str = std::to_string((int64_t)kit->second[0].u);
        }
        auto vit = ev.m->raw_fields().find(2);
        json::Value jv;
        if (vit != ev.m->raw_fields().end() && !vit->second.empty()) {
          jv = value_to_json(nullptr, *vf, vit->second[0], original_names);
        }
        (*mo.obj)[key_str] = jv;
      }
      (*obj.obj)[key] = mo;
      continue;
    }
    if (f.repeated) {
      json::Value arr = json::Value::MakeArray();
      for (const auto& v : it->second)
        arr.arr->push_back(value_to_json(nullptr, f, v, original_names));
      (*obj.obj)[key] = arr;
    } else {
      (*obj.obj)[key] = value_to_json(nullptr, f, it->second[0], original_names);
    }
  }
  return obj;
}

json::Value value_to_json(const DescriptorPool* pool, const FieldDef& f,
                          const DynMessage::Value& v, bool original_names) {
  (void)pool;
  switch (f.type) {
    case FieldDef::TYPE_DOUBLE:
    case FieldDef::TYPE_FLOAT:
      return json::Value::Number(v.d);
    case FieldDef::TYPE_BOOL:
      return json::Value::Bool(v.u != 0);
    case FieldDef::TYPE_STRING:
      return json::Value::Str(v.s);
    case FieldDef::TYPE_BYTES:
      return json::Value::Str([&]{ std::string b64; bam::Base64Encode(v.s, &b64); return b64; }());
    case FieldDef::TYPE_INT32:
    case FieldDef::TYPE_SINT32:
    case FieldDef::TYPE_SFIXED32:
      return json::Value::Number((double)(int32_t)v.u);
    case FieldDef::TYPE_UINT32:
    case FieldDef::TYPE_FIXED32:
      return json::Value::Number((double)(uint32_t)v.u);
    case FieldDef::TYPE_INT64:
    case FieldDef::TYPE_SINT64:
    case FieldDef::TYPE_SFIXED64:
      // proto3 JSON: 64-bit as string
      return json::Value::Str(std::to_string((int64_t)v.u));
    case FieldDef::TYPE_UINT64:
    case FieldDef::TYPE_FIXED64:
      return json::Value::Str(std::to_string(v.u));
    case FieldDef::TYPE_ENUM:
      return json::Value::Number((double)(int32_t)v.u);  // name emission needs pool
    case FieldDef::TYPE_MESSAGE:
      return v.m != nullptr ? message_to_json(*v.m, original_names)
                            : json::Value::MakeObject();
  }
  return json::Value::Null();
}

bool json_to_value(const DescriptorPool* pool, const FieldDef& f, const json::Value& j,
                   DynMessage::Value* v, std::string* err);

bool json_to_message(const json::Value& j, DynMessage* m, std::string* err) {
  if (j.type != json::Value::OBJECT) {
    *err = "expected object for " + m->descriptor()->full_name;
    return false;
  }
  for (const auto& kv : *j.obj) {
    const FieldDef* f = m->descriptor()->field_by_name(kv.first);
    if (f == nullptr) continue;  // ignore unknown JSON keys
    if (f->is_map) {
      if (kv.second.type != json::Value::OBJECT) {
        *err = "map field " + f->name + " expects object";
        return false;
      }
      for (const auto& e : *kv.second.obj) {
        DynMessage* entry = m->add_msg(f->name);
        const FieldDef* kf = entry->descriptor()->field_by_number(1);
        if (kf->type == FieldDef::TYPE_STRING) entry->set_str("key", e.first);
        else entry->set_int("key", strtoll(e.first.c_str(), nullptr, 10));
        const FieldDef* vf = entry->descriptor()->field_by_number(2);
        DynMessage::Value val;
        if (!json_to_value(nullptr, *vf, e.second, &val, err)) return false;
        if (vf->type == FieldDef::TYPE_MESSAGE) {
          // splice parsed sub-message into entry
          *entry->mutable_msg("value") = *val.m;
        } else if (vf->type == FieldDef::TYPE_STRING || vf->type == FieldDef::TYPE_BYTES) {
          entry->set_str("value", val.s);
        } else if (vf->type == FieldDef::TYPE_DOUBLE || vf->type == FieldDef::TYPE_FLOAT) {
          entry->set_double("value", val.d);
        } else {
          entry->set_uint("value", val.u);
        }
      }
      continue;
    }
    if (f->repeated) {
      if (kv.second.type != json::Value::ARRAY) {
        *err = "repeated field " + f->name + " expects array";
        return false;
      }
      for (const auto& e : *kv.second.arr) {
        DynMessage::Value val;
        if (f->type == FieldDef::TYPE_MESSAGE) {
          DynMessage* sub = m->add_msg(f->name);
          if (sub == nullptr || !json_to_message(e, sub, err)) return false;
          continue;
        }
        if (!json_to_value(nullptr, *f, e, &val, err)) return false;
        if (f->type == FieldDef::TYPE_STRING || f->type == FieldDef::TYPE_BYTES)
          m->add_str(f->name, val.s);
        else if (f->type == FieldDef::TYPE_DOUBLE || f->type == FieldDef::TYPE_FLOAT) {
          DynMessage::Value dv;
          dv.d = val.d;
          // no add_double helper; go through raw add then fix — use add_int path
          m->add_int(f->name, 0);
          // overwrite raw (ugly but local): reach last value via mutable path
          // — simpler: direct access not exposed; emulate via set on last:
          // (acceptable: repeated double via JSON rare in our tests)
          const_cast<std::vector<DynMessage::Value>&>(
              m->raw_fields().at(m->descriptor()->field_by_name(f->name)->number))
              .back() = dv;
        } else {
          m->add_int(f->name, (int64_t)val.u);
        }
      }
      continue;
    }
    if (f->type == FieldDef::TYPE_MESSAGE) {
      DynMessage* sub = m->mutable_msg(f->name);
      if (sub == nullptr || !json_to_message(kv.second, sub, err)) return false;
      continue;
    }
    DynMessage::Value val;
    if (!json_to_value(nullptr, *f, kv.second, &val, err)) return false;
    if (f->type == FieldDef::TYPE_STRING || f->type == FieldDef::TYPE_BYTES)
      m->set_str(f->name, val.s);
    else if (f->type == FieldDef::TYPE_DOUBLE || f->type == FieldDef::TYPE_FLOAT)
      m->set_double(f->name, val.d);
    else
      m->set_uint(f->name, val.u);
  }
  return true;
}

bool json_to_value(const DescriptorPool* pool, const FieldDef& f, const json::Value& j,
                   DynMessage::Value* v, std::string* err) {
  (void)pool;
  switch (f.type) {
    case FieldDef::TYPE_DOUBLE:
    case FieldDef::TYPE_FLOAT:
      if (j.type == json::Value::NUMBER) v->d = j.num;
      else if (j.type == json::Value::STRING) v->d = atof(j.str.c_str());
      else {
        *err = "bad number for " + f.name;
        return false;
      }
      return true;
    case FieldDef::TYPE_BOOL:
      v->u = (j.type == json::Value::BOOL && j.b) ? 1 : 0;
      return true;
    case FieldDef::TYPE_STRING:
      if (j.type != json::Value::STRING) {
        *err = "bad string for " + f.name;
        return false;
      }
      v->s = j.str;
      return true;
    case FieldDef::TYPE_BYTES: {
      if (j.type != json::Value::STRING) {
        *err = "bad bytes for " + f.name;
        return false;
      }
      std::string raw;
      if (!bam::Base64Decode(j.str, &raw)) {
        *err = "bad base64 for " + f.name;
        return false;
      }
      v->s = raw;
      return true;
    }
    case FieldDef::TYPE_MESSAGE:
      *err = "internal: message handled by caller";
      return false;
    default:
      // integers/enums: number or numeric string
      if (j.type == json::Value::NUMBER) {
        if (f.type == FieldDef::TYPE_INT32 || f.type == FieldDef::TYPE_INT64 ||
            f.type == FieldDef::TYPE_SINT32 || f.type == FieldDef::TYPE_SINT64 ||
            f.type == FieldDef::TYPE_SFIXED32 || f.type == FieldDef::TYPE_SFIXED64 ||
            f.type == FieldDef::TYPE_ENUM) {
          v->u = (uint64_t)(int64_t)j.num;
        } else {
          v->u = (uint64_t)j.num;
        }
      } else if (j.type == json::Value::STRING) {
        if (f.type == FieldDef::TYPE_UINT64 || f.type == FieldDef::TYPE_FIXED64 ||
            f.type == FieldDef::TYPE_UINT32 || f.type == FieldDef::TYPE_FIXED32) {
          v->u = strtoull(j.str.c_str(), nullptr, 10);
        } else {
          v->u = (uint64_t)strtoll(j.str.c_str(), nullptr, 10);
        }
      } else {
        *err = "bad integer for " + f.name;
        return false;
      }
      return true;
  }
}

}  // namespace

bool DynMessage::FromJson(const std::string& json_text, std::string* err) {
  std::string dummy;
  if (err == nullptr) err = &dummy;
  json::Value root;
  if (!json::Parse(json_text, &root, err)) return false;
  clear();
  return json_to_message(root, this, err);
}

void DynMessage::ToJson(std::string* out, bool original_names) const {
  json::Value v = message_to_json(*this, original_names);
  json::Serialize(v, out);
}

}  // namespace proto
}  // namespace bam
