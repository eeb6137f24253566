#include "base/json.h"

#include <math.h>
#include <string.h>
#include <stdio.h>
#include <stdlib.h>

namespace bam {
namespace json {

namespace {

struct Parser {
  const char* p;
  const char* end;
  std::string err;

  void skip_ws() {
    while (p < end && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r')) ++p;
  }

  bool fail(const std::string& what) {
    err = what;
    return false;
  }

  bool parse_value(Value* out, int depth) {
    if (depth > 100) return fail("nesting too deep");
    skip_ws();
    if (p >= end) return fail("unexpected end");
    switch (*p) {
      case '{':
        return parse_object(out, depth);
      case '[':
        return parse_array(out, depth);
      case '"': {
        out->type = Value::STRING;
        return parse_string(&out->str);
      }
      case 't':
        if (end - p >= 4 && strncmp(p, "true", 4) == 0) {
          *out = Value::Bool(true);
          p += 4;
          return true;
        }
        return fail("bad literal");
      case 'f':
        if (end - p >= 5 && strncmp(p, "false", 5) == 0) {
          *out = Value::Bool(false);
          p += 5;
          return true;
        }
        return fail("bad literal");
      case 'n':
        if (end - p >= 4 && strncmp(p, "null", 4) == 0) {
          *out = Value::Null();
          p += 4;
          return true;
        }
        return fail("bad literal");
      default: {
        char* num_end = nullptr;
        double v = strtod(p, &num_end);
        if (num_end == p) return fail("bad number");
        *out = Value::Number(v);
        p = num_end;
        return true;
      }
    }
  }

  bool parse_string(std::string* out) {
    if (*p != '"') return fail("expected string");
    ++p;
    out->clear();
    while (p < end) {
      char c = *p++;
      if (c == '"') return true;
      if (c == '\\') {
        if (p >= end) return fail("bad escape");
        char e = *p++;
        switch (e) {
          case '"': out->push_back('"'); break;
          case '\\': out->push_back('\\'); break;
          case '/': out->push_back('/'); break;
          case 'b': out->push_back('\b'); break;
          case 'f': out->push_back('\f'); break;
          case 'n': out->push_back('\n'); break;
          case 'r': out->push_back('\r'); break;
          case 't': out->push_back('\t'); break;
          case 'u': {
            if (end - p < 4) return fail("bad \\u");
            char hex[5] = {p[0], p[1], p[2], p[3], 0};
            unsigned int cp = (unsigned int)strtoul(hex, nullptr, 16);
            p += 4;
            // UTF-8 encode (BMP only; surrogate pairs combined)
            if (cp >= 0xD800 && cp <= 0xDBFF && end - p >= 6 && p[0] == '\\' && p[1] == 'u') {
              char hex2[5] = {p[2], p[3], p[4], p[5], 0};
              unsigned int lo = (unsigned int)strtoul(hex2, nullptr, 16);
              if (lo >= 0xDC00 && lo <= 0xDFFF) {
                cp = 0x10000 + ((cp - 0xD800) << 10) + (lo - 0xDC00);
                p += 6;
              }
            }
            if (cp < 0x80) {
              out->push_back((char)cp);
            } else if (cp < 0x800) {
              out->push_back((char)(0xC0 | (cp >> 6)));
              out->push_back((char)(0x80 | (cp & 0x3F)));
            } else if (cp < 0x10000) {
              out->push_back((char)(0xE0 | (cp >> 12)));
              out->push_back((char)(0x80 | ((cp >> 6) & 0x3F)));
              out->push_back((char)(0x80 | (cp & 0x3F)));
            } else {
              out->push_back((char)(0xF0 | (cp >> 18)));
              out->push_back((char)(0x80 | ((cp >> 12) & 0x3F)));
              out->push_back((char)(0x80 | ((cp >> 6) & 0x3F)));
              out->push_back((char)(0x80 | (cp & 0x3F)));
            }
            break;
          }
          default:
            return fail("bad escape char");
        }
      } else {
        out->push_back(c);
      }
    }
    return fail("unterminated string");
  }

  bool parse_object(Value* out, int depth) {
    ++p;  // {
    *out = Value::MakeObject();
    skip_ws();
    if (p < end && *p == '}') {
      ++p;
      return true;
    }
    for (;;) {
      skip_ws();
      std::string key;
      if (!parse_string(&key)) return false;
      skip_ws();
      if (p >= end || *p != ':') return fail("expected :");
      ++p;
      Value v;
      if (!parse_value(&v, depth + 1)) return false;
      (*out->obj)[key] = std::move(v);
      skip_ws();
      if (p >= end) return fail("unterminated object");
      if (*p == ',') {
        ++p;
        continue;
      }
      if (*p == '}') {
        ++p;
        return true;
      }
      return fail("expected , or }");
    }
  }

  bool parse_array(Value* out, int depth) {
    ++p;  // [
    *out = Value::MakeArray();
    skip_ws();
    if (p < end && *p == ']') {
      ++p;
      return true;
    }
    for (;;) {
      Value v;
      if (!parse_value(&v, depth + 1)) return false;
      out->arr->push_back(std::move(v));
      skip_ws();
      if (p >= end) return fail("unterminated array");
      if (*p == ',') {
        ++p;
        continue;
      }
      if (*p == ']') {
        ++p;
        return true;
      }
      return fail("expected , or ]");
    }
  }
};

void escape_to(const std::string& s, std::string* out) {
  out->push_back('"');
  for (char c : s) {
    switch (c) {
      case '"': out->append("\\\""); break;
      case '\\': out->append("\\\\"); break;
      case '\n': out->append("\\n"); break;
      case '\r': out->append("\\r"); break;
      case '\t': out->append("\\t"); break;
      case '\b': out->append("\\b"); break;
      case '\f': out->append("\\f"); break;
      default:
        if ((unsigned char)c < 0x20) {
          char buf[8];
          snprintf(buf, sizeof(buf), "\\u%04x", c);
          out->append(buf);
        } else {
          out->push_back(c);
        }
    }
  }
  out->push_back('"');
}

}  // namespace

bool Parse(const std::string& text, Value* out, std::string* error) {
  Parser parser{text.data(), text.data() + text.size(), ""};
  bool ok = parser.parse_value(out, 0);
  if (ok) {
    parser.skip_ws();
    if (parser.p != parser.end) {
      ok = false;
      parser.err = "trailing characters";
    }
  }
  if (!ok && error != nullptr) *error = parser.err;
  return ok;
}

void Serialize(const Value& v, std::string* out) {
  switch (v.type) {
    case Value::NUL:
      out->append("null");
      break;
    case Value::BOOL:
      out->append(v.b ? "true" : "false");
      break;
    case Value::NUMBER: {
      char buf[32];
      if (v.num == (double)(long long)v.num && fabs(v.num) < 1e15) {
        snprintf(buf, sizeof(buf), "%lld", (long long)v.num);
      } else {
        snprintf(buf, sizeof(buf), "%.17g", v.num);
      }
      out->append(buf);
      break;
    }
    case Value::STRING:
      escape_to(v.str, out);
      break;
    case Value::ARRAY: {
      out->push_back('[');
      bool first = true;
      for (const Value& e : *v.arr) {
        if (!first) out->push_back(',');
        first = false;
        Serialize(e, out);
      }
      out->push_back(']');
      break;
    }
    case Value::OBJECT: {
      out->push_back('{');
      bool first = true;
      for (const auto& kv : *v.obj) {
        if (!first) out->push_back(',');
        first = false;
        escape_to(kv.first, out);
        out->push_back(':');
        Serialize(kv.second, out);
      }
      out->push_back('}');
      break;
    }
  }
}

}  // namespace json
}  // namespace bam
