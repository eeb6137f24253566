// brpc_amd: minimal JSON parser/serializer (clean-room; replaces the
// reference's vendored rapidjson for json2pb purposes).
#pragma once

#include <map>
#include <memory>
#include <string>
#include <vector>

namespace bam {
namespace json {

struct Value;
typedef std::vector<Value> Array;
typedef std::map<std::string, Value> Object;

struct Value {
  enum Type { NUL, BOOL, NUMBER, STRING, ARRAY, OBJECT } type = NUL;
  bool b = false;
  double num = 0;
  std::string str;
  std::shared_ptr<Array> arr;
  std::shared_ptr<Object> obj;

  static Value Null() { return Value(); }
  static Value Bool(bool v) {
    Value x;
    x.type = BOOL;
    x.b = v;
    return x;
  }
  static Value Number(double v) {
    Value x;
    x.type = NUMBER;
    x.num = v;
    return x;
  }
  static Value Str(std::string v) {
    Value x;
    x.type = STRING;
    x.str = std::move(v);
    return x;
  }
  static Value MakeArray() {
    Value x;
    x.type = ARRAY;
    x.arr = std::make_shared<Array>();
    return x;
  }
  static Value MakeObject() {
    Value x;
    x.type = OBJECT;
    x.obj = std::make_shared<Object>();
    return x;
  }
};

// Returns false on malformed input.
bool Parse(const std::string& text, Value* out, std::string* error = nullptr);
void Serialize(const Value& v, std::string* out);

}  // namespace json
}  // namespace bam
