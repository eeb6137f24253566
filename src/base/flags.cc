#include "base/flags.h"

#include <stdlib.h>

namespace bam {
namespace flags {

namespace {
struct Registry {
  std::mutex mu;
  std::map<std::string, FlagInfo> flags;
};
Registry& registry() {
  static Registry* r = new Registry;
  return *r;
}
}  // namespace

int RegisterFlag(const FlagInfo& info) {
  Registry& r = registry();
  std::lock_guard<std::mutex> lk(r.mu);
  r.flags[info.name] = info;
  return 0;
}

std::string GetFlagValue(const std::string& name) {
  Registry& r = registry();
  std::lock_guard<std::mutex> lk(r.mu);
  auto it = r.flags.find(name);
  if (it == r.flags.end()) return "";
  const FlagInfo& f = it->second;
  switch (f.type) {
    case FLAG_BOOL:
      return *(bool*)f.ptr ? "true" : "false";
    case FLAG_INT64:
      return std::to_string(*(int64_t*)f.ptr);
    case FLAG_DOUBLE:
      return std::to_string(*(double*)f.ptr);
    case FLAG_STRING:
      return *(std::string*)f.ptr;
  }
  return "";
}

int SetFlagValue(const std::string& name, const std::string& value) {
  Registry& r = registry();
  std::lock_guard<std::mutex> lk(r.mu);
  auto it = r.flags.find(name);
  if (it == r.flags.end()) return -1;
  FlagInfo& f = it->second;
  if (f.validator && !f.validator(value)) return -2;
  switch (f.type) {
    case FLAG_BOOL:
      *(bool*)f.ptr = (value == "true" || value == "1");
      break;
    case FLAG_INT64:
      *(int64_t*)f.ptr = strtoll(value.c_str(), nullptr, 10);
      break;
    case FLAG_DOUBLE:
      *(double*)f.ptr = strtod(value.c_str(), nullptr);
      break;
    case FLAG_STRING:
      *(std::string*)f.ptr = value;
      break;
  }
  return 0;
}

void ListFlags(std::vector<FlagInfo>* out) {
  Registry& r = registry();
  std::lock_guard<std::mutex> lk(r.mu);
  out->clear();
  for (const auto& kv : r.flags) out->push_back(kv.second);
}

}  // namespace flags
}  // namespace bam
