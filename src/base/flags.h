// brpc_amd: runtime flag registry (parity: gflags usage in the reference +
// butil/reloadable_flags.h). Flags are process-global named values,
// readable/writable at runtime (surfaced rw at the /flags builtin page).
#pragma once

#include <stdint.h>

#include <functional>
#include <map>
#include <mutex>
#include <string>
#include <vector>

namespace bam {
namespace flags {

enum FlagType { FLAG_BOOL, FLAG_INT64, FLAG_DOUBLE, FLAG_STRING };

struct FlagInfo {
  std::string name;
  std::string description;
  FlagType type;
  void* ptr;
  std::function<bool(const std::string&)> validator;  // optional
  std::string default_value;
};

// Registers a flag (called by the BAM_DEFINE_* macros at static init).
int RegisterFlag(const FlagInfo& info);

// Runtime access ("" if unknown).
std::string GetFlagValue(const std::string& name);
// Returns 0 on success; -1 unknown flag; -2 validation failed.
int SetFlagValue(const std::string& name, const std::string& value);
void ListFlags(std::vector<FlagInfo>* out);

}  // namespace flags
}  // namespace bam

#define BAM_DEFINE_FLAG(type_enum, ctype, name, default_val, desc)                       \
  ctype FLAG_##name = default_val;                                                       \
  static int g_flagreg_##name = ::bam::flags::RegisterFlag(                              \
      {#name, desc, ::bam::flags::type_enum, &FLAG_##name, nullptr, #default_val});

#define BAM_DEFINE_bool(name, def, desc) BAM_DEFINE_FLAG(FLAG_BOOL, bool, name, def, desc)
#define BAM_DEFINE_int64(name, def, desc) BAM_DEFINE_FLAG(FLAG_INT64, int64_t, name, def, desc)
#define BAM_DEFINE_double(name, def, desc) BAM_DEFINE_FLAG(FLAG_DOUBLE, double, name, def, desc)
// strings need out-of-line storage
#define BAM_DEFINE_string(name, def, desc)                                               \
  std::string FLAG_##name = def;                                                         \
  static int g_flagreg_##name = ::bam::flags::RegisterFlag(                              \
      {#name, desc, ::bam::flags::FLAG_STRING, &FLAG_##name, nullptr, def});

#define BAM_DECLARE_bool(name) extern bool FLAG_##name;
#define BAM_DECLARE_int64(name) extern int64_t FLAG_##name;
#define BAM_DECLARE_string(name) extern std::string FLAG_##name;
