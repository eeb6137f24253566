// brpc_amd: error-code + message value type (parity: butil/status.h).
#pragma once

#include <string>

namespace bam {

class Status {
 public:
  Status() : code_(0) {}
  Status(int code, std::string msg) : code_(code), msg_(std::move(msg)) {}

  static Status OK() { return Status(); }

  bool ok() const { return code_ == 0; }
  int error_code() const { return code_; }
  const std::string& error_str() const { return msg_; }
  const char* error_cstr() const { return msg_.c_str(); }

  void reset() {
    code_ = 0;
    msg_.clear();
  }
  void set_error(int code, const std::string& msg) {
    code_ = code;
    msg_ = msg;
  }

 private:
  int code_;
  std::string msg_;
};

}  // namespace bam
