// brpc_amd: recordio — length-prefixed, checksummed record files.
// Parity: reference butil/recordio.h (used by rpc_dump / rpc_replay).
// Record: "RIO1" magic + u32 payload_len + u32 crc32c(payload) + payload.
#pragma once

#include <stdio.h>

#include <string>

namespace bam {

class RecordWriter {
 public:
  explicit RecordWriter(const std::string& path);
  ~RecordWriter();
  bool ok() const { return f_ != nullptr; }
  bool Write(const std::string& payload);
  void Flush();

 private:
  FILE* f_;
};

class RecordReader {
 public:
  explicit RecordReader(const std::string& path);
  ~RecordReader();
  bool ok() const { return f_ != nullptr; }
  // false at EOF or on corruption (check last_error()).
  bool Next(std::string* payload);
  const std::string& last_error() const { return err_; }

 private:
  FILE* f_;
  std::string err_;
};

}  // namespace bam
