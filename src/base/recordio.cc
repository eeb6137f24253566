#include "base/recordio.h"

#include <string.h>

#include "base/crc32c.h"

namespace bam {

static const char kMagic[4] = {'R', 'I', 'O', '1'};

RecordWriter::RecordWriter(const std::string& path) { f_ = fopen(path.c_str(), "wb"); }
RecordWriter::~RecordWriter() {
  if (f_ != nullptr) fclose(f_);
}

bool RecordWriter::Write(const std::string& payload) {
  if (f_ == nullptr) return false;
  uint32_t len = (uint32_t)payload.size();
  uint32_t crc = crc32c::Value(payload.data(), payload.size());
  if (fwrite(kMagic, 1, 4, f_) != 4) return false;
  if (fwrite(&len, 4, 1, f_) != 1) return false;
  if (fwrite(&crc, 4, 1, f_) != 1) return false;
  if (len > 0 && fwrite(payload.data(), 1, len, f_) != len) return false;
  return true;
}

void RecordWriter::Flush() {
  if (f_ != nullptr) fflush(f_);
}

RecordReader::RecordReader(const std::string& path) { f_ = fopen(path.c_str(), "rb"); }
RecordReader::~RecordReader() {
  if (f_ != nullptr) fclose(f_);
}

bool RecordReader::Next(std::string* payload) {
  if (f_ == nullptr) return false;
  char magic[4];
  if (fread(magic, 1, 4, f_) != 4) return false;  // EOF
  if (memcmp(magic, kMagic, 4) != 0) {
    err_ = "bad record magic";
    return false;
  }
  uint32_t len = 0, crc = 0;
  if (fread(&len, 4, 1, f_) != 1 || fread(&crc, 4, 1, f_) != 1) {
    err_ = "truncated record header";
    return false;
  }
  if (len > (256u << 20)) {
    err_ = "record too large";
    return false;
  }
  payload->resize(len);
  if (len > 0 && fread(&(*payload)[0], 1, len, f_) != len) {
    err_ = "truncated record payload";
    return false;
  }
  if (crc32c::Value(payload->data(), payload->size()) != crc) {
    err_ = "record crc mismatch";
    return false;
  }
  return true;
}

}  // namespace bam
