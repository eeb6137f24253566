#include "rpc/flv.h"

namespace bam {
namespace flv {

namespace {
void wr_u24(std::string* out, uint32_t v) {
  out->push_back((char)(v >> 16));
  out->push_back((char)(v >> 8));
  out->push_back((char)v);
}
void wr_u32(std::string* out, uint32_t v) {
  out->push_back((char)(v >> 24));
  wr_u24(out, v & 0xffffff);
}
uint32_t rd_u24(const uint8_t* p) {
  return ((uint32_t)p[0] << 16) | ((uint32_t)p[1] << 8) | p[2];
}
uint32_t rd_u32(const uint8_t* p) { return ((uint32_t)p[0] << 24) | rd_u24(p + 1); }
}  // namespace

void AppendHeader(std::string* out, bool has_audio, bool has_video) {
  out->append("FLV", 3);
  out->push_back(1);  // version
  out->push_back((char)((has_audio ? 0x04 : 0) | (has_video ? 0x01 : 0)));
  wr_u32(out, 9);  // data offset
  wr_u32(out, 0);  // PreviousTagSize0
}

void AppendTag(std::string* out, uint8_t type, uint32_t ts, const std::string& payload) {
  out->push_back((char)type);
  wr_u24(out, (uint32_t)payload.size());
  wr_u24(out, ts & 0xffffff);
  out->push_back((char)(ts >> 24));  // extended timestamp byte
  wr_u24(out, 0);                    // stream id
  out->append(payload);
  wr_u32(out, 11 + (uint32_t)payload.size());
}

bool Parse(const std::string& data, std::vector<Tag>* out, bool* has_audio,
           bool* has_video) {
  const uint8_t* p = (const uint8_t*)data.data();
  const uint8_t* end = p + data.size();
  if (end - p < 13 || p[0] != 'F' || p[1] != 'L' || p[2] != 'V' || p[3] != 1) return false;
  if (has_audio != nullptr) *has_audio = (p[4] & 0x04) != 0;
  if (has_video != nullptr) *has_video = (p[4] & 0x01) != 0;
  uint32_t off = rd_u32(p + 5);
  // uint64 sum: off + 4 must not wrap (a 0xffffffff data_offset slipped
  // past the old uint32 check — found by fuzz_ts_flv).
  if (off < 9 || (uint64_t)off + 4 > (uint64_t)(end - p)) return false;
  p += off;
  if (rd_u32(p) != 0) return false;  // PreviousTagSize0
  p += 4;
  while (p < end) {
    if (end - p < 11) return false;
    Tag t;
    t.type = p[0];
    uint32_t size = rd_u24(p + 1);
    t.timestamp = rd_u24(p + 4) | ((uint32_t)p[7] << 24);
    p += 11;
    if ((size_t)(end - p) < size + 4) return false;
    t.payload.assign((const char*)p, size);
    p += size;
    if (rd_u32(p) != 11 + size) return false;
    p += 4;
    out->push_back(std::move(t));
  }
  return true;
}

}  // namespace flv
}  // namespace bam
