#include "rpc/ts.h"

#include <string.h>

namespace bam {
namespace ts {

namespace {

constexpr uint16_t kPmtPid = 0x1000;
constexpr uint16_t kVideoPid = 0x100;
constexpr uint16_t kAudioPid = 0x101;
constexpr uint8_t kStreamTypeH264 = 0x1b;
constexpr uint8_t kStreamTypeAac = 0x0f;

// MPEG-2 table CRC32 (poly 0x04C11DB7, init 0xFFFFFFFF, MSB-first, no
// final xor) — distinct from the reflected zlib/crc32c variants.
uint32_t mpeg_crc32(const uint8_t* p, size_t n) {
  uint32_t crc = 0xFFFFFFFFu;
  for (size_t i = 0; i < n; ++i) {
    crc ^= (uint32_t)p[i] << 24;
    for (int b = 0; b < 8; ++b)
      crc = (crc & 0x80000000u) ? (crc << 1) ^ 0x04C11DB7u : (crc << 1);
  }
  return crc;
}

void put16(std::string* s, uint16_t v) {
  s->push_back((char)(v >> 8));
  s->push_back((char)(v & 0xff));
}

// One 188-byte packet: header + optional adaptation + payload slice.
// `pcr90` >= 0 attaches a PCR adaptation field (first packet of a video
// keyframe PES). Stuffing pads short payloads through the adaptation
// field, as the spec requires (0xFF after the flags byte).
// Returns how many payload bytes the packet actually carried.
size_t write_packet(std::string* out, uint16_t pid, bool pusi, uint8_t* cc,
                    const char* payload, size_t len, int64_t pcr90) {
  std::string af;  // adaptation-field body (flags + PCR + stuffing)
  bool have_af = false;
  if (pcr90 >= 0) {
    have_af = true;
    af.push_back(0x10);  // PCR flag
    uint64_t base = (uint64_t)pcr90 & 0x1FFFFFFFFull;
    af.push_back((char)(base >> 25));
    af.push_back((char)(base >> 17));
    af.push_back((char)(base >> 9));
    af.push_back((char)(base >> 1));
    af.push_back((char)(((base & 1) << 7) | 0x7e));  // 6 reserved bits, ext hi
    af.push_back(0);                                 // ext lo
  }
  size_t head = 4 + (have_af ? 1 + af.size() : 0);
  if (len > 188 - head) len = 188 - head;
  size_t deficit = 188 - head - len;
  if (deficit > 0) {
    if (!have_af) {
      have_af = true;
      --deficit;                       // the adaptation_field_length byte
      if (deficit > 0) {
        af.push_back(0x00);            // flags byte (no indicators)
        --deficit;
      }
      // deficit==0 with empty af => af_len 0, a bare length byte (legal).
    }
    af.append(deficit, (char)0xFF);    // stuffing at the end of the AF
  }
  out->push_back(0x47);
  out->push_back((char)((pusi ? 0x40 : 0) | ((pid >> 8) & 0x1f)));
  out->push_back((char)(pid & 0xff));
  uint8_t afc = (uint8_t)((have_af ? 0x20 : 0) | (len > 0 ? 0x10 : 0) | (*cc & 0x0f));
  out->push_back((char)afc);
  if (have_af) {
    out->push_back((char)af.size());
    *out += af;
  }
  out->append(payload, len);
  if (len > 0) *cc = (uint8_t)(*cc + 1);
  return len;
}

// Payload-unit wrapper for PSI sections: pointer_field + section bytes.
void write_section(std::string* out, uint16_t pid, uint8_t* cc,
                   const std::string& section) {
  std::string payload;
  payload.push_back(0);  // pointer_field
  payload += section;
  size_t off = 0;
  bool first = true;
  while (off < payload.size()) {
    off += write_packet(out, pid, first, cc, payload.data() + off,
                        payload.size() - off, -1);
    first = false;
  }
}

std::string finish_section(std::string body) {
  // body starts at table_id; patch section_length then append CRC.
  uint16_t sec_len = (uint16_t)(body.size() - 3 + 4);
  body[1] = (char)(0xB0 | ((sec_len >> 8) & 0x0f));
  body[2] = (char)(sec_len & 0xff);
  uint32_t crc = mpeg_crc32((const uint8_t*)body.data(), body.size());
  body.push_back((char)(crc >> 24));
  body.push_back((char)(crc >> 16));
  body.push_back((char)(crc >> 8));
  body.push_back((char)crc);
  return body;
}

void append_pts(std::string* h, uint8_t lead, int64_t v) {
  h->push_back((char)((lead << 4) | (((v >> 30) & 0x7) << 1) | 1));
  h->push_back((char)((v >> 22) & 0xff));
  h->push_back((char)((((v >> 15) & 0x7f) << 1) | 1));
  h->push_back((char)((v >> 7) & 0xff));
  h->push_back((char)(((v & 0x7f) << 1) | 1));
}

const char kAnnexB[4] = {0, 0, 0, 1};

}  // namespace

void TsMuxer::WriteTables(std::string* out) {
  // PAT: program 1 -> PMT pid.
  std::string pat;
  pat.push_back(0x00);            // table_id
  pat.push_back(0);               // section_length hi (patched)
  pat.push_back(0);               // section_length lo (patched)
  put16(&pat, 1);                 // transport_stream_id
  pat.push_back((char)0xC1);      // version 0, current
  pat.push_back(0);               // section_number
  pat.push_back(0);               // last_section_number
  put16(&pat, 1);                 // program_number
  put16(&pat, (uint16_t)(0xE000 | kPmtPid));
  write_section(out, 0, &cc_[0], finish_section(pat));

  std::string pmt;
  pmt.push_back(0x02);
  pmt.push_back(0);
  pmt.push_back(0);
  put16(&pmt, 1);                 // program_number
  pmt.push_back((char)0xC1);
  pmt.push_back(0);
  pmt.push_back(0);
  put16(&pmt, (uint16_t)(0xE000 | kVideoPid));  // PCR pid
  put16(&pmt, 0xF000);                          // program_info_length 0
  pmt.push_back((char)kStreamTypeH264);
  put16(&pmt, (uint16_t)(0xE000 | kVideoPid));
  put16(&pmt, 0xF000);
  pmt.push_back((char)kStreamTypeAac);
  put16(&pmt, (uint16_t)(0xE000 | kAudioPid));
  put16(&pmt, 0xF000);
  write_section(out, kPmtPid, &cc_[kPmtPid], finish_section(pmt));
}

void TsMuxer::WritePes(uint16_t pid, uint8_t stream_id, bool keyframe,
                       int64_t pts90, int64_t dts90, const std::string& es,
                       std::string* out) {
  std::string pes;
  pes.append("\x00\x00\x01", 3);
  pes.push_back((char)stream_id);
  bool both = dts90 != pts90;
  size_t hdr_len = both ? 10 : 5;
  size_t pkt_len = 3 + hdr_len + es.size();
  put16(&pes, pkt_len <= 0xFFFF ? (uint16_t)pkt_len : 0);  // 0: unbounded (video)
  pes.push_back((char)0x80);
  pes.push_back((char)(both ? 0xC0 : 0x80));
  pes.push_back((char)hdr_len);
  append_pts(&pes, both ? 0x3 : 0x2, pts90);
  if (both) append_pts(&pes, 0x1, dts90);
  pes += es;

  size_t off = 0;
  bool first = true;
  while (off < pes.size()) {
    int64_t pcr = (first && keyframe) ? dts90 : -1;
    off += write_packet(out, pid, first, &cc_[pid], pes.data() + off,
                        pes.size() - off, pcr);
    first = false;
  }
}

bool TsMuxer::OnVideo(const std::string& body, uint32_t ts_ms, std::string* out) {
  // FLV VIDEODATA: frame_type(4)|codec_id(4), AVCPacketType, CompositionTime(s24)
  if (body.size() < 5) return false;
  uint8_t frame_type = ((uint8_t)body[0]) >> 4;
  uint8_t codec = body[0] & 0x0f;
  if (codec != 7) return false;  // AVC only
  uint8_t pkt_type = body[1];
  int32_t ct = ((uint8_t)body[2] << 16) | ((uint8_t)body[3] << 8) | (uint8_t)body[4];
  if (ct & 0x800000) ct |= ~0xFFFFFF;  // sign-extend s24
  const char* p = body.data() + 5;
  size_t n = body.size() - 5;

  if (pkt_type == 0) {
    // AVCDecoderConfigurationRecord
    if (n < 7) return false;
    nalu_len_size_ = (p[4] & 0x3) + 1;
    size_t off = 5;
    int nsps = p[off++] & 0x1f;
    sps_.clear();
    for (int i = 0; i < nsps; ++i) {
      if (off + 2 > n) return false;
      size_t l = ((uint8_t)p[off] << 8) | (uint8_t)p[off + 1];
      off += 2;
      if (off + l > n) return false;
      if (i == 0) sps_.assign(p + off, l);
      off += l;
    }
    if (off >= n) return false;
    int npps = (uint8_t)p[off++];
    pps_.clear();
    for (int i = 0; i < npps; ++i) {
      if (off + 2 > n) return false;
      size_t l = ((uint8_t)p[off] << 8) | (uint8_t)p[off + 1];
      off += 2;
      if (off + l > n) return false;
      if (i == 0) pps_.assign(p + off, l);
      off += l;
    }
    return true;
  }
  if (pkt_type != 1) return true;  // end-of-sequence: nothing to emit

  // AVCC length-prefixed NALUs -> Annex B elementary stream.
  std::string es;
  bool key = frame_type == 1;
  if (key && !sps_.empty()) {
    es.append(kAnnexB, 4);
    es += sps_;
    es.append(kAnnexB, 4);
    es += pps_;
  }
  size_t off = 0;
  while (off + nalu_len_size_ <= n) {
    uint32_t l = 0;
    for (int i = 0; i < nalu_len_size_; ++i) l = (l << 8) | (uint8_t)p[off + i];
    off += nalu_len_size_;
    if (off + l > n) return false;
    es.append(kAnnexB, 4);
    es.append(p + off, l);
    off += l;
  }
  if (es.empty()) return false;
  int64_t dts = (int64_t)ts_ms * 90;
  int64_t pts = dts + (int64_t)ct * 90;
  WritePes(kVideoPid, 0xE0, key, pts, dts, es, out);
  return true;
}

bool TsMuxer::OnAudio(const std::string& body, uint32_t ts_ms, std::string* out) {
  // FLV AUDIODATA: format(4)|rate(2)|size(1)|type(1), AACPacketType
  if (body.size() < 2) return false;
  uint8_t fmt = ((uint8_t)body[0]) >> 4;
  if (fmt != 10) return false;  // AAC only
  if (body[1] == 0) {
    // AudioSpecificConfig: objectType(5) freqIndex(4) channels(4)
    if (body.size() < 4) return false;
    uint16_t asc = ((uint8_t)body[2] << 8) | (uint8_t)body[3];
    audio_object_type_ = (asc >> 11) & 0x1f;
    sample_rate_index_ = (asc >> 7) & 0x0f;
    channels_ = (asc >> 3) & 0x0f;
    return true;
  }
  if (audio_object_type_ == 0) return false;  // no config yet
  size_t raw = body.size() - 2;
  size_t frame_len = raw + 7;
  std::string es;
  es.push_back((char)0xFF);
  es.push_back((char)0xF1);  // MPEG-4, layer 0, no CRC
  es.push_back((char)((((audio_object_type_ - 1) & 0x3) << 6) |
                      ((sample_rate_index_ & 0xf) << 2) | ((channels_ >> 2) & 1)));
  es.push_back((char)(((channels_ & 0x3) << 6) | ((frame_len >> 11) & 0x3)));
  es.push_back((char)((frame_len >> 3) & 0xff));
  es.push_back((char)(((frame_len & 0x7) << 5) | 0x1f));
  es.push_back((char)0xFC);
  es.append(body.data() + 2, raw);
  int64_t pts = (int64_t)ts_ms * 90;
  WritePes(kAudioPid, 0xC0, false, pts, pts, es, out);
  return true;
}

bool TsMuxer::Write(const flv::Tag& tag, std::string* out) {
  if (tag.type == 9) return OnVideo(tag.payload, tag.timestamp, out);
  if (tag.type == 8) return OnAudio(tag.payload, tag.timestamp, out);
  return true;  // script data: no TS representation
}

std::string MakeHlsPlaylist(const std::vector<HlsSegment>& segments,
                            int target_duration_s, int media_sequence, bool ended) {
  std::string m3u8 = "#EXTM3U\n#EXT-X-VERSION:3\n";
  m3u8 += "#EXT-X-TARGETDURATION:" + std::to_string(target_duration_s) + "\n";
  m3u8 += "#EXT-X-MEDIA-SEQUENCE:" + std::to_string(media_sequence) + "\n";
  for (const HlsSegment& s : segments) {
    char buf[64];
    snprintf(buf, sizeof(buf), "#EXTINF:%.3f,\n", s.duration_s);
    m3u8 += buf;
    m3u8 += s.uri + "\n";
  }
  if (ended) m3u8 += "#EXT-X-ENDLIST\n";
  return m3u8;
}

bool FlvToTs(const std::vector<flv::Tag>& tags, std::string* out) {
  TsMuxer mux;
  mux.WriteTables(out);
  bool any = false;
  for (const flv::Tag& t : tags) {
    if (t.type != 8 && t.type != 9) continue;
    std::string chunk;
    if (mux.Write(t, &chunk)) {
      out->append(chunk);
      if (!chunk.empty()) any = true;
    }
  }
  return any;
}

}  // namespace ts
}  // namespace bam
