// brpc_amd: MPEG-TS muxer + HLS playlist (parity: reference brpc/ts.cpp
// TsWriter — the HLS leg of the media-server stack; FLV tags remux into
// 188-byte transport-stream packets playable as .ts / HLS segments).
//
// Wire layout written here (all public MPEG-2 Part 1 structure):
//   PAT (pid 0)  -> program 1 -> PMT pid 0x1000
//   PMT          -> H.264 (stream_type 0x1b) pid 0x100 (PCR pid),
//                   AAC ADTS (stream_type 0x0f) pid 0x101
//   PES packets with PTS(/DTS) in 90 kHz units; video keyframes start a
//   new payload-unit with an adaptation-field PCR.
// Video input is FLV AVC payloads (AVCC length-prefixed NALUs +
// AVCDecoderConfigurationRecord sequence header) converted to Annex B
// with SPS/PPS re-injected before each keyframe; audio input is FLV AAC
// payloads (AudioSpecificConfig sequence header + raw frames) wrapped in
// ADTS headers.
#pragma once

#include <stdint.h>

#include <string>
#include <vector>

#include "rpc/flv.h"

namespace bam {
namespace ts {

class TsMuxer {
 public:
  // Feeds one FLV tag (audio 8 / video 9; script tags are ignored).
  // Returns false on malformed payloads (kept: the stream continues).
  bool Write(const flv::Tag& tag, std::string* out);

  // Emits PAT+PMT (call at stream start and at each HLS segment start).
  void WriteTables(std::string* out);

  bool has_video_config() const { return !sps_.empty(); }
  bool has_audio_config() const { return audio_object_type_ != 0; }

 private:
  void WritePes(uint16_t pid, uint8_t stream_id, bool keyframe, int64_t pts90,
                int64_t dts90, const std::string& es, std::string* out);
  bool OnVideo(const std::string& body, uint32_t ts_ms, std::string* out);
  bool OnAudio(const std::string& body, uint32_t ts_ms, std::string* out);

  // AVCDecoderConfigurationRecord state
  std::string sps_, pps_;
  int nalu_len_size_ = 4;
  // AudioSpecificConfig state
  int audio_object_type_ = 0, sample_rate_index_ = 0, channels_ = 0;
  uint8_t cc_[8192] = {0};  // continuity counters per pid
};

// HLS media playlist (#EXTM3U ... #EXT-X-ENDLIST) over segment metadata.
struct HlsSegment {
  std::string uri;
  double duration_s = 0;
};
std::string MakeHlsPlaylist(const std::vector<HlsSegment>& segments,
                            int target_duration_s, int media_sequence = 0,
                            bool ended = true);

// Remuxes a parsed FLV document into one .ts blob (tables first, then
// every a/v tag). Returns false if no tag could be muxed.
bool FlvToTs(const std::vector<flv::Tag>& tags, std::string* out);

}  // namespace ts
}  // namespace bam
