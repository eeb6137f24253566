// brpc_amd: FLV container remux (parity: reference brpc/rtmp.cpp FLV
// writer + ts.cpp remux direction — RTMP media messages <-> FLV tags).
// RTMP audio(8)/video(9)/data(18) messages map 1:1 onto FLV tags, so a
// play session remuxes into a standard .flv document (ffmpeg-compatible
// layout: 9-byte header, PreviousTagSize chain, 11-byte tag headers with
// extended timestamps).
#pragma once

#include <stdint.h>

#include <string>
#include <vector>

namespace bam {
namespace flv {

struct Tag {
  uint8_t type = 0;       // 8 audio, 9 video, 18 script data
  uint32_t timestamp = 0; // ms
  std::string payload;
};

// 9-byte header + PreviousTagSize0. flags: audio/video presence bits.
void AppendHeader(std::string* out, bool has_audio = true, bool has_video = true);
// One tag + its PreviousTagSize.
void AppendTag(std::string* out, uint8_t type, uint32_t timestamp_ms,
               const std::string& payload);

// Parses a complete FLV document. Returns false on malformed input.
bool Parse(const std::string& data, std::vector<Tag>* out,
           bool* has_audio = nullptr, bool* has_video = nullptr);

}  // namespace flv
}  // namespace bam
