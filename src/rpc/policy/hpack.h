// brpc_amd: HPACK (RFC 7541) header compression, in-tree.
// Parity: reference details/hpack.cpp (910 LoC incl. Huffman) — round 1
// delegated h2 header coding to a dlopened libnghttp2; this removes that
// runtime dependency. Full static table, dynamic tables on both sides
// with size updates/eviction, integer prefix coding, and canonical
// Huffman string coding (tests/test_hpack.py extracts the code table from
// libnghttp2 as an ORACLE and cross-checks ours entry by entry).
#pragma once

#include <stdint.h>

#include <deque>
#include <string>
#include <utility>
#include <vector>

namespace bam {
namespace hpack {

typedef std::pair<std::string, std::string> Header;

// ---- Huffman (RFC 7541 Appendix B) ----
// Encodes/decodes raw octets; decode returns false on bad padding/EOS.
void HuffmanEncode(const std::string& in, std::string* out);
bool HuffmanDecode(const char* in, size_t n, std::string* out);
size_t HuffmanEncodedLength(const std::string& in);
// For the oracle test: (code, nbits) for symbols 0..256 (256 = EOS).
void HuffmanTable(std::vector<std::pair<uint32_t, int>>* out);

// ---- integer prefix coding ----
void EncodeInt(std::string* out, uint64_t value, int prefix_bits, uint8_t flags);
bool DecodeInt(const uint8_t*& p, const uint8_t* end, int prefix_bits, uint64_t* value);

// ---- encoder / decoder with dynamic tables ----

class Encoder {
 public:
  explicit Encoder(size_t max_table_size = 4096) : max_size_(max_table_size) {}
  // Appends the encoded header block for `headers` to *out. Indexes into
  // the static + dynamic tables; Huffman-codes literals when shorter.
  void Encode(const std::vector<Header>& headers, std::string* out);

 private:
  int find(const Header& h, bool* name_only) const;
  void add_dynamic(const Header& h);
  size_t max_size_;
  size_t size_ = 0;
  std::deque<Header> dynamic_;  // front = most recent (index 62)
};

class Decoder {
 public:
  explicit Decoder(size_t max_table_size = 4096) : max_size_(max_table_size) {}
  // Decodes one complete header block. false on malformed input.
  bool Decode(const char* data, size_t n, std::vector<Header>* out);

 private:
  bool lookup(uint64_t index, Header* h) const;
  void add_dynamic(const Header& h);
  size_t max_size_;
  size_t size_ = 0;
  std::deque<Header> dynamic_;
};

}  // namespace hpack
}  // namespace bam
