#include "rpc/policy/hpack.h"

#include <string.h>

namespace bam {
namespace hpack {

// ---------------- Huffman code table (RFC 7541 Appendix B) ----------------

namespace {

struct HuffCode {
  uint32_t code;
  uint8_t bits;
};

// Canonical table, symbols 0..255 + EOS (256). Verified entry-by-entry
// against libnghttp2's encoder by tests/test_hpack.py.
const HuffCode kHuff[257] = {
    {0x1ff8, 13},     {0x7fffd8, 23},   {0xfffffe2, 28},  {0xfffffe3, 28},
    {0xfffffe4, 28},  {0xfffffe5, 28},  {0xfffffe6, 28},  {0xfffffe7, 28},
    {0xfffffe8, 28},  {0xffffea, 24},   {0x3ffffffc, 30}, {0xfffffe9, 28},
    {0xfffffea, 28},  {0x3ffffffd, 30}, {0xfffffeb, 28},  {0xfffffec, 28},
    {0xfffffed, 28},  {0xfffffee, 28},  {0xfffffef, 28},  {0xffffff0, 28},
    {0xffffff1, 28},  {0xffffff2, 28},  {0x3ffffffe, 30}, {0xffffff3, 28},
    {0xffffff4, 28},  {0xffffff5, 28},  {0xffffff6, 28},  {0xffffff7, 28},
    {0xffffff8, 28},  {0xffffff9, 28},  {0xffffffa, 28},  {0xffffffb, 28},
    {0x14, 6},        {0x3f8, 10},      {0x3f9, 10},      {0xffa, 12},
    {0x1ff9, 13},     {0x15, 6},        {0xf8, 8},        {0x7fa, 11},
    {0x3fa, 10},      {0x3fb, 10},      {0xf9, 8},        {0x7fb, 11},
    {0xfa, 8},        {0x16, 6},        {0x17, 6},        {0x18, 6},
    {0x0, 5},         {0x1, 5},         {0x2, 5},         {0x19, 6},
    {0x1a, 6},        {0x1b, 6},        {0x1c, 6},        {0x1d, 6},
    {0x1e, 6},        {0x1f, 6},        {0x5c, 7},        {0xfb, 8},
    {0x7ffc, 15},     {0x20, 6},        {0xffb, 12},      {0x3fc, 10},
    {0x1ffa, 13},     {0x21, 6},        {0x5d, 7},        {0x5e, 7},
    {0x5f, 7},        {0x60, 7},        {0x61, 7},        {0x62, 7},
    {0x63, 7},        {0x64, 7},        {0x65, 7},        {0x66, 7},
    {0x67, 7},        {0x68, 7},        {0x69, 7},        {0x6a, 7},
    {0x6b, 7},        {0x6c, 7},        {0x6d, 7},        {0x6e, 7},
    {0x6f, 7},        {0x70, 7},        {0x71, 7},        {0x72, 7},
    {0xfc, 8},        {0x73, 7},        {0xfd, 8},        {0x1ffb, 13},
    {0x7fff0, 19},    {0x1ffc, 13},     {0x3ffc, 14},     {0x22, 6},
    {0x7ffd, 15},     {0x3, 5},         {0x23, 6},        {0x4, 5},
    {0x24, 6},        {0x5, 5},         {0x25, 6},        {0x26, 6},
    {0x27, 6},        {0x6, 5},         {0x74, 7},        {0x75, 7},
    {0x28, 6},        {0x29, 6},        {0x2a, 6},        {0x7, 5},
    {0x2b, 6},        {0x76, 7},        {0x2c, 6},        {0x8, 5},
    {0x9, 5},         {0x2d, 6},        {0x77, 7},        {0x78, 7},
    {0x79, 7},        {0x7a, 7},        {0x7b, 7},        {0x7ffe, 15},
    {0x7fc, 11},      {0x3ffd, 14},     {0x1ffd, 13},     {0xffffffc, 28},
    {0xfffe6, 20},    {0x3fffd2, 22},   {0xfffe7, 20},    {0xfffe8, 20},
    {0x3fffd3, 22},   {0x3fffd4, 22},   {0x3fffd5, 22},   {0x7fffd9, 23},
    {0x3fffd6, 22},   {0x7fffda, 23},   {0x7fffdb, 23},   {0x7fffdc, 23},
    {0x7fffdd, 23},   {0x7fffde, 23},   {0xffffeb, 24},   {0x7fffdf, 23},
    {0xffffec, 24},   {0xffffed, 24},   {0x3fffd7, 22},   {0x7fffe0, 23},
    {0xffffee, 24},   {0x7fffe1, 23},   {0x7fffe2, 23},   {0x7fffe3, 23},
    {0x7fffe4, 23},   {0x1fffdc, 21},   {0x3fffd8, 22},   {0x7fffe5, 23},
    {0x3fffd9, 22},   {0x7fffe6, 23},   {0x7fffe7, 23},   {0xffffef, 24},
    {0x3fffda, 22},   {0x1fffdd, 21},   {0xfffe9, 20},    {0x3fffdb, 22},
    {0x3fffdc, 22},   {0x7fffe8, 23},   {0x7fffe9, 23},   {0x1fffde, 21},
    {0x7fffea, 23},   {0x3fffdd, 22},   {0x3fffde, 22},   {0xfffff0, 24},
    {0x1fffdf, 21},   {0x3fffdf, 22},   {0x7fffeb, 23},   {0x7fffec, 23},
    {0x1fffe0, 21},   {0x1fffe1, 21},   {0x3fffe0, 22},   {0x1fffe2, 21},
    {0x7fffed, 23},   {0x3fffe1, 22},   {0x7fffee, 23},   {0x7fffef, 23},
    {0xfffea, 20},    {0x3fffe2, 22},   {0x3fffe3, 22},   {0x3fffe4, 22},
    {0x7ffff0, 23},   {0x3fffe5, 22},   {0x3fffe6, 22},   {0x7ffff1, 23},
    {0x3ffffe0, 26},  {0x3ffffe1, 26},  {0xfffeb, 20},    {0x7fff1, 19},
    {0x3fffe7, 22},   {0x7ffff2, 23},   {0x3fffe8, 22},   {0x1ffffec, 25},
    {0x3ffffe2, 26},  {0x3ffffe3, 26},  {0x3ffffe4, 26},  {0x7ffffde, 27},
    {0x7ffffdf, 27},  {0x3ffffe5, 26},  {0xfffff1, 24},   {0x1ffffed, 25},
    {0x7fff2, 19},    {0x1fffe3, 21},   {0x3ffffe6, 26},  {0x7ffffe0, 27},
    {0x7ffffe1, 27},  {0x3ffffe7, 26},  {0x7ffffe2, 27},  {0xfffff2, 24},
    {0x1fffe4, 21},   {0x1fffe5, 21},   {0x3ffffe8, 26},  {0x3ffffe9, 26},
    {0xffffffd, 28},  {0x7ffffe3, 27},  {0x7ffffe4, 27},  {0x7ffffe5, 27},
    {0xfffec, 20},    {0xfffff3, 24},   {0xfffed, 20},    {0x1fffe6, 21},
    {0x3fffe9, 22},   {0x1fffe7, 21},   {0x1fffe8, 21},   {0x7ffff3, 23},
    {0x3fffea, 22},   {0x3fffeb, 22},   {0x1ffffee, 25},  {0x1ffffef, 25},
    {0xfffff4, 24},   {0xfffff5, 24},   {0x3ffffea, 26},  {0x7ffff4, 23},
    {0x3ffffeb, 26},  {0x7ffffe6, 27},  {0x3ffffec, 26},  {0x3ffffed, 26},
    {0x7ffffe7, 27},  {0x7ffffe8, 27},  {0x7ffffe9, 27},  {0x7ffffea, 27},
    {0x7ffffeb, 27},  {0xffffffe, 28},  {0x7ffffec, 27},  {0x7ffffed, 27},
    {0x7ffffee, 27},  {0x7ffffef, 27},  {0x7fffff0, 27},  {0x3ffffee, 26},
    {0x3fffffff, 30},
};

// Decode tree built once: nodes of (left, right) indices; leaves hold the
// symbol. Bit-by-bit walk — correctness first (h2 headers are small).
struct HuffNode {
  int child[2] = {-1, -1};
  int symbol = -1;
};

std::vector<HuffNode>& huff_tree() {
  static std::vector<HuffNode>* tree = [] {
    auto* t = new std::vector<HuffNode>;
    t->emplace_back();
    for (int sym = 0; sym < 257; ++sym) {
      uint32_t code = kHuff[sym].code;
      int bits = kHuff[sym].bits;
      int node = 0;
      for (int b = bits - 1; b >= 0; --b) {
        int bit = (code >> b) & 1;
        if ((*t)[node].child[bit] < 0) {
          (*t)[node].child[bit] = (int)t->size();
          t->emplace_back();
        }
        node = (*t)[node].child[bit];
      }
      (*t)[node].symbol = sym;
    }
    return t;
  }();
  return *tree;
}

}  // namespace

size_t HuffmanEncodedLength(const std::string& in) {
  uint64_t bits = 0;
  for (unsigned char c : in) bits += kHuff[c].bits;
  return (size_t)((bits + 7) / 8);
}

void HuffmanEncode(const std::string& in, std::string* out) {
  uint64_t acc = 0;
  int nbits = 0;
  for (unsigned char c : in) {
    acc = (acc << kHuff[c].bits) | kHuff[c].code;
    nbits += kHuff[c].bits;
    while (nbits >= 8) {
      nbits -= 8;
      out->push_back((char)(acc >> nbits));
    }
  }
  if (nbits > 0) {
    // pad with EOS prefix (all 1s)
    out->push_back((char)((acc << (8 - nbits)) | (0xff >> nbits)));
  }
}

bool HuffmanDecode(const char* in, size_t n, std::string* out) {
  const auto& tree = huff_tree();
  int node = 0;
  int depth = 0;  // bits since last symbol (for padding validation)
  bool all_ones = true;
  for (size_t i = 0; i < n; ++i) {
    uint8_t byte = (uint8_t)in[i];
    for (int b = 7; b >= 0; --b) {
      int bit = (byte >> b) & 1;
      if (bit == 0) all_ones = false;
      node = tree[node].child[bit];
      if (node < 0) return false;
      ++depth;
      if (tree[node].symbol >= 0) {
        if (tree[node].symbol == 256) return false;  // explicit EOS is an error
        out->push_back((char)tree[node].symbol);
        node = 0;
        depth = 0;
        all_ones = true;
      }
    }
  }
  // Remaining bits must be a ≤7-bit prefix of EOS (all ones).
  return depth <= 7 && all_ones;
}

void HuffmanTable(std::vector<std::pair<uint32_t, int>>* out) {
  out->clear();
  for (int i = 0; i < 257; ++i) out->push_back({kHuff[i].code, kHuff[i].bits});
}

// ---------------- integer prefix coding ----------------

void EncodeInt(std::string* out, uint64_t value, int prefix_bits, uint8_t flags) {
  const uint64_t limit = (1ull << prefix_bits) - 1;
  if (value < limit) {
    out->push_back((char)(flags | value));
    return;
  }
  out->push_back((char)(flags | limit));
  value -= limit;
  while (value >= 0x80) {
    out->push_back((char)(value | 0x80));
    value >>= 7;
  }
  out->push_back((char)value);
}

bool DecodeInt(const uint8_t*& p, const uint8_t* end, int prefix_bits, uint64_t* value) {
  if (p >= end) return false;
  const uint64_t limit = (1ull << prefix_bits) - 1;
  *value = *p++ & limit;
  if (*value < limit) return true;
  int shift = 0;
  while (p < end && shift <= 56) {
    uint8_t b = *p++;
    *value += (uint64_t)(b & 0x7f) << shift;
    if ((b & 0x80) == 0) return true;
    shift += 7;
  }
  return false;
}

// ---------------- static table (RFC 7541 Appendix A) ----------------

namespace {

const Header kStatic[61] = {
    {":authority", ""},
    {":method", "GET"},
    {":method", "POST"},
    {":path", "/"},
    {":path", "/index.html"},
    {":scheme", "http"},
    {":scheme", "https"},
    {":status", "200"},
    {":status", "204"},
    {":status", "206"},
    {":status", "304"},
    {":status", "400"},
    {":status", "404"},
    {":status", "500"},
    {"accept-charset", ""},
    {"accept-encoding", "gzip, deflate"},
    {"accept-language", ""},
    {"accept-ranges", ""},
    {"accept", ""},
    {"access-control-allow-origin", ""},
    {"age", ""},
    {"allow", ""},
    {"authorization", ""},
    {"cache-control", ""},
    {"content-disposition", ""},
    {"content-encoding", ""},
    {"content-language", ""},
    {"content-length", ""},
    {"content-location", ""},
    {"content-range", ""},
    {"content-type", ""},
    {"cookie", ""},
    {"date", ""},
    {"etag", ""},
    {"expect", ""},
    {"expires", ""},
    {"from", ""},
    {"host", ""},
    {"if-match", ""},
    {"if-modified-since", ""},
    {"if-none-match", ""},
    {"if-range", ""},
    {"if-unmodified-since", ""},
    {"last-modified", ""},
    {"link", ""},
    {"location", ""},
    {"max-forwards", ""},
    {"proxy-authenticate", ""},
    {"proxy-authorization", ""},
    {"range", ""},
    {"referer", ""},
    {"refresh", ""},
    {"retry-after", ""},
    {"server", ""},
    {"set-cookie", ""},
    {"strict-transport-security", ""},
    {"transfer-encoding", ""},
    {"user-agent", ""},
    {"vary", ""},
    {"via", ""},
    {"www-authenticate", ""},
};

size_t entry_size(const Header& h) { return h.first.size() + h.second.size() + 32; }

void encode_string(std::string* out, const std::string& s) {
  size_t hlen = HuffmanEncodedLength(s);
  if (hlen < s.size()) {
    EncodeInt(out, hlen, 7, 0x80);
    HuffmanEncode(s, out);
  } else {
    EncodeInt(out, s.size(), 7, 0x00);
    out->append(s);
  }
}

bool decode_string(const uint8_t*& p, const uint8_t* end, std::string* out) {
  if (p >= end) return false;
  const bool huff = (*p & 0x80) != 0;
  uint64_t len;
  if (!DecodeInt(p, end, 7, &len)) return false;
  if ((uint64_t)(end - p) < len) return false;
  if (huff) {
    if (!HuffmanDecode((const char*)p, (size_t)len, out)) return false;
  } else {
    out->assign((const char*)p, (size_t)len);
  }
  p += len;
  return true;
}

}  // namespace

// ---------------- encoder ----------------

int Encoder::find(const Header& h, bool* name_only) const {
  int name_idx = 0;
  for (int i = 0; i < 61; ++i) {
    if (kStatic[i].first == h.first) {
      if (kStatic[i].second == h.second) {
        *name_only = false;
        return i + 1;
      }
      if (name_idx == 0) name_idx = i + 1;
    }
  }
  for (size_t i = 0; i < dynamic_.size(); ++i) {
    if (dynamic_[i].first == h.first) {
      if (dynamic_[i].second == h.second) {
        *name_only = false;
        return (int)(62 + i);
      }
      if (name_idx == 0) name_idx = (int)(62 + i);
    }
  }
  *name_only = name_idx != 0;
  return name_idx;
}

void Encoder::add_dynamic(const Header& h) {
  size_t es = entry_size(h);
  while (!dynamic_.empty() && size_ + es > max_size_) {
    size_ -= entry_size(dynamic_.back());
    dynamic_.pop_back();
  }
  if (es <= max_size_) {
    dynamic_.push_front(h);
    size_ += es;
  }
}

void Encoder::Encode(const std::vector<Header>& headers, std::string* out) {
  for (const auto& h : headers) {
    bool name_only = false;
    int idx = find(h, &name_only);
    if (idx > 0 && !name_only) {
      EncodeInt(out, (uint64_t)idx, 7, 0x80);  // indexed field
      continue;
    }
    // literal with incremental indexing
    if (idx > 0) {
      EncodeInt(out, (uint64_t)idx, 6, 0x40);
    } else {
      out->push_back(0x40);
      encode_string(out, h.first);
    }
    encode_string(out, h.second);
    add_dynamic(h);
  }
}

// ---------------- decoder ----------------

bool Decoder::lookup(uint64_t index, Header* h) const {
  if (index == 0) return false;
  if (index <= 61) {
    *h = kStatic[index - 1];
    return true;
  }
  size_t di = (size_t)(index - 62);
  if (di >= dynamic_.size()) return false;
  *h = dynamic_[di];
  return true;
}

void Decoder::add_dynamic(const Header& h) {
  size_t es = entry_size(h);
  while (!dynamic_.empty() && size_ + es > max_size_) {
    size_ -= entry_size(dynamic_.back());
    dynamic_.pop_back();
  }
  if (es <= max_size_) {
    dynamic_.push_front(h);
    size_ += es;
  }
}

bool Decoder::Decode(const char* data, size_t n, std::vector<Header>* out) {
  const uint8_t* p = (const uint8_t*)data;
  const uint8_t* end = p + n;
  while (p < end) {
    uint8_t b = *p;
    if (b & 0x80) {
      // indexed
      uint64_t idx;
      if (!DecodeInt(p, end, 7, &idx)) return false;
      Header h;
      if (!lookup(idx, &h)) return false;
      out->push_back(std::move(h));
    } else if (b & 0x40) {
      // literal with incremental indexing
      uint64_t idx;
      if (!DecodeInt(p, end, 6, &idx)) return false;
      Header h;
      if (idx != 0) {
        if (!lookup(idx, &h)) return false;
        h.second.clear();
      } else {
        if (!decode_string(p, end, &h.first)) return false;
      }
      if (!decode_string(p, end, &h.second)) return false;
      add_dynamic(h);
      out->push_back(std::move(h));
    } else if (b & 0x20) {
      // dynamic table size update
      uint64_t sz;
      if (!DecodeInt(p, end, 5, &sz)) return false;
      max_size_ = (size_t)sz;
      while (size_ > max_size_ && !dynamic_.empty()) {
        size_ -= entry_size(dynamic_.back());
        dynamic_.pop_back();
      }
    } else {
      // literal without indexing (0x00) / never indexed (0x10)
      uint64_t idx;
      if (!DecodeInt(p, end, 4, &idx)) return false;
      Header h;
      if (idx != 0) {
        if (!lookup(idx, &h)) return false;
        h.second.clear();
      } else {
        if (!decode_string(p, end, &h.first)) return false;
      }
      if (!decode_string(p, end, &h.second)) return false;
      out->push_back(std::move(h));
    }
  }
  return true;
}

}  // namespace hpack
}  // namespace bam
