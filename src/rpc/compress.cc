#include "rpc/compress.h"

#include <string.h>
#include <zlib.h>

#include <map>
#include <mutex>

#include "base/snappy.h"

namespace bam {

namespace {
struct Registry {
  std::mutex mu;
  std::map<int, CompressHandler> handlers;
};
Registry& registry() {
  static Registry* r = new Registry;
  return *r;
}

// ---- snappy ----
bool SnappyCompress(const IOBuf& in, IOBuf* out) {
  std::string flat = in.to_string();
  std::string compressed;
  snappy::Compress(flat.data(), flat.size(), &compressed);
  out->append(compressed);
  return true;
}

bool SnappyDecompress(const IOBuf& in, IOBuf* out) {
  std::string flat = in.to_string();
  std::string plain;
  if (!snappy::Uncompress(flat.data(), flat.size(), &plain)) return false;
  out->append(plain);
  return true;
}

// ---- gzip (zlib deflate with gzip wrapper) ----
bool GzipCompress(const IOBuf& in, IOBuf* out) {
  std::string flat = in.to_string();
  z_stream zs;
  memset(&zs, 0, sizeof(zs));
  if (deflateInit2(&zs, Z_DEFAULT_COMPRESSION, Z_DEFLATED, 15 + 16, 8,
                   Z_DEFAULT_STRATEGY) != Z_OK)
    return false;
  std::string buf(deflateBound(&zs, flat.size()), 0);
  zs.next_in = (Bytef*)flat.data();
  zs.avail_in = (uInt)flat.size();
  zs.next_out = (Bytef*)&buf[0];
  zs.avail_out = (uInt)buf.size();
  int rc = deflate(&zs, Z_FINISH);
  deflateEnd(&zs);
  if (rc != Z_STREAM_END) return false;
  buf.resize(zs.total_out);
  out->append(buf);
  return true;
}

bool GzipDecompress(const IOBuf& in, IOBuf* out) {
  std::string flat = in.to_string();
  z_stream zs;
  memset(&zs, 0, sizeof(zs));
  if (inflateInit2(&zs, 15 + 16) != Z_OK) return false;
  zs.next_in = (Bytef*)flat.data();
  zs.avail_in = (uInt)flat.size();
  std::string buf;
  char chunk[64 * 1024];
  int rc;
  do {
    zs.next_out = (Bytef*)chunk;
    zs.avail_out = sizeof(chunk);
    rc = inflate(&zs, Z_NO_FLUSH);
    if (rc != Z_OK && rc != Z_STREAM_END) {
      inflateEnd(&zs);
      return false;
    }
    buf.append(chunk, sizeof(chunk) - zs.avail_out);
  } while (rc != Z_STREAM_END);
  inflateEnd(&zs);
  out->append(buf);
  return true;
}

}  // namespace

int RegisterCompressHandler(CompressType type, CompressHandler handler) {
  Registry& r = registry();
  std::lock_guard<std::mutex> lk(r.mu);
  r.handlers[(int)type] = handler;
  return 0;
}

const CompressHandler* FindCompressHandler(CompressType type) {
  Registry& r = registry();
  std::lock_guard<std::mutex> lk(r.mu);
  auto it = r.handlers.find((int)type);
  return it == r.handlers.end() ? nullptr : &it->second;
}

bool ApplyCompress(CompressType type, const IOBuf& in, IOBuf* out) {
  if (type == COMPRESS_TYPE_NONE) {
    *out = in;
    return true;
  }
  const CompressHandler* h = FindCompressHandler(type);
  return h != nullptr && h->Compress(in, out);
}

bool ApplyDecompress(CompressType type, const IOBuf& in, IOBuf* out) {
  if (type == COMPRESS_TYPE_NONE) {
    *out = in;
    return true;
  }
  const CompressHandler* h = FindCompressHandler(type);
  return h != nullptr && h->Decompress(in, out);
}

void RegisterBuiltinCompressHandlers() {
  static std::once_flag flag;
  std::call_once(flag, [] {
    RegisterCompressHandler(COMPRESS_TYPE_SNAPPY,
                            {SnappyCompress, SnappyDecompress, "snappy"});
    RegisterCompressHandler(COMPRESS_TYPE_GZIP, {GzipCompress, GzipDecompress, "gzip"});
  });
}

}  // namespace bam
