// GPU-path test scenarios (run on MI355X via pytest -m gpu through the
// bindings). Each checks a gfx950 kernel against the host reference.
#include <stdlib.h>
#include <string.h>

#include <string>
#include <vector>

#include "base/crc32c.h"
#include "base/fast_rand.h"
#include "base/gpu_loader.h"
#include "base/iobuf.h"
#include "base/logging.h"

namespace bam {
namespace gputest {

// Uploads random bytes into HBM IOBuf blocks; checks copy_to round-trip.
bool hbm_iobuf_roundtrip(size_t n, int dev) {
  const gpu::GpuApi* api = gpu::api();
  if (api == nullptr) return false;
  std::string data(n, 0);
  for (size_t i = 0; i < n; ++i) data[i] = (char)fast_rand();
  IOBuf buf;
  if (buf.append_with_residency(data.data(), n, RES_HBM, dev, 64 << 10) != 0) return false;
  if (buf.hbm_bytes() != n || buf.cpu_addressable()) return false;
  std::string back;
  buf.copy_to(&back);
  return back == data;
}

// GPU crc32c vs host reference over a device buffer.
bool crc_matches(size_t n, int dev) {
  const gpu::GpuApi* api = gpu::api();
  if (api == nullptr) return false;
  std::string data(n, 0);
  for (size_t i = 0; i < n; ++i) data[i] = (char)fast_rand();
  void* d = api->alloc_hbm((uint32_t)n, dev);  // >2MiB falls to direct hipMalloc
  if (d == nullptr) return false;
  api->memcpy_res(d, 2, dev, data.data(), 0, -1, n);
  uint32_t gpu_crc = api->crc32c(d, n, 0, dev);
  uint32_t host_crc = crc32c::Value(data.data(), n);
  api->free_hbm(d, (uint32_t)n, dev);
  return gpu_crc == host_crc;
}

// crc over a device buffer with nonzero init (Extend semantics).
bool crc_extend_matches(size_t n1, size_t n2, int dev) {
  const gpu::GpuApi* api = gpu::api();
  if (api == nullptr) return false;
  std::string a(n1, 0), b(n2, 0);
  for (auto& c : a) c = (char)fast_rand();
  for (auto& c : b) c = (char)fast_rand();
  void* d = api->alloc_hbm((uint32_t)n2, dev);
  if (d == nullptr) return false;
  api->memcpy_res(d, 2, dev, b.data(), 0, -1, n2);
  uint32_t init = crc32c::Value(a.data(), n1);
  uint32_t gpu_crc = api->crc32c(d, n2, init, dev);
  std::string ab = a + b;
  uint32_t host_crc = crc32c::Value(ab.data(), ab.size());
  api->free_hbm(d, (uint32_t)n2, dev);
  return gpu_crc == host_crc;
}

// Gather scattered IOBuf HBM spans into a contiguous device buffer and
// verify via GPU crc against the host crc of the logical byte stream.
bool gather_matches(size_t total, uint32_t block, int dev) {
  const gpu::GpuApi* api = gpu::api();
  if (api == nullptr) return false;
  std::string data(total, 0);
  for (size_t i = 0; i < total; ++i) data[i] = (char)fast_rand();
  IOBuf buf;
  if (buf.append_with_residency(data.data(), total, RES_HBM, dev, block) != 0) return false;
  std::vector<const void*> srcs;
  std::vector<size_t> lens;
  for (size_t i = 0; i < buf.backing_block_num(); ++i) {
    IOBuf::Span sp = buf.span_at(i);
    srcs.push_back(sp.data);
    lens.push_back(sp.length);
  }
  void* d = api->alloc_hbm((uint32_t)total, dev);
  if (d == nullptr) return false;
  if (api->gather(d, srcs.data(), lens.data(), (int)srcs.size(), dev) != 0) return false;
  uint32_t gpu_crc = api->crc32c(d, total, 0, dev);
  return gpu_crc == crc32c::Value(data.data(), total);
}

// Pinned residency round trip.
bool pinned_roundtrip(size_t n) {
  if (!gpu::loaded() || gpu::device_count() == 0) return false;
  std::string data(n, 0);
  for (size_t i = 0; i < n; ++i) data[i] = (char)fast_rand();
  IOBuf buf;
  if (buf.append_with_residency(data.data(), n, RES_PINNED, 0, 0) != 0) return false;
  if (!buf.cpu_addressable()) return false;
  std::string back;
  buf.copy_to(&back);
  return back == data;
}

}  // namespace gputest
}  // namespace bam

// ---- perf probes (GB/s numbers for profiles/) ----

#include "base/time.h"

namespace bam {
namespace gputest {

// GPU crc32c throughput over a resident device buffer.
double crc_gbps(size_t n, int iters, int dev) {
  const gpu::GpuApi* api = gpu::api();
  if (api == nullptr) return -1;
  void* d = api->alloc_hbm((uint32_t)n, dev);
  if (d == nullptr) return -1;
  api->fill(d, n, 0x0123456789abcdefULL, dev);
  api->crc32c(d, n, 0, dev);  // warm
  int64_t t0 = monotonic_time_us();
  for (int i = 0; i < iters; ++i) api->crc32c(d, n, 0, dev);
  int64_t el = monotonic_time_us() - t0;
  api->free_hbm(d, (uint32_t)n, dev);
  return (double)n * iters / el / 1000.0;  // GB/s
}

// gather kernel throughput (spans of `block` bytes into contiguous).
double gather_gbps(size_t total, uint32_t block, int iters, int dev) {
  const gpu::GpuApi* api = gpu::api();
  if (api == nullptr) return -1;
  int nspans = (int)((total + block - 1) / block);
  std::vector<void*> blocks(nspans);
  std::vector<const void*> srcs(nspans);
  std::vector<size_t> lens(nspans);
  for (int i = 0; i < nspans; ++i) {
    blocks[i] = api->alloc_hbm(block, dev);
    if (blocks[i] == nullptr) return -1;
    srcs[i] = blocks[i];
    lens[i] = (i == nspans - 1) ? total - (size_t)(nspans - 1) * block : block;
  }
  void* dst = api->alloc_hbm((uint32_t)total, dev);
  if (dst == nullptr) return -1;
  api->gather(dst, srcs.data(), lens.data(), nspans, dev);  // warm
  int64_t t0 = monotonic_time_us();
  for (int i = 0; i < iters; ++i) api->gather(dst, srcs.data(), lens.data(), nspans, dev);
  int64_t el = monotonic_time_us() - t0;
  for (int i = 0; i < nspans; ++i) api->free_hbm(blocks[i], block, dev);
  api->free_hbm(dst, (uint32_t)total, dev);
  // count read+write bytes
  return 2.0 * total * iters / el / 1000.0;  // GB/s
}

// D2H staging throughput through the byte mover (the HBM->wire path).
double d2h_gbps(size_t n, int iters, int dev) {
  const gpu::GpuApi* api = gpu::api();
  if (api == nullptr) return -1;
  void* d = api->alloc_hbm((uint32_t)n, dev);
  void* h = malloc(n);
  if (d == nullptr || h == nullptr) return -1;
  api->memcpy_res(h, 0, -1, d, 2, dev, n);  // warm
  int64_t t0 = monotonic_time_us();
  for (int i = 0; i < iters; ++i) api->memcpy_res(h, 0, -1, d, 2, dev, n);
  int64_t el = monotonic_time_us() - t0;
  free(h);
  api->free_hbm(d, (uint32_t)n, dev);
  return (double)n * iters / el / 1000.0;
}

}  // namespace gputest
}  // namespace bam

// ---- GPU snappy vs host codec (both directions) ----

#include "base/snappy.h"

namespace bam {
namespace gputest {

// GPU-compressed stream must decompress with the HOST codec, and a
// HOST-compressed stream must decompress on the GPU.
bool snappy_cross_check(size_t n, int mode /*0 compressible, 1 random*/, int dev) {
  const gpu::GpuApi* api = gpu::api();
  if (api == nullptr) return false;
  std::string data(n, 0);
  if (mode == 0) {
    for (size_t i = 0; i < n; ++i) data[i] = (char)("abcabcab"[i % 8] + (i / 1024) % 3);
  } else {
    for (size_t i = 0; i < n; ++i) data[i] = (char)fast_rand();
  }
  void* d_in = api->alloc_hbm((uint32_t)n, dev);
  size_t comp_cap = n + n / 3 + 64;
  void* d_comp = api->alloc_hbm((uint32_t)comp_cap, dev);
  if (d_in == nullptr || d_comp == nullptr) return false;
  api->memcpy_res(d_in, 2, dev, data.data(), 0, -1, n);
  size_t comp_len = 0;
  if (api->snappy_compress(d_in, n, d_comp, comp_cap, &comp_len, dev) != 0) return false;
  // host decompress of the GPU stream
  std::string comp(comp_len, 0);
  api->memcpy_res(&comp[0], 0, -1, d_comp, 2, dev, comp_len);
  std::string back;
  if (!snappy::Uncompress(comp.data(), comp.size(), &back)) return false;
  if (back != data) return false;
  // GPU decompress of a HOST-compressed stream
  std::string host_comp;
  snappy::Compress(data.data(), data.size(), &host_comp);
  void* d_hcomp = api->alloc_hbm((uint32_t)host_comp.size(), dev);
  void* d_out = api->alloc_hbm((uint32_t)n, dev);
  if (d_hcomp == nullptr || d_out == nullptr) return false;
  api->memcpy_res(d_hcomp, 2, dev, host_comp.data(), 0, -1, host_comp.size());
  size_t out_len = 0;
  if (api->snappy_decompress(d_hcomp, host_comp.size(), d_out, n, &out_len, dev) != 0)
    return false;
  if (out_len != n) return false;
  std::string back2(n, 0);
  api->memcpy_res(&back2[0], 0, -1, d_out, 2, dev, n);
  // also decompress the GPU's own stream on the GPU
  size_t out_len2 = 0;
  if (api->snappy_decompress(d_comp, comp_len, d_out, n, &out_len2, dev) != 0) return false;
  std::string back3(n, 0);
  api->memcpy_res(&back3[0], 0, -1, d_out, 2, dev, n);
  api->free_hbm(d_in, (uint32_t)n, dev);
  api->free_hbm(d_comp, (uint32_t)comp_cap, dev);
  api->free_hbm(d_hcomp, (uint32_t)host_comp.size(), dev);
  api->free_hbm(d_out, (uint32_t)n, dev);
  return back2 == data && back3 == data;
}

double snappy_compress_gbps(size_t n, int iters, int dev) {
  const gpu::GpuApi* api = gpu::api();
  if (api == nullptr) return -1;
  void* d_in = api->alloc_hbm((uint32_t)n, dev);
  size_t cap = n + n / 3 + 64;
  void* d_comp = api->alloc_hbm((uint32_t)cap, dev);
  if (d_in == nullptr || d_comp == nullptr) return -1;
  api->fill(d_in, n, 0x6162636461626364ULL, dev);  // compressible
  size_t comp_len;
  api->snappy_compress(d_in, n, d_comp, cap, &comp_len, dev);
  int64_t t0 = monotonic_time_us();
  for (int i = 0; i < iters; ++i) api->snappy_compress(d_in, n, d_comp, cap, &comp_len, dev);
  int64_t el = monotonic_time_us() - t0;
  api->free_hbm(d_in, (uint32_t)n, dev);
  api->free_hbm(d_comp, (uint32_t)cap, dev);
  return (double)n * iters / el / 1000.0;
}

}  // namespace gputest
}  // namespace bam
