// brpc_amd: Snappy on gfx950.
//
// Compress (block-parallel, wire-compatible): the input is split into
// 2 KiB sub-chunks; one wave owns 64 consecutive sub-chunks. Chunks are
// staged into LDS with coalesced transposed loads (same tile trick as
// crc32c.hip), then each lane greedily compresses ITS chunk out of LDS
// with a per-lane 128-entry hash table (also LDS). Because every emitted
// copy references bytes inside the lane's own chunk, the concatenation of
// per-chunk element streams is a VALID standard snappy stream (offsets are
// a strict subset of what the format allows) — the host codec
// (base/snappy.cc) decompresses it unchanged. Per-chunk outputs land in a
// global scratch; the host compacts them with the span-copy kernel.
//
// Decompress (wave-cooperative): control is wave-uniform (lane 0's parse
// broadcast via readfirstlane); all 64 lanes move the literal/copy bytes
// in parallel. Overlapping copies use modular indexing (the repeating
// pattern) instead of serial byte copies.
#include <hip/hip_runtime.h>

#include <mutex>

#include "gpu_api.h"

namespace {

constexpr int kLaneChunk = 2048;          // bytes per lane
constexpr int kChunkPad = 4;              // LDS bank spread (odd word stride)
constexpr int kHashBits = 7;              // 128-entry per-lane table
constexpr int kMaxPerChunk = kLaneChunk + kLaneChunk / 6 + 16;

__device__ __forceinline__ uint32_t ld32(const uint8_t* p) {
  return (uint32_t)p[0] | ((uint32_t)p[1] << 8) | ((uint32_t)p[2] << 16) |
         ((uint32_t)p[3] << 24);
}

__device__ __forceinline__ uint32_t hash4(uint32_t v) {
  return (v * 0x1e35a7bdu) >> (32 - kHashBits);
}

// ---------------- compress ----------------

__global__ __launch_bounds__(64) void snappy_compress_kernel(const uint8_t* in, size_t n,
                                                             uint8_t* scratch,
                                                             uint32_t* sizes, int nchunks) {
  __shared__ uint8_t data[64][kLaneChunk + kChunkPad];  // ~128 KiB
  __shared__ uint16_t table[64][1 << kHashBits];        // 16 KiB
  const int lane = threadIdx.x;
  const int cbase = blockIdx.x * 64;
  const int my_chunk = cbase + lane;

  // Coalesced transposed stage: row r = chunk (cbase+r), 64 words/iter.
  const size_t group_start = (size_t)cbase * kLaneChunk;
  const bool group_full = group_start + (size_t)64 * kLaneChunk <= n &&
                          (((uintptr_t)in & 7) == 0);
  if (group_full) {
    const uint64_t* win = (const uint64_t*)(in + group_start);
    for (int t = 0; t < kLaneChunk / 8; t += 64) {
      for (int r = 0; r < 64; ++r) {
        uint64_t v = win[(size_t)r * (kLaneChunk / 8) + t + lane];
        // row base is only 4-aligned (pad=4 for bank spread): two b32 writes
        *(uint32_t*)&data[r][(t + lane) * 8] = (uint32_t)v;
        *(uint32_t*)&data[r][(t + lane) * 8 + 4] = (uint32_t)(v >> 32);
      }
    }
  } else {
    // boundary group: plain per-lane byte loads
    size_t start = (size_t)my_chunk * kLaneChunk;
    size_t end = start + kLaneChunk < n ? start + kLaneChunk : n;
    for (size_t i = start; i < end; ++i) data[lane][i - start] = in[i];
  }
  for (int i = 0; i < (1 << kHashBits); ++i) table[lane][i] = 0;
  __syncthreads();

  if (my_chunk >= nchunks) return;
  size_t start = (size_t)my_chunk * kLaneChunk;
  int len = (int)((start + kLaneChunk <= n) ? kLaneChunk : n - start);
  const uint8_t* base = data[lane];
  uint16_t* tab = table[lane];
  uint8_t* out = scratch + (size_t)my_chunk * kMaxPerChunk;
  int op = 0;

  auto emit_literal = [&](int from, int count) {
    if (count == 0) return;
    int m = count - 1;
    if (m < 60) {
      out[op++] = (uint8_t)(m << 2);
    } else if (m < 256) {
      out[op++] = (uint8_t)(60 << 2);
      out[op++] = (uint8_t)m;
    } else {
      out[op++] = (uint8_t)(61 << 2);
      out[op++] = (uint8_t)(m & 0xff);
      out[op++] = (uint8_t)(m >> 8);
    }
    for (int k = 0; k < count; ++k) out[op + k] = base[from + k];
    op += count;
  };
  auto emit_copy64 = [&](int offset, int count) {  // count in [4,64]
    if (count < 12 && offset < 2048) {
      out[op++] = (uint8_t)(1 | ((count - 4) << 2) | ((offset >> 8) << 5));
      out[op++] = (uint8_t)(offset & 0xff);
    } else {
      out[op++] = (uint8_t)(2 | ((count - 1) << 2));
      out[op++] = (uint8_t)(offset & 0xff);
      out[op++] = (uint8_t)((offset >> 8) & 0xff);
    }
  };
  auto emit_copy = [&](int offset, int count) {
    while (count >= 68) {
      emit_copy64(offset, 64);
      count -= 64;
    }
    if (count > 64) {
      emit_copy64(offset, 60);
      count -= 60;
    }
    emit_copy64(offset, count);
  };

  int ip = 1;
  int next_emit = 0;
  if (len >= 15) {
    int skip = 32;
    while (ip + 8 <= len) {
      uint32_t cur = ld32(base + ip);
      uint32_t h = hash4(cur);
      int candidate = tab[h];
      tab[h] = (uint16_t)ip;
      if (candidate != 0 && candidate < ip && ld32(base + candidate) == cur) {
        emit_literal(next_emit, ip - next_emit);
        int mlen = 4;
        while (ip + mlen < len && base[candidate + mlen] == base[ip + mlen]) ++mlen;
        emit_copy(ip - candidate, mlen);
        ip += mlen;
        next_emit = ip;
        skip = 32;
        continue;
      }
      ip += 1 + (skip >> 5);
      ++skip;
    }
  }
  emit_literal(next_emit, len - next_emit);
  sizes[my_chunk] = (uint32_t)op;
}

// ---------------- decompress ----------------

// Wave-uniform control, wave-parallel byte movement. One block (1 wave).
__global__ __launch_bounds__(64) void snappy_decompress_kernel(const uint8_t* comp,
                                                               size_t comp_len,
                                                               uint8_t* out, size_t out_cap,
                                                               uint64_t* result) {
  const int lane = threadIdx.x;
  size_t ip = 0;
  // preamble varint
  size_t expected = 0;
  int shift = 0;
  while (ip < comp_len) {
    uint8_t b = comp[ip++];
    expected |= (size_t)(b & 0x7f) << shift;
    if (!(b & 0x80)) break;
    shift += 7;
  }
  if (expected > out_cap) {
    if (lane == 0) *result = ~0ULL;
    return;
  }
  size_t op = 0;
  bool ok = true;
  while (ip < comp_len && ok) {
    uint8_t tag = comp[ip++];
    int type = tag & 3;
    if (type == 0) {
      size_t len = (tag >> 2) + 1;
      if (len > 60) {
        int extra = (int)len - 60;
        len = 0;
        for (int k = 0; k < extra; ++k) len |= (size_t)comp[ip + k] << (8 * k);
        len += 1;
        ip += extra;
      }
      if (ip + len > comp_len || op + len > expected) {
        ok = false;
        break;
      }
      for (size_t k = lane; k < len; k += 64) out[op + k] = comp[ip + k];
      ip += len;
      op += len;
    } else {
      size_t len, offset;
      if (type == 1) {
        len = ((tag >> 2) & 7) + 4;
        offset = ((size_t)(tag >> 5) << 8) | comp[ip];
        ip += 1;
      } else if (type == 2) {
        len = (tag >> 2) + 1;
        offset = (size_t)comp[ip] | ((size_t)comp[ip + 1] << 8);
        ip += 2;
      } else {
        len = (tag >> 2) + 1;
        offset = (size_t)comp[ip] | ((size_t)comp[ip + 1] << 8) |
                 ((size_t)comp[ip + 2] << 16) | ((size_t)comp[ip + 3] << 24);
        ip += 4;
      }
      if (offset == 0 || offset > op || op + len > expected) {
        ok = false;
        break;
      }
      __syncthreads();  // prior writes must be visible before reading them
      const uint8_t* from = out + (op - offset);
      if (offset >= len) {
        for (size_t k = lane; k < len; k += 64) out[op + k] = from[k];
      } else {
        for (size_t k = lane; k < len; k += 64) out[op + k] = from[k % offset];
      }
      op += len;
    }
    __syncthreads();
  }
  if (lane == 0) *result = (ok && op == expected) ? (uint64_t)op : ~0ULL;
}

// ---- device-side compaction (prefix sums + scatter; zero host loops) ----

__global__ void snappy_block_sums(const uint32_t* sizes, int nchunks, uint64_t* bsums) {
  __shared__ uint64_t sdata[256];
  int t = threadIdx.x;
  int i = blockIdx.x * 256 + t;
  sdata[t] = i < nchunks ? sizes[i] : 0;
  __syncthreads();
  for (int s = 128; s > 0; s >>= 1) {
    if (t < s) sdata[t] += sdata[t + s];
    __syncthreads();
  }
  if (t == 0) bsums[blockIdx.x] = sdata[0];
}

__global__ void snappy_scan_bsums(uint64_t* bsums, int nb, uint64_t* total) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    uint64_t acc = 0;
    for (int i = 0; i < nb; ++i) {
      uint64_t v = bsums[i];
      bsums[i] = acc;
      acc += v;
    }
    *total = acc;
  }
}

__global__ void snappy_compact(const uint8_t* scratch, const uint32_t* sizes,
                               const uint64_t* bsums, uint8_t* dst, int nchunks,
                               int max_per_chunk) {
  __shared__ uint64_t offset_sh;
  int c = blockIdx.x;
  if (c >= nchunks) return;
  int t = threadIdx.x;
  if (t == 0) {
    uint64_t off = bsums[c / 256];
    int start = (c / 256) * 256;
    for (int i = start; i < c; ++i) off += sizes[i];
    offset_sh = off;
  }
  __syncthreads();
  uint64_t off = offset_sh;
  uint32_t len = sizes[c];
  const uint8_t* src = scratch + (size_t)c * max_per_chunk;
  for (uint32_t i = t; i < len; i += blockDim.x) dst[off + i] = src[i];
}

std::mutex g_snappy_mu;
struct SnappyScratch {
  uint8_t* per_chunk = nullptr;
  uint32_t* sizes = nullptr;
  uint64_t* bsums = nullptr;
  size_t chunk_cap = 0;
  uint64_t* result = nullptr;
  uint64_t* total = nullptr;
};
SnappyScratch g_snappy[16];

void emit_varint_host(uint8_t* dst, size_t v, int* n) {
  int i = 0;
  while (v >= 0x80) {
    dst[i++] = (uint8_t)(v | 0x80);
    v >>= 7;
  }
  dst[i++] = (uint8_t)v;
  *n = i;
}

}  // namespace

extern "C" int bam_gpu_snappy_compress(const void* src_dev, size_t n, void* dst_dev,
                                       size_t dst_cap, size_t* out_len, int dev) {
  bam_gpu_quiesce(dev);  // order after any in-flight async HBM uploads
  if (n == 0) return -1;
  int old_dev = -1;
  (void)hipGetDevice(&old_dev);
  if (dev != old_dev) (void)hipSetDevice(dev);
  const int nchunks = (int)((n + kLaneChunk - 1) / kLaneChunk);
  const int nblocks = (nchunks + 63) / 64;
  SnappyScratch* sc;
  {
    std::lock_guard<std::mutex> lk(g_snappy_mu);
    sc = &g_snappy[dev < 16 ? dev : 0];
    if (sc->chunk_cap < (size_t)nchunks) {
      if (sc->per_chunk) (void)hipFree(sc->per_chunk);
      if (sc->sizes) (void)hipFree(sc->sizes);
      if (sc->bsums) (void)hipFree(sc->bsums);
      sc->per_chunk = nullptr;
      sc->sizes = nullptr;
      sc->bsums = nullptr;
      sc->chunk_cap = (size_t)nchunks * 2;
      if (hipMalloc(&sc->per_chunk, sc->chunk_cap * kMaxPerChunk) != hipSuccess ||
          hipMalloc(&sc->sizes, sc->chunk_cap * sizeof(uint32_t)) != hipSuccess ||
          hipMalloc(&sc->bsums, ((sc->chunk_cap + 255) / 256 + 1) * sizeof(uint64_t)) !=
              hipSuccess) {
        sc->chunk_cap = 0;  // a failed alloc must not look usable next call
        if (dev != old_dev && old_dev >= 0) (void)hipSetDevice(old_dev);
        return -1;
      }
    }
    if (sc->total == nullptr &&
        hipMalloc(&sc->total, sizeof(uint64_t)) != hipSuccess) {
      if (dev != old_dev && old_dev >= 0) (void)hipSetDevice(old_dev);
      return -1;
    }
  }
  hipLaunchKernelGGL(snappy_compress_kernel, dim3(nblocks), dim3(64), 0, 0,
                     (const uint8_t*)src_dev, n, sc->per_chunk, sc->sizes, nchunks);
  // Device-side compaction: block sums -> exclusive scan -> scatter.
  const int nb = (nchunks + 255) / 256;
  hipLaunchKernelGGL(snappy_block_sums, dim3(nb), dim3(256), 0, 0, sc->sizes, nchunks,
                     sc->bsums);
  hipLaunchKernelGGL(snappy_scan_bsums, dim3(1), dim3(64), 0, 0, sc->bsums, nb, sc->total);
  uint64_t payload_total = 0;
  if (hipMemcpy(&payload_total, sc->total, sizeof(payload_total),
                hipMemcpyDeviceToHost) != hipSuccess) {
    if (dev != old_dev && old_dev >= 0) (void)hipSetDevice(old_dev);
    return -1;
  }
  uint8_t pre[8];
  int pre_n;
  emit_varint_host(pre, n, &pre_n);
  if (payload_total + pre_n > dst_cap) {
    if (dev != old_dev && old_dev >= 0) (void)hipSetDevice(old_dev);
    return -1;
  }
  if (hipMemcpy(dst_dev, pre, pre_n, hipMemcpyHostToDevice) != hipSuccess) {
    if (dev != old_dev && old_dev >= 0) (void)hipSetDevice(old_dev);
    return -1;
  }
  hipLaunchKernelGGL(snappy_compact, dim3(nchunks), dim3(256), 0, 0, sc->per_chunk,
                     sc->sizes, sc->bsums, (uint8_t*)dst_dev + pre_n, nchunks, kMaxPerChunk);
  hipError_t e = hipDeviceSynchronize();
  *out_len = (size_t)payload_total + pre_n;
  if (dev != old_dev && old_dev >= 0) (void)hipSetDevice(old_dev);
  return e == hipSuccess ? 0 : -1;
}

extern "C" int bam_gpu_snappy_decompress(const void* src_dev, size_t n, void* dst_dev,
                                         size_t dst_cap, size_t* out_len, int dev) {
  bam_gpu_quiesce(dev);  // order after any in-flight async HBM uploads
  int old_dev = -1;
  (void)hipGetDevice(&old_dev);
  if (dev != old_dev) (void)hipSetDevice(dev);
  SnappyScratch* sc;
  {
    std::lock_guard<std::mutex> lk(g_snappy_mu);
    sc = &g_snappy[dev < 16 ? dev : 0];
    if (sc->result == nullptr &&
        hipMalloc(&sc->result, sizeof(uint64_t)) != hipSuccess) {
      if (dev != old_dev && old_dev >= 0) (void)hipSetDevice(old_dev);
      return -1;
    }
  }
  hipLaunchKernelGGL(snappy_decompress_kernel, dim3(1), dim3(64), 0, 0,
                     (const uint8_t*)src_dev, n, (uint8_t*)dst_dev, dst_cap, sc->result);
  uint64_t res = 0;
  if (hipMemcpy(&res, sc->result, sizeof(res), hipMemcpyDeviceToHost) != hipSuccess) {
    if (dev != old_dev && old_dev >= 0) (void)hipSetDevice(old_dev);
    return -1;
  }
  if (dev != old_dev && old_dev >= 0) (void)hipSetDevice(old_dev);
  if (res == ~0ULL) return -1;
  *out_len = (size_t)res;
  return 0;
}
