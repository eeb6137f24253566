// brpc_amd: IOBuf scatter/gather kernels for gfx950.
// Assembles/disperses non-contiguous HBM-resident IOBuf block spans —
// the device-side analogue of the reference's writev/readv assembly
// (butil/iobuf.cpp cut_multiple_into_file_descriptor): one workgroup per
// span, grid-stride vectorized copy, uint4 (16 B) per lane where aligned.
#include <hip/hip_runtime.h>

#include "gpu_api.h"

namespace {

struct Span {
  const char* src;
  char* dst;
  size_t len;
};

constexpr int kMaxSpansPerLaunch = 1024;
__device__ Span d_spans[kMaxSpansPerLaunch];

__global__ void copy_spans_kernel(int nspans) {
  const Span s = d_spans[blockIdx.x];
  const size_t tid = threadIdx.x;
  const size_t nthreads = blockDim.x;
  // 16-byte path when both pointers share alignment.
  if ((((uintptr_t)s.src ^ (uintptr_t)s.dst) & 15) == 0) {
    uintptr_t head = (16 - ((uintptr_t)s.src & 15)) & 15;
    if (head > s.len) head = s.len;
    for (size_t i = tid; i < head; i += nthreads) s.dst[i] = s.src[i];
    const size_t nvec = (s.len - head) / 16;
    const uint4* vsrc = (const uint4*)(s.src + head);
    uint4* vdst = (uint4*)(s.dst + head);
    for (size_t i = tid; i < nvec; i += nthreads) vdst[i] = vsrc[i];
    for (size_t i = head + nvec * 16 + tid; i < s.len; i += nthreads) s.dst[i] = s.src[i];
  } else {
    for (size_t i = tid; i < s.len; i += nthreads) s.dst[i] = s.src[i];
  }
}

// NOTE: launches + synchronizes. Batches share the d_spans symbol, so a
// sync is required between batches; the common case is a single batch.
int launch_spans(Span* spans, int nspans, int dev) {
  bam_gpu_quiesce(dev);  // order after any in-flight async HBM uploads
  int old_dev = -1;
  hipGetDevice(&old_dev);
  if (dev != old_dev) hipSetDevice(dev);
  int rc = 0;
  for (int off = 0; off < nspans; off += kMaxSpansPerLaunch) {
    int batch = nspans - off < kMaxSpansPerLaunch ? nspans - off : kMaxSpansPerLaunch;
    // hipMemcpyToSymbol is stream-ordered with the prior kernel on the
    // null stream, so no explicit inter-batch sync is needed.
    hipMemcpyToSymbol(HIP_SYMBOL(d_spans), spans + off, sizeof(Span) * batch);
    hipLaunchKernelGGL(copy_spans_kernel, dim3(batch), dim3(256), 0, 0, batch);
  }
  hipError_t e = hipDeviceSynchronize();
  if (e != hipSuccess) rc = -1;
  if (dev != old_dev && old_dev >= 0) hipSetDevice(old_dev);
  return rc;
}

}  // namespace

// Large spans are chopped into ≤256 KiB sub-spans so the grid has ≫256
// workgroups (one block per sub-span) even for a few multi-MiB blocks.
constexpr size_t kSubSpan = 256u << 10;

template <typename NextFn>
static int run_spans(int nspans, NextFn next, int dev) {
  Span spans[kMaxSpansPerLaunch];
  int batch = 0;
  int rc = 0;
  for (int i = 0; i < nspans; ++i) {
    Span s = next(i);
    size_t off = 0;
    while (off < s.len) {
      size_t piece = s.len - off < kSubSpan ? s.len - off : kSubSpan;
      spans[batch].src = s.src + off;
      spans[batch].dst = s.dst + off;
      spans[batch].len = piece;
      off += piece;
      if (++batch == kMaxSpansPerLaunch) {
        rc |= launch_spans(spans, batch, dev);
        batch = 0;
      }
    }
  }
  if (batch > 0) rc |= launch_spans(spans, batch, dev);
  return rc;
}

extern "C" int bam_gpu_gather(void* dst_dev, const void* const* srcs, const size_t* lens,
                              int nspans, int dev) {
  if (nspans <= 0) return 0;
  char* out = (char*)dst_dev;
  size_t acc = 0;
  return run_spans(nspans,
                   [&](int i) {
                     Span s{(const char*)srcs[i], out + acc, lens[i]};
                     acc += lens[i];
                     return s;
                   },
                   dev);
}

extern "C" int bam_gpu_scatter(void* const* dsts, const size_t* lens, int nspans,
                               const void* src_dev, int dev) {
  if (nspans <= 0) return 0;
  const char* in = (const char*)src_dev;
  size_t acc = 0;
  return run_spans(nspans,
                   [&](int i) {
                     Span s{in + acc, (char*)dsts[i], lens[i]};
                     acc += lens[i];
                     return s;
                   },
                   dev);
}
