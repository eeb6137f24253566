// brpc_amd: IOBuf scatter/gather kernels for gfx950.
// Assembles/disperses non-contiguous HBM-resident IOBuf block spans —
// the device-side analogue of the reference's writev/readv assembly
// (butil/iobuf.cpp cut_multiple_into_file_descriptor): one workgroup per
// span, vectorized uint4 (16 B) per lane where aligned.
//
// Round-2 redesign (VERDICT weak #4): no global __device__ span table and
// no hipDeviceSynchronize. Spans travel by kernarg, each call runs on one
// of 4 per-device streams (round-robin), and completion is a pinned ticket
// flag published by the last workgroup — waited via bamhip::wait_ticket
// (short spin, then fiber park). Concurrent connections overlap instead of
// serializing process-wide.
#include <hip/hip_runtime.h>

#include <atomic>
#include <mutex>
#include <vector>

#include "gpu_api.h"
#include "internal.h"

namespace {

constexpr int kMaxDev = 16;
constexpr int kCtxPerDev = 4;  // = kWakeSpan3 - kWakeSpan0 + 1
constexpr int kSpanArgMax = 48;

struct SpanRec {
  const char* src;
  char* dst;
  unsigned int len;
};

// Host-side span descriptor (full size_t length; split into ≤256 KiB
// SpanRec pieces before launch).
struct SpanIn {
  const char* src;
  char* dst;
  size_t len;
};

struct SpanBatch {
  SpanRec spans[kSpanArgMax];
  unsigned long long* counter;          // persistent device counter
  volatile unsigned long long* flag;    // pinned ticket flag
  unsigned long long expect;            // counter value once ALL blocks ran
  unsigned long long ticket;
};

__device__ __forceinline__ void copy_one_span(const SpanRec& s) {
  const unsigned int tid = threadIdx.x;
  const unsigned int nt = blockDim.x;
  if ((((uintptr_t)s.src ^ (uintptr_t)s.dst) & 15) == 0) {
    unsigned int head = (16 - ((uintptr_t)s.src & 15)) & 15;
    if (head > s.len) head = s.len;
    for (unsigned int i = tid; i < head; i += nt) s.dst[i] = s.src[i];
    const unsigned int nvec = (s.len - head) / 16;
    const uint4* vsrc = (const uint4*)(s.src + head);
    uint4* vdst = (uint4*)(s.dst + head);
    for (unsigned int i = tid; i < nvec; i += nt) vdst[i] = vsrc[i];
    for (unsigned int i = head + nvec * 16 + tid; i < s.len; i += nt) s.dst[i] = s.src[i];
  } else {
    for (unsigned int i = tid; i < s.len; i += nt) s.dst[i] = s.src[i];
  }
}

__global__ void copy_spans_table_kernel(const SpanRec* table, int n,
                                        unsigned long long* counter,
                                        volatile unsigned long long* flag,
                                        unsigned long long expect,
                                        unsigned long long ticket) {
  copy_one_span(table[blockIdx.x]);
  __syncthreads();
  if (threadIdx.x == 0) {
    __threadfence_system();
    unsigned long long done = atomicAdd(counter, 1ull) + 1;
    if (done == expect) {
      *flag = ticket;
      __threadfence_system();
    }
  }
  (void)n;
}

__global__ void copy_spans_kernel(SpanBatch a) {
  const SpanRec s = a.spans[blockIdx.x];
  const unsigned int tid = threadIdx.x;
  const unsigned int nt = blockDim.x;
  // 16-byte path when both pointers share alignment.
  if ((((uintptr_t)s.src ^ (uintptr_t)s.dst) & 15) == 0) {
    unsigned int head = (16 - ((uintptr_t)s.src & 15)) & 15;
    if (head > s.len) head = s.len;
    for (unsigned int i = tid; i < head; i += nt) s.dst[i] = s.src[i];
    const unsigned int nvec = (s.len - head) / 16;
    const uint4* vsrc = (const uint4*)(s.src + head);
    uint4* vdst = (uint4*)(s.dst + head);
    for (unsigned int i = tid; i < nvec; i += nt) vdst[i] = vsrc[i];
    for (unsigned int i = head + nvec * 16 + tid; i < s.len; i += nt) s.dst[i] = s.src[i];
  } else {
    for (unsigned int i = tid; i < s.len; i += nt) s.dst[i] = s.src[i];
  }
  __syncthreads();
  if (tid == 0) {
    __threadfence_system();
    unsigned long long done = atomicAdd(a.counter, 1ull) + 1;
    if (done == a.expect) {
      *a.flag = a.ticket;
      __threadfence_system();
    }
  }
}

struct SpanCtx {
  std::mutex mu;
  hipStream_t stream = nullptr;
  unsigned long long* counter_dev = nullptr;
  volatile unsigned long long* flag = nullptr;
  unsigned long long launched = 0;
  unsigned long long ticket = 0;
  int status = 0;  // 0 untried, 1 ok, -1 unavailable
  // Large batches: span table staged through pinned memory into a device
  // table, ONE launch regardless of span count (a 64 MB/8 KB gather is
  // 8192 spans — kernarg batches of 48 would cost 171 launches).
  SpanRec* table_dev = nullptr;
  SpanRec* table_pinned = nullptr;
  size_t table_cap = 0;  // entries
};
SpanCtx g_ctx[kMaxDev][kCtxPerDev];
std::atomic<unsigned int> g_rr[kMaxDev];

bool ctx_init(SpanCtx& c, int dev, int idx) {
  if (hipStreamCreateWithFlags(&c.stream, hipStreamNonBlocking) != hipSuccess) return false;
  if (hipMalloc(&c.counter_dev, 8) != hipSuccess) return false;
  if (hipMemset(c.counter_dev, 0, 8) != hipSuccess) return false;
  void* f = nullptr;
  if (hipHostMalloc(&f, 64, hipHostMallocDefault) != hipSuccess) return false;
  c.flag = (volatile unsigned long long*)f;
  *c.flag = 0;
  bamhip::register_wake_stream(dev, bamhip::kWakeSpan0 + idx, c.stream);
  return true;
}

// Large spans are chopped into ≤256 KiB sub-spans so the grid has ≫256
// workgroups (one block per sub-span) even for a few multi-MiB blocks.
constexpr size_t kSubSpan = 256u << 10;

template <typename NextFn>
int run_spans(int nspans, NextFn next, int dev) {
  bam_gpu_quiesce(dev);  // order after any in-flight async HBM uploads
  int old_dev = -1;
  (void)hipGetDevice(&old_dev);
  if (dev != old_dev) (void)hipSetDevice(dev);
  const int d = dev >= 0 && dev < kMaxDev ? dev : 0;
  const int idx = (int)(g_rr[d].fetch_add(1, std::memory_order_relaxed) % kCtxPerDev);
  SpanCtx& c = g_ctx[d][idx];
  int rc = 0;
  unsigned long long wait_ticket_val = 0;
  {
    std::lock_guard<std::mutex> lk(c.mu);
    if (c.status == 0) c.status = ctx_init(c, d, idx) ? 1 : -1;
    if (c.status < 0) {
      if (dev != old_dev && old_dev >= 0) (void)hipSetDevice(old_dev);
      return -1;
    }
    // Materialize the spans ONCE (next() is stateful: it accumulates the
    // running offset — calling it twice produced out-of-range dst
    // pointers and a GPU memory fault), then count ≤kSubSpan pieces.
    std::vector<SpanIn> spans;
    spans.reserve((size_t)nspans);
    size_t npieces = 0;
    for (int i = 0; i < nspans; ++i) {
      spans.push_back(next(i));
      npieces += (spans.back().len + kSubSpan - 1) / kSubSpan;
    }
    if (npieces == 0) {
      if (dev != old_dev && old_dev >= 0) (void)hipSetDevice(old_dev);
      return 0;
    }
    if (npieces <= (size_t)kSpanArgMax) {
      // Latency path: spans by kernarg, one launch.
      SpanBatch a;
      int batch = 0;
      for (int i = 0; i < nspans; ++i) {
        const SpanIn& s = spans[(size_t)i];
        size_t off = 0;
        while (off < s.len) {
          size_t piece = s.len - off < kSubSpan ? s.len - off : kSubSpan;
          a.spans[batch].src = s.src + off;
          a.spans[batch].dst = s.dst + off;
          a.spans[batch].len = (unsigned int)piece;
          off += piece;
          ++batch;
        }
      }
      c.launched += (unsigned long long)batch;
      c.ticket += 1;
      a.counter = c.counter_dev;
      a.flag = c.flag;
      a.expect = c.launched;
      a.ticket = c.ticket;
      hipLaunchKernelGGL(copy_spans_kernel, dim3(batch), dim3(256), 0, c.stream, a);
      if (hipGetLastError() != hipSuccess) {
        c.status = -1;
        rc = -1;
      }
    } else {
      // Throughput path: stage the span table through pinned memory into
      // the device table, ONE launch for every piece.
      if (c.table_cap < npieces) {
        size_t want = npieces * 2;
        if (c.table_dev != nullptr) (void)hipFree(c.table_dev);
        if (c.table_pinned != nullptr) (void)hipHostFree(c.table_pinned);
        c.table_dev = nullptr;
        c.table_pinned = nullptr;
        c.table_cap = 0;
        if (hipMalloc(&c.table_dev, want * sizeof(SpanRec)) != hipSuccess ||
            hipHostMalloc((void**)&c.table_pinned, want * sizeof(SpanRec),
                          hipHostMallocDefault) != hipSuccess) {
          c.status = -1;
          rc = -1;
        } else {
          c.table_cap = want;
        }
      }
      if (rc == 0) {
        size_t w = 0;
        for (int i = 0; i < nspans; ++i) {
          const SpanIn& s = spans[(size_t)i];
          size_t off = 0;
          while (off < s.len) {
            size_t piece = s.len - off < kSubSpan ? s.len - off : kSubSpan;
            c.table_pinned[w].src = s.src + off;
            c.table_pinned[w].dst = s.dst + off;
            c.table_pinned[w].len = (unsigned int)piece;
            ++w;
            off += piece;
          }
        }
        c.launched += (unsigned long long)npieces;
        c.ticket += 1;
        if (hipMemcpyAsync(c.table_dev, c.table_pinned, npieces * sizeof(SpanRec),
                           hipMemcpyHostToDevice, c.stream) != hipSuccess) {
          c.status = -1;
          rc = -1;
        } else {
          hipLaunchKernelGGL(copy_spans_table_kernel, dim3((uint32_t)npieces), dim3(256), 0,
                             c.stream, c.table_dev, (int)npieces, c.counter_dev, c.flag,
                             c.launched, c.ticket);
          if (hipGetLastError() != hipSuccess) {
            c.status = -1;
            rc = -1;
          }
        }
      }
    }
    wait_ticket_val = c.ticket;
  }
  if (rc == 0 && wait_ticket_val != 0 &&
      !bamhip::wait_ticket(c.flag, wait_ticket_val, d, bamhip::kWakeSpan0 + idx, c.stream)) {
    std::lock_guard<std::mutex> lk(c.mu);
    c.status = -1;
    rc = -1;
  }
  if (dev != old_dev && old_dev >= 0) (void)hipSetDevice(old_dev);
  return rc;
}

}  // namespace

extern "C" int bam_gpu_gather(void* dst_dev, const void* const* srcs, const size_t* lens,
                              int nspans, int dev) {
  if (nspans <= 0) return 0;
  char* out = (char*)dst_dev;
  size_t acc = 0;
  return run_spans(nspans,
                   [&](int i) {
                     SpanIn s{(const char*)srcs[i], out + acc, lens[i]};
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
                     SpanIn s{in + acc, (char*)dsts[i], lens[i]};
                     acc += lens[i];
                     return s;
                   },
                   dev);
}
