// brpc_amd: GPU runtime — device discovery, HBM block pools, pinned pools,
// residency-aware memcpy. gfx950-only, no CUDA paths, no torch.
//
// HBM pool design (capability parity with reference brpc/rdma/block_pool.cpp
// which registers {8K,64K,2M} blocks for verbs): hipMalloc 256 MiB slabs,
// carve into size-class blocks, per-class freelists. 288 GB HBM3E per GPU
// means we bias toward big slabs and never return memory to the runtime.
#include <hip/hip_runtime.h>
#include <string.h>

#include <mutex>
#include <vector>

#include "gpu_api.h"

static char g_err[256] = {0};

static void set_err(const char* what, hipError_t e) {
  snprintf(g_err, sizeof(g_err), "%s: %s", what, hipGetErrorString(e));
}

extern "C" const char* bam_gpu_last_error(void) { return g_err; }

extern "C" int bam_gpu_device_count(void) {
  static int count = [] {
    int n = 0;
    hipError_t e = hipGetDeviceCount(&n);
    if (e != hipSuccess) return 0;
    return n;
  }();
  return count;
}

// ---------------- HBM block pool ----------------

namespace {

constexpr uint32_t kClasses[3] = {8u << 10, 64u << 10, 2u << 20};
constexpr size_t kSlabBytes = 256u << 20;
constexpr int kMaxDev = 16;

struct DevPool {
  std::mutex mu;
  std::vector<void*> freelists[3];
  size_t slab_used = 0;
  char* slab = nullptr;
};

DevPool g_pools[kMaxDev];

int class_of(uint32_t cap) {
  for (int i = 0; i < 3; ++i)
    if (cap <= kClasses[i]) return i;
  return -1;
}

struct ScopedDevice {
  int old = -1;
  explicit ScopedDevice(int dev) {
    hipGetDevice(&old);
    if (dev != old) hipSetDevice(dev);
    else old = -1;
  }
  ~ScopedDevice() {
    if (old >= 0) hipSetDevice(old);
  }
};

}  // namespace

extern "C" void* bam_gpu_alloc_hbm(uint32_t cap, int dev) {
  if (dev < 0 || dev >= kMaxDev) return nullptr;
  int cls = class_of(cap);
  ScopedDevice sd(dev);
  if (cls < 0) {
    // big one-off allocation
    void* p = nullptr;
    hipError_t e = hipMalloc(&p, cap);
    if (e != hipSuccess) {
      set_err("hipMalloc", e);
      return nullptr;
    }
    return p;
  }
  DevPool& pool = g_pools[dev];
  std::lock_guard<std::mutex> lk(pool.mu);
  if (!pool.freelists[cls].empty()) {
    void* p = pool.freelists[cls].back();
    pool.freelists[cls].pop_back();
    return p;
  }
  uint32_t block = kClasses[cls];
  if (pool.slab == nullptr || pool.slab_used + block > kSlabBytes) {
    void* s = nullptr;
    hipError_t e = hipMalloc(&s, kSlabBytes);
    if (e != hipSuccess) {
      set_err("hipMalloc slab", e);
      return nullptr;
    }
    pool.slab = (char*)s;
    pool.slab_used = 0;
  }
  void* p = pool.slab + pool.slab_used;
  pool.slab_used += block;
  return p;
}

extern "C" void bam_gpu_free_hbm(void* p, uint32_t cap, int dev) {
  if (p == nullptr || dev < 0 || dev >= kMaxDev) return;
  int cls = class_of(cap);
  if (cls < 0) {
    ScopedDevice sd(dev);
    hipFree(p);
    return;
  }
  DevPool& pool = g_pools[dev];
  std::lock_guard<std::mutex> lk(pool.mu);
  pool.freelists[cls].push_back(p);
}

// ---------------- pinned pool ----------------

namespace {
struct PinnedPool {
  std::mutex mu;
  std::vector<std::pair<void*, uint32_t>> freelist;
};
PinnedPool g_pinned;
}  // namespace

extern "C" void* bam_gpu_alloc_pinned(uint32_t cap, int /*dev*/) {
  {
    std::lock_guard<std::mutex> lk(g_pinned.mu);
    for (size_t i = 0; i < g_pinned.freelist.size(); ++i) {
      if (g_pinned.freelist[i].second >= cap) {
        void* p = g_pinned.freelist[i].first;
        g_pinned.freelist[i] = g_pinned.freelist.back();
        g_pinned.freelist.pop_back();
        return p;
      }
    }
  }
  void* p = nullptr;
  hipError_t e = hipHostMalloc(&p, cap, hipHostMallocDefault);
  if (e != hipSuccess) {
    set_err("hipHostMalloc", e);
    return nullptr;
  }
  return p;
}

extern "C" void bam_gpu_free_pinned(void* p, uint32_t cap, int /*dev*/) {
  if (p == nullptr) return;
  std::lock_guard<std::mutex> lk(g_pinned.mu);
  if (g_pinned.freelist.size() < 64) {
    g_pinned.freelist.emplace_back(p, cap);
    return;
  }
  hipHostFree(p);
}

// ---------------- memcpy ----------------

extern "C" void bam_gpu_memcpy(void* dst, int dst_res, int dst_dev, const void* src,
                               int src_res, int src_dev, size_t n) {
  const bool dst_dev_mem = dst_res == 2;
  const bool src_dev_mem = src_res == 2;
  if (!dst_dev_mem && !src_dev_mem) {
    ::memcpy(dst, src, n);
    return;
  }
  hipMemcpyKind kind;
  int dev = 0;
  if (dst_dev_mem && src_dev_mem) {
    kind = hipMemcpyDeviceToDevice;
    dev = dst_dev;
  } else if (dst_dev_mem) {
    kind = hipMemcpyHostToDevice;
    dev = dst_dev;
  } else {
    kind = hipMemcpyDeviceToHost;
    dev = src_dev;
  }
  ScopedDevice sd(dev);
  hipError_t e = hipMemcpy(dst, src, n, kind);
  if (e != hipSuccess) set_err("hipMemcpy", e);
}

// ---------------- fill ----------------

__global__ void fill_kernel(uint64_t* dst, size_t nwords, uint64_t pattern) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < nwords; i += stride) dst[i] = pattern;
}

extern "C" int bam_gpu_fill(void* dst_dev, size_t n, uint64_t pattern, int dev) {
  ScopedDevice sd(dev);
  size_t nwords = n / 8;
  if (nwords > 0) {
    int blocks = (int)((nwords + 255) / 256);
    if (blocks > 2048) blocks = 2048;  // grid-stride beyond
    hipLaunchKernelGGL(fill_kernel, dim3(blocks), dim3(256), 0, 0, (uint64_t*)dst_dev,
                       nwords, pattern);
  }
  size_t tail = n % 8;
  if (tail != 0) {
    char tail_bytes[8];
    ::memcpy(tail_bytes, &pattern, 8);
    hipMemcpy((char*)dst_dev + n - tail, tail_bytes, tail, hipMemcpyHostToDevice);
  }
  hipError_t e = hipDeviceSynchronize();
  if (e != hipSuccess) {
    set_err("fill", e);
    return -1;
  }
  return 0;
}

// ---------------- scattered-span D2H staging ----------------
// Stages N device spans into one contiguous HOST buffer: device-side
// gather into a persistent scratch (≈4 TB/s span-copy kernel), then ONE
// D2H transfer. This is the pinned-staging-ring leg of the IOBuf write
// path (cut_into_file_descriptor with HBM-resident blocks).
extern "C" int bam_gpu_gather(void*, const void* const*, const size_t*, int, int);

namespace {
struct StageScratch {
  void* dev = nullptr;
  size_t cap = 0;
};
StageScratch g_stage[kMaxDev];
std::mutex g_stage_mu;
}  // namespace

extern "C" int bam_gpu_gather_to_host(void* host_dst, const void* const* srcs,
                                      const size_t* lens, int nspans, int dev) {
  size_t total = 0;
  for (int i = 0; i < nspans; ++i) total += lens[i];
  if (total == 0) return 0;
  ScopedDevice sd(dev);
  if (nspans == 1) {
    hipError_t e = hipMemcpy(host_dst, srcs[0], total, hipMemcpyDeviceToHost);
    return e == hipSuccess ? 0 : -1;
  }
  void* scratch;
  {
    std::lock_guard<std::mutex> lk(g_stage_mu);
    StageScratch& s = g_stage[dev < kMaxDev ? dev : 0];
    if (s.cap < total) {
      if (s.dev != nullptr) hipFree(s.dev);
      s.cap = total * 2;
      if (hipMalloc(&s.dev, s.cap) != hipSuccess) {
        s.dev = nullptr;
        s.cap = 0;
        return -1;
      }
    }
    scratch = s.dev;
  }
  if (bam_gpu_gather(scratch, srcs, lens, nspans, dev) != 0) return -1;
  hipError_t e = hipMemcpy(host_dst, scratch, total, hipMemcpyDeviceToHost);
  return e == hipSuccess ? 0 : -1;
}
