// brpc_amd: GPU runtime — device discovery, HBM block pools, pinned pools,
// residency-aware memcpy. gfx950-only, no CUDA paths, no torch.
//
// HBM pool design (capability parity with reference brpc/rdma/block_pool.cpp
// which registers {8K,64K,2M} blocks for verbs): hipMalloc 256 MiB slabs,
// carve into size-class blocks, per-class freelists. 288 GB HBM3E per GPU
// means we bias toward big slabs and never return memory to the runtime.
#include <hip/hip_runtime.h>
#include <stdlib.h>
#include <string.h>

#include <atomic>
#include <mutex>
#include <vector>

#include "gpu_api.h"
#include "internal.h"

static char g_err[256] = {0};

static void set_err(const char* what, hipError_t e) {
  snprintf(g_err, sizeof(g_err), "%s: %s", what, hipGetErrorString(e));
}

extern "C" const char* bam_gpu_last_error(void) { return g_err; }

// ---- telemetry (rendered by the /hotspots/gpu builtin page) ----
namespace {
std::atomic<uint64_t> g_stat_gathers{0};
std::atomic<uint64_t> g_stat_gather_bytes{0};
std::atomic<uint64_t> g_stat_gather_fallbacks{0};
std::atomic<uint64_t> g_stat_uploads{0};
std::atomic<uint64_t> g_stat_upload_bytes{0};
std::atomic<uint64_t> g_stat_hbm_blocks{0};
std::atomic<uint64_t> g_stat_hbm_bytes{0};
std::atomic<uint64_t> g_stat_pstage_gathers{0};
std::atomic<uint64_t> g_stat_pstage_uploads{0};
}  // namespace

extern "C" const char* bam_gpu_stats_text(void) {
  static char buf[512];
  snprintf(buf, sizeof(buf),
           "gather_batches: %llu\ngather_bytes: %llu\ngather_fallbacks: %llu\n"
           "pstage_gathers: %llu\npstage_uploads: %llu\n"
           "uploads_async: %llu\nupload_bytes: %llu\n"
           "hbm_blocks_live: %llu\nhbm_bytes_live: %llu\n",
           (unsigned long long)g_stat_gathers.load(),
           (unsigned long long)g_stat_gather_bytes.load(),
           (unsigned long long)g_stat_gather_fallbacks.load(),
           (unsigned long long)g_stat_pstage_gathers.load(),
           (unsigned long long)g_stat_pstage_uploads.load(),
           (unsigned long long)g_stat_uploads.load(),
           (unsigned long long)g_stat_upload_bytes.load(),
           (unsigned long long)g_stat_hbm_blocks.load(),
           (unsigned long long)g_stat_hbm_bytes.load());
  return buf;
}

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
    (void)hipGetDevice(&old);
    if (dev != old) (void)hipSetDevice(dev);
    else old = -1;
  }
  ~ScopedDevice() {
    if (old >= 0) (void)hipSetDevice(old);
  }
};

}  // namespace

// ---------------- fiber-wait / wake infrastructure ----------------
// See internal.h. The core runtime (src/fiber/gpu_wait.cc) registers a
// park/wake pair; until it does, waits fall back to bounded spinning.

namespace {

bam_fiber_wait_fn g_fiber_wait = nullptr;
bam_fiber_wake_fn g_fiber_wake = nullptr;

struct WakeSlot {
  hipStream_t stream = nullptr;  // set once at stream creation
  int dev = 0;
  int kind = 0;
};
WakeSlot g_wake[kMaxDev][bamhip::kWakeKinds];
std::mutex g_wake_mu;

void wake_trampoline(void* p) {
  WakeSlot* s = (WakeSlot*)p;
  bam_fiber_wake_fn wk = __atomic_load_n(&g_fiber_wake, __ATOMIC_ACQUIRE);
  if (wk != nullptr) wk(s->dev, s->kind);
}

}  // namespace

namespace bamhip {

void register_wake_stream(int dev, int kind, hipStream_t stream) {
  if (dev < 0 || dev >= kMaxDev || kind < 0 || kind >= kWakeKinds) return;
  std::lock_guard<std::mutex> lk(g_wake_mu);
  g_wake[dev][kind].dev = dev;
  g_wake[dev][kind].kind = kind;
  __atomic_store_n(&g_wake[dev][kind].stream, stream, __ATOMIC_RELEASE);
}

bool wait_ticket(const volatile unsigned long long* flag, unsigned long long want,
                 int dev, int kind, hipStream_t sync_stream) {
  // Short spin first: small staging batches complete in single-digit µs,
  // far below a park/wake round trip.
  for (int spin = 0; spin < 4000; ++spin) {
    if (*flag >= want) return true;
#if defined(__x86_64__)
    __builtin_ia32_pause();
#endif
  }
  bam_fiber_wait_fn fw = __atomic_load_n(&g_fiber_wait, __ATOMIC_ACQUIRE);
  if (fw != nullptr && fw(flag, want, dev, kind) == 0 && *flag >= want) return true;
  // No fiber runtime (plain pthread) or park unavailable: bounded spin with
  // a hard stream sync as last resort.
  for (uint64_t spin = 0; *flag < want; ++spin) {
    if (spin > 4000000) {
      if (sync_stream == nullptr) return false;
      if (hipStreamSynchronize(sync_stream) != hipSuccess || *flag < want) return false;
      break;
    }
#if defined(__x86_64__)
    __builtin_ia32_pause();
#endif
  }
  return true;
}

}  // namespace bamhip

extern "C" void bam_gpu_set_fiber_wait(bam_fiber_wait_fn wait, bam_fiber_wake_fn wake) {
  __atomic_store_n(&g_fiber_wake, wake, __ATOMIC_RELEASE);
  __atomic_store_n(&g_fiber_wait, wait, __ATOMIC_RELEASE);
}

extern "C" int bam_gpu_request_wake(int dev, int kind) {
  if (dev < 0 || dev >= kMaxDev || kind < 0 || kind >= bamhip::kWakeKinds) return -1;
  WakeSlot* s = &g_wake[dev][kind];
  hipStream_t stream = __atomic_load_n(&s->stream, __ATOMIC_ACQUIRE);
  if (stream == nullptr) return -1;
  return hipLaunchHostFunc(stream, wake_trampoline, s) == hipSuccess ? 0 : -1;
}

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
  g_stat_hbm_blocks.fetch_add(1, std::memory_order_relaxed);
  g_stat_hbm_bytes.fetch_add(kClasses[cls], std::memory_order_relaxed);
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
    (void)hipFree(p);
    return;
  }
  DevPool& pool = g_pools[dev];
  std::lock_guard<std::mutex> lk(pool.mu);
  g_stat_hbm_blocks.fetch_sub(1, std::memory_order_relaxed);
  g_stat_hbm_bytes.fetch_sub(kClasses[cls], std::memory_order_relaxed);
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
  (void)hipHostFree(p);
}

// ---------------- memcpy ----------------

// Drains any in-flight async HBM uploads on dev's staging stream (defined
// with the direct-gather state below). Every entry point that touches HBM
// outside that stream must call this first for write→read ordering.
extern "C" void bam_gpu_quiesce(int dev);

extern "C" void bam_gpu_memcpy(void* dst, int dst_res, int dst_dev, const void* src,
                               int src_res, int src_dev, size_t n) {
  const bool dst_dev_mem = dst_res == 2;
  const bool src_dev_mem = src_res == 2;
  if (!dst_dev_mem && !src_dev_mem) {
    ::memcpy(dst, src, n);
    return;
  }
  bam_gpu_quiesce(dst_dev_mem ? dst_dev : src_dev);
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
  bam_gpu_quiesce(dev);
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
    if (hipMemcpy((char*)dst_dev + n - tail, tail_bytes, tail,
                  hipMemcpyHostToDevice) != hipSuccess) {
      return -1;
    }
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

// --- direct-to-pinned fast path (latency-bound small batches) ---
// A blocking hipMemcpy / hipDeviceSynchronize costs 10-20 µs each in host
// wakeup latency; at ~100k batches/s that IS the staging ceiling. For small
// batch totals a single kernel instead writes the scattered HBM spans
// straight into the pinned bounce buffer (hipHostMalloc memory is
// fine-grained coherent on ROCm: the device dereferences the host VA) and
// then raises a pinned completion flag that the host spin-waits on — no
// runtime synchronization call at all.
namespace {

struct DirectSpan {
  const char* src;
  unsigned int len;
};
constexpr int kDirectMaxSpans = 64;  // matches IOBuf kMaxIov

struct DirectArgs {
  DirectSpan spans[kDirectMaxSpans];
  char* dst;                            // pinned host buffer (device-visible VA)
  unsigned long long* counter;          // persistent device counter, monotonic
  volatile unsigned long long* flag;    // pinned host flag (device-visible VA)
  unsigned long long expect;            // counter value once ALL blocks of this call ran
  unsigned long long ticket;            // value the last block publishes to *flag
  int nspans;
};

__global__ void gather_direct_kernel(DirectArgs a) {
  const int i = blockIdx.x;
  const DirectSpan s = a.spans[i];
  size_t off = 0;
  for (int k = 0; k < i; ++k) off += a.spans[k].len;
  const char* src = s.src;
  char* dst = a.dst + off;
  const unsigned int tid = threadIdx.x;
  const unsigned int nt = blockDim.x;
  if ((((uintptr_t)src ^ (uintptr_t)dst) & 15) == 0) {
    unsigned int head = (16 - ((uintptr_t)src & 15)) & 15;
    if (head > s.len) head = s.len;
    for (unsigned int j = tid; j < head; j += nt) dst[j] = src[j];
    const unsigned int nvec = (s.len - head) / 16;
    const uint4* vs = (const uint4*)(src + head);
    uint4* vd = (uint4*)(dst + head);
    for (unsigned int j = tid; j < nvec; j += nt) vd[j] = vs[j];
    for (unsigned int j = head + nvec * 16 + tid; j < s.len; j += nt) dst[j] = src[j];
  } else {
    for (unsigned int j = tid; j < s.len; j += nt) dst[j] = src[j];
  }
  __syncthreads();
  if (tid == 0) {
    __threadfence_system();  // publish this block's host writes before counting it done
    unsigned long long done = atomicAdd(a.counter, 1ull) + 1;
    if (done == a.expect) {
      *a.flag = a.ticket;
      __threadfence_system();
    }
  }
}

// Slim single-span variant: used for 1-span gathers and for the async
// HBM upload path (pinned ring slot -> HBM block) — 1 KiB less kernarg
// traffic per launch than the 64-span struct.
__global__ void copy1_kernel(const char* src, char* dst, unsigned int len,
                             unsigned long long* counter,
                             volatile unsigned long long* flag,
                             unsigned long long expect, unsigned long long ticket) {
  const unsigned int tid = threadIdx.x;
  const unsigned int nt = blockDim.x;
  if ((((uintptr_t)src ^ (uintptr_t)dst) & 15) == 0) {
    unsigned int head = (16 - ((uintptr_t)src & 15)) & 15;
    if (head > len) head = len;
    for (unsigned int j = tid; j < head; j += nt) dst[j] = src[j];
    const unsigned int nvec = (len - head) / 16;
    const uint4* vs = (const uint4*)(src + head);
    uint4* vd = (uint4*)(dst + head);
    for (unsigned int j = tid; j < nvec; j += nt) vd[j] = vs[j];
    for (unsigned int j = head + nvec * 16 + tid; j < len; j += nt) dst[j] = src[j];
  } else {
    for (unsigned int j = tid; j < len; j += nt) dst[j] = src[j];
  }
  __syncthreads();
  if (tid == 0) {
    __threadfence_system();
    unsigned long long done = atomicAdd(counter, 1ull) + 1;
    if (done == expect) {
      *flag = ticket;
      __threadfence_system();
    }
  }
}

constexpr int kUploadSlots = 256;
constexpr size_t kUploadSlotBytes = 32u << 10;

struct DirectState {
  std::mutex mu;
  hipStream_t stream = nullptr;
  unsigned long long* counter_dev = nullptr;
  volatile unsigned long long* flag = nullptr;  // pinned
  unsigned long long launched_blocks = 0;
  unsigned long long ticket = 0;
  int status = 0;  // 0 = untried, 1 = ok, -1 = unavailable
  // ---- async H2D upload leg: its OWN stream + bookkeeping. Uploads
  // pipeline among themselves without any host synchronization; a gather
  // (different stream) host-waits for the upload flag to catch its ticket
  // before launching, which is almost always already true. Sharing ONE
  // stream was measured 30% slower: every batch gather drained the whole
  // upload queue in stream order.
  std::mutex up_mu;
  hipStream_t up_stream = nullptr;
  unsigned long long* up_counter_dev = nullptr;
  volatile unsigned long long* up_flag = nullptr;  // pinned
  unsigned long long up_launched = 0;
  std::atomic<unsigned long long> up_ticket{0};
  char* ring = nullptr;  // pinned staging ring (slot reusable once up_flag passes its ticket)
  unsigned long long slot_ticket[kUploadSlots] = {0};
  unsigned long long nslots_used = 0;
};
DirectState g_direct[kMaxDev];

// Waits until the upload stream drained up to `want` (up_flag is pinned):
// short spin, then park the fiber (bamhip::wait_ticket).
bool upload_wait(DirectState& st, unsigned long long want, int dev) {
  return bamhip::wait_ticket(st.up_flag, want, dev, bamhip::kWakeUpload, st.up_stream);
}

// Waits until the device drained every gather launch up to `want`.
bool direct_wait(DirectState& st, unsigned long long want, int dev) {
  return bamhip::wait_ticket(st.flag, want, dev, bamhip::kWakeGather, st.stream);
}

size_t direct_max_bytes() {
  static size_t v = [] {
    const char* e = getenv("BAM_GATHER_DIRECT_MAX");
    // Default 2 MiB: spans are split into <=32 KiB sub-spans (one block
    // each), so large batches still spread across CUs. 64 blocks x 32 KiB.
    return e != nullptr ? (size_t)strtoull(e, nullptr, 10) : (size_t)(2u << 20);
  }();
  return v;
}

// Splits spans into <=kDirectSubSpan pieces so no single block serializes
// a megabyte over PCIe (measured: 1 MiB through one 256-thread block cuts
// streaming from 3.5 to 2.6 GB/s). Returns piece count, or -1 if it does
// not fit in kDirectMaxSpans.
constexpr unsigned int kDirectSubSpan = 32u << 10;

int split_spans(const void* const* srcs, const size_t* lens, int nspans, DirectArgs* a) {
  int out = 0;
  for (int i = 0; i < nspans; ++i) {
    size_t off = 0;
    while (off < lens[i]) {
      if (out >= kDirectMaxSpans) return -1;
      size_t piece = lens[i] - off;
      if (piece > kDirectSubSpan) piece = kDirectSubSpan;
      a->spans[out].src = (const char*)srcs[i] + off;
      a->spans[out].len = (unsigned int)piece;
      ++out;
      off += piece;
    }
  }
  return out;
}

// One-time per-device self-test: if the device cannot dereference pinned
// host VAs on this platform/config, disable the path for good.
bool direct_init(DirectState& st, int dev) {
  if (hipStreamCreateWithFlags(&st.stream, hipStreamNonBlocking) != hipSuccess) return false;
  bamhip::register_wake_stream(dev, bamhip::kWakeGather, st.stream);
  if (hipMalloc(&st.counter_dev, 8) != hipSuccess) return false;
  if (hipMemset(st.counter_dev, 0, 8) != hipSuccess) return false;
  void* f = nullptr;
  if (hipHostMalloc(&f, 64, hipHostMallocDefault) != hipSuccess) return false;
  st.flag = (volatile unsigned long long*)f;
  *st.flag = 0;
  void* ring = nullptr;
  if (hipHostMalloc(&ring, kUploadSlots * kUploadSlotBytes, hipHostMallocDefault) != hipSuccess)
    return false;
  st.ring = (char*)ring;
  if (hipStreamCreateWithFlags(&st.up_stream, hipStreamNonBlocking) != hipSuccess) return false;
  bamhip::register_wake_stream(dev, bamhip::kWakeUpload, st.up_stream);
  if (hipMalloc(&st.up_counter_dev, 8) != hipSuccess) return false;
  if (hipMemset(st.up_counter_dev, 0, 8) != hipSuccess) return false;
  void* uf = nullptr;
  if (hipHostMalloc(&uf, 64, hipHostMallocDefault) != hipSuccess) return false;
  st.up_flag = (volatile unsigned long long*)uf;
  *st.up_flag = 0;
  // self-test: gather 64 known bytes device -> pinned host
  char* src_dev = nullptr;
  if (hipMalloc(&src_dev, 64) != hipSuccess) return false;
  char pattern[64];
  for (int i = 0; i < 64; ++i) pattern[i] = (char)(i * 7 + 3);
  if (hipMemcpy(src_dev, pattern, 64, hipMemcpyHostToDevice) != hipSuccess) {
    (void)hipFree(src_dev);
    return false;
  }
  char* probe = nullptr;
  if (hipHostMalloc((void**)&probe, 64, hipHostMallocDefault) != hipSuccess) {
    (void)hipFree(src_dev);
    return false;
  }
  memset(probe, 0, 64);
  DirectArgs a{};
  a.spans[0].src = src_dev;
  a.spans[0].len = 64;
  a.dst = probe;
  a.counter = st.counter_dev;
  a.flag = st.flag;
  a.expect = 1;
  a.ticket = 1;
  a.nspans = 1;
  hipLaunchKernelGGL(gather_direct_kernel, dim3(1), dim3(256), 0, st.stream, a);
  bool ok = hipGetLastError() == hipSuccess &&
            hipStreamSynchronize(st.stream) == hipSuccess && *st.flag == 1 &&
            memcmp(probe, pattern, 64) == 0;
  st.launched_blocks = 1;
  st.ticket = 1;
  (void)hipFree(src_dev);
  (void)hipHostFree(probe);
  return ok;
}

// Returns 0 on success, nonzero to fall back to the scratch+memcpy path.
int gather_direct(void* host_dst, const void* const* srcs, const size_t* lens,
                  int nspans, int dev) {
  // host_dst must be pinned (device-dereferenceable): the IOBuf bounce is
  // normally RES_PINNED but falls back to malloc if hipHostMalloc failed.
  // The bounce base is thread-stable, so cache the attribute lookup.
  static thread_local struct {
    const void* p;
    bool ok;
  } tls_chk = {nullptr, false};
  if (tls_chk.p != host_dst) {
    hipPointerAttribute_t attr;
    tls_chk.ok = hipPointerGetAttributes(&attr, host_dst) == hipSuccess &&
                 attr.type == hipMemoryTypeHost;
    (void)hipGetLastError();
    tls_chk.p = host_dst;
  }
  if (!tls_chk.ok) return 1;
  DirectState& st = g_direct[dev < kMaxDev ? dev : 0];
  // Order after pending async uploads (separate stream): the blocks this
  // gather reads were uploaded strictly before the batch was packed, so
  // this wait is almost always already satisfied.
  {
    unsigned long long want = st.up_ticket.load(std::memory_order_acquire);
    if (want != 0 && *st.up_flag < want && !upload_wait(st, want, dev)) return 1;
  }
  unsigned long long ticket;
  {
    // Critical section covers ONLY ticket allocation + launch (so enqueue
    // order matches ticket order on the stream). The completion spin runs
    // outside the lock: holding st.mu while waiting would block every
    // concurrent handler upload — and with M:N fibers, a blocked pthread
    // mutex parks the whole worker thread.
    DirectArgs a{};
    const int npieces = split_spans(srcs, lens, nspans, &a);
    if (npieces < 0) return 1;  // too fragmented for the kernarg table
    std::lock_guard<std::mutex> lk(st.mu);
    if (st.status == 0) st.status = direct_init(st, dev) ? 1 : -1;
    if (st.status < 0) return 1;
    st.launched_blocks += (unsigned long long)npieces;
    const unsigned long long expect = st.launched_blocks;
    ticket = ++st.ticket;
    if (npieces == 1) {
      hipLaunchKernelGGL(copy1_kernel, dim3(1), dim3(256), 0, st.stream, a.spans[0].src,
                         (char*)host_dst, a.spans[0].len, st.counter_dev, st.flag, expect,
                         ticket);
    } else {
      a.dst = (char*)host_dst;
      a.counter = st.counter_dev;
      a.flag = st.flag;
      a.nspans = npieces;
      a.expect = expect;
      a.ticket = ticket;
      hipLaunchKernelGGL(gather_direct_kernel, dim3(npieces), dim3(256), 0, st.stream, a);
    }
    if (hipGetLastError() != hipSuccess) {
      st.status = -1;  // keep ticket/launched consistent by never using them again
      return 1;
    }
  }
  if (!direct_wait(st, ticket, dev)) {
    std::lock_guard<std::mutex> lk(st.mu);
    st.status = -1;
    return 1;
  }
  return 0;
}

// Fire-and-forget H2D: copies src into a pinned ring slot and enqueues a
// slot->HBM copy kernel on the staging stream WITHOUT waiting. Ordering
// with later reads comes from stream order (direct gather) or from
// bam_gpu_quiesce (every other HBM entry point). Returns nonzero to tell
// the caller to fall back to a synchronous copy.
int upload_direct(void* dst_dev, const void* src, size_t n, int dev) {
  if (n == 0) return 0;
  if (n > kUploadSlotBytes) return 1;
  DirectState& st = g_direct[dev < kMaxDev ? dev : 0];
  {
    std::lock_guard<std::mutex> lk(st.mu);  // init shares the gather lock
    if (st.status == 0) st.status = direct_init(st, dev) ? 1 : -1;
    if (st.status < 0) return 1;
  }
  std::lock_guard<std::mutex> lk(st.up_mu);
  const int slot = (int)(st.nslots_used++ % kUploadSlots);
  // ring full wrap: the previous occupant's kernel must have drained
  if (!upload_wait(st, st.slot_ticket[slot], dev)) return 1;
  char* sp = st.ring + (size_t)slot * kUploadSlotBytes;
  ::memcpy(sp, src, n);
  st.up_launched += 1;
  const unsigned long long expect = st.up_launched;
  const unsigned long long ticket = st.up_ticket.load(std::memory_order_relaxed) + 1;
  hipLaunchKernelGGL(copy1_kernel, dim3(1), dim3(256), 0, st.up_stream, sp, (char*)dst_dev,
                     (unsigned int)n, st.up_counter_dev, st.up_flag, expect, ticket);
  if (hipGetLastError() != hipSuccess) return 1;
  st.up_ticket.store(ticket, std::memory_order_release);
  st.slot_ticket[slot] = ticket;
  return 0;
}

}  // namespace

extern "C" void bam_gpu_quiesce(int dev) {
  if (dev < 0 || dev >= kMaxDev) dev = 0;
  DirectState& st = g_direct[dev];
  unsigned long long want;
  {
    std::lock_guard<std::mutex> lk(st.mu);
    if (st.status != 1) return;
    want = st.ticket;
  }
  unsigned long long up_want = st.up_ticket.load(std::memory_order_acquire);
  if (up_want != 0 && *st.up_flag < up_want && !upload_wait(st, up_want, dev)) {
    std::lock_guard<std::mutex> lk(st.mu);
    st.status = -1;
    return;
  }
  if (*st.flag >= want) return;
  if (!direct_wait(st, want, dev)) {
    std::lock_guard<std::mutex> lk(st.mu);
    st.status = -1;
  }
}

extern "C" int bam_gpu_pstage_copy(void* dst, const void* src, size_t n, int dev);

namespace {
// Per-thread pinned bounce for small pstage uploads (allocated once).
char* tls_upload_bounce(int dev) {
  static thread_local char* bounce = nullptr;
  if (bounce == nullptr) {
    void* p = nullptr;
    if (hipHostMalloc(&p, 4096, hipHostMallocDefault) != hipSuccess) return nullptr;
    bounce = (char*)p;
  }
  (void)dev;
  return bounce;
}
}  // namespace

extern "C" int bam_gpu_upload_async(void* dst_dev, const void* src, size_t n, int dev) {
  ScopedDevice sd(dev);
  // Synchronous persistent-kernel uploads LOST the A/B to the pipelined
  // stream ring (the wait serializes handler fibers): opt-in only.
  static const bool pstage_upload = [] {
    const char* e = getenv("BAM_PSTAGE_UPLOAD");
    return e != nullptr && e[0] == '1';
  }();
  if (pstage_upload && n <= 4096) {
    char* bounce = tls_upload_bounce(dev);
    if (bounce != nullptr) {
      ::memcpy(bounce, src, n);
      if (bam_gpu_pstage_copy(dst_dev, bounce, n, dev) == 0) {
        g_stat_uploads.fetch_add(1, std::memory_order_relaxed);
        g_stat_upload_bytes.fetch_add(n, std::memory_order_relaxed);
        g_stat_pstage_uploads.fetch_add(1, std::memory_order_relaxed);
        return 0;
      }
    }
  }
  int rc = upload_direct(dst_dev, src, n, dev);
  if (rc == 0) {
    g_stat_uploads.fetch_add(1, std::memory_order_relaxed);
    g_stat_upload_bytes.fetch_add(n, std::memory_order_relaxed);
  }
  return rc;
}

extern "C" int bam_gpu_pstage_gather(void* host_dst, const void* const* srcs,
                                     const size_t* lens, int nspans, size_t total, int dev);

extern "C" int bam_gpu_gather_to_host(void* host_dst, const void* const* srcs,
                                      const size_t* lens, int nspans, int dev) {
  size_t total = 0;
  for (int i = 0; i < nspans; ++i) total += lens[i];
  if (total == 0) return 0;
  g_stat_gathers.fetch_add(1, std::memory_order_relaxed);
  g_stat_gather_bytes.fetch_add(total, std::memory_order_relaxed);
  ScopedDevice sd(dev);
  // Small batches: the persistent staging kernel (hip/pstage.hip) skips
  // the per-batch launch entirely. It bypasses stream ordering, so drain
  // pending async uploads first (same requirement as the direct path).
  if (nspans <= 8 && total <= (4u << 10)) {
    DirectState& st = g_direct[dev >= 0 && dev < kMaxDev ? dev : 0];
    unsigned long long want = st.up_ticket.load(std::memory_order_acquire);
    if (want == 0 || *st.up_flag >= want || upload_wait(st, want, dev)) {
      if (bam_gpu_pstage_gather(host_dst, srcs, lens, nspans, total, dev) == 0) {
        g_stat_pstage_gathers.fetch_add(1, std::memory_order_relaxed);
        return 0;
      }
    }
  }
  if (total <= direct_max_bytes() && nspans <= kDirectMaxSpans &&
      gather_direct(host_dst, srcs, lens, nspans, dev) == 0) {
    return 0;
  }
  g_stat_gather_fallbacks.fetch_add(1, std::memory_order_relaxed);
  // Fallback paths read HBM outside the staging stream: drain uploads first.
  bam_gpu_quiesce(dev);
  if (nspans == 1) {
    hipError_t e = hipMemcpy(host_dst, srcs[0], total, hipMemcpyDeviceToHost);
    return e == hipSuccess ? 0 : -1;
  }
  void* scratch;
  {
    std::lock_guard<std::mutex> lk(g_stage_mu);
    StageScratch& s = g_stage[dev < kMaxDev ? dev : 0];
    if (s.cap < total) {
      if (s.dev != nullptr) (void)hipFree(s.dev);
      s.dev = nullptr;
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
