#include "base/gpu_loader.h"

#include <dlfcn.h>
#include <string.h>

#include <mutex>
#include <string>

#include "base/iobuf.h"
#include "base/logging.h"
#include "fiber/gpu_wait.h"

namespace bam {
namespace gpu {

namespace {

GpuApi g_api;
bool g_loaded = false;
int g_ndev = 0;
std::string g_error;
std::once_flag g_flag;

std::string lib_path() {
  const char* env = getenv("BAM_HIP_LIB");
  if (env != nullptr) return env;
  // Next to this shared object (the package directory).
  Dl_info info;
  if (dladdr((void*)&lib_path, &info) != 0 && info.dli_fname != nullptr) {
    std::string p = info.dli_fname;
    size_t slash = p.find_last_of('/');
    if (slash != std::string::npos) return p.substr(0, slash + 1) + "libbrpc_hip.so";
  }
  return "libbrpc_hip.so";
}

void* must_sym(void* h, const char* name) {
  void* s = dlsym(h, name);
  if (s == nullptr) g_error = std::string("missing symbol ") + name;
  return s;
}

void do_load() {
  std::string path = lib_path();
  void* h = dlopen(path.c_str(), RTLD_NOW | RTLD_GLOBAL);
  if (h == nullptr) {
    g_error = std::string("dlopen ") + path + ": " + dlerror();
    return;
  }
  g_api.device_count = (int (*)(void))must_sym(h, "bam_gpu_device_count");
  g_api.alloc_hbm = (void* (*)(uint32_t, int))must_sym(h, "bam_gpu_alloc_hbm");
  g_api.free_hbm = (void (*)(void*, uint32_t, int))must_sym(h, "bam_gpu_free_hbm");
  g_api.alloc_pinned = (void* (*)(uint32_t, int))must_sym(h, "bam_gpu_alloc_pinned");
  g_api.free_pinned = (void (*)(void*, uint32_t, int))must_sym(h, "bam_gpu_free_pinned");
  g_api.memcpy_res =
      (void (*)(void*, int, int, const void*, int, int, size_t))must_sym(h, "bam_gpu_memcpy");
  g_api.crc32c = (uint32_t (*)(const void*, size_t, uint32_t, int))must_sym(h, "bam_gpu_crc32c");
  g_api.gather =
      (int (*)(void*, const void* const*, const size_t*, int, int))must_sym(h, "bam_gpu_gather");
  g_api.scatter =
      (int (*)(void* const*, const size_t*, int, const void*, int))must_sym(h, "bam_gpu_scatter");
  g_api.fill = (int (*)(void*, size_t, uint64_t, int))must_sym(h, "bam_gpu_fill");
  g_api.gather_to_host = (int (*)(void*, const void* const*, const size_t*, int,
                                  int))must_sym(h, "bam_gpu_gather_to_host");
  g_api.upload_async =
      (int (*)(void*, const void*, size_t, int))must_sym(h, "bam_gpu_upload_async");
  g_api.quiesce = (void (*)(int))must_sym(h, "bam_gpu_quiesce");
  g_api.snappy_compress = (int (*)(const void*, size_t, void*, size_t, size_t*, int))must_sym(
      h, "bam_gpu_snappy_compress");
  g_api.snappy_decompress = (int (*)(const void*, size_t, void*, size_t, size_t*,
                                     int))must_sym(h, "bam_gpu_snappy_decompress");
  g_api.last_error = (const char* (*)(void))must_sym(h, "bam_gpu_last_error");
  if (!g_error.empty()) return;

  // RCCL comm surface — optional (older lib builds lack it).
  g_api.comm_uid = (int (*)(char*))dlsym(h, "bam_comm_uid");
  g_api.comm_create = (void* (*)(int, int, const char*, int))dlsym(h, "bam_comm_create");
  g_api.comm_destroy = (void (*)(void*))dlsym(h, "bam_comm_destroy");
  g_api.comm_rank = (int (*)(void*))dlsym(h, "bam_comm_rank");
  g_api.comm_nranks = (int (*)(void*))dlsym(h, "bam_comm_nranks");
  g_api.comm_broadcast = (int (*)(void*, void*, size_t, int))dlsym(h, "bam_comm_broadcast");
  g_api.comm_allgather =
      (int (*)(void*, const void*, void*, size_t))dlsym(h, "bam_comm_allgather");
  g_api.comm_send = (int (*)(void*, const void*, size_t, int))dlsym(h, "bam_comm_send");
  g_api.comm_recv = (int (*)(void*, void*, size_t, int))dlsym(h, "bam_comm_recv");
  g_api.comm_sendrecv = (int (*)(void*, const void*, size_t, int, void*, size_t,
                                 int))dlsym(h, "bam_comm_sendrecv");
  g_api.comm_last_error = (const char* (*)(void))dlsym(h, "bam_comm_last_error");
  g_api.stats_text = (const char* (*)(void))dlsym(h, "bam_gpu_stats_text");

  // Fiber↔stream integration (fiber/gpu_wait.h): give the HIP lib a
  // park/wake pair so its ticket waits yield the worker instead of
  // spinning; give the core the wake-marker enqueue hook.
  auto set_fiber_wait = (void (*)(int (*)(const volatile unsigned long long*,
                                          unsigned long long, int, int),
                                  void (*)(int, int)))dlsym(h, "bam_gpu_set_fiber_wait");
  auto request_wake = (int (*)(int, int))dlsym(h, "bam_gpu_request_wake");
  if (set_fiber_wait != nullptr && request_wake != nullptr) {
    gpu_wait_set_request_fn(request_wake);
    set_fiber_wait(
        [](const volatile unsigned long long* flag, unsigned long long want, int dev,
           int kind) { return gpu_fiber_wait_u64(flag, (uint64_t)want, dev, kind); },
        gpu_fiber_wake);
  }

  g_ndev = g_api.device_count();
  g_loaded = true;
  if (g_ndev > 0) {
    // Wire the residency backends into IOBuf.
    BlockMemFns hbm{g_api.alloc_hbm, g_api.free_hbm};
    set_block_allocator(RES_HBM, hbm);
    BlockMemFns pinned{g_api.alloc_pinned, g_api.free_pinned};
    set_block_allocator(RES_PINNED, pinned);
    ByteMoverFns mover{[](void* dst, Residency dres, int ddev, const void* src, Residency sres,
                          int sdev, size_t n) {
      g_api.memcpy_res(dst, (int)dres, ddev, src, (int)sres, sdev, n);
    }};
    set_byte_mover(mover);
    set_gather_to_host(g_api.gather_to_host);
    set_upload_async(g_api.upload_async);
    LOG(INFO) << "brpc_amd HIP runtime loaded: " << g_ndev << " GPU(s)";
  }
}

}  // namespace

int initialize() {
  std::call_once(g_flag, do_load);
  return g_ndev;
}

bool loaded() {
  initialize();
  return g_loaded;
}

const GpuApi* api() { return loaded() && g_ndev > 0 ? &g_api : nullptr; }

const char* load_error() {
  initialize();
  return g_error.c_str();
}

int device_count() { return initialize(); }

}  // namespace gpu
}  // namespace bam
