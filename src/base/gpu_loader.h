// brpc_amd: runtime loader for libbrpc_hip.so (the gfx950 kernel library).
// The core stays torch/HIP-free; when a GPU is present the library is
// dlopened and wired into IOBuf (PINNED/HBM allocators + byte mover) and
// the checksum registry. On a GPU machine a missing library is a HARD
// error — the HIP path must never silently fall back to host code.
#pragma once

#include <stddef.h>
#include <stdint.h>

namespace bam {
namespace gpu {

struct GpuApi {
  int (*device_count)(void);
  void* (*alloc_hbm)(uint32_t, int);
  void (*free_hbm)(void*, uint32_t, int);
  void* (*alloc_pinned)(uint32_t, int);
  void (*free_pinned)(void*, uint32_t, int);
  void (*memcpy_res)(void*, int, int, const void*, int, int, size_t);
  uint32_t (*crc32c)(const void*, size_t, uint32_t, int);
  int (*gather)(void*, const void* const*, const size_t*, int, int);
  int (*scatter)(void* const*, const size_t*, int, const void*, int);
  int (*fill)(void*, size_t, uint64_t, int);
  int (*gather_to_host)(void*, const void* const*, const size_t*, int, int);
  int (*upload_async)(void*, const void*, size_t, int);
  void (*quiesce)(int);
  int (*snappy_compress)(const void*, size_t, void*, size_t, size_t*, int);
  int (*snappy_decompress)(const void*, size_t, void*, size_t, size_t*, int);
  const char* (*last_error)(void);
  // RCCL collectives (hip/comm.hip); null on builds without librccl.
  int (*comm_uid)(char*);
  void* (*comm_create)(int, int, const char*, int);
  void (*comm_destroy)(void*);
  int (*comm_rank)(void*);
  int (*comm_nranks)(void*);
  int (*comm_broadcast)(void*, void*, size_t, int);
  int (*comm_allgather)(void*, const void*, void*, size_t);
  int (*comm_send)(void*, const void*, size_t, int);
  int (*comm_recv)(void*, void*, size_t, int);
  int (*comm_sendrecv)(void*, const void*, size_t, int, void*, size_t, int);
  const char* (*comm_last_error)(void);
  const char* (*stats_text)(void);  // telemetry for /hotspots/gpu (optional)
};

// Loads the library (idempotent). Returns device count (0 = no GPU or no
// library; check loaded() / error()). Registers IOBuf hooks on success.
int initialize();

bool loaded();
const GpuApi* api();          // nullptr until loaded
const char* load_error();     // empty if ok
int device_count();           // 0 when not loaded

}  // namespace gpu
}  // namespace bam
