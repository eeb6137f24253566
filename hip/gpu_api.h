// brpc_amd: C ABI of libbrpc_hip.so (hipcc-built, gfx950-only).
// The core runtime dlopens this library when a GPU is present and wires it
// into IOBuf (block allocators + byte movers) and the checksum/compress
// registries. No torch dependency — plain HIP runtime.
#pragma once

#include <stddef.h>
#include <stdint.h>

extern "C" {

// Returns #GPUs (0 = none / init failed). Safe to call repeatedly.
int bam_gpu_device_count(void);

// ---- block memory (IOBuf residency backends) ----
// HBM pools use size classes {8 KiB, 64 KiB, 2 MiB} carved from big slabs
// (parity: reference rdma/block_pool.cpp registered-memory pool design).
void* bam_gpu_alloc_hbm(uint32_t cap, int dev);
void bam_gpu_free_hbm(void* p, uint32_t cap, int dev);
void* bam_gpu_alloc_pinned(uint32_t cap, int dev);
void bam_gpu_free_pinned(void* p, uint32_t cap, int dev);

// ---- byte movement ----
// res codes match bam::Residency: 0 host, 1 pinned, 2 hbm.
void bam_gpu_memcpy(void* dst, int dst_res, int dst_dev, const void* src, int src_res,
                    int src_dev, size_t n);

// ---- gfx950 kernels ----
// CRC32-C of device memory; init follows the host convention (0 for fresh).
uint32_t bam_gpu_crc32c(const void* dev_ptr, size_t n, uint32_t init, int dev);

// Gather scattered device spans into one contiguous device buffer.
// srcs/lens are host arrays of device pointers/lengths.
int bam_gpu_gather(void* dst_dev, const void* const* srcs, const size_t* lens, int nspans,
                   int dev);
// Scatter a contiguous device buffer into device spans.
int bam_gpu_scatter(void* const* dsts, const size_t* lens, int nspans, const void* src_dev,
                    int dev);

// Stages N device spans into one contiguous host buffer (device gather +
// single D2H). The write path's staging ring.
int bam_gpu_gather_to_host(void* host_dst, const void* const* srcs, const size_t* lens,
                           int nspans, int dev);

// Snappy (standard wire format) on device buffers. Returns 0 on success.
int bam_gpu_snappy_compress(const void* src_dev, size_t n, void* dst_dev, size_t dst_cap,
                            size_t* out_len, int dev);
int bam_gpu_snappy_decompress(const void* src_dev, size_t n, void* dst_dev, size_t dst_cap,
                              size_t* out_len, int dev);

// Fill device memory with a repeating 64-bit pattern (tests / synthetic
// payload generation without H2D traffic).
int bam_gpu_fill(void* dst_dev, size_t n, uint64_t pattern, int dev);

// ---- fiber-wait integration (see src/fiber/gpu_wait.h) ----
// wait(flag, want, dev, kind): park the calling fiber until *flag >= want.
// Returns 0 once satisfied; nonzero = could not park (caller falls back to
// spinning). wake(dev, kind) is invoked from a HIP host-callback thread
// when the stream carrying (dev, kind) reaches a requested wake marker.
typedef int (*bam_fiber_wait_fn)(const volatile unsigned long long* flag,
                                 unsigned long long want, int dev, int kind);
typedef void (*bam_fiber_wake_fn)(int dev, int kind);
void bam_gpu_set_fiber_wait(bam_fiber_wait_fn wait, bam_fiber_wake_fn wake);
// Enqueues a wake marker on (dev, kind)'s stream; 0 on success. Stream
// order guarantees the marker fires after every op enqueued before the
// call — in particular after the kernel that publishes the waited ticket.
int bam_gpu_request_wake(int dev, int kind);

// ---- RCCL collectives over xGMI (hip/comm.hip) ----
// One communicator per (process, GPU); ops enqueue on the communicator's
// dedicated stream and PARK the calling fiber on the completion ticket.
// Buffers are device (HBM) pointers.
int bam_comm_uid(char out[128]);
void* bam_comm_create(int nranks, int rank, const char uid[128], int dev);
void bam_comm_destroy(void* comm);
int bam_comm_rank(void* comm);
int bam_comm_nranks(void* comm);
int bam_comm_broadcast(void* comm, void* buf_dev, size_t n, int root);
int bam_comm_allgather(void* comm, const void* send_dev, void* recv_dev, size_t per_rank);
int bam_comm_send(void* comm, const void* buf_dev, size_t n, int peer);
int bam_comm_recv(void* comm, void* buf_dev, size_t n, int peer);
int bam_comm_sendrecv(void* comm, const void* sbuf, size_t sn, int speer, void* rbuf,
                      size_t rn, int rpeer);
const char* bam_comm_last_error(void);

// Last error string (static buffer).
const char* bam_gpu_last_error(void);

// Fire-and-forget small H2D into HBM via the pinned staging ring + copy
// kernel on the per-device staging stream (no host synchronization).
// Returns nonzero if unavailable — caller must fall back to bam_gpu_memcpy.
// Ordering: later direct gathers are stream-ordered after it; every other
// HBM entry point drains pending uploads via bam_gpu_quiesce first.
int bam_gpu_upload_async(void* dst_dev, const void* src_host, size_t n, int dev);
void bam_gpu_quiesce(int dev);

}  // extern "C"
