// brpc_amd: IOBuf — non-contiguous zero-copy byte buffer.
//
// Capability parity with reference butil/iobuf.h (IOBuf/IOPortal/BlockRef
// model, TLS block sharing, scatter-gather fd I/O), redesigned MI355X-first:
// every Block carries a residency tag (HOST / PINNED / HBM) + device index,
// and block memory comes from pluggable per-residency allocators so payload
// blocks can live in 288 GB HBM3E while header bytes stay host-side. The
// byte-hot copy paths dispatch to gfx950 kernels (hip/) when blocks are
// device-resident; cut_into_file_descriptor stages HBM bytes through a
// pinned-host ring (see hip/gpu_api.h).
#pragma once

#include <stdint.h>
#include <sys/uio.h>

#include <atomic>
#include <string>

namespace bam {

enum Residency : uint8_t {
  RES_HOST = 0,
  RES_PINNED = 1,  // hipHostMalloc — visible to both CPU and GPU
  RES_HBM = 2,     // device memory on GPU `dev`
};

// Per-residency block payload allocator. HOST has a built-in malloc path;
// the HIP runtime library registers PINNED/HBM at load time
// (gpu::install_block_allocators).
struct BlockMemFns {
  void* (*alloc)(uint32_t cap, int dev);
  void (*dealloc)(void* ptr, uint32_t cap, int dev);
};
void set_block_allocator(Residency res, BlockMemFns fns);
bool has_block_allocator(Residency res);

// Byte movers between residencies; defaults to memcpy for host<->host and
// aborts for device until the HIP library installs real ones.
typedef int (*GatherToHostFn)(void* host_dst, const void* const* srcs, const size_t* lens,
                              int nspans, int dev);
void set_gather_to_host(GatherToHostFn fn);

struct ByteMoverFns {
  // dst_res/src_res in {HOST,PINNED,HBM}; devices are -1 for host memory.
  void (*copy)(void* dst, Residency dst_res, int dst_dev, const void* src, Residency src_res,
               int src_dev, size_t n);
};
void set_byte_mover(ByteMoverFns fns);

// Optional fire-and-forget host->HBM upload (pinned staging ring + copy
// kernel, no host sync). Returns nonzero when the caller must fall back to
// the synchronous byte mover. Installed by the HIP loader.
typedef int (*UploadAsyncFn)(void* dst_dev, const void* src_host, size_t n, int dev);
void set_upload_async(UploadAsyncFn fn);

class IOBuf {
 public:
  static const uint32_t kDefaultBlockPayload = 8192;  // multiple of 4096
  static const uint32_t kInlineRefs = 4;              // power of two

  struct Block;  // opaque outside iobuf.cc except for pool stats

  struct BlockRef {
    uint32_t offset;
    uint32_t length;
    Block* block;
  };

  IOBuf();
  ~IOBuf() { clear(); }
  IOBuf(const IOBuf& rhs);
  IOBuf& operator=(const IOBuf& rhs);
  IOBuf(IOBuf&& rhs) noexcept;
  IOBuf& operator=(IOBuf&& rhs) noexcept;

  // ---- size / introspection ----
  size_t size() const { return nbytes_; }
  bool empty() const { return nbytes_ == 0; }
  size_t backing_block_num() const { return count_; }
  // True if every referenced block is CPU-addressable (HOST or PINNED).
  bool cpu_addressable() const;
  // Sum of bytes that live in HBM blocks.
  size_t hbm_bytes() const;

  void clear();
  void swap(IOBuf& rhs);

  // ---- appending ----
  void append(const void* data, size_t n);        // copy via TLS shared blocks
  void append(const std::string& s) { append(s.data(), s.size()); }
  void append(const char* s);
  void append(const IOBuf& other);                // zero-copy ref share
  void append(IOBuf&& other);                     // steal refs
  void push_back(char c) { append(&c, 1); }
  // Zero-copy adoption of user-owned memory; deleter(ptr) runs when the
  // last reference drops. meta is carried for transports (e.g. RDMA lkey).
  int append_user_data(void* data, size_t n, void (*deleter)(void*), uint64_t meta = 0);
  // Allocate a fresh block with the given residency/device and copy host
  // data into it (uses byte mover for HBM). Appends the written range.
  int append_with_residency(const void* host_data, size_t n, Residency res, int dev,
                            uint32_t block_payload = 0);
  // Allocate ONE uninitialized block of n bytes with the given residency,
  // append it, and return its raw pointer (for transports that land
  // payloads straight into place: RCCL recv, GPUDirect). n ≤ 2 GiB.
  int append_writable_block(size_t n, Residency res, int dev, void** out_ptr);
  int append_device_block(size_t n, int dev, void** out_ptr);  // = RES_HBM
  // True if any block of `res` residency backs this buffer (used by the
  // socket write path to route HBM payloads through the KeepWrite fiber
  // so consecutive responses stage in ONE device gather).
  bool has_residency(Residency res) const;

  // ---- cutting (front) ----
  // Move up to n bytes from the front of *this to the back of *out.
  size_t cutn(IOBuf* out, size_t n);
  size_t cutn(void* out, size_t n);        // copies to host memory
  size_t cutn(std::string* out, size_t n);
  int cut1(char* c);
  // Cuts everything up to AND INCLUDING the first occurrence of `delim`
  // into *out (delim itself is consumed but not copied). Returns 0, or -1
  // if the delimiter is absent (parity: reference IOBuf::cut_until).
  int cut_until(IOBuf* out, const std::string& delim);
  size_t pop_front(size_t n);
  size_t pop_back(size_t n);

  // ---- copying without consuming ----
  size_t copy_to(void* buf, size_t n = (size_t)-1L, size_t pos = 0) const;
  size_t copy_to(std::string* s, size_t n = (size_t)-1L, size_t pos = 0) const;
  std::string to_string() const;
  // Returns a pointer to the first n bytes: either directly into the first
  // block (no copy) or after copying them into aux. nullptr if size()<n or
  // the data is not CPU-addressable.
  const void* fetch(void* aux, size_t n) const;

  // ---- fd scatter/gather I/O ----
  // writev up to front bytes; pops what was written. Returns bytes written
  // or -1 (errno set). HBM blocks are staged through the pinned ring.
  ssize_t cut_into_file_descriptor(int fd, size_t size_hint = (size_t)-1L);
  // readv-append up to max_read bytes into tail/new blocks. Returns bytes
  // read (0 on EOF) or -1.
  ssize_t append_from_file_descriptor(int fd, size_t max_read);

  // ---- internals exposed for transports & kernels ----
  const BlockRef& ref_at(size_t i) const { return refs_[(begin_ + i) & (cap_ - 1)]; }
  // Raw span of ref i: pointer (host or device), length, residency, device.
  struct Span {
    char* data;
    uint32_t length;
    Residency res;
    int dev;
  };
  Span span_at(size_t i) const;

  bool equals(const IOBuf& other) const;
  bool equals(const std::string& s) const;

  static size_t block_count();    // live blocks (gauge)
  static size_t block_memory();   // bytes held by live blocks

 private:
  void add_ref(const BlockRef& r);        // takes ownership of one refcount
  void push_ref_back(const BlockRef& r);  // ring push (refcount already owned)
  void grow(uint32_t min_cap);
  BlockRef& mutable_ref_at(size_t i) { return refs_[(begin_ + i) & (cap_ - 1)]; }
  void pop_front_ref();
  void pop_back_ref();

  BlockRef inline_[kInlineRefs];
  BlockRef* refs_;   // inline_ or heap array, capacity cap_ (power of 2)
  uint32_t cap_;
  uint32_t begin_;
  uint32_t count_;
  size_t nbytes_;
};

// Release this thread's cached partial block + freelist (tests / shutdown).
void iobuf_flush_tls_cache();

}  // namespace bam
