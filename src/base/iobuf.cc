#include "base/iobuf.h"

#include <errno.h>
#include <stdlib.h>
#include <string.h>
#include <unistd.h>

#include <algorithm>
#include <vector>

#include "base/logging.h"

namespace bam {

// ---------------- Block & allocators ----------------

static std::atomic<size_t> g_nblock{0};
static std::atomic<size_t> g_blockmem{0};

BlockMemFns g_allocators[3] = {{nullptr, nullptr}, {nullptr, nullptr}, {nullptr, nullptr}};
static ByteMoverFns g_mover = {nullptr};

void set_block_allocator(Residency res, BlockMemFns fns) { g_allocators[res] = fns; }
bool has_block_allocator(Residency res) {
  return res == RES_HOST || g_allocators[res].alloc != nullptr;
}
void set_byte_mover(ByteMoverFns fns) { g_mover = fns; }

static UploadAsyncFn g_upload_async = nullptr;
void set_upload_async(UploadAsyncFn fn) {
  // Default ON since the upload leg moved to its OWN stream (uploads
  // pipeline with no host sync; the batch gather host-waits for the upload
  // ticket, which is almost always already passed). Same-box A/B:
  // 64B/c64 150.8k vs 131.2k QPS, 16KB/c32 118.0k vs 83.3k (+42%).
  // The first single-stream attempt LOST (gathers drained the upload queue
  // in stream order) — BAM_UPLOAD_ASYNC=0 restores sync hipMemcpy uploads.
  const char* e = getenv("BAM_UPLOAD_ASYNC");
  if (e != nullptr && e[0] == '0') return;
  g_upload_async = fn;
}

static void move_bytes(void* dst, Residency dres, int ddev, const void* src, Residency sres,
                       int sdev, size_t n) {
  if (dres != RES_HBM && sres != RES_HBM) {
    memcpy(dst, src, n);
    return;
  }
  CHECK(g_mover.copy != nullptr) << "HBM byte mover not installed (HIP runtime lib not loaded)";
  g_mover.copy(dst, dres, ddev, src, sres, sdev, n);
}

enum BlockFlags : uint16_t {
  BLOCK_USER_DATA = 1,
};

struct IOBuf::Block {
  std::atomic<int32_t> nshared;
  uint16_t flags;
  Residency res;
  int8_t dev;
  uint32_t size;  // append cursor (bytes filled)
  uint32_t cap;
  char* data;
  void (*user_deleter)(void*);
  uint64_t user_meta;
  Block* next;  // TLS freelist chain

  bool full() const { return size >= cap; }
  uint32_t left_space() const { return cap - size; }
};

namespace {

// Inline payload for HOST blocks: [Block header][payload].
IOBuf::Block* create_block(uint32_t cap, Residency res, int dev) {
  IOBuf::Block* b;
  if (res == RES_HOST) {
    char* mem = (char*)malloc(sizeof(IOBuf::Block) + cap);
    if (mem == nullptr) return nullptr;
    b = (IOBuf::Block*)mem;
    b->data = mem + sizeof(IOBuf::Block);
  } else {
    CHECK(g_allocators[res].alloc != nullptr)
        << "no allocator registered for residency " << (int)res;
    void* payload = g_allocators[res].alloc(cap, dev);
    if (payload == nullptr) return nullptr;
    b = (IOBuf::Block*)malloc(sizeof(IOBuf::Block));
    if (b == nullptr) {
      g_allocators[res].dealloc(payload, cap, dev);
      return nullptr;
    }
    b->data = (char*)payload;
  }
  b->nshared.store(1, std::memory_order_relaxed);
  b->flags = 0;
  b->res = res;
  b->dev = (int8_t)dev;
  b->size = 0;
  b->cap = cap;
  b->user_deleter = nullptr;
  b->user_meta = 0;
  b->next = nullptr;
  g_nblock.fetch_add(1, std::memory_order_relaxed);
  g_blockmem.fetch_add(cap, std::memory_order_relaxed);
  return b;
}

void destroy_block(IOBuf::Block* b);

// ---- TLS block cache (amortizes allocation for append()) ----
struct TlsBlockCache {
  IOBuf::Block* current = nullptr;  // partially filled shared block
  IOBuf::Block* freelist = nullptr; // fully-released default-size blocks
  int nfree = 0;
  ~TlsBlockCache();
};

thread_local TlsBlockCache tls_cache;
const int kMaxTlsFreeBlocks = 32;

void block_dec_ref(IOBuf::Block* b) {
  if (b->nshared.fetch_sub(1, std::memory_order_acq_rel) == 1) {
    // Last reference: recycle default host blocks through TLS freelist.
    if (b->res == RES_HOST && !(b->flags & BLOCK_USER_DATA) &&
        b->cap == IOBuf::kDefaultBlockPayload && tls_cache.nfree < kMaxTlsFreeBlocks) {
      b->size = 0;
      b->nshared.store(1, std::memory_order_relaxed);
      b->next = tls_cache.freelist;
      tls_cache.freelist = b;
      ++tls_cache.nfree;
      return;
    }
    destroy_block(b);
  }
}

void destroy_block(IOBuf::Block* b) {
  g_nblock.fetch_sub(1, std::memory_order_relaxed);
  g_blockmem.fetch_sub(b->cap, std::memory_order_relaxed);
  if (b->flags & BLOCK_USER_DATA) {
    if (b->user_deleter) b->user_deleter(b->data);
    free(b);
  } else if (b->res == RES_HOST) {
    free(b);  // header+payload in one allocation
  } else {
    g_allocators[b->res].dealloc(b->data, b->cap, b->dev);
    free(b);
  }
}

TlsBlockCache::~TlsBlockCache() {
  if (current) {
    block_dec_ref(current);  // may land on freelist of a dead TLS: guard below
    current = nullptr;
  }
  IOBuf::Block* b = freelist;
  freelist = nullptr;
  nfree = kMaxTlsFreeBlocks;  // prevent re-entrant recycling into this dying cache
  while (b) {
    IOBuf::Block* next = b->next;
    destroy_block(b);
    b = next;
  }
}

// Returns the thread's shared append block with at least 1 byte free.
IOBuf::Block* acquire_tls_block() {
  TlsBlockCache& c = tls_cache;
  if (c.current != nullptr) {
    if (!c.current->full()) return c.current;
    block_dec_ref(c.current);
    c.current = nullptr;
  }
  if (c.freelist != nullptr) {
    IOBuf::Block* b = c.freelist;
    c.freelist = b->next;
    --c.nfree;
    b->next = nullptr;
    c.current = b;
    return b;
  }
  c.current = create_block(IOBuf::kDefaultBlockPayload, RES_HOST, -1);
  CHECK(c.current != nullptr) << "out of memory";
  return c.current;
}

}  // namespace

void iobuf_flush_tls_cache() {
  TlsBlockCache& c = tls_cache;
  if (c.current) {
    block_dec_ref(c.current);
    c.current = nullptr;
  }
  IOBuf::Block* b = c.freelist;
  c.freelist = nullptr;
  c.nfree = 0;
  while (b) {
    IOBuf::Block* next = b->next;
    destroy_block(b);
    b = next;
  }
}

size_t IOBuf::block_count() { return g_nblock.load(std::memory_order_relaxed); }
size_t IOBuf::block_memory() { return g_blockmem.load(std::memory_order_relaxed); }

// ---------------- IOBuf ring management ----------------

IOBuf::IOBuf() : refs_(inline_), cap_(kInlineRefs), begin_(0), count_(0), nbytes_(0) {}

IOBuf::IOBuf(const IOBuf& rhs) : IOBuf() { append(rhs); }

IOBuf& IOBuf::operator=(const IOBuf& rhs) {
  if (this != &rhs) {
    clear();
    append(rhs);
  }
  return *this;
}

IOBuf::IOBuf(IOBuf&& rhs) noexcept : IOBuf() { swap(rhs); }

IOBuf& IOBuf::operator=(IOBuf&& rhs) noexcept {
  if (this != &rhs) {
    clear();
    swap(rhs);
  }
  return *this;
}

void IOBuf::clear() {
  for (uint32_t i = 0; i < count_; ++i) block_dec_ref(mutable_ref_at(i).block);
  if (refs_ != inline_) free(refs_);
  refs_ = inline_;
  cap_ = kInlineRefs;
  begin_ = 0;
  count_ = 0;
  nbytes_ = 0;
}

void IOBuf::swap(IOBuf& rhs) {
  // Inline arrays need element-wise swap; normalize both to heap if needed.
  IOBuf* a = this;
  IOBuf* b = &rhs;
  BlockRef tmp_inline[kInlineRefs];
  bool a_inline = (a->refs_ == a->inline_);
  bool b_inline = (b->refs_ == b->inline_);
  if (a_inline) memcpy(tmp_inline, a->inline_, sizeof(tmp_inline));
  if (b_inline) memcpy(a->inline_, b->inline_, sizeof(tmp_inline));
  if (a_inline) memcpy(b->inline_, tmp_inline, sizeof(tmp_inline));
  BlockRef* ar = a->refs_;
  BlockRef* br = b->refs_;
  a->refs_ = b_inline ? a->inline_ : br;
  b->refs_ = a_inline ? b->inline_ : ar;
  std::swap(a->cap_, b->cap_);
  std::swap(a->begin_, b->begin_);
  std::swap(a->count_, b->count_);
  std::swap(a->nbytes_, b->nbytes_);
}

void IOBuf::grow(uint32_t min_cap) {
  uint32_t new_cap = cap_;
  while (new_cap < min_cap) new_cap *= 2;
  BlockRef* nr = (BlockRef*)malloc(sizeof(BlockRef) * new_cap);
  CHECK(nr != nullptr);
  for (uint32_t i = 0; i < count_; ++i) nr[i] = mutable_ref_at(i);
  if (refs_ != inline_) free(refs_);
  refs_ = nr;
  cap_ = new_cap;
  begin_ = 0;
}

void IOBuf::push_ref_back(const BlockRef& r) {
  // Merge with previous ref when contiguous in the same block (common for
  // repeated append() into the shared TLS block).
  if (count_ > 0) {
    BlockRef& last = mutable_ref_at(count_ - 1);
    if (last.block == r.block && last.offset + last.length == r.offset) {
      last.length += r.length;
      nbytes_ += r.length;
      block_dec_ref(r.block);  // we already owned a ref through `last`
      return;
    }
  }
  if (count_ == cap_) grow(cap_ * 2);
  refs_[(begin_ + count_) & (cap_ - 1)] = r;
  ++count_;
  nbytes_ += r.length;
}

void IOBuf::pop_front_ref() {
  CHECK_GT(count_, 0u);
  BlockRef& r = mutable_ref_at(0);
  nbytes_ -= r.length;
  block_dec_ref(r.block);
  begin_ = (begin_ + 1) & (cap_ - 1);
  --count_;
}

void IOBuf::pop_back_ref() {
  CHECK_GT(count_, 0u);
  BlockRef& r = mutable_ref_at(count_ - 1);
  nbytes_ -= r.length;
  block_dec_ref(r.block);
  --count_;
}

// ---------------- append ----------------

void IOBuf::append(const void* data, size_t n) {
  const char* p = (const char*)data;
  while (n > 0) {
    Block* b = acquire_tls_block();
    uint32_t copied = (uint32_t)std::min<size_t>(n, b->left_space());
    memcpy(b->data + b->size, p, copied);
    BlockRef r = {b->size, copied, b};
    b->size += copied;
    b->nshared.fetch_add(1, std::memory_order_relaxed);
    push_ref_back(r);
    p += copied;
    n -= copied;
  }
}

void IOBuf::append(const char* s) { append(s, strlen(s)); }

void IOBuf::append(const IOBuf& other) {
  for (uint32_t i = 0; i < other.count_; ++i) {
    BlockRef r = other.ref_at(i);
    r.block->nshared.fetch_add(1, std::memory_order_relaxed);
    push_ref_back(r);
  }
}

void IOBuf::append(IOBuf&& other) {
  if (empty()) {
    swap(other);
    return;
  }
  for (uint32_t i = 0; i < other.count_; ++i) {
    push_ref_back(other.mutable_ref_at(i));  // steal refcount
  }
  // Refcounts moved; reset rhs without decrementing.
  if (other.refs_ != other.inline_) free(other.refs_);
  other.refs_ = other.inline_;
  other.cap_ = kInlineRefs;
  other.begin_ = 0;
  other.count_ = 0;
  other.nbytes_ = 0;
}

int IOBuf::append_user_data(void* data, size_t n, void (*deleter)(void*), uint64_t meta) {
  if (n == 0 || n > UINT32_MAX) return -1;
  Block* b = (Block*)malloc(sizeof(Block));
  if (b == nullptr) return -1;
  b->nshared.store(1, std::memory_order_relaxed);
  b->flags = BLOCK_USER_DATA;
  b->res = RES_HOST;
  b->dev = -1;
  b->size = (uint32_t)n;
  b->cap = (uint32_t)n;
  b->data = (char*)data;
  b->user_deleter = deleter;
  b->user_meta = meta;
  b->next = nullptr;
  g_nblock.fetch_add(1, std::memory_order_relaxed);
  g_blockmem.fetch_add(n, std::memory_order_relaxed);
  push_ref_back(BlockRef{0, (uint32_t)n, b});
  return 0;
}

int IOBuf::append_with_residency(const void* host_data, size_t n, Residency res, int dev,
                                 uint32_t block_payload) {
  if (block_payload == 0) {
    block_payload = (res == RES_HOST) ? kDefaultBlockPayload
                                      : (uint32_t)std::min<size_t>(n, 1u << 21);  // ≤2 MiB
    if (block_payload == 0) block_payload = kDefaultBlockPayload;
  }
  const char* p = (const char*)host_data;
  while (n > 0) {
    uint32_t cap = (uint32_t)std::min<size_t>(n, block_payload);
    Block* b = create_block(cap, res, dev);
    if (b == nullptr) return -1;
    // HBM: prefer the async upload leg (no blocking hipMemcpy per block —
    // the response-side gather is stream-ordered after it).
    if (res != RES_HBM || g_upload_async == nullptr ||
        g_upload_async(b->data, p, cap, dev) != 0) {
      move_bytes(b->data, res, dev, p, RES_HOST, -1, cap);
    }
    b->size = cap;
    push_ref_back(BlockRef{0, cap, b});
    p += cap;
    n -= cap;
  }
  return 0;
}

int IOBuf::append_writable_block(size_t n, Residency res, int dev, void** out_ptr) {
  if (n == 0 || n > (1u << 31)) return -1;
  Block* b = create_block((uint32_t)n, res, dev);
  if (b == nullptr) return -1;
  b->size = (uint32_t)n;
  push_ref_back(BlockRef{0, (uint32_t)n, b});
  if (out_ptr != nullptr) *out_ptr = b->data;
  return 0;
}

int IOBuf::append_device_block(size_t n, int dev, void** out_ptr) {
  return append_writable_block(n, RES_HBM, dev, out_ptr);
}

// ---------------- cut / pop ----------------

size_t IOBuf::cutn(IOBuf* out, size_t n) {
  size_t moved = 0;
  while (n > 0 && count_ > 0) {
    BlockRef& r = mutable_ref_at(0);
    if (r.length <= n) {
      out->push_ref_back(r);  // transfer refcount
      n -= r.length;
      moved += r.length;
      nbytes_ -= r.length;
      begin_ = (begin_ + 1) & (cap_ - 1);
      --count_;
    } else {
      BlockRef part = {r.offset, (uint32_t)n, r.block};
      r.block->nshared.fetch_add(1, std::memory_order_relaxed);
      out->push_ref_back(part);
      r.offset += (uint32_t)n;
      r.length -= (uint32_t)n;
      nbytes_ -= n;
      moved += n;
      n = 0;
    }
  }
  return moved;
}

size_t IOBuf::cutn(void* out, size_t n) {
  size_t copied = copy_to(out, n, 0);
  pop_front(copied);
  return copied;
}

size_t IOBuf::cutn(std::string* out, size_t n) {
  n = std::min(n, nbytes_);
  size_t old = out->size();
  out->resize(old + n);
  return cutn(&(*out)[old], n);
}

int IOBuf::cut1(char* c) {
  if (empty()) return -1;
  cutn(c, 1);
  return 0;
}

size_t IOBuf::pop_front(size_t n) {
  size_t popped = 0;
  while (n > 0 && count_ > 0) {
    BlockRef& r = mutable_ref_at(0);
    if (r.length <= n) {
      n -= r.length;
      popped += r.length;
      pop_front_ref();
    } else {
      r.offset += (uint32_t)n;
      r.length -= (uint32_t)n;
      nbytes_ -= n;
      popped += n;
      n = 0;
    }
  }
  return popped;
}

size_t IOBuf::pop_back(size_t n) {
  size_t popped = 0;
  while (n > 0 && count_ > 0) {
    BlockRef& r = mutable_ref_at(count_ - 1);
    if (r.length <= n) {
      n -= r.length;
      popped += r.length;
      pop_back_ref();
    } else {
      r.length -= (uint32_t)n;
      nbytes_ -= n;
      popped += n;
      n = 0;
    }
  }
  return popped;
}

// ---------------- copy_to / fetch ----------------

size_t IOBuf::copy_to(void* buf, size_t n, size_t pos) const {
  if (pos >= nbytes_) return 0;
  n = std::min(n, nbytes_ - pos);
  char* out = (char*)buf;
  size_t remain = n;
  for (uint32_t i = 0; i < count_ && remain > 0; ++i) {
    const BlockRef& r = ref_at(i);
    if (pos >= r.length) {
      pos -= r.length;
      continue;
    }
    size_t take = std::min<size_t>(remain, r.length - pos);
    move_bytes(out, RES_HOST, -1, r.block->data + r.offset + pos, r.block->res, r.block->dev,
               take);
    out += take;
    remain -= take;
    pos = 0;
  }
  return n - remain;
}

size_t IOBuf::copy_to(std::string* s, size_t n, size_t pos) const {
  if (pos >= nbytes_) {
    s->clear();
    return 0;
  }
  n = std::min(n, nbytes_ - pos);
  s->resize(n);
  return copy_to(&(*s)[0], n, pos);
}

std::string IOBuf::to_string() const {
  std::string s;
  copy_to(&s);
  return s;
}

const void* IOBuf::fetch(void* aux, size_t n) const {
  if (n > nbytes_) return nullptr;
  if (count_ > 0) {
    const BlockRef& r = ref_at(0);
    if (r.block->res != RES_HBM && r.length >= n) return r.block->data + r.offset;
  }
  if (copy_to(aux, n, 0) != n) return nullptr;
  return aux;
}

bool IOBuf::cpu_addressable() const {
  for (uint32_t i = 0; i < count_; ++i)
    if (ref_at(i).block->res == RES_HBM) return false;
  return true;
}

size_t IOBuf::hbm_bytes() const {
  size_t n = 0;
  for (uint32_t i = 0; i < count_; ++i) {
    const BlockRef& r = ref_at(i);
    if (r.block->res == RES_HBM) n += r.length;
  }
  return n;
}

IOBuf::Span IOBuf::span_at(size_t i) const {
  const BlockRef& r = ref_at(i);
  return Span{r.block->data + r.offset, r.length, r.block->res, r.block->dev};
}

bool IOBuf::equals(const IOBuf& other) const {
  if (nbytes_ != other.nbytes_) return false;
  return to_string() == other.to_string();  // simple & correct; hot paths don't use it
}

bool IOBuf::equals(const std::string& s) const {
  if (nbytes_ != s.size()) return false;
  return to_string() == s;
}

// ---------------- fd I/O ----------------

static const size_t kMaxIov = 64;

// Staging-ring hook installed by the HIP loader: gathers scattered device
// spans into one contiguous host buffer (device gather + single D2H).
static GatherToHostFn g_gather_to_host = nullptr;
void set_gather_to_host(GatherToHostFn fn) { g_gather_to_host = fn; }

namespace {
// Per-thread pinned bounce buffer for the HBM->wire staging path. Pinned
// memory doubles D2H bandwidth vs pageable and is reused across calls.
struct TlsBounce {
  char* buf = nullptr;
  size_t cap = 0;
  bool pinned = false;
  char* get(size_t need) {
    if (cap >= need) return buf;
    release();
    cap = need < (1u << 20) ? (1u << 20) : need;
    if (has_block_allocator(RES_PINNED)) {
      buf = (char*)g_allocators[RES_PINNED].alloc((uint32_t)cap, 0);
      pinned = buf != nullptr;
    }
    if (buf == nullptr) {
      buf = (char*)malloc(cap);
      pinned = false;
    }
    return buf;
  }
  void release() {
    if (buf == nullptr) return;
    if (pinned) {
      g_allocators[RES_PINNED].dealloc(buf, (uint32_t)cap, 0);
    } else {
      free(buf);
    }
    buf = nullptr;
    cap = 0;
  }
  ~TlsBounce() { release(); }
};
thread_local TlsBounce tls_bounce;
}  // namespace

int IOBuf::cut_until(IOBuf* out, const std::string& delim) {
  if (delim.empty()) return -1;
  // KMP across block boundaries (self-overlapping delimiters like
  // "\r\n\r\n" need the real failure function).
  std::vector<size_t> lps(delim.size(), 0);
  for (size_t i = 1, len = 0; i < delim.size();) {
    if (delim[i] == delim[len]) {
      lps[i++] = ++len;
    } else if (len != 0) {
      len = lps[len - 1];
    } else {
      lps[i++] = 0;
    }
  }
  size_t match = 0;
  size_t scanned = 0;
  for (uint32_t i = 0; i < count_; ++i) {
    const BlockRef& r = ref_at(i);
    if (r.block->res != RES_HOST) return -1;  // device bytes are not scannable
    const char* p = r.block->data + r.offset;
    for (uint32_t k = 0; k < r.length; ++k) {
      ++scanned;
      while (match > 0 && p[k] != delim[match]) match = lps[match - 1];
      if (p[k] == delim[match] && ++match == delim.size()) {
        cutn(out, scanned - delim.size());
        pop_front(delim.size());
        return 0;
      }
    }
  }
  return -1;
}

bool IOBuf::has_residency(Residency res) const {
  for (uint32_t i = 0; i < count_; ++i)
    if (ref_at(i).block->res == res) return true;
  return false;
}

ssize_t IOBuf::cut_into_file_descriptor(int fd, size_t size_hint) {
  if (empty()) return 0;
  struct iovec iov[kMaxIov];
  size_t niov = 0;
  size_t queued = 0;
  // Collect spans; HBM spans are staged in ONE device-gather + D2H below.
  const void* hbm_srcs[kMaxIov];
  size_t hbm_lens[kMaxIov];
  size_t hbm_iov_idx[kMaxIov];
  int nhbm = 0;
  size_t hbm_total = 0;
  int hbm_dev = 0;
  for (uint32_t i = 0; i < count_ && niov < kMaxIov && queued < size_hint; ++i) {
    const BlockRef& r = ref_at(i);
    size_t take = std::min<size_t>(r.length, size_hint - queued);
    if (r.block->res != RES_HBM) {
      iov[niov].iov_base = r.block->data + r.offset;
      iov[niov].iov_len = take;
    } else {
      hbm_srcs[nhbm] = r.block->data + r.offset;
      hbm_lens[nhbm] = take;
      hbm_iov_idx[nhbm] = niov;
      hbm_dev = r.block->dev;
      ++nhbm;
      hbm_total += take;
      iov[niov].iov_base = nullptr;  // patched after staging
      iov[niov].iov_len = take;
    }
    ++niov;
    queued += take;
  }
  if (nhbm > 0) {
    char* bounce = tls_bounce.get(hbm_total);
    if (bounce == nullptr) {
      errno = ENOMEM;
      return -1;
    }
    if (g_gather_to_host != nullptr) {
      if (g_gather_to_host(bounce, hbm_srcs, hbm_lens, nhbm, hbm_dev) != 0) {
        errno = EIO;
        return -1;
      }
    } else {
      size_t off = 0;
      for (int k = 0; k < nhbm; ++k) {
        move_bytes(bounce + off, RES_HOST, -1, hbm_srcs[k], RES_HBM, hbm_dev, hbm_lens[k]);
        off += hbm_lens[k];
      }
    }
    size_t off = 0;
    for (int k = 0; k < nhbm; ++k) {
      iov[hbm_iov_idx[k]].iov_base = bounce + off;
      off += hbm_lens[k];
    }
  }
  ssize_t nw = ::writev(fd, iov, (int)niov);
  if (nw > 0) pop_front((size_t)nw);
  return nw;
}

ssize_t IOBuf::append_from_file_descriptor(int fd, size_t max_read) {
  // Read into the TLS shared block's tail plus fresh blocks.
  struct iovec iov[8];
  Block* blocks[8];
  size_t niov = 0;
  size_t space = 0;
  Block* tls_b = acquire_tls_block();
  iov[0].iov_base = tls_b->data + tls_b->size;
  iov[0].iov_len = std::min<size_t>(tls_b->left_space(), max_read);
  blocks[0] = tls_b;
  space = iov[0].iov_len;
  niov = 1;
  std::vector<Block*> extra;
  while (space < max_read && niov < 8) {
    Block* nb = create_block(kDefaultBlockPayload, RES_HOST, -1);
    CHECK(nb != nullptr);
    extra.push_back(nb);
    iov[niov].iov_base = nb->data;
    iov[niov].iov_len = std::min<size_t>(nb->cap, max_read - space);
    blocks[niov] = nb;
    space += iov[niov].iov_len;
    ++niov;
  }
  ssize_t nr = ::readv(fd, iov, (int)niov);
  if (nr <= 0) {
    int saved_errno = errno;
    for (Block* b : extra) destroy_block(b);
    errno = saved_errno;
    return nr;
  }
  size_t remain = (size_t)nr;
  for (size_t i = 0; i < niov && remain > 0; ++i) {
    size_t got = std::min<size_t>(remain, iov[i].iov_len);
    Block* b = blocks[i];
    if (i == 0) {
      // TLS block: data landed at its append cursor.
      BlockRef r = {b->size, (uint32_t)got, b};
      b->size += (uint32_t)got;
      b->nshared.fetch_add(1, std::memory_order_relaxed);
      push_ref_back(r);
    } else {
      b->size = (uint32_t)got;
      push_ref_back(BlockRef{0, (uint32_t)got, b});  // adopt the create refcount
      extra.erase(std::find(extra.begin(), extra.end(), b));
    }
    remain -= got;
  }
  for (Block* b : extra) destroy_block(b);
  return nr;
}

}  // namespace bam
