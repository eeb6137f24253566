// brpc_amd: in-framework RCCL collectives over xGMI (gfx950).
// The MI355X-native replacement for reference ParallelChannel TCP fan-out
// (brpc/parallel_channel.cpp:90-186) and streaming_rpc bulk transfer
// (brpc/stream.cpp:254-744): fan-out = ncclBroadcast of the HBM-resident
// request + ncclAllGather of responses; streaming = ncclSend/ncclRecv p2p.
// Each communicator owns a dedicated non-blocking stream; completion is a
// pinned ticket flag published by a marker kernel enqueued after the
// collective, waited via bamhip::wait_ticket → the calling FIBER parks and
// the worker keeps running other RPCs (never block a worker on RCCL —
// SURVEY §7 hard-part note).
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>
#include <string.h>

#include <mutex>

#include "gpu_api.h"
#include "internal.h"

namespace {

char g_comm_err[256] = {0};

void set_comm_err(const char* what, ncclResult_t r) {
  snprintf(g_comm_err, sizeof(g_comm_err), "%s: %s", what, ncclGetErrorString(r));
}

__global__ void ticket_kernel(volatile unsigned long long* flag, unsigned long long t) {
  if (threadIdx.x == 0) {
    __threadfence_system();
    *flag = t;
    __threadfence_system();
  }
}

struct BamComm {
  ncclComm_t comm = nullptr;
  int dev = 0;
  int rank = 0;
  int nranks = 0;
  hipStream_t stream = nullptr;
  volatile unsigned long long* flag = nullptr;  // pinned ticket flag
  unsigned long long ticket = 0;
  std::mutex mu;  // serializes enqueue order so tickets match stream order
};

// Enqueues the ticket marker for work already on c->stream and parks until
// it completes. Call with c->mu held for the enqueue; the wait runs inside
// too (collective rounds are serialized per communicator by design — a
// second concurrent collective on one comm would deadlock RCCL anyway).
int finish_op(BamComm* c, ncclResult_t rc, const char* what) {
  if (rc != ncclSuccess) {
    set_comm_err(what, rc);
    return -1;
  }
  c->ticket += 1;
  const unsigned long long t = c->ticket;
  hipLaunchKernelGGL(ticket_kernel, dim3(1), dim3(64), 0, c->stream, c->flag, t);
  if (hipGetLastError() != hipSuccess) {
    snprintf(g_comm_err, sizeof(g_comm_err), "%s: ticket kernel launch failed", what);
    return -1;
  }
  if (!bamhip::wait_ticket(c->flag, t, c->dev, bamhip::kWakeComm, c->stream)) {
    snprintf(g_comm_err, sizeof(g_comm_err), "%s: stream wedged", what);
    return -1;
  }
  return 0;
}

struct ScopedDev {
  int old = -1;
  explicit ScopedDev(int dev) {
    (void)hipGetDevice(&old);
    if (dev != old) (void)hipSetDevice(dev);
    else old = -1;
  }
  ~ScopedDev() {
    if (old >= 0) (void)hipSetDevice(old);
  }
};

}  // namespace

extern "C" const char* bam_comm_last_error(void) { return g_comm_err; }

extern "C" int bam_comm_uid(char out[128]) {
  static_assert(NCCL_UNIQUE_ID_BYTES == 128, "uid size drifted");
  ncclUniqueId id;
  ncclResult_t r = ncclGetUniqueId(&id);
  if (r != ncclSuccess) {
    set_comm_err("ncclGetUniqueId", r);
    return -1;
  }
  memcpy(out, id.internal, 128);
  return 0;
}

extern "C" void* bam_comm_create(int nranks, int rank, const char uid[128], int dev) {
  ScopedDev sd(dev);
  BamComm* c = new BamComm;
  c->dev = dev;
  c->rank = rank;
  c->nranks = nranks;
  if (hipStreamCreateWithFlags(&c->stream, hipStreamNonBlocking) != hipSuccess) {
    snprintf(g_comm_err, sizeof(g_comm_err), "hipStreamCreate failed");
    delete c;
    return nullptr;
  }
  void* f = nullptr;
  if (hipHostMalloc(&f, 64, hipHostMallocDefault) != hipSuccess) {
    snprintf(g_comm_err, sizeof(g_comm_err), "hipHostMalloc flag failed");
    delete c;
    return nullptr;
  }
  c->flag = (volatile unsigned long long*)f;
  *c->flag = 0;
  ncclUniqueId id;
  memcpy(id.internal, uid, 128);
  ncclResult_t r = ncclCommInitRank(&c->comm, nranks, id, rank);
  if (r != ncclSuccess) {
    set_comm_err("ncclCommInitRank", r);
    delete c;
    return nullptr;
  }
  // One communicator per device is the expected shape (one process per
  // GPU); the wake slot maps this comm's stream.
  bamhip::register_wake_stream(dev, bamhip::kWakeComm, c->stream);
  return c;
}

extern "C" void bam_comm_destroy(void* h) {
  BamComm* c = (BamComm*)h;
  if (c == nullptr) return;
  if (c->comm != nullptr) ncclCommDestroy(c->comm);
  delete c;
}

extern "C" int bam_comm_rank(void* h) { return ((BamComm*)h)->rank; }
extern "C" int bam_comm_nranks(void* h) { return ((BamComm*)h)->nranks; }

extern "C" int bam_comm_broadcast(void* h, void* buf_dev, size_t n, int root) {
  BamComm* c = (BamComm*)h;
  ScopedDev sd(c->dev);
  std::lock_guard<std::mutex> lk(c->mu);
  ncclResult_t r = ncclBroadcast(buf_dev, buf_dev, n, ncclUint8, root, c->comm, c->stream);
  return finish_op(c, r, "ncclBroadcast");
}

extern "C" int bam_comm_allgather(void* h, const void* send_dev, void* recv_dev,
                                  size_t per_rank) {
  BamComm* c = (BamComm*)h;
  ScopedDev sd(c->dev);
  std::lock_guard<std::mutex> lk(c->mu);
  ncclResult_t r = ncclAllGather(send_dev, recv_dev, per_rank, ncclUint8, c->comm, c->stream);
  return finish_op(c, r, "ncclAllGather");
}

extern "C" int bam_comm_send(void* h, const void* buf_dev, size_t n, int peer) {
  BamComm* c = (BamComm*)h;
  ScopedDev sd(c->dev);
  std::lock_guard<std::mutex> lk(c->mu);
  ncclResult_t r = ncclSend(buf_dev, n, ncclUint8, peer, c->comm, c->stream);
  return finish_op(c, r, "ncclSend");
}

extern "C" int bam_comm_recv(void* h, void* buf_dev, size_t n, int peer) {
  BamComm* c = (BamComm*)h;
  ScopedDev sd(c->dev);
  std::lock_guard<std::mutex> lk(c->mu);
  ncclResult_t r = ncclRecv(buf_dev, n, ncclUint8, peer, c->comm, c->stream);
  return finish_op(c, r, "ncclRecv");
}

// Bidirectional exchange in one grouped launch (full-duplex xGMI: both
// directions of a link run concurrently).
extern "C" int bam_comm_sendrecv(void* h, const void* sbuf, size_t sn, int speer,
                                 void* rbuf, size_t rn, int rpeer) {
  BamComm* c = (BamComm*)h;
  ScopedDev sd(c->dev);
  std::lock_guard<std::mutex> lk(c->mu);
  ncclResult_t r = ncclGroupStart();
  if (r == ncclSuccess && sn > 0) r = ncclSend(sbuf, sn, ncclUint8, speer, c->comm, c->stream);
  if (r == ncclSuccess && rn > 0) r = ncclRecv(rbuf, rn, ncclUint8, rpeer, c->comm, c->stream);
  if (r == ncclSuccess) r = ncclGroupEnd();
  return finish_op(c, r, "ncclSendRecv");
}
