// brpc_amd: CommGroup — the in-framework multi-GPU communication group.
// N processes (one per GPU) form a group; the data plane is RCCL over xGMI
// (hip/comm.hip: broadcast / all-gather / p2p send-recv on HBM buffers,
// fibers PARK on completion tickets), the control plane is a tiny TCP full
// mesh used for the RCCL uniqueId rendezvous, host-side metadata and
// barriers. Backend "tcp" runs the same collective API on host buffers so
// the group semantics are CPU-testable (and serve as the no-GPU fallback).
//
// Parity map: reference ParallelChannel fan-out (parallel_channel.cpp:90)
// → Broadcast+AllGather; streaming_rpc bulk frames (stream.cpp:254) →
// Send/Recv; UBRing peer rings (ubshm/ub_ring.h) → the xGMI p2p path.
#pragma once

#include <stddef.h>
#include <stdint.h>

#include <string>

namespace bam {

class CommGroup {
 public:
  struct Options {
    int nranks = 1;
    int rank = 0;
    // "rccl": data-plane buffers are device (HBM) pointers, moved over
    // xGMI by RCCL. "tcp": host pointers, moved over the mesh (tests /
    // no-GPU fallback).
    std::string backend = "tcp";
    std::string host = "127.0.0.1";  // all ranks on one node
    int base_port = 0;               // rank i listens on base_port + i
    int dev = 0;                     // HIP device of this rank ("rccl")
    int connect_timeout_ms = 30000;
  };

  // Blocks until every rank joined the mesh (and, for "rccl", the
  // communicator initialized). nullptr + *err on failure.
  static CommGroup* Create(const Options& opt, std::string* err);
  ~CommGroup();

  int rank() const { return opt_.rank; }
  int nranks() const { return opt_.nranks; }
  const std::string& backend() const { return opt_.backend; }

  // ---- data plane (HBM pointers for "rccl", host pointers for "tcp") ----
  // All ranks must call collectives with matching sizes; rounds on one
  // group are serialized. Calling fibers park while the GPU works.
  int Broadcast(void* buf, size_t n, int root);
  int AllGather(const void* send, void* recv, size_t per_rank);
  int Send(const void* buf, size_t n, int peer);
  int Recv(void* buf, size_t n, int peer);
  // Full-duplex exchange (both xGMI directions concurrently).
  int SendRecv(const void* sbuf, size_t sn, int speer, void* rbuf, size_t rn, int rpeer);

  // ---- host control plane (always the TCP mesh) ----
  int HostBroadcast(std::string* blob, int root);
  int HostSend(int peer, const void* data, size_t n);
  int HostRecv(int peer, std::string* out);
  int Barrier();

  const std::string& last_error() const { return err_; }

 private:
  CommGroup() {}
  struct Mesh;
  int init(const Options& opt, std::string* err);

  Options opt_;
  Mesh* mesh_ = nullptr;
  void* rccl_ = nullptr;  // BamComm* inside libbrpc_hip.so
  std::string err_;
};

}  // namespace bam
