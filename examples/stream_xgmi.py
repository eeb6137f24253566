#!/usr/bin/env python3
"""Streaming RPC with the xGMI data plane (BASELINE config 3 shape).

Two processes: rank 0 receives, rank 1 streams 1 MB frames whose payload
moves over the CommGroup data plane (RCCL p2p over xGMI on GPUs, TCP
otherwise) while the stream's socket carries descriptors + credit.

  python examples/stream_xgmi.py 0 36000 &
  python examples/stream_xgmi.py 1 36000
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import brpc_amd as b

rank, base_port = int(sys.argv[1]), int(sys.argv[2])
backend = "rccl" if b.core.gpu.initialize() > 0 else "tcp"
c = b.core.comm
h = c.create(2, rank, backend, "127.0.0.1", base_port)

if rank == 0:
    sport = c.stream_comm_serve(h, 1)
    c.send(h, 1, ("%06d" % sport).encode())
    c.host_broadcast(h, b"", 1)
    print("receiver done")
else:
    sport = int(c.recv(h, 0, 6).decode())
    gbps = c.stream_comm_send(h, "127.0.0.1:%d" % sport, 0, 256, 1 << 20)
    print("streamed 256 x 1MB frames over %s: %.2f GB/s" % (backend, gbps))
    c.host_broadcast(h, b"done", 1)
