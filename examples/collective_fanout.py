#!/usr/bin/env python3
"""ParallelChannel-over-RCCL fan-out example (BASELINE config 4 shape).

Run N processes (rank 0 is the caller, the rest run collective servers):

  for r in 0 1 2 3; do python examples/collective_fanout.py $r 4 35000 & done

On GPUs (one process per device) the payload broadcasts over xGMI via
RCCL and responses return in one all-gather; with backend "tcp" the same
machinery runs on host buffers (works anywhere).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import brpc_amd as b

rank, nranks, base_port = int(sys.argv[1]), int(sys.argv[2]), int(sys.argv[3])
backend = "rccl" if b.core.gpu.initialize() > 0 else "tcp"
c = b.core.comm
h = c.create(nranks, rank, backend, "127.0.0.1", base_port)

if rank == 0:
    addrs = [""]
    for r in range(1, nranks):
        addrs.append("127.0.0.1:%d" % int(c.recv(h, r, 6).decode()))
    payload = bytes(range(256)) * 64  # 16 KB
    comp = b.core.snappy.compress(payload)
    res = c.fanout_call(h, addrs, "snappy_echo", comp,
                        len(payload) + len(payload) // 3 + 256, 100, True)
    print("fanout (%s): %.0f rounds/s, p99 %.0f us, data_ok=%s"
          % (backend, res["qps"], res["p99_us"], res["data_ok"]))
    c.host_broadcast(h, b"done", 0)
else:
    port = c.fanout_serve(h, 0)
    c.send(h, 0, ("%06d" % port).encode())
    c.host_broadcast(h, b"", 0)  # parked until the caller finishes
