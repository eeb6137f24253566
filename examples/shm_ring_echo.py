#!/usr/bin/env python3
"""Shared-memory ring RPC (≙ reference UBRing, docs/en/ubring.md):
microsecond-scale same-host calls without sockets."""
import sys

sys.path.insert(0, ".")
import brpc_amd as b

assert b.core.rpc.start_shm_server("demo") == 0
rc, resp, err = b.core.rpc.shm_call("demo", "EchoService.Echo", b"via shm")
print("reply:", resp)
res = b.core.rpc.shm_echo_bench("demo", 64, 8, 50000)
print(f"qps={res['qps']:.0f} p99={res['p99_us']}us")
