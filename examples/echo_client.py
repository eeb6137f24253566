#!/usr/bin/env python3
"""Echo client (≙ reference example/echo_c++/client.cpp)."""
import sys
import time

sys.path.insert(0, ".")
import brpc_amd as b

addr = sys.argv[1] if len(sys.argv) > 1 else "127.0.0.1:8000"
ch = b.Channel(addr, timeout_ms=1000)
n = 0
t0 = time.time()
while time.time() - t0 < 3:
    resp, att, lat = ch.call("EchoService.Echo", b"hello world")
    assert resp == b"hello world"
    n += 1
print(f"{n} echos in 3s ({n/3:.0f} qps sequential)")
