#!/usr/bin/env python3
"""Multi-threaded echo press (≙ example/multi_threaded_echo_c++): the C++
fiber bench loop at max throughput over pooled connections."""
import sys

sys.path.insert(0, ".")
import brpc_amd as b

srv = b.Server()
srv.add_method("EchoService", "Echo", lambda req, att: req)
port = srv.start(0)
res = b.core.rpc.echo_bench(f"127.0.0.1:{port}", 64, 32, 50000, 10000,
                            "EchoService.Echo", False, True)
print(f"qps={res['qps']:.0f} p50={res['p50_us']}us p99={res['p99_us']}us")
