#!/usr/bin/env python3
"""GPU kernel perf probe: run on the MI355X box, summary goes to stdout
(redirect into gpurun_out/, then commit the summary under profiles/)."""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import brpc_amd as b

g = b.core.gpu
r = b.core.rpc

assert g.initialize() > 0, g.load_error()
out = {}
for sz in [1 << 20, 16 << 20, 256 << 20]:
    out[f"crc32c_gbps_{sz>>20}MB"] = round(g.crc_gbps(sz, 10), 1)
for total, blk in [(16 << 20, 65536), (256 << 20, 2 << 20)]:
    out[f"gather_gbps_{total>>20}MB_blk{blk>>10}K"] = round(g.gather_gbps(total, blk, 10), 1)
for sz in [1 << 20, 64 << 20]:
    out[f"d2h_gbps_{sz>>20}MB"] = round(g.d2h_gbps(sz, 10), 1)

port = r.start_echo_server(0)
addr = f"127.0.0.1:{port}"
for payload in (64, 16384):
    res = r.echo_bench(addr, payload, 32, 20000, 30000, "EchoService.Echo", False)
    out[f"echo_host_{payload}B"] = {"qps": round(res["qps"]), "p99_us": res["p99_us"]}
    res = r.echo_bench(addr, payload, 32, 10000, 30000, "EchoService.EchoHbm", True)
    out[f"echo_hbm_{payload}B"] = {"qps": round(res["qps"]), "p99_us": res["p99_us"]}
s = b.core.stream
sp = s.start_server()
out["stream_1MBframes_MBps"] = round(s.throughput(sp, 300, 1 << 20))
print(json.dumps(out, indent=1))
