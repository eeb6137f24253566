#!/usr/bin/env python3
"""Mixed-load soak for MI355X boxes: concurrent threads drive every major
path — HBM 64 B / 16 KB echo, host sync + pipelined-async echo, streaming
frames, GPU span-gather and crc32c kernels — for --seconds, then print one
JSON line with totals and error counts (all zeros expected).

Run under gpurun; redirect stdout into gpurun_out/ and commit the summary
to profiles/.
"""
import argparse
import json
import os
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import brpc_amd as b  # noqa: E402

g = b.core.gpu
r = b.core.rpc


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=int, default=300)
    ap.add_argument("--out", default="gpurun_out/soak.json")
    args = ap.parse_args()

    use_gpu = g.initialize() > 0
    port = r.start_echo_server(0)
    addr = "127.0.0.1:%d" % port
    deadline = time.monotonic() + args.seconds
    lock = threading.Lock()
    stats = {}

    def bump(tag, calls, errs, extra=0.0):
        with lock:
            s = stats.setdefault(tag, {"calls": 0, "errors": 0, "extra": 0.0})
            s["calls"] += calls
            s["errors"] += errs
            s["extra"] += extra

    def echo_loop(tag, payload, conc, method, pooled, calls):
        while time.monotonic() < deadline:
            res = r.echo_bench(addr, payload, conc, calls, 30000, method,
                               False, pooled, 1)
            bump(tag, res["total"], res["errors"])

    def async_loop(tag, payload, pipeline, calls):
        while time.monotonic() < deadline:
            res = r.async_echo_bench(addr, payload, pipeline, calls, 30000,
                                     "EchoService.Echo", True)
            bump(tag, res["total"], res["errors"])

    def kernel_loop():
        while time.monotonic() < deadline:
            gb = g.gather_gbps(64 << 20, 65536, 4)
            bump("gather_64MB", 4, 0 if gb > 0 else 1, gb)
            gb = g.crc_gbps(64 << 20, 4)
            bump("crc_64MB", 4, 0 if gb > 0 else 1, gb)
            ok = g.crc_matches(1 << 20, 0)
            bump("crc_check", 1, 0 if ok else 1)

    def stream_loop():
        s = b.core.stream
        sp = s.start_server()
        while time.monotonic() < deadline:
            mbps = s.throughput(sp, 100, 1 << 20)
            bump("stream_1MB", 100, 0 if mbps > 0 else 1, mbps)

    threads = []
    if use_gpu:
        threads += [
            threading.Thread(target=echo_loop,
                             args=("hbm64", 64, 32, "EchoService.EchoHbm", False, 20000)),
            threading.Thread(target=echo_loop,
                             args=("hbm16k", 16384, 16, "EchoService.EchoHbm", False, 4000)),
            threading.Thread(target=kernel_loop),
            threading.Thread(target=stream_loop),
        ]
    threads += [
        threading.Thread(target=echo_loop,
                         args=("host64", 64, 32, "EchoService.Echo", True, 20000)),
        threading.Thread(target=async_loop, args=("host64_async", 64, 128, 40000)),
    ]
    t0 = time.monotonic()
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    elapsed = time.monotonic() - t0

    total_calls = sum(s["calls"] for s in stats.values())
    total_errors = sum(s["errors"] for s in stats.values())
    summary = {"seconds": round(elapsed, 1), "gpu": use_gpu,
               "total_calls": total_calls, "total_errors": total_errors,
               "per_path": stats}
    print(json.dumps(summary), flush=True)
    os.makedirs(os.path.dirname(args.out), exist_ok=True)
    with open(args.out, "w") as f:
        json.dump(summary, f, indent=1)


if __name__ == "__main__":
    main()
