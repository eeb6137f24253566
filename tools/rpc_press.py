#!/usr/bin/env python3
"""rpc_press: load generator (parity: reference tools/rpc_press).

Modes:
  --qps N     paced load at N requests/s (python-paced)
  --max       max-throughput mode (C++ fiber loop, like the reference's
              unlimited mode)

Example:
  python tools/rpc_press.py --addr 127.0.0.1:8000 \
      --method EchoService.Echo --payload-size 64 --max --total 100000
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import brpc_amd as b  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--addr", required=True)
    ap.add_argument("--method", default="EchoService.Echo")
    ap.add_argument("--payload-size", type=int, default=64)
    ap.add_argument("--payload-file")
    ap.add_argument("--qps", type=int, default=0)
    ap.add_argument("--max", action="store_true")
    ap.add_argument("--total", type=int, default=10000)
    ap.add_argument("--concurrency", type=int, default=16)
    ap.add_argument("--timeout-ms", type=int, default=1000)
    ap.add_argument("--duration", type=float, default=10.0, help="seconds (qps mode)")
    args = ap.parse_args()

    payload = (open(args.payload_file, "rb").read() if args.payload_file
               else os.urandom(args.payload_size))

    if args.max or args.qps <= 0:
        res = b.core.rpc.echo_bench(args.addr, len(payload), args.concurrency,
                                    args.total, args.timeout_ms, args.method, False)
        print(f"sent={res['total']} errors={res['errors']} qps={res['qps']:.0f} "
              f"avg={res['avg_us']}us p50={res['p50_us']}us p90={res['p90_us']}us "
              f"p99={res['p99_us']}us p999={res['p999_us']}us max={res['max_us']}us")
        return 0 if res["errors"] == 0 else 1

    ch = b.Channel(args.addr, timeout_ms=args.timeout_ms)
    sent = errors = 0
    lats = []
    interval = 1.0 / args.qps
    deadline = time.time() + args.duration
    next_t = time.time()
    while time.time() < deadline:
        now = time.time()
        if now < next_t:
            time.sleep(next_t - now)
        next_t += interval
        try:
            _, _, lat = ch.call(args.method, payload)
            lats.append(lat)
        except b.RpcError:
            errors += 1
        sent += 1
    lats.sort()
    pct = lambda p: lats[int(p * (len(lats) - 1))] if lats else 0
    print(f"sent={sent} errors={errors} qps~{args.qps} p50={pct(.5)}us "
          f"p90={pct(.9)}us p99={pct(.99)}us")
    return 0


if __name__ == "__main__":
    sys.exit(main())
