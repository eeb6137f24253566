#!/usr/bin/env python3
"""benchmark_http: ab-style HTTP load generator (parity: reference
example/http_c++/benchmark_http)."""
import argparse
import threading
import time
import urllib.request


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("url")
    ap.add_argument("--threads", type=int, default=8)
    ap.add_argument("--duration", type=float, default=5.0)
    args = ap.parse_args()
    stop = time.time() + args.duration
    counts = [0] * args.threads
    errors = [0] * args.threads

    def worker(i):
        while time.time() < stop:
            try:
                with urllib.request.urlopen(args.url, timeout=5) as r:
                    r.read()
                counts[i] += 1
            except Exception:
                errors[i] += 1

    ts = [threading.Thread(target=worker, args=(i,)) for i in range(args.threads)]
    t0 = time.time()
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    el = time.time() - t0
    print(f"requests={sum(counts)} errors={sum(errors)} qps={sum(counts)/el:.0f}")


if __name__ == "__main__":
    main()
