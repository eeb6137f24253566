#!/usr/bin/env python3
"""Host 64B echo QPS push (round-2 verdict target: 500k+).

Scans the client-shape space on one box: sync fibers over pooled/single
connections, multiple client Channels, and the pipelined-async client
(completions reissue immediately, letting the wait-free write queue
coalesce requests — ≙ reference docs/cn/benchmark.md pipelined clients).
Prints one JSON line per config and writes --out.
"""
import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import brpc_amd as b  # noqa: E402

r = b.core.rpc


def run(cfg, calls):
    best = None
    for rep in range(3):
        payload = cfg.get("payload", 64)
        method = cfg.get("method", "EchoService.Echo")
        if cfg["mode"] == "sync":
            res = r.echo_bench(cfg["addr"], payload, cfg["conc"], calls, 30000,
                               method, False, cfg["pooled"],
                               cfg.get("nchannels", 1))
        else:
            res = r.async_echo_bench(cfg["addr"], payload, cfg["conc"], calls, 30000,
                                     method, cfg["pooled"])
        if res["errors"]:
            return {"qps": 0, "p99_us": -1, "errors": res["errors"],
                    "first_error": str(res.get("first_error"))}
        if rep == 0:
            continue
        if best is None or res["qps"] > best["qps"]:
            best = res
    return best


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--calls", type=int, default=40000)
    ap.add_argument("--out", default="gpurun_out/host_push.json")
    args = ap.parse_args()

    ndev = b.core.gpu.initialize()
    port = r.start_echo_server(0)
    addr = "127.0.0.1:%d" % port

    configs = []
    for c in (32, 64, 128, 256):
        configs.append(dict(mode="sync", conc=c, pooled=True, nchannels=1,
                            tag="sync_pooled_c%d" % c))
    for c in (64, 128, 256):
        configs.append(dict(mode="sync", conc=c, pooled=True, nchannels=4,
                            tag="sync_pooled_c%d_nch4" % c))
    for c in (64, 128):
        configs.append(dict(mode="sync", conc=c, pooled=False, nchannels=4,
                            tag="sync_single_c%d_nch4" % c))
    for p in (64, 128, 256):
        configs.append(dict(mode="async", conc=p, pooled=True,
                            tag="async_pooled_p%d" % p))
        configs.append(dict(mode="async", conc=p, pooled=False,
                            tag="async_single_p%d" % p))
    if ndev > 0:
        for p in (64, 128, 256):
            configs.append(dict(mode="async", conc=p, pooled=True,
                                method="EchoService.EchoHbm",
                                tag="hbm_async_pooled_p%d" % p))
        configs.append(dict(mode="async", conc=128, pooled=True, payload=16384,
                            method="EchoService.EchoHbm",
                            tag="hbm16k_async_pooled_p128"))
        configs.append(dict(mode="async", conc=128, pooled=True, payload=16384,
                            tag="host16k_async_pooled_p128"))

    results = [{"cpu_count": os.cpu_count()}]
    print(json.dumps(results[0]), flush=True)
    for cfg in configs:
        cfg["addr"] = addr
        best = run(cfg, args.calls)
        row = dict(tag=cfg["tag"], qps=round(best["qps"], 1),
                   p99_us=best.get("p99_us", -1))
        if best.get("errors"):
            row.update(errors=best["errors"], first_error=best["first_error"])
        results.append(row)
        print(json.dumps(row), flush=True)

    os.makedirs(os.path.dirname(args.out), exist_ok=True)
    with open(args.out, "w") as f:
        json.dump(results, f, indent=1)


if __name__ == "__main__":
    main()
