#!/usr/bin/env python3
"""rpc_replay: replays rpc_dump-sampled traffic against a server
(parity: reference tools/rpc_replay + brpc/rpc_dump.h)."""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import brpc_amd as b  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--dump-file", required=True)
    ap.add_argument("--addr", required=True)
    ap.add_argument("--timeout-ms", type=int, default=1000)
    ap.add_argument("--times", type=int, default=1, help="replay the file N times")
    args = ap.parse_args()

    ch = b.Channel(args.addr, timeout_ms=args.timeout_ms)
    ok = fail = 0
    for _ in range(args.times):
        reader = b.core.util.RecordReader(args.dump_file)
        if not reader.ok():
            print(f"cannot open {args.dump_file}", file=sys.stderr)
            return 1
        while True:
            rec = reader.next()
            if rec is None:
                break
            decoded = b.core.util.decode_dump_record(rec)
            if decoded is None:
                print("corrupt record", file=sys.stderr)
                continue
            service, method, body = decoded
            try:
                ch.call(f"{service}.{method}", body)
                ok += 1
            except b.RpcError as e:
                fail += 1
    print(f"replayed ok={ok} failed={fail}")
    return 0 if fail == 0 else 1


if __name__ == "__main__":
    sys.exit(main())
