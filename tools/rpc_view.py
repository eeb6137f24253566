#!/usr/bin/env python3
"""rpc_view: fetches and pretty-prints another server's builtin pages
(parity: reference tools/rpc_view)."""
import argparse
import sys
import urllib.request


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("server", help="host:port")
    ap.add_argument("page", nargs="?", default="status",
                    help="status|vars|flags|connections|rpcz|fibers|memory|health")
    ap.add_argument("--filter", default="")
    args = ap.parse_args()
    url = f"http://{args.server}/{args.page}"
    if args.filter:
        url += f"?filter={args.filter}"
    with urllib.request.urlopen(url, timeout=10) as resp:
        sys.stdout.write(resp.read().decode())
    return 0


if __name__ == "__main__":
    sys.exit(main())
