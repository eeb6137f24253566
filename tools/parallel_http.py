#!/usr/bin/env python3
"""parallel_http: fan-out fetcher for many URLs (parity: reference
tools/parallel_http)."""
import argparse
import concurrent.futures
import sys
import urllib.request


def fetch(url):
    try:
        with urllib.request.urlopen(url, timeout=10) as r:
            return url, r.status, len(r.read())
    except Exception as e:
        return url, -1, str(e)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--url-file", help="file with one URL per line (default: stdin)")
    ap.add_argument("--parallelism", type=int, default=16)
    args = ap.parse_args()
    src = open(args.url_file) if args.url_file else sys.stdin
    urls = [l.strip() for l in src if l.strip()]
    with concurrent.futures.ThreadPoolExecutor(args.parallelism) as ex:
        for url, status, info in ex.map(fetch, urls):
            print(f"{status}\t{info}\t{url}")


if __name__ == "__main__":
    main()
