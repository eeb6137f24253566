#!/usr/bin/env python3
"""Streaming RPC (≙ example/streaming_echo_c++): flow-controlled frames."""
import sys

sys.path.insert(0, ".")
import brpc_amd as b

port = b.core.stream.start_server()
rc, err = b.core.stream.echo_test(port, 100, 1 << 20)
assert rc == 0, err
mbps = b.core.stream.throughput(port, 200, 1 << 20)
print(f"streamed 100x1MiB frames round-trip OK; one-way throughput {mbps:.0f} MB/s")
