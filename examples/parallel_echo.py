#!/usr/bin/env python3
"""ParallelChannel fan-out (≙ example/parallel_echo_c++): one call fans
out to N echo servers; responses merge in order."""
import sys

sys.path.insert(0, ".")
import brpc_amd as b

ports = [b.core.rpc.start_echo_server(0) for _ in range(4)]
rc, merged, err = b.core.combo.parallel_echo(ports, b"X", -1)
assert rc == 0, err
print(f"fanned out to {len(ports)} servers; merged response: {merged}")
