#!/usr/bin/env python3
"""Backup requests (≙ example/backup_request_c++): a slow replica is raced
by a backup attempt after backup_request_ms."""
import sys

sys.path.insert(0, ".")
import brpc_amd as b

p1 = b.core.rpc.start_echo_server(0)
p2 = b.core.rpc.start_echo_server(0)
max_lat = b.core.combo.backup_request(p1, p2, 100, 3)
print(f"3 calls against a 1000ms-slow primary with 100ms backup: max latency {max_lat/1000:.0f} ms")
