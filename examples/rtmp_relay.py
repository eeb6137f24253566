#!/usr/bin/env python3
"""RTMP publish->play relay (≙ reference example/rtmp_press + media-server
basis): one publisher, one player through the built-in hub."""
import sys

sys.path.insert(0, ".")
import brpc_amd as b

port = b.core.rpc.start_rtmp_server()
pub = b.core.rpc.RtmpClient()
assert pub.connect("127.0.0.1", port, "live") == 0
assert pub.publish("cam") == 0
ply = b.core.rpc.RtmpClient()
assert ply.connect("127.0.0.1", port, "live") == 0
assert ply.play("cam") == 0
pub.push_frame(9, 40, b"\x17" + b"fake-keyframe" * 10)
t, ts, payload = ply.poll_frame(3000)
print(f"player got type={t} ts={ts} bytes={len(payload)}")
