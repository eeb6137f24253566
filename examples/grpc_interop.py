#!/usr/bin/env python3
"""gRPC interop (≙ reference h2/gRPC support): the OFFICIAL grpc python
client calling a brpc_amd server."""
import sys

sys.path.insert(0, ".")
import brpc_amd as b
import grpc

srv = b.Server()
srv.add_method("EchoService", "Echo", lambda req, att: req)
port = srv.start(0)
with grpc.insecure_channel(f"127.0.0.1:{port}") as ch:
    call = ch.unary_unary("/EchoService/Echo",
                          request_serializer=lambda x: x,
                          response_deserializer=lambda x: x)
    print("grpc says:", call(b"hello from real gRPC", timeout=5))
