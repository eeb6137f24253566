#!/usr/bin/env python3
"""Echo server (≙ reference example/echo_c++/server.cpp).

Serves the std protocol (baidu_std wire), HTTP/1.1, HTTP/2+gRPC and the
builtin pages on ONE port. Run: python examples/echo_server.py [port]"""
import sys
import time

sys.path.insert(0, ".")
import brpc_amd as b

srv = b.Server()
srv.add_method("EchoService", "Echo", lambda req, att: (req, att))
port = srv.start(int(sys.argv[1]) if len(sys.argv) > 1 else 8000)
print(f"echo server on :{port}  (try: curl localhost:{port}/status)")
while True:
    time.sleep(3600)
