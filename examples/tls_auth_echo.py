#!/usr/bin/env python3
"""TLS + authenticated echo (≙ reference SSL options + authenticator.h):
self-signed server cert, password credential verified per connection."""
import sys

sys.path.insert(0, ".")
import brpc_amd as b

cert, key = b.gen_self_signed_cert("localhost")
srv = b.Server()
srv.add_method("Echo", "Hi", lambda req, att: (req, att))
port = srv.start(0, auth_user="svc", auth_password="hunter2", ssl_cert=cert, ssl_key=key)

ch = b.Channel(f"127.0.0.1:{port}", ssl=True, auth_user="svc", auth_password="hunter2")
resp, _, lat = ch.call("Echo.Hi", b"over TLS with auth")
print(f"reply={resp!r} latency={lat}us")
srv.stop()
