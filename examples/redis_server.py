#!/usr/bin/env python3
"""Redis-protocol server (≙ example/redis_c++): speak RESP to any redis
client. Try: redis-cli -p PORT set k v"""
import sys
import time

sys.path.insert(0, ".")
import brpc_amd as b

store = {}
srv = b.RedisServer()
srv.add_handler("SET", lambda a: (store.__setitem__(a[1], a[2]), "OK")[1])
srv.add_handler("GET", lambda a: store.get(a[1]))
srv.add_handler("DEL", lambda a: 1 if store.pop(a[1], None) is not None else 0)
port = srv.start(int(sys.argv[1]) if len(sys.argv) > 1 else 0)
print(f"redis server on :{port}")
print("self-test:", b.redis_call(f"127.0.0.1:{port}", ["SET", "k", "v"]),
      b.redis_call(f"127.0.0.1:{port}", ["GET", "k"]))
if len(sys.argv) > 1:
    while True:
        time.sleep(3600)
