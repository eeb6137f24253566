#!/usr/bin/env python3
"""Generic proxy via the master catch-all + distributed tracing
(≙ reference example/baidu_master_c++ on BaiduMasterService and the rpcz
trace chain): a proxy server forwards ANY service/method it does not know
to a backend, and the trace ids stitch the hops together in /rpcz.
"""
import sys
import urllib.request

sys.path.insert(0, ".")
import brpc_amd as b

r = b.core.rpc

# backend: a plain echo server
backend_port = r.start_echo_server(0)
backend = "127.0.0.1:%d" % backend_port

# proxy: no services of its own — the master catch-all relays verbatim.
# (start_master_echo_server demonstrates the C++ ServerOptions.master_handler;
# here we use the Relay method to forward with tracing on.)
proxy_port = r.start_echo_server(0)
proxy = "127.0.0.1:%d" % proxy_port

b.core.util.set_flag("rpcz_sample_mod", "1")
rc, resp, err = r.call_method_once(
    proxy, "EchoService.Relay", ("%s|traced hop" % backend).encode(), 3000, 0)
assert rc == 0, err
assert resp == b"traced hop"

# the proxy's /rpcz shows all three spans sharing one trace id
body = urllib.request.urlopen(
    "http://127.0.0.1:%d/rpcz?verbose" % proxy_port, timeout=5).read().decode()
traced = [l for l in body.splitlines() if "trace=" in l]
assert traced, "no traced spans on /rpcz"
print("proxied 1 call via Relay; %d traced spans visible at /rpcz" % len(traced))

# the pure master-handler variant: unknown methods land in the catch-all
mport = r.start_master_echo_server()
rc, resp, err = r.call_method_once("127.0.0.1:%d" % mport,
                                   "Any.Thing", b"payload", 3000, 0)
assert rc == 0 and resp == b"master:Any.Thing:payload", (rc, resp, err)
print("master catch-all answered for an unregistered service/method")
