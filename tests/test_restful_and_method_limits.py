"""Restful URL mappings (reference AddService(..., restful_mappings)) and
per-method concurrency caps (≙ method_max_concurrency / MethodStatus gate)."""
import threading
import time
import urllib.request

import brpc_amd as b
import pytest


def test_restful_mapping_http():
    srv = b.Server()
    srv.add_method("Store", "Put", lambda req, att: (b"put:" + req, b""))
    srv.add_method("Store", "Stats", lambda req, att: (b"stats", b""))
    srv.add_restful_mapping("Store", "/v1/put => Put, /v1/stats/* => Stats")
    port = srv.start(0)
    base = "http://127.0.0.1:%d" % port
    req = urllib.request.Request(base + "/v1/put", data=b"xyz")
    with urllib.request.urlopen(req, timeout=5) as resp:
        assert resp.read() == b"put:xyz"
    with urllib.request.urlopen(base + "/v1/stats/anything/here", timeout=5) as resp:
        assert resp.read() == b"stats"
    # default /Service/Method still works alongside
    req = urllib.request.Request(base + "/Store/Put", data=b"abc")
    with urllib.request.urlopen(req, timeout=5) as resp:
        assert resp.read() == b"put:abc"
    srv.stop()


def test_bad_restful_mapping_rejected():
    srv = b.Server()
    srv.add_method("S", "M", lambda req, att: (req, b""))
    with pytest.raises(RuntimeError):
        srv.add_restful_mapping("S", "/x => NoSuchMethod")


def test_method_max_concurrency():
    srv = b.Server()

    def slow(req, att):
        time.sleep(0.05)
        return req, b""

    srv.add_method("S", "Slow", slow)
    srv.add_method("S", "Fast", lambda req, att: (req, b""))
    srv.set_method_max_concurrency("S.Slow", 2)
    port = srv.start(0)
    ok = [0]
    limited = [0]
    lock = threading.Lock()

    def run():
        ch = b.Channel("127.0.0.1:%d" % port, timeout_ms=3000, max_retry=0)
        for _ in range(4):
            try:
                ch.call("S.Slow", b"x")
                with lock:
                    ok[0] += 1
            except b.RpcError:
                with lock:
                    limited[0] += 1

    ts = [threading.Thread(target=run) for _ in range(10)]
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    assert limited[0] > 0, (ok, limited)   # cap of 2 under 10 hammers trips
    assert ok[0] > 0
    # other methods unaffected
    ch = b.Channel("127.0.0.1:%d" % port, timeout_ms=2000, max_retry=0)
    assert ch.call("S.Fast", b"f")[0] == b"f"
    srv.stop()
