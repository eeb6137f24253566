"""Memcache binary-protocol client against a minimal in-test memcached
(binary protocol) server."""
import socket
import struct
import threading

import pytest

import brpc_amd as b


def start_fake_memcached():
    """Tiny memcached-binary server: SET/GET/DELETE/VERSION on a dict."""
    store = {}
    lsock = socket.socket()
    lsock.bind(("127.0.0.1", 0))
    lsock.listen(8)
    port = lsock.getsockname()[1]

    def handle(conn):
        try:
            buf = b""
            while True:
                while len(buf) < 24:
                    chunk = conn.recv(65536)
                    if not chunk:
                        return
                    buf += chunk
                magic, op, klen = struct.unpack(">BBH", buf[:4])
                xlen = buf[4]
                blen = struct.unpack(">I", buf[8:12])[0]
                while len(buf) < 24 + blen:
                    chunk = conn.recv(65536)
                    if not chunk:
                        return
                    buf += chunk
                body = buf[24:24 + blen]
                buf = buf[24 + blen:]
                key = body[xlen:xlen + klen]
                value = body[xlen + klen:]
                status, rx, rv = 0, b"", b""
                if op == 0x01:  # set
                    store[key] = value
                elif op == 0x00:  # get
                    if key in store:
                        rx = b"\x00" * 4  # flags extras
                        rv = store[key]
                    else:
                        status = 1
                elif op == 0x04:  # delete
                    if store.pop(key, None) is None:
                        status = 1
                elif op == 0x0B:  # version
                    rv = b"1.6.0-fake"
                elif op == 0x21:  # SASL auth (couchbase-style PLAIN)
                    if key == b"PLAIN" and value == b"\x00cb_user\x00cb_pass":
                        rv = b"Authenticated"
                    else:
                        status = 0x20  # auth error
                resp_body = rx + rv
                hdr = struct.pack(">BBHBBHIIQ", 0x81, op, 0, len(rx), 0, status,
                                  len(resp_body), 0, 0)
                conn.sendall(hdr + resp_body)
        except OSError:
            pass

    def acceptor():
        while True:
            try:
                conn, _ = lsock.accept()
            except OSError:
                return
            threading.Thread(target=handle, args=(conn,), daemon=True).start()

    threading.Thread(target=acceptor, daemon=True).start()
    return port


@pytest.fixture(scope="module")
def mc_port():
    return start_fake_memcached()


def test_memcache_set_get_delete(mc_port):
    c = b.MemcacheClient(f"127.0.0.1:{mc_port}", timeout_ms=3000)
    assert c.ok()
    assert c.set("k1", b"hello memcache") == 0
    assert c.get("k1") == b"hello memcache"
    assert c.delete("k1") == 0
    assert c.get("k1") is None


def test_memcache_version(mc_port):
    c = b.MemcacheClient(f"127.0.0.1:{mc_port}", timeout_ms=3000)
    assert c.version() == "1.6.0-fake"


def test_memcache_many_pipelined(mc_port):
    c = b.MemcacheClient(f"127.0.0.1:{mc_port}", timeout_ms=3000)
    for i in range(50):
        assert c.set(f"key{i}", b"v%d" % i) == 0
    for i in range(50):
        assert c.get(f"key{i}") == b"v%d" % i


def test_couchbase_sasl_auth(mc_port):
    """Couchbase parity (reference policy/couchbase_authenticator.cpp):
    SASL PLAIN over the memcache binary protocol."""
    c = b.MemcacheClient("127.0.0.1:%d" % mc_port)
    assert c.ok()
    assert c.sasl_auth_plain("cb_user", "cb_pass") == 0
    rc = c.sasl_auth_plain("cb_user", "wrong")
    assert rc == 10000 + 0x20


def start_vbucket_memcached(expected_owner, owner_id, redirect):
    """memcached-binary mock that CHECKS the request's vbucket id: serves
    keys whose vbucket it owns (per `expected_owner` map), answers
    NOT_MY_VBUCKET (0x0007) when `redirect[0]` is set or it is not the
    owner. Returns (port, store, seen_vbuckets)."""
    store = {}
    seen = []
    lsock = socket.socket()
    lsock.bind(("127.0.0.1", 0))
    lsock.listen(8)
    port = lsock.getsockname()[1]

    def handle(conn):
        try:
            buf = b""
            while True:
                while len(buf) < 24:
                    chunk = conn.recv(65536)
                    if not chunk:
                        return
                    buf += chunk
                magic, op, klen = struct.unpack(">BBH", buf[:4])
                xlen = buf[4]
                vbucket = struct.unpack(">H", buf[6:8])[0]
                blen = struct.unpack(">I", buf[8:12])[0]
                while len(buf) < 24 + blen:
                    chunk = conn.recv(65536)
                    if not chunk:
                        return
                    buf += chunk
                body = buf[24:24 + blen]
                buf = buf[24 + blen:]
                key = body[xlen:xlen + klen].decode()
                val = body[xlen + klen:]
                seen.append((op, key, vbucket))
                status = 0
                out_val = b""
                if redirect[0] or expected_owner.get(vbucket) != owner_id:
                    status = 0x0007  # NOT_MY_VBUCKET
                elif op == 0x01:
                    store[key] = val
                elif op == 0x00:
                    if key in store:
                        out_val = store[key]
                    else:
                        status = 1
                elif op == 0x04:
                    store.pop(key, None)
                extras = b"\x00" * 4 if op == 0x00 and status == 0 else b""
                resp = struct.pack(">BBHBBHIIQ", 0x81, op, 0, len(extras), 0,
                                   status, len(extras) + len(out_val), 0, 0)
                conn.sendall(resp + extras + out_val)
        except OSError:
            pass

    def acceptor():
        while True:
            try:
                conn, _ = lsock.accept()
            except OSError:
                return
            threading.Thread(target=handle, args=(conn,), daemon=True).start()

    threading.Thread(target=acceptor, daemon=True).start()
    return port, store, seen


def test_couchbase_vbucket_routing():
    """CouchbaseClient (rpc/couchbase.*): REST vBucketServerMap config,
    CRC32-based key->vbucket mapping, per-request vbucket ids, and
    NOT_MY_VBUCKET -> config refetch + retry (rebalance)."""
    NVB = 64
    # phase 1: node A owns even vbuckets, node B odd.
    ownerA = {vb: 0 for vb in range(0, NVB, 2)}
    ownerA.update({vb: 1 for vb in range(1, NVB, 2)})
    redirA = [False]
    redirB = [False]
    ownerB = dict(ownerA)
    portA, storeA, seenA = start_vbucket_memcached(ownerA, 0, redirA)
    portB, storeB, seenB = start_vbucket_memcached(ownerB, 1, redirB)

    phase = [1]
    cfg_srv = b.Server()

    def config(req, att):
        import json as pyjson
        if phase[0] == 1:
            vmap = [[vb % 2] for vb in range(NVB)]
        else:
            vmap = [[1] for _ in range(NVB)]  # everything moved to B
        body = pyjson.dumps({"vBucketServerMap": {
            "serverList": ["127.0.0.1:%d" % portA, "127.0.0.1:%d" % portB],
            "vBucketMap": vmap,
        }}).encode()
        return body, b""

    cfg_srv.add_method("pools", "cfg", config)
    cfg_srv.add_restful_mapping("pools", "/pools/default/b/app => cfg")
    cfg_port = cfg_srv.start(0)

    c = b.core.CouchbaseClient()
    assert c.init("127.0.0.1:%d" % cfg_port, "app") == 0, c.last_error()
    assert c.nvbuckets() == NVB and c.nservers() == 2

    keys = ["user:%d" % i for i in range(40)]
    for k in keys:
        assert c.set(k, ("v-" + k).encode()) == 0
    for k in keys:
        assert c.get(k) == ("v-" + k).encode()
    # requests carried the right vbucket ids and split across both nodes
    for op, key, vb in seenA + seenB:
        assert vb == b.core.CouchbaseClient.vbucket_of(key.encode(), NVB)
    assert storeA and storeB

    # phase 2: rebalance — node A starts refusing; map moves all to B.
    phase[0] = 2
    redirA[0] = True
    ownerB.update({vb: 1 for vb in range(NVB)})  # B owns everything now
    k = next(k for k in keys if k in storeA)
    storeB[k] = b"moved"  # B owns it after rebalance
    assert c.get(k) == b"moved"  # NOT_MY_VBUCKET -> refetch -> retry on B
    cfg_srv.stop()
