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
