"""TLS parity (reference brpc ChannelSSLOptions/ServerSSLOptions): the
whole socket byte-path (handshake, read, write) runs over OpenSSL when
enabled; certificates are PEM strings or file paths."""
import brpc_amd as b
import pytest


@pytest.fixture(scope="module")
def cert_key():
    return b.gen_self_signed_cert("localhost")


def _tls_server(cert_key):
    cert, key = cert_key
    srv = b.Server()
    srv.add_method("Echo", "Hi", lambda req, att: (req + b"!", att))
    port = srv.start(0, ssl_cert=cert, ssl_key=key)
    return srv, port


def test_tls_echo(cert_key):
    srv, port = _tls_server(cert_key)
    ch = b.Channel("127.0.0.1:%d" % port, ssl=True)
    resp, att, lat = ch.call("Echo.Hi", b"hello", attachment=b"raw")
    assert resp == b"hello!"
    assert att == b"raw"
    srv.stop()


def test_tls_many_calls_and_big_payload(cert_key):
    srv, port = _tls_server(cert_key)
    # generous deadline: the suite leaves leaked background fibers/servers
    # from earlier tests competing for workers
    ch = b.Channel("127.0.0.1:%d" % port, ssl=True, timeout_ms=20000)
    big = bytes(range(256)) * 1024  # 256 KiB crosses the 16 KiB TLS record chunking
    resp, _, _ = ch.call("Echo.Hi", big)
    assert resp == big + b"!"
    for i in range(20):
        resp, _, _ = ch.call("Echo.Hi", b"x%d" % i)
        assert resp == b"x%d!" % i
    srv.stop()


def test_plain_client_rejected_by_tls_server(cert_key):
    srv, port = _tls_server(cert_key)
    ch = b.Channel("127.0.0.1:%d" % port, max_retry=0, timeout_ms=2000)
    with pytest.raises(b.RpcError):
        ch.call("Echo.Hi", b"hello")
    srv.stop()


def test_tls_with_auth(cert_key):
    cert, key = cert_key
    srv = b.Server()
    srv.add_method("Echo", "Hi", lambda req, att: (req, b""))
    port = srv.start(0, auth_user="u", auth_password="p", ssl_cert=cert, ssl_key=key)
    ch = b.Channel("127.0.0.1:%d" % port, ssl=True, auth_user="u", auth_password="p")
    assert ch.call("Echo.Hi", b"ok")[0] == b"ok"
    srv.stop()
    # Bad credential must hit a FRESH connection (auth is per-connection and
    # channels to the same endpoint share sockets — reference semantics), so
    # use a second server instance.
    srv2 = b.Server()
    srv2.add_method("Echo", "Hi", lambda req, att: (req, b""))
    port2 = srv2.start(0, auth_user="u", auth_password="p", ssl_cert=cert, ssl_key=key)
    bad = b.Channel("127.0.0.1:%d" % port2, ssl=True, auth_user="u", auth_password="x",
                    max_retry=0)
    with pytest.raises(b.RpcError) as ei:
        bad.call("Echo.Hi", b"ok")
    assert ei.value.args[0] == 1004
    srv2.stop()


def test_bad_cert_rejected_at_start():
    srv = b.Server()
    srv.add_method("Echo", "Hi", lambda req, att: (req, b""))
    with pytest.raises(RuntimeError):
        srv.start(0, ssl_cert="-----BEGIN CERTIFICATE-----\ngarbage\n-----END CERTIFICATE-----\n",
                  ssl_key="nope")
