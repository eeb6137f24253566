"""Authentication parity (reference brpc/authenticator.h): the client sends
RpcMeta.authentication_data (baidu_std field 7); the server verifies once
per connection and rejects bad credentials with ERPCAUTH (1004)."""
import brpc_amd as b
import pytest

ERPCAUTH = 1004


def _start_server():
    srv = b.Server()
    srv.add_method("Echo", "Hi", lambda req, att: (req, b""))
    port = srv.start(0, auth_user="alice", auth_password="sesame")
    return srv, port


def test_good_credential():
    srv, port = _start_server()
    ch = b.Channel("127.0.0.1:%d" % port, auth_user="alice", auth_password="sesame")
    resp, att, _ = ch.call("Echo.Hi", b"ping")
    assert resp == b"ping"
    srv.stop()


def test_bad_credential_rejected():
    srv, port = _start_server()
    ch = b.Channel("127.0.0.1:%d" % port, auth_user="alice", auth_password="wrong",
                   max_retry=0)
    with pytest.raises(b.RpcError) as ei:
        ch.call("Echo.Hi", b"ping")
    assert ei.value.args[0] == ERPCAUTH
    srv.stop()


def test_missing_credential_rejected():
    srv, port = _start_server()
    ch = b.Channel("127.0.0.1:%d" % port, max_retry=0)
    with pytest.raises(b.RpcError) as ei:
        ch.call("Echo.Hi", b"ping")
    assert ei.value.args[0] == ERPCAUTH
    srv.stop()


def test_auth_cached_per_connection():
    srv, port = _start_server()
    ch = b.Channel("127.0.0.1:%d" % port, auth_user="alice", auth_password="sesame")
    for i in range(20):
        resp, _, _ = ch.call("Echo.Hi", b"x%d" % i)
        assert resp == b"x%d" % i
    srv.stop()


def test_no_auth_server_ignores_credential():
    srv = b.Server()
    srv.add_method("Echo", "Hi", lambda req, att: (req, b""))
    port = srv.start(0)
    ch = b.Channel("127.0.0.1:%d" % port, auth_user="bob", auth_password="pw")
    resp, _, _ = ch.call("Echo.Hi", b"ok")
    assert resp == b"ok"
    srv.stop()
