"""Python-facing Server/Channel API: Python-defined service handlers run on
the usercode pthread pool (≙ reference usercode_in_pthread), never on
fiber workers."""
import threading

import pytest

import brpc_amd as b


@pytest.fixture(scope="module")
def py_server():
    srv = b.Server()

    def upper(request, attachment):
        return request.upper()

    def with_attachment(request, attachment):
        return request[::-1], attachment + b"!"

    def boom(request, attachment):
        raise ValueError("kaboom")

    threads_seen = set()

    def which_thread(request, attachment):
        threads_seen.add(threading.get_ident())
        return b"ok"

    srv.add_method("PySvc", "Upper", upper)
    srv.add_method("PySvc", "WithAtt", with_attachment)
    srv.add_method("PySvc", "Boom", boom)
    srv.add_method("PySvc", "WhichThread", which_thread)
    port = srv.start(0)
    assert port > 0
    return srv, port, threads_seen


def test_python_handler_roundtrip(py_server):
    _, port, _ = py_server
    ch = b.Channel(f"127.0.0.1:{port}", timeout_ms=3000)
    resp, att, lat = ch.call("PySvc.Upper", b"hello")
    assert resp == b"HELLO"
    assert lat > 0


def test_python_handler_attachment(py_server):
    _, port, _ = py_server
    ch = b.Channel(f"127.0.0.1:{port}", timeout_ms=3000)
    resp, att, _ = ch.call("PySvc.WithAtt", b"abc", attachment=b"bin")
    assert resp == b"cba"
    assert att == b"bin!"


def test_python_handler_exception_propagates(py_server):
    _, port, _ = py_server
    ch = b.Channel(f"127.0.0.1:{port}", timeout_ms=3000)
    with pytest.raises(b.RpcError) as ei:
        ch.call("PySvc.Boom", b"x")
    code, msg = ei.value.args
    assert code == 2001  # EINTERNAL
    assert "kaboom" in msg


def test_handlers_not_on_fiber_workers(py_server):
    _, port, threads_seen = py_server
    ch = b.Channel(f"127.0.0.1:{port}", timeout_ms=3000)
    for _ in range(10):
        ch.call("PySvc.WhichThread", b"")
    assert threading.get_ident() not in threads_seen  # ran on pool threads


def test_concurrent_python_calls(py_server):
    _, port, _ = py_server
    ch = b.Channel(f"127.0.0.1:{port}", timeout_ms=5000)
    errs = []

    def worker():
        try:
            for i in range(50):
                resp, _, _ = ch.call("PySvc.Upper", b"x%d" % i)
                assert resp == b"X%d" % i
        except Exception as e:  # pragma: no cover
            errs.append(e)

    ts = [threading.Thread(target=worker) for _ in range(4)]
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    assert not errs


def test_unknown_method_raises(py_server):
    _, port, _ = py_server
    ch = b.Channel(f"127.0.0.1:{port}", timeout_ms=2000)
    with pytest.raises(b.RpcError) as ei:
        ch.call("PySvc.Nope", b"x")
    assert ei.value.args[0] == 1002  # ENOMETHOD


def test_session_local_data_per_connection():
    """≙ reference ServerOptions::session_local_data_factory +
    Controller::session_local_data: one lazily-created object per
    connection, destroyed at connection recycle."""
    r = b.core.rpc
    port = r.start_session_counter_server()
    # same (cached single) connection: counter grows
    vals = [r.protocol_call("127.0.0.1:%d" % port, "std", "Sess.Count", b"")[1]
            for _ in range(3)]
    assert vals == [b"1", b"2", b"3"], vals


def test_retry_policy_hook():
    """ChannelOptions::retry_policy (≙ reference brpc/retry_policy.h):
    a policy returning false stops retries after the first failure."""
    n = b.core.rpc.retry_policy_test(3)
    assert n == 1, n


def test_start_cancel():
    """Controller::StartCancel (≙ reference StartCancel/IsCanceled):
    cancels an in-flight call with ECANCELED well before the server's
    800ms handler or the 5s deadline."""
    port = b.core.rpc.start_echo_server(0)
    lat_us = b.core.rpc.cancel_test(port)
    assert lat_us > 0, "expected ECANCELED, got errno %d" % -lat_us
    assert lat_us < 700000, lat_us  # canceled long before the sleep finished


def test_channel_http_call_headers():
    """Channel(protocol='http').http_call: custom verb + headers out,
    (status, headers, body) back (python face of the HttpHeaderExt views)."""
    port = b.core.rpc.start_echo_server(0)
    ch = b.Channel("127.0.0.1:%d" % port, protocol="http", timeout_ms=3000)
    status, headers, body = ch.http_call("/health")
    assert status == 200 and body == b"OK\n"
    assert "content-type" in headers
    # RPC over http with a custom header (server echoes the body)
    status, headers, body = ch.http_call("/EchoService/Echo", b"ping",
                                         method="POST",
                                         headers={"X-Probe": "1"})
    assert status == 200 and body == b"ping"


def test_channel_request_code_kwarg():
    port = b.core.rpc.start_echo_server(0)
    ch = b.Channel("list://127.0.0.1:%d" % port, lb="c_hash", timeout_ms=3000)
    resp, att, lat = ch.call("EchoService.Echo", b"x", request_code=42,
                             has_request_code=True)
    assert resp == b"x"


def test_channel_http_call_404_and_error_paths():
    """http_call surfaces HTTP error statuses as data (status, headers,
    body), and transport failures as RpcError."""
    port = b.core.rpc.start_echo_server(0)
    ch = b.Channel("127.0.0.1:%d" % port, protocol="http", timeout_ms=3000)
    status, headers, body = ch.http_call("/no/such/page")
    assert status == 404 and b"no such page" in body
    bad = b.Channel("127.0.0.1:1", protocol="http", timeout_ms=300)
    try:
        bad.http_call("/x")
        assert False, "expected RpcError"
    except b.RpcError:
        pass


def test_short_and_pooled_coexist():
    """A short-conn channel and a pooled channel to the same server work
    side by side (no registry cross-talk)."""
    port = b.core.rpc.start_echo_server(0)
    addr = "127.0.0.1:%d" % port
    ok, err = b.core.rpc.short_connection_test()  # uses its own server
    assert ok, err
    pooled = b.Channel(addr, timeout_ms=2000)
    for _ in range(4):
        resp, _, _ = pooled.call("EchoService.Echo", b"both")
        assert resp == b"both"
