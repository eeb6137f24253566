"""HTTP/2 + gRPC server interop, tested with the OFFICIAL grpc python
client against our server (parity: reference h2/gRPC support)."""
import pytest

grpc = pytest.importorskip("grpc")

import brpc_amd as b  # noqa: E402

r = b.core.rpc


@pytest.fixture(scope="module")
def addr():
    port = r.start_echo_server(0)
    return f"127.0.0.1:{port}"


def _stub(channel, method):
    return channel.unary_unary(method,
                               request_serializer=lambda x: x,
                               response_deserializer=lambda x: x)


def test_grpc_unary_echo(addr):
    with grpc.insecure_channel(addr) as ch:
        call = _stub(ch, "/EchoService/Echo")
        resp = call(b"grpc interop payload", timeout=10)
        assert resp == b"grpc interop payload"


def test_grpc_multiple_calls_one_channel(addr):
    with grpc.insecure_channel(addr) as ch:
        call = _stub(ch, "/EchoService/Echo")
        for i in range(20):
            assert call(b"msg %d" % i, timeout=10) == b"msg %d" % i


def test_grpc_unknown_method_unimplemented(addr):
    with grpc.insecure_channel(addr) as ch:
        call = _stub(ch, "/EchoService/NoSuch")
        with pytest.raises(grpc.RpcError) as ei:
            call(b"x", timeout=10)
        assert ei.value.code() == grpc.StatusCode.UNIMPLEMENTED


def test_grpc_server_error_maps_to_internal(addr):
    with grpc.insecure_channel(addr) as ch:
        call = _stub(ch, "/EchoService/Fail")
        with pytest.raises(grpc.RpcError) as ei:
            call(b"x", timeout=10)
        assert ei.value.code() == grpc.StatusCode.INTERNAL


def test_grpc_large_payload(addr):
    import os
    data = os.urandom(1 << 20)
    with grpc.insecure_channel(addr) as ch:
        call = _stub(ch, "/EchoService/Echo")
        assert call(data, timeout=20) == data


# ---- OUR gRPC client ----

def test_our_grpc_client_to_our_server(addr):
    rc, resp, err = b.core.combo.grpc_call(addr, "EchoService/Echo", b"c2s payload")
    assert rc == 0, err
    assert resp == b"c2s payload"


def test_our_grpc_client_to_real_grpc_server():
    """Ultimate interop: OUR h2/gRPC client against the OFFICIAL grpc
    python server."""
    from concurrent import futures

    class Handler(grpc.GenericRpcHandler):
        def service(self, handler_call_details):
            if handler_call_details.method == "/Real/Upper":
                return grpc.unary_unary_rpc_method_handler(
                    lambda req, ctx: req.upper(),
                    request_deserializer=lambda x: x,
                    response_serializer=lambda x: x)
            return None

    server = grpc.server(futures.ThreadPoolExecutor(max_workers=4))
    server.add_generic_rpc_handlers((Handler(),))
    port = server.add_insecure_port("127.0.0.1:0")
    server.start()
    try:
        rc, resp, err = b.core.combo.grpc_call(f"127.0.0.1:{port}", "Real/Upper",
                                               b"interop!", 5000)
        assert rc == 0, err
        assert resp == b"INTEROP!"
    finally:
        server.stop(0)


def test_our_grpc_client_unknown_method(addr):
    rc, resp, err = b.core.combo.grpc_call(addr, "EchoService/Missing", b"x")
    assert rc != 0
    assert "12" in err or "grpc-status" in err


def test_builtin_grpc_health_check(addr):
    """Builtin grpc.health.v1.Health/Check (≙ reference grpc_health_check):
    responds SERVING without a user-registered service."""
    with grpc.insecure_channel(addr) as ch:
        call = _stub(ch, "/grpc.health.v1.Health/Check")
        resp = call(b"", timeout=5)
        assert resp == b"\x08\x01"  # status: SERVING


def test_grpc_flow_control_large_response(addr):
    """>1 MB response exceeds the client's default 64 KB h2 windows: the
    in-tree session must queue DATA and resume on WINDOW_UPDATE."""
    import grpc
    channel = grpc.insecure_channel(addr)
    big = bytes(range(256)) * 5000  # 1.25 MB
    fut = channel.unary_unary("/EchoService/Echo",
                              request_serializer=lambda x: x,
                              response_deserializer=lambda x: x)
    resp = fut(big, timeout=20)
    assert resp == big
    channel.close()


def test_grpc_health_check(addr):
    """grpc.health.v1.Health/Check (≙ reference grpc_health_check): the
    official grpc client probes health and gets SERVING (field 1 = 1)."""
    with grpc.insecure_channel(addr) as ch:
        call = ch.unary_unary("/grpc.health.v1.Health/Check")
        resp = call(b"", timeout=10)
        assert resp == b"\x08\x01"
