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
