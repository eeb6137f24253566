"""Native HTTP/1.1 and plain-h2 client protocols (reference
http_rpc_protocol.cpp client half + http2_rpc_protocol.cpp "h2"):
CallMethod over protocol="http"/"h2" hits RPC methods at /Service/Method
and builtin pages at /page, FIFO (h1) / stream (h2) correlation."""
import brpc_amd as b
import pytest

r = b.core.rpc

EHTTP = 1010


@pytest.fixture(scope="module")
def port():
    return r.start_echo_server(0)


@pytest.mark.parametrize("proto", ["http", "h2"])
def test_rpc_over_http_semantics(port, proto):
    rc, resp, err = r.protocol_call("127.0.0.1:%d" % port, proto,
                                    "EchoService.Echo", b"body-" + proto.encode())
    assert rc == 0, err
    assert resp == b"body-" + proto.encode()


@pytest.mark.parametrize("proto", ["http", "h2"])
def test_builtin_page_over(port, proto):
    rc, resp, err = r.protocol_call("127.0.0.1:%d" % port, proto, "health", b"")
    assert rc == 0, err
    assert resp == b"OK\n"


@pytest.mark.parametrize("proto", ["http", "h2"])
def test_unknown_path_is_http_error(port, proto):
    rc, resp, err = r.protocol_call("127.0.0.1:%d" % port, proto, "no/such", b"x")
    assert rc == EHTTP, (rc, err)


def test_http_large_body_roundtrip(port):
    big = bytes(range(256)) * 2000  # 512 KB
    rc, resp, err = r.protocol_call("127.0.0.1:%d" % port, "http",
                                    "EchoService.Echo", big)
    assert rc == 0, err
    assert resp == big


def test_http_pipelined_sequence(port):
    for i in range(30):
        rc, resp, err = r.protocol_call("127.0.0.1:%d" % port, "http",
                                        "EchoService.Echo", b"n%d" % i)
        assert rc == 0, err
        assert resp == b"n%d" % i


def test_server_error_maps_to_500(port):
    rc, resp, err = r.protocol_call("127.0.0.1:%d" % port, "http",
                                    "EchoService.Fail", b"x")
    assert rc == EHTTP
    assert "500" in err
