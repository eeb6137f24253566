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


def test_hostile_content_length_rejected(port):
    """a negative or absurd Content-Length must close the
    connection instead of growing read_buf_ forever (reference enforces
    FLAGS_max_body_size, http_message.cpp)."""
    import socket as pysock
    for cl in ("-1", "1152921504606846976", "99999999999999999999", "12x"):
        s = pysock.create_connection(("127.0.0.1", port), timeout=5)
        req = ("POST /EchoService/Echo HTTP/1.1\r\nHost: x\r\n"
               "Content-Length: %s\r\n\r\n" % cl).encode()
        s.sendall(req)
        s.settimeout(5)
        # Server must drop the connection (parse error), not wait for a body.
        try:
            data = s.recv(4096)
        except pysock.timeout:
            raise AssertionError("server kept connection open for Content-Length=%s" % cl)
        assert data == b"", (cl, data)
        s.close()


def test_normal_content_length_still_works(port):
    import socket as pysock
    s = pysock.create_connection(("127.0.0.1", port), timeout=5)
    body = b"hello"
    s.sendall(b"POST /EchoService/Echo HTTP/1.1\r\nHost: x\r\n"
              b"Content-Length: %d\r\n\r\n%s" % (len(body), body))
    s.settimeout(5)
    data = s.recv(65536)
    assert b"200" in data.split(b"\r\n", 1)[0]
    assert data.endswith(body)
    s.close()


def test_progressive_response_reading(port):
    """Controller::response_read_progressively (≙ reference
    ProgressiveReader / response_read_progressively): body bytes stream to
    the reader as they arrive; the buffered response stays empty."""
    big = bytes(range(256)) * 4096  # 1 MB
    srv = b.Server()
    srv.add_method("Big", "Blob", lambda req, att: (big, b""))
    p = srv.start(0)
    chunks, saw_done, resp = b.core.http_call_progressive(
        "127.0.0.1:%d" % p, "/Big/Blob", b"x", 10000)
    assert saw_done
    assert resp == b""          # nothing buffered
    assert b"".join(chunks) == big
    srv.stop()


def test_progressive_chunked_response():
    """Progressive delivery of a CHUNKED body from a raw scripted server,
    including a chunk split across TCP segments."""
    import socket as pysock
    import threading
    srv = pysock.socket()
    srv.bind(("127.0.0.1", 0))
    srv.listen(1)

    def run():
        c, _ = srv.accept()
        c.recv(4096)
        c.sendall(b"HTTP/1.1 200 OK\r\nTransfer-Encoding: chunked\r\n\r\n")
        c.sendall(b"5\r\nhello\r\n")
        c.sendall(b"8\r\nwor")   # chunk split mid-payload
        import time; time.sleep(0.05)
        c.sendall(b"ld!!!\r\n")
        c.sendall(b"0\r\n\r\n")
        import time as t2; t2.sleep(0.2)
        c.close()

    threading.Thread(target=run, daemon=True).start()
    sport = srv.getsockname()[1]
    chunks, saw_done, resp = b.core.http_call_progressive(
        "127.0.0.1:%d" % sport, "/stream/x", b"y", 8000)
    assert saw_done
    assert b"".join(chunks) == b"helloworld!!!"
    srv.close()


def test_http_header_ext():
    """Controller::http_request()/http_response() custom verb, headers,
    status and content-type (≙ reference HttpHeader accessors)."""
    ok, err = r.http_header_ext_test()
    assert ok, err


def test_h2_http_call_headers():
    """Channel(protocol='h2').http_call: the HttpHeaderExt views work on
    the in-tree h2 client too (custom header out, status/headers back)."""
    import brpc_amd
    port = r.start_echo_server(0)
    ch = brpc_amd.Channel("127.0.0.1:%d" % port, protocol="h2", timeout_ms=3000)
    status, headers, body = ch.http_call("/health")
    assert status == 200 and body == b"OK\n"
    status, headers, body = ch.http_call("/EchoService/Echo", b"h2ping",
                                         headers={"X-H2-Probe": "yes"})
    assert status == 200 and body == b"h2ping"
