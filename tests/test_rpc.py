"""End-to-end RPC tests over loopback TCP in one process.

Models the reference's ChannelTest fixture (test/brpc_channel_unittest.cpp):
in-process server + client, sync calls, errors, timeouts, attachments.
BASELINE config 1 (echo over loopback on CPU) is exercised here.
"""
import pytest

import brpc_amd as b

r = b.core.rpc

# error-code parity with reference brpc/errno.proto
ENOSERVICE = 1001
ENOMETHOD = 1002
ERPCTIMEDOUT = 1008
EINTERNAL = 2001


@pytest.fixture(scope="module")
def server_addr():
    port = r.start_echo_server(0)
    assert port > 0
    return f"127.0.0.1:{port}"


def test_echo_roundtrip(server_addr):
    rc, resp, lat = r.echo_once(server_addr, b"hello rpc world", 2000)
    assert rc == 0
    assert resp == b"hello rpc world"
    assert 0 < lat < 2_000_000


def test_echo_empty_payload(server_addr):
    rc, resp, _ = r.echo_once(server_addr, b"", 2000)
    assert rc == 0
    assert resp == b""


def test_echo_large_payload(server_addr):
    import os

    payload = os.urandom(1 << 20)  # 1 MiB spans many IOBuf blocks
    rc, resp, _ = r.echo_once(server_addr, payload, 10000)
    assert rc == 0
    assert resp == payload


def test_attachment_roundtrip(server_addr):
    assert r.attachment_test(server_addr)


def test_unknown_method(server_addr):
    rc, _, err = r.call_method_once(server_addr, "EchoService.NoSuch", b"x", 1000, 0)
    assert rc == ENOMETHOD
    assert "NoSuch" in err


def test_unknown_service(server_addr):
    rc, _, err = r.call_method_once(server_addr, "Nope.Echo", b"x", 1000, 0)
    assert rc in (ENOSERVICE, ENOMETHOD)


def test_server_side_failure(server_addr):
    rc, _, err = r.call_method_once(server_addr, "EchoService.Fail", b"x", 1000, 0)
    assert rc == EINTERNAL
    assert "asked for it" in err


def test_timeout(server_addr):
    rc, _, err = r.call_method_once(server_addr, "EchoService.Sleep", b"300", 100, 0)
    assert rc == ERPCTIMEDOUT


def test_slow_call_within_deadline(server_addr):
    rc, resp, _ = r.call_method_once(server_addr, "EchoService.Sleep", b"50", 2000, 0)
    assert rc == 0
    assert resp == b"slept"


def test_connection_refused_with_retries():
    rc, _, err = r.call_method_once("127.0.0.1:1", "EchoService.Echo", b"x", 2000, 2)
    assert rc != 0 and rc != ERPCTIMEDOUT


def test_concurrent_echo_bench_small(server_addr):
    res = r.echo_bench(server_addr, 64, 8, 2000, 5000)
    assert res["errors"] == 0
    assert res["total"] == 2000
    assert res["qps"] > 100


def test_concurrent_echo_16k(server_addr):
    res = r.echo_bench(server_addr, 16384, 8, 500, 10000)
    assert res["errors"] == 0


def test_interceptor_admission():
    """ServerOptions.interceptor rejects requests lacking the credential
    (parity: reference brpc/interceptor.h)."""
    port = b.core.combo.start_intercepted_server("777")
    addr = f"127.0.0.1:{port}"
    rc, err = b.core.combo.call_with_logid(addr, 777)
    assert rc == 0, err
    rc, err = b.core.combo.call_with_logid(addr, 123)
    assert rc == 1004  # EAUTH
    assert "credential" in err


def test_async_pipelined_bench():
    """Pipelined async client (completions reissue; ≙ reference async
    CallMethod w/ done, test/brpc_channel_unittest.cpp async paths):
    all calls complete, none lost, latencies recorded."""
    port = r.start_echo_server(0)
    addr = "127.0.0.1:%d" % port
    res = r.async_echo_bench(addr, 64, 16, 2000, 10000, "EchoService.Echo", True)
    assert res["errors"] == 0, res["first_error"]
    assert res["total"] == 2000
    assert res["qps"] > 0 and res["p99_us"] > 0


def test_echo_bench_multi_channel():
    """nchannels>1 spreads sync workers over independent Channels."""
    port = r.start_echo_server(0)
    addr = "127.0.0.1:%d" % port
    res = r.echo_bench(addr, 64, 8, 1000, 10000, "EchoService.Echo",
                       False, True, 4)
    assert res["errors"] == 0, res["first_error"]
    assert res["total"] == 1000


def test_channel_options_tail():
    """ns_filter, succeed_without_server, enable_circuit_breaker
    (≙ reference ChannelOptions, brpc/channel.h:52-163)."""
    ok, err = r.channel_options_tail_test()
    assert ok, err


def test_thread_local_data():
    """Controller::thread_local_data (≙ reference ServerOptions
    thread_local_data_factory): per-worker lazily created, stable."""
    ok, err = r.thread_local_data_test()
    assert ok, err


def test_request_code_consistent_hash():
    """set_request_code routes c_hash deterministically (≙ reference
    Controller::set_request_code + consistent_hashing LB)."""
    ok, err = r.request_code_test()
    assert ok, err


def test_short_connection_type():
    """connection_type='short': fresh connection per call, closed after
    (≙ reference CONNECTION_TYPE_SHORT)."""
    ok, err = r.short_connection_test()
    assert ok, err


@pytest.mark.timeout(60)
def test_idle_timeout_reaps_connections():
    """ServerOptions.idle_timeout_sec closes idle server-side connections
    (≙ reference idle_timeout_sec, brpc/server.h:62): an open socket with
    no traffic dies; an active channel keeps working."""
    import socket as pysock
    import time
    port = r.start_idle_timeout_server(1)  # 1-second idle timeout
    s = pysock.create_connection(("127.0.0.1", port), timeout=5)
    # connection should be closed by the reaper within ~2.5s
    s.settimeout(4)
    t0 = time.time()
    data = s.recv(1)  # blocks until the server closes (returns b"")
    assert data == b"", "expected server-side close"
    assert time.time() - t0 < 3.5
    s.close()
    # server still serves fresh connections
    rc, resp, _ = r.echo_once("127.0.0.1:%d" % port, b"alive", 3000)
    assert rc == 0 and resp == b"alive"
