"""Streaming RPC tests (≙ reference brpc_streaming_rpc_unittest.cpp).

Stream established through a normal RPC, DATA/CLOSE/FEEDBACK frames with
credit windows multiplexed on the same socket.
"""
import pytest

import brpc_amd as b

s = b.core.stream


@pytest.fixture(scope="module")
def port():
    p = s.start_server()
    assert p > 0
    return p


def test_stream_echo_small(port):
    rc, err = s.echo_test(port, 10, 1000)
    assert rc == 0, err


def test_stream_echo_many_frames(port):
    rc, err = s.echo_test(port, 500, 4096)
    assert rc == 0, err


def test_stream_echo_1mb_frames(port):
    # BASELINE config 3 shape: 1 MiB frames (host path here; xGMI in bench)
    rc, err = s.echo_test(port, 20, 1 << 20)
    assert rc == 0, err


def test_stream_flow_control_window(port):
    # 64 MiB through an 8 MiB window forces FEEDBACK-driven flow control.
    mbps = s.throughput(port, 64, 1 << 20)
    assert mbps > 1, mbps


def test_stream_throughput_sane(port):
    mbps = s.throughput(port, 200, 1 << 20)
    assert mbps > 50, f"stream throughput too low: {mbps} MB/s"
