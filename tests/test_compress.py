"""Snappy codec + wire compression tests.

The host snappy codec (base/snappy.cc) is the oracle for the gfx950
snappy kernel; wire tests check compress_type end-to-end on the std
protocol (compressed request AND compressed response).
"""
import os
import random

import brpc_amd as b

sn = b.core.snappy
r = b.core.rpc

COMPRESS_SNAPPY = 1
COMPRESS_GZIP = 2


def test_snappy_roundtrip_simple():
    data = b"hello hello hello hello hello world" * 100
    comp = sn.compress(data)
    assert len(comp) < len(data) // 2  # repetitive: must compress well
    assert sn.uncompress(comp) == data


def test_snappy_roundtrip_random():
    data = os.urandom(100_000)  # incompressible
    comp = sn.compress(data)
    assert sn.uncompress(comp) == data


def test_snappy_roundtrip_sizes():
    random.seed(7)
    for n in [0, 1, 3, 59, 60, 61, 64, 1000, 65535, 65536, 65537, 300_000]:
        data = bytes(random.randrange(4) for _ in range(n))  # compressible
        assert sn.uncompress(sn.compress(data)) == data, n


def test_snappy_empty_and_tiny():
    assert sn.uncompress(sn.compress(b"")) == b""
    assert sn.uncompress(sn.compress(b"a")) == b"a"


def test_snappy_corrupt_returns_none():
    assert sn.uncompress(b"\xff\xff\xff\xff\xff\xff") is None


def test_wire_snappy_echo():
    port = r.start_echo_server(0)
    addr = f"127.0.0.1:{port}"
    payload = b"compress me " * 5000
    rc, resp = b.core.combo.compressed_echo(addr, payload, COMPRESS_SNAPPY)
    assert rc == 0
    assert resp == payload


def test_wire_gzip_request():
    port = r.start_echo_server(0)
    addr = f"127.0.0.1:{port}"
    payload = b"gzip payload " * 3000
    rc, resp = b.core.combo.compressed_echo(addr, payload, COMPRESS_GZIP)
    assert rc == 0
    assert resp == payload
