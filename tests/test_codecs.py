"""base64 / sha1 / murmur3 codecs + small containers + multidim vars."""
import base64
import hashlib
import os

import brpc_amd as b

c = b.core.codecs


def test_base64_matches_python():
    for n in (0, 1, 2, 3, 100, 1000):
        data = os.urandom(n)
        assert c.base64_encode(data) == base64.b64encode(data).decode()
        assert c.base64_decode(base64.b64encode(data).decode()) == data


def test_base64_reject_bad():
    assert c.base64_decode("!!!!") is None


def test_sha1_matches_hashlib():
    for data in (b"", b"abc", os.urandom(1000), b"x" * 10000):
        assert c.sha1_hex(data) == hashlib.sha1(data).hexdigest()


def test_murmur3_known_vector():
    # public murmur3_x86_32 test vector
    assert c.murmur3_32(b"", 0) == 0
    assert c.murmur3_32(b"hello", 0) == 0x248BFA47


def test_containers():
    assert c.containers_selftest()


def test_multidimension_vars():
    assert c.multidim_selftest()
    dump = b.core.var.describe("selftest_mdim")
    assert "echo" in dump and "7" in dump
