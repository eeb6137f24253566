"""Base-layer tests: IOBuf, crc32c, EndPoint, fast_rand.

Models the reference's test/iobuf_unittest.cpp + crc32c tests:
pure in-process unit tests against the C++ core through the bindings.
"""
import os
import random

import pytest

import brpc_amd as b


class TestIOBuf:
    def test_append_and_read(self):
        buf = b.IOBuf()
        assert buf.empty() and len(buf) == 0
        buf.append(b"hello ")
        buf.append(b"world")
        assert buf.to_bytes() == b"hello world"
        assert len(buf) == 11
        assert not buf.empty()
        assert buf.cpu_addressable()

    def test_large_append_spans_blocks(self):
        data = os.urandom(1 << 20)
        buf = b.IOBuf()
        buf.append(data)
        assert len(buf) == len(data)
        assert buf.backing_block_num() >= len(data) // 8192
        assert buf.to_bytes() == data

    def test_cutn_zero_copy(self):
        data = os.urandom(50000)
        buf = b.IOBuf()
        buf.append(data)
        out = b.IOBuf()
        moved = buf.cutn_to_iobuf(out, 12345)
        assert moved == 12345
        assert out.to_bytes() == data[:12345]
        assert buf.to_bytes() == data[12345:]

    def test_cutn_bytes(self):
        buf = b.IOBuf()
        buf.append(b"abcdefgh")
        assert buf.cutn(3) == b"abc"
        assert buf.to_bytes() == b"defgh"
        # cutting more than size returns what's there
        assert buf.cutn(100) == b"defgh"
        assert buf.empty()

    def test_pop_front_back(self):
        data = os.urandom(30000)
        buf = b.IOBuf()
        buf.append(data)
        assert buf.pop_front(100) == 100
        assert buf.pop_back(200) == 200
        assert buf.to_bytes() == data[100:-200]

    def test_copy_to_with_pos(self):
        data = os.urandom(100000)
        buf = b.IOBuf()
        buf.append(data)
        for _ in range(20):
            pos = random.randrange(0, len(data))
            n = random.randrange(0, len(data) - pos + 10)
            assert buf.copy_to(n, pos) == data[pos:pos + n]
        # copy does not consume
        assert len(buf) == len(data)

    def test_append_iobuf_shares_blocks(self):
        buf = b.IOBuf()
        buf.append(os.urandom(20000))
        buf2 = b.IOBuf()
        buf2.append_iobuf(buf)
        buf2.append_iobuf(buf)
        assert len(buf2) == 2 * len(buf)
        assert buf2.to_bytes() == buf.to_bytes() * 2

    def test_many_small_appends_merge_refs(self):
        buf = b.IOBuf()
        chunks = [bytes([i % 256]) * 7 for i in range(1000)]
        for c in chunks:
            buf.append(c)
        assert buf.to_bytes() == b"".join(chunks)
        # consecutive appends into the shared TLS block must merge refs
        assert buf.backing_block_num() <= len(buf) // 8192 + 2

    def test_clear_releases(self):
        buf = b.IOBuf()
        buf.append(os.urandom(100000))
        buf.clear()
        assert buf.empty() and len(buf) == 0


class TestCrc32c:
    def test_known_vectors(self):
        # standard CRC32-C test vectors
        assert b.crc32c(b"") == 0
        assert b.crc32c(b"123456789") == 0xE3069283
        assert b.crc32c(b"a") == 0xC1D04330

    def test_extend_matches_whole(self):
        data = os.urandom(10000)
        for split in (0, 1, 7, 8, 9, 4096, 9999, 10000):
            c = b.crc32c(data[split:], b.crc32c(data[:split]))
            assert c == b.crc32c(data)

    def test_combine(self):
        for la, lb in [(0, 5), (5, 0), (1, 1), (1000, 777), (65536, 3)]:
            x, y = os.urandom(la), os.urandom(lb)
            assert b.crc32c_combine(b.crc32c(x), b.crc32c(y), lb) == b.crc32c(x + y)


class TestEndPoint:
    def test_parse(self):
        ep = b.str2endpoint("127.0.0.1:8080")
        assert ep.port == 8080
        assert str(ep) == "127.0.0.1:8080"

    def test_bad(self):
        with pytest.raises(RuntimeError):
            b.str2endpoint("no-port-here")


def test_fast_rand():
    vals = {b.fast_rand() for _ in range(100)}
    assert len(vals) == 100


def test_iobuf_cut_until():
    """IOBuf::cut_until (≙ reference iobuf cut_until): delimiter consumed,
    body returned; KMP handles self-overlapping delimiters across blocks."""
    buf = b.IOBuf()
    buf.append(b"GET / HTTP/1.1\r\nHost: x\r\n\r\nBODY")
    line = buf.cut_until(b"\r\n")
    assert line == b"GET / HTTP/1.1"
    rest = buf.cut_until(b"\r\n\r\n")
    assert rest == b"Host: x"
    assert buf.cutn(100) == b"BODY"
    # self-overlap: "aab" inside "aaab"
    buf2 = b.IOBuf()
    buf2.append(b"aa")
    buf2.append(b"ab!")  # crosses block boundary
    assert buf2.cut_until(b"aab") == b"a"
    assert buf2.cutn(10) == b"!"
    # absent delimiter
    buf3 = b.IOBuf()
    buf3.append(b"xyz")
    assert buf3.cut_until(b"\r\n") is None
    assert buf3.cutn(10) == b"xyz"
