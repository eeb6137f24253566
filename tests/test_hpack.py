"""In-tree HPACK (rpc/policy/hpack.*, ≙ reference details/hpack.cpp)
verified against libnghttp2 as an ORACLE via ctypes: every Huffman code,
header blocks both directions, dynamic-table behavior."""
import ctypes
import ctypes.util

import brpc_amd as b
import pytest

H = b.core.hpack


def load_nghttp2():
    try:
        return ctypes.CDLL("libnghttp2.so.14")
    except OSError:
        return None


NG = load_nghttp2()
needs_oracle = pytest.mark.skipif(NG is None, reason="libnghttp2 oracle unavailable")


class NV(ctypes.Structure):
    _fields_ = [("name", ctypes.c_char_p), ("value", ctypes.c_char_p),
                ("namelen", ctypes.c_size_t), ("valuelen", ctypes.c_size_t),
                ("flags", ctypes.c_uint8)]


class NVOut(ctypes.Structure):
    _fields_ = [("name", ctypes.POINTER(ctypes.c_uint8)),
                ("value", ctypes.POINTER(ctypes.c_uint8)),
                ("namelen", ctypes.c_size_t), ("valuelen", ctypes.c_size_t),
                ("flags", ctypes.c_uint8)]


def ng_deflate(headers, table_size=4096):
    d = ctypes.c_void_p()
    assert NG.nghttp2_hd_deflate_new(ctypes.byref(d), ctypes.c_size_t(table_size)) == 0
    nva = (NV * len(headers))()
    keep = []
    for i, (k, v) in enumerate(headers):
        keep.append((k, v))
        nva[i].name = k
        nva[i].value = v
        nva[i].namelen = len(k)
        nva[i].valuelen = len(v)
        nva[i].flags = 0
    buf = ctypes.create_string_buffer(1 << 16)
    NG.nghttp2_hd_deflate_hd.restype = ctypes.c_ssize_t
    n = NG.nghttp2_hd_deflate_hd(d, buf, ctypes.c_size_t(len(buf)), nva,
                                 ctypes.c_size_t(len(headers)))
    assert n >= 0, n
    NG.nghttp2_hd_deflate_del(d)
    return buf.raw[:n]


def ng_inflate_new():
    i = ctypes.c_void_p()
    assert NG.nghttp2_hd_inflate_new(ctypes.byref(i)) == 0
    return i


def ng_inflate(inflater, block):
    out = []
    NG.nghttp2_hd_inflate_hd2.restype = ctypes.c_ssize_t
    data = (ctypes.c_uint8 * len(block)).from_buffer_copy(block)
    pos = 0
    while pos < len(block):
        nv = NVOut()
        flags = ctypes.c_int(0)
        rv = NG.nghttp2_hd_inflate_hd2(
            inflater, ctypes.byref(nv), ctypes.byref(flags),
            ctypes.byref(data, pos), ctypes.c_size_t(len(block) - pos), 1)
        assert rv >= 0, rv
        pos += rv
        if flags.value & 0x02:  # NGHTTP2_HD_INFLATE_EMIT
            name = bytes(bytearray(nv.name[j] for j in range(nv.namelen)))
            val = bytes(bytearray(nv.value[j] for j in range(nv.valuelen)))
            out.append((name, val))
        if flags.value & 0x01:  # NGHTTP2_HD_INFLATE_FINAL
            break
        if rv == 0:
            break
    NG.nghttp2_hd_inflate_end_headers(inflater)
    return out


def hpack_literal_block(name, value_huff_bytes, value_orig_len):
    """Builds: literal-with-incremental-indexing, raw name, huffman value."""
    out = bytearray([0x40])
    # raw name
    out += H.encode_int(len(name), 7, 0x00)
    out += name
    out += H.encode_int(len(value_huff_bytes), 7, 0x80)
    out += value_huff_bytes
    return bytes(out)


@needs_oracle
def test_huffman_encode_table_canonical_all_symbols():
    """Our Huffman ENCODING of all 256 octets must decode exactly in
    nghttp2 — proves every entry of our code table is canonical."""
    value = bytes(range(256))
    huff = H.huffman_encode(value)
    block = hpack_literal_block(b"x-all", huff, len(value))
    inf = ng_inflate_new()
    headers = ng_inflate(inf, block)
    assert headers == [(b"x-all", value)]


def test_huffman_roundtrip_self():
    for payload in [b"", b"a", b"www.example.com", b"no-cache",
                    bytes(range(256)) * 3, b"\x00\xff" * 100]:
        assert H.huffman_decode(H.huffman_encode(payload)) == payload


@needs_oracle
def test_decode_nghttp2_deflated_blocks():
    """nghttp2's deflater output (indexed fields, huffman literals,
    dynamic-table references across blocks) decodes with our Decoder."""
    dec = H.Decoder(4096)
    d = ctypes.c_void_p()
    assert NG.nghttp2_hd_deflate_new(ctypes.byref(d), 4096) == 0
    NG.nghttp2_hd_deflate_hd.restype = ctypes.c_ssize_t
    blocks = [
        [(b":method", b"GET"), (b":path", b"/index.html"), (b"custom-key", b"custom-value")],
        [(b":method", b"GET"), (b"custom-key", b"custom-value"), (b"x-n", b"\x01\x02\xfe")],
        [(b":status", b"200"), (b"content-type", b"application/grpc")],
    ]
    for hs in blocks:
        nva = (NV * len(hs))()
        for i, (k, v) in enumerate(hs):
            nva[i].name = k; nva[i].value = v
            nva[i].namelen = len(k); nva[i].valuelen = len(v); nva[i].flags = 0
        buf = ctypes.create_string_buffer(1 << 16)
        n = NG.nghttp2_hd_deflate_hd(d, buf, len(buf), nva, len(hs))
        assert n > 0
        got = dec.decode(buf.raw[:n])
        assert got == hs
    NG.nghttp2_hd_deflate_del(d)


@needs_oracle
def test_our_encoder_inflates_in_nghttp2():
    enc = H.Encoder(4096)
    inf = ng_inflate_new()
    blocks = [
        [(b":method", b"POST"), (b":path", b"/Svc/Method"), (b"te", b"trailers")],
        [(b":method", b"POST"), (b":path", b"/Svc/Method"), (b"te", b"trailers")],
        [(b":status", b"200"), (b"grpc-status", b"0"), (b"x-bin", bytes(range(64)))],
    ]
    for hs in blocks:
        block = enc.encode([(k.decode("latin1"), v.decode("latin1")) for k, v in hs])
        got = ng_inflate(inf, block)
        assert got == hs
    # second identical block should be tiny (dynamic-table indexed)
    b1 = enc.encode([("x-repeated", "vvvv"), ("x-repeated2", "wwww")])
    b2 = enc.encode([("x-repeated", "vvvv"), ("x-repeated2", "wwww")])
    assert len(b2) <= 4  # two indexed fields


def test_integer_coding_rfc_examples():
    # RFC 7541 C.1: 10 with 5-bit prefix -> 0x0a; 1337 with 5-bit prefix
    assert H.encode_int(10, 5, 0) == b"\x0a"
    assert H.encode_int(1337, 5, 0) == b"\x1f\x9a\x0a"
    assert H.encode_int(42, 8, 0) == b"\x2a"


def test_hpack_roundtrip_self():
    enc = H.Encoder(4096)
    dec = H.Decoder(4096)
    for _ in range(3):
        hs = [("content-type", "text/html"), (":status", "404"),
              ("x-custom", "a" * 300), ("x-bin", "\x00\x01\x02")]
        block = enc.encode(hs)
        got = dec.decode(block)
        assert [(k.decode("latin1"), v.decode("latin1")) for k, v in got] == hs
