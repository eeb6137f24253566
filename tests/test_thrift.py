"""Thrift framed-transport protocol: server dispatch of TBinary-enveloped
calls + pipelined client, with hand-crafted TBinary payloads (the struct
layer passes through opaquely, like the reference's thrift adaptor)."""
import struct

import pytest

import brpc_amd as b


def tbinary_string_struct(field_id, s):
    """struct { <field_id>: string s }"""
    return (struct.pack(">bhI", 11, field_id, len(s)) + s + b"\x00")


@pytest.fixture(scope="module")
def thrift_port():
    srv = b.Server()

    # handler sees raw TBinary args struct; replies with a result struct
    def echo(req, att):
        # parse: expect field 1 string
        assert req[0:1] == b"\x0b"
        (ln,) = struct.unpack(">I", req[3:7])
        val = req[7:7 + ln]
        return tbinary_string_struct(0, val.upper())

    srv.add_method("thrift", "Echo", echo)
    port = srv.start(0)
    return port


def test_thrift_roundtrip(thrift_port):
    payload = tbinary_string_struct(1, b"hello thrift")
    resp = b.thrift_call(f"127.0.0.1:{thrift_port}", "Echo", payload, 3000)
    assert resp == tbinary_string_struct(0, b"HELLO THRIFT")


def test_thrift_unknown_method(thrift_port):
    with pytest.raises(b.RpcError):
        b.thrift_call(f"127.0.0.1:{thrift_port}", "NoSuch",
                      tbinary_string_struct(1, b"x"), 2000)


def test_thrift_raw_socket_client(thrift_port):
    """A plain socket speaking framed TBinary (i.e. any real thrift client)."""
    import socket
    args = tbinary_string_struct(1, b"raw")
    name = b"Echo"
    msg = struct.pack(">I", 0x80010001) + struct.pack(">I", len(name)) + name \
        + struct.pack(">I", 7) + args
    frame = struct.pack(">I", len(msg)) + msg
    s = socket.create_connection(("127.0.0.1", thrift_port), timeout=5)
    s.sendall(frame)
    data = b""
    while len(data) < 4:
        data += s.recv(4096)
    (flen,) = struct.unpack(">I", data[:4])
    while len(data) < 4 + flen:
        data += s.recv(4096)
    s.close()
    ver, = struct.unpack(">I", data[4:8])
    assert ver == 0x80010002  # REPLY
    nlen, = struct.unpack(">I", data[8:12])
    assert data[12:12 + nlen] == b"Echo"
    seqid, = struct.unpack(">I", data[12 + nlen:16 + nlen])
    assert seqid == 7
    assert data[16 + nlen:] == tbinary_string_struct(0, b"RAW")
