"""Mongo wire protocol, server side (reference policy/mongo_protocol.cpp +
mongo_service_adaptor.h): 16-byte LE header with op_code as magic; raw body
handed to ServerOptions::mongo_handler; OP_QUERY -> OP_REPLY envelope,
OP_MSG -> OP_MSG envelope, legacy writes are fire-and-forget. The same port
still answers baidu_std (protocol sniffing)."""
import socket
import struct

import brpc_amd as b
import pytest

r = b.core.rpc

OP_REPLY, OP_INSERT, OP_QUERY, OP_MSG = 1, 2002, 2004, 2013


@pytest.fixture(scope="module")
def port():
    return r.start_mongo_server()


def _mongo_packet(opcode, body, request_id=7):
    return struct.pack("<iiii", 16 + len(body), request_id, 0, opcode) + body


def _read_exact(s, n):
    out = b""
    while len(out) < n:
        chunk = s.recv(n - len(out))
        assert chunk, "connection closed early"
        out += chunk
    return out


def _parse_bson_ok_n(doc):
    (doclen,) = struct.unpack_from("<i", doc, 0)
    assert doclen == len(doc)
    ok = struct.unpack_from("<d", doc, 4 + 1 + 3)[0]  # 0x01 'ok\0' double
    n = struct.unpack_from("<i", doc, 4 + 12 + 1 + 2)[0]  # 0x10 'n\0' int32
    return ok, n


def test_op_query_gets_op_reply(port):
    s = socket.create_connection(("127.0.0.1", port), timeout=5)
    # OP_QUERY body: flags, cstring collection, skip, return, query doc
    body = struct.pack("<i", 0) + b"admin.$cmd\0" + struct.pack("<ii", 0, 1) + b"\x05\x00\x00\x00\x00"
    s.sendall(_mongo_packet(OP_QUERY, body, request_id=41))
    head = struct.unpack("<iiii", _read_exact(s, 16))
    assert head[3] == OP_REPLY
    assert head[2] == 41  # responseTo echoes our request id
    flags, cursor, start, nret = struct.unpack("<iqii", _read_exact(s, 20))
    assert nret == 1
    doc = _read_exact(s, head[0] - 36)
    ok, _ = _parse_bson_ok_n(doc)
    assert ok == 1.0
    s.close()


def test_op_msg_roundtrip_and_insert_counter(port):
    s = socket.create_connection(("127.0.0.1", port), timeout=5)
    # two fire-and-forget inserts (no reply)
    ins_body = struct.pack("<i", 0) + b"db.coll\0" + b"\x05\x00\x00\x00\x00"
    s.sendall(_mongo_packet(OP_INSERT, ins_body))
    s.sendall(_mongo_packet(OP_INSERT, ins_body))
    # then an OP_MSG; its reply reports the insert count
    msg_body = struct.pack("<i", 0) + b"\x00" + b"\x05\x00\x00\x00\x00"
    s.sendall(_mongo_packet(OP_MSG, msg_body, request_id=9))
    head = struct.unpack("<iiii", _read_exact(s, 16))
    assert head[3] == OP_MSG
    assert head[2] == 9
    _flags = _read_exact(s, 4)
    section = _read_exact(s, head[0] - 20)
    assert section[0] == 0  # kind-0 section
    ok, n = _parse_bson_ok_n(section[1:])
    assert ok == 1.0
    assert n >= 2
    s.close()


def test_std_rpc_still_works_on_mongo_port(port):
    rc, resp, err = r.protocol_call("127.0.0.1:%d" % port, "std",
                                    "EchoService.Echo", b"mixed")
    assert rc == 0, err
    assert resp == b"mixed"
