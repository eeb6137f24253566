"""esp client protocol (reference policy/esp_protocol.cpp + esp_head.h):
32-byte packed LE head {from u64, to u64, msg u32, msg_id u64, body_len
i32} + raw body; no magic (parse gated to esp-created sockets); FIFO
correlation. Tested against a scripted socket server speaking the format."""
import socket
import struct
import threading

import brpc_amd as b

r = b.core.rpc

HEAD = struct.Struct("<QQIQi")


def _fake_esp_server():
    srv = socket.socket()
    srv.bind(("127.0.0.1", 0))
    srv.listen(4)

    def run():
        while True:
            try:
                c, _ = srv.accept()
            except OSError:
                return
            def handle(c):
                buf = b""
                while True:
                    try:
                        chunk = c.recv(65536)
                    except OSError:
                        return
                    if not chunk:
                        return
                    buf += chunk
                    while len(buf) >= HEAD.size:
                        frm, to, msg, msg_id, blen = HEAD.unpack_from(buf)
                        if len(buf) < HEAD.size + blen:
                            break
                        body = buf[HEAD.size:HEAD.size + blen]
                        buf = buf[HEAD.size + blen:]
                        reply = b"esp:%d:" % msg + body
                        c.sendall(HEAD.pack(to, frm, msg, msg_id, len(reply)) + reply)
            threading.Thread(target=handle, args=(c,), daemon=True).start()

    threading.Thread(target=run, daemon=True).start()
    return srv, srv.getsockname()[1]


def test_esp_call_roundtrip():
    srv, port = _fake_esp_server()
    rc, resp, err = r.protocol_call("127.0.0.1:%d" % port, "esp", "42", b"payload")
    assert rc == 0, err
    assert resp == b"esp:42:payload"
    srv.close()


def test_esp_sequential_calls_fifo():
    srv, port = _fake_esp_server()
    for i in range(10):
        rc, resp, err = r.protocol_call("127.0.0.1:%d" % port, "esp", "7", b"n%d" % i)
        assert rc == 0, err
        assert resp == b"esp:7:n%d" % i
    srv.close()
