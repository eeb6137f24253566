"""MySQL client protocol (reference policy/mysql/, clean-room subset):
packet framing, HandshakeV10 + HandshakeResponse41 with
mysql_native_password scramble, COM_QUERY (OK/ERR/resultset), COM_PING.
Validated against a scripted server that checks the scramble bytes."""
import hashlib
import socket
import struct
import threading

import brpc_amd as b

SALT = b"abcdefgh12345678abcd"  # 20 bytes: part1(8) + part2(12)
USER, PASSWORD = "alice", "sekrit"


def _scramble(password, salt):
    h1 = hashlib.sha1(password.encode()).digest()
    h2 = hashlib.sha1(h1).digest()
    h3 = hashlib.sha1(salt + h2).digest()
    return bytes(a ^ b for a, b in zip(h1, h3))


def _packet(payload, seq):
    return struct.pack("<I", len(payload))[:3] + bytes([seq]) + payload


def _lenc(n):
    if n < 0xFB:
        return bytes([n])
    return b"\xfc" + struct.pack("<H", n)


def _lenc_str(s):
    return _lenc(len(s)) + s


def _coldef(name):
    parts = b"".join(_lenc_str(x) for x in
                     [b"def", b"db", b"t", b"t", name, name])
    parts += b"\x0c" + struct.pack("<HIBHB", 33, 255, 0xfd, 0, 0) + b"\x00\x00"
    return parts


def _fake_mysql_server(results):
    srv = socket.socket()
    srv.bind(("127.0.0.1", 0))
    srv.listen(2)

    def handle(c):
        # HandshakeV10
        hs = bytes([10]) + b"8.0.99-fake\0" + struct.pack("<I", 1234)
        hs += SALT[:8] + b"\x00"
        hs += struct.pack("<H", 0xFFFF)        # capability low (all)
        hs += bytes([33]) + struct.pack("<H", 2) + struct.pack("<H", 0xFFFF >> 16)
        hs += bytes([21]) + b"\x00" * 10       # auth data len (8+12+1), reserved
        hs += SALT[8:20] + b"\x00"
        hs += b"mysql_native_password\x00"
        c.sendall(_packet(hs, 0))
        # HandshakeResponse41
        head = c.recv(4)
        ln = struct.unpack("<I", head[:3] + b"\x00")[0]
        resp = b""
        while len(resp) < ln:
            resp += c.recv(ln - len(resp))
        # parse username + auth
        p = 4 + 4 + 1 + 23
        z = resp.index(b"\x00", p)
        user = resp[p:z].decode()
        alen = resp[z + 1]
        auth = resp[z + 2:z + 2 + alen]
        if user != USER or auth != _scramble(PASSWORD, SALT):
            err = b"\xff" + struct.pack("<H", 1045) + b"#28000Access denied"
            c.sendall(_packet(err, 2))
            c.close()
            return
        c.sendall(_packet(b"\x00\x00\x00\x02\x00\x00\x00", 2))  # OK
        # command loop
        while True:
            head = c.recv(4)
            if len(head) < 4:
                return
            ln = struct.unpack("<I", head[:3] + b"\x00")[0]
            body = b""
            while len(body) < ln:
                chunk = c.recv(ln - len(body))
                if not chunk:
                    return
                body += chunk
            cmd = body[0]
            if cmd == 0x01:  # QUIT
                c.close()
                return
            if cmd == 0x0e:  # PING
                c.sendall(_packet(b"\x00\x00\x00\x02\x00\x00\x00", 1))
                continue
            if cmd == 0x16:  # STMT_PREPARE
                sql = body[1:].decode()
                nparams = sql.count("?")
                # PREPARE_OK: 00 stmt_id nr_cols nr_params 00 warnings
                ok = struct.pack("<BIHHBH", 0, 77, 1, nparams, 0, 0)
                c.sendall(_packet(ok, 1))
                seqn = 2
                for _ in range(nparams):
                    c.sendall(_packet(_coldef(b"?"), seqn)); seqn += 1
                if nparams:
                    c.sendall(_packet(b"\xfe\x00\x00\x02\x00", seqn)); seqn += 1
                c.sendall(_packet(_coldef(b"res"), seqn)); seqn += 1
                c.sendall(_packet(b"\xfe\x00\x00\x02\x00", seqn))
                continue
            if cmd == 0x17:  # STMT_EXECUTE: echo params joined as one row
                # parse: stmt_id u32, flags u8, iter u32, then bitmap+types+values
                nparams_guess = 2
                pos = 1 + 4 + 1 + 4
                nb = (nparams_guess + 7) // 8
                pos += nb + 1 + 2 * nparams_guess
                vals = []
                for _ in range(nparams_guess):
                    ln = body[pos]; pos += 1
                    vals.append(body[pos:pos + ln]); pos += ln
                joined = b"|".join(vals)
                seqn = 1
                c.sendall(_packet(_lenc(1), seqn)); seqn += 1
                c.sendall(_packet(_coldef(b"res"), seqn)); seqn += 1
                c.sendall(_packet(b"\xfe\x00\x00\x02\x00", seqn)); seqn += 1
                # binary row: 00 header, null bitmap (1 col -> 1 byte), lenc value
                row = b"\x00" + b"\x00" + _lenc_str(joined)
                c.sendall(_packet(row, seqn)); seqn += 1
                c.sendall(_packet(b"\xfe\x00\x00\x02\x00", seqn))
                continue
            if cmd == 0x19:  # STMT_CLOSE: no response
                continue
            if cmd == 0x03:  # QUERY
                sql = body[1:].decode()
                if sql.startswith("SELECT"):
                    cols, rows = results
                    seq = 1
                    c.sendall(_packet(_lenc(len(cols)), seq)); seq += 1
                    for name in cols:
                        c.sendall(_packet(_coldef(name), seq)); seq += 1
                    c.sendall(_packet(b"\xfe\x00\x00\x02\x00", seq)); seq += 1  # EOF
                    for row in rows:
                        payload = b"".join(b"\xfb" if v is None else _lenc_str(v)
                                           for v in row)
                        c.sendall(_packet(payload, seq)); seq += 1
                    c.sendall(_packet(b"\xfe\x00\x00\x02\x00", seq))
                elif sql.startswith("INSERT"):
                    ok = b"\x00" + _lenc(3) + _lenc(42) + struct.pack("<HH", 2, 0)
                    c.sendall(_packet(ok, 1))
                else:
                    err = b"\xff" + struct.pack("<H", 1064) + b"#42000You have an error"
                    c.sendall(_packet(err, 1))

    def accept_loop():
        while True:
            try:
                conn, _ = srv.accept()
            except OSError:
                return
            threading.Thread(target=handle, args=(conn,), daemon=True).start()

    threading.Thread(target=accept_loop, daemon=True).start()
    return srv, srv.getsockname()[1]


def test_mysql_connect_query_resultset():
    srv, port = _fake_mysql_server(([b"id", b"name"], [[b"1", b"bob"], [b"2", None]]))
    c = b.core.rpc.MysqlClient()
    assert c.connect("127.0.0.1", port, USER, PASSWORD) == 0
    assert c.server_version() == "8.0.99-fake"
    assert c.ping() == 0
    res = c.query("SELECT id, name FROM t")
    assert res.ok
    assert res.columns == ["id", "name"]
    assert res.rows == [["1", "bob"], ["2", ""]]
    c.close()
    srv.close()


def test_mysql_insert_and_error():
    srv, port = _fake_mysql_server(([b"x"], []))
    c = b.core.rpc.MysqlClient()
    assert c.connect("127.0.0.1", port, USER, PASSWORD) == 0
    res = c.query("INSERT INTO t VALUES (1)")
    assert res.ok
    assert res.affected_rows == 3
    assert res.last_insert_id == 42
    res = c.query("GARBAGE")
    assert res.error_code == 1064
    assert "error" in res.error_message
    srv.close()


def test_mysql_bad_password_rejected():
    srv, port = _fake_mysql_server(([b"x"], []))
    c = b.core.rpc.MysqlClient()
    rc = c.connect("127.0.0.1", port, USER, "wrong")
    assert rc == 1045
    assert not c.connected()
    srv.close()


def test_mysql_prepared_statements():
    srv, port = _fake_mysql_server(([b"x"], []))
    c = b.core.rpc.MysqlClient()
    assert c.connect("127.0.0.1", port, USER, PASSWORD) == 0
    sid, nparams = c.prepare("SELECT concat(?, ?)")
    assert sid == 77
    assert nparams == 2
    res = c.execute_prepared(sid, ["abc", "xyz"])
    assert res.ok, (res.error_code, res.error_message)
    assert res.columns == ["res"]
    assert res.rows == [["abc|xyz"]]
    c.close_statement(sid)
    srv.close()


def test_mysql_caching_sha2_fast_auth():
    """caching_sha2_password fast path: server advertises the plugin in
    HandshakeV10; client scramble = XOR(SHA256(pwd),
    SHA256(SHA256(SHA256(pwd)) + nonce)); server replies AuthMoreData
    0x03 (fast auth success) then OK."""
    import socket
    import threading

    def sha2_scramble(password, salt):
        h1 = hashlib.sha256(password.encode()).digest()
        h2 = hashlib.sha256(hashlib.sha256(h1).digest() + salt).digest()
        return bytes(a ^ b_ for a, b_ in zip(h1, h2))

    srv = socket.socket()
    srv.bind(("127.0.0.1", 0))
    srv.listen(1)
    result = {}

    def run():
        c, _ = srv.accept()
        hs = bytes([10]) + b"8.4.0-fake\0" + struct.pack("<I", 5)
        hs += SALT[:8] + b"\x00"
        hs += struct.pack("<H", 0xFFFF)
        hs += bytes([33]) + struct.pack("<H", 2) + struct.pack("<H", 0xFFFF >> 16)
        hs += bytes([21]) + b"\x00" * 10
        hs += SALT[8:20] + b"\x00"
        hs += b"caching_sha2_password\x00"
        c.sendall(_packet(hs, 0))
        head = c.recv(4)
        ln = struct.unpack("<I", head[:3] + b"\x00")[0]
        resp = b""
        while len(resp) < ln:
            resp += c.recv(ln - len(resp))
        p = 4 + 4 + 1 + 23
        z = resp.index(b"\x00", p)
        alen = resp[z + 1]
        auth = resp[z + 2:z + 2 + alen]
        result["auth_ok"] = auth == sha2_scramble(PASSWORD, SALT)
        c.sendall(_packet(b"\x01\x03", 2))                      # fast auth success
        c.sendall(_packet(b"\x00\x00\x00\x02\x00\x00\x00", 3))  # OK
        # stay open for COM_QUIT
        c.recv(64)
        c.close()

    threading.Thread(target=run, daemon=True).start()
    port = srv.getsockname()[1]
    c = b.core.rpc.MysqlClient()
    rc = c.connect("127.0.0.1", port, USER, PASSWORD)
    assert rc == 0
    assert result["auth_ok"], "caching_sha2 scramble mismatch"
    c.close()
    srv.close()


def test_mysql_caching_sha2_full_auth_rsa():
    """caching_sha2 FULL auth over plain TCP (round-1 gap): server demands
    full auth (0x01 0x04), client requests the RSA public key (0x02),
    server sends PEM, client sends RSA-OAEP(password||NUL XOR nonce).
    The mock decrypts with openssl pkeyutl and verifies."""
    import socket
    import subprocess
    import tempfile
    import threading
    import os

    tmp = tempfile.mkdtemp()
    priv = os.path.join(tmp, "k.pem")
    pub = os.path.join(tmp, "k.pub")
    subprocess.run(["openssl", "genrsa", "-out", priv, "2048"], check=True,
                   capture_output=True)
    subprocess.run(["openssl", "rsa", "-in", priv, "-pubout", "-out", pub],
                   check=True, capture_output=True)
    pem_pub = open(pub, "rb").read()

    srv = socket.socket()
    srv.bind(("127.0.0.1", 0))
    srv.listen(1)
    result = {}

    def read_pkt(c):
        head = c.recv(4)
        ln = struct.unpack("<I", head[:3] + b"\x00")[0]
        data = b""
        while len(data) < ln:
            data += c.recv(ln - len(data))
        return data

    def run():
        c, _ = srv.accept()
        hs = bytes([10]) + b"8.4.0-fake\0" + struct.pack("<I", 5)
        hs += SALT[:8] + b"\x00"
        hs += struct.pack("<H", 0xFFFF)
        hs += bytes([33]) + struct.pack("<H", 2) + struct.pack("<H", 0xFFFF >> 16)
        hs += bytes([21]) + b"\x00" * 10
        hs += SALT[8:20] + b"\x00"
        hs += b"caching_sha2_password\x00"
        c.sendall(_packet(hs, 0))
        read_pkt(c)                      # HandshakeResponse41
        c.sendall(_packet(b"\x01\x04", 2))  # full auth required
        req = read_pkt(c)
        result["asked_key"] = req == b"\x02"
        c.sendall(_packet(b"\x01" + pem_pub, 4))
        enc = read_pkt(c)
        with open(os.path.join(tmp, "enc.bin"), "wb") as f:
            f.write(enc)
        out = subprocess.run(
            ["openssl", "pkeyutl", "-decrypt", "-inkey", priv,
             "-in", os.path.join(tmp, "enc.bin"),
             "-pkeyopt", "rsa_padding_mode:oaep"],
            capture_output=True)
        plain = out.stdout
        expect = bytes((PASSWORD.encode() + b"\x00")[i] ^ SALT[i % len(SALT)]
                       for i in range(len(PASSWORD) + 1))
        result["decrypted_ok"] = plain == expect
        c.sendall(_packet(b"\x00\x00\x00\x02\x00\x00\x00", 6))  # OK
        c.recv(64)
        c.close()

    threading.Thread(target=run, daemon=True).start()
    port = srv.getsockname()[1]
    c = b.core.rpc.MysqlClient()
    rc = c.connect("127.0.0.1", port, USER, PASSWORD)
    assert rc == 0
    assert result["asked_key"]
    assert result["decrypted_ok"], "RSA-OAEP password blob mismatch"
    c.close()
    srv.close()
