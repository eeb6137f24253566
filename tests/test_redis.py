"""Redis (RESP) protocol tests — server-side command dispatch (parity:
reference RedisService) checked with a RAW python socket speaking real
RESP, plus the pipelined client through Channel."""
import socket

import pytest

import brpc_amd as b


@pytest.fixture(scope="module")
def redis_server():
    store = {}
    srv = b.RedisServer()

    def cmd_set(args):
        if len(args) != 3:
            raise ValueError("wrong number of arguments for 'set'")
        store[args[1]] = args[2]
        return "OK"

    def cmd_get(args):
        return store.get(args[1])  # None -> nil

    def cmd_incr(args):
        v = int(store.get(args[1], b"0")) + 1
        store[args[1]] = str(v).encode()
        return v

    def cmd_keys(args):
        return sorted(store.keys())

    srv.add_handler("SET", cmd_set)
    srv.add_handler("GET", cmd_get)
    srv.add_handler("INCR", cmd_incr)
    srv.add_handler("KEYS", cmd_keys)
    port = srv.start(0)
    return port


def raw_cmd(port, *args):
    out = b"*%d\r\n" % len(args)
    for a in args:
        out += b"$%d\r\n%s\r\n" % (len(a), a)
    s = socket.create_connection(("127.0.0.1", port), timeout=5)
    s.sendall(out)
    data = b""
    s.settimeout(5)
    while not data.endswith(b"\r\n"):
        chunk = s.recv(4096)
        if not chunk:
            break
        data += chunk
    s.close()
    return data


def test_raw_set_get(redis_server):
    port = redis_server
    assert raw_cmd(port, b"SET", b"k", b"v") == b"+OK\r\n"
    assert raw_cmd(port, b"GET", b"k") == b"$1\r\nv\r\n"
    assert raw_cmd(port, b"GET", b"missing") == b"$-1\r\n"


def test_raw_unknown_command(redis_server):
    port = redis_server
    assert raw_cmd(port, b"NOPE").startswith(b"-ERR unknown command")


def test_client_channel(redis_server):
    port = redis_server
    addr = f"127.0.0.1:{port}"
    assert b.redis_call(addr, ["SET", "x", "42"]) == "OK"
    assert b.redis_call(addr, ["GET", "x"]) == b"42"
    assert b.redis_call(addr, ["INCR", "x"]) == 43
    assert b.redis_call(addr, ["GET", "x"]) == b"43"


def test_client_array_reply(redis_server):
    port = redis_server
    addr = f"127.0.0.1:{port}"
    b.redis_call(addr, ["SET", "a", "1"])
    b.redis_call(addr, ["SET", "b", "2"])
    keys = b.redis_call(addr, ["KEYS", "*"])
    assert b"a" in keys and b"b" in keys


def test_client_error_reply(redis_server):
    port = redis_server
    addr = f"127.0.0.1:{port}"
    with pytest.raises(b.RpcError):
        b.redis_call(addr, ["SET", "only-key"])


def test_pipelined_commands(redis_server):
    """Multiple commands on one connection answered in order."""
    port = redis_server
    out = b""
    for i in range(5):
        k = b"p%d" % i
        out += b"*3\r\n$3\r\nSET\r\n$%d\r\n%s\r\n$1\r\n%d\r\n" % (len(k), k, i)
    s = socket.create_connection(("127.0.0.1", port), timeout=5)
    s.sendall(out)
    data = b""
    while data.count(b"+OK\r\n") < 5:
        data += s.recv(4096)
    s.close()
    assert data == b"+OK\r\n" * 5
