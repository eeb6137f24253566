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


def test_redis_cluster_slot_of():
    """CRC16-CCITT slot mapping incl. hash tags (redis cluster spec
    values: 'foo'->12182, '123456789'->12739, '{user1000}.following' ==
    '{user1000}.followers')."""
    C = b.core.RedisClusterClient
    assert C.slot_of(b"foo") == 12182
    assert C.slot_of(b"123456789") == 12739
    assert C.slot_of(b"{user1000}.following") == C.slot_of(b"{user1000}.followers")
    assert C.slot_of(b"{user1000}.following") == C.slot_of(b"user1000")


def test_redis_cluster_routing_and_moved():
    """Slot routing from CLUSTER SLOTS + -MOVED redirect with remap +
    -ASK one-shot redirect, against two scripted RESP nodes."""
    storeA, storeB = {}, {}
    moved_mode = [False]

    def make_node(store, my_half, ports):
        srv = b.RedisServer()

        def cluster(args):
            lo = args[1].decode().lower() if len(args) > 1 else ""
            if lo != "slots":
                raise RuntimeError("unsupported")
            if not moved_mode[0]:
                return [[0, 8191, ["127.0.0.1", ports[0]]],
                        [8192, 16383, ["127.0.0.1", ports[1]]]]
            return [[0, 16383, ["127.0.0.1", ports[1]]]]

        def set_(args):
            k = args[1].decode()
            slot = b.core.RedisClusterClient.slot_of(args[1])
            owned = (slot < 8192) == (my_half == 0)
            if moved_mode[0] and my_half == 0:
                return "-MOVED %d 127.0.0.1:%d" % (slot, ports[1])
            if not owned and not moved_mode[0]:
                return "-MOVED %d 127.0.0.1:%d" % (slot, ports[1 - my_half])
            store[k] = args[2]
            return "OK"

        def get(args):
            k = args[1].decode()
            slot = b.core.RedisClusterClient.slot_of(args[1])
            if moved_mode[0] and my_half == 0:
                return "-MOVED %d 127.0.0.1:%d" % (slot, ports[1])
            return store.get(k)

        def asking(args):
            return "OK"

        srv.add_handler("cluster", cluster)
        srv.add_handler("set", set_)
        srv.add_handler("get", get)
        srv.add_handler("asking", asking)
        return srv

    ports = [0, 0]
    srvA = make_node(storeA, 0, ports)
    srvB = make_node(storeB, 1, ports)
    ports[0] = srvA.start(0)
    ports[1] = srvB.start(0)

    c = b.core.RedisClusterClient()
    assert c.init("127.0.0.1:%d" % ports[0]) == 0
    assert c.nslots_mapped() == 16384

    # keys landing on both halves route correctly
    keys = [b"foo", b"bar", b"{user1000}.x", b"k%d" % 7]
    for k in keys:
        assert c.command(["SET", k.decode(), "v-" + k.decode()]) == "OK"
        assert c.command(["GET", k.decode()]) == b"v-" + k
    assert storeA and storeB

    # rebalance: node A answers MOVED for everything; client follows and
    # remaps (subsequent calls go straight to B)
    moved_mode[0] = True
    kA = next(iter(storeA))
    storeB[kA] = b"after-move"
    assert c.command(["GET", kA]) == b"after-move"
