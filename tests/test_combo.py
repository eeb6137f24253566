"""Combo channels (Parallel/Selective/Partition) + load balancers.

Models reference test/brpc_parallel_channel_unittest.cpp and LB unittests:
multiple in-process servers on distinct loopback ports.
"""
import pytest

import brpc_amd as b

r = b.core.rpc
c = b.core.combo


@pytest.fixture(scope="module")
def three_ports():
    return [r.start_echo_server(0) for _ in range(3)]


def test_parallel_fanout_merges_in_order(three_ports):
    rc, merged, err = c.parallel_echo(three_ports, b"PAY", -1)
    assert rc == 0, err
    assert merged == b"PAY" * 3


def test_parallel_with_dead_sub_fails(three_ports):
    rc, merged, err = c.parallel_echo(three_ports + [1], b"x", -1)  # port 1: dead
    assert rc != 0


def test_parallel_fail_limit_tolerates(three_ports):
    rc, merged, err = c.parallel_echo(three_ports + [1], b"Q", 1)
    assert rc == 0, err
    assert merged == b"Q" * 3


def test_selective_failover(three_ports):
    rc, resp = c.selective(1, three_ports[0])
    assert rc == 0
    assert resp == b"sel"


def test_partition_channel(three_ports):
    rc, merged = c.partition(three_ports)
    assert rc == 0
    assert merged == b"P" * 3


@pytest.mark.parametrize("lb", ["rr", "random", "p2c", "la", "wrr", "c_hash"])
def test_lb_spreads_load(three_ports, lb):
    n = c.lb_spread(lb, three_ports, 60)
    assert n > 0, f"lb {lb} failed: {n}"
    if lb in ("rr", "random", "wrr", "c_hash"):
        assert n == 3  # all servers hit
    else:
        assert n >= 1  # la/p2c may legitimately prefer one fast server
