"""Naming services beyond list/file (reference policy/domain_naming_service
+ consul_naming_service): dns:// resolves every A record; consul:// pulls
Service.Address/Port from the health API — served here by a scripted
consul agent running on our own HTTP server."""
import json

import brpc_amd as b
import pytest

r = b.core.rpc


def test_dns_naming_resolves_localhost():
    port = 7777
    eps = r.resolve_naming("dns://localhost:%d" % port)
    assert eps and all(e.endswith(":%d" % port) for e in eps), eps


def test_consul_naming():
    # scripted consul agent: our own Server answering the health API
    echo_port = r.start_echo_server(0)
    agent = b.Server()

    def health(req, att):
        body = json.dumps([
            {"Service": {"Address": "127.0.0.1", "Port": echo_port}},
            {"Service": {"Address": "127.0.0.1", "Port": echo_port}},
        ]).encode()
        return body, b""

    agent.add_method("v1", "health/service/echo", health)  # path form below
    agent_port = agent.start(0)
    # resolution goes through /v1/health/service/echo?passing=1 — register
    # a restful mapping so the GET lands on the handler
    agent.add_restful_mapping("v1", "/v1/health/service/echo => health/service/echo")
    eps = b.core.rpc.resolve_naming("consul://127.0.0.1:%d/echo" % agent_port)
    assert eps == ["127.0.0.1:%d" % echo_port] * 2, eps
    agent.stop()


def test_dns_resolver_direct():
    eps = b.core.rpc.resolve_naming("dns://localhost:8123")
    assert "127.0.0.1:8123" in eps


def test_nacos_naming():
    """nacos:// (parity: reference policy/nacos_naming_service.cpp):
    instance list API against a scripted nacos on our own HTTP server;
    unhealthy/disabled instances filtered."""
    echo_port = r.start_echo_server(0)
    agent = b.Server()

    def instances(req, att):
        body = json.dumps({"hosts": [
            {"ip": "127.0.0.1", "port": echo_port, "healthy": True, "enabled": True},
            {"ip": "127.0.0.1", "port": 1, "healthy": False, "enabled": True},
            {"ip": "127.0.0.1", "port": 2, "healthy": True, "enabled": False},
            {"ip": "127.0.0.1", "port": echo_port + 0, "healthy": True, "enabled": True},
        ]}).encode()
        return body, b""

    agent.add_method("nacos", "instances", instances)
    agent.add_restful_mapping("nacos", "/nacos/v1/ns/instance/list => instances")
    agent_port = agent.start(0)
    eps = b.core.rpc.resolve_naming("nacos://127.0.0.1:%d/echo-svc" % agent_port)
    assert eps == ["127.0.0.1:%d" % echo_port] * 2, eps
    agent.stop()


def test_remotefile_naming():
    """remotefile:// (parity: reference remote_file_naming_service): a
    server-list file fetched over HTTP, same format as file://."""
    echo_port = r.start_echo_server(0)
    agent = b.Server()

    def listing(req, att):
        return ("127.0.0.1:%d\n# comment\n127.0.0.1:%d\n" %
                (echo_port, echo_port)).encode(), b""

    agent.add_method("files", "servers", listing)
    agent.add_restful_mapping("files", "/lists/servers.txt => servers")
    agent_port = agent.start(0)
    eps = b.core.rpc.resolve_naming(
        "remotefile://127.0.0.1:%d/lists/servers.txt" % agent_port)
    assert eps == ["127.0.0.1:%d" % echo_port] * 2, eps
    agent.stop()


def test_dlist_naming():
    """dlist:// (parity: reference DomainListNamingService,
    policy/list_naming_service.cpp:106): entries may be DNS names,
    resolved to (possibly several) A records on every refresh."""
    eps = b.core.rpc.resolve_naming("dlist://localhost:8000,127.0.0.2:9000")
    assert "127.0.0.1:8000" in eps, eps
    assert "127.0.0.2:9000" in eps, eps


def test_discovery_naming():
    """discovery:// (parity: reference policy/discovery_naming_service.cpp
    fetchs API): JSON data.<appid>.instances[].addrs with scheme prefixes
    stripped."""
    echo_port = r.start_echo_server(0)
    agent = b.Server()

    def fetchs(req, att):
        body = json.dumps({"code": 0, "data": {"my.app": {"instances": [
            {"addrs": ["grpc://127.0.0.1:%d" % echo_port,
                       "http://127.0.0.1:%d" % (echo_port + 1)]},
            {"addrs": ["127.0.0.1:%d" % echo_port]},
        ]}}}).encode()
        return body, b""

    agent.add_method("disc", "fetchs", fetchs)
    agent.add_restful_mapping("disc", "/discovery/fetchs => fetchs")
    agent_port = agent.start(0)
    eps = b.core.rpc.resolve_naming(
        "discovery://127.0.0.1:%d/my.app?env=prod&status=1" % agent_port)
    assert eps.count("127.0.0.1:%d" % echo_port) == 2, eps
    assert "127.0.0.1:%d" % (echo_port + 1) in eps, eps
    agent.stop()


def test_dlist_mixed_and_empty():
    """dlist:// resolves mixed literal+hostname entries; an unresolvable
    list fails instead of returning an empty server set."""
    eps = b.core.rpc.resolve_naming("dlist://127.0.0.3:7,localhost:9")
    assert "127.0.0.3:7" in eps and "127.0.0.1:9" in eps, eps
    # wholly-unresolvable list -> empty (the refresher keeps the last
    # good set; ResolveNamingUrl returns nonzero, binding maps to [])
    assert b.core.rpc.resolve_naming("dlist://no.such.host.invalid:1") == []
