"""Circuit breaker + health check tests (≙ reference circuit_breaker /
health_check unittests): repeated connect failures isolate an endpoint;
a background probe revives it once a server appears."""
import socket
import time

import brpc_amd as b

r = b.core.rpc


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def test_isolation_and_revival():
    port = free_port()
    addr = f"127.0.0.1:{port}"
    # 3 consecutive connection failures isolate the endpoint
    for _ in range(3):
        rc, _, _ = r.call_method_once(addr, "EchoService.Echo", b"x", 500, 0)
        assert rc != 0
    # now isolated: the call fails fast (EHOSTDOWN conducted as EFAILEDSOCKET)
    t0 = time.time()
    rc, _, err = r.call_method_once(addr, "EchoService.Echo", b"x", 2000, 0)
    assert rc != 0
    assert time.time() - t0 < 1.0, "isolated endpoint should fail fast"
    # bring a server up on that port; the health checker revives it
    got = r.start_echo_server(port)
    assert got == port
    deadline = time.time() + 5
    ok = False
    while time.time() < deadline:
        rc, resp, _ = r.call_method_once(addr, "EchoService.Echo", b"revive", 1000, 0)
        if rc == 0 and resp == b"revive":
            ok = True
            break
        time.sleep(0.3)
    assert ok, "endpoint was not revived by health check"


def test_server_death_fails_inflight_calls_fast():
    """Killing the server mid-flight must conduct failures to pending
    calls quickly (pending-session registry), not strand them to their
    full deadline."""
    import threading
    import time
    srv = b.Server()

    def slow(req, att):
        time.sleep(3)
        return req, b""

    srv.add_method("D", "Slow", slow)
    port = srv.start(0)
    ch = b.Channel("127.0.0.1:%d" % port, timeout_ms=20000, max_retry=0)
    errs = []

    def call():
        t0 = time.monotonic()
        try:
            ch.call("D.Slow", b"x")
            errs.append(("ok", time.monotonic() - t0))
        except b.RpcError as e:
            errs.append((e.args[0], time.monotonic() - t0))

    ts = [threading.Thread(target=call) for _ in range(4)]
    for t in ts:
        t.start()
    time.sleep(0.3)
    srv.stop()  # closes the listener; in-flight handlers keep running
    for t in ts:
        t.join()
    # handlers complete after ~3s (server object still alive) OR the calls
    # fail fast — either way nothing may run to the 20s deadline
    assert all(dur < 10 for _c, dur in errs), errs
