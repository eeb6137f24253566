"""HTTP/1.1 protocol + builtin services tests.

Models reference brpc_builtin_service_unittest.cpp: hit the builtin pages
over a raw HTTP client against the same port that serves std-protocol RPC
(protocols multiplex on one listener, like the reference).
"""
import urllib.request

import pytest

import brpc_amd as b

r = b.core.rpc


@pytest.fixture(scope="module")
def base():
    port = r.start_echo_server(0)
    # prime an RPC so /status /rpcz have content
    rc, _, _ = r.echo_once(f"127.0.0.1:{port}", b"prime", 2000)
    assert rc == 0
    return f"http://127.0.0.1:{port}"


def get(url, timeout=5):
    with urllib.request.urlopen(url, timeout=timeout) as resp:
        return resp.status, resp.read().decode()


def test_health(base):
    st, body = get(base + "/health")
    assert st == 200 and body == "OK\n"


def test_index(base):
    st, body = get(base + "/index")
    assert st == 200 and "/vars" in body


def test_status_lists_methods(base):
    st, body = get(base + "/status")
    assert st == 200
    assert "EchoService" in body
    assert "processed_requests" in body


def test_vars(base):
    st, body = get(base + "/vars")
    assert st == 200
    assert "_count" in body or len(body) > 0


def test_flags_get_and_set(base):
    st, body = get(base + "/flags")
    assert st == 200 and "enable_rpcz" in body
    st, body = get(base + "/flags/rpcz_max_spans?setvalue=1024")
    assert st == 200
    st, body = get(base + "/flags/rpcz_max_spans")
    assert "1024" in body


def test_connections(base):
    st, body = get(base + "/connections")
    assert st == 200 and "socket_count" in body


def test_protobufs(base):
    st, body = get(base + "/protobufs")
    assert st == 200 and "EchoService.Echo" in body


def test_fibers_page(base):
    st, body = get(base + "/fibers")
    assert st == 200 and "workers" in body


def test_memory_page(base):
    st, body = get(base + "/memory")
    assert st == 200 and "iobuf_block_count" in body


def test_rpcz_records_spans(base):
    st, body = get(base + "/rpcz")
    assert st == 200
    assert "EchoService.Echo" in body  # both client+server spans recorded


def test_prometheus_metrics(base):
    st, body = get(base + "/brpc_metrics")
    assert st == 200
    assert "# TYPE" in body


def test_404(base):
    import urllib.error
    with pytest.raises(urllib.error.HTTPError) as ei:
        get(base + "/nope")
    assert ei.value.code == 404


def test_rpc_over_http(base):
    req = urllib.request.Request(base + "/EchoService/Echo", data=b"http payload",
                                 method="POST")
    with urllib.request.urlopen(req, timeout=5) as resp:
        assert resp.status == 200
        assert resp.read() == b"http payload"


def test_fiber_stack_tracer(base):
    """/fibers?st=1 captures suspended fibers' stacks (≙ bthread tracer)."""
    import brpc_amd as bb
    # park some fibers so there is something to trace
    import threading
    t = threading.Thread(target=lambda: bb.core.fiber.usleep_test(600_000))
    t.start()
    import time
    time.sleep(0.15)  # let the fiber park in fiber_usleep
    st, body = get(base + "/fibers?st=1")
    t.join()
    assert st == 200
    assert "fiber #" in body
    # the parked fiber's stack should include the butex/usleep path
    assert "butex" in body or "usleep" in body or "sched" in body, body[:2000]


def test_threads_page(base):
    code, body = get(base + "/threads")
    assert code == 200
    assert "Threads:" in body
    assert "fiber_workers:" in body


def test_hotspots_cpu_profile(base):
    import threading
    # burn CPU in the background so the PROF timer has something to sample
    stop = [False]

    def burn():
        x = 0
        while not stop[0]:
            x = (x * 1103515245 + 12345) & 0xFFFFFFFF

    ts = [threading.Thread(target=burn) for _ in range(2)]
    for t in ts:
        t.start()
    try:
        code, body = get(base + "/hotspots/cpu?seconds=1", timeout=15)
        assert code == 200
        assert "cpu profile:" in body
        assert "samples @" in body
    finally:
        stop[0] = True
        for t in ts:
            t.join()


def test_contention_page():
    """/hotspots/contention: sampled butex park sites with symbolized
    callsites (≙ reference contention profiler)."""
    import urllib.request
    port = r.start_echo_server(0)
    addr = "127.0.0.1:%d" % port
    for _ in range(300):
        r.echo_once(addr, b"x" * 64, 2000)
    body = urllib.request.urlopen(
        "http://127.0.0.1:%d/hotspots/contention" % port, timeout=10).read().decode()
    assert "contention profile" in body
    assert "sampled parks" in body


def test_gpu_stats_page():
    """/hotspots/gpu: HIP runtime telemetry + gpu_wait counters (shows
    'no GPU runtime' on CPU boxes, real counters on the MI355X)."""
    import urllib.request
    port = r.start_echo_server(0)
    body = urllib.request.urlopen(
        "http://127.0.0.1:%d/hotspots/gpu" % port, timeout=5).read().decode()
    assert "gpu_wait_parks" in body


def test_memory_page_has_malloc_stats():
    import urllib.request
    port = r.start_echo_server(0)
    body = urllib.request.urlopen(
        "http://127.0.0.1:%d/memory" % port, timeout=5).read().decode()
    assert "malloc_in_use_bytes" in body
    assert "iobuf_block_count" in body


def test_rpcz_persistent_spandb(tmp_path):
    """rpcz SpanDB (≙ reference leveldb-backed span store, brpc/span.cpp):
    sampled spans persist to a recordio file and query back at /rpcz?db=N
    across what would be a process restart (file survives)."""
    import urllib.request
    db = str(tmp_path / "spans.rio")
    assert b.core.util.set_flag("rpcz_db_path", db) == 0
    assert b.core.util.set_flag("rpcz_sample_mod", "1") == 0
    try:
        port = r.start_echo_server(0)
        addr = "127.0.0.1:%d" % port
        for i in range(50):
            r.echo_once(addr, b"z" * 32, 2000)
        body = urllib.request.urlopen(
            "http://127.0.0.1:%d/rpcz?db=20" % port, timeout=5).read().decode()
        assert "persisted_spans" in body
        assert "EchoService.Echo" in body
        # the file itself is recordio: readable independently
        lines = [l for l in body.splitlines() if "EchoService.Echo" in l]
        assert len(lines) >= 10
        # a FRESH process (≙ restart) can read the SpanDB file directly
        import subprocess, sys as _sys
        out = subprocess.run(
            [_sys.executable, "-c", (
                "import sys; sys.path.insert(0, %r)\n"
                "import brpc_amd as b\n"
                "rr = b.core.util.RecordReader(%r)\n"
                "n = 0\n"
                "while True:\n"
                "    rec = rr.next()\n"
                "    if rec is None: break\n"
                "    assert b'EchoService.Echo' in rec\n"
                "    n += 1\n"
                "print('records', n)") % (
                    __import__('os').path.dirname(
                        __import__('os').path.dirname(
                            __import__('os').path.abspath(__file__))), db)],
            capture_output=True, text=True, timeout=120)
        assert out.returncode == 0, out.stderr[-800:]
        assert int(out.stdout.split()[-1]) >= 10, out.stdout
    finally:
        b.core.util.set_flag("rpcz_db_path", "")
        b.core.util.set_flag("rpcz_sample_mod", "16")


def test_trace_id_chains_through_nested_calls():
    """Trace propagation (≙ reference brpc/span.h: trace_id/span_id/
    parent_span_id ride RpcRequestMeta fields 4-6; TLS parent chaining
    span.h:153): client -> A.Relay -> nested Echo on B. The nested client
    span and B's server span must share A's trace_id, and the nested
    call's parent_span_id must be A's inbound span id."""
    import re
    import urllib.request
    assert b.core.util.set_flag("rpcz_sample_mod", "1") == 0
    try:
        port_b = r.start_echo_server(0)
        port_a = r.start_echo_server(0)
        payload = b"trace-me-7391"
        rc, resp, _ = r.call_method_once("127.0.0.1:%d" % port_a,
                                         "EchoService.Relay",
                                         ("127.0.0.1:%d|" % port_b).encode() + payload,
                                         3000, 0)
        assert rc == 0 and resp == payload
        body = urllib.request.urlopen(
            "http://127.0.0.1:%d/rpcz?verbose" % port_a, timeout=5).read().decode()
        # rows: "... | trace=<hex> span=<hex> [parent=<hex>]"
        rows = [l for l in body.splitlines() if "trace=" in l]
        relay_s = [l for l in rows if "EchoService.Relay" in l and "| S |" in l]
        echo_c = [l for l in rows if "EchoService.Echo" in l and "| C |" in l]
        echo_s = [l for l in rows if "EchoService.Echo" in l and "| S |" in l]
        assert relay_s and echo_c and echo_s

        def ids(line):
            m = re.search(r"trace=([0-9a-f]+) span=([0-9a-f]+)(?: parent=([0-9a-f]+))?",
                          line)
            return m.group(1), m.group(2), m.group(3)
        rt, rs, _ = ids(relay_s[-1])
        ct, cs, cp = ids(echo_c[-1])
        st, ss, sp = ids(echo_s[-1])
        assert ct == rt == st, (rt, ct, st)
        assert cp == rs, "nested client parent %s != relay server span %s" % (cp, rs)
        assert ss == cs and sp == cp  # B's server sees the nested call's ids
    finally:
        b.core.util.set_flag("rpcz_sample_mod", "16")


def test_list_sockets_dir_pages(base):
    import json as _json
    st, body = get(base + "/list")
    assert st == 200
    services = _json.loads(body)
    assert any(s["service"] == "EchoService" and "Echo" in s["methods"]
               for s in services)
    st, body = get(base + "/sockets")
    assert st == 200 and "socket_count" in body
    # per-id detail: take an id from /connections
    conn_line = [l for l in body.splitlines() if " | " in l and l[0].isdigit()]
    if conn_line:
        sid = conn_line[0].split(" | ")[0]
        st, det = get(base + "/sockets?id=%s" % sid)
        assert st == 200 and "remote:" in det and "in_bytes:" in det
    st, body = get(base + "/dir?path=/root/repo/tests")
    assert st == 200 and "test_http_builtin.py" in body


def test_master_handler_catch_all():
    """Generic/proxy pass-through (≙ reference BaiduMasterService,
    baidu_master_service.h:36): unknown service/method lands in the
    catch-all with the original names and raw request."""
    port = r.start_master_echo_server()
    rc, resp, err = r.call_method_once("127.0.0.1:%d" % port,
                                       "NoSuch.Service", b"raw-bytes", 3000, 0)
    assert rc == 0, err
    assert resp == b"master:NoSuch.Service:raw-bytes"


def test_ids_page(base):
    st, body = get(base + "/ids")
    assert st == 200 and "sessions_created:" in body
    created = int([l for l in body.splitlines()
                   if l.startswith("sessions_created:")][0].split(":")[1])
    assert created > 0  # the priming RPC used a session


def test_pprof_endpoints(base):
    """Remote pprof attach endpoints (≙ reference builtin/pprof_service.cpp):
    /pprof/profile returns the legacy gperftools binary format (8-byte LE
    words: [0,3,0,period,0] ... [0,1,0]); /pprof/symbol symbolizes
    POSTed addresses; /pprof/cmdline is raw /proc/self/cmdline."""
    import struct as _struct
    with urllib.request.urlopen(base + "/pprof/profile?seconds=1",
                                timeout=30) as resp:
        prof = resp.read()
    assert len(prof) >= 64 and len(prof) % 8 == 0
    words = _struct.unpack("<%dQ" % (len(prof) // 8), prof)
    assert words[0:3] == (0, 3, 0) and words[4] == 0
    assert words[-3:] == (0, 1, 0)
    # symbol: GET advertises support; POST symbolizes an address
    st, body = get(base + "/pprof/symbol")
    assert "num_symbols: 1" in body
    addr = None
    i = 5
    while i + 1 < len(words) - 3:
        count, depth = words[i], words[i + 1]
        if count == 1 and depth > 0:
            addr = words[i + 2]
            break
        i += 2 + depth
    if addr:
        req = urllib.request.Request(base + "/pprof/symbol",
                                     data=("0x%x" % addr).encode(),
                                     method="POST")
        with urllib.request.urlopen(req, timeout=10) as resp:
            sym = resp.read().decode()
        assert ("0x%x" % addr) in sym and "\t" in sym
    st, body = get(base + "/pprof/cmdline")
    assert st == 200 and "python" in body
    st, body = get(base + "/pprof/heap")
    assert st == 200 and body.startswith("heap profile:")
    assert "MAPPED_LIBRARIES:" in body
