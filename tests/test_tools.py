"""Tools: rpc_press / rpc_dump + rpc_replay / rpc_view round trips."""
import os
import subprocess
import sys
import tempfile

import brpc_amd as b

r = b.core.rpc
u = b.core.util

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_recordio_roundtrip(tmp_path):
    path = str(tmp_path / "x.recordio")
    w = u.RecordWriter(path)
    assert w.ok()
    recs = [os.urandom(n) for n in (0, 1, 100, 10000)]
    for rec in recs:
        assert w.write(rec)
    w.flush()
    rd = u.RecordReader(path)
    got = []
    while True:
        rec = rd.next()
        if rec is None:
            break
        got.append(rec)
    assert got == recs


def test_rpc_dump_and_replay(tmp_path):
    port = r.start_echo_server(0)
    addr = f"127.0.0.1:{port}"
    dump_file = str(tmp_path / "dump.recordio")
    u.set_flag("rpc_dump_file", dump_file)
    u.set_flag("rpc_dump_ratio", "1")
    u.set_flag("rpc_dump", "true")
    before = u.rpc_dump_count()
    for i in range(5):
        rc, _, _ = r.echo_once(addr, b"dumpme%d" % i, 2000)
        assert rc == 0
    u.set_flag("rpc_dump", "false")
    assert u.rpc_dump_count() - before == 5
    # replay against the same server
    out = subprocess.run([sys.executable, "tools/rpc_replay.py", "--dump-file", dump_file,
                          "--addr", addr], cwd=REPO, capture_output=True, text=True,
                         timeout=60)
    assert out.returncode == 0, out.stderr
    assert "ok=5" in out.stdout


def test_rpc_press_max_mode():
    port = r.start_echo_server(0)
    out = subprocess.run([sys.executable, "tools/rpc_press.py", "--addr",
                          f"127.0.0.1:{port}", "--max", "--total", "2000",
                          "--concurrency", "8"],
                         cwd=REPO, capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr
    assert "errors=0" in out.stdout


def test_rpc_view():
    port = r.start_echo_server(0)
    out = subprocess.run([sys.executable, "tools/rpc_view.py", f"127.0.0.1:{port}",
                          "status"], cwd=REPO, capture_output=True, text=True, timeout=60)
    assert out.returncode == 0, out.stderr
    assert "brpc_amd" in out.stdout
