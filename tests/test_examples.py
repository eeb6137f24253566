"""Every example must run clean (examples double as smoke benchmarks,
like the reference's example/ apps)."""
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

EXAMPLES = ["multi_threaded_echo.py", "parallel_echo.py", "streaming_echo.py",
            "tls_auth_echo.py", "rtmp_relay.py", "shm_ring_echo.py",
            "redis_server.py", "grpc_interop.py", "backup_request.py",
            "proxy_master.py"]


@pytest.mark.parametrize("name", EXAMPLES)
def test_example_runs(name):
    out = subprocess.run([sys.executable, f"examples/{name}"], cwd=REPO,
                         capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stderr[-1500:]
    assert out.stdout.strip()


def test_example_collective_fanout():
    import socket
    s = socket.socket(); s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1] + 500
    s.close()
    procs = [subprocess.Popen([sys.executable, "examples/collective_fanout.py",
                               str(r), "3", str(port)], cwd=REPO,
                              stdout=subprocess.PIPE, stderr=subprocess.PIPE,
                              text=True)
             for r in range(3)]
    outs = []
    for p in procs:
        out, err = p.communicate(timeout=120)
        assert p.returncode == 0, err[-1500:]
        outs.append(out)
    assert "data_ok=True" in outs[0]


def test_example_stream_xgmi():
    import socket
    s = socket.socket(); s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1] + 600
    s.close()
    procs = [subprocess.Popen([sys.executable, "examples/stream_xgmi.py",
                               str(r), str(port)], cwd=REPO,
                              stdout=subprocess.PIPE, stderr=subprocess.PIPE,
                              text=True)
             for r in range(2)]
    outs = []
    for p in procs:
        out, err = p.communicate(timeout=120)
        assert p.returncode == 0, err[-1500:]
        outs.append(out)
    assert "GB/s" in outs[1]
