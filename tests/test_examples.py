"""Every example must run clean (examples double as smoke benchmarks,
like the reference's example/ apps)."""
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

EXAMPLES = ["multi_threaded_echo.py", "parallel_echo.py", "streaming_echo.py",
            "tls_auth_echo.py", "rtmp_relay.py", "shm_ring_echo.py",
            "redis_server.py", "grpc_interop.py", "backup_request.py"]


@pytest.mark.parametrize("name", EXAMPLES)
def test_example_runs(name):
    out = subprocess.run([sys.executable, f"examples/{name}"], cwd=REPO,
                         capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stderr[-1500:]
    assert out.stdout.strip()
