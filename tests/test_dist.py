"""Distributed bench-path test: 2 processes over gloo on CPU (the same
rendezvous/aggregation path the driver uses with RCCL on GPUs)."""
import json
import os
import socket
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_two_ranks_gloo():
    env = dict(os.environ)
    env["BAM_BENCH_BACKEND"] = "gloo"  # ranks > GPUs on a 1-GPU box must agree
    env["MASTER_ADDR"] = "127.0.0.1"
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node=2",
        "--master-addr", "127.0.0.1", "--master-port", str(port),
        "bench.py", "--gpus", "2", "--steps", "2", "--warmup", "1",
        "--calls-per-step", "500", "--concurrency", "8",
    ]
    out = subprocess.run(cmd, cwd=REPO, env=env, capture_output=True, text=True,
                         timeout=300)
    assert out.returncode == 0, out.stderr[-3000:] + out.stdout[-1000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    res = json.loads(line)
    assert res["metric"] == "echo_qps"
    assert res["n_gpus"] == 2
    assert res["value"] > 0
    assert res["scaling"] == "weak"


def test_bench_single_process():
    out = subprocess.run([sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
                          "--calls-per-step", "500"],
                         cwd=REPO, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    res = json.loads(line)
    assert res["value"] > 0
    assert res["config"]["p99_us"] > 0


def test_stream_p2p_two_ranks_gloo():
    """Config-3 harness on CPU: 2 ranks stream frames over gloo (the same
    script rides RCCL/xGMI on GPUs)."""
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node=2",
        "--master-addr", "127.0.0.1", "--master-port", str(port),
        "tools/stream_xgmi_bench.py", "--frames", "20", "--warmup", "2",
    ]
    env = dict(os.environ)
    env["BAM_BENCH_BACKEND"] = "gloo"  # 2 ranks on a 1-GPU box must agree
    out = subprocess.run(cmd, cwd=REPO, env=env, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-3000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    res = json.loads(line)
    assert res["value"] > 0


def test_bench_stream_mode_two_ranks_gloo():
    """bench.py --mode stream: config-3 headline through the framework's
    stream layer + CommGroup data plane (RCCL/xGMI on GPUs, tcp here)."""
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    env = dict(os.environ)
    env["BAM_BENCH_BACKEND"] = "gloo"
    env["MASTER_ADDR"] = "127.0.0.1"
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node=2",
        "--master-addr", "127.0.0.1", "--master-port", str(port),
        "bench.py", "--gpus", "2", "--steps", "2", "--warmup", "1",
        "--mode", "stream", "--frames-per-step", "8",
    ]
    out = subprocess.run(cmd, cwd=REPO, env=env, capture_output=True, text=True,
                         timeout=300)
    assert out.returncode == 0, out.stderr[-3000:]
    res = json.loads([l for l in out.stdout.splitlines() if l.startswith("{")][-1])
    assert res["metric"] == "stream_gbps"
    assert res["value"] > 0


def test_bench_fanout_mode_two_ranks_gloo():
    """bench.py --mode fanout: config-4 headline (collective fan-out)."""
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    env = dict(os.environ)
    env["BAM_BENCH_BACKEND"] = "gloo"
    env["MASTER_ADDR"] = "127.0.0.1"
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node=2",
        "--master-addr", "127.0.0.1", "--master-port", str(port),
        "bench.py", "--gpus", "2", "--steps", "2", "--warmup", "1",
        "--mode", "fanout", "--calls-per-step", "200",
    ]
    out = subprocess.run(cmd, cwd=REPO, env=env, capture_output=True, text=True,
                         timeout=300)
    assert out.returncode == 0, out.stderr[-3000:]
    res = json.loads([l for l in out.stdout.splitlines() if l.startswith("{")][-1])
    assert res["metric"] == "fanout_qps"
    assert res["value"] > 0
