"""Fuzz-harness parity (SURVEY §4 ≙ reference test/fuzzing/): each protocol
parser has a libFuzzer target under tests/fuzz/ built by `make fuzz` with
ASan; this test builds them and runs each for a few seconds. Regression
memo: fuzz_snappy found the preamble malloc-bomb fixed in base/snappy.cc;
fuzz_mcpack found a parse_primitive stack overflow (type 0x3a);
fuzz_ts_flv found the FLV data_offset uint32-overflow OOB read."""
import os
import subprocess

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
TARGETS = ["fuzz_rpc_meta", "fuzz_http", "fuzz_redis", "fuzz_json", "fuzz_snappy",
           "fuzz_mcpack", "fuzz_cut_until", "fuzz_hpack", "fuzz_ts_flv", "fuzz_h2"]


@pytest.mark.slow
@pytest.mark.timeout(900)  # make fuzz cold build + 10 targets
def test_build_and_run_fuzzers():
    build = subprocess.run(["make", "-j16", "fuzz"], cwd=REPO, capture_output=True,
                           text=True, timeout=900)
    assert build.returncode == 0, build.stderr[-2000:]
    env = dict(os.environ)
    env["ASAN_OPTIONS"] = "detect_leaks=0"
    for t in TARGETS:
        exe = os.path.join(REPO, "build", "fuzz", "bin", t)
        assert os.path.exists(exe), t
        run = subprocess.run([exe, "-max_total_time=3", "-rss_limit_mb=2048"],
                             capture_output=True, text=True, timeout=120, env=env)
        assert run.returncode == 0, t + ": " + run.stderr[-2000:]
