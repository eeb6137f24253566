"""Race/memory detection parity: the fiber runtime builds and
passes its stress scenarios under AddressSanitizer with fiber-switch
annotations (≙ reference ASan fiber support in bthread/stack_inl.h)."""
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.slow
@pytest.mark.timeout(900)  # cold ASan rebuild can exceed the global 180 s
def test_fiber_suite_under_asan(tmp_path):
    main = tmp_path / "asan_main.cc"
    main.write_text("""
#include <cstdio>
#include <cstdlib>
#include "fiber/fiber.h"
namespace bam { namespace selftest {
int64_t start_join_test(int, int);
bool urgent_test();
int64_t usleep_test(int64_t);
bool butex_wake_test();
bool butex_timeout_test();
int64_t mutex_test(int, int);
bool countdown_test(int);
bool fiber_key_test();
}}
using namespace bam::selftest;
#define CHECK_STAGE(n, cond) \\
  do { if (!(cond)) { fprintf(stderr, "stage %d failed: %s\\n", n, #cond); return n; } } while (0)
int main() {
  CHECK_STAGE(1, start_join_test(50, 500) == 25000);
  CHECK_STAGE(2, urgent_test());
  CHECK_STAGE(3, usleep_test(10000) >= 8000);
  CHECK_STAGE(4, butex_wake_test());
  CHECK_STAGE(5, butex_timeout_test());
  CHECK_STAGE(6, mutex_test(8, 500) == 4000);
  CHECK_STAGE(7, countdown_test(20));
  CHECK_STAGE(8, fiber_key_test());
  printf("asan fiber suite OK\\n");
  return 0;
}
""")
    exe = tmp_path / "asan_fibers"
    srcs = ["src/testsupport/fiber_selftest.cc"] + \
           [f"src/fiber/{f}" for f in os.listdir(f"{REPO}/src/fiber")
            if f.endswith((".cc", ".S"))]
    cmd = ["g++", "-O1", "-g", "-std=c++17", "-fsanitize=address",
           "-fno-omit-frame-pointer", "-pthread", f"-I{REPO}/src", str(main),
           *srcs, "src/base/logging.cc", "-o", str(exe)]
    build = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True, timeout=300)
    assert build.returncode == 0, build.stderr[-2000:]
    env = dict(os.environ)
    env["ASAN_OPTIONS"] = "detect_leaks=0"
    # One retry: the suite includes wall-clock-sensitive stages (usleep
    # accuracy) that can spuriously fail on a loaded CI box. A real ASan
    # report or logic bug fails deterministically on both attempts.
    for attempt in range(2):
        run = subprocess.run([str(exe)], capture_output=True, text=True, timeout=120, env=env)
        if run.returncode == 0:
            break
    assert run.returncode == 0, run.stdout[-500:] + run.stderr[-2000:]
    assert "OK" in run.stdout
