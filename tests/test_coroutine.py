"""C++20 coroutine adapter (reference brpc/coroutine.h Awaitable): built
as a separate -std=c++20 binary against the C++17 core objects; a chain
of two co_await'ed echoes must complete with correct payloads."""
import os
import subprocess

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.slow
def test_coroutine_rpc_chain(tmp_path):
    main = tmp_path / "co_main.cc"
    main.write_text("""
#include <atomic>
#include <cstdio>
#include <string>
#include <thread>

#include "rpc/channel.h"
#include "rpc/coroutine.h"

namespace bam { namespace rpctest { int start_echo_server(int); } }
using namespace bam;

std::atomic<int> g_done{0};
std::string g_out;

co::Task run(Channel* ch) {
  IOBuf req1;
  req1.append("first");
  co::RpcResult r1 = co_await co::AwaitRpc(ch, "EchoService.Echo", std::move(req1));
  if (r1.error_code != 0) { g_out = "err1:" + r1.error_text; g_done = 1; co_return; }
  IOBuf req2;
  req2.append(r1.response.to_string() + "+second");
  co::RpcResult r2 = co_await co::AwaitRpc(ch, "EchoService.Echo", std::move(req2));
  g_out = r2.error_code == 0 ? r2.response.to_string() : "err2";
  g_done = 1;
  co_return;
}

int main() {
  int port = rpctest::start_echo_server(0);
  Channel ch;
  ChannelOptions opts;
  opts.timeout_ms = 3000;
  if (ch.Init(("127.0.0.1:" + std::to_string(port)).c_str(), &opts) != 0) return 2;
  run(&ch);
  for (int i = 0; i < 500 && !g_done.load(); ++i)
    std::this_thread::sleep_for(std::chrono::milliseconds(10));
  if (!g_done.load()) { printf("timeout\\n"); return 3; }
  printf("result=%s\\n", g_out.c_str());
  return g_out == "first+second" ? 0 : 4;
}
""")
    exe = tmp_path / "co_rpc"
    core_objs = []
    for root, _dirs, files in os.walk(os.path.join(REPO, "build", "src")):
        for f in files:
            if f.endswith(".o") and "bindings" not in root:
                core_objs.append(os.path.join(root, f))
    if not core_objs:
        pytest.skip("build/ objects absent (e.g. gpurun snapshot excludes them); "
                    "the CPU CI machine builds them via make")
    cmd = ["g++", "-O1", "-std=c++20", "-fcoroutines", "-pthread", f"-I{REPO}/src",
           str(main), *core_objs, "-ldl", "-lz", "-lssl", "-lcrypto", "-o", str(exe)]
    build = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True, timeout=300)
    assert build.returncode == 0, build.stderr[-2000:]
    run_p = subprocess.run([str(exe)], capture_output=True, text=True, timeout=60)
    assert run_p.returncode == 0, run_p.stdout + run_p.stderr[-1000:]
    assert "result=first+second" in run_p.stdout
