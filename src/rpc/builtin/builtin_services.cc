// brpc_amd: builtin HTTP services.
// Parity: reference brpc/builtin/ (27 services; registration at
// server.cpp:501-603). This file serves the core set: /index, /status,
// /vars, /flags, /health, /version, /connections, /protobufs, /fibers
// (≙ /bthreads), /memory, /brpc_metrics (Prometheus), /rpcz (spans).
#include <string.h>
#include <dirent.h>
#include <errno.h>

#include <algorithm>
#include <vector>

#include <sstream>

#include "base/flags.h"
#include "base/gpu_loader.h"
#include "base/iobuf.h"
#include "base/time.h"
#include "fiber/fiber.h"
#include "fiber/session.h"
#include <malloc.h>

#include "rpc/policy/http_protocol.h"
#include "fiber/gpu_wait.h"
#include "rpc/rpcz.h"
#include "rpc/server.h"
#include "rpc/socket.h"
#include "var/variable.h"

namespace bam {
std::string dump_fiber_stacks(int max_fibers);  // fiber/tracer.cc
std::string CpuProfile(int seconds, int hz);    // rpc/cpu_profiler.cc
std::string CpuProfileBinary(int seconds, int hz);
std::string SymbolizeAddresses(const std::string& body);
std::string ContentionProfile();                // rpc/cpu_profiler.cc
namespace policy {

namespace {

int64_t g_start_time_us = monotonic_time_us();

void page_index(HttpResponse* resp) {
  // Dashboard (≙ reference /index with embedded flot plots; ours is a
  // dependency-free canvas sparkline polling /vars once a second).
  resp->content_type = "text/html";
  std::ostringstream os;
  os << "<html><head><title>brpc_amd</title><style>"
        "body{font-family:monospace;margin:24px}"
        "ul{columns:3;max-width:720px}canvas{border:1px solid #ccc}"
        "#live{margin:12px 0;padding:8px;background:#f6f6f6}"
        "</style></head><body><h1>brpc_amd server</h1>"
        "<div id=live>loading /vars…</div>"
        "<canvas id=plot width=600 height=120></canvas>"
        "<div>requests/s (60 s window, 1 Hz poll of /vars)</div><ul>";
  const char* pages[] = {"status", "vars",   "flags",  "health",       "version",
                         "connections", "sockets", "list", "dir", "ids", "pprof/profile", "pprof/symbol", "protobufs", "fibers", "memory", "threads",
                         "hotspots/cpu", "hotspots/contention", "hotspots/gpu",
                         "rpcz", "brpc_metrics"};
  for (const char* p : pages) os << "<li><a href=\"/" << p << "\">/" << p << "</a></li>";
  os << "</ul><script>\n"
        "const hist=[];\n"
        "async function tick(){\n"
        "  try{\n"
        "    const t=await (await fetch('/vars')).text();\n"
        "    const kv={};\n"
        "    for(const line of t.split('\\n')){\n"
        "      const i=line.indexOf(':');\n"
        "      if(i>0)kv[line.slice(0,i).trim()]=line.slice(i+1).trim();\n"
        "    }\n"
        "    const qps=[];\n"
        "    for(const k in kv)if(k.endsWith('_qps'))qps.push(k+'='+kv[k]);\n"
        "    document.getElementById('live').textContent=\n"
        "      qps.length?qps.join('  '):Object.keys(kv).length+' vars';\n"
        "    let total=0;\n"
        "    for(const k in kv)if(k.endsWith('_qps'))total+=parseFloat(kv[k])||0;\n"
        "    hist.push(total);if(hist.length>60)hist.shift();\n"
        "    const c=document.getElementById('plot'),g=c.getContext('2d');\n"
        "    g.clearRect(0,0,c.width,c.height);\n"
        "    const max=Math.max(1,...hist);\n"
        "    g.beginPath();\n"
        "    hist.forEach((v,i)=>{const x=i*c.width/60,y=c.height-4-v/max*(c.height-8);\n"
        "      i?g.lineTo(x,y):g.moveTo(x,y);});\n"
        "    g.strokeStyle='#06c';g.stroke();\n"
        "    g.fillText(max.toFixed(0)+' qps peak',6,12);\n"
        "  }catch(e){}\n"
        "  setTimeout(tick,1000);\n"
        "}\n"
        "tick();\n"
        "</script></body></html>";
  resp->body.append(os.str());
}

void page_status(Server* server, HttpResponse* resp) {
  std::ostringstream os;
  os << "version: brpc_amd/1.0 (MI355X-native)\n";
  os << "uptime_s: " << (monotonic_time_us() - g_start_time_us) / 1000000 << "\n";
  os << "fiber_workers: " << fiber_get_concurrency() << "\n";
  os << "fibers_created: " << fiber_count_created() << "\n";
  os << "fibers_active: " << fiber_count_active() << "\n";
  os << "gpu_devices: " << gpu::device_count() << "\n";
  if (server != nullptr) {
    os << "listen: " << endpoint2str(server->listen_address()) << "\n";
    os << "processed_requests: " << server->nprocessed.load() << "\n";
    os << "concurrency: " << server->concurrency.load() << "\n";
    for (const auto& kv : server->services()) {
      os << "service: " << kv.first << "\n";
      for (const auto& m : kv.second->methods()) {
        var::LatencyRecorder* rec = server->method_status(kv.first, m.first);
        os << "  " << m.first << "  count=" << rec->count() << " qps=" << rec->qps()
           << " latency_avg_us=" << rec->latency_avg()
           << " p99_us=" << rec->latency_percentile(0.99) << "\n";
      }
    }
  }
  resp->body.append(os.str());
}

void page_vars(const HttpRequest& req, HttpResponse* resp) {
  std::string filter;
  auto it = req.query.find("filter");
  if (it != req.query.end()) filter = it->second;
  // /vars/some_prefix form
  if (req.path.size() > 6) filter = req.path.substr(6);
  std::ostringstream os;
  var::Variable::dump_exposed(os, filter);
  resp->body.append(os.str());
}

void page_flags(const HttpRequest& req, HttpResponse* resp) {
  // /flags/NAME?setvalue=V modifies (parity: reloadable gflags at /flags)
  if (req.path.size() > 7) {
    std::string name = req.path.substr(7);
    auto it = req.query.find("setvalue");
    if (it != req.query.end()) {
      int rc = flags::SetFlagValue(name, it->second);
      if (rc == 0) {
        resp->body.append("set " + name + "=" + it->second + "\n");
      } else {
        resp->status = rc == -1 ? 404 : 403;
        resp->body.append(rc == -1 ? "unknown flag\n" : "validation failed\n");
      }
      return;
    }
    resp->body.append(name + " : " + flags::GetFlagValue(name) + "\n");
    return;
  }
  std::vector<flags::FlagInfo> all;
  flags::ListFlags(&all);
  std::ostringstream os;
  for (const auto& f : all) {
    os << f.name << " : " << flags::GetFlagValue(f.name) << "  (default: " << f.default_value
       << ")  # " << f.description << "\n";
  }
  resp->body.append(os.str());
}

void page_connections(HttpResponse* resp) {
  std::vector<SocketId> ids;
  ListSockets(&ids);
  std::ostringstream os;
  os << "socket_count: " << ids.size() << "\n";
  os << "id | remote | in_bytes | out_bytes | in_msgs | out_msgs\n";
  for (SocketId id : ids) {
    SocketUniquePtr s;
    if (Socket::Address(id, &s) != 0) continue;
    os << id << " | " << endpoint2str(s->remote_side()) << " | " << s->in_bytes.load()
       << " | " << s->out_bytes.load() << " | " << s->in_messages.load() << " | "
       << s->out_messages.load() << "\n";
  }
  resp->body.append(os.str());
}

void page_list(Server* server, HttpResponse* resp) {
  // Machine-readable service list (≙ reference builtin/list_service.cpp,
  // which returns a proto; ours is JSON).
  std::ostringstream os;
  os << "[";
  bool first_s = true;
  if (server != nullptr) {
    for (const auto& kv : server->services()) {
      if (!first_s) os << ",";
      first_s = false;
      os << "{\"service\":\"" << kv.first << "\",\"methods\":[";
      bool first_m = true;
      for (const auto& m : kv.second->methods()) {
        if (!first_m) os << ",";
        first_m = false;
        os << "\"" << m.first << "\"";
      }
      os << "]}";
    }
  }
  os << "]\n";
  resp->headers["Content-Type"] = "application/json";
  resp->body.append(os.str());
}

void page_sockets(const HttpRequest& req, HttpResponse* resp) {
  // Per-socket detail (≙ reference builtin/sockets_service.cpp: one
  // Socket's DebugString by id; summary without one).
  auto it = req.query.find("id");
  if (it == req.query.end()) {
    page_connections(resp);
    return;
  }
  SocketId id = (SocketId)strtoull(it->second.c_str(), nullptr, 10);
  SocketUniquePtr s;
  if (Socket::Address(id, &s) != 0) {
    resp->status = 404;
    resp->body.append("no such socket (recycled or never existed)\n");
    return;
  }
  std::ostringstream os;
  os << "socket_id: " << id << "\n"
     << "fd: " << s->fd() << "\n"
     << "remote: " << endpoint2str(s->remote_side()) << "\n"
     << "local: " << endpoint2str(s->local_side()) << "\n"
     << "in_bytes: " << s->in_bytes.load() << "\n"
     << "out_bytes: " << s->out_bytes.load() << "\n"
     << "in_messages: " << s->in_messages.load() << "\n"
     << "out_messages: " << s->out_messages.load() << "\n"
     << "failed: " << (s->Failed() ? 1 : 0) << "\n";
  resp->body.append(os.str());
}

void page_dir(const HttpRequest& req, HttpResponse* resp) {
  // Directory listing (≙ reference builtin/dir_service.cpp).
  std::string path = "/";
  auto it = req.query.find("path");
  if (it != req.query.end() && !it->second.empty()) path = it->second;
  DIR* d = opendir(path.c_str());
  if (d == nullptr) {
    resp->status = 404;
    resp->body.append("cannot open " + path + ": " + strerror(errno) + "\n");
    return;
  }
  std::vector<std::string> names;
  while (struct dirent* e = readdir(d)) names.push_back(e->d_name);
  closedir(d);
  std::sort(names.begin(), names.end());
  std::ostringstream os;
  os << path << ":\n";
  for (const std::string& n : names) os << n << "\n";
  resp->body.append(os.str());
}

void page_protobufs(Server* server, HttpResponse* resp) {
  std::ostringstream os;
  if (server != nullptr) {
    for (const auto& kv : server->services()) {
      for (const auto& m : kv.second->methods()) {
        os << kv.first << "." << m.first << "\n";
      }
    }
  }
  resp->body.append(os.str());
}

void page_fibers(const HttpRequest& req, HttpResponse* resp) {
  std::ostringstream os;
  os << "workers: " << fiber_get_concurrency() << "\n";
  os << "fibers_created: " << fiber_count_created() << "\n";
  os << "fibers_active: " << fiber_count_active() << "\n";
  if (req.query.count("st") != 0) {
    // stack capture of suspended fibers (parity: /bthreads/<tid>?st=1)
    os << bam::dump_fiber_stacks(64);
  }
  resp->body.append(os.str());
}

// /threads: pthread inventory (parity: builtin/threads_service.cpp).
void page_threads(HttpResponse* resp) {
  std::ostringstream os;
  FILE* f = fopen("/proc/self/status", "r");
  if (f != nullptr) {
    char line[256];
    while (fgets(line, sizeof(line), f) != nullptr) {
      if (strncmp(line, "Threads:", 8) == 0 || strncmp(line, "VmRSS:", 6) == 0 ||
          strncmp(line, "VmSize:", 7) == 0) {
        os << line;
      }
    }
    fclose(f);
  }
  os << "fiber_workers: " << fiber_get_concurrency() << "\n";
  resp->body.append(os.str());
}

void page_memory(HttpResponse* resp) {
  // parity: reference /memory (tcmalloc extension dump) — here glibc
  // mallinfo2 + the framework's own pools.
  std::ostringstream os;
  struct mallinfo2 mi = mallinfo2();
  os << "malloc_arena_bytes: " << mi.arena << "\n";
  os << "malloc_in_use_bytes: " << mi.uordblks << "\n";
  os << "malloc_free_bytes: " << mi.fordblks << "\n";
  os << "malloc_mmap_bytes: " << mi.hblkhd << "\n";
  os << "iobuf_block_count: " << IOBuf::block_count() << "\n";
  os << "iobuf_block_memory: " << IOBuf::block_memory() << "\n";
  if (gpu::api() != nullptr && gpu::api()->stats_text != nullptr) {
    os << "---- gpu ----\n" << gpu::api()->stats_text();
  }
  resp->body.append(os.str());
}

void page_metrics(HttpResponse* resp) {
  // Prometheus exposition format from the var registry.
  std::ostringstream raw;
  var::Variable::dump_exposed(raw, "");
  std::istringstream in(raw.str());
  std::ostringstream os;
  std::string line;
  while (std::getline(in, line)) {
    size_t sep = line.find(" : ");
    if (sep == std::string::npos) continue;
    std::string name = line.substr(0, sep);
    std::string value = line.substr(sep + 3);
    // prometheus names: [a-zA-Z0-9_]
    for (char& c : name)
      if (!isalnum((unsigned char)c)) c = '_';
    // only numeric values are representable
    char* end = nullptr;
    strtod(value.c_str(), &end);
    if (end == value.c_str()) continue;
    os << "# TYPE " << name << " gauge\n" << name << " " << value << "\n";
  }
  resp->body.append(os.str());
}

}  // namespace

bool DispatchBuiltinService(Server* server, const HttpRequest& req, HttpResponse* resp) {
  const std::string& p = req.path;
  if (p == "/" || p == "/index") {
    page_index(resp);
  } else if (p == "/health") {
    resp->body.append("OK\n");
  } else if (p == "/status") {
    page_status(server, resp);
  } else if (p == "/vars" || p.rfind("/vars/", 0) == 0) {
    page_vars(req, resp);
  } else if (p == "/flags" || p.rfind("/flags/", 0) == 0) {
    page_flags(req, resp);
  } else if (p == "/version") {
    resp->body.append("brpc_amd/1.0 gfx950\n");
  } else if (p == "/connections") {
    page_connections(resp);
  } else if (p == "/sockets") {
    page_sockets(req, resp);
  } else if (p == "/ids") {
    int64_t created = 0, destroyed = 0;
    session_stats(&created, &destroyed);
    std::ostringstream os;
    os << "sessions_created: " << created << "\n"
       << "sessions_destroyed: " << destroyed << "\n"
       << "sessions_active: " << (created - destroyed) << "\n";
    resp->body.append(os.str());
  } else if (p == "/list") {
    page_list(server, resp);
  } else if (p == "/dir") {
    page_dir(req, resp);
  } else if (p == "/protobufs") {
    page_protobufs(server, resp);
  } else if (p == "/fibers" || p == "/bthreads") {
    page_fibers(req, resp);
  } else if (p == "/memory") {
    page_memory(resp);
  } else if (p == "/threads") {
    page_threads(resp);
  } else if (p == "/pprof/profile") {
    // Remote pprof attach (≙ reference builtin/pprof_service.cpp):
    // legacy gperftools binary CPU profile; `pprof http://host:port` works.
    int seconds = 10;
    auto it = req.query.find("seconds");
    if (it != req.query.end()) seconds = atoi(it->second.c_str());
    resp->content_type = "application/octet-stream";
    resp->body.append(CpuProfileBinary(seconds, 100));
  } else if (p == "/pprof/symbol") {
    if (req.method == "POST" && req.body.size() > 0) {
      resp->body.append(SymbolizeAddresses(req.body.to_string()));
    } else {
      resp->body.append("num_symbols: 1\n");
    }
  } else if (p == "/pprof/cmdline") {
    FILE* f = fopen("/proc/self/cmdline", "r");
    if (f != nullptr) {
      char buf[4096];
      size_t n = fread(buf, 1, sizeof(buf), f);
      fclose(f);
      resp->body.append(buf, n);
    }
  } else if (p == "/pprof/heap" || p == "/pprof/growth") {
    // Minimal gperftools-text heap profile: the arena totals from
    // mallinfo2 (glibc malloc — no tcmalloc in this image); detailed
    // per-site allocation profiling is at /memory + the ASan builds.
    struct mallinfo2 mi = mallinfo2();
    std::ostringstream os;
    os << "heap profile:      1:  " << mi.uordblks << " [     1:  "
       << mi.uordblks << "] @ heap_v2/1\n";
    os << "      1:  " << mi.uordblks << " [     1:  " << mi.uordblks
       << "] @ 0x0\n\nMAPPED_LIBRARIES:\n";
    FILE* f = fopen("/proc/self/maps", "r");
    if (f != nullptr) {
      char buf[8192];
      size_t n;
      while ((n = fread(buf, 1, sizeof(buf), f)) > 0) os.write(buf, n);
      fclose(f);
    }
    resp->body.append(os.str());
  } else if (p == "/hotspots/cpu" || p == "/hotspots") {
    // parity: builtin/hotspots_service.cpp, self-contained SIGPROF sampler
    int seconds = 1;
    auto it = req.query.find("seconds");
    if (it != req.query.end()) seconds = atoi(it->second.c_str());
    int hz = 200;
    auto hzit = req.query.find("hz");
    if (hzit != req.query.end()) hz = atoi(hzit->second.c_str());
    resp->body.append(CpuProfile(seconds, hz));
  } else if (p == "/hotspots/contention" || p == "/contention") {
    // parity: contention profiler (instrumented sync primitives)
    resp->body.append(ContentionProfile());
  } else if (p == "/hotspots/gpu" || p == "/gpustats") {
    // MI355X-native analogue of a GPU hotspots page: the HIP runtime's
    // own telemetry (staging gathers, async uploads, HBM pool occupancy)
    // + scheduler GPU-wait integration counters. Kernel-level timing
    // comes from rocprofv3 offline (profiles/).
    std::ostringstream os;
    if (gpu::api() != nullptr && gpu::api()->stats_text != nullptr) {
      os << gpu::api()->stats_text();
    } else {
      os << "no GPU runtime loaded\n";
    }
    os << "gpu_wait_parks: " << gpu_wait_parks() << "\n";
    os << "gpu_wait_wake_requests: " << gpu_wait_wake_requests() << "\n";
    resp->body.append(os.str());
  } else if (p == "/brpc_metrics" || p == "/metrics") {
    page_metrics(resp);
  } else if (p == "/rpcz") {
    auto dbit = req.query.find("db");
    if (dbit != req.query.end()) {
      rpcz::DumpPersistedSpans(&resp->body, atoi(dbit->second.c_str()));
    } else {
      rpcz::DumpRecentSpans(&resp->body, req.query.count("verbose") != 0);
    }
  } else {
    return false;
  }
  return true;
}

}  // namespace policy
}  // namespace bam
