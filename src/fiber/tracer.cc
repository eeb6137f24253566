// brpc_amd: fiber tracer — stack capture of suspended fibers.
// Parity: reference bthread/task_tracer.h (/bthreads/<tid>?st=1): walks a
// suspended fiber's saved frame-pointer chain from its switch context
// (context.S stores rbp at sp+48, return address at sp+56) and symbolizes
// frames with dladdr. Running fibers report their worker; only parked
// fibers have a stable stack to walk.
#include <dlfcn.h>

#include <sstream>
#include <vector>

#include "base/resource_pool.h"
#include "fiber/scheduler.h"

namespace bam {

namespace {

struct Frame {
  uintptr_t ip;
};

// Walks an rbp chain constrained to [stack_base, stack_top).
void walk_frames(uintptr_t rbp, uintptr_t ip0, uintptr_t lo, uintptr_t hi,
                 std::vector<Frame>* out) {
  out->push_back(Frame{ip0});
  uintptr_t rbp_cur = rbp;
  for (int depth = 0; depth < 32; ++depth) {
    if (rbp_cur < lo || rbp_cur + 16 > hi || (rbp_cur & 7) != 0) break;
    uintptr_t next_rbp = *(uintptr_t*)rbp_cur;
    uintptr_t ret = *(uintptr_t*)(rbp_cur + 8);
    if (ret < 0x1000) break;
    out->push_back(Frame{ret});
    if (next_rbp <= rbp_cur) break;
    rbp_cur = next_rbp;
  }
}

std::string symbolize(uintptr_t ip) {
  Dl_info info;
  if (dladdr((void*)ip, &info) != 0 && info.dli_sname != nullptr) {
    char buf[512];
    snprintf(buf, sizeof(buf), "%#lx %s+%#lx", ip, info.dli_sname,
             ip - (uintptr_t)info.dli_saddr);
    return buf;
  }
  char buf[32];
  snprintf(buf, sizeof(buf), "%#lx", ip);
  return buf;
}

}  // namespace

// Dumps all live (suspended) fibers with best-effort stacks. Racy by
// design (like the reference tracer): a fiber may resume mid-walk; frames
// are bounds-checked against its own stack so the walker cannot fault on
// its memory, but output may be garbage for just-resumed fibers.
std::string dump_fiber_stacks(int max_fibers) {
  std::ostringstream os;
  int found = 0;
  for (uint32_t rid = 0; rid < 4096 && found < max_fibers; ++rid) {
    FiberMeta* m = address_resource<FiberMeta>(rid);
    if (m == nullptr) break;  // past the allocated blocks
    char* stack_base = m->stack_base;
    void* sp = m->ctx_sp;
    if (stack_base == nullptr || sp == nullptr) continue;
    uintptr_t lo = (uintptr_t)stack_base;
    uintptr_t hi = lo + m->stack_size;
    uintptr_t usp = (uintptr_t)sp;
    if (usp < lo || usp + 64 > hi) continue;  // running or being recycled
    ++found;
    // context.S frame: [fpu 8][r15][r14][r13][r12][rbx][rbp][ret]
    uintptr_t rbp = *(uintptr_t*)(usp + 48);
    uintptr_t ip = *(uintptr_t*)(usp + 56);
    std::vector<Frame> frames;
    walk_frames(rbp, ip, lo, hi, &frames);
    os << "fiber #" << (rid + 1) << " version=" << m->version.load() << " stack=" << (void*)lo
       << "\n";
    for (const Frame& f : frames) os << "    " << symbolize(f.ip) << "\n";
  }
  if (found == 0) os << "no suspended fibers\n";
  return os.str();
}

}  // namespace bam
