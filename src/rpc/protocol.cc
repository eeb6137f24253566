#include "rpc/protocol.h"

#include "base/logging.h"

#include <atomic>
#include <cstring>

#include "base/logging.h"

namespace bam {

static Protocol g_protocols[kMaxProtocols];
static std::atomic<int> g_nprotocols{0};

int RegisterProtocol(const Protocol& p) {
  int n = g_nprotocols.load(std::memory_order_acquire);
  CHECK(n < kMaxProtocols) << "protocol registry full (" << n
                           << "): raise kMaxProtocols";  // silent -1 once cost rtmp its slot
  if (n >= kMaxProtocols) return -1;
  g_protocols[n] = p;
  g_nprotocols.store(n + 1, std::memory_order_release);
  return n;
}

const Protocol* GetProtocol(int index) {
  if (index < 0 || index >= g_nprotocols.load(std::memory_order_acquire)) return nullptr;
  return &g_protocols[index];
}

int FindProtocolIndex(const std::string& name) {
  int n = g_nprotocols.load(std::memory_order_acquire);
  for (int i = 0; i < n; ++i) {
    if (name == g_protocols[i].name) return i;
  }
  return -1;
}

int FindClientProtocolIndex(const std::string& name) {
  // A name can be registered twice (e.g. "h2": server parse entry AND the
  // nghttp2 client entry). Channels must bind to one that can issue
  // client requests.
  int n = g_nprotocols.load(std::memory_order_acquire);
  int any = -1;
  for (int i = 0; i < n; ++i) {
    if (name == g_protocols[i].name) {
      if (g_protocols[i].support_client) return i;
      if (any < 0) any = i;
    }
  }
  return any;
}

int ProtocolCount() { return g_nprotocols.load(std::memory_order_acquire); }

}  // namespace bam
