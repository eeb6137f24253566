#include "rpc/socket_map.h"

#include <map>
#include <mutex>

#include "rpc/input_messenger.h"

namespace bam {

namespace {

InputMessenger* client_messenger() {
  static InputMessenger* m = new InputMessenger(/*server_side=*/false);
  return m;
}

struct ClientSocketMap {
  std::mutex mu;
  std::map<EndPoint, SocketId> sockets;
};

ClientSocketMap& the_map() {
  static ClientSocketMap* m = new ClientSocketMap;
  return *m;
}

}  // namespace

int GetClientSocket(const EndPoint& ep, SocketUniquePtr* out) {
  ClientSocketMap& m = the_map();
  {
    std::lock_guard<std::mutex> lk(m.mu);
    auto it = m.sockets.find(ep);
    if (it != m.sockets.end()) {
      if (Socket::Address(it->second, out) == 0 && !(*out)->Failed()) return 0;
      m.sockets.erase(it);
      out->reset(nullptr);
    }
  }
  // Create outside the lock (connect may take time), then publish.
  InputMessenger* messenger = client_messenger();
  SocketOptions opts;
  opts.remote_side = ep;
  opts.connect_on_create = true;
  opts.on_edge_triggered_events = [messenger](Socket* s) { messenger->OnNewMessages(s); };
  SocketId sid;
  if (Socket::Create(opts, &sid) != 0) return -1;
  if (Socket::Address(sid, out) != 0) return -1;
  {
    std::lock_guard<std::mutex> lk(m.mu);
    auto it = m.sockets.find(ep);
    if (it != m.sockets.end()) {
      // Raced with another creator: prefer the existing healthy one.
      SocketUniquePtr existing;
      if (Socket::Address(it->second, &existing) == 0 && !existing->Failed()) {
        (*out)->SetFailed(ECANCELED, "duplicate connection");
        out->reset(existing.release());
        return 0;
      }
      it->second = sid;
      return 0;
    }
    m.sockets[ep] = sid;
  }
  return 0;
}

void RemoveClientSocket(const EndPoint& ep, SocketId expected) {
  ClientSocketMap& m = the_map();
  std::lock_guard<std::mutex> lk(m.mu);
  auto it = m.sockets.find(ep);
  if (it != m.sockets.end() && it->second == expected) m.sockets.erase(it);
}

}  // namespace bam
