// brpc_amd: EventDispatcher — edge-triggered epoll loop on a dedicated
// pthread; consumers are Sockets addressed by versioned id.
// Parity: reference brpc/event_dispatcher.h (epoll variant).
#pragma once

#include <atomic>

#include "rpc/socket.h"

namespace bam {

class EventDispatcher {
 public:
  static EventDispatcher* singleton();

  // Registers fd with EPOLLIN|EPOLLOUT|EPOLLET, data = socket id.
  int add_consumer(SocketId sid, int fd);
  int remove_consumer(int fd);

 private:
  EventDispatcher();
  void run();

  int epfd_;
};

}  // namespace bam
