// brpc_amd: EventDispatcher — edge-triggered epoll loops on dedicated
// pthreads; consumers are Sockets addressed by versioned id. N loops
// (BAM_EVENT_DISPATCHERS, default 1; ≙ reference -event_dispatcher_num,
// brpc/event_dispatcher.h:197) shard sockets by id so one epoll thread
// is not the ceiling at several hundred kQPS.
#pragma once

#include <atomic>

#include "rpc/socket.h"

namespace bam {

class EventDispatcher {
 public:
  static EventDispatcher* singleton();          // shard 0 (legacy callers)
  static EventDispatcher* dispatcher_for(SocketId sid);

  // Registers fd with EPOLLIN|EPOLLOUT|EPOLLET, data = socket id.
  int add_consumer(SocketId sid, int fd);
  int remove_consumer(int fd);

 private:
  friend struct DispatcherSetAccess;
  EventDispatcher();
  void run();

  int epfd_;
};

}  // namespace bam
