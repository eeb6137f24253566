#include "rpc/event_dispatcher.h"

#include <stdlib.h>
#include <sys/epoll.h>
#include <unistd.h>

#include <thread>

#include "base/logging.h"

namespace bam {

EventDispatcher::EventDispatcher() {
  epfd_ = epoll_create1(EPOLL_CLOEXEC);
  CHECK_GE(epfd_, 0) << "epoll_create failed";
  std::thread([this] { run(); }).detach();
}

struct DispatcherSetAccess {
  static EventDispatcher* make() { return new EventDispatcher; }
};

namespace {
struct DispatcherSet {
  int n;
  EventDispatcher** d;
  DispatcherSet() {
    const char* e = getenv("BAM_EVENT_DISPATCHERS");
    n = e != nullptr ? atoi(e) : 1;
    if (n < 1) n = 1;
    if (n > 16) n = 16;
    d = new EventDispatcher*[n];
    for (int i = 0; i < n; ++i) d[i] = DispatcherSetAccess::make();
  }
};
DispatcherSet& dispatchers() {
  static DispatcherSet* s = new DispatcherSet;
  return *s;
}
}  // namespace

EventDispatcher* EventDispatcher::singleton() { return dispatchers().d[0]; }

EventDispatcher* EventDispatcher::dispatcher_for(SocketId sid) {
  DispatcherSet& s = dispatchers();
  // low id bits are the pool slot: spreads adjacent sockets round-robin
  return s.d[(uint32_t)sid % (uint32_t)s.n];
}

int EventDispatcher::add_consumer(SocketId sid, int fd) {
  struct epoll_event ev;
  ev.events = EPOLLIN | EPOLLOUT | EPOLLET | EPOLLRDHUP;
  ev.data.u64 = sid;
  return epoll_ctl(epfd_, EPOLL_CTL_ADD, fd, &ev);
}

int EventDispatcher::remove_consumer(int fd) {
  return epoll_ctl(epfd_, EPOLL_CTL_DEL, fd, nullptr);
}

void EventDispatcher::run() {
  const int kMaxEvents = 64;
  struct epoll_event events[kMaxEvents];
  for (;;) {
    int n = epoll_wait(epfd_, events, kMaxEvents, -1);
    if (n < 0) {
      if (errno == EINTR) continue;
      PLOG(ERROR) << "epoll_wait failed";
      return;
    }
    for (int i = 0; i < n; ++i) {
      SocketId sid = events[i].data.u64;
      SocketUniquePtr s;
      if (Socket::Address(sid, &s) != 0) continue;
      uint32_t ev = events[i].events;
      if (ev & EPOLLOUT) s->on_output_event();
      if (ev & (EPOLLIN | EPOLLRDHUP | EPOLLHUP | EPOLLERR)) s->on_input_event();
    }
  }
}

}  // namespace bam
