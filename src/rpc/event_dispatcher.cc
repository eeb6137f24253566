#include "rpc/event_dispatcher.h"

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

EventDispatcher* EventDispatcher::singleton() {
  static EventDispatcher* d = new EventDispatcher;
  return d;
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
