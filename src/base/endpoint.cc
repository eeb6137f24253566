#include "base/endpoint.h"

#include <fcntl.h>
#include <netdb.h>
#include <netinet/tcp.h>
#include <stdio.h>
#include <string.h>
#include <sys/socket.h>
#include <unistd.h>

namespace bam {

int str2endpoint(const char* str, EndPoint* ep) {
  const char* colon = strrchr(str, ':');
  if (colon == nullptr) return -1;
  std::string host(str, colon - str);
  int port = atoi(colon + 1);
  if (port < 0 || port > 65535) return -1;
  return hostname2endpoint(host.c_str(), port, ep);
}

int hostname2endpoint(const char* host, int port, EndPoint* ep) {
  if (host[0] == '\0' || strcmp(host, "0.0.0.0") == 0) {
    ep->ip.s_addr = INADDR_ANY;
    ep->port = port;
    return 0;
  }
  if (inet_pton(AF_INET, host, &ep->ip) == 1) {
    ep->port = port;
    return 0;
  }
  struct addrinfo hints;
  memset(&hints, 0, sizeof(hints));
  hints.ai_family = AF_INET;
  hints.ai_socktype = SOCK_STREAM;
  struct addrinfo* res = nullptr;
  if (getaddrinfo(host, nullptr, &hints, &res) != 0 || res == nullptr) return -1;
  ep->ip = ((struct sockaddr_in*)res->ai_addr)->sin_addr;
  ep->port = port;
  freeaddrinfo(res);
  return 0;
}

std::string endpoint2str(const EndPoint& ep) {
  char buf[32];
  char ipbuf[INET_ADDRSTRLEN];
  inet_ntop(AF_INET, &ep.ip, ipbuf, sizeof(ipbuf));
  snprintf(buf, sizeof(buf), "%s:%d", ipbuf, ep.port);
  return std::string(buf);
}

int make_non_blocking(int fd) {
  int flags = fcntl(fd, F_GETFL, 0);
  if (flags < 0) return -1;
  return fcntl(fd, F_SETFL, flags | O_NONBLOCK);
}

int make_no_delay(int fd) {
  int one = 1;
  return setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
}

int make_close_on_exec(int fd) { return fcntl(fd, F_SETFD, FD_CLOEXEC); }

int tcp_listen(const EndPoint& ep, int backlog) {
  int fd = socket(AF_INET, SOCK_STREAM | SOCK_NONBLOCK | SOCK_CLOEXEC, 0);
  if (fd < 0) return -1;
  int one = 1;
  setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
  struct sockaddr_in addr;
  memset(&addr, 0, sizeof(addr));
  addr.sin_family = AF_INET;
  addr.sin_addr = ep.ip;
  addr.sin_port = htons((uint16_t)ep.port);
  if (bind(fd, (struct sockaddr*)&addr, sizeof(addr)) != 0 || listen(fd, backlog) != 0) {
    close(fd);
    return -1;
  }
  return fd;
}

int tcp_connect(const EndPoint& ep, bool* in_progress) {
  if (in_progress) *in_progress = false;
  int fd = socket(AF_INET, SOCK_STREAM | SOCK_NONBLOCK | SOCK_CLOEXEC, 0);
  if (fd < 0) return -1;
  struct sockaddr_in addr;
  memset(&addr, 0, sizeof(addr));
  addr.sin_family = AF_INET;
  addr.sin_addr = ep.ip;
  addr.sin_port = htons((uint16_t)ep.port);
  int rc = connect(fd, (struct sockaddr*)&addr, sizeof(addr));
  if (rc != 0) {
    if (errno == EINPROGRESS) {
      if (in_progress) *in_progress = true;
      return fd;
    }
    close(fd);
    return -1;
  }
  return fd;
}

static int get_side(int fd, EndPoint* ep, bool local) {
  struct sockaddr_in addr;
  socklen_t len = sizeof(addr);
  int rc = local ? getsockname(fd, (struct sockaddr*)&addr, &len)
                 : getpeername(fd, (struct sockaddr*)&addr, &len);
  if (rc != 0) return -1;
  ep->ip = addr.sin_addr;
  ep->port = ntohs(addr.sin_port);
  return 0;
}

int get_local_side(int fd, EndPoint* ep) { return get_side(fd, ep, true); }
int get_remote_side(int fd, EndPoint* ep) { return get_side(fd, ep, false); }

}  // namespace bam
