// brpc_amd: EndPoint (ip:port) + TCP helpers.
// Parity: reference butil/endpoint.h + fd_utility.h.
#pragma once

#include <arpa/inet.h>
#include <netinet/in.h>
#include <stdint.h>

#include <string>

namespace bam {

struct EndPoint {
  in_addr ip;     // network byte order
  int port = 0;   // host byte order

  EndPoint() { ip.s_addr = 0; }
  EndPoint(in_addr ip2, int port2) : ip(ip2), port(port2) {}

  bool operator==(const EndPoint& o) const { return ip.s_addr == o.ip.s_addr && port == o.port; }
  bool operator!=(const EndPoint& o) const { return !(*this == o); }
  bool operator<(const EndPoint& o) const {
    return ip.s_addr != o.ip.s_addr ? ip.s_addr < o.ip.s_addr : port < o.port;
  }
};

// "1.2.3.4:80" or "hostname:80" -> EndPoint. Returns 0 on success.
int str2endpoint(const char* str, EndPoint* ep);
int hostname2endpoint(const char* host, int port, EndPoint* ep);
std::string endpoint2str(const EndPoint& ep);

// Returns listen fd (nonblocking, CLOEXEC, SO_REUSEADDR) or -1.
int tcp_listen(const EndPoint& ep, int backlog = 1024);
// Nonblocking connect; returns fd or -1. If connecting is in progress,
// *in_progress is set and the fd must be waited for EPOLLOUT.
int tcp_connect(const EndPoint& ep, bool* in_progress);
int make_non_blocking(int fd);
int make_no_delay(int fd);
int make_close_on_exec(int fd);
// Local address of a bound/connected socket.
int get_local_side(int fd, EndPoint* ep);
int get_remote_side(int fd, EndPoint* ep);

struct EndPointHasher {
  size_t operator()(const EndPoint& ep) const {
    return (size_t)ep.ip.s_addr * 101 + (size_t)ep.port;
  }
};

}  // namespace bam
