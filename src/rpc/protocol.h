// brpc_amd: pluggable wire-protocol seam.
// Parity: reference brpc/protocol.h (Protocol struct of function pointers,
// RegisterProtocol, InputMessenger tries registered parsers in order and
// remembers the socket's preferred index).
#pragma once

#include <functional>
#include <memory>
#include <string>

#include "base/iobuf.h"

namespace bam {

class Socket;
typedef uint64_t SocketId;

enum ParseErrorCode {
  PARSE_OK = 0,
  PARSE_ERROR_TRY_OTHERS,
  PARSE_ERROR_NOT_ENOUGH_DATA,
  PARSE_ERROR_NO_RESOURCE,
  PARSE_ERROR_ABSOLUTELY_WRONG,
};

// A parsed-but-not-yet-processed inbound message.
struct InputMessageBase {
  SocketId socket_id = 0;
  int protocol_index = -1;
  virtual ~InputMessageBase() {}
};

struct ParseResult {
  ParseErrorCode error = PARSE_ERROR_ABSOLUTELY_WRONG;
  InputMessageBase* msg = nullptr;

  static ParseResult make_ok(InputMessageBase* m) { return ParseResult{PARSE_OK, m}; }
  static ParseResult make_error(ParseErrorCode e) { return ParseResult{e, nullptr}; }
};

class Controller;
typedef uint64_t SessionId2;  // mirrors fiber SessionId without the include

struct Protocol {
  // Cuts one complete message from `source` (the socket's read buffer).
  ParseResult (*parse)(IOBuf* source, Socket* sock, bool read_eof) = nullptr;
  // Runs in a fiber; must delete/recycle msg. Server side.
  void (*process_request)(InputMessageBase* msg) = nullptr;
  // Runs in a fiber; must delete/recycle msg. Client side.
  void (*process_response)(InputMessageBase* msg) = nullptr;
  // Client-side request packing (header+meta+payload). nullptr = std only.
  void (*pack_request)(IOBuf* out, Controller* cntl, uint64_t correlation_id) = nullptr;
  // Stateful client protocols (h2): take over the whole issue step
  // (session bookkeeping + socket write). Returns 0 on success.
  int (*issue_request)(Socket* sock, Controller* cntl, uint64_t correlation_id) = nullptr;
  // True for protocols without correlation ids: responses match requests
  // FIFO on the connection (redis/memcache pipelining).
  bool client_pipelined = false;
  // True if this protocol can appear on server connections.
  bool support_server = false;
  bool support_client = false;
  const char* name = "unknown";
};

static const int kMaxProtocols = 32;

// Registration order = parse attempt order. Returns index or -1.
int RegisterProtocol(const Protocol& p);
const Protocol* GetProtocol(int index);
int FindProtocolIndex(const std::string& name);
// Prefers an entry with support_client (names can be double-registered).
int FindClientProtocolIndex(const std::string& name);
int ProtocolCount();

}  // namespace bam
