// brpc_amd: InputMessenger — protocol-agnostic reader.
// Parity: reference brpc/input_messenger.h: drains the socket, cuts
// messages with registered protocol parsers (remembering the preferred
// index per socket), then runs each message's process callback in a fiber.
#pragma once

#include "rpc/protocol.h"
#include "rpc/socket.h"

namespace bam {

class InputMessenger {
 public:
  explicit InputMessenger(bool server_side) : server_side_(server_side) {}

  // The socket's on_edge_triggered_events callback: reads until EAGAIN and
  // dispatches every complete message.
  void OnNewMessages(Socket* s);

  bool server_side() const { return server_side_; }

 private:
  void DispatchMessage(InputMessageBase* msg, int protocol_index);

  bool server_side_;
};

}  // namespace bam
