#include "rpc/input_messenger.h"

#include "base/time.h"

#include <errno.h>

#include "base/flags.h"
#include "base/logging.h"
#include "fiber/fiber.h"

namespace bam {

// Processing responses inline in the parse fiber skips one fiber spawn per
// call (response processing is framework code: it completes a session and
// wakes the joiner). Requests keep fiber-per-message isolation by default
// (user handlers may block) — parity with the reference's ProcessInplace
// option.
BAM_DEFINE_bool(process_response_in_place, true,
                "run client response processing inline in the parse fiber");
BAM_DEFINE_bool(process_request_in_place, false,
                "run server request handlers inline in the parse fiber");

namespace {

struct ProcessArg {
  InputMessageBase* msg;
  const Protocol* proto;
  bool server_side;
};

void ProcessMessageFiber(void* raw) {
  ProcessArg* a = (ProcessArg*)raw;
  if (a->server_side) {
    a->proto->process_request(a->msg);
  } else {
    a->proto->process_response(a->msg);
  }
  delete a;
}

}  // namespace

void InputMessenger::DispatchMessage(InputMessageBase* msg, int protocol_index) {
  const Protocol* proto = GetProtocol(protocol_index);
  msg->protocol_index = protocol_index;
  const bool in_place =
      server_side_ ? FLAG_process_request_in_place : FLAG_process_response_in_place;
  if (in_place) {
    if (server_side_) proto->process_request(msg);
    else proto->process_response(msg);
    return;
  }
  ProcessArg* a = new ProcessArg{msg, proto, server_side_};
  fiber_t th;
  // Urgent start: the parsing fiber is requeued, the message handler runs
  // immediately (the reference's latency trick).
  if (fiber_start_urgent(&th, ProcessMessageFiber, a) != 0) {
    ProcessMessageFiber(a);
  }
}

void InputMessenger::OnNewMessages(Socket* s) {
  const size_t kOnceRead = 256 * 1024;
  // TLS: input edges drive the (non-blocking) handshake until it is done;
  // no application bytes exist before that.
  if (s->ssl_state() == 1) {
    if (s->ssl_continue_handshake() != 0) return;  // failed -> SetFailed
    if (s->ssl_state() == 1) return;               // wants more transport data
  }
  bool eof = false;
  while (!s->Failed()) {
    ssize_t nr = s->read_bytes(&s->read_buf(), kOnceRead);
    if (nr < 0) {
      if (errno == EAGAIN || errno == EWOULDBLOCK) {
        // drained; fall through to parse what we have, then return
      } else if (errno == EINTR) {
        continue;
      } else {
        int err = errno;
        s->SetFailed(err, "read failed");
        return;
      }
    } else if (nr == 0) {
      eof = true;
    } else {
      s->in_bytes.fetch_add(nr, std::memory_order_relaxed);
      s->last_active_us.store(monotonic_time_us(), std::memory_order_relaxed);
    }

    // Parse as many complete messages as possible.
    while (!s->read_buf().empty()) {
      int idx = s->preferred_protocol_index;
      InputMessageBase* msg = nullptr;
      int matched = -1;
      if (idx >= 0) {
        const Protocol* p = GetProtocol(idx);
        ParseResult r = p->parse(&s->read_buf(), s, eof);
        if (r.error == PARSE_OK) {
          msg = r.msg;
          matched = idx;
        } else if (r.error == PARSE_ERROR_NOT_ENOUGH_DATA) {
          break;
        } else if (r.error == PARSE_ERROR_ABSOLUTELY_WRONG) {
          s->SetFailed(EPROTO, "protocol parse error");
          return;
        }
        // TRY_OTHERS falls through to scan
      }
      if (msg == nullptr) {
        bool not_enough = false;
        int n = ProtocolCount();
        for (int i = 0; i < n && msg == nullptr; ++i) {
          if (i == idx) continue;
          const Protocol* p = GetProtocol(i);
          if (p->parse == nullptr) continue;
          if (server_side_ && !p->support_server) continue;
          if (!server_side_ && !p->support_client) continue;
          ParseResult r = p->parse(&s->read_buf(), s, eof);
          switch (r.error) {
            case PARSE_OK:
              msg = r.msg;
              matched = i;
              s->preferred_protocol_index = i;
              break;
            case PARSE_ERROR_NOT_ENOUGH_DATA:
              not_enough = true;
              break;
            default:
              break;  // try next protocol
          }
        }
        if (msg == nullptr) {
          if (not_enough) break;  // wait for more bytes
          s->SetFailed(EPROTO, "no protocol matched input");
          return;
        }
      }
      msg->socket_id = s->id();
      s->in_messages.fetch_add(1, std::memory_order_relaxed);
      DispatchMessage(msg, matched);
    }

    if (eof) {
      s->SetFailed(ECONNRESET, "remote closed connection");
      return;
    }
    if (nr < 0) return;  // EAGAIN: wait for the next edge
    if ((size_t)nr < kOnceRead) {
      // Socket likely drained; one more read attempt will hit EAGAIN —
      // loop continues to confirm (edge-triggered requires full drain).
    }
  }
}

}  // namespace bam
