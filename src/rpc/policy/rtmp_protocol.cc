// brpc_amd: RTMP server protocol.
// Parity: reference brpc/rtmp.cpp + policy/rtmp_protocol.cpp (clean-room
// subset): plain C0C1C2/S0S1S2 handshake, chunk streams (rtmp_chunk.h),
// AMF0 commands connect / createStream / publish / play / deleteStream,
// control messages WindowAckSize / SetPeerBandwidth / SetChunkSize, and a
// built-in publish→play relay hub (the media-server basis: audio(8),
// video(9), data(18) messages from a publisher fan out to every player of
// the same stream name). Digest-authenticated handshakes and FLV/HLS
// remuxing are out of scope this round.
// Enabled by ServerOptions::enable_rtmp.
#include <string.h>

#include <map>
#include <mutex>
#include <string>
#include <vector>

#include "base/fast_rand.h"
#include "base/iobuf.h"
#include "base/logging.h"
#include "rpc/amf.h"
#include "rpc/policy/http_protocol.h"
#include "rpc/protocol.h"
#include "rpc/rtmp_chunk.h"
#include "rpc/server.h"
#include "rpc/socket.h"

namespace bam {
namespace policy {

namespace {

using rtmp::ChunkReader;
using rtmp::Message;

constexpr size_t kHsSize = 1536;

struct RtmpCtx {
  int hs_state = 0;  // 0 await C0C1, 1 await C2, 2 streaming
  ChunkReader reader;
  uint32_t out_chunk = 4096;
  std::string app;
  std::string publishing;  // stream name if this connection publishes
  std::string playing;     // stream name if this connection plays
  SocketId socket_id = 0;  // for hub cleanup at recycle
};

// ---- relay hub ----
struct Hub {
  std::mutex mu;
  // stream name -> player socket ids
  std::map<std::string, std::vector<SocketId>> players;
};
Hub& hub() {
  static Hub* h = new Hub;
  return *h;
}

void hub_add_player(const std::string& name, SocketId sid) {
  Hub& h = hub();
  std::lock_guard<std::mutex> lk(h.mu);
  h.players[name].push_back(sid);
}
void hub_remove_player(const std::string& name, SocketId sid) {
  Hub& h = hub();
  std::lock_guard<std::mutex> lk(h.mu);
  auto it = h.players.find(name);
  if (it == h.players.end()) return;
  for (size_t i = 0; i < it->second.size(); ++i) {
    if (it->second[i] == sid) {
      it->second[i] = it->second.back();
      it->second.pop_back();
      break;
    }
  }
}
std::vector<SocketId> hub_players(const std::string& name) {
  Hub& h = hub();
  std::lock_guard<std::mutex> lk(h.mu);
  auto it = h.players.find(name);
  return it == h.players.end() ? std::vector<SocketId>() : it->second;
}

void send_raw(Socket* sock, const std::string& bytes) {
  IOBuf out;
  out.append(bytes);
  sock->Write(&out);
}

void send_message(Socket* sock, RtmpCtx* ctx, uint32_t csid, uint8_t type, uint32_t msid,
                  uint32_t ts, const std::string& payload) {
  std::string bytes;
  rtmp::write_message(&bytes, csid, type, msid, ts, payload, ctx->out_chunk);
  send_raw(sock, bytes);
}

void handle_message(Socket* sock_raw, RtmpCtx* ctx, Server* server, Message& m);

int g_rtmp_protocol_index = -1;

ParseResult ParseRtmp(IOBuf* source, Socket* sock, bool) {
  Server* server = (Server*)sock->user();
  if (server == nullptr || !server->options().enable_rtmp)
    return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);

  if (sock->protocol_ctx != nullptr && sock->protocol_ctx_owner != g_rtmp_protocol_index)
    return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
  RtmpCtx* ctx = (RtmpCtx*)sock->protocol_ctx;
  if (ctx == nullptr) {
    // C0 must be version 3 before we claim the connection.
    char c0;
    if (source->size() < 1) return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
    source->copy_to(&c0, 1, 0);
    if (c0 != 0x03) return ParseResult::make_error(PARSE_ERROR_TRY_OTHERS);
    ctx = new RtmpCtx;
    ctx->socket_id = sock->id();
    sock->protocol_ctx = ctx;
    sock->protocol_ctx_deleter = [](void* p) {
      RtmpCtx* c = (RtmpCtx*)p;
      // a player that vanished without deleteStream must leave the hub
      if (!c->playing.empty()) hub_remove_player(c->playing, c->socket_id);
      delete c;
    };
    sock->protocol_ctx_owner = g_rtmp_protocol_index;
  }
  if (ctx->hs_state == 0) {
    if (source->size() < 1 + kHsSize)
      return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
    std::string c0c1;
    source->cutn(&c0c1, 1 + kHsSize);
    // S0 + S1 (our random) + S2 (echo C1)
    std::string out(1, '\x03');
    std::string s1(kHsSize, '\0');
    for (size_t i = 8; i < kHsSize; ++i) s1[i] = (char)fast_rand();
    out += s1;
    out.append(c0c1, 1, kHsSize);
    send_raw(sock, out);
    ctx->hs_state = 1;
    return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
  }
  if (ctx->hs_state == 1) {
    if (source->size() < kHsSize) return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
    source->pop_front(kHsSize);  // C2 (unvalidated, like plain handshake)
    ctx->hs_state = 2;
  }
  // Pump: drain source into the chunk reader and handle every complete
  // message inline (like the h2 session pump) — the InputMessenger loop
  // stops when read_buf empties, so parse cannot leave completed messages
  // behind in the reader.
  if (!source->empty()) {
    std::string bytes;
    source->copy_to(&bytes, (size_t)-1, 0);
    source->pop_front(bytes.size());
    ctx->reader.feed(bytes.data(), bytes.size());
  }
  for (;;) {
    Message m;
    int rc = ctx->reader.next(&m);
    if (rc < 0) return ParseResult::make_error(PARSE_ERROR_ABSOLUTELY_WRONG);
    if (rc == 0) return ParseResult::make_error(PARSE_ERROR_NOT_ENOUGH_DATA);
    handle_message(sock, ctx, server, m);
  }
}

void reply_command(Socket* sock, RtmpCtx* ctx, const std::vector<amf::Value>& vals) {
  std::string payload;
  for (const amf::Value& v : vals) amf::Encode(v, &payload);
  send_message(sock, ctx, 3, rtmp::kMsgCommandAmf0, 0, 0, payload);
}

amf::Value status_object(const char* level, const char* code, const char* desc) {
  amf::Value st = amf::Value::Object();
  st.obj["level"] = amf::Value::Str(level);
  st.obj["code"] = amf::Value::Str(code);
  st.obj["description"] = amf::Value::Str(desc);
  return st;
}

void handle_message(Socket* sock_raw, RtmpCtx* ctx, Server* server, Message& m) {
  SocketUniquePtr sock;
  if (Socket::Address(sock_raw->id(), &sock) != 0) return;
  switch (m.type) {
    case rtmp::kMsgCommandAmf0: {
      std::vector<amf::Value> vals;
      if (!amf::DecodeAll(m.payload.data(), m.payload.size(), &vals) || vals.empty() ||
          vals[0].type != amf::Value::STRING) {
        break;
      }
      const std::string& cmd = vals[0].str;
      double txn = vals.size() > 1 && vals[1].type == amf::Value::NUMBER ? vals[1].num : 0;
      if (cmd == "connect") {
        if (vals.size() > 2 && vals[2].type == amf::Value::OBJECT) {
          auto it = vals[2].obj.find("app");
          if (it != vals[2].obj.end()) ctx->app = it->second.str;
        }
        // control preamble: WindowAckSize, SetPeerBandwidth, SetChunkSize
        send_message(sock.get(), ctx, 2, rtmp::kMsgWindowAckSize, 0, 0,
                     rtmp::u32_be(2500000));
        send_message(sock.get(), ctx, 2, rtmp::kMsgSetPeerBandwidth, 0, 0,
                     rtmp::u32_be(2500000) + std::string(1, '\x02'));
        send_message(sock.get(), ctx, 2, rtmp::kMsgSetChunkSize, 0, 0,
                     rtmp::u32_be(ctx->out_chunk));
        amf::Value props = amf::Value::Object();
        props.obj["fmsVer"] = amf::Value::Str("BAM/1.0");
        props.obj["capabilities"] = amf::Value::Number(31);
        amf::Value info = status_object("status", "NetConnection.Connect.Success",
                                        "Connection succeeded.");
        info.obj["objectEncoding"] = amf::Value::Number(0);
        reply_command(sock.get(), ctx,
                      {amf::Value::Str("_result"), amf::Value::Number(txn), props, info});
      } else if (cmd == "createStream") {
        reply_command(sock.get(), ctx,
                      {amf::Value::Str("_result"), amf::Value::Number(txn),
                       amf::Value::Null(), amf::Value::Number(1)});
      } else if (cmd == "publish") {
        if (vals.size() > 3 && vals[3].type == amf::Value::STRING) {
          ctx->publishing = ctx->app + "/" + vals[3].str;
        }
        reply_command(sock.get(), ctx,
                      {amf::Value::Str("onStatus"), amf::Value::Number(0),
                       amf::Value::Null(),
                       status_object("status", "NetStream.Publish.Start",
                                     "Start publishing")});
      } else if (cmd == "play") {
        if (vals.size() > 3 && vals[3].type == amf::Value::STRING) {
          ctx->playing = ctx->app + "/" + vals[3].str;
          hub_add_player(ctx->playing, sock->id());
        }
        reply_command(sock.get(), ctx,
                      {amf::Value::Str("onStatus"), amf::Value::Number(0),
                       amf::Value::Null(),
                       status_object("status", "NetStream.Play.Start", "Start playing")});
      } else if (cmd == "deleteStream" || cmd == "closeStream") {
        if (!ctx->playing.empty()) {
          hub_remove_player(ctx->playing, sock->id());
          ctx->playing.clear();
        }
        ctx->publishing.clear();
      }
      server->nprocessed.fetch_add(1, std::memory_order_relaxed);
      break;
    }
    case rtmp::kMsgAudio:
    case rtmp::kMsgVideo:
    case rtmp::kMsgDataAmf0: {
      if (ctx->publishing.empty()) break;
      for (SocketId pid : hub_players(ctx->publishing)) {
        SocketUniquePtr player;
        if (Socket::Address(pid, &player) != 0) {
          hub_remove_player(ctx->publishing, pid);
          continue;
        }
        RtmpCtx* pctx = (RtmpCtx*)player->protocol_ctx;
        if (pctx != nullptr) {
          send_message(player.get(), pctx, m.type == rtmp::kMsgAudio ? 4u : 5u, m.type,
                       1, m.timestamp, m.payload);
        }
      }
      break;
    }
    default:
      break;  // acks, user control: ignored
  }
  // player teardown on socket failure is handled lazily (Address() fails)
}

}  // namespace

void RegisterRtmpProtocol() {
  static std::once_flag once;
  std::call_once(once, [] {
    Protocol p;
    p.parse = ParseRtmp;  // self-contained pump; no message objects emitted
    p.support_server = true;
    p.support_client = false;
    p.name = "rtmp";
    g_rtmp_protocol_index = RegisterProtocol(p);
  });
}

}  // namespace policy
}  // namespace bam
