// brpc_amd: HTTP/1.1 types + registration (see http_protocol.cc).
#pragma once

#include <map>
#include <string>

#include "base/iobuf.h"

namespace bam {

class Server;

namespace policy {

struct HttpRequest {
  std::string method;
  std::string path;
  std::map<std::string, std::string> query;
  std::map<std::string, std::string> headers;  // lower-cased keys
  IOBuf body;
  bool keep_alive = true;
};

struct HttpResponse {
  int status = 200;
  std::string content_type = "text/plain";
  std::map<std::string, std::string> headers;
  IOBuf body;
};

void RegisterHttpProtocol();
void RegisterH2Protocol();  // h2 + gRPC (policy/h2_protocol.cc)
void RegisterThriftProtocol();  // framed TBinary (policy/thrift_protocol.cc)
void RegisterGrpcClientProtocol();  // h2/gRPC client (policy/h2_client.cc)
// Legacy Baidu protocols (policy/legacy_protocols.cc): hulu_pbrpc,
// sofa_pbrpc, nshead (raw-body service, FIFO-correlated).
void RegisterHuluProtocol();
void RegisterSofaProtocol();
void RegisterNsheadProtocol();
void RegisterMongoProtocol();  // server-side (policy/mongo_protocol.cc)
void RegisterEspProtocol();    // client-side (policy/legacy_protocols.cc)
void RegisterNovaProtocol();   // client-side nshead variant (method index in reserved)
void RegisterUbrpcProtocol();  // client-side nshead+mcpack (id-correlated)
void RegisterPublicPbrpcProtocol();  // client-side nshead+PublicPbrpc pb
void RegisterRtmpProtocol();   // server-side (policy/rtmp_protocol.cc)

// Implemented in rpc/builtin/builtin_services.cc; returns true if the path
// matched a builtin page.
bool DispatchBuiltinService(Server* server, const HttpRequest& req, HttpResponse* resp);

// Parses a request head ("METHOD /path HTTP/1.1\r\nHeader: v\r\n...").
// Public for the fuzz harness (tests/fuzz/fuzz_http.cc).
bool ParseHttpHead(const std::string& head, HttpRequest* out);

}  // namespace policy
}  // namespace bam
