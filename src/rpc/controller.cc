#include "rpc/controller.h"

#include <ctype.h>

#include <mutex>

#include "fiber/key.h"

#include "rpc/server.h"

namespace bam {

void Controller::Reset() {
  start_us_ = end_us_ = 0;
  error_code_ = 0;
  error_text_.clear();
  timeout_ms_ = -1;
  backup_request_ms_ = -1;
  max_retry_ = 3;
  retry_count_ = 0;
  log_id_ = 0;
  request_compress_ = COMPRESS_TYPE_NONE;
  response_compress_ = COMPRESS_TYPE_NONE;
  request_attachment_.clear();
  response_attachment_.clear();
  cid_ = 0;
  server_ = nullptr;
  server_socket_ = 0;
  server_cid_ = 0;
  remote_stream_id_ = 0;
  response_stream_id_ = 0;
  auth_context_ = nullptr;
  concurrency_counted_ = false;
  method_gate_entered_ = false;
  call = Call();
}

Controller::~Controller() {}



void* Controller::session_local_data() {
  if (server_ == nullptr || !server_->options().session_local_data_factory) return nullptr;
  SocketUniquePtr sock;
  if (Socket::Address(server_socket_, &sock) != 0) return nullptr;
  void* d = sock->session_local_data.load(std::memory_order_acquire);
  if (d != nullptr) return d;
  void* fresh = server_->options().session_local_data_factory();
  if (fresh == nullptr) return nullptr;
  void* expected = nullptr;
  sock->session_local_deleter = server_->options().session_local_data_deleter;
  if (sock->session_local_data.compare_exchange_strong(expected, fresh,
                                                       std::memory_order_acq_rel)) {
    return fresh;
  }
  // lost the install race
  if (server_->options().session_local_data_deleter)
    server_->options().session_local_data_deleter(fresh);
  return expected;
}

namespace {
struct TldEntry {
  void* data = nullptr;
  Server* server = nullptr;  // returns data to the server pool at fiber exit
};
fiber_key_t g_tld_key;
std::once_flag g_tld_once;
void tld_dtor(void* p) {
  TldEntry* e = (TldEntry*)p;
  if (e->server != nullptr) e->server->ReturnTld(e->data);
  delete e;
}
}  // namespace

void* Controller::thread_local_data() {
  if (server_ == nullptr || !server_->options().thread_local_data_factory) return nullptr;
  std::call_once(g_tld_once, [] { fiber_key_create(&g_tld_key, tld_dtor); });
  TldEntry* e = (TldEntry*)fiber_getspecific(g_tld_key);
  if (e == nullptr) {
    e = new TldEntry;
    e->server = server_;
    e->data = server_->BorrowTld();
    fiber_setspecific(g_tld_key, e);
  }
  return e->data;
}

static std::string lower(const std::string& s) {
  std::string o = s;
  for (char& c : o) c = (char)tolower((unsigned char)c);
  return o;
}

void HttpHeaderExt::SetHeader(const std::string& k, const std::string& v) {
  headers[lower(k)] = v;
}

const std::string* HttpHeaderExt::GetHeader(const std::string& k) const {
  auto it = headers.find(lower(k));
  return it == headers.end() ? nullptr : &it->second;
}

HttpHeaderExt& Controller::http_request() {
  if (http_request_ == nullptr) http_request_.reset(new HttpHeaderExt);
  return *http_request_;
}

HttpHeaderExt& Controller::http_response() {
  if (http_response_ == nullptr) http_response_.reset(new HttpHeaderExt);
  return *http_response_;
}

void Controller::StartCancel() {
  if (cid_ != 0) session_error(cid_, ECANCELED_RPC);
}

}  // namespace bam
