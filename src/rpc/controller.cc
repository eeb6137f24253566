#include "rpc/controller.h"

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

}  // namespace bam
