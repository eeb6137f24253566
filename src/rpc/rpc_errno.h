// brpc_amd: RPC error codes (numeric parity with reference brpc/errno.proto
// so wire-level error_code values are interchangeable).
#pragma once

namespace bam {

enum RpcError {
  ENOSERVICE = 1001,    // service not found
  ENOMETHOD = 1002,     // method not found
  EREQUEST = 1003,      // bad request
  ERPCAUTH = 1004,      // auth failed
  ETOOMANYFAILS = 1005, // too many sub-channel failures (ParallelChannel)
  EPCHANFINISH = 1006,  // ParallelChannel finished
  EBACKUPREQUEST = 1007,// trigger backup request
  ERPCTIMEDOUT = 1008,  // RPC deadline exceeded
  EFAILEDSOCKET = 1009, // broken socket
  EHTTP = 1010,         // http error
  EOVERCROWDED = 1011,  // too many buffered writes
  ERTMPPUBLISHABLE = 1012,
  ERTMPCREATESTREAM = 1013,
  EEOF = 1014,
  EUNUSED = 1015,
  ESSL_ERR = 1016,
  ECANCELED_RPC = 1017,  // StartCancel (client-side cancellation)
  EINTERNAL = 2001,     // server internal error
  ERESPONSE = 2002,     // bad response
  ELOGOFF = 2003,       // server stopping
  ELIMIT = 2004,        // concurrency limit reached
};

// alias used in errno positions
const int EOVERCROWDED_ERRNO = EOVERCROWDED;

const char* rpc_strerror(int code);

}  // namespace bam
