#include "rpc/closure.h"
#include "rpc/load_balancer.h"
#include "rpc/rpc_errno.h"

namespace bam {

namespace {
class DoNothingClosure : public Closure {
 public:
  void Run() override {}  // static singleton; never deleted
};
}  // namespace

Closure* DoNothing() {
  static DoNothingClosure c;
  return &c;
}

const char* rpc_strerror(int code) {
  switch (code) {
    case ENOSERVICE: return "ENOSERVICE: service not found";
    case ENOMETHOD: return "ENOMETHOD: method not found";
    case EREQUEST: return "EREQUEST: bad request";
    case ERPCAUTH: return "EAUTH: authentication failed";
    case ETOOMANYFAILS: return "ETOOMANYFAILS: too many sub-channel failures";
    case ERPCTIMEDOUT: return "ERPCTIMEDOUT: deadline exceeded";
    case EFAILEDSOCKET: return "EFAILEDSOCKET: broken socket";
    case EHTTP: return "EHTTP: http error";
    case EOVERCROWDED: return "EOVERCROWDED: too many buffered writes";
    case EINTERNAL: return "EINTERNAL: server internal error";
    case ERESPONSE: return "ERESPONSE: bad response";
    case ELOGOFF: return "ELOGOFF: server stopping";
    case ELIMIT: return "ELIMIT: concurrency limit reached";
    default: return "";
  }
}

__attribute__((weak)) LoadBalancer* CreateExtendedLoadBalancer(const std::string&) {
  return nullptr;
}

}  // namespace bam
