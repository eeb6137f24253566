#include "bindings/bind.h"

PYBIND11_MODULE(_core, m) {
  m.doc() = "brpc_amd core: MI355X-native RPC runtime (bRPC capability rebuild)";
  bind_base(m);
  bind_fiber(m);
  bind_rpc(m);
  bind_var(m);
  bind_gpu(m);
  bind_rpc_combo(m);
  bind_rpc_stream(m);
  bind_snappy(m);
  bind_api(m);
  bind_redis(m);
  bind_util(m);
  bind_memcache(m);
  bind_json2pb(m);
  bind_thrift(m);
  bind_codecs(m);
  bind_comm(m);
  bind_proto(m);
  bind_hpack(m);
}
