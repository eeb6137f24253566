#pragma once
#include <pybind11/pybind11.h>
namespace py = pybind11;

void bind_base(py::module_& m);
void bind_fiber(py::module_& m);
void bind_rpc(py::module_& m);
void bind_var(py::module_& m);
void bind_gpu(py::module_& m);
void bind_rpc_combo(py::module_& m);
void bind_rpc_stream(py::module_& m);
void bind_snappy(py::module_& m);
void bind_api(py::module_& m);
void bind_redis(py::module_& m);
void bind_util(py::module_& m);
void bind_memcache(py::module_& m);
void bind_json2pb(py::module_& m);
void bind_thrift(py::module_& m);
void bind_codecs(py::module_& m);
void bind_comm(py::module_& m);
void bind_proto(py::module_& m);
void bind_hpack(py::module_& m);
