#include <pybind11/pybind11.h>

#include <sstream>

#include "bindings/bind.h"
#include "var/variable.h"

void bind_var(py::module_& m) {
  auto v = m.def_submodule("var");
  v.def("dump_exposed", [](const std::string& filter) {
    std::ostringstream os;
    bam::var::Variable::dump_exposed(os, filter);
    return os.str();
  }, py::arg("filter") = "");
  v.def("count_exposed", &bam::var::Variable::count_exposed);
  v.def("describe", [](const std::string& name) -> py::object {
    bam::var::Variable* var = bam::var::Variable::find_exposed(name);
    if (var == nullptr) return py::none();
    return py::str(var->get_description());
  });
  // C++-side scenario: adder from many fibers, verify combine.
  v.def("adder_selftest", [](int nfibers, int iters) {
    static bam::var::Adder<int64_t> adder("selftest_adder");
    adder.reset();
    struct Arg { bam::var::Adder<int64_t>* a; int iters; };
    // run on fibers to populate multiple thread cells
    extern int64_t var_adder_fiber_test(bam::var::Adder<int64_t>*, int, int);
    return var_adder_fiber_test(&adder, nfibers, iters);
  }, py::call_guard<py::gil_scoped_release>());
  v.def("latency_recorder_selftest", []() {
    bam::var::LatencyRecorder rec;
    for (int i = 1; i <= 1000; ++i) rec << i;
    bool ok = rec.count() == 1000;
    ok = ok && rec.latency_avg() == 500;
    int64_t p50 = rec.latency_percentile(0.5);
    ok = ok && p50 >= 450 && p50 <= 550;
    int64_t p99 = rec.latency_percentile(0.99);
    ok = ok && p99 >= 950 && p99 <= 1000;
    ok = ok && rec.latency_max() == 1000;
    return ok;
  }, py::call_guard<py::gil_scoped_release>());
  v.def("latency_histogram_selftest", []() {
    // Burst bias regression: 200 slow samples then 100k fast ones. The old
    // last-8192 ring would have evicted every slow sample; the histogram
    // keeps them weighted exactly.
    bam::var::LatencyRecorder rec;
    for (int i = 0; i < 200; ++i) rec << 5000;
    for (int i = 0; i < 100000; ++i) rec << 10;
    bool ok = rec.count() == 100200;
    int64_t p50 = rec.latency_percentile(0.5);
    ok = ok && p50 >= 5 && p50 <= 15;
    int64_t p99 = rec.latency_percentile(0.99);
    ok = ok && p99 <= 20;  // fast samples dominate the 99th
    int64_t p9995 = rec.latency_percentile(0.9995);
    ok = ok && p9995 >= 4000 && p9995 <= 6000;  // the slow tail is NOT lost
    return ok;
  }, py::call_guard<py::gil_scoped_release>());
}
