#include <pybind11/pybind11.h>

#include "fiber/fiber.h"
#include "fiber/gpu_wait.h"
#include "bindings/bind.h"

namespace bam {
namespace selftest {
int64_t start_join_test(int nfibers, int iters);
bool urgent_test();
int64_t usleep_test(int64_t us);
bool butex_wake_test();
bool butex_timeout_test();
int64_t mutex_test(int nfibers, int iters);
bool countdown_test(int n);
bool semaphore_test();
int rwlock_test(int nreaders, int nwriters, int iters);
bool timer_test();
bool fiber_key_test();
bool fiber_interrupt_test(std::string* err);
bool execution_queue_urgent_test(std::string* err);
bool gpu_wait_selftest();
int64_t fd_wait_selftest();
bool stack_class_selftest();
}  // namespace selftest
}  // namespace bam

void bind_fiber(py::module_& m) {
  auto f = m.def_submodule("fiber");
  // Release the GIL: these block on fiber completion.
  f.def("start_join_test", &bam::selftest::start_join_test,
        py::call_guard<py::gil_scoped_release>());
  f.def("urgent_test", &bam::selftest::urgent_test, py::call_guard<py::gil_scoped_release>());
  f.def("usleep_test", &bam::selftest::usleep_test, py::call_guard<py::gil_scoped_release>());
  f.def("butex_wake_test", &bam::selftest::butex_wake_test,
        py::call_guard<py::gil_scoped_release>());
  f.def("butex_timeout_test", &bam::selftest::butex_timeout_test,
        py::call_guard<py::gil_scoped_release>());
  f.def("mutex_test", &bam::selftest::mutex_test, py::call_guard<py::gil_scoped_release>());
  f.def("semaphore_test", &bam::selftest::semaphore_test,
        py::call_guard<py::gil_scoped_release>());
  f.def("rwlock_test", &bam::selftest::rwlock_test,
        py::call_guard<py::gil_scoped_release>());
  f.def("countdown_test", &bam::selftest::countdown_test,
        py::call_guard<py::gil_scoped_release>());
  f.def("timer_test", &bam::selftest::timer_test, py::call_guard<py::gil_scoped_release>());
  f.def("key_test", &bam::selftest::fiber_key_test, py::call_guard<py::gil_scoped_release>());
  f.def("execution_queue_urgent_test", []() {
    std::string err;
    bool ok;
    {
      py::gil_scoped_release rel;
      ok = bam::selftest::execution_queue_urgent_test(&err);
    }
    return py::make_tuple(ok, err);
  });
  f.def("interrupt_test", []() {
    std::string err;
    bool ok;
    {
      py::gil_scoped_release rel;
      ok = bam::selftest::fiber_interrupt_test(&err);
    }
    return py::make_tuple(ok, err);
  });
  f.def("gpu_wait_test", &bam::selftest::gpu_wait_selftest,
        py::call_guard<py::gil_scoped_release>());
  f.def("fd_wait_test", &bam::selftest::fd_wait_selftest,
        py::call_guard<py::gil_scoped_release>());
  f.def("stack_class_test", &bam::selftest::stack_class_selftest,
        py::call_guard<py::gil_scoped_release>());
  f.def("gpu_wait_parks", &bam::gpu_wait_parks);
  f.def("gpu_wait_wake_requests", &bam::gpu_wait_wake_requests);
  f.def("concurrency", &bam::fiber_get_concurrency);
  f.def("set_concurrency", &bam::fiber_set_concurrency);
  f.def("count_created", &bam::fiber_count_created);
  f.def("count_active", &bam::fiber_count_active);
}
