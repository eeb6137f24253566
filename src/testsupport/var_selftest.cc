#include "var/variable.h"

#include <vector>

#include "fiber/fiber.h"

namespace bam {
struct VarArg {
  var::Adder<int64_t>* a;
  int iters;
};
static void var_fiber_fn(void* raw) {
  VarArg* a = (VarArg*)raw;
  for (int i = 0; i < a->iters; ++i) {
    (*a->a) << 1;
    if ((i & 255) == 0) fiber_yield();
  }
}
}  // namespace bam

int64_t var_adder_fiber_test(bam::var::Adder<int64_t>* a, int nfibers, int iters) {
  bam::VarArg arg{a, iters};
  std::vector<bam::fiber_t> tids(nfibers);
  for (int i = 0; i < nfibers; ++i) bam::fiber_start_background(&tids[i], bam::var_fiber_fn, &arg);
  for (int i = 0; i < nfibers; ++i) bam::fiber_join(tids[i]);
  return a->get_value();
}
