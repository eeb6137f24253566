#include "bindings/bind.h"
// Filled in as the fiber / rpc / var layers land.
void bind_var(py::module_& m) { (void)m; }
