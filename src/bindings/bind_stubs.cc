#include "bindings/bind.h"
// Filled in as the fiber / rpc / var layers land.
