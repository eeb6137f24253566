// brpc_amd: user-code backup pool — a bounded pthread pool that runs user
// callbacks which must not execute on fiber workers (parity: reference
// details/usercode_backup_pool.h, usercode_in_pthread). The Python service
// bindings route handlers here (the GIL and fiber stacks don't mix).
#pragma once

#include <functional>

namespace bam {

// Enqueues fn to the pool (threads started lazily; count from
// -usercode_pool_threads). Never blocks the caller.
void SubmitUserCode(std::function<void()> fn);

int UserCodePoolThreads();

}  // namespace bam
