// brpc_amd: GPU-event waits integrated with the fiber scheduler — the
// "scheduler rebuilt around HIP streams" requirement (BASELINE north star;
// parity concept: reference bthread/butex.cpp:675 park semantics fused with
// stream completion).
//
// Model: every async GPU leg (staging gathers, async uploads, span copies,
// collectives) publishes a monotonically increasing 64-bit ticket into a
// pinned host flag from device code. A fiber that must wait for ticket T:
//   1. spins briefly (µs-scale batches; handled inside libbrpc_hip.so),
//   2. then parks HERE on a butex keyed by (device, stream-kind),
//   3. a hipLaunchHostFunc marker — enqueued on the same stream, therefore
//      ordered after the ticket's kernel — wakes the butex.
// The worker pthread is free while the fiber is parked: other RPC fibers
// run. This replaces the round-1 unbounded host spin in the socket write
// path (hip/gpu_runtime.hip direct_wait).
#pragma once

#include <stdint.h>

namespace bam {

// Parks the calling fiber until *flag >= want. Returns 0 once the
// condition holds; -1 if parking is unavailable (no wake stream registered
// for (dev, kind), or wake-marker enqueue failed) — the caller (HIP lib)
// then falls back to spinning. Callable from non-fiber pthreads too (they
// park on the underlying futex instead of a fiber switch).
int gpu_fiber_wait_u64(const volatile unsigned long long* flag, uint64_t want,
                       int dev, int kind);

// Wake callback: bumps the (dev, kind) butex and wakes all parked waiters.
// Runs on a HIP host-callback thread.
void gpu_fiber_wake(int dev, int kind);

// Registers the HIP lib's wake-marker enqueue hook
// (bam_gpu_request_wake). Done by gpu_loader at dlopen time; tests may
// install a fake to exercise park/wake without a GPU.
void gpu_wait_set_request_fn(int (*request_wake)(int dev, int kind));
int (*gpu_wait_get_request_fn())(int dev, int kind);  // for save/restore in tests

// Diagnostics (exposed as /vars gpu_wait_*): fiber parks taken and wake
// markers enqueued since process start.
int64_t gpu_wait_parks();
int64_t gpu_wait_wake_requests();

}  // namespace bam
