#include "fiber/gpu_wait.h"

#include <atomic>

#include "base/time.h"
#include "fiber/butex.h"

namespace bam {

namespace {

constexpr int kMaxDev = 16;
constexpr int kKinds = 8;  // mirrors bamhip::kWakeKinds

struct WaitSlot {
  std::atomic<int>* butex = nullptr;
  // True while a wake marker is known to be in flight on the stream. A
  // marker enqueued BEFORE a waiter's ticket kernel can fire early — the
  // wake clears the bit, the waiter re-checks its flag and enqueues a
  // fresh marker, so dedup never loses a wake.
  std::atomic<bool> wake_pending{false};
  std::atomic<int> once{0};
};
WaitSlot g_slots[kMaxDev][kKinds];

std::atomic<int (*)(int, int)> g_request_wake{nullptr};
std::atomic<int64_t> g_parks{0};
std::atomic<int64_t> g_wake_reqs{0};

std::atomic<int>* slot_butex(WaitSlot& s) {
  // Lazy butex creation (butex_create is cheap but not constexpr).
  if (s.butex == nullptr) {
    int expected = 0;
    if (s.once.compare_exchange_strong(expected, 1)) {
      std::atomic<int>* b = butex_create();
      b->store(0, std::memory_order_relaxed);
      __atomic_store_n(&s.butex, b, __ATOMIC_RELEASE);
      s.once.store(2, std::memory_order_release);
    } else {
      while (s.once.load(std::memory_order_acquire) != 2) {
      }
    }
  }
  return __atomic_load_n(&s.butex, __ATOMIC_ACQUIRE);
}

}  // namespace

void gpu_wait_set_request_fn(int (*request_wake)(int, int)) {
  g_request_wake.store(request_wake, std::memory_order_release);
}

int (*gpu_wait_get_request_fn())(int, int) {
  return g_request_wake.load(std::memory_order_acquire);
}

int gpu_fiber_wait_u64(const volatile unsigned long long* flag, uint64_t want,
                       int dev, int kind) {
  if (dev < 0 || dev >= kMaxDev || kind < 0 || kind >= kKinds) return -1;
  int (*req)(int, int) = g_request_wake.load(std::memory_order_acquire);
  if (req == nullptr) return -1;
  WaitSlot& s = g_slots[dev][kind];
  std::atomic<int>* b = slot_butex(s);
  bool parked = false;
  // Hard deadline: if the ticket NEVER lands (faulted kernel, wedged
  // stream), hand control back to the HIP lib's bounded-spin +
  // hipStreamSynchronize fallback, which surfaces the device error
  // instead of waiting forever.
  const int64_t give_up_at = monotonic_time_us() + 10 * 1000000;
  while (*flag < want) {
    if (monotonic_time_us() >= give_up_at) return -1;
    // Capture the butex value BEFORE the final flag check: if the wake
    // fires in between, butex_wait returns EWOULDBLOCK instead of parking
    // past the wake.
    const int v = b->load(std::memory_order_acquire);
    if (*flag >= want) break;
    if (!s.wake_pending.exchange(true, std::memory_order_acq_rel)) {
      if (req(dev, kind) != 0) {
        s.wake_pending.store(false, std::memory_order_release);
        return parked ? 0 : -1;  // no wake stream yet: caller spins
      }
      g_wake_reqs.fetch_add(1, std::memory_order_relaxed);
    }
    // 5 ms backstop: a lost wake (host-callback thread wedged) degrades to
    // a periodic poll instead of a hang.
    int64_t abst = monotonic_time_us() + 5000;
    butex_wait(b, v, &abst);
    parked = true;
  }
  if (parked) g_parks.fetch_add(1, std::memory_order_relaxed);
  return 0;
}

void gpu_fiber_wake(int dev, int kind) {
  if (dev < 0 || dev >= kMaxDev || kind < 0 || kind >= kKinds) return;
  WaitSlot& s = g_slots[dev][kind];
  s.wake_pending.store(false, std::memory_order_release);
  std::atomic<int>* b = slot_butex(s);
  b->fetch_add(1, std::memory_order_release);
  butex_wake_all(b);
}

int64_t gpu_wait_parks() { return g_parks.load(std::memory_order_relaxed); }
int64_t gpu_wait_wake_requests() { return g_wake_reqs.load(std::memory_order_relaxed); }

}  // namespace bam
