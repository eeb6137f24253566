#include "fiber/timer_thread.h"

#include <condition_variable>
#include <mutex>
#include <queue>
#include <thread>
#include <vector>

#include "base/resource_pool.h"
#include "base/time.h"

namespace bam {

namespace {

struct TimerMeta {
  std::atomic<uint32_t> version{1};
  void (*fn)(void*, void*);
  void* a;
  void* b;
};

struct HeapEntry {
  int64_t when;
  ResourceId rid;
  uint32_t ver;
  bool operator>(const HeapEntry& o) const { return when > o.when; }
};

class TimerThread {
 public:
  static TimerThread& instance() {
    static TimerThread* t = new TimerThread;  // leaked singleton
    return *t;
  }

  TimerId add(int64_t when_us, void (*fn)(void*, void*), void* a, void* b) {
    ResourceId rid;
    TimerMeta* m = get_resource<TimerMeta>(&rid);
    uint32_t ver = m->version.load(std::memory_order_relaxed);
    m->fn = fn;
    m->a = a;
    m->b = b;
    {
      std::lock_guard<std::mutex> lk(mu_);
      heap_.push(HeapEntry{when_us, rid, ver});
      if (when_us < earliest_) {
        earliest_ = when_us;
        cv_.notify_one();
      }
    }
    return ((uint64_t)ver << 32) | (rid + 1);
  }

  int del(TimerId id) {
    if (id == 0) return -1;
    ResourceId rid = (uint32_t)(id & 0xffffffffu) - 1;
    uint32_t ver = (uint32_t)(id >> 32);
    TimerMeta* m = address_resource<TimerMeta>(rid);
    if (m == nullptr) return -1;
    uint32_t expected = ver;
    if (m->version.compare_exchange_strong(expected, ver + 1, std::memory_order_acq_rel)) {
      return 0;  // cancelled; heap entry is skipped + recycled when popped
    }
    return -1;
  }

 private:
  TimerThread() : earliest_(INT64_MAX) {
    std::thread([this] { run(); }).detach();
  }

  void run() {
    std::vector<HeapEntry> due;
    std::unique_lock<std::mutex> lk(mu_);
    for (;;) {
      int64_t now = monotonic_time_us();
      while (!heap_.empty() && heap_.top().when <= now) {
        due.push_back(heap_.top());
        heap_.pop();
      }
      earliest_ = heap_.empty() ? INT64_MAX : heap_.top().when;
      if (!due.empty()) {
        lk.unlock();
        for (const HeapEntry& e : due) {
          TimerMeta* m = address_resource<TimerMeta>(e.rid);
          uint32_t expected = e.ver;
          if (m->version.compare_exchange_strong(expected, e.ver + 1,
                                                 std::memory_order_acq_rel)) {
            m->fn(m->a, m->b);  // we own the run
          }
          return_resource<TimerMeta>(e.rid);
        }
        due.clear();
        lk.lock();
        continue;
      }
      if (heap_.empty()) {
        cv_.wait(lk);
      } else {
        cv_.wait_for(lk, std::chrono::microseconds(heap_.top().when - now));
      }
    }
  }

  std::mutex mu_;
  std::condition_variable cv_;
  std::priority_queue<HeapEntry, std::vector<HeapEntry>, std::greater<HeapEntry>> heap_;
  int64_t earliest_;
};

}  // namespace

TimerId timer_add(int64_t abstime_us, void (*fn)(void*, void*), void* a, void* b) {
  return TimerThread::instance().add(abstime_us, fn, a, b);
}

int timer_delete(TimerId id) { return TimerThread::instance().del(id); }

}  // namespace bam
