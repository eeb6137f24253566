#include "rpc/usercode_pool.h"

#include <condition_variable>
#include <deque>
#include <mutex>
#include <thread>
#include <vector>

#include "base/flags.h"

namespace bam {

BAM_DEFINE_int64(usercode_pool_threads, 8,
                 "pthreads running user callbacks that must not block fiber workers");

namespace {

class UserCodePool {
 public:
  static UserCodePool& instance() {
    static UserCodePool* p = new UserCodePool;
    return *p;
  }

  void submit(std::function<void()> fn) {
    {
      std::lock_guard<std::mutex> lk(mu_);
      ensure_started();
      queue_.push_back(std::move(fn));
    }
    cv_.notify_one();
  }

  int threads() const { return nthreads_; }

 private:
  void ensure_started() {
    if (started_) return;
    started_ = true;
    nthreads_ = (int)FLAG_usercode_pool_threads;
    if (nthreads_ < 1) nthreads_ = 1;
    for (int i = 0; i < nthreads_; ++i) {
      std::thread([this] { run(); }).detach();
    }
  }

  void run() {
    for (;;) {
      std::function<void()> fn;
      {
        std::unique_lock<std::mutex> lk(mu_);
        cv_.wait(lk, [this] { return !queue_.empty(); });
        fn = std::move(queue_.front());
        queue_.pop_front();
      }
      fn();
    }
  }

  std::mutex mu_;
  std::condition_variable cv_;
  std::deque<std::function<void()>> queue_;
  bool started_ = false;
  int nthreads_ = 0;
};

}  // namespace

void SubmitUserCode(std::function<void()> fn) { UserCodePool::instance().submit(std::move(fn)); }

int UserCodePoolThreads() { return UserCodePool::instance().threads(); }

}  // namespace bam
