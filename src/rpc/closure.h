// brpc_amd: google-style one-shot Closure (parity: protobuf Closure used
// throughout the reference API surface).
#pragma once

#include <utility>

namespace bam {

class Closure {
 public:
  virtual ~Closure() {}
  virtual void Run() = 0;  // typically deletes itself
};

template <typename F>
class FunctionClosure : public Closure {
 public:
  explicit FunctionClosure(F&& f) : f_(std::forward<F>(f)) {}
  void Run() override {
    f_();
    delete this;
  }

 private:
  F f_;
};

template <typename F>
Closure* NewCallback(F&& f) {
  return new FunctionClosure<F>(std::forward<F>(f));
}

// A closure that does nothing on Run (for fire-and-forget RPC).
Closure* DoNothing();

}  // namespace bam
