// brpc_amd: C++20 coroutine adapter for asynchronous RPC.
// Parity: reference brpc/coroutine.h (Awaitable / co_await on async
// calls). Header-only and guarded: the core library builds as C++17; user
// translation units compiled with -std=c++20 get `co_await
// AwaitRpc(channel, method, request)` which suspends the coroutine and
// resumes it from the RPC's done closure (on the usercode/fiber
// completion path — the coroutine body must not block a fiber worker for
// long, same rule as any done callback).
#pragma once

#if defined(__cpp_impl_coroutine) || (defined(__cplusplus) && __cplusplus >= 202002L)

#include <coroutine>
#include <string>
#include <utility>

#include "rpc/channel.h"
#include "rpc/closure.h"
#include "rpc/controller.h"

namespace bam {
namespace co {

// Result of one awaited RPC.
struct RpcResult {
  int error_code = 0;
  std::string error_text;
  IOBuf response;
};

// Awaitable wrapping one asynchronous CallMethod.
class RpcAwaitable {
 public:
  RpcAwaitable(ChannelBase* channel, std::string full_method, IOBuf request)
      : channel_(channel), full_method_(std::move(full_method)),
        request_(std::move(request)) {}

  bool await_ready() const noexcept { return false; }

  void await_suspend(std::coroutine_handle<> h) {
    handle_ = h;
    Closure* done = NewCallback([this] {
      result_.error_code = cntl_.ErrorCode();
      result_.error_text = cntl_.ErrorText();
      result_.response.swap(response_);
      handle_.resume();
    });
    channel_->CallMethod(full_method_, &cntl_, &request_, &response_, done);
  }

  RpcResult await_resume() noexcept { return std::move(result_); }

 private:
  ChannelBase* channel_;
  std::string full_method_;
  IOBuf request_;
  IOBuf response_;
  Controller cntl_;
  RpcResult result_;
  std::coroutine_handle<> handle_;
};

inline RpcAwaitable AwaitRpc(ChannelBase* channel, std::string full_method, IOBuf request) {
  return RpcAwaitable(channel, std::move(full_method), std::move(request));
}

// Minimal fire-and-forget task type for driving awaited RPC chains.
struct Task {
  struct promise_type {
    Task get_return_object() { return {}; }
    std::suspend_never initial_suspend() noexcept { return {}; }
    std::suspend_never final_suspend() noexcept { return {}; }
    void return_void() {}
    void unhandled_exception() { std::terminate(); }
  };
};

}  // namespace co
}  // namespace bam

#endif  // C++20
