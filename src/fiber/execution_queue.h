// brpc_amd: ExecutionQueue — MPSC serialized executor.
// Parity: reference bthread/execution_queue.h: producers push tasks
// lock-free (CAS stack); a single consumer fiber drains them in order
// through the user handler. Used by streams, LB feedback, media paths.
#pragma once

#include <atomic>
#include <functional>
#include <vector>

#include "base/logging.h"
#include "fiber/butex.h"
#include "fiber/fiber.h"

namespace bam {

template <typename T>
class ExecutionQueue {
 public:
  // Receives tasks in submission order (possibly batched).
  typedef std::function<void(std::vector<T>& batch)> Handler;

  ExecutionQueue() : idle_butex_(butex_create()) {}
  ~ExecutionQueue() {
    stop();
    join();
    butex_destroy(idle_butex_);
  }

  int start(Handler h) {
    handler_ = std::move(h);
    started_ = true;
    return 0;
  }

  // Thread/fiber-safe. Returns 0, or EINVAL after stop().
  int execute(T task) { return push(&head_, std::move(task)); }

  // High-priority lane (≙ reference TASK_OPTIONS_URGENT,
  // bthread/execution_queue.h:78): urgent tasks run before anything
  // still waiting in the normal lane, preserving order within each lane.
  int execute_urgent(T task) { return push(&urgent_head_, std::move(task)); }

  void stop() { stopped_.store(true, std::memory_order_release); }

  // Waits until all submitted tasks ran and the consumer exited.
  void join() {
    for (;;) {
      int v = idle_butex_->load(std::memory_order_acquire);
      if (inflight_.load(std::memory_order_acquire) == 0 &&
          head_.load(std::memory_order_acquire) == nullptr &&
          urgent_head_.load(std::memory_order_acquire) == nullptr)
        return;
      butex_wait(idle_butex_, v, nullptr);
    }
  }

 private:
  struct Node {
    Node* next;
    T value;
  };

  int push(std::atomic<Node*>* lane, T task) {
    if (!started_ || stopped_.load(std::memory_order_acquire)) return EINVAL;
    Node* node = new Node{nullptr, std::move(task)};
    Node* old = lane->load(std::memory_order_relaxed);
    do {
      node->next = old;
    } while (!lane->compare_exchange_weak(old, node, std::memory_order_release));
    if (events_.fetch_add(1, std::memory_order_acq_rel) == 0) {
      inflight_.fetch_add(1, std::memory_order_acq_rel);
      fiber_t th;
      if (fiber_start_background(&th, consumer_entry, this) != 0) consumer_entry(this);
    }
    return 0;
  }

  // Pops a lane into submission order and appends to *batch.
  void drain_lane(std::atomic<Node*>* lane, std::vector<T>* batch) {
    Node* h = lane->exchange(nullptr, std::memory_order_acq_rel);
    Node* fifo = nullptr;
    while (h != nullptr) {
      Node* nx = h->next;
      h->next = fifo;
      fifo = h;
      h = nx;
    }
    while (fifo != nullptr) {
      batch->push_back(std::move(fifo->value));
      Node* nx = fifo->next;
      delete fifo;
      fifo = nx;
    }
  }

  static void consumer_entry(void* raw) {
    ((ExecutionQueue*)raw)->consume();
  }

  void consume() {
    for (;;) {
      std::vector<T> batch;
      drain_lane(&urgent_head_, &batch);  // urgent lane jumps the queue
      drain_lane(&head_, &batch);
      if (!batch.empty() && handler_) handler_(batch);
      int v = events_.load(std::memory_order_acquire);
      if (head_.load(std::memory_order_acquire) == nullptr &&
          urgent_head_.load(std::memory_order_acquire) == nullptr &&
          events_.compare_exchange_strong(v, 0, std::memory_order_acq_rel)) {
        break;
      }
    }
    inflight_.fetch_sub(1, std::memory_order_acq_rel);
    idle_butex_->fetch_add(1, std::memory_order_release);
    butex_wake_all(idle_butex_);
  }

  std::atomic<Node*> head_{nullptr};
  std::atomic<Node*> urgent_head_{nullptr};
  std::atomic<int> events_{0};
  std::atomic<int> inflight_{0};
  std::atomic<bool> stopped_{false};
  bool started_ = false;
  Handler handler_;
  std::atomic<int>* idle_butex_;
};

}  // namespace bam
