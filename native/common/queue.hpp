// queue.hpp — bounded MPMC queue with close semantics.
//
// The MI355X-native equivalent of the reference's
// tokio::sync::mpsc::channel::<ScaleKind>(100) (reference main.rs:284):
// the producer (query task) blocks when the consumer (scale task) falls
// behind; closing the queue drains-and-stops the consumer.
#pragma once

#include <condition_variable>
#include <deque>
#include <mutex>
#include <optional>

namespace qx {

template <typename T>
class BoundedQueue {
public:
  explicit BoundedQueue(size_t capacity) : cap_(capacity) {}

  // Blocks while full. Returns false if the queue was closed.
  bool push(T v) {
    std::unique_lock<std::mutex> lock(mu_);
    cv_push_.wait(lock, [&] { return closed_ || q_.size() < cap_; });
    if (closed_) return false;
    q_.push_back(std::move(v));
    cv_pop_.notify_one();
    return true;
  }

  // Blocks while empty. Returns nullopt once closed AND drained.
  std::optional<T> pop() {
    std::unique_lock<std::mutex> lock(mu_);
    cv_pop_.wait(lock, [&] { return closed_ || !q_.empty(); });
    if (q_.empty()) return std::nullopt;
    T v = std::move(q_.front());
    q_.pop_front();
    cv_push_.notify_one();
    return v;
  }

  void close() {
    std::lock_guard<std::mutex> lock(mu_);
    closed_ = true;
    cv_push_.notify_all();
    cv_pop_.notify_all();
  }

  size_t size() const {
    std::lock_guard<std::mutex> lock(mu_);
    return q_.size();
  }

private:
  mutable std::mutex mu_;
  std::condition_variable cv_push_, cv_pop_;
  std::deque<T> q_;
  size_t cap_;
  bool closed_ = false;
};

}  // namespace qx
