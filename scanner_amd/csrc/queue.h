// Bounded blocking MPMC queue connecting pipeline stages (reference:
// scanner/util/queue.h). Simple mutex+condvar implementation — stage threads
// block on it by design (backpressure), so lock-free buys nothing here.
#pragma once

#include <condition_variable>
#include <deque>
#include <mutex>
#include <optional>

#include "common.h"

namespace sca {

template <typename T>
class BoundedQueue {
 public:
  explicit BoundedQueue(size_t max_size = 0) : max_size_(max_size) {}

  // Returns false if the queue was closed.
  bool push(T item) {
    std::unique_lock<std::mutex> l(mu_);
    not_full_.wait(l, [&] {
      return closed_ || max_size_ == 0 || q_.size() < max_size_;
    });
    if (closed_) return false;
    q_.push_back(std::move(item));
    not_empty_.notify_one();
    return true;
  }

  // Blocks until an item is available or the queue is closed+drained.
  std::optional<T> pop() {
    std::unique_lock<std::mutex> l(mu_);
    not_empty_.wait(l, [&] { return closed_ || !q_.empty(); });
    if (q_.empty()) return std::nullopt;
    T item = std::move(q_.front());
    q_.pop_front();
    not_full_.notify_one();
    return item;
  }

  std::optional<T> try_pop() {
    std::lock_guard<std::mutex> l(mu_);
    if (q_.empty()) return std::nullopt;
    T item = std::move(q_.front());
    q_.pop_front();
    not_full_.notify_one();
    return item;
  }

  void close() {
    std::lock_guard<std::mutex> l(mu_);
    closed_ = true;
    not_empty_.notify_all();
    not_full_.notify_all();
  }

  size_t size() const {
    std::lock_guard<std::mutex> l(mu_);
    return q_.size();
  }

  bool closed() const {
    std::lock_guard<std::mutex> l(mu_);
    return closed_;
  }

 private:
  size_t max_size_;
  mutable std::mutex mu_;
  std::condition_variable not_empty_, not_full_;
  std::deque<T> q_;
  bool closed_ = false;
};

}  // namespace sca
