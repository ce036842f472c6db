// Minimal thread pool (reference: scanner/util/thread_pool.h). Used by the
// save workers' multi-sink fan-out and storage prefetch.
#pragma once

#include <functional>
#include <future>
#include <thread>
#include <vector>

#include "queue.h"

namespace sca {

class ThreadPool {
 public:
  explicit ThreadPool(size_t n) {
    for (size_t i = 0; i < n; ++i) {
      threads_.emplace_back([this] {
        while (auto task = tasks_.pop()) {
          (*task)();
        }
      });
    }
  }

  ~ThreadPool() {
    tasks_.close();
    for (auto& t : threads_) t.join();
  }

  template <typename F>
  std::future<void> submit(F&& f) {
    auto task = std::make_shared<std::packaged_task<void()>>(std::forward<F>(f));
    auto fut = task->get_future();
    tasks_.push([task] { (*task)(); });
    return fut;
  }

 private:
  BoundedQueue<std::function<void()>> tasks_;
  std::vector<std::thread> threads_;
};

}  // namespace sca
