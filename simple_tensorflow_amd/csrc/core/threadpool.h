// Simple work-stealing-free thread pool (capability analog of the reference's
// core/lib/core/threadpool.h; we do not need Eigen's non-blocking pool for the
// GPU-centric hot path — GPU work is stream-async and CPU kernels are few).
#pragma once

#include <condition_variable>
#include <deque>
#include <functional>
#include <mutex>
#include <thread>
#include <vector>

namespace stf {

class ThreadPool {
 public:
  explicit ThreadPool(int num_threads, const std::string& name = "stf") {
    for (int i = 0; i < num_threads; ++i) {
      threads_.emplace_back([this]() { WorkerLoop(); });
    }
  }
  ~ThreadPool() {
    {
      std::lock_guard<std::mutex> l(mu_);
      shutdown_ = true;
    }
    cv_.notify_all();
    for (auto& t : threads_) t.join();
  }

  void Schedule(std::function<void()> fn) {
    {
      std::lock_guard<std::mutex> l(mu_);
      queue_.push_back(std::move(fn));
    }
    cv_.notify_one();
  }

  int NumThreads() const { return (int)threads_.size(); }

 private:
  void WorkerLoop() {
    for (;;) {
      std::function<void()> fn;
      {
        std::unique_lock<std::mutex> l(mu_);
        cv_.wait(l, [this]() { return shutdown_ || !queue_.empty(); });
        if (shutdown_ && queue_.empty()) return;
        fn = std::move(queue_.front());
        queue_.pop_front();
      }
      fn();
    }
  }

  std::mutex mu_;
  std::condition_variable cv_;
  std::deque<std::function<void()>> queue_;
  std::vector<std::thread> threads_;
  bool shutdown_ = false;
};

}  // namespace stf
