// Minimal futures-based thread pool with stable per-thread ids.
// Capability parity: vendor/thread_pool as used by reference
// src/polisher.cpp:173-179 (thread_identifiers for per-thread scratch lookup).
#pragma once

#include <atomic>
#include <condition_variable>
#include <functional>
#include <future>
#include <memory>
#include <mutex>
#include <queue>
#include <thread>
#include <unordered_map>
#include <vector>

namespace rga {

class ThreadPool {
 public:
  explicit ThreadPool(uint32_t num_threads) : done_(false) {
    num_threads = std::max(1u, num_threads);
    for (uint32_t i = 0; i < num_threads; ++i) {
      workers_.emplace_back([this] { run(); });
    }
    for (uint32_t i = 0; i < num_threads; ++i) {
      thread_ids_[workers_[i].get_id()] = i;
    }
  }

  ThreadPool(const ThreadPool&) = delete;
  ThreadPool& operator=(const ThreadPool&) = delete;

  ~ThreadPool() {
    {
      std::unique_lock<std::mutex> lock(mutex_);
      done_ = true;
    }
    cv_.notify_all();
    for (auto& t : workers_) {
      t.join();
    }
  }

  uint32_t num_threads() const { return static_cast<uint32_t>(workers_.size()); }

  // Stable 0-based id of the calling worker thread; ~0u for foreign threads.
  uint32_t this_thread_id() const {
    auto it = thread_ids_.find(std::this_thread::get_id());
    return it == thread_ids_.end() ? ~0u : it->second;
  }

  template <typename F, typename... Args>
  auto submit(F&& f, Args&&... args) -> std::future<std::invoke_result_t<F, Args...>> {
    using R = std::invoke_result_t<F, Args...>;
    auto task = std::make_shared<std::packaged_task<R()>>(
        std::bind(std::forward<F>(f), std::forward<Args>(args)...));
    std::future<R> result = task->get_future();
    {
      std::unique_lock<std::mutex> lock(mutex_);
      tasks_.emplace([task] { (*task)(); });
    }
    cv_.notify_one();
    return result;
  }

 private:
  void run() {
    while (true) {
      std::function<void()> task;
      {
        std::unique_lock<std::mutex> lock(mutex_);
        cv_.wait(lock, [this] { return done_ || !tasks_.empty(); });
        if (done_ && tasks_.empty()) {
          return;
        }
        task = std::move(tasks_.front());
        tasks_.pop();
      }
      task();
    }
  }

  std::vector<std::thread> workers_;
  std::unordered_map<std::thread::id, uint32_t> thread_ids_;
  std::queue<std::function<void()>> tasks_;
  std::mutex mutex_;
  std::condition_variable cv_;
  bool done_;
};

}  // namespace rga
