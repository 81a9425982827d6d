// Persistent host thread pool — avoids spawning threads per read call
// (the GPU boxes have 256 cores; spawn cost per call would dominate).
#pragma once

#include <atomic>
#include <condition_variable>
#include <functional>
#include <mutex>
#include <thread>
#include <vector>

namespace lakesoul {

class ThreadPool {
 public:
  static ThreadPool& instance() {
    static ThreadPool pool;
    return pool;
  }

  // Run fn(i) for i in [0, n) across the pool; blocks until done.
  // Concurrent callers serialize (each call still uses the whole pool).
  void parallel_for(int64_t n, const std::function<void(int64_t)>& fn) {
    if (n <= 0) return;
    if (n == 1) {
      fn(0);
      return;
    }
    std::lock_guard<std::mutex> job_lock(job_mu_);
    std::unique_lock<std::mutex> lk(mu_);
    job_fn_ = &fn;
    job_n_ = n;
    job_next_ = 0;
    job_done_ = 0;
    gen_++;
    cv_.notify_all();
    // this thread participates too
    lk.unlock();
    work();
    lk.lock();
    done_cv_.wait(lk, [&] { return job_done_ == job_n_; });
    job_fn_ = nullptr;
  }

 private:
  ThreadPool() {
    int n = (int)std::thread::hardware_concurrency();
    if (n > 64) n = 64;
    if (n < 2) n = 2;
    for (int i = 0; i < n - 1; i++) {
      threads_.emplace_back([this] { loop(); });
    }
  }

  ~ThreadPool() {
    {
      std::lock_guard<std::mutex> lk(mu_);
      stop_ = true;
      cv_.notify_all();
    }
    for (auto& t : threads_) t.join();
  }

  void loop() {
    uint64_t seen = 0;
    while (true) {
      const std::function<void(int64_t)>* fn = nullptr;
      {
        std::unique_lock<std::mutex> lk(mu_);
        cv_.wait(lk, [&] { return stop_ || (job_fn_ && gen_ != seen); });
        if (stop_) return;
        seen = gen_;
        fn = job_fn_;
      }
      if (fn) work();
    }
  }

  void work() {
    const std::function<void(int64_t)>* fn;
    {
      std::lock_guard<std::mutex> lk(mu_);
      fn = job_fn_;
    }
    if (!fn) return;
    while (true) {
      int64_t i = job_next_.fetch_add(1);
      if (i >= job_n_) break;
      (*fn)(i);
      int64_t d = job_done_.fetch_add(1) + 1;
      if (d == job_n_) {
        std::lock_guard<std::mutex> lk(mu_);
        done_cv_.notify_all();
      }
    }
  }

  std::vector<std::thread> threads_;
  std::mutex job_mu_;
  std::mutex mu_;
  std::condition_variable cv_, done_cv_;
  const std::function<void(int64_t)>* job_fn_ = nullptr;
  std::atomic<int64_t> job_next_{0}, job_done_{0};
  int64_t job_n_ = 0;
  uint64_t gen_ = 0;
  bool stop_ = false;
};

}  // namespace lakesoul
