// Persistent host thread pool — avoids spawning threads per read call
// (the GPU boxes have 256 cores; spawn cost per call would dominate).
// Safe under concurrent parallel_for callers: jobs are reference-counted
// and a task index is only dispatched while its own job is live, so a
// straggler worker can never touch a completed job's (stack-allocated)
// function object.
#pragma once

#include <atomic>
#include <condition_variable>
#include <functional>
#include <memory>
#include <mutex>
#include <thread>
#include <vector>

namespace lakesoul {

class ThreadPool {
 public:
  static ThreadPool& instance() {
    static ThreadPool* pool = new ThreadPool();  // leaked: workers may
    return *pool;  // outlive static destruction order at process exit
  }

  // Run fn(i) for i in [0, n) across the pool; blocks until done.
  // Concurrent callers serialize (each still uses the whole pool).
  void parallel_for(int64_t n, const std::function<void(int64_t)>& fn) {
    if (n <= 0) return;
    if (n == 1) {
      fn(0);
      return;
    }
    std::lock_guard<std::mutex> job_lock(job_mu_);
    auto job = std::make_shared<Job>();
    job->fn = &fn;
    job->n = n;
    {
      std::lock_guard<std::mutex> lk(mu_);
      cur_ = job;
      cv_.notify_all();
    }
    run_job(*job);  // calling thread participates
    {
      std::unique_lock<std::mutex> lk(mu_);
      done_cv_.wait(lk, [&] { return job->done.load() == job->n; });
      cur_.reset();
    }
  }

 private:
  struct Job {
    const std::function<void(int64_t)>* fn = nullptr;
    std::atomic<int64_t> next{0};
    std::atomic<int64_t> done{0};
    int64_t n = 0;
  };

  ThreadPool() {
    int n = (int)std::thread::hardware_concurrency();
    if (n > 64) n = 64;
    if (n < 2) n = 2;
    for (int i = 0; i < n - 1; i++) threads_.emplace_back([this] { loop(); });
  }

  void loop() {
    std::shared_ptr<Job> last;
    while (true) {
      std::shared_ptr<Job> j;
      {
        std::unique_lock<std::mutex> lk(mu_);
        cv_.wait(lk, [&] { return cur_ && cur_ != last; });
        j = cur_;
      }
      last = j;
      run_job(*j);
    }
  }

  void run_job(Job& j) {
    while (true) {
      int64_t i = j.next.fetch_add(1);
      if (i >= j.n) break;
      (*j.fn)(i);
      if (j.done.fetch_add(1) + 1 == j.n) {
        std::lock_guard<std::mutex> lk(mu_);
        done_cv_.notify_all();
      }
    }
  }

  std::vector<std::thread> threads_;
  std::mutex job_mu_;
  std::mutex mu_;
  std::condition_variable cv_, done_cv_;
  std::shared_ptr<Job> cur_;
};

}  // namespace lakesoul
