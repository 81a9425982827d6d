// Persistent host thread pool — avoids spawning threads per read call
// (the GPU boxes have 256 cores; spawn cost per call would dominate).
//
// Supports CONCURRENT parallel_for callers: the scan pipeline prefetches
// several units at once and each fetch issues its own job; workers drain
// every active job, so overlapping fetches genuinely overlap instead of
// serializing on a single-job mutex. Jobs are reference-counted; a task
// index is only dispatched while its job is live, so a straggler worker
// can never touch a completed job's (stack-allocated) function object.
#pragma once

#include <algorithm>
#include <atomic>
#include <cstdio>
#include <cstdlib>
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
  void parallel_for(int64_t n, const std::function<void(int64_t)>& fn) {
    if (n <= 0) return;
    if (n == 1) {
      fn(0);
      return;
    }
    auto job = std::make_shared<Job>();
    job->fn = &fn;
    job->n = n;
    {
      std::lock_guard<std::mutex> lk(mu_);
      active_.push_back(job);
      gen_++;
      cv_.notify_all();
    }
    run_job(*job);  // calling thread participates in its own job
    {
      std::unique_lock<std::mutex> lk(mu_);
      done_cv_.wait(lk, [&] { return job->done.load() == job->n; });
      active_.erase(std::remove(active_.begin(), active_.end(), job),
                    active_.end());
    }
  }

 private:
  struct Job {
    const std::function<void(int64_t)>* fn = nullptr;
    std::atomic<int64_t> next{0};
    std::atomic<int64_t> done{0};
    int64_t n = 0;
  };

  static int effective_cpus() {
    // the GPU boxes report 256 cores but the job cgroup may cap the
    // quota far lower — oversubscribing the pool thrashes. cgroup v2:
    // /sys/fs/cgroup/cpu.max = "<quota> <period>" or "max <period>".
    int hw = (int)std::thread::hardware_concurrency();
    if (const char* env = std::getenv("LAKESOUL_POOL_THREADS")) {
      int v = atoi(env);
      if (v > 0) return v;
    }
    FILE* f = std::fopen("/sys/fs/cgroup/cpu.max", "r");
    if (f) {
      char buf[64] = {0};
      if (std::fgets(buf, sizeof(buf), f)) {
        long quota, period;
        if (std::sscanf(buf, "%ld %ld", &quota, &period) == 2 && quota > 0 &&
            period > 0) {
          int q = (int)(quota / period);
          if (q > 0 && q < hw) hw = q;
        }
      }
      std::fclose(f);
    }
    // one process per GPU shares the node's quota: divide by the rank
    // count so 8 ranks don't each spin up a full-quota pool and thrash
    if (const char* ws = std::getenv("WORLD_SIZE")) {
      int w = atoi(ws);
      if (w > 1) hw = hw / w;
    }
    return hw;
  }

  ThreadPool() {
    int n = effective_cpus();
    if (n > 64) n = 64;
    if (n < 2) n = 2;
    for (int i = 0; i < n - 1; i++) threads_.emplace_back([this] { loop(); });
  }

  void loop() {
    uint64_t seen = 0;
    while (true) {
      std::vector<std::shared_ptr<Job>> jobs;
      {
        std::unique_lock<std::mutex> lk(mu_);
        cv_.wait(lk, [&] { return gen_.load() != seen; });
        seen = gen_.load();
        jobs = active_;
      }
      for (auto& j : jobs) run_job(*j);
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
  std::mutex mu_;
  std::condition_variable cv_, done_cv_;
  std::vector<std::shared_ptr<Job>> active_;
  std::atomic<uint64_t> gen_{0};
};

}  // namespace lakesoul
