// ThreadPool race check — built with -fsanitize=thread by
// tests/test_threadpool.py::test_tsan_clean (the
// reference has no sanitizer CI; here the pool's job-queue handoff is
// TSAN-verified since it carries every scan's host decode).
#include <atomic>
#include <cstdio>
#include <vector>

#include "../thread_pool.h"

using lakesoul::ThreadPool;

int main() {
  std::atomic<long> sum{0};
  std::vector<long> out(100000, 0);
  for (int round = 0; round < 20; round++) {
    ThreadPool::instance().parallel_for((long)out.size(), [&](long i) {
      out[i] = i * 2;
      sum.fetch_add(1, std::memory_order_relaxed);
    });
  }
  // concurrent jobs from multiple submitter threads
  std::vector<std::thread> subs;
  std::atomic<long> total{0};
  for (int t = 0; t < 4; t++) {
    subs.emplace_back([&] {
      for (int r = 0; r < 10; r++) {
        ThreadPool::instance().parallel_for(5000, [&](long i) {
          total.fetch_add(i, std::memory_order_relaxed);
        });
      }
    });
  }
  for (auto& s : subs) s.join();
  long expect = 4L * 10 * (5000L * 4999 / 2);
  if (total.load() != expect) {
    std::printf("FAIL total=%ld expect=%ld\n", total.load(), expect);
    return 1;
  }
  std::printf("OK sum=%ld\n", sum.load());
  return 0;
}
