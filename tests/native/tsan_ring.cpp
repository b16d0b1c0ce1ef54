// ThreadSanitizer stress for the lock-free MPMC ring core
// ("TSAN on the C++ ring-buffer unit tests").
//
//   g++ -O1 -g -std=c++17 -fsanitize=thread -pthread \
//       tests/native/tsan_ring.cpp -o /tmp/tsan_ring && /tmp/tsan_ring
//
// 4 producers x 4 consumers over a 64-cell queue; verifies every value
// delivered exactly once while TSAN watches for data races. Run by
// tests/test_ringbuffer.py::test_ring_core_tsan when g++ supports TSAN.

#include <atomic>
#include <cstdio>
#include <cstdlib>
#include <thread>
#include <vector>

#include "../../torched_impala_amd/csrc/ring_core.h"

int main() {
  constexpr int64_t CAP = 64;
  constexpr int NPROD = 4, NCONS = 4;
  constexpr int64_t PER_PROD = 20000;
  std::vector<int64_t> words(impala_ring::words_needed(CAP));
  impala_ring::init(words.data(), CAP);

  std::vector<std::atomic<int>> seen(NPROD * PER_PROD);
  for (auto& s : seen) s.store(0);
  std::atomic<int64_t> consumed{0};

  std::vector<std::thread> threads;
  for (int p = 0; p < NPROD; ++p) {
    threads.emplace_back([&, p] {
      for (int64_t i = 0; i < PER_PROD; ++i) {
        const int64_t v = p * PER_PROD + i;
        while (!impala_ring::push(words.data(), v)) std::this_thread::yield();
      }
    });
  }
  for (int c = 0; c < NCONS; ++c) {
    threads.emplace_back([&] {
      while (consumed.load(std::memory_order_relaxed) < NPROD * PER_PROD) {
        const int64_t v = impala_ring::pop(words.data());
        if (v < 0) {
          std::this_thread::yield();
          continue;
        }
        if (seen[v].fetch_add(1) != 0) {
          std::fprintf(stderr, "DUPLICATE delivery of %lld\n", (long long)v);
          std::exit(2);
        }
        consumed.fetch_add(1, std::memory_order_relaxed);
      }
    });
  }
  for (auto& t : threads) t.join();
  for (int64_t v = 0; v < NPROD * PER_PROD; ++v) {
    if (seen[v].load() != 1) {
      std::fprintf(stderr, "LOST value %lld\n", (long long)v);
      return 3;
    }
  }
  std::puts("tsan_ring OK");
  return 0;
}
