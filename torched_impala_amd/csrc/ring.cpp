// Lock-free bounded MPMC index queue over a shared-memory int64 tensor.
//
// The actor->learner transport control plane (SURVEY.md §1b layer 3;
// BASELINE.json:5 "lock-free pinned-host trajectory ring buffer"): slot
// payloads live in big shared arenas (ringbuffer.py); free/full slot
// INDICES flow through two of these queues. The queue core (Vyukov
// bounded MPMC with per-cell monotonic sequence numbers for ABA-safety
// and tear detection — SURVEY.md §7 hard-part 2) lives in ring_core.h,
// torch-free, and is additionally stress-tested under ThreadSanitizer
// (tests/native/tsan_ring.cpp).
//
// All cross-process synchronisation is std::atomic<int64_t> on the
// shared mmap (lock-free on x86-64). Blocking pops release the GIL and
// spin with sched_yield + short sleeps; an abort flag unblocks waiters.

#include <torch/extension.h>

#include <chrono>
#include <thread>

#include "ring_core.h"

namespace {

using impala_ring::Hdr;

inline int64_t* words(torch::Tensor& t) {
  TORCH_CHECK(t.scalar_type() == torch::kInt64 && t.is_contiguous(),
              "ring tensor must be contiguous int64");
  return t.data_ptr<int64_t>();
}

int64_t words_needed(int64_t capacity) {
  return impala_ring::words_needed(capacity);
}

void ring_init(torch::Tensor t, int64_t capacity) {
  TORCH_CHECK((capacity & (capacity - 1)) == 0 && capacity > 0,
              "capacity must be a power of two");
  TORCH_CHECK(t.numel() >= words_needed(capacity), "tensor too small");
  impala_ring::init(words(t), capacity);
}

bool ring_push(torch::Tensor t, int64_t value) {
  return impala_ring::push(words(t), value);
}

int64_t ring_pop(torch::Tensor t) { return impala_ring::pop(words(t)); }

// Blocking pop; releases the GIL. timeout_ms < 0 => wait forever.
// Returns -1 on timeout, -2 on abort.
int64_t ring_pop_wait(torch::Tensor t, double timeout_ms) {
  int64_t* w = words(t);
  Hdr* h = impala_ring::hdr_of(w);
  auto deadline = std::chrono::steady_clock::now() +
                  std::chrono::duration<double, std::milli>(
                      timeout_ms < 0 ? 1e12 : timeout_ms);
  int spins = 0;
  for (;;) {
    int64_t v = impala_ring::pop(w);
    if (v >= 0) return v;
    if (h->abort_flag.load(std::memory_order_acquire)) return -2;
    if (timeout_ms >= 0 && std::chrono::steady_clock::now() > deadline)
      return -1;
    if (++spins < 64) {
      std::this_thread::yield();
    } else {
      std::this_thread::sleep_for(std::chrono::microseconds(200));
    }
  }
}

void ring_abort(torch::Tensor t) {
  impala_ring::hdr_of(words(t))
      ->abort_flag.store(1, std::memory_order_release);
}

int64_t ring_size_approx(torch::Tensor t) {
  Hdr* h = impala_ring::hdr_of(words(t));
  return h->enqueue_pos.load(std::memory_order_relaxed) -
         h->dequeue_pos.load(std::memory_order_relaxed);
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "lock-free MPMC slot-index queue over shared memory";
  m.def("words_needed", &words_needed);
  m.def("init", &ring_init);
  m.def("push", &ring_push);
  m.def("pop", &ring_pop);
  m.def("pop_wait", &ring_pop_wait,
        py::call_guard<py::gil_scoped_release>());
  m.def("abort", &ring_abort);
  m.def("size_approx", &ring_size_approx);
}
