// Lock-free bounded MPMC index queue over a shared-memory int64 tensor.
//
// The actor->learner transport control plane (SURVEY.md §1b layer 3;
// BASELINE.json:5 "lock-free pinned-host trajectory ring buffer"): slot
// payloads live in big shared arenas (ringbuffer.py); free/full slot
// INDICES flow through two of these queues. Classic Vyukov bounded MPMC:
// per-cell monotonic sequence numbers give ABA-safety and tear-detection
// (SURVEY.md §7 hard-part 2).
//
// Memory layout inside the caller-provided int64 tensor (shared mmap):
//   [0] enqueue_pos  [1] dequeue_pos  [2] capacity(pow2)  [3] epoch/abort
//   [4 + 2*i] cell_i.seq   [5 + 2*i] cell_i.data     i in [0, capacity)
//
// All cross-process synchronisation is std::atomic<int64_t> on the shared
// mapping (lock-free on x86-64). Blocking pops release the GIL and spin
// with sched_yield + short sleeps; an abort flag unblocks all waiters.

#include <torch/extension.h>

#include <atomic>
#include <chrono>
#include <cstdint>
#include <thread>

namespace {

struct Hdr {
  std::atomic<int64_t> enqueue_pos;
  std::atomic<int64_t> dequeue_pos;
  int64_t capacity;
  std::atomic<int64_t> abort_flag;
};

struct Cell {
  std::atomic<int64_t> seq;
  int64_t data;
};

static_assert(sizeof(Hdr) == 4 * sizeof(int64_t), "layout");
static_assert(sizeof(Cell) == 2 * sizeof(int64_t), "layout");

inline Hdr* hdr(torch::Tensor& t) {
  TORCH_CHECK(t.scalar_type() == torch::kInt64 && t.is_contiguous(),
              "ring tensor must be contiguous int64");
  return reinterpret_cast<Hdr*>(t.data_ptr<int64_t>());
}

inline Cell* cells(Hdr* h) { return reinterpret_cast<Cell*>(h + 1); }

int64_t words_needed(int64_t capacity) { return 4 + 2 * capacity; }

void ring_init(torch::Tensor t, int64_t capacity) {
  TORCH_CHECK((capacity & (capacity - 1)) == 0 && capacity > 0,
              "capacity must be a power of two");
  TORCH_CHECK(t.numel() >= words_needed(capacity), "tensor too small");
  Hdr* h = hdr(t);
  h->enqueue_pos.store(0, std::memory_order_relaxed);
  h->dequeue_pos.store(0, std::memory_order_relaxed);
  h->capacity = capacity;
  h->abort_flag.store(0, std::memory_order_relaxed);
  Cell* c = cells(h);
  for (int64_t i = 0; i < capacity; ++i) {
    c[i].seq.store(i, std::memory_order_relaxed);
    c[i].data = -1;
  }
  std::atomic_thread_fence(std::memory_order_seq_cst);
}

bool ring_push(torch::Tensor t, int64_t value) {
  Hdr* h = hdr(t);
  Cell* c = cells(h);
  const int64_t mask = h->capacity - 1;
  int64_t pos = h->enqueue_pos.load(std::memory_order_relaxed);
  for (;;) {
    Cell* cell = &c[pos & mask];
    int64_t seq = cell->seq.load(std::memory_order_acquire);
    int64_t dif = seq - pos;
    if (dif == 0) {
      if (h->enqueue_pos.compare_exchange_weak(pos, pos + 1,
                                               std::memory_order_relaxed)) {
        cell->data = value;
        cell->seq.store(pos + 1, std::memory_order_release);
        return true;
      }
    } else if (dif < 0) {
      return false;  // full
    } else {
      pos = h->enqueue_pos.load(std::memory_order_relaxed);
    }
  }
}

int64_t ring_pop(torch::Tensor t) {
  Hdr* h = hdr(t);
  Cell* c = cells(h);
  const int64_t mask = h->capacity - 1;
  int64_t pos = h->dequeue_pos.load(std::memory_order_relaxed);
  for (;;) {
    Cell* cell = &c[pos & mask];
    int64_t seq = cell->seq.load(std::memory_order_acquire);
    int64_t dif = seq - (pos + 1);
    if (dif == 0) {
      if (h->dequeue_pos.compare_exchange_weak(pos, pos + 1,
                                               std::memory_order_relaxed)) {
        int64_t v = cell->data;
        cell->seq.store(pos + mask + 1, std::memory_order_release);
        return v;
      }
    } else if (dif < 0) {
      return -1;  // empty
    } else {
      pos = h->dequeue_pos.load(std::memory_order_relaxed);
    }
  }
}

// Blocking pop; releases the GIL. timeout_ms < 0 => wait forever.
// Returns -1 on timeout, -2 on abort.
int64_t ring_pop_wait(torch::Tensor t, double timeout_ms) {
  Hdr* h = hdr(t);
  auto deadline = std::chrono::steady_clock::now() +
                  std::chrono::duration<double, std::milli>(
                      timeout_ms < 0 ? 1e12 : timeout_ms);
  int spins = 0;
  for (;;) {
    int64_t v = ring_pop(t);
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
  hdr(t)->abort_flag.store(1, std::memory_order_release);
}

int64_t ring_size_approx(torch::Tensor t) {
  Hdr* h = hdr(t);
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
