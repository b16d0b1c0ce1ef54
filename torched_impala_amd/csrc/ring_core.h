// Lock-free bounded MPMC queue core (Vyukov) over a caller-provided
// int64 word array — torch-free so it can be unit/TSAN-tested
// standalone (tests/native/tsan_ring.cpp) and bound into the extension
// (ring.cpp).
//
// Word layout:
//   [0] enqueue_pos  [1] dequeue_pos  [2] capacity(pow2)  [3] abort
//   [4 + 2*i] cell_i.seq   [5 + 2*i] cell_i.data   i in [0, capacity)
#pragma once

#include <atomic>
#include <cstdint>

namespace impala_ring {

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

inline int64_t words_needed(int64_t capacity) { return 4 + 2 * capacity; }

inline Hdr* hdr_of(int64_t* words) { return reinterpret_cast<Hdr*>(words); }
inline Cell* cells_of(Hdr* h) { return reinterpret_cast<Cell*>(h + 1); }

inline void init(int64_t* words, int64_t capacity) {
  Hdr* h = hdr_of(words);
  h->enqueue_pos.store(0, std::memory_order_relaxed);
  h->dequeue_pos.store(0, std::memory_order_relaxed);
  h->capacity = capacity;
  h->abort_flag.store(0, std::memory_order_relaxed);
  Cell* c = cells_of(h);
  for (int64_t i = 0; i < capacity; ++i) {
    c[i].seq.store(i, std::memory_order_relaxed);
    c[i].data = -1;
  }
  std::atomic_thread_fence(std::memory_order_seq_cst);
}

inline bool push(int64_t* words, int64_t value) {
  Hdr* h = hdr_of(words);
  Cell* c = cells_of(h);
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

inline int64_t pop(int64_t* words) {
  Hdr* h = hdr_of(words);
  Cell* c = cells_of(h);
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

}  // namespace impala_ring
