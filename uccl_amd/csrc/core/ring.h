// Lockless bounded rings.
//
// Parity role: the reference's DPDK-style jring (include/util/jring.h) —
// the universal app<->engine queue. Two shapes:
//   * SpscRing: single-producer/single-consumer, head/tail
//     acquire-release only (the common engine-thread pairing);
//   * MpmcRing: bounded multi-producer/multi-consumer with per-slot
//     sequence numbers (Vyukov's classic bounded queue) — one CAS per
//     producer, no locks, FIFO per producer.
// Capacity must be a power of two. T must be trivially copyable.

#pragma once

#include <atomic>
#include <cstddef>
#include <cstdint>
#include <stdexcept>
#include <type_traits>
#include <vector>

namespace uccl {

template <typename T>
class SpscRing {
  static_assert(std::is_trivially_copyable<T>::value, "POD payloads only");

 public:
  explicit SpscRing(size_t capacity_pow2)
      : mask_(capacity_pow2 - 1), buf_(capacity_pow2) {
    if (capacity_pow2 == 0 || (capacity_pow2 & mask_) != 0)
      throw std::invalid_argument("capacity must be a power of two");
  }

  bool push(T const& v) {
    size_t const h = head_.load(std::memory_order_relaxed);
    if (h - tail_.load(std::memory_order_acquire) > mask_) return false;
    buf_[h & mask_] = v;
    head_.store(h + 1, std::memory_order_release);
    return true;
  }

  bool pop(T* out) {
    size_t const t = tail_.load(std::memory_order_relaxed);
    if (t == head_.load(std::memory_order_acquire)) return false;
    *out = buf_[t & mask_];
    tail_.store(t + 1, std::memory_order_release);
    return true;
  }

  size_t size() const {
    return head_.load(std::memory_order_acquire) -
           tail_.load(std::memory_order_acquire);
  }

 private:
  alignas(64) std::atomic<size_t> head_{0};
  alignas(64) std::atomic<size_t> tail_{0};
  size_t const mask_;
  std::vector<T> buf_;
};

template <typename T>
class MpmcRing {
  static_assert(std::is_trivially_copyable<T>::value, "POD payloads only");

 public:
  explicit MpmcRing(size_t capacity_pow2)
      : mask_(capacity_pow2 - 1), cells_(capacity_pow2) {
    if (capacity_pow2 == 0 || (capacity_pow2 & mask_) != 0)
      throw std::invalid_argument("capacity must be a power of two");
    for (size_t i = 0; i < capacity_pow2; ++i)
      cells_[i].seq.store(i, std::memory_order_relaxed);
  }

  bool push(T const& v) {
    size_t pos = head_.load(std::memory_order_relaxed);
    for (;;) {
      Cell& c = cells_[pos & mask_];
      size_t const seq = c.seq.load(std::memory_order_acquire);
      intptr_t const d = static_cast<intptr_t>(seq) -
                         static_cast<intptr_t>(pos);
      if (d == 0) {
        if (head_.compare_exchange_weak(pos, pos + 1,
                                        std::memory_order_relaxed))
          break;
      } else if (d < 0) {
        return false;  // full
      } else {
        pos = head_.load(std::memory_order_relaxed);
      }
    }
    Cell& c = cells_[pos & mask_];
    c.val = v;
    c.seq.store(pos + 1, std::memory_order_release);
    return true;
  }

  bool pop(T* out) {
    size_t pos = tail_.load(std::memory_order_relaxed);
    for (;;) {
      Cell& c = cells_[pos & mask_];
      size_t const seq = c.seq.load(std::memory_order_acquire);
      intptr_t const d = static_cast<intptr_t>(seq) -
                         static_cast<intptr_t>(pos + 1);
      if (d == 0) {
        if (tail_.compare_exchange_weak(pos, pos + 1,
                                        std::memory_order_relaxed))
          break;
      } else if (d < 0) {
        return false;  // empty
      } else {
        pos = tail_.load(std::memory_order_relaxed);
      }
    }
    Cell& c = cells_[pos & mask_];
    *out = c.val;
    c.seq.store(pos + mask_ + 1, std::memory_order_release);
    return true;
  }

  size_t size_approx() const {
    return head_.load(std::memory_order_relaxed) -
           tail_.load(std::memory_order_relaxed);
  }

 private:
  struct Cell {
    std::atomic<size_t> seq;
    T val;
  };
  alignas(64) std::atomic<size_t> head_{0};
  alignas(64) std::atomic<size_t> tail_{0};
  size_t const mask_;
  std::vector<Cell> cells_;
};

}  // namespace uccl
