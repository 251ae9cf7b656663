// Lock-free latency percentile recorder.
//
// Parity role: the reference's include/util/latency.h percentile
// recorder. Log-spaced ns buckets (2 per octave, ~41% resolution) so
// record() is a couple of instructions and percentile queries need no
// stored samples.

#pragma once

#include <array>
#include <atomic>
#include <cmath>
#include <cstdint>

namespace uccl {

class LatencyHist {
 public:
  static constexpr int kBuckets = 128;  // 2 buckets/octave, covers ns..hours

  void record_ns(uint64_t ns) {
    b_[bucket(ns)].fetch_add(1, std::memory_order_relaxed);
    count_.fetch_add(1, std::memory_order_relaxed);
  }
  void record_us(double us) {
    record_ns(static_cast<uint64_t>(us * 1000.0));
  }

  uint64_t count() const { return count_.load(std::memory_order_relaxed); }

  // approximate value (ns) at percentile p in [0,100]
  uint64_t percentile_ns(double p) const {
    uint64_t const n = count();
    if (!n) return 0;
    uint64_t target = static_cast<uint64_t>(p / 100.0 * (n - 1)) + 1;
    uint64_t seen = 0;
    for (int i = 0; i < kBuckets; ++i) {
      seen += b_[i].load(std::memory_order_relaxed);
      if (seen >= target) return upper_bound(i);
    }
    return upper_bound(kBuckets - 1);
  }
  double percentile_us(double p) const { return percentile_ns(p) / 1000.0; }

  void reset() {
    for (auto& b : b_) b.store(0, std::memory_order_relaxed);
    count_.store(0, std::memory_order_relaxed);
  }

 private:
  static int bucket(uint64_t ns) {
    if (ns < 2) return 0;
    int const lz = __builtin_clzll(ns);
    int const octave = 63 - lz;
    // second bit below the MSB halves the octave
    int const half = (ns >> (octave - 1)) & 1;
    int const idx = octave * 2 + half;
    return idx < kBuckets ? idx : kBuckets - 1;
  }
  static uint64_t upper_bound(int idx) {
    int const octave = idx / 2;
    uint64_t const lo = 1ull << octave;
    return idx % 2 ? (lo | (lo >> 1)) + (lo >> 1) : lo + (lo >> 1);
  }

  std::array<std::atomic<uint64_t>, kBuckets> b_{};
  std::atomic<uint64_t> count_{0};
};

}  // namespace uccl
