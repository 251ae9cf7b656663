// Carousel-style timing wheel (SIGCOMM'17) for paced egress release.
// Role parity: the reference's collective/rdma timing_wheel.h — the
// sender schedules a flow's next transmission instant and the engine
// releases due flows in O(slots touched), instead of re-scanning every
// flow on a coarse timer. Entries beyond the wheel horizon park in an
// overflow list that re-files on each advance.
//
// Thread model: schedule() may be called from any thread holding no
// other wheel state (internal mutex); advance() is called by the
// engine-0 loop. Duplicate schedules of the same flow are fine — the
// consumer re-checks the flow's own release time before sending, so a
// stale release is a cheap no-op.
#pragma once

#include <atomic>
#include <cstdint>
#include <mutex>
#include <utility>
#include <vector>

namespace uccl {

class TimingWheel {
 public:
  explicit TimingWheel(uint64_t gran_ns = 100'000, size_t nslots = 1024)
      : gran_(gran_ns), slots_(nslots) {}

  // Schedule `flow` for release at absolute time `at_ns`.
  void schedule(uint64_t flow, uint64_t at_ns) {
    std::lock_guard<std::mutex> g(mu_);
    if (!inited_) {
      base_ = at_ns;
      inited_ = true;
    }
    uint64_t const horizon = base_ + gran_ * slots_.size();
    if (at_ns >= horizon) {
      overflow_.emplace_back(at_ns, flow);
    } else {
      size_t const idx =
          (at_ns <= base_ ? cursor_
                          : (cursor_ + (at_ns - base_) / gran_) %
                                slots_.size());
      slots_[idx].push_back(flow);
    }
    ++pending_;
  }

  bool empty() const { return pending_.load(std::memory_order_relaxed) == 0; }

  // Release every entry due at `now`; calls due(flow) outside the lock.
  template <typename F>
  void advance(uint64_t now, F&& due) {
    if (empty()) return;
    std::vector<uint64_t> fire;
    {
      std::lock_guard<std::mutex> g(mu_);
      if (!inited_) return;
      while (base_ <= now) {
        auto& s = slots_[cursor_];
        for (uint64_t f : s) fire.push_back(f);
        s.clear();
        base_ += gran_;
        cursor_ = (cursor_ + 1) % slots_.size();
        // refile overflow entries that fell inside the horizon
        uint64_t const horizon = base_ + gran_ * slots_.size();
        for (size_t i = 0; i < overflow_.size();) {
          if (overflow_[i].first < horizon) {
            uint64_t const at = overflow_[i].first;
            size_t const idx =
                (at <= base_ ? cursor_
                             : (cursor_ + (at - base_) / gran_) %
                                   slots_.size());
            slots_[idx].push_back(overflow_[i].second);
            overflow_[i] = overflow_.back();
            overflow_.pop_back();
          } else {
            ++i;
          }
        }
      }
      pending_ -= fire.size();
    }
    for (uint64_t f : fire) due(f);
  }

 private:
  uint64_t gran_;
  std::vector<std::vector<uint64_t>> slots_;
  std::vector<std::pair<uint64_t, uint64_t>> overflow_;  // {at, flow}
  uint64_t base_ = 0;
  size_t cursor_ = 0;
  bool inited_ = false;
  std::atomic<size_t> pending_{0};
  std::mutex mu_;
};

}  // namespace uccl
