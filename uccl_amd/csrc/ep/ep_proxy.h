// CPU proxy for GPU-initiated EP transfers.
//
// Parity role: the reference's ep/src/proxy.cpp run_dual loop — drain the
// device→host command rings, execute the network writes, complete back
// into peer GPU memory. MI355X re-design: the wire is
// the uccl_amd multipath reliable transport (csrc/transport) instead of
// raw ibverbs, the completion write-back is a host→device copy of the
// seq-tagged count (system-visible to the spinning wait kernel), and
// combine returns are shipped host-side off a HIP event (the dispatch
// direction is the GPU-initiated one, via the D2H ring).
#pragma once

#include <hip/hip_runtime.h>

#include <atomic>
#include <condition_variable>
#include <deque>
#include <map>
#include <memory>
#include <mutex>
#include <thread>
#include <vector>

#include "../transport/reliable.h"
#include "d2h_ring.h"
#include "ep_layout.h"

namespace uccl {
namespace ep {

class EpProxy {
 public:
  EpProxy(const EpView& view, void* heap, D2HRing* ring_host, int device);
  ~EpProxy();

  std::string transport_metadata() const { return tp_->metadata(); }
  void set_view(const EpView& v) { v_ = v; }
  // flows[r] for every proxy peer r; establish with connect/accept by rank
  // order convention (lower rank accepts, higher rank connects).
  void establish_flows(const std::vector<std::string>& tp_md,
                       uint32_t proxy_mask);

  // combine returns: ship expert_out rows for every proxy peer once
  // `ready` has been recorded on the compute stream; counts[le*world+src]
  // is the pinned host copy of the dispatch counts.
  void enqueue_combine(void const* expert_out, uint64_t seq,
                       hipEvent_t ready, int const* counts);

  void start();

 private:
  struct WireHdr {
    uint32_t kind;   // 1=dispatch, 2=combine, 3=combine-done,
                     // 4=barrier, 5=atomic-add
    uint32_t seq32;
    uint32_t le;
    uint32_t src;
    uint64_t count;  // barrier: unused; atomic: the add value
    uint64_t aux;    // atomic: target heap byte offset
  };

  // Per-thread pinned staging + non-blocking stream: the ring thread,
  // the combine-tx thread, and every per-peer rx thread run concurrently
  // and must never share host staging (corruption observed at world=4).
  struct Lane {
    void* buf = nullptr;
    void* buf2 = nullptr;  // double buffer: D2H of chunk i+1 overlaps the
                           // transport send of chunk i (batched posting)
    hipStream_t stream = nullptr;
    Lane(int device, size_t bytes);
    ~Lane();
  };

  void ring_loop();
  void rx_loop(int peer);
  void comb_tx_loop();
  void handle_barrier_arrival(Lane& lane, uint64_t seq);
  void write_sync_flag(Lane& lane, int idx, uint64_t seq);
  int num_proxy_peers() const;
  int flow_peer(uint64_t flow) const;
  void ship_rows(Lane& lane, uint64_t flow, WireHdr const& h,
                 void const* dev_rows, uint32_t const* dev_metas_or_null,
                 std::vector<uint32_t> const* host_metas,
                 size_t row_bytes = 0, void const* dev_scales = nullptr,
                 size_t scale_row_bytes = 0);
  void ship_batch(Lane& lane, uint64_t flow, uint32_t seq, uint64_t row0,
                  std::vector<uint32_t> const& les,
                  std::vector<uint32_t> const& cnts, uint64_t total_rows);

  EpView v_;
  void* heap_;
  D2HRing* ring_;
  int device_;
  uint32_t proxy_mask_ = 0;

  std::unique_ptr<transport::TransportEndpoint> tp_;
  std::vector<uint64_t> flows_;
  // serializes multi-message sequences (hdr, payload..., metas) per flow:
  // the ring thread and the combine-tx thread share peer flows
  std::vector<std::unique_ptr<std::timed_mutex>> flow_mu_;

  size_t stage_bytes_ = 0;

  struct CombTask {
    void const* expert_out;
    uint64_t seq;
    hipEvent_t ready;
    int const* counts;  // pinned host buffer; valid to read once `ready`
                        // has completed (the D2H precedes it in-stream)
  };
  std::deque<CombTask> comb_q_;
  bool comb_busy_ = false;  // a task is being shipped right now
  // barrier bookkeeping: arrivals per seq (self + every proxy peer)
  std::mutex bar_mu_;
  std::map<uint64_t, int> bar_seen_;
  std::map<uint64_t, Lane*> bar_lane_;
  // serializes atomic-add read-modify-writes into the heap
  std::mutex atomic_mu_;
  std::mutex mu_;
  std::condition_variable cv_;
  std::atomic<bool> stop_{false};

  std::thread ring_thread_;
  std::thread comb_thread_;
  std::vector<std::thread> rx_threads_;
};

}  // namespace ep
}  // namespace uccl
