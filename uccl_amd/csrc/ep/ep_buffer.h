#pragma once

#include <hip/hip_runtime.h>

#include <array>
#include <string>
#include <vector>

#include <memory>

#include "ep_layout.h"

namespace uccl {
namespace ep {

class EpProxy;
struct D2HRing;

class EpBuffer {
 public:
  EpBuffer(int rank, int world, int device, int num_experts, int topk,
           int hidden, int max_tokens, int elem_size, bool use_fp8 = false,
           bool with_normal = true);
  ~EpBuffer();
  EpBuffer(const EpBuffer&) = delete;

  std::string handle_bytes() const;
  void connect(const std::vector<std::string>& handles);

  // x: [num_tokens, hidden] (elem_size); topk_idx: [num_tokens, topk] i64.
  // out_counts: device int32 [local_experts, world].
  void dispatch(void const* x, int64_t const* topk_idx, int num_tokens,
                int* out_counts, hipStream_t stream);

  // Phase-split dispatch (DeepEP SEND|RECV split, internode_ll.cu:62):
  // send launches plan/copy/publish; recv launches the count wait —
  // callers overlap compute between the two (recv-hook support).
  // reuse_plan=true skips the plan/prefix kernels and replays the
  // previous dispatch's compaction lists (DeepEP cached-handle mode,
  // intranode.cu:150) — only valid when topk_idx is unchanged.
  void dispatch_send(void const* x, int64_t const* topk_idx,
                     int num_tokens, bool reuse_plan, hipStream_t stream);
  void dispatch_recv(int* out_counts, hipStream_t stream);

  // expert_out: [local_experts, world*max_tokens, hidden];
  // out: [num_tokens, hidden]; topk_w: [num_tokens, topk] f32.
  void combine(void const* expert_out, void* out, int64_t const* topk_idx,
               float const* topk_w, hipStream_t stream);

  // Phase-split combine: send returns expert outputs to source cells and
  // signals; recv waits for all ranks' returns then reduces.
  void combine_send(void const* expert_out, hipStream_t stream);
  void combine_recv(void* out, int64_t const* topk_idx,
                    float const* topk_w, hipStream_t stream);

  // true when a plan from a prior dispatch can be replayed (same
  // topk shape; the heap's plan scratch is untouched since)
  bool plan_cached() const { return last_num_tokens_ >= 0; }

  // Normal (rank-granular) mode: DeepEP HT dispatch/combine. Each token
  // ships once per destination rank with its topk row + weights; the
  // receiver runs its local experts over the per-source rows and
  // combine returns one processed row per received token, reduced at
  // the source over contributing ranks.
  void nrm_dispatch_send(void const* x, int64_t const* topk_idx,
                         float const* topk_w, int num_tokens,
                         hipStream_t stream);
  void nrm_dispatch_recv(int* out_counts, hipStream_t stream);
  void nrm_combine_send(void const* x, hipStream_t stream);
  void nrm_combine_recv(void* out, int64_t const* topk_idx,
                        hipStream_t stream);
  void* nrm_x_base() const {
    return static_cast<char*>(heap_) + v_.off_nrm_x;
  }
  void* nrm_meta_base() const {
    return static_cast<char*>(heap_) + v_.off_nrm_meta;
  }
  void* nrm_topk_base() const {
    return static_cast<char*>(heap_) + v_.off_nrm_topk;
  }
  void* nrm_w_base() const {
    return static_cast<char*>(heap_) + v_.off_nrm_w;
  }

  // Proxy-path synchronization commands (parity: the reference proxy's
  // ATOMIC/BARRIER/QUIET cmd types, ep/src/proxy.cpp:1629-1718 + D2H
  // cmd switch). All are GPU-initiated via the D2H ring and complete by
  // a device-visible flag the paired wait kernel spins on.
  void barrier(hipStream_t stream);       // across all proxy peers
  void quiet(hipStream_t stream);         // prior ring cmds fully shipped
  void atomic_add(int dst, uint64_t value, hipStream_t stream);
  uint64_t read_sync_word(int idx);       // test/diagnostic accessor
  // raises (with the decoded wait id) if any device wait timed out and
  // recorded a diagnostic instead of wedging/trapping the queue
  void check_error();

  const EpView& view() const { return v_; }
  void* recv_x_ptr() const {
    return static_cast<char*>(heap_) + v_.off_disp_x;
  }
  void* recv_scale_ptr() const {
    return static_cast<char*>(heap_) + v_.off_disp_scale;
  }
  void* recv_meta_ptr() const {
    return static_cast<char*>(heap_) + v_.off_disp_meta;
  }
  int rank() const { return rank_; }
  int world() const { return world_; }
  int device() const { return device_; }

  bool proxy_enabled() const { return proxy_ != nullptr; }

 private:
  int rank_, world_, device_;
  void* heap_ = nullptr;
  EpView v_{};
  std::array<bool, kMaxRanks> ipc_opened_{};
  bool connected_ = false;
  int last_num_tokens_ = -1;
  uint64_t sync_seq_ = 0;
  int nrm_last_tokens_ = -1;

  // proxy path (internode / forced): D2H command ring + CPU proxy over
  // the reliable transport
  std::unique_ptr<EpProxy> proxy_;
  D2HRing* ring_host_ = nullptr;
  // pinned copies of dispatch counts, rotated per generation so a
  // combine in flight never reads a buffer the next dispatch_recv is
  // overwriting
  static constexpr int kCountSlots = 4;
  int* host_counts_[kCountSlots] = {};
};

}  // namespace ep
}  // namespace uccl
