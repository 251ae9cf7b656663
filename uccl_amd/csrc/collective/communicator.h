// Host-side Communicator for the xGMI collective engine.
//
// Role parity: the comm-lifecycle half of the reference's lite-collective
// NCCL drop-in (experimental/lite/lite-collective/nccl/nccl.cu:1455
// ncclCommInitRank + collective entry points), re-designed for MI355X:
// symmetric HIP-IPC heaps per rank, algorithm selection by message size
// (LL packet / one-shot fullmesh / two-shot RS+AG push), host-sequenced
// send/recv with per-destination staging slots and ack credits.
//
// Bootstrap (exchanging the 64-byte IPC handles + device ids) is left to
// the caller — in Python, torch.distributed (gloo or any backend) or any
// out-of-band channel; this class is transport-agnostic about rendezvous,
// the same split as the reference's TCP-OOB bootstrap.
#pragma once

#include <hip/hip_runtime.h>

#include <array>
#include <atomic>
#include <cstdint>
#include <string>
#include <thread>
#include <vector>

#include "kernels.h"
#include "layout.h"

namespace uccl {

struct IpcInfo {
  hipIpcMemHandle_t handle;
  int device;
  int pid;
};

class Communicator {
 public:
  Communicator(int rank, int world, int device, size_t heap_bytes);
  ~Communicator();

  Communicator(const Communicator&) = delete;
  Communicator& operator=(const Communicator&) = delete;

  // Serialized IpcInfo for this rank, to be exchanged out-of-band.
  std::string handle_bytes() const;

  // Install all ranks' serialized IpcInfo blobs (ordered by rank).
  void connect(const std::vector<std::string>& all_handles);

  int rank() const { return rank_; }
  int world() const { return world_; }
  int device() const { return device_; }
  size_t scratch_capacity_bytes() const { return scratch_cap_; }

  // --- symmetric (zero-copy) user region ---------------------------------
  // Bump-allocates `bytes` in the registered symmetric region; every rank
  // must perform the same allocations in the same order (standard
  // symmetric-heap contract). Collectives on tensors inside this region
  // skip all staging copies.
  size_t sym_alloc(size_t bytes);
  void* heap_base() const { return heap_; }
  size_t user_region_offset() const { return user_off_; }
  size_t user_region_capacity() const { return user_cap_; }
  bool is_symmetric_ptr(void const* p) const {
    auto const u = reinterpret_cast<uintptr_t>(p);
    auto const b = reinterpret_cast<uintptr_t>(heap_);
    return u >= b + user_off_ && u < b + heap_bytes_;
  }

  // All ops are asynchronous on `stream` and in-place where natural.
  void all_reduce(void* data, size_t count, Dtype dt, hipStream_t stream,
                  RedOp op = RedOp::kSum);
  void all_gather(void* out, void const* in, size_t count_per_rank, Dtype dt,
                  hipStream_t stream);
  void reduce_scatter(void* out, void const* in, size_t count_per_rank,
                      Dtype dt, hipStream_t stream, RedOp op = RedOp::kSum);
  void broadcast(void* data, size_t count, Dtype dt, int root,
                 hipStream_t stream);
  void all_to_all(void* out, void const* in, size_t count_per_rank, Dtype dt,
                  hipStream_t stream);
  void send(void const* data, size_t bytes, int dst, hipStream_t stream);
  void recv(void* data, size_t bytes, int src, hipStream_t stream);
  void barrier(hipStream_t stream);

  struct OpStats {
    uint64_t calls = 0;
    uint64_t bytes = 0;
  };
  // per-op host-side tallies (parity: the reference's #ifdef STATS
  // per-engine counters, collective/rdma/transport.cc:1797)
  std::array<OpStats, 8> const& stats() const { return stats_; }

 private:
  CommView view(uint64_t seq) const;
  uint64_t next_seq() {
    uint64_t s = seq_;
    seq_ += 2;
    return s;
  }

  int rank_, world_, device_;
  size_t heap_bytes_, scratch_cap_;
  void* heap_ = nullptr;
  std::array<void*, kMaxRanks> peers_{};
  std::array<bool, kMaxRanks> ipc_opened_{};
  uint64_t seq_ = 2;  // flag regions start zeroed; first live seq must be >0
  size_t user_off_ = 0;   // symmetric user region offset in the heap
  size_t user_cap_ = 0;
  size_t user_bump_ = 0;
  // per-destination send / per-source recv sequence counters (p2p channel)
  std::array<uint64_t, kMaxRanks> send_seq_{};
  std::array<uint64_t, kMaxRanks> recv_seq_{};
  bool connected_ = false;

  // optional periodic stats dump (UCCL_ENGINE_STATS=1), the analog of the
  // reference's per-engine stats thread (transport.cc:1797)
  std::thread stats_thread_;
  std::atomic<bool> stop_stats_{false};

  std::array<OpStats, 8> stats_{};
  void tally(int op, size_t bytes) {
    ++stats_[op].calls;
    stats_[op].bytes += bytes;
  }

  // thresholds (env-tunable)
  size_t ll_threshold_;
  size_t oneshot_threshold_;

  static constexpr int kCollCh = 0;
  static constexpr int kP2PDataCh = 2;
  static constexpr int kP2PAckCh = 3;
};

}  // namespace uccl
