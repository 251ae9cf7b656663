#include "communicator.h"

#include <hip/hip_runtime.h>
#include <unistd.h>

#include <cstring>
#include <sstream>

#include "../core/env.h"
#include "../core/log.h"

namespace uccl {

static inline unsigned all_mask(int world) {
  return world >= 32 ? 0xffffffffu : ((1u << world) - 1u);
}

Communicator::Communicator(int rank, int world, int device, size_t heap_bytes)
    : rank_(rank), world_(world), device_(device) {
  UCCL_CHECK(world >= 1 && world <= kMaxRanks)
      << "world=" << world << " (intranode engine supports up to "
      << kMaxRanks << ")";
  UCCL_CHECK(rank >= 0 && rank < world) << "bad rank " << rank;
  size_t const base_bytes =
      heap_bytes ? heap_bytes
                 : static_cast<size_t>(env_int("UCCL_SYM_HEAP_MB", 1184)) *
                       (1 << 20);
  UCCL_CHECK(base_bytes > kScratchAOffset + (4 << 20))
      << "heap too small: " << base_bytes;
  // per-parity capacity: the A/B scratch regions are each split into two
  // parity halves (see layout.h CommView comment)
  scratch_cap_ = (scratch_capacity(base_bytes) / 2) & ~size_t(255);
  // symmetric user region (zero-copy collectives) appended after scratch
  user_cap_ = static_cast<size_t>(env_int("UCCL_SYM_USER_MB", 512))
              << 20;
  user_off_ = kScratchAOffset + 4 * scratch_cap_;
  heap_bytes_ = user_off_ + user_cap_;
  ll_threshold_ = env_int("UCCL_LL_THRESHOLD", 32 * 1024);
  if (ll_threshold_ > kLLMaxMsgBytes) ll_threshold_ = kLLMaxMsgBytes;
  oneshot_threshold_ = env_int("UCCL_ONESHOT_THRESHOLD", 2 * 1024 * 1024);

  UCCL_CHECK_HIP(hipSetDevice(device_));
  UCCL_CHECK_HIP(hipMalloc(&heap_, heap_bytes_));
  // zero flags + LL regions so stale values can never match a live seq
  UCCL_CHECK_HIP(hipMemset(heap_, 0, kScratchAOffset));
  UCCL_CHECK_HIP(hipDeviceSynchronize());
  peers_.fill(nullptr);
  peers_[rank_] = heap_;
  UCCL_LOG_INFO << "Communicator rank " << rank_ << "/" << world_
                << " heap=" << (heap_bytes_ >> 20) << "MB scratch_cap="
                << (scratch_cap_ >> 20) << "MB";
  if (env_bool("UCCL_ENGINE_STATS", false)) {
    stats_thread_ = std::thread([this] {
      static const char* names[8] = {"ar", "ag", "rs", "bc",
                                     "a2a", "snd", "rcv", "bar"};
      while (!stop_stats_) {
        for (int i = 0; i < 20 && !stop_stats_; ++i)
          usleep(100 * 1000);
        std::ostringstream os;
        for (int i = 0; i < 8; ++i)
          if (stats_[i].calls)
            os << names[i] << "=" << stats_[i].calls << "/"
               << (stats_[i].bytes >> 20) << "MB ";
        if (!os.str().empty())
          UCCL_LOG_INFO << "[stats rank " << rank_ << "] " << os.str();
      }
    });
  }
}

Communicator::~Communicator() {
  stop_stats_ = true;
  if (stats_thread_.joinable()) stats_thread_.join();
  for (int r = 0; r < world_; ++r) {
    if (ipc_opened_[r] && peers_[r]) (void)hipIpcCloseMemHandle(peers_[r]);
  }
  if (heap_) (void)hipFree(heap_);
}

std::string Communicator::handle_bytes() const {
  IpcInfo info{};
  UCCL_CHECK_HIP(hipIpcGetMemHandle(const_cast<hipIpcMemHandle_t*>(&info.handle),
                                    heap_));
  info.device = device_;
  info.pid = static_cast<int>(getpid());
  return std::string(reinterpret_cast<const char*>(&info), sizeof(info));
}

void Communicator::connect(const std::vector<std::string>& all_handles) {
  UCCL_CHECK(static_cast<int>(all_handles.size()) == world_)
      << "expected " << world_ << " handles, got " << all_handles.size();
  UCCL_CHECK_HIP(hipSetDevice(device_));
  int const my_pid = static_cast<int>(getpid());
  for (int r = 0; r < world_; ++r) {
    if (r == rank_) continue;
    IpcInfo info{};
    UCCL_CHECK(all_handles[r].size() == sizeof(IpcInfo))
        << "bad handle blob from rank " << r;
    memcpy(&info, all_handles[r].data(), sizeof(info));
    if (info.device != device_) {
      int can = 0;
      UCCL_CHECK_HIP(hipDeviceCanAccessPeer(&can, device_, info.device));
      UCCL_CHECK(can) << "GPU " << device_ << " cannot access peer GPU "
                      << info.device << " (rank " << r << ")";
      hipError_t e = hipDeviceEnablePeerAccess(info.device, 0);
      UCCL_CHECK(e == hipSuccess || e == hipErrorPeerAccessAlreadyEnabled)
          << "enable peer access " << device_ << "->" << info.device << ": "
          << hipGetErrorString(e);
    }
    UCCL_CHECK(info.pid != my_pid)
        << "rank " << r << " is the same process; one process per rank "
        << "is required for HIP IPC";
    void* p = nullptr;
    UCCL_CHECK_HIP(hipIpcOpenMemHandle(&p, info.handle,
                                       hipIpcMemLazyEnablePeerAccess));
    peers_[r] = p;
    ipc_opened_[r] = true;
  }
  connected_ = true;
}

size_t Communicator::sym_alloc(size_t bytes) {
  size_t const off = (user_bump_ + 255) & ~size_t(255);
  UCCL_CHECK(off + bytes <= user_cap_)
      << "symmetric region exhausted (" << (user_cap_ >> 20)
      << "MB; raise UCCL_SYM_USER_MB)";
  user_bump_ = off + bytes;
  return user_off_ + off;
}

CommView Communicator::view(uint64_t seq) const {
  CommView cv{};
  cv.rank = rank_;
  cv.world = world_;
  cv.seq = seq;
  cv.channel = kCollCh;
  cv.scratch_cap = scratch_cap_;
  size_t const parity = (seq >> 1) & 1;
  cv.sa_off = kScratchAOffset + parity * scratch_cap_;
  cv.sb_off = kScratchAOffset + 2 * scratch_cap_ + parity * scratch_cap_;
  for (int r = 0; r < kMaxRanks; ++r) cv.peers[r] = peers_[r];
  return cv;
}

void Communicator::all_reduce(void* data, size_t count, Dtype dt,
                              hipStream_t stream, RedOp op) {
  tally(0, count * dtype_size(dt));
  if (world_ == 1) {
    // Sum over one rank is the identity. By default this is a no-op; with
    // UCCL_WORLD1_STAGED=1 we still run the full staged kernel path
    // (copy-in + signal/wait + fullmesh reduce over {self}) so that
    // single-GPU benchmarks and smoke tests exercise the real engine.
    static bool const staged = env_bool("UCCL_WORLD1_STAGED", false);
    if (!staged) return;
    size_t const es = dtype_size(dt);
    size_t const chunk_elems = (scratch_cap_ / es) & ~size_t(63);
    for (size_t off = 0; off < count; off += chunk_elems) {
      size_t const n = std::min(chunk_elems, count - off);
      char* p = static_cast<char*>(data) + off * es;
      CommView const cv = view(next_seq());
      launch_copy(static_cast<char*>(heap_) + cv.sa_off, p, n * es, stream);
      launch_signal_wait(cv, cv.seq, all_mask(world_), stream);
      launch_oneshot_allreduce(cv, p, n, dt, op, stream);
    }
    return;
  }
  UCCL_CHECK(connected_) << "connect() not called";
  UCCL_CHECK(dt != Dtype::kU8) << "allreduce needs a typed dtype";
  size_t const es = dtype_size(dt);
  size_t const bytes = count * es;

  if (bytes <= ll_threshold_ && es < 8) {
    // (8-byte dtypes skip LL: its 4B packet lanes can't carry i64/f64)
    launch_ll_allreduce(view(next_seq()), data, data, count, dt, op,
                        stream);
    return;
  }
  if (is_symmetric_ptr(data)) {
    // zero-copy path: reduce straight out of every rank's symmetric user
    // region, push results back in place (no staging copies at all)
    size_t const shard_bytes = (count * es + world_ - 1) / world_ + 256;
    UCCL_CHECK(shard_bytes <= scratch_cap_)
        << "symmetric allreduce shard exceeds scratch";
    size_t const uoff =
        static_cast<char const*>(data) - static_cast<char*>(heap_);
    CommView const cv = view(next_seq());
    CommView const cv2 = view(next_seq());
    // entry barrier: every rank's input is produced + published
    launch_signal_wait(cv, cv.seq, all_mask(world_), stream);
    launch_twoshot_sym_rs(cv, uoff, count, dt, op, stream);
    // mid barrier: nobody reads user inputs any more -> pushes may land
    launch_signal_wait(cv, cv.seq + 1, all_mask(world_), stream);
    launch_twoshot_sym_push(cv, uoff, count, dt, stream);
    // exit barrier: all pushes into my buffer are visible before my
    // stream continues (the caller reads the result right after)
    launch_signal_wait(cv2, cv2.seq, all_mask(world_), stream);
    return;
  }
  bool const oneshot = bytes <= oneshot_threshold_;
  size_t const chunk_elems = (scratch_cap_ / es) & ~size_t(63);
  for (size_t off = 0; off < count; off += chunk_elems) {
    size_t const n = std::min(chunk_elems, count - off);
    char* p = static_cast<char*>(data) + off * es;
    CommView const cv = view(next_seq());
    launch_copy(static_cast<char*>(heap_) + cv.sa_off, p, n * es, stream);
    launch_signal_wait(cv, cv.seq, all_mask(world_), stream);
    if (oneshot) {
      launch_oneshot_allreduce(cv, p, n, dt, op, stream);
    } else {
      launch_twoshot_rs_push(cv, n, dt, op, stream);
      launch_signal_wait(cv, cv.seq + 1, all_mask(world_), stream);
      launch_twoshot_copyout(cv, p, n * es, stream);
    }
  }
}

void Communicator::all_gather(void* out, void const* in, size_t count_per_rank,
                              Dtype dt, hipStream_t stream) {
  tally(1, count_per_rank * dtype_size(dt));
  size_t const es = dtype_size(dt);
  if (world_ == 1) {
    if (out != in) launch_copy(out, in, count_per_rank * es, stream);
    return;
  }
  UCCL_CHECK(connected_) << "connect() not called";
  if (is_symmetric_ptr(out)) {
    // zero-copy: entry flag round first — remote writers must not touch a
    // peer's `out` before that peer has entered the call (its preceding
    // stream work may still be reading the buffer) — then push my slice
    // into every rank's slot, then the completion round.
    size_t const uoff =
        static_cast<char const*>(out) - static_cast<char*>(heap_);
    CommView const cv = view(next_seq());
    launch_signal_wait(cv, cv.seq, all_mask(world_), stream);
    launch_allgather_sym_push(cv, in, uoff, count_per_rank * es, stream);
    launch_signal_wait(cv, cv.seq + 1, all_mask(world_), stream);
    return;
  }
  size_t const chunk_elems = (scratch_cap_ / es) & ~size_t(63);
  for (size_t off = 0; off < count_per_rank; off += chunk_elems) {
    size_t const n = std::min(chunk_elems, count_per_rank - off);
    CommView const cv = view(next_seq());
    launch_copy(static_cast<char*>(heap_) + cv.sa_off,
                static_cast<char const*>(in) + off * es, n * es, stream);
    launch_signal_wait(cv, cv.seq, all_mask(world_), stream);
    if (n == count_per_rank) {
      launch_allgather_pull(cv, out, n * es, stream);
    } else {
      // chunked: strided per-peer pulls into out[p][off..]
      for (int p = 0; p < world_; ++p) {
        launch_copy_from_peer(
            cv, p, cv.sa_off,
            static_cast<char*>(out) +
                (static_cast<size_t>(p) * count_per_rank + off) * es,
            n * es, stream);
      }
    }
  }
}

void Communicator::reduce_scatter(void* out, void const* in,
                                  size_t count_per_rank, Dtype dt,
                                  hipStream_t stream, RedOp op) {
  tally(2, count_per_rank * dtype_size(dt) * world_);
  size_t const es = dtype_size(dt);
  if (world_ == 1) {
    if (out != in) launch_copy(out, in, count_per_rank * es, stream);
    return;
  }
  UCCL_CHECK(connected_) << "connect() not called";
  UCCL_CHECK(dt != Dtype::kU8) << "reduce_scatter needs a typed dtype";
  if (is_symmetric_ptr(in)) {
    // zero-copy: reduce my shard straight out of every rank's symmetric
    // input; entry barrier publishes inputs, exit barrier frees them
    size_t const uoff =
        static_cast<char const*>(in) - static_cast<char*>(heap_);
    CommView const cv = view(next_seq());
    launch_signal_wait(cv, cv.seq, all_mask(world_), stream);
    launch_reducescatter_sym(cv, uoff, out, count_per_rank, dt, op,
                             stream);
    launch_signal_wait(cv, cv.seq + 1, all_mask(world_), stream);
    return;
  }
  size_t const chunk_elems = (scratch_cap_ / es / world_) & ~size_t(63);
  for (size_t off = 0; off < count_per_rank; off += chunk_elems) {
    size_t const n = std::min(chunk_elems, count_per_rank - off);
    CommView const cv = view(next_seq());
    // stage [world][n] slices: slice p from in[p*count_per_rank + off]
    for (int p = 0; p < world_; ++p) {
      launch_copy(static_cast<char*>(heap_) + cv.sa_off +
                      static_cast<size_t>(p) * n * es,
                  static_cast<char const*>(in) +
                      (static_cast<size_t>(p) * count_per_rank + off) * es,
                  n * es, stream);
    }
    launch_signal_wait(cv, cv.seq, all_mask(world_), stream);
    launch_reducescatter_pull(cv, static_cast<char*>(out) + off * es, n, dt,
                              op, stream);
  }
}

void Communicator::broadcast(void* data, size_t count, Dtype dt, int root,
                             hipStream_t stream) {
  tally(3, count * dtype_size(dt));
  if (world_ == 1) return;
  UCCL_CHECK(connected_) << "connect() not called";
  size_t const es = dtype_size(dt);
  size_t const chunk_elems = (scratch_cap_ / es) & ~size_t(63);
  for (size_t off = 0; off < count; off += chunk_elems) {
    size_t const n = std::min(chunk_elems, count - off);
    CommView const cv = view(next_seq());
    if (rank_ == root) {
      launch_copy(static_cast<char*>(heap_) + cv.sa_off,
                  static_cast<char*>(data) + off * es, n * es, stream);
    }
    // wait ALL (not just root): a rank's next-call signal is what proves
    // its reads of this parity finished (scratch-reuse safety)
    launch_signal_wait(cv, cv.seq, all_mask(world_), stream);
    launch_broadcast_pull(cv, root, static_cast<char*>(data) + off * es,
                          n * es, stream);
  }
}

void Communicator::all_to_all(void* out, void const* in, size_t count_per_rank,
                              Dtype dt, hipStream_t stream) {
  tally(4, count_per_rank * dtype_size(dt) * world_);
  size_t const es = dtype_size(dt);
  if (world_ == 1) {
    if (out != in) launch_copy(out, in, count_per_rank * es, stream);
    return;
  }
  UCCL_CHECK(connected_) << "connect() not called";
  if (is_symmetric_ptr(out)) {
    // entry round before remote writes (see all_gather comment)
    size_t const uoff =
        static_cast<char const*>(out) - static_cast<char*>(heap_);
    CommView const cv = view(next_seq());
    launch_signal_wait(cv, cv.seq, all_mask(world_), stream);
    launch_alltoall_sym_push(cv, in, uoff, count_per_rank * es, stream);
    launch_signal_wait(cv, cv.seq + 1, all_mask(world_), stream);
    return;
  }
  size_t const chunk_elems = (scratch_cap_ / es / world_) & ~size_t(63);
  UCCL_CHECK(chunk_elems > 0) << "scratch too small for alltoall";
  for (size_t off = 0; off < count_per_rank; off += chunk_elems) {
    size_t const n = std::min(chunk_elems, count_per_rank - off);
    CommView const cv = view(next_seq());
    for (int p = 0; p < world_; ++p) {
      launch_copy(static_cast<char*>(heap_) + cv.sa_off +
                      static_cast<size_t>(p) * n * es,
                  static_cast<char const*>(in) +
                      (static_cast<size_t>(p) * count_per_rank + off) * es,
                  n * es, stream);
    }
    launch_signal_wait(cv, cv.seq, all_mask(world_), stream);
    if (n == count_per_rank) {
      launch_alltoall_pull(cv, out, n * es, stream);
    } else {
      // chunked: strided per-peer pulls
      for (int p = 0; p < world_; ++p) {
        launch_copy_from_peer(
            cv, p, cv.sa_off + static_cast<size_t>(rank_) * n * es,
            static_cast<char*>(out) +
                (static_cast<size_t>(p) * count_per_rank + off) * es,
            n * es, stream);
      }
    }
  }
}

void Communicator::send(void const* data, size_t bytes, int dst,
                        hipStream_t stream) {
  tally(5, bytes);
  UCCL_CHECK(connected_ && dst != rank_ && dst >= 0 && dst < world_)
      << "bad send dst " << dst;
  size_t const slot_off = kP2POffset + static_cast<size_t>(dst) * kP2PSlotBytes;
  for (size_t off = 0; off < bytes || (bytes == 0 && off == 0);
       off += kP2PSlotBytes) {
    size_t const n = std::min(kP2PSlotBytes, bytes - off);
    CommView cv = view(0);
    if (send_seq_[dst] > 0) {
      // wait for receiver ack of the previous chunk before reusing the slot
      launch_wait_peer(cv, dst, kP2PAckCh, send_seq_[dst], stream);
    }
    launch_copy(static_cast<char*>(heap_) + slot_off,
                static_cast<char const*>(data) + off, n, stream);
    ++send_seq_[dst];
    launch_signal_peer(cv, dst, kP2PDataCh, send_seq_[dst], stream);
    if (bytes == 0) break;
  }
}

void Communicator::recv(void* data, size_t bytes, int src,
                        hipStream_t stream) {
  tally(6, bytes);
  UCCL_CHECK(connected_ && src != rank_ && src >= 0 && src < world_)
      << "bad recv src " << src;
  size_t const slot_off =
      kP2POffset + static_cast<size_t>(rank_) * kP2PSlotBytes;
  for (size_t off = 0; off < bytes || (bytes == 0 && off == 0);
       off += kP2PSlotBytes) {
    size_t const n = std::min(kP2PSlotBytes, bytes - off);
    CommView cv = view(0);
    ++recv_seq_[src];
    launch_wait_peer(cv, src, kP2PDataCh, recv_seq_[src], stream);
    launch_copy_from_peer(cv, src, slot_off,
                          static_cast<char*>(data) + off, n, stream);
    launch_signal_peer(cv, src, kP2PAckCh, recv_seq_[src], stream);
    if (bytes == 0) break;
  }
}

void Communicator::barrier(hipStream_t stream) {
  tally(7, 0);
  if (world_ == 1) return;
  UCCL_CHECK(connected_) << "connect() not called";
  launch_barrier(view(next_seq()), stream);
}

}  // namespace uccl
