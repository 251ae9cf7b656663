// Host-side EP Buffer: symmetric heap allocation, IPC exchange, dispatch /
// combine orchestration. Parity role: reference ep/src/uccl_ep.cc Buffer
// (:348) intranode path, without proxies (pure xGMI on one MI355X node).

#include "ep_buffer.h"

#include <hip/hip_runtime.h>
#include <unistd.h>

#include <cstring>

#include "../core/env.h"
#include "../core/log.h"
#include "ep_kernels.h"

namespace uccl {
namespace ep {

static size_t align256(size_t x) { return (x + 255) & ~size_t(255); }

EpBuffer::EpBuffer(int rank, int world, int device, int num_experts,
                   int topk, int hidden, int max_tokens, int elem_size)
    : rank_(rank), world_(world), device_(device) {
  UCCL_CHECK(world >= 1 && world <= kMaxRanks) << "world=" << world;
  UCCL_CHECK(num_experts % world == 0)
      << "num_experts " << num_experts << " must divide world " << world;
  UCCL_CHECK(elem_size == 2 || elem_size == 4) << "elem_size 2|4 only";
  UCCL_CHECK(max_tokens <= (1 << 24)) << "max_tokens too large for meta";
  UCCL_CHECK(topk <= 64) << "topk too large for meta (k stored in 8 bits)";

  v_.rank = rank;
  v_.world = world;
  v_.num_experts = num_experts;
  v_.local_experts = num_experts / world;
  v_.topk = topk;
  v_.hidden = hidden;
  v_.max_tokens = max_tokens;
  v_.elem_size = elem_size;
  v_.seq = 0;

  size_t off = 0;
  v_.off_disp_count = off;
  off = align256(off + sizeof(uint64_t) * v_.local_experts * world);
  v_.off_comb_flag = off;
  off = align256(off + sizeof(uint64_t) * world);
  v_.off_disp_meta = off;
  off = align256(off + sizeof(uint32_t) * v_.local_experts * world *
                           static_cast<size_t>(max_tokens));
  v_.off_disp_x = off;
  off = align256(off + static_cast<size_t>(v_.local_experts) * world *
                           max_tokens * hidden * elem_size);
  v_.off_comb_x = off;
  off = align256(off + static_cast<size_t>(max_tokens) * topk * hidden *
                           elem_size);
  v_.off_plan = off;
  off = align256(off + sizeof(uint32_t) * num_experts *
                           (1 + static_cast<size_t>(max_tokens)));
  v_.heap_bytes = off;

  UCCL_CHECK_HIP(hipSetDevice(device_));
  UCCL_CHECK_HIP(hipMalloc(&heap_, v_.heap_bytes));
  UCCL_CHECK_HIP(hipMemset(heap_, 0, v_.off_disp_meta));
  UCCL_CHECK_HIP(hipDeviceSynchronize());
  for (int r = 0; r < kMaxRanks; ++r) v_.peers[r] = nullptr;
  v_.peers[rank_] = heap_;
  UCCL_LOG_INFO << "EpBuffer rank " << rank << "/" << world << " experts="
                << num_experts << " hidden=" << hidden << " max_tokens="
                << max_tokens << " heap=" << (v_.heap_bytes >> 20) << "MB";
}

EpBuffer::~EpBuffer() {
  for (int r = 0; r < world_; ++r)
    if (ipc_opened_[r] && v_.peers[r]) (void)hipIpcCloseMemHandle(v_.peers[r]);
  if (heap_) (void)hipFree(heap_);
}

std::string EpBuffer::handle_bytes() const {
  struct Blob {
    hipIpcMemHandle_t h;
    int device;
    int pid;
  } b{};
  UCCL_CHECK_HIP(hipIpcGetMemHandle(&b.h, heap_));
  b.device = device_;
  b.pid = static_cast<int>(getpid());
  return std::string(reinterpret_cast<char*>(&b), sizeof(b));
}

void EpBuffer::connect(const std::vector<std::string>& handles) {
  UCCL_CHECK(static_cast<int>(handles.size()) == world_);
  UCCL_CHECK_HIP(hipSetDevice(device_));
  for (int r = 0; r < world_; ++r) {
    if (r == rank_) continue;
    struct Blob {
      hipIpcMemHandle_t h;
      int device;
      int pid;
    } b{};
    UCCL_CHECK(handles[r].size() == sizeof(b)) << "bad ep handle blob";
    memcpy(&b, handles[r].data(), sizeof(b));
    if (b.device != device_) {
      hipError_t e = hipDeviceEnablePeerAccess(b.device, 0);
      UCCL_CHECK(e == hipSuccess || e == hipErrorPeerAccessAlreadyEnabled)
          << hipGetErrorString(e);
    }
    void* p = nullptr;
    UCCL_CHECK_HIP(hipIpcOpenMemHandle(&p, b.h,
                                       hipIpcMemLazyEnablePeerAccess));
    v_.peers[r] = p;
    ipc_opened_[r] = true;
  }
  connected_ = true;
}

void EpBuffer::dispatch(void const* x, int64_t const* topk_idx,
                        int num_tokens, int* out_counts,
                        hipStream_t stream) {
  UCCL_CHECK(connected_ || world_ == 1) << "connect() not called";
  UCCL_CHECK(num_tokens <= v_.max_tokens)
      << num_tokens << " tokens > max_tokens " << v_.max_tokens;
  ++v_.seq;
  launch_ep_dispatch(v_, x, topk_idx, num_tokens, out_counts, stream);
  last_num_tokens_ = num_tokens;
}

void EpBuffer::combine(void const* expert_out, void* out,
                       int64_t const* topk_idx, float const* topk_w,
                       hipStream_t stream) {
  UCCL_CHECK(last_num_tokens_ >= 0) << "combine without a prior dispatch";
  launch_ep_combine(v_, expert_out, out, topk_idx, topk_w, last_num_tokens_,
                    stream);
}

}  // namespace ep
}  // namespace uccl
