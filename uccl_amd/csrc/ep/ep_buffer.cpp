// Host-side EP Buffer: symmetric heap allocation, IPC exchange, dispatch /
// combine orchestration. Parity role: reference ep/src/uccl_ep.cc Buffer
// (:348) intranode path, without proxies (pure xGMI on one MI355X node).

#include "ep_buffer.h"

#include <hip/hip_runtime.h>
#include <unistd.h>

#include <cstring>

#include "../core/env.h"
#include "../core/log.h"
#include "d2h_ring.h"
#include "ep_kernels.h"
#include "ep_proxy.h"

namespace uccl {
namespace ep {

static size_t align256(size_t x) { return (x + 255) & ~size_t(255); }

EpBuffer::EpBuffer(int rank, int world, int device, int num_experts,
                   int topk, int hidden, int max_tokens, int elem_size,
                   bool use_fp8, bool with_normal)
    : rank_(rank), world_(world), device_(device) {
  if (use_fp8) {
    UCCL_CHECK(hidden % 128 == 0) << "fp8 dispatch needs hidden % 128 == 0";
    UCCL_CHECK(elem_size == 2) << "fp8 dispatch quantizes bf16/fp16 input";
  }
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
  v_.disp_fp8 = use_fp8 ? 1 : 0;
  v_.disp_elem = use_fp8 ? 1 : elem_size;
  v_.seq = 0;

  size_t off = 0;
  v_.off_disp_count = off;
  off = align256(off + sizeof(uint64_t) * v_.local_experts * world);
  v_.off_comb_flag = off;
  off = align256(off + sizeof(uint64_t) * world);
  v_.off_sync = off;
  off = align256(off + sizeof(uint64_t) * 4);
  v_.off_consumed = off;
  off = align256(off + sizeof(uint64_t) * world);
  v_.off_disp_meta = off;
  off = align256(off + sizeof(uint32_t) * v_.local_experts * world *
                           static_cast<size_t>(max_tokens));
  v_.off_disp_x = off;
  off = align256(off + static_cast<size_t>(v_.local_experts) * world *
                           max_tokens * hidden * v_.disp_elem);
  v_.off_disp_scale = off;
  if (use_fp8)
    off = align256(off + static_cast<size_t>(v_.local_experts) * world *
                             max_tokens * (hidden / 128) * sizeof(float));
  v_.off_comb_x = off;
  off = align256(off + static_cast<size_t>(max_tokens) * topk * hidden *
                           elem_size);
  v_.off_plan = off;
  off = align256(off + sizeof(uint32_t) * num_experts *
                           (2 + static_cast<size_t>(max_tokens)));
  v_.off_plan_chunks = off;
  off = align256(off + sizeof(uint32_t) * num_experts * kPlanChunks);
  size_t const egress_rows = static_cast<size_t>(max_tokens) * topk;
  v_.off_egress = off;
  off = align256(off + egress_rows * hidden * elem_size);
  v_.off_egress_meta = off;
  off = align256(off + egress_rows * sizeof(uint32_t));
  // ingress: per-source-peer staging slices of max_tokens rows (each
  // proxy rx message covers one (expert, src) pair <= max_tokens rows)
  size_t const ingress_rows =
      static_cast<size_t>(world) * max_tokens;
  v_.off_ingress = off;
  off = align256(off + ingress_rows * hidden * elem_size);
  v_.off_ingress_meta = off;
  off = align256(off + ingress_rows * sizeof(uint32_t));
  // normal (rank-granular) mode regions — DeepEP HT dispatch/combine
  v_.with_normal = with_normal ? 1 : 0;
  if (with_normal) {
    size_t const rows = static_cast<size_t>(world) * max_tokens;
    v_.off_nrm_count = off;
    off = align256(off + sizeof(uint64_t) * world);
    v_.off_nrm_flag = off;
    off = align256(off + sizeof(uint64_t) * world);
    v_.off_nrm_x = off;
    off = align256(off + rows * hidden * elem_size);
    v_.off_nrm_meta = off;
    off = align256(off + rows * sizeof(uint32_t));
    v_.off_nrm_topk = off;
    off = align256(off + rows * topk * sizeof(int64_t));
    v_.off_nrm_w = off;
    off = align256(off + rows * topk * sizeof(float));
    v_.off_nrm_plan = off;
    off = align256(off + sizeof(uint32_t) * world *
                             (1 + static_cast<size_t>(max_tokens)));
    v_.off_nrm_ret = off;
    off = align256(off + static_cast<size_t>(max_tokens) * world * hidden *
                             elem_size);
  }
  v_.heap_bytes = off;

  UCCL_CHECK_HIP(hipSetDevice(device_));
  UCCL_CHECK_HIP(hipMalloc(&heap_, v_.heap_bytes));
  UCCL_CHECK_HIP(hipMemset(heap_, 0, v_.off_disp_meta));
  if (v_.with_normal) {
    UCCL_CHECK_HIP(hipMemset(static_cast<char*>(heap_) + v_.off_nrm_count,
                             0, sizeof(uint64_t) * world_));
    UCCL_CHECK_HIP(hipMemset(static_cast<char*>(heap_) + v_.off_nrm_flag,
                             0, sizeof(uint64_t) * world_));
  }
  UCCL_CHECK_HIP(hipDeviceSynchronize());
  for (int r = 0; r < kMaxRanks; ++r) v_.peers[r] = nullptr;
  v_.peers[rank_] = heap_;
  v_.proxy_mask = 0;
  v_.ring = nullptr;
  // fp8 + proxy: the copy kernel quantizes into split egress streams
  // (rows + scales) and the proxy ships both — half the wire bytes of
  // raw bf16 (see egress_x_fp8/egress_scale_fp8).
  if (env_bool("UCCL_EP_FORCE_PROXY", false) && world_ > 1) {
    UCCL_CHECK_HIP(hipHostMalloc(&ring_host_, sizeof(D2HRing),
                                 hipHostMallocMapped));
    memset(ring_host_, 0, sizeof(D2HRing));
    void* dev = nullptr;
    UCCL_CHECK_HIP(hipHostGetDevicePointer(&dev, ring_host_, 0));
    v_.ring = static_cast<D2HRing*>(dev);
  }
  for (int i = 0; i < kCountSlots; ++i)
    UCCL_CHECK_HIP(hipHostMalloc(
        reinterpret_cast<void**>(&host_counts_[i]),
        sizeof(int) * v_.local_experts * world_));
  UCCL_LOG_INFO << "EpBuffer rank " << rank << "/" << world << " experts="
                << num_experts << " hidden=" << hidden << " max_tokens="
                << max_tokens << " heap=" << (v_.heap_bytes >> 20) << "MB";
}

EpBuffer::~EpBuffer() {
  proxy_.reset();  // joins proxy threads before the heap goes away
  for (int r = 0; r < world_; ++r)
    if (ipc_opened_[r] && v_.peers[r]) (void)hipIpcCloseMemHandle(v_.peers[r]);
  if (heap_) (void)hipFree(heap_);
  if (ring_host_) (void)hipHostFree(ring_host_);
  for (int i = 0; i < kCountSlots; ++i)
    if (host_counts_[i]) (void)hipHostFree(host_counts_[i]);
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
  std::string out(reinterpret_cast<char*>(&b), sizeof(b));
  if (v_.ring) {
    // proxy mode: append the transport rendezvous metadata (the proxy is
    // created lazily here so its endpoint exists before exchange)
    auto* self = const_cast<EpBuffer*>(this);
    if (!self->proxy_)
      self->proxy_ = std::make_unique<EpProxy>(v_, heap_, ring_host_,
                                               device_);
    out += proxy_->transport_metadata();
  }
  return out;
}

void EpBuffer::connect(const std::vector<std::string>& handles) {
  UCCL_CHECK(static_cast<int>(handles.size()) == world_);
  UCCL_CHECK_HIP(hipSetDevice(device_));
  bool const force_proxy = v_.ring != nullptr;
  std::vector<std::string> tp_md(world_);
  for (int r = 0; r < world_; ++r) {
    if (r == rank_) continue;
    struct Blob {
      hipIpcMemHandle_t h;
      int device;
      int pid;
    } b{};
    UCCL_CHECK(handles[r].size() >= sizeof(b)) << "bad ep handle blob";
    memcpy(&b, handles[r].data(), sizeof(b));
    tp_md[r] = handles[r].substr(sizeof(b));
    if (force_proxy) {
      // emulate internode: no IPC mapping, all traffic via proxy
      v_.proxy_mask |= 1u << r;
      continue;
    }
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
  if (force_proxy) {
    UCCL_CHECK(proxy_) << "handle_bytes() must run before connect()";
    proxy_->set_view(v_);
    proxy_->establish_flows(tp_md, v_.proxy_mask);
    proxy_->start();
  }
  connected_ = true;
}

void EpBuffer::dispatch_send(void const* x, int64_t const* topk_idx,
                             int num_tokens, bool reuse_plan,
                             hipStream_t stream) {
  UCCL_CHECK(connected_ || world_ == 1) << "connect() not called";
  UCCL_CHECK(num_tokens <= v_.max_tokens)
      << num_tokens << " tokens > max_tokens " << v_.max_tokens;
  UCCL_CHECK(!reuse_plan || last_num_tokens_ == num_tokens)
      << "cached dispatch requires the same token count as the plan";
  ++v_.seq;
  launch_ep_dispatch_send(v_, x, topk_idx, num_tokens, reuse_plan, stream);
  last_num_tokens_ = num_tokens;
}

void EpBuffer::dispatch_recv(int* out_counts, hipStream_t stream) {
  launch_ep_dispatch_recv(v_, out_counts, stream);
  // pinned host copy of counts (combine proxy shipping needs them);
  // rotate slots so an in-flight combine's pointer stays valid
  UCCL_CHECK_HIP(hipMemcpyAsync(host_counts_[v_.seq % kCountSlots],
                                out_counts,
                                sizeof(int) * v_.local_experts * world_,
                                hipMemcpyDeviceToHost, stream));
}

void EpBuffer::dispatch(void const* x, int64_t const* topk_idx,
                        int num_tokens, int* out_counts,
                        hipStream_t stream) {
  dispatch_send(x, topk_idx, num_tokens, /*reuse_plan=*/false, stream);
  dispatch_recv(out_counts, stream);
}

void EpBuffer::nrm_dispatch_send(void const* x, int64_t const* topk_idx,
                                 float const* topk_w, int num_tokens,
                                 hipStream_t stream) {
  UCCL_CHECK(v_.with_normal) << "buffer built without normal mode";
  UCCL_CHECK(connected_ || world_ == 1) << "connect() not called";
  UCCL_CHECK(!v_.proxy_mask) << "normal mode is xGMI-only (intranode)";
  UCCL_CHECK(num_tokens <= v_.max_tokens) << "too many tokens";
  ++v_.seq;
  launch_ep_nrm_dispatch_send(v_, x, topk_idx, topk_w, num_tokens, stream);
  nrm_last_tokens_ = num_tokens;
}

void EpBuffer::nrm_dispatch_recv(int* out_counts, hipStream_t stream) {
  launch_ep_nrm_dispatch_recv(v_, out_counts, stream);
}

void EpBuffer::nrm_combine_send(void const* x, hipStream_t stream) {
  UCCL_CHECK(nrm_last_tokens_ >= 0) << "combine without dispatch";
  launch_ep_nrm_combine_send(v_, x, stream);
}

void EpBuffer::nrm_combine_recv(void* out, int64_t const* topk_idx,
                                hipStream_t stream) {
  launch_ep_nrm_combine_recv(v_, out, topk_idx, nrm_last_tokens_, stream);
}

void EpBuffer::barrier(hipStream_t stream) {
  UCCL_CHECK(v_.ring) << "barrier requires the proxy path";
  ++sync_seq_;
  launch_ep_barrier(v_, sync_seq_, stream);
}

void EpBuffer::quiet(hipStream_t stream) {
  UCCL_CHECK(v_.ring) << "quiet requires the proxy path";
  ++sync_seq_;
  launch_ep_quiet(v_, sync_seq_, stream);
}

void EpBuffer::atomic_add(int dst, uint64_t value, hipStream_t stream) {
  UCCL_CHECK(v_.ring) << "atomic_add requires the proxy path";
  launch_ep_atomic_add(v_, dst, v_.off_sync + 2 * sizeof(uint64_t), value,
                       stream);
}

void EpBuffer::check_error() {
  uint64_t const e = read_sync_word(3);
  if (!e) return;
  static char const* names[] = {"?",            "dispatch_wait",
                                "combine_wait", "consume_gate",
                                "sync_wait",    "nrm_wait",
                                "nrm_ret_wait", "ring_drain"};
  int const code = static_cast<int>(e >> 56);
  UCCL_CHECK(false) << "EP device wait TIMED OUT: "
                    << names[code >= 0 && code <= 7 ? code : 0]
                    << " rank=" << ((e >> 48) & 0xff) << " aux="
                    << ((e >> 32) & 0xffff) << " seq=" << (e & 0xffffffffull);
}

uint64_t EpBuffer::read_sync_word(int idx) {
  uint64_t v = 0;
  UCCL_CHECK_HIP(hipMemcpy(&v, sync_ptr(heap_, v_, idx), sizeof(v),
                           hipMemcpyDeviceToHost));
  return v;
}

void EpBuffer::combine_send(void const* expert_out, hipStream_t stream) {
  UCCL_CHECK(last_num_tokens_ >= 0) << "combine without a prior dispatch";
  launch_ep_combine_send(v_, expert_out, stream);
  if (v_.proxy_mask) {
    // proxy peers: ship expert_out rows host-side once the stream reaches
    // this point (expert_out fully produced + host_counts_ landed)
    hipEvent_t ev = nullptr;
    UCCL_CHECK_HIP(hipEventCreateWithFlags(&ev, hipEventDisableTiming));
    UCCL_CHECK_HIP(hipEventRecord(ev, stream));
    proxy_->enqueue_combine(expert_out, v_.seq, ev,
                            host_counts_[v_.seq % kCountSlots]);
  }
}

void EpBuffer::combine_recv(void* out, int64_t const* topk_idx,
                            float const* topk_w, hipStream_t stream) {
  launch_ep_combine_finish(v_, out, topk_idx, topk_w, last_num_tokens_,
                           stream);
}

void EpBuffer::combine(void const* expert_out, void* out,
                       int64_t const* topk_idx, float const* topk_w,
                       hipStream_t stream) {
  combine_send(expert_out, stream);
  combine_recv(out, topk_idx, topk_w, stream);
}

}  // namespace ep
}  // namespace uccl
