// CDNA4 expert-parallel dispatch/combine kernels (intranode xGMI).
//
// Protocol (see ep_layout.h): per destination expert e, one block scans the
// local top-k table, compacts matching (token, k) pairs in token order
// (stable two-pass prefix compaction in LDS), copies each token's hidden
// vector into the destination rank's slot array over xGMI, then publishes a
// seq-tagged count with a system-scope release store. Combine returns each
// processed slot to its source (token, k) cell and the source reduces with
// top-k weights in fp32. Waits are isolated in single-block kernels (same
// no-spinning-workers rule as the collective engine).
//
// Functional parity: reference ep/src/internode_ll.cu dispatch/combine
// (phases, per-expert slots, src-idx meta) minus the RDMA/proxy path, which
// a single 8-OAM MI355X node does not need.

#include <hip/hip_runtime.h>

#include <algorithm>

#include "../device/primitives.h"
#include "d2h_ring.h"
#include "ep_kernels.h"

namespace uccl {
namespace ep {

using namespace uccl::device;

namespace {

__device__ inline uint64_t tag_count(uint64_t seq, uint32_t count) {
  return (seq << 32) | count;
}

// Wait-timeout diagnostics: instead of trapping (which loses the device
// printf and kills the queue with an opaque 0x1016), record WHICH wait
// stalled into the heap's sync[3] word and return; the host checks the
// word after its next sync and raises with the decoded context
// (EpBuffer::check_error). err = code:8 | rank:8 | aux:16 | seq:32.
enum EpWaitCode : uint64_t {
  kErrDispatchWait = 1,
  kErrCombineWait = 2,
  kErrConsumeGate = 3,
  kErrSyncWait = 4,
  kErrNrmWait = 5,
  kErrNrmRetWait = 6,
  kErrRingDrain = 7,
};

__device__ inline void record_wait_err(const EpView& v, uint64_t code,
                                       uint64_t aux) {
  uint64_t const e = (code << 56) |
                     (static_cast<uint64_t>(v.rank & 0xff) << 48) |
                     ((aux & 0xffff) << 32) |
                     (v.seq & 0xffffffffull);
  st_release_sys(sync_ptr(v.peers[v.rank], v, 3), e);
}

// Quantize one hidden row (bf16) to fp8-e4m3 with per-128-element scales,
// cooperatively with a 256-thread block (two 128-element groups in
// flight; cross-wave amax via LDS). DeepEP LL fp8 semantics
// (reference ep/src/internode_ll.cu:96-200): scale = amax/448.
__device__ inline void block_quant_row_fp8(
    char* __restrict__ dst, float* __restrict__ dscale,
    __hip_bfloat16 const* __restrict__ src, int hidden, float* red) {
  int const ngroups = hidden / 128;
  for (int g0 = 0; g0 < ngroups; g0 += 2) {
    int const g = g0 + (threadIdx.x >> 7);
    int const lane = threadIdx.x & 127;
    bool const active = g < ngroups;
    float v = 0.f;
    if (active) v = __bfloat162float(src[g * 128 + lane]);
    red[threadIdx.x] = fabsf(v);
    __syncthreads();
    // tree-reduce amax within each 128-thread half
#pragma unroll
    for (int off = 64; off > 0; off >>= 1) {
      if (lane < off)
        red[threadIdx.x] =
            fmaxf(red[threadIdx.x], red[threadIdx.x + off]);
      __syncthreads();
    }
    float const amax = red[(threadIdx.x >> 7) << 7];
    float const scale = amax > 0.f ? amax / 448.f : 1.f;
    if (active) {
      reinterpret_cast<__hip_fp8_e4m3*>(dst)[g * 128 + lane] =
          __hip_fp8_e4m3(v / scale);
      if (lane == 0) dscale[g] = scale;
    }
    __syncthreads();
  }
}

// Copy for read-once sources (combine returns): NT on both sides.
__device__ inline void block_copy_nt(char* __restrict__ dst,
                                     char const* __restrict__ src,
                                     size_t bytes) {
  size_t const nvec = bytes / 16;
  auto* d = reinterpret_cast<V16*>(dst);
  auto const* s = reinterpret_cast<V16 const*>(src);
  for (size_t i = threadIdx.x; i < nvec; i += blockDim.x)
    nt_store(&d[i], nt_load(&s[i]));
  size_t const tail = bytes & 15;
  if (tail && threadIdx.x < tail)
    dst[bytes - tail + threadIdx.x] = src[bytes - tail + threadIdx.x];
}

// Copy `bytes` from src to dst cooperatively with the whole block.
__device__ inline void block_copy(char* __restrict__ dst,
                                  char const* __restrict__ src,
                                  size_t bytes) {
  size_t const nvec = bytes / 16;
  auto* d = reinterpret_cast<V16*>(dst);
  auto const* s = reinterpret_cast<V16 const*>(src);
  // NT stores (dest written once) but cached loads: a token row is read
  // up to top-k times by different expert blocks, so load-side NT would
  // force HBM re-reads (measured: 122us -> 134us dispatch p50)
  for (size_t i = threadIdx.x; i < nvec; i += blockDim.x)
    nt_store(&d[i], s[i]);
  size_t const tail = bytes & 15;
  if (tail && threadIdx.x < tail)
    dst[bytes - tail + threadIdx.x] = src[bytes - tail + threadIdx.x];
}

}  // namespace

// ---------------------------------------------------------------------------
// dispatch: three phases for full-chip parallelism (the naive one-block-
// per-expert variant measured only ~70 GB/s on MI355X — 8 blocks on 256
// CUs; see profiles/).
//   plan    (grid=E):        stable compaction of (token,k) lists into
//                            private plan scratch
//   copy    (grid=E*fanout): parallel slot copies over xGMI
//   publish (grid=1):        seq-tagged release-store of counts
// ---------------------------------------------------------------------------

// copy blocks per (expert|pair): launchers size the grid for ~2048 blocks
// total; kernels derive the fanout from gridDim
__host__ __device__ inline int fanout_for(int pairs) {
  int f = 2048 / (pairs > 0 ? pairs : 1);
  return f < 1 ? 1 : (f > 256 ? 256 : f);
}

// Back-to-back dispatch safety: a new dispatch generation must not
// overwrite peers' count/meta/slot words while they are still reading
// the previous one (the combine round-trip provides this implicitly in
// the dispatch;combine;dispatch pattern, but DeepEP's phase-split and
// cached modes allow consecutive dispatches). dispatch_recv SIGNALS
// "generation consumed" into every peer; the next dispatch_send GATES
// on all peers having consumed the prior generation.
__global__ void k_ep_consume_signal(EpView v, uint64_t seq) {
  if (threadIdx.x < static_cast<unsigned>(v.world) &&
      !((v.proxy_mask >> threadIdx.x) & 1u))
    st_release_sys(consumed_ptr(v.peers[threadIdx.x], v, v.rank), seq);
  // proxied peers have no IPC mapping: route the signal through the CPU
  // proxy (ring cmd -> tiny wire message -> peer rx writes our consumed
  // word). Without this, a fast peer's generation N+1 counts overwrote
  // generation N tags before this rank's wait kernel sampled them —
  // observed as "last-shipped expert slot never matches its seq tag"
  // under UCCL_EP_FORCE_PROXY with back-to-back dispatches.
  __syncthreads();
  if (threadIdx.x == 0 && v.proxy_mask && v.ring)
    ring_push(v.ring, TransferCmd{static_cast<uint32_t>(CmdOp::kConsume),
                                  static_cast<uint32_t>(seq), seq, 0, 0});
}

__global__ void k_ep_consume_gate(EpView v, uint64_t prev_seq) {
  // gate on EVERY peer (IPC-signalled or proxy-relayed): the consumed
  // word for proxied peers is written host-side by the proxy rx loop
  if (threadIdx.x < static_cast<unsigned>(v.world)) {
    uint64_t const* p = consumed_ptr(v.peers[v.rank], v, threadIdx.x);
    for (uint64_t it = 0;; ++it) {
      if (ld_acquire_sys(p) >= prev_seq) return;
      if (it > (1ull << 28)) {
        record_wait_err(v, kErrConsumeGate, threadIdx.x);
        return;
      }
      backoff();
    }
  }
}

// proxy egress reuse gate: spin until the CPU proxy has shipped every
// queued ring command (head catches tail) so the egress staging rows
// can be overwritten by the next generation
__global__ void k_ep_ring_wait_empty(EpView v) {
  if (threadIdx.x != 0 || blockIdx.x != 0 || !v.ring) return;
  for (uint64_t it = 0;; ++it) {
    uint64_t const t = __hip_atomic_load(
        const_cast<uint64_t*>(&v.ring->tail), __ATOMIC_RELAXED,
        __HIP_MEMORY_SCOPE_SYSTEM);
    uint64_t const h = __hip_atomic_load(
        const_cast<uint64_t*>(&v.ring->head), __ATOMIC_ACQUIRE,
        __HIP_MEMORY_SCOPE_SYSTEM);
    if (h == t) return;
    if (it > (1ull << 28)) {
      record_wait_err(v, kErrRingDrain, static_cast<uint64_t>(t - h));
      return;
    }
    backoff();
  }
}

// Plan build, parallel over (expert, token-chunk): grid E*kPlanChunks.
// One-block-per-expert left >95% of the 256 CUs idle at DeepEP's small
// expert counts (rocprof r02: 48.7 us = 25% of dispatch at E=8). Token
// order inside the plan list is preserved (chunks are contiguous token
// ranges, filled at chunk-prefix bases) — the slot arrays stay in token
// order, which the torch-reference tests assert bit-exactly.
__global__ void k_ep_dispatch_plan_count(
    EpView v, int64_t const* __restrict__ topk_idx, int num_tokens) {
  int const e = blockIdx.x / kPlanChunks;
  int const c = blockIdx.x % kPlanChunks;
  int const chunk = (num_tokens + kPlanChunks - 1) / kPlanChunks;
  int const t0 = c * chunk;
  int const t1 = min(t0 + chunk, num_tokens);
  __shared__ uint32_t red[256];
  uint32_t mine = 0;
  // strided-coalesced over the flattened (t,k) range (order irrelevant
  // for counting)
  size_t const lo = static_cast<size_t>(t0) * v.topk;
  size_t const hi = static_cast<size_t>(t1) * v.topk;
  for (size_t i = lo + threadIdx.x; i < hi; i += blockDim.x)
    if (topk_idx[i] == e) ++mine;
  red[threadIdx.x] = mine;
  __syncthreads();
  for (int off = blockDim.x / 2; off; off >>= 1) {
    if (threadIdx.x < static_cast<unsigned>(off))
      red[threadIdx.x] += red[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0)
    plan_chunks_ptr(v.peers[v.rank], v, e)[c] = red[0];
}

__global__ void k_ep_dispatch_plan_fill(
    EpView v, int64_t const* __restrict__ topk_idx, int num_tokens) {
  int const e = blockIdx.x / kPlanChunks;
  int const c = blockIdx.x % kPlanChunks;
  int const chunk = (num_tokens + kPlanChunks - 1) / kPlanChunks;
  int const ct0 = c * chunk;
  int const ct1 = min(ct0 + chunk, num_tokens);
  uint32_t* plan = plan_ptr(v.peers[v.rank], v, e);
  uint32_t const base = plan_chunks_ptr(v.peers[v.rank], v, e)[c];
  extern __shared__ uint32_t prefix[];  // [blockDim.x + 1]
  // ordered within the chunk: per-thread contiguous token segments
  int const seg = (ct1 - ct0 + blockDim.x - 1) / blockDim.x;
  int const t0 = ct0 + threadIdx.x * seg;
  int const t1 = min(t0 + seg, ct1);
  uint32_t mine = 0;
  for (int t = t0; t < t1; ++t)
    for (int k = 0; k < v.topk; ++k)
      if (topk_idx[static_cast<size_t>(t) * v.topk + k] == e) ++mine;
  prefix[threadIdx.x + 1] = mine;
  __syncthreads();
  // Hillis-Steele inclusive scan over per-thread counts
  if (threadIdx.x == 0) prefix[0] = 0;
  for (int off = 1; off < static_cast<int>(blockDim.x); off <<= 1) {
    uint32_t const add =
        threadIdx.x + 1 > static_cast<unsigned>(off)
            ? prefix[threadIdx.x + 1 - off]
            : 0;
    __syncthreads();
    prefix[threadIdx.x + 1] += add;
    __syncthreads();
  }
  uint32_t pos = base + prefix[threadIdx.x];
  for (int t = t0; t < t1; ++t)
    for (int k = 0; k < v.topk; ++k)
      if (topk_idx[static_cast<size_t>(t) * v.topk + k] == e)
        plan[2 + pos++] = static_cast<uint32_t>(t) |
                          (static_cast<uint32_t>(k) << 24);
}

// Exclusive prefix of per-expert counts -> plan[e][1] (egress row
// offsets for the proxy path). 256-thread Hillis-Steele scan in LDS per
// tile of experts — the round-1 single-thread walk serialized at 256
// experts.
__global__ void k_ep_plan_prefix(EpView v) {
  __shared__ uint32_t tile[256];
  // fold the per-(expert,chunk) counts: plan[e][0] = total, and rewrite
  // chunks[e][c] into the exclusive in-plan base for chunk c (consumed
  // by k_ep_dispatch_plan_fill)
  for (int e = threadIdx.x; e < v.num_experts; e += blockDim.x) {
    uint32_t* ch = plan_chunks_ptr(v.peers[v.rank], v, e);
    uint32_t run = 0;
#pragma unroll
    for (int c = 0; c < kPlanChunks; ++c) {
      uint32_t const n = ch[c];
      ch[c] = run;
      run += n;
    }
    plan_ptr(v.peers[v.rank], v, e)[0] = run;
  }
  __syncthreads();
  uint32_t carry = 0;
  for (int base = 0; base < v.num_experts; base += 256) {
    int const e = base + static_cast<int>(threadIdx.x);
    uint32_t const cnt =
        e < v.num_experts ? plan_ptr(v.peers[v.rank], v, e)[0] : 0;
    tile[threadIdx.x] = cnt;
    __syncthreads();
#pragma unroll
    for (int off = 1; off < 256; off <<= 1) {
      uint32_t const add = threadIdx.x >= static_cast<unsigned>(off)
                               ? tile[threadIdx.x - off]
                               : 0;
      __syncthreads();
      tile[threadIdx.x] += add;
      __syncthreads();
    }
    if (e < v.num_experts)
      plan_ptr(v.peers[v.rank], v, e)[1] = carry + tile[threadIdx.x] - cnt;
    uint32_t const tot = tile[255];
    __syncthreads();
    carry += tot;
  }
}

__global__ void k_ep_dispatch_copy(EpView v, void const* __restrict__ x) {
  int const fanout = gridDim.x / v.num_experts;
  int const e = blockIdx.x / fanout;
  int const b = blockIdx.x % fanout;
  if (e >= v.num_experts) return;
  int const dst = e / v.local_experts;
  int const le = e % v.local_experts;
  bool const via_proxy = (v.proxy_mask >> dst) & 1u;
  void* me = v.peers[v.rank];
  void* dbase = v.peers[dst];
  uint32_t const* plan = plan_ptr(me, v, e);
  uint32_t const count = plan[0];
  uint32_t const pfx = plan[1];
  size_t const row_bytes = static_cast<size_t>(v.hidden) * v.elem_size;
  for (uint32_t i = b; i < count; i += fanout) {
    uint32_t const tk = plan[2 + i];
    uint32_t const t = tk & kMetaTokMask;
    char const* src_row =
        static_cast<char const*>(x) + static_cast<size_t>(t) * row_bytes;
    if (via_proxy) {
      // stage into packed local egress; the CPU proxy ships it. fp8
      // mode quantizes HERE so the wire carries 1B/elem + scales
      // (half the D2H + transport bytes of raw bf16).
      if (v.disp_fp8) {
        __shared__ float redp[256];
        block_quant_row_fp8(
            egress_x_fp8(me, v, pfx + i), egress_scale_fp8(me, v, pfx + i),
            reinterpret_cast<__hip_bfloat16 const*>(src_row), v.hidden,
            redp);
      } else {
        block_copy(egress_row(me, v, pfx + i), src_row, row_bytes);
      }
      if (threadIdx.x == 0) *egress_meta(me, v, pfx + i) = tk;
    } else {
      size_t const slot = static_cast<size_t>(v.rank) * v.max_tokens + i;
      if (v.disp_fp8) {
        __shared__ float red[256];
        block_quant_row_fp8(
            disp_x_ptr(dbase, v, le, slot),
            disp_scale_ptr(dbase, v, le, slot),
            reinterpret_cast<__hip_bfloat16 const*>(src_row), v.hidden,
            red);
      } else {
        block_copy(disp_x_ptr(dbase, v, le, slot), src_row, row_bytes);
      }
      if (threadIdx.x == 0) *disp_meta_ptr(dbase, v, le, slot) = tk;
    }
  }
}

// publish counts after the copy kernel's dispatch-boundary flush.
// Direct peers get the seq-tagged count release-store; proxy peers get a
// 32-byte TransferCmd pushed into the D2H ring (GPU-initiated transfer,
// executed by the CPU proxy — the IBGDA-replacement design).
__global__ void k_ep_dispatch_publish(EpView v) {
  for (int e = threadIdx.x; e < v.num_experts; e += blockDim.x) {
    int const dst = e / v.local_experts;
    if ((v.proxy_mask >> dst) & 1u) continue;
    int const le = e % v.local_experts;
    uint32_t const count = plan_ptr(v.peers[v.rank], v, e)[0];
    st_release_sys(disp_count_ptr(v.peers[dst], v, le, v.rank),
                   tag_count(v.seq, count));
  }
  __syncthreads();
  if (threadIdx.x == 0 && v.proxy_mask) {
    for (int e = 0; e < v.num_experts; ++e) {
      int const dst = e / v.local_experts;
      if (!((v.proxy_mask >> dst) & 1u)) continue;
      uint32_t const* plan = plan_ptr(v.peers[v.rank], v, e);
      TransferCmd c{static_cast<uint32_t>(CmdOp::kDispatchWrite),
                    static_cast<uint32_t>(v.seq), static_cast<uint64_t>(e),
                    plan[1], plan[0]};
      ring_push(v.ring, c);
    }
  }
}

// sharded wait: spin until every (local_expert, src) count carries this
// seq tag, then write the plain counts into out_counts
// [local_experts][world]. Grid-strided so 256-expert shapes spread the
// spinning across blocks instead of serializing in one.
__global__ void k_ep_dispatch_wait(EpView v,
                                   int* __restrict__ out_counts) {
  int const n = v.local_experts * v.world;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * blockDim.x) {
    int const le = i / v.world;
    int const src = i % v.world;
    uint64_t const* p = disp_count_ptr(v.peers[v.rank], v, le, src);
    uint64_t got = 0;
    for (uint64_t it = 0;; ++it) {
      got = ld_acquire_sys(p);
      if ((got >> 32) == v.seq) break;
      if (it > (1ull << 28)) {
        record_wait_err(v, kErrDispatchWait,
                        static_cast<uint64_t>(le) * v.world + src);
        break;
      }
      backoff();
    }
    out_counts[i] = static_cast<int>(got & 0xffffffffu);
  }
}

// ---------------------------------------------------------------------------
// combine send: grid = local_experts * world blocks; block (le, src) walks
// its received slots and writes each expert output row back into the source
// rank's (token, k) cell.
// ---------------------------------------------------------------------------

__global__ void k_ep_combine_send(EpView v,
                                  void const* __restrict__ expert_out) {
  int const npairs = v.local_experts * v.world;
  int const fanout = gridDim.x / npairs;
  int const pair = blockIdx.x / fanout;
  int const b = blockIdx.x % fanout;
  if (pair >= npairs) return;
  int const le = pair / v.world;
  int const src = pair % v.world;
  if ((v.proxy_mask >> src) & 1u) return;  // proxy ships these host-side
  void* me = v.peers[v.rank];
  uint64_t const tagged = *disp_count_ptr(me, v, le, src);
  // count published at dispatch time with this seq (combine reuses it)
  uint32_t const count = static_cast<uint32_t>(tagged & 0xffffffffu);
  void* sbase = v.peers[src];
  size_t const row_bytes = static_cast<size_t>(v.hidden) * v.elem_size;
  for (uint32_t i = b; i < count; i += fanout) {
    size_t const slot = static_cast<size_t>(src) * v.max_tokens + i;
    uint32_t const meta = *disp_meta_ptr(me, v, le, slot);
    uint32_t const t = meta & kMetaTokMask;
    uint32_t const k = meta >> 24;
    // defensive: corrupt meta (e.g. after an error-path early return)
    // must never index outside the return region
    if (t >= static_cast<uint32_t>(v.max_tokens) ||
        k >= static_cast<uint32_t>(v.topk))
      continue;
    char const* srcrow =
        static_cast<char const*>(expert_out) +
        ((static_cast<size_t>(le) * v.world * v.max_tokens + slot) *
         v.hidden) *
            v.elem_size;
    // expert outputs are read exactly once -> NT both sides
    block_copy_nt(comb_x_ptr(sbase, v, t, k), srcrow, row_bytes);
  }
}

// one block: publish "my returns to you are complete" to every rank
// (launched after k_ep_combine_send; dispatch boundary flushed the writes)
__global__ void k_ep_combine_signal(EpView v) {
  if (threadIdx.x < static_cast<unsigned>(v.world) &&
      !((v.proxy_mask >> threadIdx.x) & 1u))
    st_release_sys(comb_flag_ptr(v.peers[threadIdx.x], v, v.rank), v.seq);
}

__global__ void k_ep_combine_wait(EpView v) {
  if (threadIdx.x < static_cast<unsigned>(v.world)) {
    uint64_t const* p = comb_flag_ptr(v.peers[v.rank], v, threadIdx.x);
    for (uint64_t it = 0;; ++it) {
      if (ld_acquire_sys(p) >= v.seq) break;
      if (it > (1ull << 28)) {
        record_wait_err(v, kErrCombineWait, threadIdx.x);
        break;
      }
      backoff();
    }
  }
}

// proxy-ingress scatter: rows staged at ingress[row0..row0+count) are
// moved into their (token, k) comb_x cells (proxy RX path).
__global__ void k_ep_comb_scatter(EpView v, size_t row0, size_t count) {
  void* me = v.peers[v.rank];
  size_t const row_bytes = static_cast<size_t>(v.hidden) * v.elem_size;
  for (size_t i = blockIdx.x; i < count; i += gridDim.x) {
    uint32_t const meta = *ingress_meta(me, v, row0 + i);
    uint32_t const t = meta & kMetaTokMask;
    uint32_t const k = meta >> 24;
    if (t >= static_cast<uint32_t>(v.max_tokens) ||
        k >= static_cast<uint32_t>(v.topk))
      continue;
    block_copy(comb_x_ptr(me, v, t, k), ingress_row(me, v, row0 + i),
               row_bytes);
  }
}

// reduce: grid = num_tokens blocks; out[t] = sum_k w[t][k] * comb_x[t][k]
// in fp32, skipping masked (idx<0) entries.
template <typename T>
__global__ void k_ep_combine_reduce(EpView v, void* __restrict__ out,
                                    int64_t const* __restrict__ topk_idx,
                                    float const* __restrict__ topk_w,
                                    int num_tokens) {
  int const t = blockIdx.x;
  if (t >= num_tokens) return;
  void* me = v.peers[v.rank];
  size_t const vper = 16 / sizeof(T);
  size_t const nvec = static_cast<size_t>(v.hidden) / vper;
  auto* orow = reinterpret_cast<V16*>(static_cast<char*>(out) +
                                      static_cast<size_t>(t) * v.hidden *
                                          sizeof(T));
  for (size_t i = threadIdx.x; i < nvec; i += blockDim.x) {
    float acc[16 / sizeof(T)] = {};
    for (int k = 0; k < v.topk; ++k) {
      if (topk_idx[static_cast<size_t>(t) * v.topk + k] < 0) continue;
      float const w = topk_w[static_cast<size_t>(t) * v.topk + k];
      V16 const val =
          reinterpret_cast<V16 const*>(comb_x_ptr(me, v, t, k))[i];
#pragma unroll
      for (size_t j = 0; j < vper; ++j)
        acc[j] += w * static_cast<float>(reinterpret_cast<T const*>(
                          &val)[j]);
    }
    V16 r;
#pragma unroll
    for (size_t j = 0; j < vper; ++j)
      reinterpret_cast<T*>(&r)[j] = static_cast<T>(acc[j]);
    orow[i] = r;
  }
  // hidden not divisible by vec width: scalar tail
  size_t const tail = v.hidden - nvec * vper;
  if (tail && threadIdx.x < tail) {
    size_t const j = nvec * vper + threadIdx.x;
    float acc = 0.f;
    for (int k = 0; k < v.topk; ++k) {
      if (topk_idx[static_cast<size_t>(t) * v.topk + k] < 0) continue;
      acc += topk_w[static_cast<size_t>(t) * v.topk + k] *
             static_cast<float>(reinterpret_cast<T const*>(
                 comb_x_ptr(me, v, t, 0))[k * v.hidden + j]);
    }
    reinterpret_cast<T*>(orow)[j] = static_cast<T>(acc);
  }
}

// ---------------------------------------------------------------------------
// normal (rank-granular) mode: DeepEP HT dispatch/combine. Each token is
// shipped ONCE per destination rank (deduped over its top-k experts)
// with its topk row + weights; combine returns one processed row per
// received token to its source, which reduces over contributing ranks.
// Reference: ep/src/intranode.cu:186 (dispatch) / :722 (combine).
// ---------------------------------------------------------------------------

__global__ void k_nrm_plan(EpView v, int64_t const* __restrict__ topk_idx,
                           int num_tokens) {
  int const dst = blockIdx.x;  // destination rank
  extern __shared__ uint32_t smem[];
  uint32_t* prefix = smem;  // [blockDim.x + 1]
  uint32_t* plan = nrm_plan_ptr(v.peers[v.rank], v, dst);  // [count, toks..]

  int const seg = (num_tokens + blockDim.x - 1) / blockDim.x;
  int const t0 = threadIdx.x * seg;
  int const t1 = min(t0 + seg, num_tokens);
  auto hits_rank = [&](int t) {
    for (int k = 0; k < v.topk; ++k) {
      int64_t const e = topk_idx[static_cast<size_t>(t) * v.topk + k];
      if (e >= 0 && static_cast<int>(e / v.local_experts) == dst)
        return true;
    }
    return false;
  };
  uint32_t mine = 0;
  for (int t = t0; t < t1; ++t)
    if (hits_rank(t)) ++mine;
  prefix[threadIdx.x + 1] = mine;
  __syncthreads();
  if (threadIdx.x == 0) {
    prefix[0] = 0;
    for (unsigned i = 1; i <= blockDim.x; ++i) prefix[i] += prefix[i - 1];
    plan[0] = prefix[blockDim.x];
  }
  __syncthreads();
  uint32_t pos = prefix[threadIdx.x];
  for (int t = t0; t < t1; ++t)
    if (hits_rank(t)) plan[1 + pos++] = static_cast<uint32_t>(t);
}

__global__ void k_nrm_copy(EpView v, void const* __restrict__ x,
                           int64_t const* __restrict__ topk_idx,
                           float const* __restrict__ topk_w) {
  int const fanout = gridDim.x / v.world;
  int const dst = blockIdx.x / fanout;
  int const b = blockIdx.x % fanout;
  if (dst >= v.world) return;
  if ((v.proxy_mask >> dst) & 1u) return;  // normal mode is xGMI-only
  void* me = v.peers[v.rank];
  void* dbase = v.peers[dst];
  uint32_t const* plan = nrm_plan_ptr(me, v, dst);
  uint32_t const count = plan[0];
  size_t const row_bytes = static_cast<size_t>(v.hidden) * v.elem_size;
  for (uint32_t i = b; i < count; i += fanout) {
    uint32_t const t = plan[1 + i];
    block_copy(nrm_x_ptr(dbase, v, v.rank, i),
               static_cast<char const*>(x) +
                   static_cast<size_t>(t) * row_bytes,
               row_bytes);
    if (threadIdx.x == 0) *nrm_meta_ptr(dbase, v, v.rank, i) = t;
    // topk row (global expert ids) + weights travel with the token
    if (threadIdx.x < static_cast<unsigned>(v.topk)) {
      nrm_topk_ptr(dbase, v, v.rank, i)[threadIdx.x] =
          topk_idx[static_cast<size_t>(t) * v.topk + threadIdx.x];
      nrm_w_ptr(dbase, v, v.rank, i)[threadIdx.x] =
          topk_w ? topk_w[static_cast<size_t>(t) * v.topk + threadIdx.x]
                 : 0.f;
    }
  }
}

__global__ void k_nrm_publish(EpView v) {
  if (threadIdx.x < static_cast<unsigned>(v.world) &&
      !((v.proxy_mask >> threadIdx.x) & 1u)) {
    uint32_t const count = nrm_plan_ptr(v.peers[v.rank], v, threadIdx.x)[0];
    st_release_sys(nrm_count_ptr(v.peers[threadIdx.x], v, v.rank),
                   tag_count(v.seq, count));
  }
}

__global__ void k_nrm_wait(EpView v, int* __restrict__ out_counts) {
  if (threadIdx.x < static_cast<unsigned>(v.world)) {
    uint64_t const* p = nrm_count_ptr(v.peers[v.rank], v, threadIdx.x);
    uint64_t got = 0;
    for (uint64_t it = 0;; ++it) {
      got = ld_acquire_sys(p);
      if ((got >> 32) == v.seq) break;
      if (it > (1ull << 28)) {
        record_wait_err(v, kErrNrmWait, threadIdx.x);
        break;
      }
      backoff();
    }
    out_counts[threadIdx.x] = static_cast<int>(got & 0xffffffffu);
  }
}

// combine returns: block (src, b) walks rows received from `src` and
// writes the processed row back into src's per-(token, my_rank) cell.
__global__ void k_nrm_return(EpView v, void const* __restrict__ x) {
  int const fanout = gridDim.x / v.world;
  int const src = blockIdx.x / fanout;
  int const b = blockIdx.x % fanout;
  if (src >= v.world) return;
  if ((v.proxy_mask >> src) & 1u) return;
  void* me = v.peers[v.rank];
  uint64_t const tagged = *nrm_count_ptr(me, v, src);
  uint32_t const count = static_cast<uint32_t>(tagged & 0xffffffffu);
  void* sbase = v.peers[src];
  size_t const row_bytes = static_cast<size_t>(v.hidden) * v.elem_size;
  for (uint32_t i = b; i < count; i += fanout) {
    uint32_t const t = *nrm_meta_ptr(me, v, src, i);
    if (t >= static_cast<uint32_t>(v.max_tokens)) continue;
    block_copy_nt(nrm_ret_ptr(sbase, v, t, v.rank),
                  static_cast<char const*>(x) +
                      (static_cast<size_t>(src) * v.max_tokens + i) *
                          row_bytes,
                  row_bytes);
  }
}

__global__ void k_nrm_ret_signal(EpView v) {
  if (threadIdx.x < static_cast<unsigned>(v.world) &&
      !((v.proxy_mask >> threadIdx.x) & 1u))
    st_release_sys(nrm_flag_ptr(v.peers[threadIdx.x], v, v.rank), v.seq);
}

__global__ void k_nrm_ret_wait(EpView v) {
  if (threadIdx.x < static_cast<unsigned>(v.world)) {
    uint64_t const* p = nrm_flag_ptr(v.peers[v.rank], v, threadIdx.x);
    for (uint64_t it = 0;; ++it) {
      if (ld_acquire_sys(p) >= v.seq) return;
      if (it > (1ull << 28)) {
        record_wait_err(v, kErrNrmRetWait, threadIdx.x);
        return;
      }
      backoff();
    }
  }
}

// reduce: out[t] = sum over ranks r that received token t of ret[t][r]
// (fp32 accumulation; contributing set recomputed from topk_idx)
template <typename T>
__global__ void k_nrm_reduce(EpView v, void* __restrict__ out,
                             int64_t const* __restrict__ topk_idx,
                             int num_tokens) {
  int const t = blockIdx.x;
  if (t >= num_tokens) return;
  void* me = v.peers[v.rank];
  uint32_t contributes = 0;  // bitmask of ranks that got this token
  for (int k = 0; k < v.topk; ++k) {
    int64_t const e = topk_idx[static_cast<size_t>(t) * v.topk + k];
    if (e >= 0) contributes |= 1u << static_cast<int>(e / v.local_experts);
  }
  size_t const vper = 16 / sizeof(T);
  size_t const nvec = static_cast<size_t>(v.hidden) / vper;
  auto* orow = reinterpret_cast<V16*>(
      static_cast<char*>(out) +
      static_cast<size_t>(t) * v.hidden * sizeof(T));
  for (size_t i = threadIdx.x; i < nvec; i += blockDim.x) {
    float acc[16 / sizeof(T)] = {};
    for (int r = 0; r < v.world; ++r) {
      if (!((contributes >> r) & 1u)) continue;
      V16 const val = reinterpret_cast<V16 const*>(
          nrm_ret_ptr(me, v, t, r))[i];
#pragma unroll
      for (size_t j = 0; j < vper; ++j)
        acc[j] += static_cast<float>(reinterpret_cast<T const*>(&val)[j]);
    }
    V16 rvec;
#pragma unroll
    for (size_t j = 0; j < vper; ++j)
      reinterpret_cast<T*>(&rvec)[j] = static_cast<T>(acc[j]);
    orow[i] = rvec;
  }
  size_t const tail = v.hidden - nvec * vper;
  if (tail && threadIdx.x < tail) {
    size_t const j = nvec * vper + threadIdx.x;
    float acc = 0.f;
    for (int r = 0; r < v.world; ++r) {
      if (!((contributes >> r) & 1u)) continue;
      acc += static_cast<float>(
          reinterpret_cast<T const*>(nrm_ret_ptr(me, v, t, r))[j]);
    }
    static_cast<T*>(out)[static_cast<size_t>(t) * v.hidden + j] =
        static_cast<T>(acc);
  }
}

// ---------------------------------------------------------------------------
// proxy sync commands (ATOMIC/BARRIER/QUIET): one-thread push into the
// D2H ring + a paired single-block wait on the device-visible flag the
// proxy writes back (same trap-not-wedge discipline as the other waits).
// ---------------------------------------------------------------------------

__global__ void k_ep_sync_push(EpView v, uint32_t op, uint64_t a,
                               uint64_t b, uint64_t c) {
  if (threadIdx.x == 0 && blockIdx.x == 0)
    ring_push(v.ring, TransferCmd{op, static_cast<uint32_t>(v.seq), a, b, c});
}

__global__ void k_ep_sync_wait(EpView v, int idx, uint64_t seq) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    uint64_t const* p = sync_ptr(v.peers[v.rank], v, idx);
    for (uint64_t it = 0;; ++it) {
      if (ld_acquire_sys(p) >= seq) return;
      if (it > (1ull << 28)) {
        record_wait_err(v, kErrSyncWait, idx);
        return;
      }
      backoff();
    }
  }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------

void launch_ep_dispatch_send(const EpView& v, void const* x,
                             int64_t const* topk_idx, int num_tokens,
                             bool reuse_plan, hipStream_t s) {
  if (v.seq > 1 && v.world > 1)
    k_ep_consume_gate<<<1, 64, 0, s>>>(v, v.seq - 1);
  if (v.ring && v.proxy_mask) k_ep_ring_wait_empty<<<1, 64, 0, s>>>(v);
  if (!reuse_plan) {
    size_t const smem = 257 * sizeof(uint32_t);
    int const pgrid = v.num_experts * kPlanChunks;
    k_ep_dispatch_plan_count<<<pgrid, 256, 0, s>>>(v, topk_idx,
                                                   num_tokens);
    k_ep_plan_prefix<<<1, 256, 0, s>>>(v);
    k_ep_dispatch_plan_fill<<<pgrid, 256, smem, s>>>(v, topk_idx,
                                                     num_tokens);
  }
  k_ep_dispatch_copy<<<v.num_experts * fanout_for(v.num_experts), 256, 0,
                       s>>>(v, x);
  k_ep_dispatch_publish<<<1, 256, 0, s>>>(v);
}

void launch_ep_dispatch_recv(const EpView& v, int* out_counts,
                             hipStream_t s) {
  int const n = v.local_experts * v.world;
  int const grid = std::min(64, (n + 255) / 256 + 1);
  k_ep_dispatch_wait<<<grid, 256, 0, s>>>(v, out_counts);
  if (v.world > 1) k_ep_consume_signal<<<1, 64, 0, s>>>(v, v.seq);
}

void launch_ep_dispatch(const EpView& v, void const* x,
                        int64_t const* topk_idx, int num_tokens,
                        int* out_counts, hipStream_t s) {
  launch_ep_dispatch_send(v, x, topk_idx, num_tokens, false, s);
  launch_ep_dispatch_recv(v, out_counts, s);
}

void launch_ep_nrm_dispatch_send(const EpView& v, void const* x,
                                 int64_t const* topk_idx,
                                 float const* topk_w, int num_tokens,
                                 hipStream_t s) {
  if (v.seq > 1 && v.world > 1)
    k_ep_consume_gate<<<1, 64, 0, s>>>(v, v.seq - 1);
  size_t const smem = 257 * sizeof(uint32_t);
  k_nrm_plan<<<v.world, 256, smem, s>>>(v, topk_idx, num_tokens);
  k_nrm_copy<<<v.world * fanout_for(v.world), 256, 0, s>>>(v, x, topk_idx,
                                                           topk_w);
  k_nrm_publish<<<1, 64, 0, s>>>(v);
}

void launch_ep_nrm_dispatch_recv(const EpView& v, int* out_counts,
                                 hipStream_t s) {
  k_nrm_wait<<<1, 64, 0, s>>>(v, out_counts);
  if (v.world > 1) k_ep_consume_signal<<<1, 64, 0, s>>>(v, v.seq);
}

void launch_ep_nrm_combine_send(const EpView& v, void const* x,
                                hipStream_t s) {
  k_nrm_return<<<v.world * fanout_for(v.world), 256, 0, s>>>(v, x);
  k_nrm_ret_signal<<<1, 64, 0, s>>>(v);
}

void launch_ep_nrm_combine_recv(const EpView& v, void* out,
                                int64_t const* topk_idx, int num_tokens,
                                hipStream_t s) {
  k_nrm_ret_wait<<<1, 64, 0, s>>>(v);
  if (v.elem_size == 2) {
    k_nrm_reduce<__hip_bfloat16>
        <<<num_tokens, 256, 0, s>>>(v, out, topk_idx, num_tokens);
  } else {
    k_nrm_reduce<float>
        <<<num_tokens, 256, 0, s>>>(v, out, topk_idx, num_tokens);
  }
}

void launch_ep_barrier(const EpView& v, uint64_t seq, hipStream_t s) {
  k_ep_sync_push<<<1, 64, 0, s>>>(
      v, static_cast<uint32_t>(CmdOp::kBarrier), seq, 0, 0);
  k_ep_sync_wait<<<1, 64, 0, s>>>(v, 0, seq);
}

void launch_ep_quiet(const EpView& v, uint64_t seq, hipStream_t s) {
  k_ep_sync_push<<<1, 64, 0, s>>>(
      v, static_cast<uint32_t>(CmdOp::kQuiet), seq, 0, 0);
  k_ep_sync_wait<<<1, 64, 0, s>>>(v, 1, seq);
}

void launch_ep_atomic_add(const EpView& v, int dst, uint64_t off,
                          uint64_t value, hipStream_t s) {
  k_ep_sync_push<<<1, 64, 0, s>>>(
      v, static_cast<uint32_t>(CmdOp::kAtomicAdd),
      static_cast<uint64_t>(dst), off, value);
}

void launch_ep_comb_scatter(const EpView& v, size_t row0, size_t count,
                            hipStream_t s) {
  int const grid = count ? static_cast<int>(std::min<size_t>(count, 512)) : 1;
  k_ep_comb_scatter<<<grid, 256, 0, s>>>(v, row0, count);
}

void launch_ep_combine_send(const EpView& v, void const* expert_out,
                            hipStream_t s) {
  int const npairs = v.local_experts * v.world;
  k_ep_combine_send<<<npairs * fanout_for(npairs), 256, 0, s>>>(v,
                                                                expert_out);
  k_ep_combine_signal<<<1, 64, 0, s>>>(v);
}

void launch_ep_combine_finish(const EpView& v, void* out,
                              int64_t const* topk_idx, float const* topk_w,
                              int num_tokens, hipStream_t s) {
  k_ep_combine_wait<<<1, 64, 0, s>>>(v);
  if (v.elem_size == 2) {
    k_ep_combine_reduce<__hip_bfloat16>
        <<<num_tokens, 256, 0, s>>>(v, out, topk_idx, topk_w, num_tokens);
  } else {
    k_ep_combine_reduce<float>
        <<<num_tokens, 256, 0, s>>>(v, out, topk_idx, topk_w, num_tokens);
  }
}

}  // namespace ep
}  // namespace uccl
