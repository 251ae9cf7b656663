// Expert-parallel (DeepEP-compatible) symmetric-heap layout.
//
// Parity role: the reference's ep/ Buffer + internode_ll kernels
// (ep/src/uccl_ep.cc:348, ep/src/internode_ll.cu:62/:747), re-designed for
// a single MI355X node: no NIC, no proxy threads — dispatch tokens are
// written straight into the destination rank's HBM over xGMI, per-expert
// per-source slot arrays carry seq-tagged counts, and combine returns ride
// the same path back. 288 GB HBM3E per GPU makes the fully-partitioned
// slot layout ([local_expert][src_rank*max_tokens]) affordable, which
// removes all cross-source contention.
#pragma once

#include <cstddef>
#include <cstdint>

#include "../collective/layout.h"  // kMaxRanks

namespace uccl {
namespace ep {

// Runtime-sized layout; offsets are computed host-side once.
struct D2HRing;

struct EpView {
  int rank;
  int world;
  uint32_t proxy_mask;   // bit r set => traffic to rank r goes via proxy
  D2HRing* ring;         // device-visible pointer (proxy mode), else null
  int num_experts;     // global
  int local_experts;   // num_experts / world
  int topk;
  int hidden;          // elements per token
  int max_tokens;      // per source rank (DeepEP num_max_dispatch_tokens_per_rank)
  int elem_size;       // bytes per element of x (combine payload dtype)
  int disp_fp8;        // 1 => dispatch payload is fp8-e4m3 with per-128
                       //   scales (DeepEP LL fp8 mode); else raw elem_size
  int disp_elem;       // bytes per dispatched element (1 if disp_fp8)
  uint64_t seq;
  void* peers[kMaxRanks];

  // byte offsets into every rank's heap:
  size_t off_disp_count;  // u64 [local_experts][world]    (seq<<32 | count)
  size_t off_disp_x;      // [local_experts][world*max_tokens][hidden] disp_elem
  size_t off_disp_scale;  // f32 [local_experts][world*max_tokens][hidden/128]
                          //   (fp8 mode only)
  size_t off_disp_meta;   // u32 [local_experts][world*max_tokens]
                          //   meta = src_token_idx | (k << 24)
  size_t off_comb_flag;   // u64 [world]                   (seq)
  size_t off_sync;        // u64 [4]: [0] barrier seq, [1] quiet seq,
                          //          [2] atomic scratch, [3] reserved
  size_t off_consumed;    // u64 [world]: consumed[r] on MY heap = the
                          //   dispatch generation rank r has finished
                          //   reading from me (next-dispatch gate)
  size_t off_comb_x;      // [max_tokens][topk][hidden] elems
  size_t off_plan;        // u32 [num_experts][2 + max_tokens]  (private
                          //   per-rank scratch: count, egress prefix, list)
  size_t off_plan_chunks; // u32 [num_experts][kPlanChunks] per-(expert,
                          //   token-chunk) counts, then in-chunk bases
                          //   (the dispatch plan parallelizes over E*C
                          //   blocks; one block per expert left 97% of
                          //   the chip idle at small expert counts)
  size_t off_egress;      // packed egress rows [max_tokens*topk][hidden]
                          //   (proxy mode: rows destined to remote ranks)
  size_t off_egress_meta; // u32 [max_tokens*topk]
  size_t off_ingress;     // proxy RX staging rows (combine returns)
  size_t off_ingress_meta;

  // --- normal (rank-granular) mode: DeepEP's HT dispatch ships each
  // token ONCE per destination RANK (deduped across its top-k experts)
  // together with its topk row + weights; combine returns one row per
  // received token to its source (reference ep/src/intranode.cu:186
  // dispatch / :722 combine). Regions exist when with_normal.
  int with_normal;
  size_t off_nrm_count;   // u64 [world]            (seq<<32 | count)
  size_t off_nrm_x;       // [world][max_tokens][hidden] elem
  size_t off_nrm_meta;    // u32 [world][max_tokens]  src token idx
  size_t off_nrm_topk;    // i64 [world][max_tokens][topk]  (global ids)
  size_t off_nrm_w;       // f32 [world][max_tokens][topk]
  size_t off_nrm_plan;    // u32 [world][1 + max_tokens] private scratch
  size_t off_nrm_ret;     // [max_tokens][world][hidden] elem (returns)
  size_t off_nrm_flag;    // u64 [world]            (return seq)
  size_t heap_bytes;
};

constexpr uint32_t kMetaTokMask = 0x00ffffffu;
constexpr int kPlanChunks = 16;  // token chunks per expert in plan build

__host__ __device__ inline uint32_t* plan_chunks_ptr(void* base,
                                                     const EpView& v,
                                                     int e) {
  return reinterpret_cast<uint32_t*>(static_cast<char*>(base) +
                                     v.off_plan_chunks) +
         static_cast<size_t>(e) * kPlanChunks;
}

__host__ __device__ inline uint64_t* disp_count_ptr(void* base,
                                                    const EpView& v, int le,
                                                    int src) {
  return reinterpret_cast<uint64_t*>(static_cast<char*>(base) +
                                     v.off_disp_count) +
         static_cast<size_t>(le) * v.world + src;
}

__host__ __device__ inline char* disp_x_ptr(void* base, const EpView& v,
                                            int le, size_t slot) {
  return static_cast<char*>(base) + v.off_disp_x +
         ((static_cast<size_t>(le) * v.world * v.max_tokens + slot) *
          v.hidden) *
             v.disp_elem;
}

__host__ __device__ inline float* disp_scale_ptr(void* base, const EpView& v,
                                                 int le, size_t slot) {
  return reinterpret_cast<float*>(static_cast<char*>(base) +
                                  v.off_disp_scale) +
         (static_cast<size_t>(le) * v.world * v.max_tokens + slot) *
             (v.hidden / 128);
}

__host__ __device__ inline uint32_t* disp_meta_ptr(void* base,
                                                   const EpView& v, int le,
                                                   size_t slot) {
  return reinterpret_cast<uint32_t*>(static_cast<char*>(base) +
                                     v.off_disp_meta) +
         static_cast<size_t>(le) * v.world * v.max_tokens + slot;
}

__host__ __device__ inline uint32_t* plan_ptr(void* base, const EpView& v,
                                              int e) {
  return reinterpret_cast<uint32_t*>(static_cast<char*>(base) + v.off_plan) +
         static_cast<size_t>(e) * (2 + v.max_tokens);
}

__host__ __device__ inline char* egress_row(void* base, const EpView& v,
                                            size_t row) {
  return static_cast<char*>(base) + v.off_egress +
         row * static_cast<size_t>(v.hidden) * v.elem_size;
}

// fp8 proxy egress: quantized rows and their per-128 scales live as two
// SEPARATE contiguous streams inside the (bf16-sized, so 2x larger than
// needed) egress region — the receiver then lands each with one
// contiguous H2D per expert span instead of per-row unpacking.
__host__ __device__ inline char* egress_x_fp8(void* base, const EpView& v,
                                              size_t row) {
  return static_cast<char*>(base) + v.off_egress +
         row * static_cast<size_t>(v.hidden);
}

__host__ __device__ inline float* egress_scale_fp8(void* base,
                                                   const EpView& v,
                                                   size_t row) {
  size_t const nrows =
      static_cast<size_t>(v.max_tokens) * v.topk;  // region capacity
  return reinterpret_cast<float*>(static_cast<char*>(base) + v.off_egress +
                                  nrows * static_cast<size_t>(v.hidden)) +
         row * (v.hidden / 128);
}

__host__ __device__ inline uint32_t* egress_meta(void* base, const EpView& v,
                                                 size_t row) {
  return reinterpret_cast<uint32_t*>(static_cast<char*>(base) +
                                     v.off_egress_meta) +
         row;
}

__host__ __device__ inline char* ingress_row(void* base, const EpView& v,
                                             size_t row) {
  return static_cast<char*>(base) + v.off_ingress +
         row * static_cast<size_t>(v.hidden) * v.elem_size;
}

__host__ __device__ inline uint32_t* ingress_meta(void* base,
                                                  const EpView& v,
                                                  size_t row) {
  return reinterpret_cast<uint32_t*>(static_cast<char*>(base) +
                                     v.off_ingress_meta) +
         row;
}

// --- normal-mode helpers ---------------------------------------------------

__host__ __device__ inline uint64_t* nrm_count_ptr(void* base,
                                                   const EpView& v,
                                                   int src) {
  return reinterpret_cast<uint64_t*>(static_cast<char*>(base) +
                                     v.off_nrm_count) +
         src;
}

__host__ __device__ inline char* nrm_x_ptr(void* base, const EpView& v,
                                           int src, size_t row) {
  return static_cast<char*>(base) + v.off_nrm_x +
         ((static_cast<size_t>(src) * v.max_tokens + row) * v.hidden) *
             v.elem_size;
}

__host__ __device__ inline uint32_t* nrm_meta_ptr(void* base,
                                                  const EpView& v, int src,
                                                  size_t row) {
  return reinterpret_cast<uint32_t*>(static_cast<char*>(base) +
                                     v.off_nrm_meta) +
         static_cast<size_t>(src) * v.max_tokens + row;
}

__host__ __device__ inline int64_t* nrm_topk_ptr(void* base,
                                                 const EpView& v, int src,
                                                 size_t row) {
  return reinterpret_cast<int64_t*>(static_cast<char*>(base) +
                                    v.off_nrm_topk) +
         (static_cast<size_t>(src) * v.max_tokens + row) * v.topk;
}

__host__ __device__ inline float* nrm_w_ptr(void* base, const EpView& v,
                                            int src, size_t row) {
  return reinterpret_cast<float*>(static_cast<char*>(base) + v.off_nrm_w) +
         (static_cast<size_t>(src) * v.max_tokens + row) * v.topk;
}

__host__ __device__ inline uint32_t* nrm_plan_ptr(void* base,
                                                  const EpView& v, int dst) {
  return reinterpret_cast<uint32_t*>(static_cast<char*>(base) +
                                     v.off_nrm_plan) +
         static_cast<size_t>(dst) * (1 + v.max_tokens);
}

__host__ __device__ inline char* nrm_ret_ptr(void* base, const EpView& v,
                                             size_t tok, int src) {
  return static_cast<char*>(base) + v.off_nrm_ret +
         ((tok * v.world + src) * static_cast<size_t>(v.hidden)) *
             v.elem_size;
}

__host__ __device__ inline uint64_t* nrm_flag_ptr(void* base,
                                                  const EpView& v,
                                                  int src) {
  return reinterpret_cast<uint64_t*>(static_cast<char*>(base) +
                                     v.off_nrm_flag) +
         src;
}

__host__ __device__ inline uint64_t* comb_flag_ptr(void* base,
                                                   const EpView& v,
                                                   int src) {
  return reinterpret_cast<uint64_t*>(static_cast<char*>(base) +
                                     v.off_comb_flag) +
         src;
}

__host__ __device__ inline uint64_t* sync_ptr(void* base, const EpView& v,
                                              int idx) {
  return reinterpret_cast<uint64_t*>(static_cast<char*>(base) + v.off_sync) +
         idx;
}

__host__ __device__ inline uint64_t* consumed_ptr(void* base,
                                                  const EpView& v, int r) {
  return reinterpret_cast<uint64_t*>(static_cast<char*>(base) +
                                     v.off_consumed) +
         r;
}

__host__ __device__ inline char* comb_x_ptr(void* base, const EpView& v,
                                            size_t tok, int k) {
  return static_cast<char*>(base) + v.off_comb_x +
         ((tok * v.topk + k) * static_cast<size_t>(v.hidden)) * v.elem_size;
}

}  // namespace ep
}  // namespace uccl
