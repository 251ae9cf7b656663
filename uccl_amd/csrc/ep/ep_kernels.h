#pragma once

#include <hip/hip_runtime.h>

#include "ep_layout.h"

namespace uccl {
namespace ep {

void launch_ep_dispatch(const EpView& v, void const* x,
                        int64_t const* topk_idx, int num_tokens,
                        int* out_counts, hipStream_t s);
// phase-split halves of launch_ep_dispatch (SEND|RECV, recv-hook support)
void launch_ep_dispatch_send(const EpView& v, void const* x,
                             int64_t const* topk_idx, int num_tokens,
                             bool reuse_plan, hipStream_t s);
void launch_ep_dispatch_recv(const EpView& v, int* out_counts,
                             hipStream_t s);
void launch_ep_combine_send(const EpView& v, void const* expert_out,
                            hipStream_t s);
void launch_ep_combine_finish(const EpView& v, void* out,
                              int64_t const* topk_idx, float const* topk_w,
                              int num_tokens, hipStream_t s);
void launch_ep_comb_scatter(const EpView& v, size_t row0, size_t count,
                            hipStream_t s);
// normal (rank-granular) mode — DeepEP HT dispatch/combine
void launch_ep_nrm_dispatch_send(const EpView& v, void const* x,
                                 int64_t const* topk_idx,
                                 float const* topk_w, int num_tokens,
                                 hipStream_t s);
void launch_ep_nrm_dispatch_recv(const EpView& v, int* out_counts,
                                 hipStream_t s);
void launch_ep_nrm_combine_send(const EpView& v, void const* x,
                                hipStream_t s);
void launch_ep_nrm_combine_recv(const EpView& v, void* out,
                                int64_t const* topk_idx, int num_tokens,
                                hipStream_t s);
// proxy sync commands (D2H ring push + device flag wait)
void launch_ep_barrier(const EpView& v, uint64_t seq, hipStream_t s);
void launch_ep_quiet(const EpView& v, uint64_t seq, hipStream_t s);
void launch_ep_atomic_add(const EpView& v, int dst, uint64_t off,
                          uint64_t value, hipStream_t s);

}  // namespace ep
}  // namespace uccl
