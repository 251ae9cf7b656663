// Host-side launcher declarations for the CDNA4 collective kernels.
#pragma once

#include <hip/hip_runtime.h>

#include "layout.h"

namespace uccl {

// kU8 is the raw-byte dtype used by copy-shaped collectives (broadcast,
// allgather, alltoall, send/recv) so any tensor dtype can ride them; the
// reducing collectives reject it.
enum class Dtype : int { kF32 = 0, kF16 = 1, kBF16 = 2, kI32 = 3, kU8 = 4, kF8E4M3 = 5, kI64 = 6, kF64 = 7 };

inline size_t dtype_size(Dtype d) {
  switch (d) {
    case Dtype::kF16:
    case Dtype::kBF16: return 2;
    case Dtype::kU8:
    case Dtype::kF8E4M3: return 1;
    case Dtype::kI64:
    case Dtype::kF64: return 8;
    default: return 4;
  }
}

// Reduction op for the reducing collectives. Values match the device-side
// red_apply OP template parameter (device/primitives.h). Avg/premulsum are
// host-side compositions: scale + kSum (see launch_scale).
enum class RedOp : int { kSum = 0, kProd = 1, kMin = 2, kMax = 3 };

void launch_copy(void* dst, void const* src, size_t bytes, hipStream_t s);
// in-place elementwise multiply by `factor` (avg = sum + scale 1/world;
// premulsum = scale by the rank's scalar + sum)
void launch_scale(void* data, size_t count, Dtype dt, double factor,
                  hipStream_t s);
void launch_oneshot_allreduce(const CommView& cv, void* out, size_t count,
                              Dtype dt, RedOp op, hipStream_t s);
void launch_twoshot_rs_push(const CommView& cv, size_t count, Dtype dt,
                            RedOp op, hipStream_t s);
void launch_twoshot_copyout(const CommView& cv, void* out, size_t bytes,
                            hipStream_t s);
void launch_twoshot_sym_rs(const CommView& cv, size_t uoff, size_t count,
                           Dtype dt, RedOp op, hipStream_t s);
void launch_twoshot_sym_push(const CommView& cv, size_t uoff, size_t count,
                             Dtype dt, hipStream_t s);
void launch_allgather_sym_push(const CommView& cv, void const* in,
                               size_t uoff, size_t slot_bytes,
                               hipStream_t s);
void launch_reducescatter_sym(const CommView& cv, size_t uoff, void* out,
                              size_t count, Dtype dt, RedOp op,
                              hipStream_t s);
void launch_alltoall_sym_push(const CommView& cv, void const* in,
                              size_t uoff, size_t chunk_bytes,
                              hipStream_t s);
void launch_ll_allreduce(const CommView& cv, void const* in, void* out,
                         size_t count, Dtype dt, RedOp op, hipStream_t s);
void launch_allgather_pull(const CommView& cv, void* out, size_t chunk_bytes,
                           hipStream_t s);
void launch_reducescatter_pull(const CommView& cv, void* out, size_t count,
                               Dtype dt, RedOp op, hipStream_t s);
void launch_broadcast_pull(const CommView& cv, int root, void* out,
                           size_t bytes, hipStream_t s);
void launch_alltoall_pull(const CommView& cv, void* out, size_t chunk_bytes,
                          hipStream_t s);
void launch_signal_wait(const CommView& cv, uint64_t val, unsigned wait_mask,
                        hipStream_t s);
void launch_barrier(const CommView& cv, hipStream_t s);
void launch_signal_peer(const CommView& cv, int dst, int ch, uint64_t val,
                        hipStream_t s);
void launch_wait_peer(const CommView& cv, int src, int ch, uint64_t val,
                      hipStream_t s);
void launch_copy_from_peer(const CommView& cv, int src, size_t src_off,
                           void* dst, size_t bytes, hipStream_t s);

}  // namespace uccl
