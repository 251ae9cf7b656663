// Device→host command ring for GPU-initiated proxy transfers.
//
// Parity role: the reference's ep/include/ring_buffer.cuh TransferCmd
// ring (GPU kernels push commands; pinned-host head/tail; CPU proxy
// drains and executes the network operations) — the portable replacement
// for NVIDIA-only IBGDA that defines UCCL-EP. Re-designed
// minimal for CDNA4: 32-byte commands, system-scope release/acquire on
// the tail, s_sleep backoff on a full ring.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

namespace uccl {
namespace ep {

enum class CmdOp : uint32_t {
  kNone = 0,
  kDispatchWrite = 1,  // a = global expert, b = egress row offset, c = count
  // parity with the reference proxy's command set
  // (ep/src/proxy.cpp:828 post_gpu_command op switch):
  kAtomicAdd = 2,  // a = dst rank, b = heap byte offset, c = add value
  kBarrier = 3,    // a = seq: CPU-side barrier across proxy peers
  kQuiet = 4,      // a = seq: all prior ring cmds fully shipped
  kConsume = 5,    // a = seq: tell proxied peers this generation's counts
                   // were consumed (their next dispatch may overwrite)
};

struct TransferCmd {
  uint32_t op;
  uint32_t seq32;
  uint64_t a, b, c;
};
static_assert(sizeof(TransferCmd) == 32, "cmd size");

constexpr uint32_t kRingSlots = 1024;

// Lives in pinned host memory; the device sees it through
// hipHostGetDevicePointer.
struct D2HRing {
  volatile uint64_t tail;  // device producer
  uint64_t pad0[7];
  volatile uint64_t head;  // host consumer
  uint64_t pad1[7];
  TransferCmd cmds[kRingSlots];
};

#if defined(__HIP_DEVICE_COMPILE__) || defined(__HIPCC__)
__device__ inline void ring_push(D2HRing* r, TransferCmd const& c) {
  uint64_t t = __hip_atomic_load(const_cast<uint64_t*>(&r->tail),
                                 __ATOMIC_RELAXED,
                                 __HIP_MEMORY_SCOPE_SYSTEM);
  // bounded wait for space, then trap (never wedge the GPU)
  for (uint64_t i = 0;; ++i) {
    uint64_t h = __hip_atomic_load(const_cast<uint64_t*>(&r->head),
                                   __ATOMIC_ACQUIRE,
                                   __HIP_MEMORY_SCOPE_SYSTEM);
    if (t - h < kRingSlots) break;
    if (i > (1ull << 28)) __builtin_trap();
    __builtin_amdgcn_s_sleep(2);
  }
  TransferCmd* slot = const_cast<TransferCmd*>(&r->cmds[t % kRingSlots]);
  __hip_atomic_store(reinterpret_cast<uint64_t*>(slot),
                     reinterpret_cast<uint64_t const*>(&c)[0],
                     __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
  __hip_atomic_store(reinterpret_cast<uint64_t*>(slot) + 1,
                     reinterpret_cast<uint64_t const*>(&c)[1],
                     __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
  __hip_atomic_store(reinterpret_cast<uint64_t*>(slot) + 2,
                     reinterpret_cast<uint64_t const*>(&c)[2],
                     __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
  __hip_atomic_store(reinterpret_cast<uint64_t*>(slot) + 3,
                     reinterpret_cast<uint64_t const*>(&c)[3],
                     __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
  __hip_atomic_store(const_cast<uint64_t*>(&r->tail), t + 1,
                     __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_SYSTEM);
}
#endif

}  // namespace ep
}  // namespace uccl
