// CDNA4 (gfx950) device-side primitives for the uccl_amd collective engine:
// system-scope flag signal/wait, LL (low-latency) 8-byte flagged packets,
// vectorized copy/reduce helpers. This is the uccl_amd analog of the
// reference's device channel substrate (experimental/lite/lite-collective
// core/*_device.hpp and ep/include/ring_buffer.cuh) re-designed for the
// CDNA4 memory model: wave64, per-XCD non-coherent L2, xGMI peer-HBM
// access. All cross-GPU ordering goes through __hip_atomic_* at
// __HIP_MEMORY_SCOPE_SYSTEM (release stores emit the L2 writeback, acquire
// loads emit the invalidate) — the verified-fence discipline the reference
// learned the hard way (ep/README.md:139 "aggressive atomics").
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_fp8.h>

#include <cstdint>

namespace uccl {
namespace device {

constexpr int kWave = 64;  // CDNA wavefront width (not 32!)

// ---------------------------------------------------------------------------
// System-scope atomics: the only legal way to order data across GPUs on xGMI.
// ---------------------------------------------------------------------------

__device__ __forceinline__ void st_release_sys(uint64_t* p, uint64_t v) {
  __hip_atomic_store(p, v, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_SYSTEM);
}

__device__ __forceinline__ void st_relaxed_sys(uint64_t* p, uint64_t v) {
  __hip_atomic_store(p, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
}

__device__ __forceinline__ uint64_t ld_acquire_sys(uint64_t const* p) {
  return __hip_atomic_load(p, __ATOMIC_ACQUIRE, __HIP_MEMORY_SCOPE_SYSTEM);
}

__device__ __forceinline__ uint64_t ld_relaxed_sys(uint64_t const* p) {
  return __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
}

// Scope "" means system scope in the amdgcn fence builtin.
__device__ __forceinline__ void fence_release_sys() {
  __builtin_amdgcn_fence(__ATOMIC_RELEASE, "");
}

__device__ __forceinline__ void fence_acquire_sys() {
  __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
}

// Nanosleep-style backoff for spin loops (CDNA has s_sleep, no __nanosleep).
__device__ __forceinline__ void backoff() { __builtin_amdgcn_s_sleep(2); }

// Bounded spin-wait: waits until *flag >= target.  Traps (aborting the
// kernel, and with it the process) instead of hanging the GPU forever —
// a deadlocked collective should kill the job, not wedge the node.
// rank/peer/ch identify WHICH wait stalled in the abort log (real-xGMI
// bring-up debuggability; printf costs nothing until the timeout path).
__device__ __forceinline__ void wait_flag_ge(uint64_t const* flag,
                                             uint64_t target, int rank = -1,
                                             int peer = -1, int ch = -1) {
  // ~2.7e8 polls with s_sleep backoff ≈ tens of seconds: far beyond any
  // sane collective wait, but short enough that a deadlock aborts the
  // kernel instead of wedging the GPU (gpurun strike avoidance).
  for (uint64_t i = 0; i < (1ull << 28); ++i) {
    if (ld_acquire_sys(flag) >= target) return;
    backoff();
  }
  if (threadIdx.x % 64 == 0)
    printf(
        "uccl: device wait TIMEOUT rank=%d peer=%d ch=%d want>=%llu "
        "last=%llu\n",
        rank, peer, ch, (unsigned long long)target,
        (unsigned long long)ld_relaxed_sys(flag));
  __builtin_trap();
}

// ---------------------------------------------------------------------------
// LL packet: 8 bytes = {u32 data, u32 flag}, written/read as one system-scope
// 64-bit atomic so data+flag arrive together over xGMI (same design point as
// NCCL LL / lite-collective allreduce_packet.cu, sized for wave64 here).
// ---------------------------------------------------------------------------

union LLPacket {
  uint64_t u64;
  struct {
    uint32_t data;
    uint32_t flag;
  };
};

__device__ __forceinline__ void ll_write(uint64_t* slot, uint32_t data,
                                         uint32_t flag) {
  LLPacket p;
  p.data = data;
  p.flag = flag;
  st_relaxed_sys(slot, p.u64);
}

__device__ __forceinline__ uint32_t ll_read(uint64_t const* slot,
                                            uint32_t flag) {
  LLPacket p;
  for (uint64_t i = 0; i < (1ull << 28); ++i) {
    p.u64 = ld_relaxed_sys(slot);
    if (p.flag == flag) return p.data;
    backoff();
  }
  __builtin_trap();
}

// ---------------------------------------------------------------------------
// 16-byte vector type for coalesced HBM/xGMI traffic (16 B/lane × wave64 =
// 1 KiB per instruction).
// ---------------------------------------------------------------------------

union alignas(16) V16 {
  uint4 u;
  float f32[4];
  __hip_bfloat16 bf16[8];
  __half f16[8];
  unsigned short u16[8];
};

// Non-temporal 16B load/store: streaming collectives touch each line once,
// so bypassing L2 wins ~16% HBM bandwidth on MI355X (tools/probe_copy:
// 5.32 -> 6.16 TB/s r+w at grid 8192x256). Visibility is unaffected — the
// dispatch-boundary release still drains all outstanding stores.
using v4u = unsigned __attribute__((ext_vector_type(4)));

__device__ __forceinline__ V16 nt_load(V16 const* p) {
  V16 r;
  *reinterpret_cast<v4u*>(&r) =
      __builtin_nontemporal_load(reinterpret_cast<v4u const*>(p));
  return r;
}

__device__ __forceinline__ void nt_store(V16* p, V16 v) {
  __builtin_nontemporal_store(*reinterpret_cast<v4u*>(&v),
                              reinterpret_cast<v4u*>(p));
}

// ---------------------------------------------------------------------------
// Reduction ops. OP values match uccl::RedOp host-side (kernels.h):
// 0=sum 1=prod 2=min 3=max. Applied in the accumulator domain (fp32 for the
// 16-bit/8-bit float types, native for int/i64/f64), so min/max are exact
// and prod rounds once at pack.
// ---------------------------------------------------------------------------

template <int OP, typename A>
__device__ __forceinline__ A red_apply(A a, A b) {
  if constexpr (OP == 1) return a * b;
  else if constexpr (OP == 2) return a < b ? a : b;
  else if constexpr (OP == 3) return a > b ? a : b;
  else return a + b;
}

// Elementwise fp32-accumulate add of two 16B vectors of T.
template <typename T>
__device__ __forceinline__ V16 v16_add(V16 a, V16 b);

template <>
__device__ __forceinline__ V16 v16_add<float>(V16 a, V16 b) {
  V16 r;
#pragma unroll
  for (int i = 0; i < 4; ++i) r.f32[i] = a.f32[i] + b.f32[i];
  return r;
}

template <>
__device__ __forceinline__ V16 v16_add<__hip_bfloat16>(V16 a, V16 b) {
  V16 r;
#pragma unroll
  for (int i = 0; i < 8; ++i)
    r.bf16[i] = __float2bfloat16(__bfloat162float(a.bf16[i]) +
                                 __bfloat162float(b.bf16[i]));
  return r;
}

template <>
__device__ __forceinline__ V16 v16_add<__half>(V16 a, V16 b) {
  V16 r;
#pragma unroll
  for (int i = 0; i < 8; ++i)
    r.f16[i] = __float2half(__half2float(a.f16[i]) + __half2float(b.f16[i]));
  return r;
}

template <>
__device__ __forceinline__ V16 v16_add<int>(V16 a, V16 b) {
  V16 r;
#pragma unroll
  for (int i = 0; i < 4; ++i)
    reinterpret_cast<int*>(r.f32)[i] = reinterpret_cast<int*>(a.f32)[i] +
                                       reinterpret_cast<int*>(b.f32)[i];
  return r;
}

// Accumulator lane type per element type: fp32 for the small float types
// (exact sums for world ≤ 8, one rounding at pack; min/max exact because
// every bf16/f16/fp8 value is representable in fp32), native for the rest.
template <typename T>
struct AccOf {
  using type = float;
};
template <>
struct AccOf<int> {
  using type = int;
};
template <>
struct AccOf<long long> {
  using type = long long;
};
template <>
struct AccOf<double> {
  using type = double;
};

// fp32-accumulator reduction over 16B vectors: unpack T lanes into the
// accumulator domain, apply OP per lane, pack once at the end (so bf16
// sums don't lose bits per-step with world_size up to 8). OP: see
// red_apply. All T here have a float conversion operator + a
// from-float constructor (fp8 pack saturates).
template <typename T, int OP = 0>
struct AccumV16 {
  using A = typename AccOf<T>::type;
  static constexpr int kN = 16 / sizeof(T);
  A v[kN];
  __device__ __forceinline__ void init(V16 a) {
    T const* e = reinterpret_cast<T const*>(&a);
#pragma unroll
    for (int i = 0; i < kN; ++i) v[i] = static_cast<A>(e[i]);
  }
  __device__ __forceinline__ void add(V16 a) {
    T const* e = reinterpret_cast<T const*>(&a);
#pragma unroll
    for (int i = 0; i < kN; ++i)
      v[i] = red_apply<OP>(v[i], static_cast<A>(e[i]));
  }
  __device__ __forceinline__ V16 pack() const {
    V16 r;
    T* e = reinterpret_cast<T*>(&r);
#pragma unroll
    for (int i = 0; i < kN; ++i) e[i] = static_cast<T>(v[i]);
    return r;
  }
};

// scalar accumulator type for ragged tails (exact for integers/doubles)
template <typename T>
struct TailAcc {
  using type = float;
};
template <>
struct TailAcc<int> {
  using type = long long;
};
template <>
struct TailAcc<long long> {
  using type = long long;
};
template <>
struct TailAcc<double> {
  using type = double;
};

}  // namespace device
}  // namespace uccl
