// CDNA4 xGMI collective kernels for uccl_amd.
//
// Design (MI355X-first, not a port): every rank owns a symmetric heap
// (layout.h) exported over HIP IPC. Collectives are staged: a vectorized
// copy-in to the local heap, then signal/wait flag rounds with system-scope
// atomics, then kernels that read/write *peer* HBM directly over the 7
// point-to-point xGMI links. Three algorithm families, selected by size:
//   - LL packet allreduce: single kernel, 8-byte flagged packets, no
//     barriers at all (latency path, small messages)
//   - one-shot fullmesh: each rank pulls all peers' staged input and
//     reduces locally (mid sizes)
//   - two-shot RS+AG ("push"): each rank reduces its 1/N shard from all
//     peers then pushes the result to every peer (bandwidth path; per-GPU
//     xGMI traffic ~2S(N-1)/N spread evenly across links)
// Functional parity targets the reference's lite-collective kernel family
// (experimental/lite/lite-collective/collective/*.cu) minus NVLS, which has
// no MI355X analog.
//
// Cross-kernel visibility relies on the HSA dispatch-completion system-scope
// release (kernel K's writes are system-visible before stream-ordered K+1
// runs); in-kernel signaling always uses explicit release fences.

#include <hip/hip_runtime.h>

#include <type_traits>

#include "kernels.h"
#include "../device/primitives.h"

namespace uccl {

using namespace uccl::device;

// ---------------------------------------------------------------------------
// Heap address helpers
// ---------------------------------------------------------------------------

__host__ __device__ inline uint64_t* flag_ptr(void* base, int writer, int ch) {
  return reinterpret_cast<uint64_t*>(base) + writer * kMaxChannels + ch;
}

__host__ __device__ inline char* ll_slot(void* base, int parity, int src) {
  return reinterpret_cast<char*>(base) + kLLOffset +
         (static_cast<size_t>(parity) * kMaxRanks + src) * kLLSlotBytes;
}

__device__ inline char* scratch_a(void* base, const CommView& cv) {
  return reinterpret_cast<char*>(base) + cv.sa_off;
}

__device__ inline char* scratch_b(void* base, const CommView& cv) {
  return reinterpret_cast<char*>(base) + cv.sb_off;
}

// Flag round = one tiny single-block kernel: signal `val` on our writer
// slot in every rank's heap, then spin until every rank in `wait_mask` has
// signalled `val` into ours. Runs as its own dispatch so that (a) worker
// kernels never spin — no co-residency/occupancy deadlock risk at any
// grid size, and (b) cross-GPU visibility is inherited from the HSA
// dispatch boundaries: the dispatch-completion system release publishes
// the preceding stream-ordered kernel's stores, and the next dispatch's
// system acquire invalidates stale caches before worker kernels read
// peer HBM.
__global__ void k_signal_wait(CommView cv, uint64_t val, unsigned wait_mask) {
  if (threadIdx.x < static_cast<unsigned>(cv.world)) {
    st_release_sys(flag_ptr(cv.peers[threadIdx.x], cv.rank, cv.channel), val);
    if ((wait_mask >> threadIdx.x) & 1u) {
      wait_flag_ge(flag_ptr(cv.peers[cv.rank], threadIdx.x, cv.channel), val,
                   cv.rank, static_cast<int>(threadIdx.x), cv.channel);
    }
  }
}

// ---------------------------------------------------------------------------
// Generic vectorized copy (grid-stride, 16 B/lane main loop + byte tail)
// ---------------------------------------------------------------------------

__global__ void k_copy(void* __restrict__ dst, void const* __restrict__ src,
                       size_t bytes) {
  size_t const nvec = bytes / 16;
  auto* d = reinterpret_cast<V16*>(dst);
  auto const* s = reinterpret_cast<V16 const*>(src);
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  for (; i < nvec; i += stride) nt_store(&d[i], nt_load(&s[i]));
  // byte tail
  size_t const tail = bytes & 15;
  if (tail && blockIdx.x == 0 && threadIdx.x < tail) {
    reinterpret_cast<char*>(dst)[bytes - tail + threadIdx.x] =
        reinterpret_cast<char const*>(src)[bytes - tail + threadIdx.x];
  }
}

// ---------------------------------------------------------------------------
// One-shot fullmesh allreduce: wait for all ranks' staged input, then each
// thread pulls the same offset from every peer's scratchA and reduces.
// ---------------------------------------------------------------------------

// Scalar-tail reduction helper shared by the reduce kernels: reduce element
// j of each rank's T-typed region (regions indexed per rank via `reg`).
template <typename T, int OP, typename RegionFn>
__device__ __forceinline__ T tail_reduce(const CommView& cv, RegionFn reg,
                                         size_t j) {
  using A = typename TailAcc<T>::type;
  A a = static_cast<A>(reinterpret_cast<T const*>(reg(0))[j]);
  for (int p = 1; p < cv.world; ++p)
    a = red_apply<OP>(a,
                      static_cast<A>(reinterpret_cast<T const*>(reg(p))[j]));
  return static_cast<T>(a);
}

template <typename T, int OP>
__global__ void k_oneshot_allreduce(CommView cv, void* __restrict__ out,
                                    size_t count) {
  size_t const vper = 16 / sizeof(T);
  size_t const nvec = count / vper;
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  for (; i < nvec; i += stride) {
    AccumV16<T, OP> acc;
    // fixed rank order => bitwise-identical results on every rank
    acc.init(nt_load(
        reinterpret_cast<V16 const*>(scratch_a(cv.peers[0], cv)) + i));
#pragma unroll 7
    for (int p = 1; p < cv.world; ++p) {
      acc.add(nt_load(
          reinterpret_cast<V16 const*>(scratch_a(cv.peers[p], cv)) + i));
    }
    nt_store(reinterpret_cast<V16*>(out) + i, acc.pack());
  }
  // scalar tail
  size_t const tail = count - nvec * vper;
  if (tail && blockIdx.x == 0 && threadIdx.x < tail) {
    size_t const j = nvec * vper + threadIdx.x;
    reinterpret_cast<T*>(out)[j] = tail_reduce<T, OP>(
        cv, [&](int p) { return scratch_a(cv.peers[p], cv); }, j);
  }
}

// ---------------------------------------------------------------------------
// Two-shot: phase 1 kernel — signal staged input ready (seq), wait, reduce
// my 1/N shard across all peers' scratchA, push the result into every
// rank's scratchB at my shard offset. The next kernel (stream-ordered)
// signals seq+1 after these pushes are dispatch-flushed.
// Shard s covers vec range [s*shard_nvec, ...); shards are by-vector so all
// remote traffic stays 16B-aligned. The scalar tail of the whole buffer is
// handled by the last shard owner.
// ---------------------------------------------------------------------------

template <typename T, int OP>
__global__ void k_twoshot_rs_push(CommView cv, size_t count) {
  size_t const vper = 16 / sizeof(T);
  size_t const nvec = count / vper;
  size_t const shard = (nvec + cv.world - 1) / cv.world;
  size_t const beg = cv.rank * shard;
  size_t const end = min(beg + shard, nvec);

  size_t i = beg + blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  for (; i < end; i += stride) {
    AccumV16<T, OP> acc;
    // fixed rank order => bitwise-identical results on every rank
    acc.init(nt_load(
        reinterpret_cast<V16 const*>(scratch_a(cv.peers[0], cv)) + i));
#pragma unroll 7
    for (int p = 1; p < cv.world; ++p) {
      acc.add(nt_load(
          reinterpret_cast<V16 const*>(scratch_a(cv.peers[p], cv)) + i));
    }
    V16 const r = acc.pack();
    // push to every rank's scratchB (spread across links; self included)
#pragma unroll 8
    for (int k = 0; k < cv.world; ++k) {
      int const p = (cv.rank + k) % cv.world;
      nt_store(reinterpret_cast<V16*>(scratch_b(cv.peers[p], cv)) + i, r);
    }
  }

  // scalar tail: owned by last rank
  size_t const tail = count - nvec * vper;
  if (tail && cv.rank == cv.world - 1 && blockIdx.x == 0 &&
      threadIdx.x < tail) {
    size_t const j = nvec * vper + threadIdx.x;
    T const r = tail_reduce<T, OP>(
        cv, [&](int p) { return scratch_a(cv.peers[p], cv); }, j);
    for (int p = 0; p < cv.world; ++p)
      reinterpret_cast<T*>(scratch_b(cv.peers[p], cv))[j] = r;
  }
}

// ---------------------------------------------------------------------------
// Symmetric (zero-copy) two-shot: the tensor lives in every rank's
// registered symmetric user region at the same offset, so there is no
// copy-in/copy-out at all. Phase 1 reduces my 1/N shard straight from all
// peers' user buffers into my parity scratchB; after the flag round
// (nobody reads user inputs any more), phase 2 pushes the reduced shard
// into every rank's user buffer in place.
// ---------------------------------------------------------------------------

template <typename T, int OP>
__global__ void k_twoshot_sym_rs(CommView cv, size_t uoff, size_t count) {
  size_t const vper = 16 / sizeof(T);
  size_t const nvec = count / vper;
  size_t const shard = (nvec + cv.world - 1) / cv.world;
  size_t const beg = cv.rank * shard;
  size_t const end = min(beg + shard, nvec);
  size_t i = beg + blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  auto* sb = reinterpret_cast<V16*>(scratch_b(cv.peers[cv.rank], cv));
  for (; i < end; i += stride) {
    AccumV16<T, OP> acc;
    acc.init(nt_load(reinterpret_cast<V16 const*>(
                 static_cast<char*>(cv.peers[0]) + uoff) + i));
#pragma unroll 7
    for (int p = 1; p < cv.world; ++p) {
      acc.add(nt_load(reinterpret_cast<V16 const*>(
                  static_cast<char*>(cv.peers[p]) + uoff) + i));
    }
    nt_store(&sb[i - beg], acc.pack());
  }
  size_t const tail = count - nvec * vper;
  if (tail && cv.rank == cv.world - 1 && blockIdx.x == 0 &&
      threadIdx.x < tail) {
    size_t const j = nvec * vper + threadIdx.x;
    // stash tail results after the vector shard in scratchB
    reinterpret_cast<T*>(sb)[(end - beg) * vper + threadIdx.x] =
        tail_reduce<T, OP>(
            cv, [&](int p) { return static_cast<char*>(cv.peers[p]) + uoff; },
            j);
  }
}

template <typename T>
__global__ void k_twoshot_sym_push(CommView cv, size_t uoff, size_t count) {
  size_t const vper = 16 / sizeof(T);
  size_t const nvec = count / vper;
  size_t const shard = (nvec + cv.world - 1) / cv.world;
  size_t const beg = cv.rank * shard;
  size_t const end = min(beg + shard, nvec);
  size_t i = beg + blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  auto const* sb =
      reinterpret_cast<V16 const*>(scratch_b(cv.peers[cv.rank], cv));
  for (; i < end; i += stride) {
    V16 const r = sb[i - beg];
#pragma unroll 8
    for (int k = 0; k < cv.world; ++k) {
      int const p = (cv.rank + k) % cv.world;
      nt_store(reinterpret_cast<V16*>(
                   static_cast<char*>(cv.peers[p]) + uoff) + i,
               r);
    }
  }
  size_t const tail = count - nvec * vper;
  if (tail && cv.rank == cv.world - 1 && blockIdx.x == 0 &&
      threadIdx.x < tail) {
    size_t const j = nvec * vper + threadIdx.x;
    T const r = reinterpret_cast<T const*>(sb)[(end - beg) * vper +
                                               threadIdx.x];
    for (int p = 0; p < cv.world; ++p)
      reinterpret_cast<T*>(static_cast<char*>(cv.peers[p]) + uoff)[j] = r;
  }
}

// Symmetric all_gather: `out` (symmetric, world*n elems) receives rank r's
// local `in` at slot r on EVERY rank — pure pushes, no read dependencies,
// so the only flag round is the completion wait after the push.
__global__ void k_allgather_sym_push(CommView cv, void const* __restrict__ in,
                                     size_t uoff, size_t slot_bytes) {
  size_t const nvec = slot_bytes / 16;
  size_t const tail = slot_bytes & 15;
  auto const* s = reinterpret_cast<V16 const*>(in);
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  for (int k = 0; k < cv.world; ++k) {
    int const p = (cv.rank + k) % cv.world;
    auto* d = reinterpret_cast<V16*>(static_cast<char*>(cv.peers[p]) + uoff +
                                     static_cast<size_t>(cv.rank) *
                                         slot_bytes);
    for (size_t j = i; j < nvec; j += stride) nt_store(&d[j], nt_load(&s[j]));
    if (tail && blockIdx.x == 0 && threadIdx.x < tail)
      reinterpret_cast<char*>(d)[slot_bytes - tail + threadIdx.x] =
          reinterpret_cast<char const*>(s)[slot_bytes - tail + threadIdx.x];
  }
}

// Symmetric all_to_all: `out` symmetric [world*n]; rank r pushes its
// input chunk p into peer p's out at slot r. Pure pushes + completion
// round, like the symmetric all_gather.
__global__ void k_alltoall_sym_push(CommView cv, void const* __restrict__ in,
                                    size_t uoff, size_t chunk_bytes) {
  size_t const nvec = chunk_bytes / 16;
  size_t const tail = chunk_bytes & 15;
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  for (int k = 0; k < cv.world; ++k) {
    int const p = (cv.rank + k) % cv.world;
    auto const* s = reinterpret_cast<V16 const*>(
        static_cast<char const*>(in) + static_cast<size_t>(p) * chunk_bytes);
    auto* d = reinterpret_cast<V16*>(static_cast<char*>(cv.peers[p]) + uoff +
                                     static_cast<size_t>(cv.rank) *
                                         chunk_bytes);
    for (size_t j = i; j < nvec; j += stride) nt_store(&d[j], nt_load(&s[j]));
    if (tail && blockIdx.x == 0 && threadIdx.x < tail)
      reinterpret_cast<char*>(d)[chunk_bytes - tail + threadIdx.x] =
          reinterpret_cast<char const*>(s)[chunk_bytes - tail + threadIdx.x];
  }
}

// Symmetric reduce_scatter: `in` symmetric [world*count]; out[i] =
// sum_p in_p[rank*count + i]. Entry barrier (inputs published) before the
// remote reads; exit barrier so callers may overwrite `in` afterwards.
template <typename T, int OP>
__global__ void k_reducescatter_sym(CommView cv, size_t uoff,
                                    void* __restrict__ out, size_t count) {
  size_t const vper = 16 / sizeof(T);
  size_t const nvec = count / vper;
  size_t const elem_off = static_cast<size_t>(cv.rank) * count;
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  for (; i < nvec; i += stride) {
    AccumV16<T, OP> acc;
    acc.init(nt_load(reinterpret_cast<V16 const*>(
        reinterpret_cast<T const*>(static_cast<char*>(cv.peers[0]) + uoff) +
        elem_off) + i));
    for (int p = 1; p < cv.world; ++p) {
      acc.add(nt_load(reinterpret_cast<V16 const*>(
          reinterpret_cast<T const*>(static_cast<char*>(cv.peers[p]) +
                                     uoff) +
          elem_off) + i));
    }
    nt_store(reinterpret_cast<V16*>(out) + i, acc.pack());
  }
  size_t const tail = count - nvec * vper;
  if (tail && blockIdx.x == 0 && threadIdx.x < tail) {
    size_t const j = nvec * vper + threadIdx.x;
    reinterpret_cast<T*>(out)[j] = tail_reduce<T, OP>(
        cv,
        [&](int p) {
          return reinterpret_cast<T const*>(
                     static_cast<char*>(cv.peers[p]) + uoff) +
                 elem_off;
        },
        j);
  }
}

// Phase 2 kernel — signal seq+1 (my pushes are visible: dispatch boundary),
// wait for everyone's pushes, copy assembled scratchB to the output.
__global__ void k_twoshot_copyout(CommView cv, void* __restrict__ out,
                                  size_t bytes) {
  char const* src = scratch_b(cv.peers[cv.rank], cv);
  size_t const nvec = bytes / 16;
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  auto* d = reinterpret_cast<V16*>(out);
  auto const* s = reinterpret_cast<V16 const*>(src);
  for (; i < nvec; i += stride) nt_store(&d[i], nt_load(&s[i]));
  size_t const tail = bytes & 15;
  if (tail && blockIdx.x == 0 && threadIdx.x < tail)
    reinterpret_cast<char*>(out)[bytes - tail + threadIdx.x] =
        src[bytes - tail + threadIdx.x];
}

// ---------------------------------------------------------------------------
// LL packet allreduce: single kernel, no staging, no flag rounds. Each
// thread reads its input vec (4B granules), writes {data, seq32} packets
// into every peer's LL slot for src=rank (parity-alternating buffers, skew
// bounded at 1 by construction — see docs/design.md), then reduces packets
// from all ranks out of its own LL region.
// ---------------------------------------------------------------------------

template <typename T, int OP>
__global__ void k_ll_allreduce(CommView cv, void const* __restrict__ in,
                               void* __restrict__ out, size_t count) {
  uint32_t const flag = static_cast<uint32_t>(cv.seq);
  // host seq advances by 2 per collective -> (seq>>1) alternates per call
  int const parity = static_cast<int>((cv.seq >> 1) & 1);
  size_t const bytes = count * sizeof(T);
  size_t const nw = (bytes + 3) / 4;  // number of 4B payload words
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;

  for (size_t w = i; w < nw; w += stride) {
    uint32_t d;
    if ((w + 1) * 4 <= bytes) {
      d = reinterpret_cast<uint32_t const*>(in)[w];
    } else {  // ragged last word
      d = 0;
      char const* cin = reinterpret_cast<char const*>(in);
      for (size_t b = w * 4; b < bytes; ++b)
        reinterpret_cast<char*>(&d)[b - w * 4] = cin[b];
    }
#pragma unroll 8
    for (int k = 0; k < cv.world; ++k) {
      int const p = (cv.rank + k) % cv.world;
      ll_write(reinterpret_cast<uint64_t*>(ll_slot(cv.peers[p], parity,
                                                   cv.rank)) + w,
               d, flag);
    }
  }

  for (size_t w = i; w < nw; w += stride) {
    float accf[4] = {0.f, 0.f, 0.f, 0.f};
    int acci[1] = {0};
    for (int p = 0; p < cv.world; ++p) {
      uint32_t const d = ll_read(
          reinterpret_cast<uint64_t const*>(
              ll_slot(cv.peers[cv.rank], parity, p)) + w,
          flag);
      // p==0 initializes (prod/min/max have no cheap identity)
      if constexpr (sizeof(T) == 4) {
        if constexpr (__is_same(T, int)) {
          int const x = static_cast<int>(d);
          acci[0] = p ? red_apply<OP>(acci[0], x) : x;
        } else {
          float const x = __uint_as_float(d);
          accf[0] = p ? red_apply<OP>(accf[0], x) : x;
        }
      } else if constexpr (sizeof(T) == 2) {  // two lanes per word
        T lo, hi;
        reinterpret_cast<uint16_t&>(lo) = d & 0xffff;
        reinterpret_cast<uint16_t&>(hi) = d >> 16;
        float const xl = static_cast<float>(lo), xh = static_cast<float>(hi);
        accf[0] = p ? red_apply<OP>(accf[0], xl) : xl;
        accf[1] = p ? red_apply<OP>(accf[1], xh) : xh;
      } else {  // 1-byte types (fp8): four lanes per word
#pragma unroll
        for (int b = 0; b < 4; ++b) {
          T e;
          reinterpret_cast<uint8_t&>(e) =
              static_cast<uint8_t>((d >> (8 * b)) & 0xff);
          float const x = static_cast<float>(e);
          accf[b] = p ? red_apply<OP>(accf[b], x) : x;
        }
      }
    }
    uint32_t r;
    if constexpr (sizeof(T) == 4) {
      if constexpr (__is_same(T, int)) {
        r = static_cast<uint32_t>(acci[0]);
      } else {
        r = __float_as_uint(accf[0]);
      }
    } else if constexpr (sizeof(T) == 2) {
      T lo = static_cast<T>(accf[0]), hi = static_cast<T>(accf[1]);
      r = reinterpret_cast<uint16_t&>(lo) |
          (static_cast<uint32_t>(reinterpret_cast<uint16_t&>(hi)) << 16);
    } else {
      r = 0;
#pragma unroll
      for (int b = 0; b < 4; ++b) {
        T e = static_cast<T>(accf[b]);
        r |= static_cast<uint32_t>(reinterpret_cast<uint8_t&>(e))
             << (8 * b);
      }
    }
    if ((w + 1) * 4 <= bytes) {
      reinterpret_cast<uint32_t*>(out)[w] = r;
    } else {
      char* cout = reinterpret_cast<char*>(out);
      for (size_t b = w * 4; b < bytes; ++b)
        cout[b] = reinterpret_cast<char const*>(&r)[b - w * 4];
    }
  }
}

// ---------------------------------------------------------------------------
// AllGather / ReduceScatter / Broadcast / AllToAll phase-2 kernels
// (phase 1 is always k_copy into scratchA).
// ---------------------------------------------------------------------------

// out[r*chunk .. ] = rank r's scratchA chunk, pulled from each peer.
__global__ void k_allgather_pull(CommView cv, void* __restrict__ out,
                                 size_t chunk_bytes) {
  size_t const nvec = chunk_bytes / 16;
  size_t const tail = chunk_bytes & 15;
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  for (int k = 0; k < cv.world; ++k) {
    int const p = (cv.rank + k) % cv.world;
    auto const* s = reinterpret_cast<V16 const*>(scratch_a(cv.peers[p], cv));
    auto* d = reinterpret_cast<V16*>(reinterpret_cast<char*>(out) +
                                     static_cast<size_t>(p) * chunk_bytes);
    for (size_t j = i; j < nvec; j += stride) nt_store(&d[j], nt_load(&s[j]));
    if (tail && blockIdx.x == 0 && threadIdx.x < tail)
      reinterpret_cast<char*>(d)[chunk_bytes - tail + threadIdx.x] =
          reinterpret_cast<char const*>(s)[chunk_bytes - tail + threadIdx.x];
  }
}

// out = sum over ranks of scratchA[p][rank*count .. +count] (my shard).
template <typename T, int OP>
__global__ void k_reducescatter_pull(CommView cv, void* __restrict__ out,
                                     size_t count) {
  size_t const vper = 16 / sizeof(T);
  size_t const nvec = count / vper;
  size_t const elem_off = static_cast<size_t>(cv.rank) * count;
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  for (; i < nvec; i += stride) {
    AccumV16<T, OP> acc;
    acc.init(nt_load(reinterpret_cast<V16 const*>(
        reinterpret_cast<T const*>(scratch_a(cv.peers[0], cv)) +
        elem_off) + i));
    for (int p = 1; p < cv.world; ++p) {
      acc.add(nt_load(reinterpret_cast<V16 const*>(
          reinterpret_cast<T const*>(scratch_a(cv.peers[p], cv)) +
          elem_off) + i));
    }
    nt_store(reinterpret_cast<V16*>(out) + i, acc.pack());
  }
  size_t const tail = count - nvec * vper;
  if (tail && blockIdx.x == 0 && threadIdx.x < tail) {
    size_t const j = nvec * vper + threadIdx.x;
    reinterpret_cast<T*>(out)[j] = tail_reduce<T, OP>(
        cv,
        [&](int p) {
          return reinterpret_cast<T const*>(scratch_a(cv.peers[p], cv)) +
                 elem_off;
        },
        j);
  }
}

// Broadcast pull: every rank copies root's scratchA into out. Root must
// have staged; only root's flag is awaited.
__global__ void k_broadcast_pull(CommView cv, int root, void* __restrict__ out,
                                 size_t bytes) {
  auto const* s = reinterpret_cast<V16 const*>(scratch_a(cv.peers[root], cv));
  auto* d = reinterpret_cast<V16*>(out);
  size_t const nvec = bytes / 16;
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  for (; i < nvec; i += stride) nt_store(&d[i], nt_load(&s[i]));
  size_t const tail = bytes & 15;
  if (tail && blockIdx.x == 0 && threadIdx.x < tail)
    reinterpret_cast<char*>(out)[bytes - tail + threadIdx.x] =
        reinterpret_cast<char const*>(s)[bytes - tail + threadIdx.x];
}

// AllToAll pull: rank r's output chunk p comes from peer p's scratchA at
// chunk offset r.
__global__ void k_alltoall_pull(CommView cv, void* __restrict__ out,
                                size_t chunk_bytes) {
  size_t const nvec = chunk_bytes / 16;
  size_t const tail = chunk_bytes & 15;
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  for (int k = 0; k < cv.world; ++k) {
    int const p = (cv.rank + k) % cv.world;
    auto const* s = reinterpret_cast<V16 const*>(
        scratch_a(cv.peers[p], cv) + static_cast<size_t>(cv.rank) * chunk_bytes);
    auto* d = reinterpret_cast<V16*>(reinterpret_cast<char*>(out) +
                                     static_cast<size_t>(p) * chunk_bytes);
    for (size_t j = i; j < nvec; j += stride) nt_store(&d[j], nt_load(&s[j]));
    if (tail && blockIdx.x == 0 && threadIdx.x < tail)
      reinterpret_cast<char*>(d)[chunk_bytes - tail + threadIdx.x] =
          reinterpret_cast<char const*>(s)[chunk_bytes - tail + threadIdx.x];
  }
}

// In-place elementwise scale (avg = sum + 1/world; premulsum = scale + sum).
// Multiplies in the accumulator domain (fp32 / double) like the reducers.
template <typename T>
__global__ void k_scale(T* __restrict__ p, size_t count, double factor) {
  using A = typename AccOf<T>::type;
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  for (; i < count; i += stride)
    p[i] = static_cast<T>(static_cast<A>(static_cast<A>(p[i]) *
                                         static_cast<A>(factor)));
}

// ---------------------------------------------------------------------------
// Barrier and point-to-point signal/wait (host-sequenced send/recv staging)
// ---------------------------------------------------------------------------

// Signal `val` into rank dst's flags[rank][ch]
__global__ void k_signal_peer(CommView cv, int dst, int ch, uint64_t val) {
  if (threadIdx.x == 0 && blockIdx.x == 0)
    st_release_sys(flag_ptr(cv.peers[dst], cv.rank, ch), val);
}

// Wait until our flags[src][ch] >= val
__global__ void k_wait_peer(CommView cv, int src, int ch, uint64_t val) {
  if (threadIdx.x == 0 && blockIdx.x == 0)
    wait_flag_ge(flag_ptr(cv.peers[cv.rank], src, ch), val, cv.rank, src,
                 ch);
}

// Copy from a peer's heap (+byte offset from heap base) into local memory.
__global__ void k_copy_from_peer(CommView cv, int src, size_t src_off,
                                 void* __restrict__ dst, size_t bytes) {
  auto const* s = reinterpret_cast<V16 const*>(
      reinterpret_cast<char*>(cv.peers[src]) + src_off);
  auto* d = reinterpret_cast<V16*>(dst);
  size_t const nvec = bytes / 16;
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  for (; i < nvec; i += stride) nt_store(&d[i], nt_load(&s[i]));
  size_t const tail = bytes & 15;
  if (tail && blockIdx.x == 0 && threadIdx.x < tail)
    reinterpret_cast<char*>(dst)[bytes - tail + threadIdx.x] =
        reinterpret_cast<char const*>(s)[bytes - tail + threadIdx.x];
}

// ---------------------------------------------------------------------------
// Launchers
// ---------------------------------------------------------------------------

static inline int grid_for(size_t bytes) {
  // memory-bound: probe_copy shows peak HBM bw at ~8192 workgroups
  size_t const want = (bytes / 16 + 255) / 256;
  size_t const g = want < 8 ? 8 : (want > 8192 ? 8192 : want);
  return static_cast<int>(g);
}

void launch_copy(void* dst, void const* src, size_t bytes, hipStream_t s) {
  k_copy<<<grid_for(bytes), 256, 0, s>>>(dst, src, bytes);
}

template <typename T>
static void l_scale(void* data, size_t count, double factor, hipStream_t s) {
  k_scale<T><<<grid_for(count * sizeof(T)), 256, 0, s>>>(
      static_cast<T*>(data), count, factor);
}

// runtime RedOp -> compile-time OP template argument
template <typename F>
static inline void op_switch(RedOp op, F&& f) {
  switch (op) {
    case RedOp::kProd: f(std::integral_constant<int, 1>{}); break;
    case RedOp::kMin: f(std::integral_constant<int, 2>{}); break;
    case RedOp::kMax: f(std::integral_constant<int, 3>{}); break;
    default: f(std::integral_constant<int, 0>{}); break;
  }
}

#define DT_DISPATCH(dt, fn, ...)                        \
  switch (dt) {                                         \
    case Dtype::kF32: fn<float>(__VA_ARGS__); break;    \
    case Dtype::kF16: fn<__half>(__VA_ARGS__); break;   \
    case Dtype::kBF16: fn<__hip_bfloat16>(__VA_ARGS__); break; \
    case Dtype::kI32: fn<int>(__VA_ARGS__); break;      \
    case Dtype::kU8: /* reducing collectives reject kU8 host-side */ break; \
    case Dtype::kF8E4M3: fn<__hip_fp8_e4m3>(__VA_ARGS__); break; \
    case Dtype::kI64: fn<long long>(__VA_ARGS__); break; \
    case Dtype::kF64: fn<double>(__VA_ARGS__); break; \
  }

template <typename T>
static void l_oneshot(const CommView& cv, void* out, size_t count, RedOp op,
                      hipStream_t s) {
  op_switch(op, [&](auto o) {
    k_oneshot_allreduce<T, decltype(o)::value>
        <<<grid_for(count * sizeof(T)), 256, 0, s>>>(cv, out, count);
  });
}

template <typename T>
static void l_twoshot_rs(const CommView& cv, size_t count, RedOp op,
                         hipStream_t s) {
  op_switch(op, [&](auto o) {
    k_twoshot_rs_push<T, decltype(o)::value>
        <<<grid_for(count * sizeof(T) / (cv.world ? cv.world : 1)), 256, 0,
           s>>>(cv, count);
  });
}

template <typename T>
static void l_ll(const CommView& cv, void const* in, void* out, size_t count,
                 RedOp op, hipStream_t s) {
  op_switch(op, [&](auto o) {
    k_ll_allreduce<T, decltype(o)::value>
        <<<grid_for(count * sizeof(T) * 2), 256, 0, s>>>(cv, in, out, count);
  });
}

template <typename T>
static void l_rs_pull(const CommView& cv, void* out, size_t count, RedOp op,
                      hipStream_t s) {
  op_switch(op, [&](auto o) {
    k_reducescatter_pull<T, decltype(o)::value>
        <<<grid_for(count * sizeof(T)), 256, 0, s>>>(cv, out, count);
  });
}

void launch_scale(void* data, size_t count, Dtype dt, double factor,
                  hipStream_t s) {
  DT_DISPATCH(dt, l_scale, data, count, factor, s);
}

void launch_oneshot_allreduce(const CommView& cv, void* out, size_t count,
                              Dtype dt, RedOp op, hipStream_t s) {
  DT_DISPATCH(dt, l_oneshot, cv, out, count, op, s);
}

void launch_twoshot_rs_push(const CommView& cv, size_t count, Dtype dt,
                            RedOp op, hipStream_t s) {
  DT_DISPATCH(dt, l_twoshot_rs, cv, count, op, s);
}

template <typename T>
static void l_sym_rs(const CommView& cv, size_t uoff, size_t count, RedOp op,
                     hipStream_t s) {
  op_switch(op, [&](auto o) {
    k_twoshot_sym_rs<T, decltype(o)::value>
        <<<grid_for(count * sizeof(T) / (cv.world ? cv.world : 1)), 256, 0,
           s>>>(cv, uoff, count);
  });
}

template <typename T>
static void l_sym_push(const CommView& cv, size_t uoff, size_t count,
                       hipStream_t s) {
  k_twoshot_sym_push<T>
      <<<grid_for(count * sizeof(T) / (cv.world ? cv.world : 1)), 256, 0,
         s>>>(cv, uoff, count);
}

void launch_twoshot_sym_rs(const CommView& cv, size_t uoff, size_t count,
                           Dtype dt, RedOp op, hipStream_t s) {
  DT_DISPATCH(dt, l_sym_rs, cv, uoff, count, op, s);
}

void launch_allgather_sym_push(const CommView& cv, void const* in,
                               size_t uoff, size_t slot_bytes,
                               hipStream_t s) {
  k_allgather_sym_push<<<grid_for(slot_bytes * cv.world), 256, 0, s>>>(
      cv, in, uoff, slot_bytes);
}

void launch_alltoall_sym_push(const CommView& cv, void const* in,
                              size_t uoff, size_t chunk_bytes,
                              hipStream_t s) {
  k_alltoall_sym_push<<<grid_for(chunk_bytes * cv.world), 256, 0, s>>>(
      cv, in, uoff, chunk_bytes);
}

template <typename T>
static void l_rs_sym2(const CommView& cv, size_t uoff, void* out,
                      size_t count, RedOp op, hipStream_t s) {
  op_switch(op, [&](auto o) {
    k_reducescatter_sym<T, decltype(o)::value>
        <<<grid_for(count * sizeof(T)), 256, 0, s>>>(cv, uoff, out, count);
  });
}

void launch_reducescatter_sym(const CommView& cv, size_t uoff, void* out,
                              size_t count, Dtype dt, RedOp op,
                              hipStream_t s) {
  DT_DISPATCH(dt, l_rs_sym2, cv, uoff, out, count, op, s);
}

void launch_twoshot_sym_push(const CommView& cv, size_t uoff, size_t count,
                             Dtype dt, hipStream_t s) {
  DT_DISPATCH(dt, l_sym_push, cv, uoff, count, s);
}

void launch_twoshot_copyout(const CommView& cv, void* out, size_t bytes,
                            hipStream_t s) {
  k_twoshot_copyout<<<grid_for(bytes), 256, 0, s>>>(cv, out, bytes);
}

void launch_ll_allreduce(const CommView& cv, void const* in, void* out,
                         size_t count, Dtype dt, RedOp op, hipStream_t s) {
  DT_DISPATCH(dt, l_ll, cv, in, out, count, op, s);
}

void launch_allgather_pull(const CommView& cv, void* out, size_t chunk_bytes,
                           hipStream_t s) {
  k_allgather_pull<<<grid_for(chunk_bytes * cv.world), 256, 0, s>>>(
      cv, out, chunk_bytes);
}

void launch_reducescatter_pull(const CommView& cv, void* out, size_t count,
                               Dtype dt, RedOp op, hipStream_t s) {
  DT_DISPATCH(dt, l_rs_pull, cv, out, count, op, s);
}

void launch_broadcast_pull(const CommView& cv, int root, void* out,
                           size_t bytes, hipStream_t s) {
  k_broadcast_pull<<<grid_for(bytes), 256, 0, s>>>(cv, root, out, bytes);
}

void launch_alltoall_pull(const CommView& cv, void* out, size_t chunk_bytes,
                          hipStream_t s) {
  k_alltoall_pull<<<grid_for(chunk_bytes * cv.world), 256, 0, s>>>(
      cv, out, chunk_bytes);
}

void launch_signal_wait(const CommView& cv, uint64_t val, unsigned wait_mask,
                        hipStream_t s) {
  k_signal_wait<<<1, 64, 0, s>>>(cv, val, wait_mask);
}

void launch_barrier(const CommView& cv, hipStream_t s) {
  unsigned const mask = (cv.world >= 32) ? 0xffffffffu
                                         : ((1u << cv.world) - 1u);
  k_signal_wait<<<1, 64, 0, s>>>(cv, cv.seq, mask);
}

void launch_signal_peer(const CommView& cv, int dst, int ch, uint64_t val,
                        hipStream_t s) {
  k_signal_peer<<<1, 64, 0, s>>>(cv, dst, ch, val);
}

void launch_wait_peer(const CommView& cv, int src, int ch, uint64_t val,
                      hipStream_t s) {
  k_wait_peer<<<1, 64, 0, s>>>(cv, src, ch, val);
}

void launch_copy_from_peer(const CommView& cv, int src, size_t src_off,
                           void* dst, size_t bytes, hipStream_t s) {
  k_copy_from_peer<<<grid_for(bytes), 256, 0, s>>>(cv, src, src_off, dst,
                                                   bytes);
}

}  // namespace uccl
