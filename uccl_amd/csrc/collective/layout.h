// Symmetric-heap layout shared by the host Communicator and the CDNA4
// collective kernels. Every rank allocates one identical hipMalloc block
// ("symmetric heap") and exports it over HIP IPC; kernels address peer
// heaps at identical offsets. Sized for MI355X: 288 GB HBM3E makes a
// few-hundred-MB resident scratch free, which buys zero-rendezvous staged
// collectives (copy-in → spray over 7 xGMI links → copy-out).
#pragma once

#include <cstddef>
#include <cstdint>

namespace uccl {

constexpr int kMaxRanks = 8;      // one MI355X node (8 OAMs, fullmesh xGMI)
constexpr int kMaxChannels = 16;  // independent flag channels

// --- offsets into each rank's symmetric heap --------------------------------
// [0, kFlagsBytes)                : u64 flags[kMaxRanks][kMaxChannels]
//                                   flags[w][c]: seq written by peer w
// [kLLOffset, +2*kLLSlotBytes*R)  : LL packet slots [2 parity][rank][bytes]
// [kScratchAOffset, +cap)         : staging for this rank's input chunks
// [kScratchBOffset, +cap)         : staging for reduced / gathered output
// ---------------------------------------------------------------------------

constexpr size_t kFlagsBytes = 64 * 1024;  // generously padded
constexpr size_t kLLSlotBytes = 1 * 1024 * 1024;  // per (parity, src-rank)
constexpr size_t kLLOffset = kFlagsBytes;
constexpr size_t kLLBytes = 2ull * kMaxRanks * kLLSlotBytes;  // 16 MB
// p2p send/recv staging: one slot per destination rank, so concurrent
// sends to different peers never share staging with each other or with
// the collective scratch regions.
constexpr size_t kP2PSlotBytes = 2 * 1024 * 1024;
constexpr size_t kP2POffset = kLLOffset + kLLBytes;
constexpr size_t kP2PBytes = static_cast<size_t>(kMaxRanks) * kP2PSlotBytes;
constexpr size_t kScratchAOffset = kP2POffset + kP2PBytes;

// LL packets carry 4B payload per 8B packet -> max LL message bytes:
constexpr size_t kLLMaxMsgBytes = kLLSlotBytes / 2;

inline size_t scratch_capacity(size_t heap_bytes) {
  return (heap_bytes - kScratchAOffset) / 2;
}

inline size_t scratch_b_offset(size_t heap_bytes) {
  return kScratchAOffset + scratch_capacity(heap_bytes);
}

// Kernel-argument view of the communicator (passed by value; <4KB).
//
// Scratch regions are parity double-buffered: collective call with host
// sequence `seq` (which advances by 2 per call) uses parity (seq>>1)&1.
// A rank can be at most one call ahead of any peer that has not yet
// signalled (every collective kernel begins with signal_all+wait_all), so
// two parities suffice to make cross-call scratch reuse race-free — the
// same skew argument as the LL packet slots.
struct CommView {
  int rank;
  int world;
  uint64_t seq;          // round-1 sequence value; round-2 uses seq+1
  int channel;           // flag channel for this collective
  size_t scratch_cap;    // per-parity scratch capacity in bytes
  size_t sa_off;         // byte offset of scratchA (this call's parity)
  size_t sb_off;         // byte offset of scratchB (this call's parity)
  void* peers[kMaxRanks];  // peer heap bases mapped into this process
};

}  // namespace uccl
