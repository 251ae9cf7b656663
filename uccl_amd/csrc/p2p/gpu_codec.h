// GPU lossless float codec: plane split + 64-lane interleaved rANS.
//
// Parity role: the reference's DietGPU integration — GPU-resident
// lossless compression inside the p2p transfer pipeline
// (/root/reference/p2p/rdma/compression.cc; strategies at
// p2p/README.md:83). Re-designed for CDNA4 instead of ported:
//
//   - byte-plane split of f32/bf16/f16 tensors (exponent bytes compress
//     far better than mantissa bytes on model data) as a grid-stride
//     kernel
//   - entropy stage = rANS (the same family DietGPU uses), 12-bit
//     normalized frequencies, 64 INTERLEAVED streams per 64KB block —
//     one stream per wave64 lane, so encode/decode are wave-parallel
//     with zero divergence (DietGPU interleaves 32 ways for warps)
//   - per-block raw fallback when entropy coding does not pay
//
// Wire format (little endian):
//   FileHdr { u32 magic, u32 nplanes, u32 elem_size, u32 block_bytes,
//             u64 orig_bytes }
//   per plane: PlaneHdr { u64 comp_bytes, u32 nblocks, u32 pad }
//              u64 block_off[nblocks]   (into this plane's payload)
//              payload: per block BlockHdr { u32 orig, u32 comp,
//                u32 mode(0 raw | 1 rans), u32 pad }
//                mode 1: u16 freq[256]; u32 lane_bytes[64]; lane streams
//                mode 0: raw bytes
//
// Everything here runs on the GPU; the host only sizes buffers and
// launches. A __host__-compiled mirror of the rANS math backs the CPU
// unit tests (tests/test_gpu_codec_cpu.py) so the coder logic is
// validated off-GPU too.
#pragma once

#include <hip/hip_runtime.h>

#include <cstddef>
#include <cstdint>
#include <vector>

namespace uccl {
namespace p2p {
namespace gpu {

constexpr uint32_t kMagic = 0x55475a31;  // "UGZ1"
constexpr uint32_t kProbBits = 12;
constexpr uint32_t kProbScale = 1u << kProbBits;
constexpr uint32_t kRansL = 1u << 23;
constexpr int kLanes = 64;                 // one rANS stream per wave lane
constexpr uint32_t kBlockBytes = 64 << 10;  // input block per workgroup

// worst case for one encoded block (raw fallback bounds it, plus hdrs)
constexpr size_t kBlockWorst =
    16 + 512 + 256 + kBlockBytes + kBlockBytes / 8;

// Upper bound for a whole compressed buffer of `bytes` input split into
// `nplanes` planes.
size_t compress_bound(size_t bytes, int nplanes);

// Compress `bytes` of device memory at `src` (elem_size in {1,2,4};
// nplanes == elem_size for split modes, 1 for no split). Returns the
// compressed size written to the device buffer `dst` (capacity
// dst_cap >= compress_bound). Synchronous on `stream`.
size_t compress(void const* src, size_t bytes, int elem_size, int nplanes,
                void* dst, size_t dst_cap, hipStream_t stream);

// Decompress a frame at device `src` into device `dst` (orig_cap bytes).
// Returns the decompressed size.
size_t decompress(void const* src, size_t src_bytes, void* dst,
                  size_t orig_cap, hipStream_t stream);

// CPU mirror round-trip of the rANS coder (CPU test tier).
bool host_rans_selftest(std::vector<uint8_t> const& data);

}  // namespace gpu
}  // namespace p2p
}  // namespace uccl
