// Lossless float codec for p2p transfers.
//
// Parity role: the reference's DietGPU-based compression layer
// (p2p/rdma/compression.{h,cc}; strategies kNone/kSplitOnly/kSplitEncode
// via UCCL_P2P_COMPRESS_STRATEGY, applied to >2MB float transfers).
// Re-designed for this stack: the hosts on the TCP/multipath plane stage
// through pinned host memory anyway, so the codec runs on CPU there —
// byte-plane split (sign/exponent bytes of fp32/fp16/bf16 are
// low-entropy on real model data) + DEFLATE on the compressible planes,
// raw passthrough for planes that don't shrink. Bitwise-lossless.
//
// Wire format (little-endian):
//   u32 magic 'UCZF' | u8 ver | u8 strategy | u8 elem_size | u8 dtype
//   u64 orig_bytes | u8 nplanes
//   nplanes x { u8 encoding (0 raw, 1 deflate) | u64 stored_bytes }
//   plane payloads, then the (bytes % elem_size) raw tail.

#pragma once

#include <cstddef>
#include <cstdint>
#include <string>

namespace uccl {
namespace p2p {
namespace comp {

enum Strategy : int {
  kNone = 0,        // framed raw passthrough
  kSplitOnly = 1,   // plane split, no entropy stage
  kSplitDeflate = 2 // plane split + deflate on exponent-bearing planes
};

// elem_size 2 or 4; dtype_code is carried opaquely (0 f32, 1 f16, 2 bf16,
// 3 raw bytes) so the receiver can reconstruct a typed tensor.
std::string compress(void const* data, size_t bytes, int elem_size,
                     int dtype_code, int strategy);

// original payload size encoded in `hdr` (first 24 bytes suffice)
size_t orig_bytes(void const* hdr, size_t avail);
int dtype_code(void const* hdr, size_t avail);

// decompress the full frame into out (out_cap >= orig_bytes). Returns
// bytes written; throws std::runtime_error on corrupt input.
size_t decompress(void const* frame, size_t frame_bytes, void* out,
                  size_t out_cap);

}  // namespace comp
}  // namespace p2p
}  // namespace uccl
