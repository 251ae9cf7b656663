// GPU plane-split + 64-lane interleaved rANS codec (gpu_codec.h).
//
// Block pipeline (one 256-thread workgroup per 64KB input block):
//   encode: LDS histogram -> 12-bit frequency normalization -> 64 lanes
//           each rANS-encode their interleaved symbol sub-sequence in
//           reverse into per-lane scratch -> block assembled (raw
//           fallback when rANS does not pay) -> host-side exclusive scan
//           compacts blocks into the final frame
//   decode: LDS cum2sym alias table (4096 B) -> 64 lanes decode forward
//
// The rANS math is the public-domain byte-wise variant (state in
// [2^23, 2^31), byte renormalization); __host__ __device__ so the CPU
// test tier validates the coder logic without a GPU.

#include <hip/hip_runtime.h>

#include <cstring>
#include <vector>

#include "../core/log.h"
#include "gpu_codec.h"

namespace uccl {
namespace p2p {
namespace gpu {

namespace {

struct FileHdr {
  uint32_t magic;
  uint32_t nplanes;
  uint32_t elem_size;
  uint32_t block_bytes;
  uint64_t orig_bytes;
};

struct PlaneHdr {
  uint64_t comp_bytes;
  uint32_t nblocks;
  uint32_t pad;
};

struct BlockHdr {
  uint32_t orig;
  uint32_t comp;  // payload bytes after this header
  uint32_t mode;  // 0 raw, 1 rans
  uint32_t pad;
};

// --- rANS core (host+device) ----------------------------------------------

struct EncSym {
  uint16_t freq;
  uint16_t cum;
};

__host__ __device__ inline void rans_enc_put(uint32_t* x, uint8_t* buf,
                                             int* pos, EncSym s) {
  // emit bytes (reverse stream: pos decrements) until x fits
  uint32_t const x_max =
      ((kRansL >> kProbBits) << 8) * static_cast<uint32_t>(s.freq);
  uint32_t v = *x;
  while (v >= x_max) {
    buf[--(*pos)] = static_cast<uint8_t>(v & 0xff);
    v >>= 8;
  }
  *x = ((v / s.freq) << kProbBits) + (v % s.freq) + s.cum;
}

__host__ __device__ inline void rans_enc_flush(uint32_t x, uint8_t* buf,
                                               int* pos) {
  buf[--(*pos)] = static_cast<uint8_t>(x >> 0);
  buf[--(*pos)] = static_cast<uint8_t>(x >> 8);
  buf[--(*pos)] = static_cast<uint8_t>(x >> 16);
  buf[--(*pos)] = static_cast<uint8_t>(x >> 24);
}

__host__ __device__ inline uint32_t rans_dec_init(uint8_t const* buf,
                                                  int* pos) {
  uint32_t x = 0;
  x |= static_cast<uint32_t>(buf[(*pos)++]) << 24;
  x |= static_cast<uint32_t>(buf[(*pos)++]) << 16;
  x |= static_cast<uint32_t>(buf[(*pos)++]) << 8;
  x |= static_cast<uint32_t>(buf[(*pos)++]) << 0;
  return x;
}

__host__ __device__ inline uint8_t rans_dec_get(uint32_t* x,
                                                uint8_t const* buf,
                                                int* pos,
                                                uint16_t const* freq,
                                                uint16_t const* cum,
                                                uint8_t const* cum2sym) {
  uint32_t const slot = *x & (kProbScale - 1);
  uint8_t const s = cum2sym[slot];
  uint32_t v = static_cast<uint32_t>(freq[s]) * (*x >> kProbBits) + slot -
               cum[s];
  while (v < kRansL) v = (v << 8) | buf[(*pos)++];
  *x = v;
  return s;
}

// Normalize a 256-entry histogram to sum kProbScale with every nonzero
// count >= 1 (serial; 256 entries).
__host__ __device__ inline void normalize_freqs(uint32_t const* hist,
                                                uint32_t total,
                                                uint16_t* freq) {
  uint32_t assigned = 0;
  int nz = 0;
  for (int i = 0; i < 256; ++i)
    if (hist[i]) ++nz;
  for (int i = 0; i < 256; ++i) {
    if (!hist[i]) {
      freq[i] = 0;
      continue;
    }
    uint64_t f = (static_cast<uint64_t>(hist[i]) * kProbScale) / total;
    if (f == 0) f = 1;
    freq[i] = static_cast<uint16_t>(f);
    assigned += f;
  }
  // settle the rounding drift: donate surplus to / steal deficit from
  // the currently-largest bucket, never letting any nonzero bucket hit
  // zero (heavy-tailed histograms can make a single adjustment
  // underflow)
  int32_t diff = static_cast<int32_t>(kProbScale) -
                 static_cast<int32_t>(assigned);
  while (diff != 0) {
    int big = 0;
    for (int i = 1; i < 256; ++i)
      if (freq[i] > freq[big]) big = i;
    if (diff > 0) {
      freq[big] = static_cast<uint16_t>(freq[big] + diff);
      diff = 0;
    } else {
      int32_t const take =
          diff < -(static_cast<int32_t>(freq[big]) - 1)
              ? -(static_cast<int32_t>(freq[big]) - 1)
              : diff;
      freq[big] = static_cast<uint16_t>(freq[big] + take);
      diff -= take;
    }
  }
  (void)nz;
}

// --- kernels ---------------------------------------------------------------

// byte-plane split: out[pl][i] = in[i*elem_size + pl]
__global__ void k_plane_split(uint8_t const* __restrict__ in,
                              uint8_t* __restrict__ out, size_t elems,
                              int elem_size) {
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  for (; i < elems; i += stride)
    for (int pl = 0; pl < elem_size; ++pl)
      out[static_cast<size_t>(pl) * elems + i] = in[i * elem_size + pl];
}

__global__ void k_plane_merge(uint8_t const* __restrict__ in,
                              uint8_t* __restrict__ out, size_t elems,
                              int elem_size) {
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  for (; i < elems; i += stride)
    for (int pl = 0; pl < elem_size; ++pl)
      out[i * elem_size + pl] = in[static_cast<size_t>(pl) * elems + i];
}

// Encode one block per workgroup into scratch[blockIdx * kBlockWorst];
// comp_sizes[b] = bytes produced (header included).
// per-lane encode scratch: capacity per lane for a full 64KB block
constexpr uint32_t kPerLaneCap = 2 * (kBlockBytes / kLanes) + 16;

__global__ void __launch_bounds__(256) k_rans_encode(
    uint8_t const* __restrict__ src, size_t total_bytes,
    uint8_t* __restrict__ scratch, uint8_t* __restrict__ lane_scratch,
    uint32_t* __restrict__ comp_sizes) {
  __shared__ uint32_t hist[256];
  __shared__ uint16_t freq[256];
  __shared__ uint16_t cum[257];
  __shared__ uint32_t lane_len[kLanes];

  size_t const b0 = static_cast<size_t>(blockIdx.x) * kBlockBytes;
  if (b0 >= total_bytes) return;
  uint32_t const n =
      static_cast<uint32_t>(min(static_cast<size_t>(kBlockBytes),
                                total_bytes - b0));
  uint8_t const* in = src + b0;
  uint8_t* out = scratch + static_cast<size_t>(blockIdx.x) * kBlockWorst;

  hist[threadIdx.x] = 0;
  if (threadIdx.x < 1) {
    // (256 threads; only [0,256) slots — all covered by the line above)
  }
  __syncthreads();
  for (uint32_t i = threadIdx.x; i < n; i += blockDim.x)
    atomicAdd(&hist[in[i]], 1u);
  __syncthreads();
  if (threadIdx.x == 0) {
    normalize_freqs(hist, n, freq);
    cum[0] = 0;
    for (int i = 0; i < 256; ++i)
      cum[i + 1] = cum[i] + freq[i];
  }
  __syncthreads();

  // 64 interleaved streams: lane l encodes symbols l, l+64, ... in
  // REVERSE into its slice of the SEPARATE lane scratch (a shared
  // buffer would overlap the assembly region and race the compaction)
  uint32_t const per_lane_cap = kPerLaneCap;
  uint8_t* lane_base = lane_scratch + static_cast<size_t>(blockIdx.x) *
                                          kLanes * kPerLaneCap;
  if (threadIdx.x < kLanes) {
    int const l = threadIdx.x;
    uint8_t* buf = lane_base + static_cast<size_t>(l) * per_lane_cap;
    int pos = static_cast<int>(per_lane_cap);
    uint32_t x = kRansL;
    // highest index owned by this lane, walking down
    int32_t const nsym = (static_cast<int32_t>(n) - 1 - l) / kLanes + 1;
    for (int32_t k = nsym - 1; k >= 0; --k) {
      uint8_t const s = in[k * kLanes + l];
      rans_enc_put(&x, buf, &pos, EncSym{freq[s], cum[s]});
    }
    rans_enc_flush(x, buf, &pos);
    lane_len[l] = per_lane_cap - static_cast<uint32_t>(pos);
  }
  __syncthreads();

  // total compressed payload
  __shared__ uint32_t tot;
  if (threadIdx.x == 0) {
    uint32_t t = 512 + kLanes * 4;  // freq table + lane lengths
    for (int l = 0; l < kLanes; ++l) t += lane_len[l];
    tot = t;
  }
  __syncthreads();

  if (tot >= n) {
    // raw fallback
    __syncthreads();
    if (threadIdx.x == 0) {
      BlockHdr h{n, n, 0, 0};
      memcpy(out, &h, sizeof(h));
      comp_sizes[blockIdx.x] = sizeof(BlockHdr) + n;
    }
    __syncthreads();
    for (uint32_t i = threadIdx.x; i < n; i += blockDim.x)
      out[sizeof(BlockHdr) + i] = in[i];
    return;
  }

  // assemble: hdr | freq[256]u16 | lane_len[64]u32 | lane streams
  if (threadIdx.x == 0) {
    BlockHdr h{n, tot, 1, 0};
    memcpy(out, &h, sizeof(h));
    memcpy(out + sizeof(BlockHdr), freq, 512);
    memcpy(out + sizeof(BlockHdr) + 512, lane_len, kLanes * 4);
  }
  __syncthreads();
  // compact lane streams (serial offsets, parallel byte copies)
  __shared__ uint32_t lane_off[kLanes + 1];
  if (threadIdx.x == 0) {
    lane_off[0] = sizeof(BlockHdr) + 512 + kLanes * 4;
    for (int l = 0; l < kLanes; ++l)
      lane_off[l + 1] = lane_off[l] + lane_len[l];
  }
  __syncthreads();
  for (int l = 0; l < kLanes; ++l) {
    uint8_t const* sbuf = lane_base + static_cast<size_t>(l) * per_lane_cap +
                          (per_lane_cap - lane_len[l]);
    for (uint32_t i = threadIdx.x; i < lane_len[l]; i += blockDim.x)
      out[lane_off[l] + i] = sbuf[i];
  }
  __syncthreads();
  if (threadIdx.x == 0)
    comp_sizes[blockIdx.x] = sizeof(BlockHdr) + tot;
}

// gather encoded blocks into the final contiguous plane payload
__global__ void k_compact(uint8_t const* __restrict__ scratch,
                          uint32_t const* __restrict__ comp_sizes,
                          uint64_t const* __restrict__ offs,
                          uint8_t* __restrict__ dst, int nblocks) {
  int const b = blockIdx.x;
  if (b >= nblocks) return;
  uint8_t const* s = scratch + static_cast<size_t>(b) * kBlockWorst;
  uint8_t* d = dst + offs[b];
  uint32_t const n = comp_sizes[b];
  for (uint32_t i = threadIdx.x; i < n; i += blockDim.x) d[i] = s[i];
}

__global__ void __launch_bounds__(256) k_rans_decode(
    uint8_t const* __restrict__ payload,
    uint64_t const* __restrict__ offs, uint8_t* __restrict__ dst,
    size_t total_bytes, int nblocks) {
  __shared__ uint16_t freq[256];
  __shared__ uint16_t cum[257];
  __shared__ uint8_t cum2sym[kProbScale];
  __shared__ uint32_t lane_off[kLanes + 1];

  int const b = blockIdx.x;
  if (b >= nblocks) return;
  uint8_t const* blk = payload + offs[b];
  BlockHdr h;
  memcpy(&h, blk, sizeof(h));
  uint8_t* out = dst + static_cast<size_t>(b) * kBlockBytes;

  if (h.mode == 0) {
    for (uint32_t i = threadIdx.x; i < h.orig; i += blockDim.x)
      out[i] = blk[sizeof(BlockHdr) + i];
    return;
  }

  if (threadIdx.x == 0) {
    memcpy(freq, blk + sizeof(BlockHdr), 512);
    cum[0] = 0;
    for (int i = 0; i < 256; ++i) cum[i + 1] = cum[i] + freq[i];
    uint32_t const* ll = reinterpret_cast<uint32_t const*>(
        blk + sizeof(BlockHdr) + 512);
    lane_off[0] = sizeof(BlockHdr) + 512 + kLanes * 4;
    for (int l = 0; l < kLanes; ++l) lane_off[l + 1] = lane_off[l] + ll[l];
  }
  __syncthreads();
  // alias table: cum2sym[slot] = symbol
  for (int s = 0; s < 256; ++s) {
    for (uint32_t i = cum[s] + threadIdx.x; i < cum[s + 1];
         i += blockDim.x)
      cum2sym[i] = static_cast<uint8_t>(s);
  }
  __syncthreads();

  if (threadIdx.x < kLanes) {
    int const l = threadIdx.x;
    uint8_t const* buf = blk;
    int pos = static_cast<int>(lane_off[l]);
    uint32_t x = rans_dec_init(buf, &pos);
    int32_t const nsym =
        (static_cast<int32_t>(h.orig) - 1 - l) / kLanes + 1;
    for (int32_t k = 0; k < nsym; ++k)
      out[k * kLanes + l] =
          rans_dec_get(&x, buf, &pos, freq, cum, cum2sym);
  }
}

size_t align8(size_t x) { return (x + 7) & ~size_t(7); }

// device scratch pool per call (simple: allocate/free per call; codec
// calls move MBs, the alloc cost is noise)
struct DevBuf {
  void* p = nullptr;
  ~DevBuf() {
    if (p) (void)hipFree(p);
  }
  void* alloc(size_t n) {
    UCCL_CHECK_HIP(hipMalloc(&p, n));
    return p;
  }
};

// encode one contiguous plane; returns compressed plane size (payload +
// block table + PlaneHdr), writing at dst.
size_t encode_plane(uint8_t const* src, size_t bytes, uint8_t* dst,
                    hipStream_t stream) {
  int const nblocks =
      static_cast<int>((bytes + kBlockBytes - 1) / kBlockBytes);
  DevBuf scratch_b, lanes_b, sizes_b, offs_b;
  auto* scratch = static_cast<uint8_t*>(
      scratch_b.alloc(static_cast<size_t>(nblocks) * kBlockWorst));
  auto* lanescr = static_cast<uint8_t*>(lanes_b.alloc(
      static_cast<size_t>(nblocks) * kLanes * kPerLaneCap));
  auto* sizes = static_cast<uint32_t*>(
      sizes_b.alloc(sizeof(uint32_t) * nblocks));
  auto* offs = static_cast<uint64_t*>(
      offs_b.alloc(sizeof(uint64_t) * nblocks));
  k_rans_encode<<<nblocks, 256, 0, stream>>>(src, bytes, scratch, lanescr,
                                             sizes);
  // host-side exclusive scan of comp sizes (nblocks is small)
  std::vector<uint32_t> hsizes(nblocks);
  UCCL_CHECK_HIP(hipMemcpyAsync(hsizes.data(), sizes,
                                sizeof(uint32_t) * nblocks,
                                hipMemcpyDeviceToHost, stream));
  UCCL_CHECK_HIP(hipStreamSynchronize(stream));
  std::vector<uint64_t> hoffs(nblocks);
  uint64_t run = 0;
  for (int i = 0; i < nblocks; ++i) {
    hoffs[i] = run;
    // 8-align every block so the decoder's typed u64/u32 header loads
    // are aligned (unaligned global loads FAULT on CDNA; found the hard
    // way at the first multi-block plane)
    run += align8(hsizes[i]);
  }
  PlaneHdr ph{run, static_cast<uint32_t>(nblocks), 0};
  UCCL_CHECK_HIP(hipMemcpyAsync(dst, &ph, sizeof(ph),
                                hipMemcpyHostToDevice, stream));
  UCCL_CHECK_HIP(hipMemcpyAsync(dst + sizeof(ph), hoffs.data(),
                                sizeof(uint64_t) * nblocks,
                                hipMemcpyHostToDevice, stream));
  UCCL_CHECK_HIP(hipMemcpyAsync(offs, hoffs.data(),
                                sizeof(uint64_t) * nblocks,
                                hipMemcpyHostToDevice, stream));
  uint8_t* payload = dst + sizeof(ph) + sizeof(uint64_t) * nblocks;
  k_compact<<<nblocks, 256, 0, stream>>>(scratch, sizes, offs, payload,
                                         nblocks);
  UCCL_CHECK_HIP(hipStreamSynchronize(stream));
  return sizeof(ph) + sizeof(uint64_t) * nblocks + run;
}

size_t decode_plane(uint8_t const* src, uint8_t* dst, size_t plane_bytes,
                    hipStream_t stream) {
  PlaneHdr ph{};
  UCCL_CHECK_HIP(hipMemcpy(&ph, src, sizeof(ph), hipMemcpyDeviceToHost));
  auto const* offs =
      reinterpret_cast<uint64_t const*>(src + sizeof(ph));
  uint8_t const* payload =
      src + sizeof(ph) + sizeof(uint64_t) * ph.nblocks;
  k_rans_decode<<<ph.nblocks, 256, 0, stream>>>(payload, offs, dst,
                                                plane_bytes, ph.nblocks);
  UCCL_CHECK_HIP(hipStreamSynchronize(stream));
  return sizeof(ph) + sizeof(uint64_t) * ph.nblocks + ph.comp_bytes;
}

}  // namespace

// Host-compiled mirror test of the rANS coder (the same
// __host__ __device__ functions the kernels use): encode a 64-lane
// interleaved block on the CPU, decode it back, compare. Lets the CPU
// test tier validate the coder math without a GPU.
bool host_rans_selftest(std::vector<uint8_t> const& data) {
  uint32_t const n = static_cast<uint32_t>(data.size());
  if (!n) return true;
  uint32_t hist[256] = {};
  for (uint8_t b : data) ++hist[b];
  uint16_t freq[256];
  normalize_freqs(hist, n, freq);
  uint16_t cum[257];
  cum[0] = 0;
  for (int i = 0; i < 256; ++i) cum[i + 1] = cum[i] + freq[i];
  if (cum[256] != kProbScale) return false;
  std::vector<uint8_t> cum2sym(kProbScale);
  for (int sym = 0; sym < 256; ++sym)
    for (uint32_t i = cum[sym]; i < cum[sym + 1]; ++i)
      cum2sym[i] = static_cast<uint8_t>(sym);

  uint32_t const per_lane_cap = 2 * ((n + kLanes - 1) / kLanes) + 16;
  std::vector<std::vector<uint8_t>> lanes(kLanes);
  std::vector<int> pos(kLanes);
  for (int l = 0; l < kLanes; ++l) {
    lanes[l].assign(per_lane_cap, 0);
    int p = static_cast<int>(per_lane_cap);
    uint32_t x = kRansL;
    int32_t const nsym = (static_cast<int32_t>(n) - 1 - l) / kLanes + 1;
    for (int32_t k = nsym - 1; k >= 0; --k) {
      if (l >= static_cast<int>(n)) break;
      uint8_t const sym = data[k * kLanes + l];
      rans_enc_put(&x, lanes[l].data(), &p, EncSym{freq[sym], cum[sym]});
    }
    rans_enc_flush(x, lanes[l].data(), &p);
    pos[l] = p;
  }
  std::vector<uint8_t> out(n, 0);
  for (int l = 0; l < kLanes; ++l) {
    int p = pos[l];
    uint32_t x = rans_dec_init(lanes[l].data(), &p);
    int32_t const nsym = (static_cast<int32_t>(n) - 1 - l) / kLanes + 1;
    for (int32_t k = 0; k < nsym; ++k) {
      if (l >= static_cast<int>(n)) break;
      out[k * kLanes + l] =
          rans_dec_get(&x, lanes[l].data(), &p, freq, cum,
                       cum2sym.data());
    }
  }
  return out == data;
}

size_t compress_bound(size_t bytes, int nplanes) {
  size_t const per_plane_blocks =
      (bytes / (nplanes ? nplanes : 1) + kBlockBytes - 1) / kBlockBytes + 1;
  return sizeof(FileHdr) +
         static_cast<size_t>(nplanes) *
             (sizeof(PlaneHdr) + 8 * per_plane_blocks +
              per_plane_blocks * kBlockWorst) +
         4096;
}

size_t compress(void const* src, size_t bytes, int elem_size, int nplanes,
                void* dst, size_t dst_cap, hipStream_t stream) {
  UCCL_CHECK(elem_size == 1 || elem_size == 2 || elem_size == 4);
  UCCL_CHECK(nplanes == 1 || nplanes == elem_size);
  UCCL_CHECK(bytes % elem_size == 0) << "partial element";
  UCCL_CHECK(dst_cap >= compress_bound(bytes, nplanes)) << "dst too small";
  auto* out = static_cast<uint8_t*>(dst);
  FileHdr fh{kMagic, static_cast<uint32_t>(nplanes),
             static_cast<uint32_t>(elem_size), kBlockBytes, bytes};
  UCCL_CHECK_HIP(hipMemcpyAsync(out, &fh, sizeof(fh),
                                hipMemcpyHostToDevice, stream));
  size_t w = sizeof(fh);
  if (nplanes == 1) {
    w += encode_plane(static_cast<uint8_t const*>(src), bytes, out + w,
                      stream);
    return w;
  }
  size_t const elems = bytes / elem_size;
  DevBuf planes_b;
  auto* planes = static_cast<uint8_t*>(planes_b.alloc(bytes));
  int const grid =
      static_cast<int>(std::min<size_t>((elems + 255) / 256, 4096));
  k_plane_split<<<grid, 256, 0, stream>>>(
      static_cast<uint8_t const*>(src), planes, elems, elem_size);
  for (int pl = 0; pl < nplanes; ++pl)
    w += encode_plane(planes + static_cast<size_t>(pl) * elems, elems,
                      out + w, stream);
  return w;
}

size_t decompress(void const* src, size_t src_bytes, void* dst,
                  size_t orig_cap, hipStream_t stream) {
  FileHdr fh{};
  UCCL_CHECK_HIP(hipMemcpy(&fh, src, sizeof(fh), hipMemcpyDeviceToHost));
  UCCL_CHECK(fh.magic == kMagic) << "bad gpu codec frame";
  UCCL_CHECK(fh.orig_bytes <= orig_cap) << "output too small";
  UCCL_CHECK(fh.nplanes == 1 || fh.nplanes == fh.elem_size)
      << "bad plane count";
  auto const* in = static_cast<uint8_t const*>(src);
  size_t r = sizeof(fh);
  if (fh.nplanes == 1) {
    r += decode_plane(in + r, static_cast<uint8_t*>(dst), fh.orig_bytes,
                      stream);
    (void)src_bytes;
    return fh.orig_bytes;
  }
  size_t const elems = fh.orig_bytes / fh.elem_size;
  DevBuf planes_b;
  auto* planes = static_cast<uint8_t*>(planes_b.alloc(fh.orig_bytes));
  for (uint32_t pl = 0; pl < fh.nplanes; ++pl)
    r += decode_plane(in + r, planes + static_cast<size_t>(pl) * elems,
                      elems, stream);
  int const grid =
      static_cast<int>(std::min<size_t>((elems + 255) / 256, 4096));
  k_plane_merge<<<grid, 256, 0, stream>>>(planes, static_cast<uint8_t*>(dst),
                                          elems, fh.elem_size);
  UCCL_CHECK_HIP(hipStreamSynchronize(stream));
  return fh.orig_bytes;
}

}  // namespace gpu
}  // namespace p2p
}  // namespace uccl
