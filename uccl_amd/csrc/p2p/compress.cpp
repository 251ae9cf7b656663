#include "compress.h"

#include <zlib.h>

#include <cstring>
#include <future>
#include <stdexcept>
#include <thread>
#include <vector>

namespace uccl {
namespace p2p {
namespace comp {

namespace {
constexpr uint32_t kMagic = 0x555A4346u;  // 'UCZF'
constexpr uint8_t kVersion = 1;

struct PlaneMeta {
  uint8_t encoding;  // 0 raw, 1 deflate
  uint64_t stored;
};

void put_u32(std::string& s, uint32_t v) { s.append((char const*)&v, 4); }
void put_u64(std::string& s, uint64_t v) { s.append((char const*)&v, 8); }

uint32_t get_u32(uint8_t const* p) {
  uint32_t v;
  std::memcpy(&v, p, 4);
  return v;
}
uint64_t get_u64(uint8_t const* p) {
  uint64_t v;
  std::memcpy(&v, p, 8);
  return v;
}

// deflate `n` bytes; empty result means "did not shrink, store raw"
std::string deflate_block(uint8_t const* p, size_t n) {
  uLongf cap = compressBound(n);
  std::string out;
  out.resize(cap);
  // level 1: this codec sits on the transfer hot path; level 1 captures
  // most of the exponent redundancy at several hundred MB/s per core
  if (compress2((Bytef*)out.data(), &cap, (Bytef const*)p, n, 1) != Z_OK)
    return {};
  out.resize(cap);
  return out;
}

constexpr size_t kBlock = 4u << 20;  // per-thread deflate granule


// a 64KB sample that fails to shrink by >3% predicts an incompressible
// plane — skip the full (expensive) deflate of megabytes of noise
bool probe_compressible(uint8_t const* p, size_t n) {
  size_t const sample = n < (64u << 10) ? n : (64u << 10);
  std::string enc = deflate_block(p + (n - sample) / 2, sample);
  return !enc.empty() && enc.size() < sample * 97 / 100;
}

// block-parallel deflate of a whole plane; empty => did not shrink.
// encoding 2 layout: u32 nblocks | nblocks x u64 sizes | blobs
std::string deflate_plane(uint8_t const* p, size_t n) {
  size_t const nblocks = (n + kBlock - 1) / kBlock;
  std::vector<std::future<std::string>> futs;
  futs.reserve(nblocks);
  for (size_t b = 0; b < nblocks; ++b) {
    size_t const off = b * kBlock;
    size_t const len = std::min(kBlock, n - off);
    futs.push_back(std::async(
        nblocks > 1 ? std::launch::async : std::launch::deferred,
        [p, off, len] { return deflate_block(p + off, len); }));
  }
  std::vector<std::string> blobs(nblocks);
  size_t total = 4 + nblocks * 8;
  for (size_t b = 0; b < nblocks; ++b) {
    blobs[b] = futs[b].get();
    if (blobs[b].empty()) return {};
    total += blobs[b].size();
  }
  if (total >= n) return {};
  std::string out;
  out.reserve(total);
  uint32_t const nb32 = static_cast<uint32_t>(nblocks);
  out.append((char const*)&nb32, 4);
  for (auto const& b : blobs) {
    uint64_t const sz = b.size();
    out.append((char const*)&sz, 8);
  }
  for (auto const& b : blobs) out.append(b);
  return out;
}

bool inflate_block(uint8_t const* p, size_t n, uint8_t* out, size_t out_n) {
  uLongf got = out_n;
  return uncompress((Bytef*)out, &got, (Bytef const*)p, n) == Z_OK &&
         got == out_n;
}

bool inflate_plane(uint8_t const* p, size_t n, uint8_t* out, size_t out_n) {
  if (n < 4) return false;
  uint32_t nblocks;
  std::memcpy(&nblocks, p, 4);
  if (nblocks == 0 || n < 4 + size_t(nblocks) * 8) return false;
  std::vector<uint64_t> sizes(nblocks);
  std::memcpy(sizes.data(), p + 4, size_t(nblocks) * 8);
  size_t off = 4 + size_t(nblocks) * 8;
  size_t out_off = 0;
  std::vector<std::future<bool>> futs;
  for (uint32_t b = 0; b < nblocks; ++b) {
    // subtraction form: sizes[] comes off the wire; off + sizes[b] could
    // wrap a u64 and bypass the bound
    if (sizes[b] > n - off) return false;
    size_t const blen = b + 1 < nblocks ? kBlock : out_n - out_off;
    if (out_off + blen > out_n) return false;
    uint8_t const* src = p + off;
    uint8_t* dst = out + out_off;
    uint64_t const sz = sizes[b];
    futs.push_back(std::async(
        nblocks > 1 ? std::launch::async : std::launch::deferred,
        [src, sz, dst, blen] { return inflate_block(src, sz, dst, blen); }));
    off += sizes[b];
    out_off += blen;
  }
  bool ok = out_off == out_n;
  for (auto& f : futs) ok = f.get() && ok;
  return ok;
}

// which planes carry sign/exponent bits (worth an entropy stage)
bool plane_compressible(int elem_size, int plane) {
  if (elem_size == 2) return plane == 1;              // bf16/f16 high byte
  if (elem_size == 4) return plane >= 2;              // f32 top two bytes
  return false;
}
}  // namespace

std::string compress(void const* data, size_t bytes, int elem_size,
                     int dtype_code, int strategy) {
  if (elem_size != 2 && elem_size != 4) {
    elem_size = 1;
    strategy = kNone;
  }
  int const nplanes = strategy == kNone ? 1 : elem_size;
  size_t const elems = strategy == kNone ? bytes : bytes / elem_size;
  size_t const tail = strategy == kNone ? 0 : bytes - elems * elem_size;
  uint8_t const* src = static_cast<uint8_t const*>(data);

  std::vector<std::vector<uint8_t>> planes(nplanes);
  std::vector<PlaneMeta> meta(nplanes);
  std::vector<std::string> encoded(nplanes);

  if (strategy == kNone) {
    meta[0] = {0, bytes};
  } else {
    for (int pl = 0; pl < nplanes; ++pl) {
      planes[pl].resize(elems);
      uint8_t* d = planes[pl].data();
      for (size_t i = 0; i < elems; ++i) d[i] = src[i * elem_size + pl];
      meta[pl] = {0, elems};
      if (strategy == kSplitDeflate && elems >= 512 &&
          plane_compressible(elem_size, pl) &&
          probe_compressible(d, elems)) {
        encoded[pl] = deflate_plane(d, elems);
        if (!encoded[pl].empty()) meta[pl] = {1, encoded[pl].size()};
      }
    }
  }

  std::string out;
  size_t total = 24 + 1 + nplanes * 9 + tail;
  for (auto const& m : meta) total += m.stored;
  out.reserve(total);
  put_u32(out, kMagic);
  out.push_back((char)kVersion);
  out.push_back((char)strategy);
  out.push_back((char)elem_size);
  out.push_back((char)dtype_code);
  put_u64(out, bytes);
  put_u64(out, 0);  // reserved
  out.push_back((char)nplanes);
  for (auto const& m : meta) {
    out.push_back((char)m.encoding);
    put_u64(out, m.stored);
  }
  if (strategy == kNone) {
    out.append((char const*)src, bytes);
  } else {
    for (int pl = 0; pl < nplanes; ++pl) {
      if (meta[pl].encoding)
        out.append(encoded[pl]);
      else
        out.append((char const*)planes[pl].data(), planes[pl].size());
    }
    if (tail) out.append((char const*)src + elems * elem_size, tail);
  }
  return out;
}

size_t orig_bytes(void const* hdr, size_t avail) {
  auto const* p = static_cast<uint8_t const*>(hdr);
  if (avail < 16 || get_u32(p) != kMagic)
    throw std::runtime_error("compress: bad frame header");
  return get_u64(p + 8);
}

int dtype_code(void const* hdr, size_t avail) {
  auto const* p = static_cast<uint8_t const*>(hdr);
  if (avail < 8 || get_u32(p) != kMagic)
    throw std::runtime_error("compress: bad frame header");
  return p[7];
}

size_t decompress(void const* frame, size_t frame_bytes, void* out,
                  size_t out_cap) {
  auto const* p = static_cast<uint8_t const*>(frame);
  if (frame_bytes < 25 || get_u32(p) != kMagic || p[4] != kVersion)
    throw std::runtime_error("compress: bad frame");
  int const strategy = p[5];
  int const elem_size = p[6];
  // elem_size and nplanes are wire-controlled and feed a division and the
  // interleave store stride below: reject anything but the shapes the
  // encoder emits before they can divide by zero or write past out_cap.
  if (elem_size != 1 && elem_size != 2 && elem_size != 4)
    throw std::runtime_error("compress: bad elem size");
  size_t const bytes = get_u64(p + 8);
  if (bytes > out_cap) throw std::runtime_error("compress: output too small");
  size_t off = 24;
  int const nplanes = p[off++];
  if (nplanes < 1 || nplanes > 8 || frame_bytes < off + nplanes * 9)
    throw std::runtime_error("compress: bad plane table");
  if (strategy == kNone ? nplanes != 1 : nplanes != elem_size)
    throw std::runtime_error("compress: plane count mismatch");
  std::vector<PlaneMeta> meta(nplanes);
  for (int pl = 0; pl < nplanes; ++pl) {
    meta[pl].encoding = p[off];
    meta[pl].stored = get_u64(p + off + 1);
    off += 9;
  }
  uint8_t* dst = static_cast<uint8_t*>(out);

  if (strategy == kNone) {
    if (meta[0].stored != bytes || frame_bytes < off + bytes)
      throw std::runtime_error("compress: truncated raw frame");
    std::memcpy(dst, p + off, bytes);
    return bytes;
  }
  size_t const elems = bytes / elem_size;
  size_t const tail = bytes - elems * elem_size;
  std::vector<uint8_t> plane(elems);
  for (int pl = 0; pl < nplanes; ++pl) {
    if (meta[pl].stored > frame_bytes - off)  // subtraction: no u64 wrap
      throw std::runtime_error("compress: truncated plane");
    uint8_t const* stored = p + off;
    uint8_t const* plane_data;
    if (meta[pl].encoding == 1) {
      if (!inflate_plane(stored, meta[pl].stored, plane.data(), elems))
        throw std::runtime_error("compress: inflate failed");
      plane_data = plane.data();
    } else {
      if (meta[pl].stored != elems)
        throw std::runtime_error("compress: bad raw plane size");
      plane_data = stored;
    }
    for (size_t i = 0; i < elems; ++i)
      dst[i * elem_size + pl] = plane_data[i];
    off += meta[pl].stored;
  }
  if (tail) {
    if (frame_bytes < off + tail)
      throw std::runtime_error("compress: truncated tail");
    std::memcpy(dst + elems * elem_size, p + off, tail);
  }
  return bytes;
}

}  // namespace comp
}  // namespace p2p
}  // namespace uccl
