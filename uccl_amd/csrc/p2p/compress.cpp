#include "compress.h"

#include <zlib.h>

#include <cstring>
#include <stdexcept>
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
std::string deflate_plane(uint8_t const* p, size_t n) {
  uLongf cap = compressBound(n);
  std::string out;
  out.resize(cap);
  // level 1: this codec sits on the transfer hot path; zlib level 1 on a
  // single byte plane is ~300 MB/s and captures most of the exponent
  // redundancy
  if (compress2((Bytef*)out.data(), &cap, (Bytef const*)p, n, 1) != Z_OK)
    return {};
  if (cap >= n) return {};
  out.resize(cap);
  return out;
}

bool inflate_plane(uint8_t const* p, size_t n, uint8_t* out, size_t out_n) {
  uLongf got = out_n;
  return uncompress((Bytef*)out, &got, (Bytef const*)p, n) == Z_OK &&
         got == out_n;
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
          plane_compressible(elem_size, pl)) {
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
  size_t const bytes = get_u64(p + 8);
  if (bytes > out_cap) throw std::runtime_error("compress: output too small");
  size_t off = 24;
  int const nplanes = p[off++];
  if (nplanes < 1 || nplanes > 8 || frame_bytes < off + nplanes * 9)
    throw std::runtime_error("compress: bad plane table");
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
    if (frame_bytes < off + meta[pl].stored)
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
