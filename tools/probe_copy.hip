// Standalone probe: copy-kernel variants for the staged collective paths.
// Finds the best grid/width/store flavor for 256MB-class streaming copies
// on MI355X (results feed kernels.hip's launch configuration).
//   hipcc --offload-arch=gfx950 -O3 tools/probe_copy.hip -o gpurun_out/probe_copy

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdio>
#include <vector>

#define CHECK(x)                                          \
  do {                                                    \
    hipError_t e = (x);                                   \
    if (e != hipSuccess) {                                \
      printf("HIP error %s @%d\n", hipGetErrorString(e), __LINE__); \
      return 1;                                           \
    }                                                     \
  } while (0)

using V16 = uint4;
struct V32 {
  uint4 a, b;
};

__global__ void k_copy_v16(void* __restrict__ dst, void const* __restrict__ src,
                           size_t bytes) {
  size_t const n = bytes / 16;
  auto* d = reinterpret_cast<V16*>(dst);
  auto const* s = reinterpret_cast<V16 const*>(src);
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  for (; i < n; i += stride) d[i] = s[i];
}

__global__ void k_copy_v32(void* __restrict__ dst, void const* __restrict__ src,
                           size_t bytes) {
  size_t const n = bytes / 32;
  auto* d = reinterpret_cast<V32*>(dst);
  auto const* s = reinterpret_cast<V32 const*>(src);
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  for (; i < n; i += stride) d[i] = s[i];
}

__global__ void k_copy_nt(void* __restrict__ dst, void const* __restrict__ src,
                          size_t bytes) {
  size_t const n = bytes / 16;
  auto* d = reinterpret_cast<V16*>(dst);
  auto const* s = reinterpret_cast<V16 const*>(src);
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  using VU = unsigned __attribute__((ext_vector_type(4)));
  for (; i < n; i += stride) {
    VU v = __builtin_nontemporal_load(reinterpret_cast<VU const*>(s) + i);
    __builtin_nontemporal_store(v, reinterpret_cast<VU*>(d) + i);
  }
}

// reduce shape: out[i] = a[i] (+ unpack/accumulate cost model: bf16->fp32->bf16)
__global__ void k_reduce1(void* __restrict__ dst, void const* __restrict__ src,
                          size_t bytes) {
  size_t const n = bytes / 16;
  auto* d = reinterpret_cast<V16*>(dst);
  auto const* s = reinterpret_cast<V16 const*>(src);
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t const stride = gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    V16 v = s[i];
    float acc[8];
    auto const* h = reinterpret_cast<__hip_bfloat16 const*>(&v);
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] = __bfloat162float(h[j]);
    V16 r;
    auto* o = reinterpret_cast<__hip_bfloat16*>(&r);
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = __float2bfloat16(acc[j]);
    d[i] = r;
  }
}

template <typename K>
double bench(K kernel, void* d, void* s, size_t bytes, int grid, int block,
             int iters) {
  for (int i = 0; i < 3; ++i)
    kernel<<<grid, block>>>(d, s, bytes);
  (void)hipDeviceSynchronize();
  hipEvent_t a, b;
  (void)hipEventCreate(&a);
  (void)hipEventCreate(&b);
  (void)hipEventRecord(a);
  for (int i = 0; i < iters; ++i)
    kernel<<<grid, block>>>(d, s, bytes);
  (void)hipEventRecord(b);
  (void)hipEventSynchronize(b);
  float ms = 0;
  (void)hipEventElapsedTime(&ms, a, b);
  (void)hipEventDestroy(a);
  (void)hipEventDestroy(b);
  return 2.0 * bytes * iters / (ms / 1e3) / 1e12;  // TB/s (r+w)
}

int main() {
  size_t const bytes = 256ull << 20;
  void *d, *s;
  CHECK(hipMalloc(&d, bytes));
  CHECK(hipMalloc(&s, bytes));
  CHECK(hipMemset(s, 7, bytes));
  struct Cfg {
    const char* name;
    int grid, block;
  };
  std::vector<Cfg> cfgs = {{"g2048b256", 2048, 256}, {"g4096b256", 4096, 256},
                           {"g8192b256", 8192, 256}, {"g2048b512", 2048, 512},
                           {"g1024b1024", 1024, 1024}};
  for (auto const& c : cfgs) {
    printf("v16 %-11s %.2f TB/s\n", c.name,
           bench(k_copy_v16, d, s, bytes, c.grid, c.block, 10));
    printf("v32 %-11s %.2f TB/s\n", c.name,
           bench(k_copy_v32, d, s, bytes, c.grid, c.block, 10));
    printf("nt  %-11s %.2f TB/s\n", c.name,
           bench(k_copy_nt, d, s, bytes, c.grid, c.block, 10));
  }
  printf("reduce1 g4096b256 %.2f TB/s\n",
         bench(k_reduce1, d, s, bytes, 4096, 256, 10));
  (void)hipFree(d);
  (void)hipFree(s);
  return 0;
}
