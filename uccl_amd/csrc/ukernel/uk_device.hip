// Persistent device workers for the spray executor (uk_device.h).
//
// k_uk_worker: grid = world blocks, one per rank lane. Each block spins
// on its lane's pinned tail, executes tasks cooperatively with all 256
// threads (fp32 copy/reduce/put), raises signal flags with system-scope
// release stores (pinned memory: host poll() sees them without a sync),
// and publishes the per-lane done counter the host producers block on.
// Exit: a kUkStop task. Reference analog:
// experimental/ukernel/src/device/persistent_kernel_ops.cu:233.

#include <hip/hip_runtime.h>

#include <cstring>

#include "../core/log.h"
#include "../device/primitives.h"
#include "uk_device.h"

namespace uccl {
namespace uk {

using namespace uccl::device;

namespace {

__device__ void dev_copy(float* dst, float const* src, uint64_t bytes) {
  uint64_t const n = bytes / 4;
  for (uint64_t i = threadIdx.x; i < n; i += blockDim.x) dst[i] = src[i];
}

__device__ void dev_reduce(float* dst, float const* src, uint64_t bytes) {
  uint64_t const n = bytes / 4;
  for (uint64_t i = threadIdx.x; i < n; i += blockDim.x) dst[i] += src[i];
}

__global__ void k_uk_worker(UkLane* lanes, uint32_t* flags) {
  UkLane* lane = &lanes[blockIdx.x];
  __shared__ DevTask task;
  uint64_t next = 0;
  for (;;) {
    // wait for a produced task (bounded; trap instead of wedging)
    if (threadIdx.x == 0) {
      for (uint64_t it = 0;; ++it) {
        uint64_t t = __hip_atomic_load(const_cast<uint64_t*>(&lane->tail),
                                       __ATOMIC_ACQUIRE,
                                       __HIP_MEMORY_SCOPE_SYSTEM);
        if (t > next) break;
        if (it > (1ull << 30)) {
          printf("uccl_uk: worker lane %d idle timeout\n",
                 static_cast<int>(blockIdx.x));
          __builtin_trap();
        }
        backoff();
      }
      // copy the task into LDS for the whole block
      DevTask const* slot =
          const_cast<DevTask const*>(&lane->tasks[next % kUkLaneSlots]);
      task = *slot;
    }
    __syncthreads();
    DevTask const t = task;
    __syncthreads();
    if (t.op == kUkStop) return;
    switch (static_cast<Op>(t.op)) {
      case Op::kCopy:
        dev_copy(reinterpret_cast<float*>(t.dst),
                 reinterpret_cast<float const*>(t.src), t.bytes);
        break;
      case Op::kReduce:
        dev_reduce(reinterpret_cast<float*>(t.dst),
                   reinterpret_cast<float const*>(t.src), t.bytes);
        break;
      case Op::kPut:
        // cross-rank put: plain copy (peer HBM pointer on multi-GPU)
        dev_copy(reinterpret_cast<float*>(t.dst),
                 reinterpret_cast<float const*>(t.src), t.bytes);
        break;
      case Op::kSignal:
        break;  // the flag store below is the whole task
      default:
        break;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      if (static_cast<Op>(t.op) == Op::kSignal) {
        // release: all prior lane work is visible before the flag rises
        __hip_atomic_store(&flags[t.flag], 1u, __ATOMIC_RELEASE,
                           __HIP_MEMORY_SCOPE_SYSTEM);
      }
      ++next;
      __hip_atomic_store(const_cast<uint64_t*>(&lane->done), next,
                         __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_SYSTEM);
    }
    __syncthreads();
  }
}

}  // namespace

DeviceBackend::DeviceBackend(int world, uint64_t in_bytes,
                             uint64_t out_bytes, uint64_t scratch_bytes,
                             int device)
    : world_(world),
      device_(device),
      in_bytes_(in_bytes),
      out_bytes_(out_bytes),
      scratch_bytes_(scratch_bytes) {
  UCCL_CHECK(world >= 1 && world <= 16) << "device backend world 1..16";
  UCCL_CHECK_HIP(hipSetDevice(device_));
  for (int r = 0; r < world_; ++r) {
    void *i = nullptr, *o = nullptr, *s = nullptr;
    UCCL_CHECK_HIP(hipMalloc(&i, in_bytes ? in_bytes : 4));
    UCCL_CHECK_HIP(hipMalloc(&o, out_bytes ? out_bytes : 4));
    UCCL_CHECK_HIP(hipMalloc(&s, scratch_bytes ? scratch_bytes : 4));
    UCCL_CHECK_HIP(hipMemset(o, 0, out_bytes ? out_bytes : 4));
    UCCL_CHECK_HIP(hipMemset(s, 0, scratch_bytes ? scratch_bytes : 4));
    in_.push_back(i);
    out_.push_back(o);
    scratch_.push_back(s);
  }
  UCCL_CHECK_HIP(hipHostMalloc(reinterpret_cast<void**>(&lanes_host_),
                               sizeof(UkLane) * world_,
                               hipHostMallocMapped));
  memset(lanes_host_, 0, sizeof(UkLane) * world_);
  UCCL_CHECK_HIP(hipHostGetDevicePointer(
      reinterpret_cast<void**>(&lanes_dev_), lanes_host_, 0));
  UCCL_CHECK_HIP(hipHostMalloc(
      reinterpret_cast<void**>(const_cast<uint32_t**>(&flags_host_)),
      sizeof(uint32_t) * 65536, hipHostMallocMapped));
  memset(const_cast<uint32_t*>(flags_host_), 0, sizeof(uint32_t) * 65536);
  UCCL_CHECK_HIP(hipHostGetDevicePointer(
      reinterpret_cast<void**>(&flags_dev_),
      const_cast<uint32_t*>(flags_host_), 0));
  pushed_.assign(world_, 0);
  for (int r = 0; r < world_; ++r)
    lane_mu_.emplace_back(new std::mutex());
  UCCL_CHECK_HIP(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking));
  hipLaunchKernelGGL(k_uk_worker, dim3(world_), dim3(256), 0, stream_,
                     lanes_dev_, flags_dev_);
}

DeviceBackend::~DeviceBackend() {
  DevTask stop{};
  stop.op = kUkStop;
  for (int r = 0; r < world_; ++r) push(r, stop);
  (void)hipStreamSynchronize(stream_);
  (void)hipStreamDestroy(stream_);
  for (void* p : in_) (void)hipFree(p);
  for (void* p : out_) (void)hipFree(p);
  for (void* p : scratch_) (void)hipFree(p);
  if (lanes_host_) (void)hipHostFree(lanes_host_);
  if (flags_host_) (void)hipHostFree(const_cast<uint32_t*>(flags_host_));
}

void DeviceBackend::push(int lane, DevTask const& t) {
  std::lock_guard<std::mutex> g(*lane_mu_[lane]);
  UkLane* l = &lanes_host_[lane];
  uint64_t const idx = pushed_[lane];
  // bounded ring: wait for the worker to drain before overwriting
  while (idx - __atomic_load_n(const_cast<uint64_t*>(&l->done),
                               __ATOMIC_ACQUIRE) >=
         kUkLaneSlots) {
    sched_yield();
  }
  l->tasks[idx % kUkLaneSlots] = t;
  __atomic_store_n(const_cast<uint64_t*>(&l->tail), idx + 1,
                   __ATOMIC_RELEASE);
  pushed_[lane] = idx + 1;
  // blocking completion per task keeps the host-side executor's
  // dependency accounting exact (its worker threads run per rank, so
  // cross-lane parallelism is preserved)
  if (t.op != kUkStop) {
    while (__atomic_load_n(const_cast<uint64_t*>(&l->done),
                           __ATOMIC_ACQUIRE) < idx + 1) {
      sched_yield();
    }
  }
}

void* DeviceBackend::resolve(BufRef const& b) {
  void* base = nullptr;
  uint64_t cap = 0;
  switch (b.space) {
    case Space::kInput: base = in_[b.rank]; cap = in_bytes_; break;
    case Space::kOutput: base = out_[b.rank]; cap = out_bytes_; break;
    case Space::kScratch: base = scratch_[b.rank]; cap = scratch_bytes_; break;
  }
  UCCL_CHECK(b.offset <= cap) << "ukernel device ref out of range";
  return static_cast<char*>(base) + b.offset;
}

void DeviceBackend::copy(Task const& t) {
  DevTask d{};
  d.op = static_cast<uint32_t>(Op::kCopy);
  d.src = reinterpret_cast<uint64_t>(resolve(t.src));
  d.dst = reinterpret_cast<uint64_t>(resolve(t.dst));
  d.bytes = t.bytes;
  push(t.rank, d);
}

void DeviceBackend::reduce(Task const& t) {
  DevTask d{};
  d.op = static_cast<uint32_t>(Op::kReduce);
  d.src = reinterpret_cast<uint64_t>(resolve(t.src));
  d.dst = reinterpret_cast<uint64_t>(resolve(t.dst));
  d.bytes = t.bytes;
  push(t.rank, d);
}

void DeviceBackend::put(Task const& t) {
  DevTask d{};
  d.op = static_cast<uint32_t>(Op::kPut);
  d.src = reinterpret_cast<uint64_t>(resolve(t.src));
  d.dst = reinterpret_cast<uint64_t>(resolve(t.dst));
  d.bytes = t.bytes;
  push(t.rank, d);
}

void DeviceBackend::signal(uint64_t flag) {
  // signal rides the lane of the signalling task's rank; the executor
  // calls signal() from that rank's worker thread. Lane choice only
  // affects ordering, which the per-lane FIFO preserves; route by flag
  // to keep the call context-free.
  DevTask d{};
  d.op = static_cast<uint32_t>(Op::kSignal);
  d.flag = flag % 65536;
  push(static_cast<int>(flag) % world_, d);
}

bool DeviceBackend::poll(uint64_t flag) {
  return __atomic_load_n(const_cast<uint32_t*>(&flags_host_[flag % 65536]),
                         __ATOMIC_ACQUIRE) != 0;
}

void DeviceBackend::upload_input(int rank, float const* src,
                                 uint64_t bytes) {
  UCCL_CHECK_HIP(
      hipMemcpy(in_[rank], src, bytes, hipMemcpyHostToDevice));
}

void DeviceBackend::download_output(int rank, float* dst, uint64_t bytes) {
  UCCL_CHECK_HIP(
      hipMemcpy(dst, out_[rank], bytes, hipMemcpyDeviceToHost));
}

}  // namespace uk
}  // namespace uccl
