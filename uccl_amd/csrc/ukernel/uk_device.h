// DeviceBackend for the spray executor: persistent HIP worker kernels
// draining per-rank C2D (CPU->GPU) task FIFOs.
//
// Parity role: the reference ukernel's persistent device workers
// (experimental/ukernel/src/device/persistent_kernel_ops.cu:233
// singlePersistentKernel / :324 multiPersistentKernel polling C2D FIFOs
// for copy/reduce Tasks). MI355X re-design: one worker BLOCK per rank
// lane (all lanes co-resident by construction: grid = world <= 8 on 256
// CUs), tasks in pinned host memory with system-scope release/acquire
// handoff, flags in pinned memory so the host-side executor's poll()
// reads them directly without a sync.
//
// Single-process model: each "rank"'s Input/Output/Scratch spaces are
// device buffers on one GPU (exactly how the HostBackend models ranks in
// host memory); kPut between ranks is a device-to-device copy executed
// by the source rank's lane. On real multi-GPU the same task stream
// binds to IPC peer pointers — the lane/FIFO machinery is unchanged.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>
#include <mutex>
#include <vector>

#include "ukernel.h"

namespace uccl {
namespace uk {

struct DevTask {
  uint32_t op;       // Op cast; 100 = stop
  uint32_t pad;
  uint64_t src;      // device pointer
  uint64_t dst;      // device pointer
  uint64_t bytes;
  uint64_t flag;     // kSignal: flag index
};

constexpr uint32_t kUkStop = 100;
constexpr uint32_t kUkLaneSlots = 256;

// One C2D lane (pinned host memory, device-visible).
struct UkLane {
  volatile uint64_t tail;   // host producer
  uint64_t pad0[7];
  volatile uint64_t done;   // device consumer: tasks fully completed
  uint64_t pad1[7];
  DevTask tasks[kUkLaneSlots];
};

class DeviceBackend : public Backend {
 public:
  // world lanes; per-rank device spaces sized like HostBackend's.
  DeviceBackend(int world, uint64_t in_bytes, uint64_t out_bytes,
                uint64_t scratch_bytes, int device = 0);
  ~DeviceBackend() override;

  // host<->device staging for test setup/verification
  void upload_input(int rank, float const* src, uint64_t bytes);
  void download_output(int rank, float* dst, uint64_t bytes);

  void copy(Task const& t) override;
  void reduce(Task const& t) override;
  void put(Task const& t) override;
  void signal(uint64_t flag) override;
  bool poll(uint64_t flag) override;

 private:
  void push(int lane, DevTask const& t);
  void* resolve(BufRef const& b);

  int world_;
  int device_;
  uint64_t in_bytes_, out_bytes_, scratch_bytes_;
  std::vector<void*> in_, out_, scratch_;  // device buffers per rank
  UkLane* lanes_host_ = nullptr;           // pinned array [world]
  UkLane* lanes_dev_ = nullptr;            // device view of the same
  volatile uint32_t* flags_host_ = nullptr;  // pinned flag words
  uint32_t* flags_dev_ = nullptr;
  std::vector<uint64_t> pushed_;           // per-lane produced count
  std::vector<std::unique_ptr<std::mutex>> lane_mu_;
  hipStream_t stream_ = nullptr;           // persistent kernel stream
};

}  // namespace uk
}  // namespace uccl
