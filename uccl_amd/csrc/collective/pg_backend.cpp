// Native torch.distributed backend ("uccl") over the xGMI collective
// engine: c10d::Backend subclass so that dist.init_process_group("uccl")
// routes all_reduce / broadcast / all_gather / reduce_scatter /
// all_to_all / send / recv / barrier through uccl_amd's CDNA4 kernels
// unmodified — the drop-in role the reference fills with its RCCL plugin
// and lite-collective NCCL API.
//
// Work semantics match ProcessGroupNCCL's stream-ordered model: ops are
// enqueued on the current HIP stream; Work records a HIP event at enqueue;
// wait() blocks the (current) stream-consumer via event sync.

#include <torch/python.h>

#include <pybind11/chrono.h>

#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <hip/hip_runtime.h>
#include <torch/csrc/distributed/c10d/Backend.hpp>
#include <torch/csrc/distributed/c10d/Store.hpp>
#include <torch/csrc/distributed/c10d/Types.hpp>
#include <torch/csrc/distributed/c10d/Work.hpp>

#include "../core/env.h"
#include "../core/log.h"
#include "communicator.h"

namespace uccl {

namespace {

hipStream_t cur_stream(int device) {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA(device).stream();
}

Dtype to_dtype(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kFloat: return Dtype::kF32;
    case at::kHalf: return Dtype::kF16;
    case at::kBFloat16: return Dtype::kBF16;
    case at::kInt: return Dtype::kI32;
    case at::kLong: return Dtype::kI64;
    case at::kDouble: return Dtype::kF64;
    case at::kFloat8_e4m3fn: return Dtype::kF8E4M3;
    default:
      TORCH_CHECK(false, "uccl backend: unsupported dtype ",
                  t.scalar_type());
  }
}

// Work with ProcessGroupNCCL-style semantics: ops are enqueued on the
// caller's stream; wait()/synchronize() make the *current stream* wait on
// the op's completion event (no host block), so DDP keeps compute/comm
// overlap; getFuture() completes asynchronously off a watcher thread.
class UcclWork : public c10d::Work {
 public:
  UcclWork(int rank, c10d::OpType op, hipStream_t stream, int device,
           std::vector<at::Tensor> results)
      : c10d::Work(rank, op),
        device_(device),
        event_(new EventHolder),
        results_(std::move(results)) {
    (void)hipEventCreateWithFlags(&event_->ev, hipEventDisableTiming);
    (void)hipEventRecord(event_->ev, stream);
  }

  bool isCompleted() override {
    return hipEventQuery(event_->ev) == hipSuccess;
  }
  bool isSuccess() const override { return true; }
  std::vector<at::Tensor> result() override { return results_; }
  void synchronize() override {
    hipStream_t cur =
        at::hip::getCurrentHIPStreamMasqueradingAsCUDA(device_).stream();
    (void)hipStreamWaitEvent(cur, event_->ev, 0);
  }
  bool wait(std::chrono::milliseconds) override {
    synchronize();
    // read per call, not once per process (see STATUS.md lesson 8)
    if (uccl::env_bool("UCCL_BLOCKING_WAIT", false))
      (void)hipEventSynchronize(event_->ev);
    return true;
  }
  c10::intrusive_ptr<c10::ivalue::Future> getFuture() override {
    auto fut = c10::make_intrusive<c10::ivalue::Future>(
        c10::ListType::create(c10::TensorType::get()));
    auto holder = event_;  // keeps the event alive past Work destruction
    auto results = results_;
    std::thread([fut, holder, results]() {
      (void)hipEventSynchronize(holder->ev);
      fut->markCompleted(c10::IValue(results));
    }).detach();
    return fut;
  }

 private:
  struct EventHolder {
    hipEvent_t ev = nullptr;
    ~EventHolder() {
      if (ev) (void)hipEventDestroy(ev);
    }
  };
  int device_;
  std::shared_ptr<EventHolder> event_;
  std::vector<at::Tensor> results_;
};

class UcclBackend : public c10d::Backend {
 public:
  UcclBackend(c10::intrusive_ptr<c10d::Store> store, int rank, int size)
      : c10d::Backend(rank, size), store_(std::move(store)) {
    int device = 0;
    (void)hipGetDevice(&device);
    comm_ = std::make_unique<Communicator>(rank, size, device,
                                           /*heap_bytes=*/0);
    // bootstrap: exchange IPC handles through the c10d store
    if (size > 1) {
      std::string mine = comm_->handle_bytes();
      store_->set("uccl_h_" + std::to_string(rank),
                  std::vector<uint8_t>(mine.begin(), mine.end()));
      std::vector<std::string> handles(size);
      for (int r = 0; r < size; ++r) {
        auto v = store_->get("uccl_h_" + std::to_string(r));
        handles[r] = std::string(v.begin(), v.end());
      }
      comm_->connect(handles);
    }
  }


  static RedOp to_redop(const c10d::ReduceOp& op) {
    if (op == c10d::ReduceOp::SUM || op == c10d::ReduceOp::AVG)
      return RedOp::kSum;  // AVG = SUM + post-scale at the call site
    if (op == c10d::ReduceOp::PRODUCT) return RedOp::kProd;
    if (op == c10d::ReduceOp::MIN) return RedOp::kMin;
    if (op == c10d::ReduceOp::MAX) return RedOp::kMax;
    TORCH_CHECK(false, "uccl: unsupported reduce op");
  }

  const std::string getBackendName() const override { return "uccl"; }

  c10::intrusive_ptr<c10d::Work> allreduce(
      std::vector<at::Tensor>& tensors,
      const c10d::AllreduceOptions& opts) override {
    TORCH_CHECK(tensors.size() == 1, "uccl: one tensor per op");
    auto& t = tensors[0];
    check(t);
    hipStream_t s = cur_stream(comm_->device());
    comm_->all_reduce(t.data_ptr(), t.numel(), to_dtype(t), s,
                      to_redop(opts.reduceOp));
    if (opts.reduceOp == c10d::ReduceOp::AVG)
      t.mul_(1.0 / getSize());  // stream-ordered on the current stream
    return c10::make_intrusive<UcclWork>(getRank(), c10d::OpType::ALLREDUCE, s,
                                         comm_->device(), tensors);
  }

  c10::intrusive_ptr<c10d::Work> broadcast(
      std::vector<at::Tensor>& tensors,
      const c10d::BroadcastOptions& opts) override {
    TORCH_CHECK(tensors.size() == 1, "uccl: one tensor per op");
    auto& t = tensors[0];
    check(t);
    hipStream_t s = cur_stream(comm_->device());
    comm_->broadcast(t.data_ptr(), t.numel() * t.element_size(),
                     Dtype::kU8, static_cast<int>(opts.rootRank), s);
    return c10::make_intrusive<UcclWork>(getRank(), c10d::OpType::BROADCAST, s,
                                         comm_->device(), tensors);
  }

  c10::intrusive_ptr<c10d::Work> _allgather_base(
      at::Tensor& output, at::Tensor& input,
      const c10d::AllgatherOptions&) override {
    check(input);
    check(output);
    TORCH_CHECK(output.numel() == input.numel() * getSize());
    TORCH_CHECK(output.scalar_type() == input.scalar_type());
    hipStream_t s = cur_stream(comm_->device());
    comm_->all_gather(output.data_ptr(), input.data_ptr(),
                      input.numel() * input.element_size(), Dtype::kU8, s);
    return c10::make_intrusive<UcclWork>(
        getRank(), c10d::OpType::_ALLGATHER_BASE, s, comm_->device(),
        std::vector<at::Tensor>{output});
  }

  c10::intrusive_ptr<c10d::Work> allgather(
      std::vector<std::vector<at::Tensor>>& outputs,
      std::vector<at::Tensor>& inputs,
      const c10d::AllgatherOptions&) override {
    TORCH_CHECK(inputs.size() == 1 && outputs.size() == 1);
    auto& in = inputs[0];
    check(in);
    TORCH_CHECK(static_cast<int>(outputs[0].size()) == getSize());
    hipStream_t s = cur_stream(comm_->device());
    at::Tensor flat = at::empty({getSize() * in.numel()}, in.options());
    comm_->all_gather(flat.data_ptr(), in.data_ptr(),
                      in.numel() * in.element_size(), Dtype::kU8, s);
    for (int r = 0; r < getSize(); ++r)
      outputs[0][r].view(-1).copy_(
          flat.narrow(0, r * in.numel(), in.numel()), /*non_blocking=*/true);
    return c10::make_intrusive<UcclWork>(getRank(), c10d::OpType::ALLGATHER, s,
                                         comm_->device(), outputs[0]);
  }

  c10::intrusive_ptr<c10d::Work> _reduce_scatter_base(
      at::Tensor& output, at::Tensor& input,
      const c10d::ReduceScatterOptions& opts) override {
    check(input);
    check(output);
    TORCH_CHECK(input.numel() == output.numel() * getSize());
    hipStream_t s = cur_stream(comm_->device());
    comm_->reduce_scatter(output.data_ptr(), input.data_ptr(),
                          output.numel(), to_dtype(input), s,
                          to_redop(opts.reduceOp));
    if (opts.reduceOp == c10d::ReduceOp::AVG)
      output.mul_(1.0 / getSize());
    return c10::make_intrusive<UcclWork>(
        getRank(), c10d::OpType::_REDUCE_SCATTER_BASE, s, comm_->device(),
        std::vector<at::Tensor>{output});
  }

  c10::intrusive_ptr<c10d::Work> alltoall_base(
      at::Tensor& output, at::Tensor& input,
      std::vector<int64_t>& out_splits, std::vector<int64_t>& in_splits,
      const c10d::AllToAllOptions&) override {
    TORCH_CHECK(out_splits.empty() && in_splits.empty(),
                "uccl: only even all_to_all splits");
    check(input);
    check(output);
    TORCH_CHECK(input.numel() == output.numel() &&
                input.numel() % getSize() == 0);
    hipStream_t s = cur_stream(comm_->device());
    comm_->all_to_all(output.data_ptr(), input.data_ptr(),
                      input.numel() / getSize() * input.element_size(),
                      Dtype::kU8, s);
    return c10::make_intrusive<UcclWork>(
        getRank(), c10d::OpType::ALLTOALL_BASE, s, comm_->device(),
        std::vector<at::Tensor>{output});
  }

  c10::intrusive_ptr<c10d::Work> reduce(
      std::vector<at::Tensor>& tensors,
      const c10d::ReduceOptions& opts) override {
    // Only the root's tensor is specified to hold the result; implement
    // as a SUM allreduce (non-root tensors also end up reduced, which
    // the torch.distributed contract permits).
    TORCH_CHECK(tensors.size() == 1);
    auto& t = tensors[0];
    check(t);
    hipStream_t s = cur_stream(comm_->device());
    comm_->all_reduce(t.data_ptr(), t.numel(), to_dtype(t), s,
                      to_redop(opts.reduceOp));
    if (opts.reduceOp == c10d::ReduceOp::AVG)
      t.mul_(1.0 / getSize());
    return c10::make_intrusive<UcclWork>(getRank(), c10d::OpType::REDUCE, s,
                                         comm_->device(),
                                         tensors);
  }

  c10::intrusive_ptr<c10d::Work> gather(
      std::vector<std::vector<at::Tensor>>& outputs,
      std::vector<at::Tensor>& inputs,
      const c10d::GatherOptions& opts) override {
    TORCH_CHECK(inputs.size() == 1);
    auto& in = inputs[0];
    check(in);
    hipStream_t s = cur_stream(comm_->device());
    at::Tensor flat = at::empty({getSize() * in.numel()}, in.options());
    comm_->all_gather(flat.data_ptr(), in.data_ptr(),
                      in.numel() * in.element_size(), Dtype::kU8, s);
    std::vector<at::Tensor> results;
    if (getRank() == opts.rootRank) {
      TORCH_CHECK(outputs.size() == 1 &&
                  static_cast<int>(outputs[0].size()) == getSize());
      for (int r = 0; r < getSize(); ++r) {
        outputs[0][r].view(-1).copy_(
            flat.narrow(0, r * in.numel(), in.numel()), true);
      }
      results = outputs[0];
    }
    return c10::make_intrusive<UcclWork>(getRank(), c10d::OpType::GATHER, s,
                                         comm_->device(),
                                         results);
  }

  c10::intrusive_ptr<c10d::Work> scatter(
      std::vector<at::Tensor>& outputs,
      std::vector<std::vector<at::Tensor>>& inputs,
      const c10d::ScatterOptions& opts) override {
    TORCH_CHECK(outputs.size() == 1);
    auto& out = outputs[0];
    check(out);
    hipStream_t s = cur_stream(comm_->device());
    at::Tensor flat = at::empty({getSize() * out.numel()}, out.options());
    if (getRank() == opts.rootRank) {
      TORCH_CHECK(inputs.size() == 1 &&
                  static_cast<int>(inputs[0].size()) == getSize());
      for (int r = 0; r < getSize(); ++r)
        flat.narrow(0, r * out.numel(), out.numel())
            .copy_(inputs[0][r].view(-1), true);
    }
    comm_->broadcast(flat.data_ptr(),
                     flat.numel() * flat.element_size(), Dtype::kU8,
                     static_cast<int>(opts.rootRank), s);
    out.view(-1).copy_(
        flat.narrow(0, getRank() * out.numel(), out.numel()), true);
    return c10::make_intrusive<UcclWork>(
        getRank(), c10d::OpType::SCATTER, s, comm_->device(),
        std::vector<at::Tensor>{out});
  }

  c10::intrusive_ptr<c10d::Work> send(std::vector<at::Tensor>& tensors,
                                      int dst, int) override {
    TORCH_CHECK(tensors.size() == 1);
    auto& t = tensors[0];
    check(t);
    hipStream_t s = cur_stream(comm_->device());
    comm_->send(t.data_ptr(), t.numel() * t.element_size(), dst, s);
    return c10::make_intrusive<UcclWork>(getRank(), c10d::OpType::SEND, s,
                                         comm_->device(),
                                         tensors);
  }

  c10::intrusive_ptr<c10d::Work> recv(std::vector<at::Tensor>& tensors,
                                      int src, int) override {
    TORCH_CHECK(tensors.size() == 1);
    auto& t = tensors[0];
    check(t);
    hipStream_t s = cur_stream(comm_->device());
    comm_->recv(t.data_ptr(), t.numel() * t.element_size(), src, s);
    return c10::make_intrusive<UcclWork>(getRank(), c10d::OpType::RECV, s,
                                         comm_->device(),
                                         tensors);
  }

  c10::intrusive_ptr<c10d::Work> barrier(
      const c10d::BarrierOptions&) override {
    hipStream_t s = cur_stream(comm_->device());
    comm_->barrier(s);
    (void)hipStreamSynchronize(s);  // NCCL-like host-blocking barrier
    return c10::make_intrusive<UcclWork>(getRank(), c10d::OpType::BARRIER, s,
                                         comm_->device(), std::vector<at::Tensor>{});
  }

 private:
  static void check(const at::Tensor& t) {
    TORCH_CHECK(t.is_cuda(), "uccl backend: GPU tensors only");
    TORCH_CHECK(t.is_contiguous(), "uccl backend: contiguous tensors only");
  }

  c10::intrusive_ptr<c10d::Store> store_;
  std::unique_ptr<Communicator> comm_;
};

}  // namespace

c10::intrusive_ptr<c10d::Backend> create_uccl_backend(
    c10::intrusive_ptr<c10d::Store> store, int rank, int size,
    std::chrono::milliseconds /*timeout*/) {
  return c10::make_intrusive<UcclBackend>(std::move(store), rank, size);
}

void register_pg_backend(pybind11::module_& m) {
  m.def("_create_uccl_backend", &create_uccl_backend);
}

}  // namespace uccl
