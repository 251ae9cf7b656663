// pybind11 module `uccl_amd._C` — Python surface over the native engine.
// Parity role: the reference's nanobind modules uccl.p2p (p2p/engine_api.cc)
// and uccl.ep (ep/src/uccl_ep.cc:1783+); here a single extension hosts the
// collective Communicator (and, as they land, the p2p Endpoint and EP
// Buffer), built HIP-native for gfx950.

#include <torch/extension.h>

#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <hip/hip_runtime.h>

#include "../collective/communicator.h"
#include "../core/log.h"
#include "../ep/ep_buffer.h"
#include "../p2p/endpoint.h"
#include "../transport/reliable.h"
#include "../p2p/compress.h"
#include "../p2p/gpu_codec.h"
#include "../core/latency.h"
#include "../core/ring.h"
#include "../core/trace.h"
#include "../ukernel/ukernel.h"
#include "../ukernel/uk_device.h"

namespace py = pybind11;

namespace uccl {
void register_pg_backend(pybind11::module_& m);
}

using uccl::Communicator;
using uccl::Dtype;
using uccl::p2p::Endpoint;

namespace {

hipStream_t current_stream(int device) {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA(device).stream();
}

Dtype dtype_of(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kFloat: return Dtype::kF32;
    case at::kHalf: return Dtype::kF16;
    case at::kBFloat16: return Dtype::kBF16;
    case at::kInt: return Dtype::kI32;
    case at::kLong: return Dtype::kI64;
    case at::kDouble: return Dtype::kF64;
    case at::kFloat8_e4m3fn: return Dtype::kF8E4M3;
    default:
      TORCH_CHECK(false, "uccl_amd: unsupported dtype ", t.scalar_type());
  }
}

void check_tensor(const at::Tensor& t) {
  TORCH_CHECK(t.is_cuda(), "uccl_amd: tensor must be on GPU");
  TORCH_CHECK(t.is_contiguous(), "uccl_amd: tensor must be contiguous");
}

uccl::RedOp redop_of(const std::string& op) {
  if (op == "sum") return uccl::RedOp::kSum;
  if (op == "prod") return uccl::RedOp::kProd;
  if (op == "min") return uccl::RedOp::kMin;
  if (op == "max") return uccl::RedOp::kMax;
  TORCH_CHECK(false, "uccl_amd: unknown reduce op '", op,
              "' (sum|prod|min|max)");
}

}  // namespace

PYBIND11_MODULE(_C, m) {
  m.doc() = "uccl_amd native engine (MI355X / gfx950)";

  uccl::register_pg_backend(m);

  m.def("device_count", [] {
    int n = 0;
    (void)hipGetDeviceCount(&n);
    return n;
  });

  // peer-access matrix + link attributes (xGMI topology introspection;
  // parity role: the reference's NIC/GPU topology discovery, util.h)
  m.def("peer_matrix", [] {
    int n = 0;
    (void)hipGetDeviceCount(&n);
    std::vector<std::vector<int>> can(n, std::vector<int>(n, 0));
    for (int a = 0; a < n; ++a)
      for (int b = 0; b < n; ++b) {
        if (a == b) {
          can[a][b] = 1;
          continue;
        }
        int ok = 0;
        (void)hipDeviceCanAccessPeer(&ok, a, b);
        can[a][b] = ok;
      }
    return can;
  });

  py::class_<Communicator>(m, "Communicator")
      .def(py::init([](int rank, int world, int device, size_t heap_bytes) {
             return new Communicator(rank, world, device, heap_bytes);
           }),
           py::arg("rank"), py::arg("world"), py::arg("device"),
           py::arg("heap_bytes") = 0)
      .def_property_readonly("rank", &Communicator::rank)
      .def_property_readonly("world", &Communicator::world)
      .def_property_readonly("device", &Communicator::device)
      .def_property_readonly("scratch_capacity",
                             &Communicator::scratch_capacity_bytes)
      .def("handle_bytes",
           [](Communicator& c) { return py::bytes(c.handle_bytes()); })
      .def("connect",
           [](Communicator& c, const std::vector<std::string>& handles) {
             c.connect(handles);
           })
      .def("all_reduce",
           [](Communicator& c, at::Tensor t, const std::string& op) {
             check_tensor(t);
             c.all_reduce(t.data_ptr(), t.numel(), dtype_of(t),
                          current_stream(c.device()), redop_of(op));
           },
           py::arg("tensor"), py::arg("op") = "sum")
      .def("all_gather",
           [](Communicator& c, at::Tensor out, at::Tensor in) {
             check_tensor(out);
             check_tensor(in);
             TORCH_CHECK(out.numel() == in.numel() * c.world(),
                         "all_gather: out must have world*numel(in) elems");
             TORCH_CHECK(out.scalar_type() == in.scalar_type());
             c.all_gather(out.data_ptr(), in.data_ptr(), in.numel(),
                          dtype_of(in), current_stream(c.device()));
           })
      .def("reduce_scatter",
           [](Communicator& c, at::Tensor out, at::Tensor in,
              const std::string& op) {
             check_tensor(out);
             check_tensor(in);
             TORCH_CHECK(in.numel() == out.numel() * c.world(),
                         "reduce_scatter: in must have world*numel(out)");
             TORCH_CHECK(out.scalar_type() == in.scalar_type());
             c.reduce_scatter(out.data_ptr(), in.data_ptr(), out.numel(),
                              dtype_of(in), current_stream(c.device()),
                              redop_of(op));
           },
           py::arg("out"), py::arg("in"), py::arg("op") = "sum")
      .def("broadcast",
           [](Communicator& c, at::Tensor t, int root) {
             check_tensor(t);
             c.broadcast(t.data_ptr(), t.numel(), dtype_of(t), root,
                         current_stream(c.device()));
           })
      .def("all_to_all",
           [](Communicator& c, at::Tensor out, at::Tensor in) {
             check_tensor(out);
             check_tensor(in);
             TORCH_CHECK(out.numel() == in.numel());
             TORCH_CHECK(in.numel() % c.world() == 0,
                         "all_to_all: numel must divide world");
             TORCH_CHECK(out.scalar_type() == in.scalar_type());
             c.all_to_all(out.data_ptr(), in.data_ptr(),
                          in.numel() / c.world(), dtype_of(in),
                          current_stream(c.device()));
           })
      .def("send",
           [](Communicator& c, at::Tensor t, int dst) {
             check_tensor(t);
             c.send(t.data_ptr(), t.numel() * t.element_size(), dst,
                    current_stream(c.device()));
           })
      .def("recv",
           [](Communicator& c, at::Tensor t, int src) {
             check_tensor(t);
             c.recv(t.data_ptr(), t.numel() * t.element_size(), src,
                    current_stream(c.device()));
           })
      .def("barrier", [](Communicator& c) {
        c.barrier(current_stream(c.device()));
      })
      .def("symmetric_tensor",
           [](Communicator& c, std::vector<int64_t> sizes,
              py::object dtype_obj) {
             auto dtype = torch::python::detail::py_object_to_dtype(
                 dtype_obj);
             int64_t numel = 1;
             for (auto s : sizes) numel *= s;
             size_t const bytes =
                 static_cast<size_t>(numel) *
                 c10::elementSize(dtype);
             size_t const off = c.sym_alloc(bytes);
             return at::from_blob(
                 static_cast<char*>(c.heap_base()) + off, sizes,
                 at::TensorOptions().dtype(dtype).device(
                     at::Device(at::kCUDA, c.device())));
           },
           py::arg("sizes"), py::arg("dtype"))
      .def("is_symmetric",
           [](Communicator& c, at::Tensor t) {
             return c.is_symmetric_ptr(t.data_ptr());
           })
      .def("stats", [](Communicator& c) {
        static const char* names[8] = {"all_reduce", "all_gather",
                                       "reduce_scatter", "broadcast",
                                       "all_to_all", "send", "recv",
                                       "barrier"};
        py::dict d;
        auto const& st = c.stats();
        for (int i = 0; i < 8; ++i) {
          py::dict e;
          e["calls"] = st[i].calls;
          e["bytes"] = st[i].bytes;
          d[names[i]] = e;
        }
        return d;
      });

  // --- P2P engine --------------------------------------------------------
  auto dev_of = [](const at::Tensor& t) {
    return t.is_cuda() ? static_cast<int>(t.get_device()) : -1;
  };

  py::class_<Endpoint>(m, "Endpoint")
      .def(py::init<int, int>(), py::arg("gpu") = -1,
           py::arg("num_workers") = 2)
      .def("metadata",
           [](Endpoint& e) { return py::bytes(e.metadata()); })
      .def("connect", &Endpoint::connect,
           py::call_guard<py::gil_scoped_release>())
      .def("accept", &Endpoint::accept,
           py::call_guard<py::gil_scoped_release>())
      .def("num_conns", &Endpoint::num_conns)
      .def("close_conn", &Endpoint::close_conn,
           py::call_guard<py::gil_scoped_release>())
      .def("reg",
           [](Endpoint& e, at::Tensor t) {
             TORCH_CHECK(t.is_contiguous());
             return e.reg(t.data_ptr(), t.numel() * t.element_size(),
                          t.is_cuda() ? t.get_device() : -1);
           })
      .def("dereg", &Endpoint::dereg)
      .def("advertise",
           [](Endpoint& e, uint64_t mr, uint64_t off, uint64_t bytes) {
             return py::bytes(e.advertise(mr, off, bytes));
           },
           py::arg("mr_id"), py::arg("offset") = 0, py::arg("bytes") = 0)
      .def("stats",
           [](Endpoint& e) {
             py::dict d;
             for (auto& [name, st] : e.stats()) {
               py::dict o;
               o["calls"] = st.calls;
               o["bytes"] = st.bytes;
               o["p50_us"] = st.p50_us;
               o["p99_us"] = st.p99_us;
               d[py::str(name)] = o;
             }
             return d;
           })
      .def("send",
           [dev_of](Endpoint& e, uint64_t cid, at::Tensor t) {
             TORCH_CHECK(t.is_contiguous());
             int dev = dev_of(t);
             void* p = t.data_ptr();
             size_t n = t.numel() * t.element_size();
             py::gil_scoped_release rel;
             e.send(cid, p, n, dev);
           })
      .def("recv",
           [dev_of](Endpoint& e, uint64_t cid, at::Tensor t) {
             TORCH_CHECK(t.is_contiguous());
             int dev = dev_of(t);
             void* p = t.data_ptr();
             size_t n = t.numel() * t.element_size();
             py::gil_scoped_release rel;
             e.recv(cid, p, n, dev);
           })
      .def("write",
           [dev_of](Endpoint& e, uint64_t cid, at::Tensor t,
                    const std::string& ad) {
             TORCH_CHECK(t.is_contiguous());
             int dev = dev_of(t);
             void* p = t.data_ptr();
             size_t n = t.numel() * t.element_size();
             py::gil_scoped_release rel;
             e.write(cid, p, n, dev, ad);
           })
      .def("read",
           [dev_of](Endpoint& e, uint64_t cid, at::Tensor t,
                    const std::string& ad) {
             TORCH_CHECK(t.is_contiguous());
             int dev = dev_of(t);
             void* p = t.data_ptr();
             size_t n = t.numel() * t.element_size();
             py::gil_scoped_release rel;
             e.read(cid, p, n, dev, ad);
           })
      .def("send_async",
           [dev_of](Endpoint& e, uint64_t cid, at::Tensor t) {
             TORCH_CHECK(t.is_contiguous());
             return e.send_async(cid, t.data_ptr(),
                                 t.numel() * t.element_size(), dev_of(t));
           })
      .def("recv_async",
           [dev_of](Endpoint& e, uint64_t cid, at::Tensor t) {
             TORCH_CHECK(t.is_contiguous());
             return e.recv_async(cid, t.data_ptr(),
                                 t.numel() * t.element_size(), dev_of(t));
           })
      .def("write_async",
           [dev_of](Endpoint& e, uint64_t cid, at::Tensor t,
                    const std::string& ad) {
             TORCH_CHECK(t.is_contiguous());
             return e.write_async(cid, t.data_ptr(),
                                  t.numel() * t.element_size(), dev_of(t),
                                  ad);
           })
      .def("read_async",
           [dev_of](Endpoint& e, uint64_t cid, at::Tensor t,
                    const std::string& ad) {
             TORCH_CHECK(t.is_contiguous());
             return e.read_async(cid, t.data_ptr(),
                                 t.numel() * t.element_size(), dev_of(t),
                                 ad);
           })
      .def("poll_async", &Endpoint::poll_async,
           py::call_guard<py::gil_scoped_release>());

  // --- EP (DeepEP-compatible expert parallel) -----------------------------
  py::class_<uccl::ep::EpBuffer>(m, "EpBuffer")
      .def(py::init([](int rank, int world, int device, int num_experts,
                       int topk, int hidden, int max_tokens, int elem_size,
                       bool use_fp8, bool with_normal) {
             return new uccl::ep::EpBuffer(rank, world, device, num_experts,
                                           topk, hidden, max_tokens,
                                           elem_size, use_fp8, with_normal);
           }),
           py::arg("rank"), py::arg("world"), py::arg("device"),
           py::arg("num_experts"), py::arg("topk"), py::arg("hidden"),
           py::arg("max_tokens"), py::arg("elem_size"),
           py::arg("use_fp8") = false, py::arg("with_normal") = true)
      .def("handle_bytes",
           [](uccl::ep::EpBuffer& b) { return py::bytes(b.handle_bytes()); })
      .def("connect",
           [](uccl::ep::EpBuffer& b, const std::vector<std::string>& h) {
             b.connect(h);
           })
      .def("dispatch",
           [](uccl::ep::EpBuffer& b, at::Tensor x, at::Tensor topk_idx) {
             TORCH_CHECK(x.is_cuda() && x.is_contiguous());
             TORCH_CHECK(topk_idx.scalar_type() == at::kLong &&
                         topk_idx.is_cuda() && topk_idx.is_contiguous());
             auto const& v = b.view();
             TORCH_CHECK(x.dim() == 2 && x.size(1) == v.hidden);
             TORCH_CHECK(x.element_size() == v.elem_size);
             TORCH_CHECK(topk_idx.dim() == 2 && topk_idx.size(1) == v.topk &&
                         topk_idx.size(0) == x.size(0));
             auto counts = at::empty(
                 {v.local_experts, v.world},
                 at::TensorOptions().dtype(at::kInt).device(x.device()));
             b.dispatch(x.data_ptr(), topk_idx.data_ptr<int64_t>(),
                        static_cast<int>(x.size(0)), counts.data_ptr<int>(),
                        current_stream(b.device()));
             if (v.disp_fp8) {
               auto recv_x = at::from_blob(
                   b.recv_x_ptr(),
                   {v.local_experts,
                    static_cast<int64_t>(v.world) * v.max_tokens, v.hidden},
                   at::TensorOptions().dtype(at::kFloat8_e4m3fn)
                       .device(x.device()));
               auto recv_scale = at::from_blob(
                   b.recv_scale_ptr(),
                   {v.local_experts,
                    static_cast<int64_t>(v.world) * v.max_tokens,
                    v.hidden / 128},
                   at::TensorOptions().dtype(at::kFloat).device(x.device()));
               return py::make_tuple(recv_x, counts, recv_scale);
             }
             auto recv_x = at::from_blob(
                 b.recv_x_ptr(),
                 {v.local_experts,
                  static_cast<int64_t>(v.world) * v.max_tokens, v.hidden},
                 at::TensorOptions().dtype(x.scalar_type())
                     .device(x.device()));
             return py::make_tuple(recv_x, counts);
           })
      // phase-split halves (DeepEP SEND|RECV split + cached-handle
      // replay; see EpBuffer::dispatch_send)
      .def("dispatch_send",
           [](uccl::ep::EpBuffer& b, at::Tensor x, at::Tensor topk_idx,
              bool reuse_plan) {
             TORCH_CHECK(x.is_cuda() && x.is_contiguous());
             TORCH_CHECK(topk_idx.scalar_type() == at::kLong &&
                         topk_idx.is_cuda() && topk_idx.is_contiguous());
             auto const& v = b.view();
             TORCH_CHECK(x.dim() == 2 && x.size(1) == v.hidden);
             TORCH_CHECK(topk_idx.dim() == 2 && topk_idx.size(1) == v.topk);
             b.dispatch_send(x.data_ptr(), topk_idx.data_ptr<int64_t>(),
                             static_cast<int>(x.size(0)), reuse_plan,
                             current_stream(b.device()));
           },
           py::arg("x"), py::arg("topk_idx"),
           py::arg("reuse_plan") = false)
      .def("dispatch_recv",
           [](uccl::ep::EpBuffer& b, at::Tensor counts) {
             auto const& v = b.view();
             TORCH_CHECK(counts.is_cuda() && counts.is_contiguous() &&
                         counts.scalar_type() == at::kInt &&
                         counts.numel() == v.local_experts * v.world);
             b.dispatch_recv(counts.data_ptr<int>(),
                             current_stream(b.device()));
             return counts;
           })
      .def("recv_x_view",
           [](uccl::ep::EpBuffer& b, py::object dtype_obj) {
             auto const& v = b.view();
             auto dt = v.disp_fp8
                           ? at::kFloat8_e4m3fn
                           : (v.elem_size == 2 ? at::kBFloat16 : at::kFloat);
             return at::from_blob(
                 b.recv_x_ptr(),
                 {v.local_experts,
                  static_cast<int64_t>(v.world) * v.max_tokens, v.hidden},
                 at::TensorOptions().dtype(dt).device(
                     at::Device(at::kCUDA, b.device())));
           },
           py::arg("dtype") = py::none())
      .def("recv_scale_view",
           [](uccl::ep::EpBuffer& b) {
             auto const& v = b.view();
             TORCH_CHECK(v.disp_fp8, "scales exist only in fp8 mode");
             return at::from_blob(
                 b.recv_scale_ptr(),
                 {v.local_experts,
                  static_cast<int64_t>(v.world) * v.max_tokens,
                  v.hidden / 128},
                 at::TensorOptions().dtype(at::kFloat).device(
                     at::Device(at::kCUDA, b.device())));
           })
      .def("combine_send",
           [](uccl::ep::EpBuffer& b, at::Tensor expert_out) {
             TORCH_CHECK(expert_out.is_cuda() && expert_out.is_contiguous());
             TORCH_CHECK(expert_out.element_size() == b.view().elem_size);
             b.combine_send(expert_out.data_ptr(),
                            current_stream(b.device()));
           })
      .def("combine_recv",
           [](uccl::ep::EpBuffer& b, at::Tensor out, at::Tensor topk_idx,
              at::Tensor topk_w) {
             auto const& v = b.view();
             TORCH_CHECK(out.is_cuda() && out.is_contiguous() &&
                         out.element_size() == v.elem_size &&
                         out.dim() == 2 && out.size(0) == topk_idx.size(0) &&
                         out.size(1) == v.hidden);
             TORCH_CHECK(topk_idx.scalar_type() == at::kLong &&
                         topk_idx.is_contiguous());
             TORCH_CHECK(topk_w.scalar_type() == at::kFloat &&
                         topk_w.is_contiguous());
             b.combine_recv(out.data_ptr(), topk_idx.data_ptr<int64_t>(),
                            topk_w.data_ptr<float>(),
                            current_stream(b.device()));
             return out;
           })
      // normal (rank-granular) mode — DeepEP HT dispatch/combine
      .def("nrm_dispatch_send",
           [](uccl::ep::EpBuffer& b, at::Tensor x, at::Tensor topk_idx,
              py::object topk_w) {
             auto const& v = b.view();
             TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
                         x.dim() == 2 && x.size(1) == v.hidden &&
                         x.element_size() == v.elem_size);
             TORCH_CHECK(topk_idx.scalar_type() == at::kLong &&
                         topk_idx.is_cuda() && topk_idx.is_contiguous() &&
                         topk_idx.size(1) == v.topk);
             float const* wp = nullptr;
             at::Tensor w;
             if (!topk_w.is_none()) {
               w = topk_w.cast<at::Tensor>();
               TORCH_CHECK(w.scalar_type() == at::kFloat &&
                           w.is_contiguous() && w.is_cuda());
               wp = w.data_ptr<float>();
             }
             b.nrm_dispatch_send(x.data_ptr(), topk_idx.data_ptr<int64_t>(),
                                 wp, static_cast<int>(x.size(0)),
                                 current_stream(b.device()));
           },
           py::arg("x"), py::arg("topk_idx"),
           py::arg("topk_w") = py::none())
      .def("nrm_dispatch_recv",
           [](uccl::ep::EpBuffer& b, at::Tensor counts) {
             auto const& v = b.view();
             TORCH_CHECK(counts.is_cuda() && counts.is_contiguous() &&
                         counts.scalar_type() == at::kInt &&
                         counts.numel() == v.world);
             b.nrm_dispatch_recv(counts.data_ptr<int>(),
                                 current_stream(b.device()));
             return counts;
           })
      .def("nrm_combine_send",
           [](uccl::ep::EpBuffer& b, at::Tensor x) {
             auto const& v = b.view();
             TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
                         x.element_size() == v.elem_size);
             b.nrm_combine_send(x.data_ptr(), current_stream(b.device()));
           })
      .def("nrm_combine_recv",
           [](uccl::ep::EpBuffer& b, at::Tensor out, at::Tensor topk_idx) {
             auto const& v = b.view();
             TORCH_CHECK(out.is_cuda() && out.is_contiguous() &&
                         out.dim() == 2 && out.size(1) == v.hidden);
             TORCH_CHECK(topk_idx.scalar_type() == at::kLong &&
                         topk_idx.is_cuda() && topk_idx.is_contiguous());
             b.nrm_combine_recv(out.data_ptr(),
                                topk_idx.data_ptr<int64_t>(),
                                current_stream(b.device()));
             return out;
           })
      .def("nrm_x_view",
           [](uccl::ep::EpBuffer& b) {
             auto const& v = b.view();
             auto dt = v.elem_size == 2 ? at::kBFloat16 : at::kFloat;
             return at::from_blob(
                 b.nrm_x_base(),
                 {v.world, v.max_tokens, v.hidden},
                 at::TensorOptions().dtype(dt).device(
                     at::Device(at::kCUDA, b.device())));
           })
      .def("nrm_meta_view",
           [](uccl::ep::EpBuffer& b) {
             auto const& v = b.view();
             return at::from_blob(
                 b.nrm_meta_base(), {v.world, v.max_tokens},
                 at::TensorOptions().dtype(at::kInt).device(
                     at::Device(at::kCUDA, b.device())));
           })
      .def("nrm_topk_view",
           [](uccl::ep::EpBuffer& b) {
             auto const& v = b.view();
             return at::from_blob(
                 b.nrm_topk_base(), {v.world, v.max_tokens, v.topk},
                 at::TensorOptions().dtype(at::kLong).device(
                     at::Device(at::kCUDA, b.device())));
           })
      .def("nrm_w_view",
           [](uccl::ep::EpBuffer& b) {
             auto const& v = b.view();
             return at::from_blob(
                 b.nrm_w_base(), {v.world, v.max_tokens, v.topk},
                 at::TensorOptions().dtype(at::kFloat).device(
                     at::Device(at::kCUDA, b.device())));
           })
      // proxy sync commands (reference proxy ATOMIC/BARRIER/QUIET parity)
      .def("barrier",
           [](uccl::ep::EpBuffer& b) {
             b.barrier(current_stream(b.device()));
           })
      .def("quiet",
           [](uccl::ep::EpBuffer& b) {
             b.quiet(current_stream(b.device()));
           })
      .def("atomic_add",
           [](uccl::ep::EpBuffer& b, int dst, uint64_t value) {
             b.atomic_add(dst, value, current_stream(b.device()));
           })
      .def("read_sync_word",
           [](uccl::ep::EpBuffer& b, int idx) {
             return b.read_sync_word(idx);
           })
      .def("check_error",
           [](uccl::ep::EpBuffer& b) { b.check_error(); })
      .def("combine",
           [](uccl::ep::EpBuffer& b, at::Tensor expert_out,
              at::Tensor topk_idx, at::Tensor topk_w) {
             auto const& v = b.view();
             TORCH_CHECK(expert_out.is_cuda() && expert_out.is_contiguous());
             TORCH_CHECK(expert_out.element_size() == v.elem_size);
             TORCH_CHECK(topk_idx.scalar_type() == at::kLong &&
                         topk_idx.is_contiguous());
             TORCH_CHECK(topk_w.scalar_type() == at::kFloat &&
                         topk_w.is_contiguous());
             auto out = at::empty(
                 {topk_idx.size(0), v.hidden},
                 at::TensorOptions().dtype(expert_out.scalar_type())
                     .device(expert_out.device()));
             b.combine(expert_out.data_ptr(), out.data_ptr(),
                       topk_idx.data_ptr<int64_t>(),
                       topk_w.data_ptr<float>(),
                       current_stream(b.device()));
             return out;
           });

  // --- software multipath reliable transport ------------------------------
  using uccl::transport::TransportEndpoint;
  py::class_<uccl::transport::Stats>(m, "TransportStats")
      .def_readonly("data_sent", &uccl::transport::Stats::data_sent)
      .def_readonly("data_recv", &uccl::transport::Stats::data_recv)
      .def_readonly("acks_sent", &uccl::transport::Stats::acks_sent)
      .def_readonly("acks_recv", &uccl::transport::Stats::acks_recv)
      .def_readonly("retransmits", &uccl::transport::Stats::retransmits)
      .def_readonly("rto_retransmits",
                    &uccl::transport::Stats::rto_retransmits)
      .def_readonly("injected_drops",
                    &uccl::transport::Stats::injected_drops)
      .def_readonly("dup_recv", &uccl::transport::Stats::dup_recv)
      .def_readonly("send_fail", &uccl::transport::Stats::send_fail)
      .def_readonly("msgs_sent", &uccl::transport::Stats::msgs_sent)
      .def_readonly("msgs_recv", &uccl::transport::Stats::msgs_recv)
      .def_readonly("srtt_us", &uccl::transport::Stats::srtt_us)
      .def_readonly("cwnd", &uccl::transport::Stats::cwnd)
      .def_readonly("rtt_p50_us", &uccl::transport::Stats::rtt_p50_us)
      .def_readonly("rtt_p99_us", &uccl::transport::Stats::rtt_p99_us);

  py::class_<TransportEndpoint>(m, "TransportEndpoint")
      .def(py::init<int, size_t>(), py::arg("num_paths") = 8,
           py::arg("chunk_bytes") = 8192)
      .def("metadata", &TransportEndpoint::metadata)
      .def("connect",
           [](TransportEndpoint& e, const std::string& md, uint64_t tag) {
             py::gil_scoped_release rel;
             return e.connect(md, tag);
           },
           py::arg("metadata"), py::arg("tag") = 0)
      .def("accept",
           [](TransportEndpoint& e) {
             py::gil_scoped_release rel;
             return e.accept(nullptr);
           })
      .def("send",
           [](TransportEndpoint& e, uint64_t flow, at::Tensor t) {
             TORCH_CHECK(t.is_contiguous() && !t.is_cuda(),
                         "transport send: host tensors");
             void* p = t.data_ptr();
             size_t n = t.numel() * t.element_size();
             py::gil_scoped_release rel;
             e.send_msg(flow, p, n);
           })
      .def("post_send",
           [](TransportEndpoint& e, uint64_t flow, at::Tensor t) {
             TORCH_CHECK(t.is_contiguous() && !t.is_cuda(),
                         "transport post_send: host tensors");
             // caller keeps the tensor alive until flush()
             void* p = t.data_ptr();
             size_t n = t.numel() * t.element_size();
             py::gil_scoped_release rel;
             e.send_msg_async(flow, p, n);
           })
      .def("flush",
           [](TransportEndpoint& e, uint64_t flow) {
             py::gil_scoped_release rel;
             e.flush_sends(flow);
           })
      .def("close_flow",
           [](TransportEndpoint& e, uint64_t flow) {
             py::gil_scoped_release rel;
             e.close_flow(flow);
           })
      .def("recv",
           [](TransportEndpoint& e, uint64_t flow, at::Tensor t) {
             TORCH_CHECK(t.is_contiguous() && !t.is_cuda(),
                         "transport recv: host tensors");
             void* p = t.data_ptr();
             size_t n = t.numel() * t.element_size();
             py::gil_scoped_release rel;
             e.recv_msg(flow, p, n);
           })
      .def("stats", &TransportEndpoint::stats);

  // --- tracing / latency observability (csrc/core) -------------------------
  // reference analogs: NPKit event tracing (lite-collective core/npkit.cc)
  // and the latency percentile recorder (include/util/latency.h)
  m.def("trace_enabled", &uccl::trace::enabled);
  m.def("trace_set_enabled", &uccl::trace::set_enabled);
  m.def("trace_dump_json", &uccl::trace::dump_json);
  m.def("trace_clear", &uccl::trace::clear);
  m.def("trace_num_events", &uccl::trace::num_events);
  py::class_<uccl::SpscRing<uint64_t>>(m, "SpscRingU64")
      .def(py::init<size_t>(), py::arg("capacity_pow2"))
      .def("push", &uccl::SpscRing<uint64_t>::push)
      .def("pop",
           [](uccl::SpscRing<uint64_t>& r) -> py::object {
             uint64_t v;
             if (r.pop(&v)) return py::cast(v);
             return py::none();
           })
      .def("size", &uccl::SpscRing<uint64_t>::size);
  py::class_<uccl::MpmcRing<uint64_t>>(m, "MpmcRingU64")
      .def(py::init<size_t>(), py::arg("capacity_pow2"))
      .def("push", &uccl::MpmcRing<uint64_t>::push,
           py::call_guard<py::gil_scoped_release>())
      .def("pop",
           [](uccl::MpmcRing<uint64_t>& r) -> py::object {
             uint64_t v;
             bool ok;
             {
               py::gil_scoped_release rel;
               ok = r.pop(&v);
             }
             if (ok) return py::cast(v);
             return py::none();
           })
      .def("size_approx", &uccl::MpmcRing<uint64_t>::size_approx);
  py::class_<uccl::LatencyHist>(m, "LatencyHist")
      .def(py::init<>())
      .def("record_us", &uccl::LatencyHist::record_us)
      .def("record_ns", &uccl::LatencyHist::record_ns)
      .def("count", &uccl::LatencyHist::count)
      .def("percentile_us", &uccl::LatencyHist::percentile_us)
      .def("reset", &uccl::LatencyHist::reset);

  // --- lossless float codec (csrc/p2p/compress) ----------------------------
  // reference analog: DietGPU compression layer, p2p/rdma/compression.cc
  namespace comp = uccl::p2p::comp;
  m.def("comp_compress",
        [](at::Tensor t, int strategy) {
          TORCH_CHECK(t.is_contiguous() && !t.is_cuda(),
                      "compress: host contiguous tensors");
          int elem = static_cast<int>(t.element_size());
          int code;
          switch (t.scalar_type()) {
            case at::kFloat: code = 0; break;
            case at::kHalf: code = 1; break;
            case at::kBFloat16: code = 2; break;
            default: code = 3; elem = 1;
          }
          size_t const nbytes = t.numel() * t.element_size();
          std::string s;
          {
            py::gil_scoped_release rel;
            s = comp::compress(t.data_ptr(), nbytes, elem, code, strategy);
          }
          auto out = at::empty({static_cast<int64_t>(s.size())}, at::kByte);
          std::memcpy(out.data_ptr(), s.data(), s.size());
          return out;
        },
        py::arg("tensor"), py::arg("strategy") = static_cast<int>(comp::kSplitDeflate));
  m.def("comp_decompress", [](at::Tensor frame) {
    TORCH_CHECK(frame.is_contiguous() && !frame.is_cuda() &&
                    frame.scalar_type() == at::kByte,
                "decompress: host uint8 frame");
    size_t const fb = frame.numel();
    size_t const n = comp::orig_bytes(frame.data_ptr(), fb);
    int const code = comp::dtype_code(frame.data_ptr(), fb);
    at::ScalarType const st = code == 0   ? at::kFloat
                              : code == 1 ? at::kHalf
                              : code == 2 ? at::kBFloat16
                                          : at::kByte;
    int const es = code == 3 ? 1 : (code == 0 ? 4 : 2);
    auto out = at::empty({static_cast<int64_t>(n / es)}, st);
    {
      py::gil_scoped_release rel;
      comp::decompress(frame.data_ptr(), fb, out.data_ptr(), n);
    }
    return out;
  });

  // --- chunk-graph planner / spray executor (csrc/ukernel) -----------------
  // reference analog: experimental/ukernel planner->lower->SprayExecutor,
  // unit-tested the same way (host mock backend, single process)
  namespace uk = uccl::uk;
  py::class_<uk::Topology>(m, "UkTopology")
      .def(py::init<int>(), py::arg("world"))
      .def_readonly("world", &uk::Topology::world)
      .def("set_link_weight",
           [](uk::Topology& t, int src, int dst, double w) {
             TORCH_CHECK(src >= 0 && src < t.world && dst >= 0 &&
                         dst < t.world && src != dst, "bad link");
             t.link_weight[size_t(src) * t.world + dst] = w;
           });
  py::class_<uk::ChunkGraph>(m, "UkGraph")
      .def_property_readonly(
          "num_tasks",
          [](uk::ChunkGraph const& g) { return g.tasks.size(); })
      .def_readonly("world", &uk::ChunkGraph::world)
      .def_readonly("scratch_bytes", &uk::ChunkGraph::scratch_bytes)
      .def("dump", &uk::ChunkGraph::dump);
  m.def("uk_plan_sendrecv", &uk::plan_sendrecv_spray, py::arg("topo"),
        py::arg("src"), py::arg("dst"), py::arg("nbytes"),
        py::arg("chunk_bytes"));
  m.def("uk_plan_allreduce_rsag", &uk::plan_allreduce_rsag, py::arg("topo"),
        py::arg("nbytes"), py::arg("elem_bytes") = 4,
        py::arg("chunk_bytes") = 1 << 20);
  m.def("uk_plan_allreduce_oneshot", &uk::plan_allreduce_oneshot,
        py::arg("topo"), py::arg("nbytes"), py::arg("elem_bytes") = 4);
  m.def("uk_plan_broadcast", &uk::plan_broadcast, py::arg("topo"),
        py::arg("root"), py::arg("nbytes"), py::arg("chunk_bytes"));
  m.def("uk_plan_allgather", &uk::plan_allgather, py::arg("topo"),
        py::arg("nbytes"), py::arg("chunk_bytes") = 1 << 20);
  m.def("uk_plan_reducescatter", &uk::plan_reducescatter, py::arg("topo"),
        py::arg("shard_bytes"), py::arg("elem_bytes") = 4,
        py::arg("chunk_bytes") = 1 << 20);
  m.def("uk_plan_alltoall", &uk::plan_alltoall, py::arg("topo"),
        py::arg("seg_bytes"), py::arg("chunk_bytes") = 1 << 20);
  m.def("uk_lower", &uk::lower);
  m.def("uk_estimate_us", &uk::estimate_us, py::arg("graph"),
        py::arg("topo"), py::arg("link_gbps") = 150.0,
        py::arg("local_gbps") = 1500.0, py::arg("overhead_us") = 4.0);
  m.def("uk_plan_allreduce_auto", &uk::plan_allreduce_auto, py::arg("topo"),
        py::arg("nbytes"), py::arg("elem_bytes") = 4,
        py::arg("chunk_bytes") = 1 << 20);
  m.def("uk_execute_host",
        [](uk::ChunkGraph const& g, std::vector<at::Tensor> inputs,
           int64_t out_bytes) {
          int const world = g.world;
          TORCH_CHECK(static_cast<int>(inputs.size()) == world,
                      "one input tensor per rank");
          for (auto const& t : inputs)
            TORCH_CHECK(t.is_contiguous() && !t.is_cuda() &&
                            t.scalar_type() == at::kFloat,
                        "host float32 contiguous inputs");
          uint64_t const in_bytes = inputs[0].numel() * 4;
          uk::HostBackend hb(world, in_bytes, out_bytes, g.scratch_bytes);
          for (int r = 0; r < world; ++r)
            std::memcpy(hb.input(r), inputs[r].data_ptr<float>(), in_bytes);
          uk::ExecStats st;
          {
            py::gil_scoped_release rel;
            st = uk::execute(g, hb);
          }
          std::vector<at::Tensor> outs;
          for (int r = 0; r < world; ++r) {
            auto t = at::empty({out_bytes / 4}, at::kFloat);
            std::memcpy(t.data_ptr<float>(), hb.output(r), out_bytes);
            outs.push_back(t);
          }
          py::dict d;
          d["tasks_run"] = st.tasks_run;
          d["wait_requeues"] = st.wait_requeues;
          d["link_bytes"] = st.link_bytes;
          return py::make_tuple(outs, d);
        },
        py::arg("graph"), py::arg("inputs"), py::arg("out_bytes"));

  // --- GPU lossless codec (DietGPU-role: plane split + 64-lane rANS) ------
  m.def("gpu_compress",
        [](at::Tensor t, int nplanes) {
          TORCH_CHECK(t.is_cuda() && t.is_contiguous());
          int const es = static_cast<int>(t.element_size());
          TORCH_CHECK(es == 1 || es == 2 || es == 4, "elem size 1|2|4");
          if (nplanes <= 0) nplanes = es;
          size_t const bytes = t.numel() * es;
          size_t const cap = uccl::p2p::gpu::compress_bound(bytes, nplanes);
          auto out = at::empty({static_cast<int64_t>(cap)},
                               at::TensorOptions().dtype(at::kByte)
                                   .device(t.device()));
          size_t w;
          {
            py::gil_scoped_release rel;
            w = uccl::p2p::gpu::compress(
                t.data_ptr(), bytes, es, nplanes, out.data_ptr(), cap,
                current_stream(t.get_device()));
          }
          return out.narrow(0, 0, static_cast<int64_t>(w));
        },
        py::arg("tensor"), py::arg("nplanes") = 0)
  ;
  m.def("gpu_decompress",
        [](at::Tensor frame, at::Tensor out) {
          TORCH_CHECK(frame.is_cuda() && frame.is_contiguous() &&
                      frame.scalar_type() == at::kByte);
          TORCH_CHECK(out.is_cuda() && out.is_contiguous());
          size_t const cap = out.numel() * out.element_size();
          size_t r;
          {
            py::gil_scoped_release rel;
            r = uccl::p2p::gpu::decompress(
                frame.data_ptr(), frame.numel(), out.data_ptr(), cap,
                current_stream(frame.get_device()));
          }
          return static_cast<int64_t>(r);
        },
        py::arg("frame"), py::arg("out"));
  m.def("gpu_codec_host_selftest", [](at::Tensor t) {
    TORCH_CHECK(!t.is_cuda() && t.scalar_type() == at::kByte &&
                t.is_contiguous());
    std::vector<uint8_t> v(t.data_ptr<uint8_t>(),
                           t.data_ptr<uint8_t>() + t.numel());
    return uccl::p2p::gpu::host_rans_selftest(v);
  });

  // Same contract on the DEVICE backend: persistent worker kernels drain
  // per-rank C2D task FIFOs (the reference's persistent_kernel_ops.cu
  // role); outputs round-trip through the device so a spray-planned
  // transfer demonstrably runs on the GPU.
  m.def("uk_execute_device",
        [](uk::ChunkGraph const& g, std::vector<at::Tensor> inputs,
           int64_t out_bytes) {
          int const world = g.world;
          TORCH_CHECK(static_cast<int>(inputs.size()) == world,
                      "one input tensor per rank");
          for (auto const& t : inputs)
            TORCH_CHECK(t.is_contiguous() && !t.is_cuda() &&
                            t.scalar_type() == at::kFloat,
                        "host float32 contiguous inputs");
          uint64_t const in_bytes = inputs[0].numel() * 4;
          uk::DeviceBackend db(world, in_bytes, out_bytes, g.scratch_bytes);
          for (int r = 0; r < world; ++r)
            db.upload_input(r, inputs[r].data_ptr<float>(), in_bytes);
          uk::ExecStats st;
          {
            py::gil_scoped_release rel;
            st = uk::execute(g, db);
          }
          std::vector<at::Tensor> outs;
          for (int r = 0; r < world; ++r) {
            auto t = at::empty({out_bytes / 4}, at::kFloat);
            db.download_output(r, t.data_ptr<float>(), out_bytes);
            outs.push_back(t);
          }
          py::dict d;
          d["tasks_run"] = st.tasks_run;
          d["wait_requeues"] = st.wait_requeues;
          d["link_bytes"] = st.link_bytes;
          return py::make_tuple(outs, d);
        },
        py::arg("graph"), py::arg("inputs"), py::arg("out_bytes"));
}
