// libuccl_nccl.so — NCCL/RCCL C-ABI drop-in over the uccl_amd xGMI
// collective engine.
//
// Parity role: the reference's lite-collective NCCL API library
// (experimental/lite/lite-collective/nccl/nccl.cu:1455+ — ncclCommInitRank,
// ncclAllReduce, ... implemented on its own device kernels). Here the same
// surface (comm lifecycle via ncclUniqueId TCP bootstrap, the collective
// entry points, group semantics) sits on uccl_amd's Communicator. Types
// come from the system rccl.h, so the ABI matches what rccl-tests and
// NCCL-linked apps expect; LD_PRELOAD (or plain linking) swaps the data
// plane.
//
// Scope: single node (up to 8 ranks over xGMI), sum/prod/min/max/avg and
// premulsum reductions, the dtypes the engine supports, NCCL group
// semantics for p2p (deferred + per-peer streams so group{send;recv}
// pairs >2MB can't deadlock on the slot-credit pacing). Everything else
// returns a clean error.

#include <rccl/rccl.h>

#include <algorithm>
#include <atomic>
#include <cmath>
#include <condition_variable>
#include <cstring>
#include <map>
#include <mutex>
#include <thread>
#include <vector>

#include "../collective/communicator.h"
#include "../core/log.h"
#include "../core/net.h"

namespace {

using uccl::Communicator;
using uccl::Dtype;

// --- uniqueId bootstrap ----------------------------------------------------
// ncclUniqueId (128 B) carries {magic, ip, port} of a one-shot exchange
// server started by ncclGetUniqueId in the calling process: every rank
// connects, submits {rank, nranks, ipc blob}, and receives all ranks'
// blobs once the last one arrives.

constexpr uint64_t kIdMagic = 0x7563636c6e636cULL;

struct IdPayload {
  uint64_t magic;
  char ip[48];
  uint16_t port;
};

struct ExchangeServer {
  int listen_fd = -1;
  std::thread th;
  ~ExchangeServer() {
    if (listen_fd >= 0) {
      ::shutdown(listen_fd, SHUT_RDWR);
      ::close(listen_fd);
    }
    if (th.joinable()) th.join();
  }
};

std::mutex g_mu;
std::vector<std::unique_ptr<ExchangeServer>> g_servers;

struct WireBlob {
  uint64_t magic;  // rejects stray connections on recycled ports
  int rank;
  int nranks;
  uint32_t len;
};
constexpr uint64_t kBlobMagic = 0x7563636c2d696478ULL;
constexpr uint32_t kMaxBlobLen = 1u << 20;  // IPC blobs are ~KB

void serve(int listen_fd) {
  // collect blobs until all nranks have reported, then broadcast
  std::vector<int> fds;
  std::vector<std::string> blobs;
  int expected = -1;
  int got = 0;
  while (expected < 0 || got < expected) {
    int fd = ::accept(listen_fd, nullptr, nullptr);
    if (fd < 0) return;  // shut down
    WireBlob h{};
    if (!uccl::net::recv_all(fd, &h, sizeof(h)) || h.magic != kBlobMagic ||
        h.nranks < 1 || h.nranks > 8 || h.len > kMaxBlobLen) {
      ::close(fd);
      continue;
    }
    if (expected < 0) {
      expected = h.nranks;
      blobs.resize(expected);
    }
    std::string b(h.len, '\0');
    uccl::net::recv_all(fd, b.data(), h.len);
    if (h.rank >= 0 && h.rank < expected && blobs[h.rank].empty()) {
      blobs[h.rank] = std::move(b);
      ++got;
    }
    fds.push_back(fd);
  }
  // reply: nranks, then len+blob per rank
  for (int fd : fds) {
    uint32_t n = static_cast<uint32_t>(blobs.size());
    uccl::net::send_all(fd, &n, sizeof(n));
    for (auto const& b : blobs) {
      uint32_t l = static_cast<uint32_t>(b.size());
      uccl::net::send_all(fd, &l, sizeof(l));
      uccl::net::send_all(fd, b.data(), l);
    }
    ::close(fd);
  }
}

struct PreMulScalar {
  bool device_resident = false;
  void const* dev_ptr = nullptr;  // ncclScalarDevice: read at execution
  double value = 1.0;             // ncclScalarHostImmediate: copied at create
  ncclDataType_t dtype = ncclFloat32;
};

struct UcclComm {
  std::unique_ptr<Communicator> comm;
  int device = 0;
  std::mutex mu;
  std::vector<PreMulScalar> premul;  // ncclRedOpCreatePreMulSum handles
};

// --- group semantics -------------------------------------------------------
// NCCL requires ops inside ncclGroupStart/End to be able to progress
// concurrently. Collectives are stream-async on the engine already; p2p ops
// are deferred here and re-issued on per-(peer,dir) streams at GroupEnd,
// bracketed by events so the user stream observes them in order.

struct PendingP2P {
  UcclComm* uc;
  bool is_send;
  void* buff;
  size_t bytes;
  int peer;
  hipStream_t stream;
};

thread_local int g_group_depth = 0;
thread_local std::vector<PendingP2P> g_pending;

ncclResult_t flush_group() {
  // self-pairs (rank sending to itself) become plain DtoD copies
  std::vector<PendingP2P> ops;
  ops.swap(g_pending);
  std::vector<size_t> self_sends, self_recvs;
  for (size_t i = 0; i < ops.size(); ++i) {
    auto& o = ops[i];
    if (o.peer == o.uc->comm->rank()) {
      (o.is_send ? self_sends : self_recvs).push_back(i);
    }
  }
  if (self_sends.size() != self_recvs.size()) return ncclInvalidUsage;
  for (size_t k = 0; k < self_sends.size(); ++k) {
    auto& s = ops[self_sends[k]];
    auto& r = ops[self_recvs[k]];
    if (s.bytes != r.bytes) return ncclInvalidUsage;
    (void)hipMemcpyAsync(r.buff, s.buff, s.bytes, hipMemcpyDeviceToDevice,
                         s.stream);
    if (r.stream != s.stream) {
      hipEvent_t e;
      (void)hipEventCreateWithFlags(&e, hipEventDisableTiming);
      (void)hipEventRecord(e, s.stream);
      (void)hipStreamWaitEvent(r.stream, e, 0);
      (void)hipEventDestroy(e);
    }
  }
  // Peer ops are re-issued on the USER stream in a canonical order both
  // sides agree on: sort by peer rank, and within a pairing the lower
  // rank enqueues its sends before its recvs while the higher rank
  // enqueues recvs first. Each matched send/recv chain then completes
  // pair-by-pair on a single stream — no credit-window deadlock, and no
  // extra streams (per-(peer,dir) streams deadlocked on ROCm when many
  // spin-wait kernels shared the ~4 HW queues of one device).
  std::vector<PendingP2P> peer_ops;
  for (auto& o : ops)
    if (o.peer != o.uc->comm->rank()) peer_ops.push_back(o);
  std::stable_sort(peer_ops.begin(), peer_ops.end(),
                   [](PendingP2P const& a, PendingP2P const& b) {
                     if (a.peer != b.peer) return a.peer < b.peer;
                     int const ra = a.uc->comm->rank();
                     int const pa =
                         (a.is_send == (ra < a.peer)) ? 0 : 1;
                     int const pb =
                         (b.is_send == (ra < b.peer)) ? 0 : 1;
                     return pa < pb;
                   });
  for (auto& o : peer_ops) {
    try {
      if (o.is_send)
        o.uc->comm->send(o.buff, o.bytes, o.peer, o.stream);
      else
        o.uc->comm->recv(o.buff, o.bytes, o.peer, o.stream);
    } catch (std::exception const& e) {
      UCCL_LOG_ERROR << "grouped p2p failed: " << e.what();
      return ncclInternalError;
    }
  }
  return ncclSuccess;
}

ncclResult_t to_dtype(ncclDataType_t t, Dtype* out) {
  switch (t) {
    case ncclFloat32: *out = Dtype::kF32; return ncclSuccess;
    case ncclFloat16: *out = Dtype::kF16; return ncclSuccess;
    case ncclBfloat16: *out = Dtype::kBF16; return ncclSuccess;
    case ncclInt32: *out = Dtype::kI32; return ncclSuccess;
    case ncclInt8: *out = Dtype::kU8; return ncclSuccess;  // copy ops only
    case ncclInt64: *out = Dtype::kI64; return ncclSuccess;
    case ncclFloat64: *out = Dtype::kF64; return ncclSuccess;
    default: return ncclInvalidArgument;
  }
}

bool is_float_dtype(Dtype d) {
  return d == Dtype::kF32 || d == Dtype::kF16 || d == Dtype::kBF16 ||
         d == Dtype::kF64 || d == Dtype::kF8E4M3;
}

// Decompose an ncclRedOp into {engine op, pre-scale, post-scale}:
//   avg       = sum, post-scale 1/world
//   premulsum = pre-scale by this rank's scalar, sum
struct OpPlan {
  uccl::RedOp op = uccl::RedOp::kSum;
  bool prescale = false;
  double prescale_by = 1.0;
  bool postscale = false;
  double postscale_by = 1.0;
};

ncclResult_t plan_op(UcclComm* uc, ncclRedOp_t op, Dtype dt, OpPlan* plan) {
  *plan = OpPlan{};
  if (op >= ncclNumOps) {
    size_t const idx = static_cast<size_t>(op) - ncclNumOps;
    std::lock_guard<std::mutex> g(uc->mu);
    if (idx >= uc->premul.size()) return ncclInvalidArgument;
    PreMulScalar const& s = uc->premul[idx];
    double v = s.value;
    if (s.device_resident) {
      // ncclScalarDevice: the scalar lives in device memory and is read at
      // execution time; small sync copy here keeps the semantics.
      if (s.dtype == ncclFloat32) {
        float f = 1.f;
        (void)hipMemcpy(&f, s.dev_ptr, sizeof(f), hipMemcpyDeviceToHost);
        v = f;
      } else if (s.dtype == ncclFloat64) {
        (void)hipMemcpy(&v, s.dev_ptr, sizeof(v), hipMemcpyDeviceToHost);
      } else if (s.dtype == ncclFloat16) {
        uint16_t h = 0;
        (void)hipMemcpy(&h, s.dev_ptr, sizeof(h), hipMemcpyDeviceToHost);
        // fp16 -> double on host
        int const sign = (h >> 15) & 1, exp = (h >> 10) & 0x1f,
                  man = h & 0x3ff;
        double m = exp == 0 ? man / 1024.0 / 16384.0
                            : (1.0 + man / 1024.0) * std::pow(2.0, exp - 15);
        v = sign ? -m : m;
      } else {
        return ncclInvalidArgument;
      }
    }
    plan->prescale = true;
    plan->prescale_by = v;
    return ncclSuccess;
  }
  switch (op) {
    case ncclSum: return ncclSuccess;
    case ncclProd: plan->op = uccl::RedOp::kProd; return ncclSuccess;
    case ncclMin: plan->op = uccl::RedOp::kMin; return ncclSuccess;
    case ncclMax: plan->op = uccl::RedOp::kMax; return ncclSuccess;
    case ncclAvg:
      if (!is_float_dtype(dt)) return ncclInvalidUsage;
      plan->postscale = true;
      return ncclSuccess;  // postscale_by filled by caller (1/world)
    default: return ncclInvalidArgument;
  }
}

#define COMM(c) (reinterpret_cast<UcclComm*>(c))

}  // namespace

extern "C" {

ncclResult_t ncclGetVersion(int* version) {
  if (version) *version = 22807;  // claims NCCL 2.28-era API
  return ncclSuccess;
}

const char* ncclGetErrorString(ncclResult_t code) {
  switch (code) {
    case ncclSuccess: return "no error";
    case ncclInvalidArgument: return "invalid argument (uccl_amd shim)";
    case ncclInvalidUsage: return "invalid usage (uccl_amd shim)";
    default: return "error (uccl_amd shim)";
  }
}

ncclResult_t ncclGetUniqueId(ncclUniqueId* id) {
  static_assert(sizeof(IdPayload) <= sizeof(ncclUniqueId), "id size");
  auto srv = std::make_unique<ExchangeServer>();
  uint16_t port = 0;
  srv->listen_fd = uccl::net::listen_on(&port);
  int lfd = srv->listen_fd;
  srv->th = std::thread([lfd] { serve(lfd); });
  IdPayload p{};
  p.magic = kIdMagic;
  std::string ip = uccl::net::local_ip();
  strncpy(p.ip, ip.c_str(), sizeof(p.ip) - 1);
  p.port = port;
  memset(id, 0, sizeof(*id));
  memcpy(id, &p, sizeof(p));
  std::lock_guard<std::mutex> g(g_mu);
  g_servers.push_back(std::move(srv));
  return ncclSuccess;
}

ncclResult_t ncclCommInitRank(ncclComm_t* comm, int nranks,
                              ncclUniqueId commId, int rank) {
  IdPayload p{};
  memcpy(&p, &commId, sizeof(p));
  if (p.magic != kIdMagic || nranks < 1 || nranks > 8)
    return ncclInvalidArgument;
  int device = 0;
  (void)hipGetDevice(&device);
  auto* uc = new UcclComm();
  uc->device = device;
  try {
    uc->comm = std::make_unique<Communicator>(rank, nranks, device, 0);
    std::string blob = uc->comm->handle_bytes();
    int fd = uccl::net::connect_to(p.ip, p.port);
    WireBlob h{kBlobMagic, rank, nranks,
               static_cast<uint32_t>(blob.size())};
    uccl::net::send_all(fd, &h, sizeof(h));
    uccl::net::send_all(fd, blob.data(), blob.size());
    uint32_t n = 0;
    if (!uccl::net::recv_all(fd, &n, sizeof(n)) ||
        static_cast<int>(n) != nranks) {
      ::close(fd);
      delete uc;
      return ncclInternalError;
    }
    std::vector<std::string> handles(nranks);
    for (int r = 0; r < nranks; ++r) {
      uint32_t l = 0;
      uccl::net::recv_all(fd, &l, sizeof(l));
      if (l > kMaxBlobLen) {
        ::close(fd);
        return ncclSystemError;
      }
      handles[r].resize(l);
      uccl::net::recv_all(fd, handles[r].data(), l);
    }
    ::close(fd);
    if (nranks > 1) uc->comm->connect(handles);
  } catch (std::exception const& e) {
    UCCL_LOG_ERROR << "ncclCommInitRank failed: " << e.what();
    delete uc;
    return ncclInternalError;
  }
  *comm = reinterpret_cast<ncclComm_t>(uc);
  return ncclSuccess;
}

ncclResult_t ncclCommDestroy(ncclComm_t comm) {
  delete COMM(comm);
  return ncclSuccess;
}

ncclResult_t ncclCommAbort(ncclComm_t comm) { return ncclCommDestroy(comm); }

ncclResult_t ncclCommCount(const ncclComm_t comm, int* count) {
  *count = COMM(comm)->comm->world();
  return ncclSuccess;
}

ncclResult_t ncclCommCuDevice(const ncclComm_t comm, int* dev) {
  *dev = COMM(comm)->device;
  return ncclSuccess;
}

ncclResult_t ncclCommUserRank(const ncclComm_t comm, int* rank) {
  *rank = COMM(comm)->comm->rank();
  return ncclSuccess;
}

ncclResult_t ncclCommGetAsyncError(ncclComm_t, ncclResult_t* asyncError) {
  if (asyncError) *asyncError = ncclSuccess;
  return ncclSuccess;
}

// Group semantics: collectives are independently stream-async; p2p ops are
// deferred and re-issued on per-(peer,dir) streams at GroupEnd (see
// flush_group) so paired send/recv >2MB can't deadlock on slot credits.
ncclResult_t ncclGroupStart(void) {
  ++g_group_depth;
  return ncclSuccess;
}

ncclResult_t ncclGroupEnd(void) {
  if (g_group_depth <= 0) return ncclInvalidUsage;
  if (--g_group_depth == 0 && !g_pending.empty()) return flush_group();
  return ncclSuccess;
}

ncclResult_t ncclAllReduce(const void* sendbuff, void* recvbuff,
                           size_t count, ncclDataType_t datatype,
                           ncclRedOp_t op, ncclComm_t comm,
                           hipStream_t stream) {
  Dtype dt;
  if (to_dtype(datatype, &dt) != ncclSuccess || dt == Dtype::kU8)
    return ncclInvalidArgument;
  OpPlan plan;
  ncclResult_t r = plan_op(COMM(comm), op, dt, &plan);
  if (r != ncclSuccess) return r;
  auto& c = *COMM(comm)->comm;
  if (sendbuff != recvbuff) {
    size_t bytes = count * uccl::dtype_size(dt);
    (void)hipMemcpyAsync(recvbuff, sendbuff, bytes,
                         hipMemcpyDeviceToDevice, stream);
  }
  if (plan.prescale)
    uccl::launch_scale(recvbuff, count, dt, plan.prescale_by, stream);
  c.all_reduce(recvbuff, count, dt, stream, plan.op);
  if (plan.postscale)
    uccl::launch_scale(recvbuff, count, dt, 1.0 / c.world(), stream);
  return ncclSuccess;
}

ncclResult_t ncclBroadcast(const void* sendbuff, void* recvbuff,
                           size_t count, ncclDataType_t datatype, int root,
                           ncclComm_t comm, hipStream_t stream) {
  Dtype dt;
  if (to_dtype(datatype, &dt) != ncclSuccess) return ncclInvalidArgument;
  auto& c = *COMM(comm)->comm;
  if (sendbuff != recvbuff && c.rank() == root) {
    (void)hipMemcpyAsync(recvbuff, sendbuff,
                         count * uccl::dtype_size(dt),
                         hipMemcpyDeviceToDevice, stream);
  }
  c.broadcast(recvbuff, count * uccl::dtype_size(dt), Dtype::kU8, root,
              stream);
  return ncclSuccess;
}

ncclResult_t ncclBcast(void* buff, size_t count, ncclDataType_t datatype,
                       int root, ncclComm_t comm, hipStream_t stream) {
  return ncclBroadcast(buff, buff, count, datatype, root, comm, stream);
}

ncclResult_t ncclAllGather(const void* sendbuff, void* recvbuff,
                           size_t sendcount, ncclDataType_t datatype,
                           ncclComm_t comm, hipStream_t stream) {
  Dtype dt;
  if (to_dtype(datatype, &dt) != ncclSuccess) return ncclInvalidArgument;
  COMM(comm)->comm->all_gather(recvbuff, sendbuff,
                               sendcount * uccl::dtype_size(dt), Dtype::kU8,
                               stream);
  return ncclSuccess;
}

ncclResult_t ncclReduceScatter(const void* sendbuff, void* recvbuff,
                               size_t recvcount, ncclDataType_t datatype,
                               ncclRedOp_t op, ncclComm_t comm,
                               hipStream_t stream) {
  Dtype dt;
  if (to_dtype(datatype, &dt) != ncclSuccess || dt == Dtype::kU8)
    return ncclInvalidArgument;
  OpPlan plan;
  ncclResult_t r = plan_op(COMM(comm), op, dt, &plan);
  if (r != ncclSuccess) return r;
  // premulsum would need to scale the (larger, caller-owned) send buffer;
  // not supported on this entry point.
  if (plan.prescale) return ncclInvalidUsage;
  auto& c = *COMM(comm)->comm;
  c.reduce_scatter(recvbuff, sendbuff, recvcount, dt, stream, plan.op);
  if (plan.postscale)
    uccl::launch_scale(recvbuff, recvcount, dt, 1.0 / c.world(), stream);
  return ncclSuccess;
}

ncclResult_t ncclReduce(const void* sendbuff, void* recvbuff, size_t count,
                        ncclDataType_t datatype, ncclRedOp_t op, int root,
                        ncclComm_t comm, hipStream_t stream) {
  // allreduce everywhere; the root's buffer ends up with the reduction
  // (non-root recvbuffs are also reduced, which the API permits to vary).
  // NCCL allows recvbuff == nullptr on non-roots: reduce into a scratch
  // buffer there (stream-ordered alloc/free).
  if (recvbuff == nullptr) {
    if (COMM(comm)->comm->rank() == root) return ncclInvalidArgument;
    Dtype dt;
    if (to_dtype(datatype, &dt) != ncclSuccess) return ncclInvalidArgument;
    void* tmp = nullptr;
    if (hipMallocAsync(&tmp, count * uccl::dtype_size(dt), stream) !=
        hipSuccess)
      return ncclUnhandledCudaError;
    ncclResult_t r =
        ncclAllReduce(sendbuff, tmp, count, datatype, op, comm, stream);
    (void)hipFreeAsync(tmp, stream);
    return r;
  }
  return ncclAllReduce(sendbuff, recvbuff, count, datatype, op, comm,
                       stream);
}

ncclResult_t ncclSend(const void* sendbuff, size_t count,
                      ncclDataType_t datatype, int peer, ncclComm_t comm,
                      hipStream_t stream) {
  Dtype dt;
  if (to_dtype(datatype, &dt) != ncclSuccess) return ncclInvalidArgument;
  size_t const bytes = count * uccl::dtype_size(dt);
  if (g_group_depth > 0) {
    g_pending.push_back(PendingP2P{COMM(comm), true,
                                   const_cast<void*>(sendbuff), bytes, peer,
                                   stream});
    return ncclSuccess;
  }
  COMM(comm)->comm->send(sendbuff, bytes, peer, stream);
  return ncclSuccess;
}

ncclResult_t ncclRecv(void* recvbuff, size_t count,
                      ncclDataType_t datatype, int peer, ncclComm_t comm,
                      hipStream_t stream) {
  Dtype dt;
  if (to_dtype(datatype, &dt) != ncclSuccess) return ncclInvalidArgument;
  size_t const bytes = count * uccl::dtype_size(dt);
  if (g_group_depth > 0) {
    g_pending.push_back(
        PendingP2P{COMM(comm), false, recvbuff, bytes, peer, stream});
    return ncclSuccess;
  }
  COMM(comm)->comm->recv(recvbuff, bytes, peer, stream);
  return ncclSuccess;
}

// premulsum: result = sum_r (scalar_r * x_r); each rank pre-scales its own
// contribution (plan_op) before the sum.
ncclResult_t ncclRedOpCreatePreMulSum(ncclRedOp_t* op, void* scalar,
                                      ncclDataType_t datatype,
                                      ncclScalarResidence_t residence,
                                      ncclComm_t comm) {
  if (!op || !scalar) return ncclInvalidArgument;
  PreMulScalar s;
  s.dtype = datatype;
  if (residence == ncclScalarDevice) {
    s.device_resident = true;
    s.dev_ptr = scalar;
  } else {
    switch (datatype) {
      case ncclFloat32: s.value = *static_cast<float*>(scalar); break;
      case ncclFloat64: s.value = *static_cast<double*>(scalar); break;
      default: return ncclInvalidArgument;  // host-immediate fp16 unused
    }
  }
  auto* uc = COMM(comm);
  std::lock_guard<std::mutex> g(uc->mu);
  uc->premul.push_back(s);
  *op = static_cast<ncclRedOp_t>(ncclNumOps + uc->premul.size() - 1);
  return ncclSuccess;
}

ncclResult_t ncclRedOpDestroy(ncclRedOp_t op, ncclComm_t comm) {
  (void)comm;
  return op >= ncclNumOps ? ncclSuccess : ncclInvalidArgument;
}

// RCCL extension, also provided by the reference's drop-in
// (lite-collective nccl.cu:2069): symmetric all-to-all, count elements
// per peer.
ncclResult_t ncclAllToAll(const void* sendbuff, void* recvbuff,
                          size_t count, ncclDataType_t datatype,
                          ncclComm_t comm, hipStream_t stream) {
  Dtype dt;
  if (to_dtype(datatype, &dt) != ncclSuccess) return ncclInvalidArgument;
  COMM(comm)->comm->all_to_all(recvbuff, sendbuff,
                               count * uccl::dtype_size(dt), Dtype::kU8,
                               stream);
  return ncclSuccess;
}

ncclResult_t ncclCommFinalize(ncclComm_t comm) {
  // flush outstanding engine work; destruction stays with CommDestroy
  if (comm) COMM(comm)->comm->barrier(nullptr);
  return ncclSuccess;
}

const char* ncclGetLastError(ncclComm_t) {
  return "";  // entry points fail fast with result codes; no deferred log
}

ncclResult_t ncclMemAlloc(void** ptr, size_t size) {
  if (!ptr) return ncclInvalidArgument;
  return hipMalloc(ptr, size) == hipSuccess ? ncclSuccess
                                            : ncclUnhandledCudaError;
}

ncclResult_t ncclMemFree(void* ptr) {
  return hipFree(ptr) == hipSuccess ? ncclSuccess : ncclUnhandledCudaError;
}

}  // extern "C"
