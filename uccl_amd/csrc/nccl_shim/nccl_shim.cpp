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
// Scope: single node (up to 8 ranks over xGMI), SUM reductions, the
// dtypes the engine supports. Everything else returns a clean error.

#include <rccl/rccl.h>

#include <atomic>
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

struct UcclComm {
  std::unique_ptr<Communicator> comm;
  int device = 0;
};

ncclResult_t to_dtype(ncclDataType_t t, Dtype* out) {
  switch (t) {
    case ncclFloat32: *out = Dtype::kF32; return ncclSuccess;
    case ncclFloat16: *out = Dtype::kF16; return ncclSuccess;
    case ncclBfloat16: *out = Dtype::kBF16; return ncclSuccess;
    case ncclInt32: *out = Dtype::kI32; return ncclSuccess;
    case ncclInt8: *out = Dtype::kU8; return ncclSuccess;  // copy ops only
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

// group semantics: ops are independently stream-async; nothing to defer
ncclResult_t ncclGroupStart(void) { return ncclSuccess; }
ncclResult_t ncclGroupEnd(void) { return ncclSuccess; }

ncclResult_t ncclAllReduce(const void* sendbuff, void* recvbuff,
                           size_t count, ncclDataType_t datatype,
                           ncclRedOp_t op, ncclComm_t comm,
                           hipStream_t stream) {
  if (op != ncclSum) return ncclInvalidUsage;
  Dtype dt;
  if (to_dtype(datatype, &dt) != ncclSuccess || dt == Dtype::kU8)
    return ncclInvalidArgument;
  auto& c = *COMM(comm)->comm;
  if (sendbuff != recvbuff) {
    size_t bytes = count * uccl::dtype_size(dt);
    (void)hipMemcpyAsync(recvbuff, sendbuff, bytes,
                         hipMemcpyDeviceToDevice, stream);
  }
  c.all_reduce(recvbuff, count, dt, stream);
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
  if (op != ncclSum) return ncclInvalidUsage;
  Dtype dt;
  if (to_dtype(datatype, &dt) != ncclSuccess || dt == Dtype::kU8)
    return ncclInvalidArgument;
  COMM(comm)->comm->reduce_scatter(recvbuff, sendbuff, recvcount, dt,
                                   stream);
  return ncclSuccess;
}

ncclResult_t ncclReduce(const void* sendbuff, void* recvbuff, size_t count,
                        ncclDataType_t datatype, ncclRedOp_t op, int root,
                        ncclComm_t comm, hipStream_t stream) {
  // allreduce everywhere; the root's buffer ends up with the reduction
  // (non-root recvbuffs are also reduced, which the API permits to vary)
  return ncclAllReduce(sendbuff, recvbuff, count, datatype, op, comm,
                       stream);
}

ncclResult_t ncclSend(const void* sendbuff, size_t count,
                      ncclDataType_t datatype, int peer, ncclComm_t comm,
                      hipStream_t stream) {
  Dtype dt;
  if (to_dtype(datatype, &dt) != ncclSuccess) return ncclInvalidArgument;
  COMM(comm)->comm->send(sendbuff, count * uccl::dtype_size(dt), peer,
                         stream);
  return ncclSuccess;
}

ncclResult_t ncclRecv(void* recvbuff, size_t count,
                      ncclDataType_t datatype, int peer, ncclComm_t comm,
                      hipStream_t stream) {
  Dtype dt;
  if (to_dtype(datatype, &dt) != ncclSuccess) return ncclInvalidArgument;
  COMM(comm)->comm->recv(recvbuff, count * uccl::dtype_size(dt), peer,
                         stream);
  return ncclSuccess;
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
