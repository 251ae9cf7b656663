#include "rccl_plane.h"

#include <dlfcn.h>

#include "../core/log.h"

namespace uccl {
namespace p2p {

namespace {
// stable RCCL/NCCL ABI constants we rely on
constexpr int kNcclSuccess = 0;
constexpr int kNcclInt8 = 0;
}  // namespace

RcclPlane::RcclPlane() {
  lib_ = dlopen("librccl.so.1", RTLD_NOW | RTLD_GLOBAL);
  if (!lib_) lib_ = dlopen("librccl.so", RTLD_NOW | RTLD_GLOBAL);
  if (!lib_) return;
  p_get_unique_id_ = reinterpret_cast<int (*)(void*)>(
      dlsym(lib_, "ncclGetUniqueId"));
  p_comm_init_rank_ = reinterpret_cast<int (*)(void**, int, void*, int)>(
      dlsym(lib_, "ncclCommInitRank"));
  p_comm_destroy_ =
      reinterpret_cast<int (*)(void*)>(dlsym(lib_, "ncclCommDestroy"));
  p_send_ = reinterpret_cast<int (*)(void const*, size_t, int, int, void*,
                                     hipStream_t)>(dlsym(lib_, "ncclSend"));
  p_recv_ = reinterpret_cast<int (*)(void*, size_t, int, int, void*,
                                     hipStream_t)>(dlsym(lib_, "ncclRecv"));
  p_err_str_ = reinterpret_cast<char const* (*)(int)>(
      dlsym(lib_, "ncclGetErrorString"));
  if (!p_get_unique_id_ || !p_comm_init_rank_ || !p_send_ || !p_recv_) {
    dlclose(lib_);
    lib_ = nullptr;
  }
}

RcclPlane& RcclPlane::get() {
  static RcclPlane plane;
  return plane;
}

bool RcclPlane::available() { return get().lib_ != nullptr; }

bool RcclPlane::create_unique_id(UniqueId* out) {
  std::lock_guard<std::mutex> g(mu_);
  return p_get_unique_id_(out->data) == kNcclSuccess;
}

void* RcclPlane::comm_init(int rank2, UniqueId const& id, int device) {
  // collective across the connection's two ranks; a failure (e.g. RCCL's
  // duplicate-device rejection when both ranks share one GPU) happens on
  // both sides, so both fall back consistently
  (void)hipSetDevice(device);
  void* comm = nullptr;
  int const rc = p_comm_init_rank_(&comm, 2,
                                   const_cast<char*>(id.data), rank2);
  if (rc != kNcclSuccess) {
    UCCL_LOG_WARN << "p2p rccl plane: ncclCommInitRank failed: "
                  << (p_err_str_ ? p_err_str_(rc) : "?");
    return nullptr;
  }
  return comm;
}

void RcclPlane::comm_destroy(void* comm) {
  if (comm && p_comm_destroy_) (void)p_comm_destroy_(comm);
}

bool RcclPlane::send(void* comm, void const* dev_ptr, size_t bytes,
                     int peer, hipStream_t stream) {
  int rc;
  {
    // serialize ENQUEUES only; the stream sync happens outside the lock
    // so simultaneous bidirectional traffic cannot deadlock
    std::lock_guard<std::mutex> g(mu_);
    rc = p_send_(dev_ptr, bytes, kNcclInt8, peer, comm, stream);
  }
  if (rc != kNcclSuccess) return false;
  return hipStreamSynchronize(stream) == hipSuccess;
}

bool RcclPlane::recv(void* comm, void* dev_ptr, size_t bytes, int peer,
                     hipStream_t stream) {
  int rc;
  {
    std::lock_guard<std::mutex> g(mu_);
    rc = p_recv_(dev_ptr, bytes, kNcclInt8, peer, comm, stream);
  }
  if (rc != kNcclSuccess) return false;
  return hipStreamSynchronize(stream) == hipSuccess;
}

}  // namespace p2p
}  // namespace uccl
