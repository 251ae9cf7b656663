// RCCL-as-transport data plane for the p2p engine.
//
// Parity role: the reference's p2p NCCL backend
// (/root/reference/p2p/nccl/nccl_endpoint.h:58, dlopen'd via nccl_dl.cc)
// — the portability fallback that makes the p2p API usable wherever the
// stock CCL works. Here the stock CCL is RCCL: librccl.so.1 is dlopen'd
// at first use, each p2p connection owns a 2-rank ncclComm bootstrapped
// by exchanging the uniqueId over the connection's existing TCP channel,
// and GPU payloads move as ncclSend/ncclRecv directly between device
// buffers (no host staging).
//
// Threading: RCCL comms are not thread-safe, so enqueues are serialized
// per comm; sends ride a dedicated tx stream and recvs a rx stream so
// simultaneous bidirectional traffic cannot deadlock on stream order.
#pragma once

#include <hip/hip_runtime.h>

#include <cstddef>
#include <mutex>
#include <string>

namespace uccl {
namespace p2p {

class RcclPlane {
 public:
  // 128-byte opaque id, matching ncclUniqueId
  struct UniqueId {
    char data[128];
  };

  // process-wide availability (librccl.so.1 resolvable)
  static bool available();
  static RcclPlane& get();

  bool create_unique_id(UniqueId* out);
  // collective over the 2 ranks of one p2p connection; returns nullptr on
  // failure (e.g. both ranks on one device: RCCL rejects duplicates, the
  // caller falls back to another plane)
  void* comm_init(int rank2, UniqueId const& id, int device);
  void comm_destroy(void* comm);

  bool send(void* comm, void const* dev_ptr, size_t bytes, int peer,
            hipStream_t stream);
  bool recv(void* comm, void* dev_ptr, size_t bytes, int peer,
            hipStream_t stream);

 private:
  RcclPlane();
  void* lib_ = nullptr;
  std::mutex mu_;  // serialize enqueues (RCCL comms are not thread-safe)
  // resolved symbols
  int (*p_get_unique_id_)(void*) = nullptr;
  int (*p_comm_init_rank_)(void**, int, void*, int) = nullptr;
  int (*p_comm_destroy_)(void*) = nullptr;
  int (*p_send_)(void const*, size_t, int, int, void*, hipStream_t) =
      nullptr;
  int (*p_recv_)(void*, size_t, int, int, void*, hipStream_t) = nullptr;
  char const* (*p_err_str_)(int) = nullptr;
};

}  // namespace p2p
}  // namespace uccl
