// Narrow RDMA provider ABI for the verbs fabric.
//
// Why an indirection instead of #include <infiniband/verbs.h>: this image
// (and the MI355X GPU boxes) ship no rdma-core headers or library, and
// hand-vendoring rdma-core's struct layouts (ibv_qp/ibv_context_ops and
// the inline fast-path wrappers) cannot be validated here — an ABI guess
// that is silently wrong is worse than an explicit seam. So the fabric
// talks to this C function table with 1:1 verbs semantics, and two
// providers implement it:
//
//   verbs_adapter.c     — the REAL thing: ibv_open_device / ibv_reg_mr /
//                         ibv_create_qp(RC) / ibv_modify_qp INIT->RTR->RTS
//                         / ibv_post_send(IBV_WR_RDMA_WRITE_WITH_IMM) /
//                         ibv_poll_cq. Compiled against the system
//                         <infiniband/verbs.h> into libuccl_verbs_ib.so
//                         when rdma-core is present (build-gated, exactly
//                         like the reference dlopens libibverbs —
//                         /root/reference/p2p/rdma/ibverbs_dl.cc).
//   mock provider       — in-process software loopback with the same
//                         semantics (placement writes, CQEs, QP wiring,
//                         configurable drop rate) used by the CPU test
//                         tier to exercise the whole verbs fabric logic.
//
// The fabric dlopens UCCL_VERBS_PROVIDER (default libuccl_verbs_ib.so)
// and resolves `uccl_verbs_provider`.
#pragma once

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct UvMr UvMr;    // opaque: wraps ibv_mr
typedef struct UvCq UvCq;    // opaque: wraps ibv_cq
typedef struct UvQp UvQp;    // opaque: wraps ibv_qp (RC)

enum UvOpcode {
  UV_WC_SEND = 0,       // send WR completed (tx side)
  UV_WC_RECV = 1,       // SEND arrived into a posted recv buffer
  UV_WC_RECV_IMM = 2,   // RDMA_WRITE_WITH_IMM arrived (consumes a recv WR)
  UV_WC_WRITE = 3,      // RDMA write WR completed (tx side)
};

typedef struct UvCompletion {
  uint64_t wr_id;
  int32_t status;    // 0 = success (ibv_wc_status)
  int32_t opcode;    // UvOpcode
  uint32_t imm;      // valid for UV_WC_RECV_IMM (network byte order undone)
  uint32_t byte_len; // recv-side completions
  uint32_t qp_num;   // local QP the completion belongs to
} UvCompletion;

typedef struct UvAddr {
  uint8_t gid[16];   // RoCE GID (or zero on IB with lid routing)
  uint16_t lid;
  uint8_t port;
  uint8_t gid_index;
} UvAddr;

typedef struct UvProvider UvProvider;
struct UvProvider {
  void* impl;

  // device/PD lifetime (first active device/port unless hint names one)
  int (*open)(UvProvider* p, char const* dev_hint);
  void (*close)(UvProvider* p);
  int (*query_addr)(UvProvider* p, UvAddr* out);

  // memory registration (local read + remote write access)
  UvMr* (*reg_mr)(UvProvider* p, void* addr, size_t len);
  uint32_t (*mr_lkey)(UvMr* mr);
  uint32_t (*mr_rkey)(UvMr* mr);
  void (*dereg_mr)(UvMr* mr);

  // completion queues
  UvCq* (*create_cq)(UvProvider* p, int depth);
  void (*destroy_cq)(UvCq* cq);
  int (*poll_cq)(UvCq* cq, int max, UvCompletion* out);

  // RC queue pairs
  UvQp* (*create_qp)(UvProvider* p, UvCq* send_cq, UvCq* recv_cq,
                     int max_send_wr, int max_recv_wr);
  uint32_t (*qp_num)(UvQp* qp);
  int (*connect_qp)(UvQp* qp, UvAddr const* peer, uint32_t peer_qpn);
  void (*destroy_qp)(UvQp* qp);

  // work requests (wr_id is returned in the matching completion)
  int (*post_recv)(UvQp* qp, uint64_t wr_id, void* addr, uint32_t len,
                   UvMr* mr);
  int (*post_send)(UvQp* qp, uint64_t wr_id, void const* addr, uint32_t len,
                   UvMr* mr);
  int (*post_write_imm)(UvQp* qp, uint64_t wr_id, void const* laddr,
                        uint32_t len, UvMr* lmr, uint64_t raddr,
                        uint32_t rkey, uint32_t imm);
};

// every provider .so exports exactly this
UvProvider* uccl_verbs_provider(void);
typedef UvProvider* (*uccl_verbs_provider_fn)(void);

#ifdef __cplusplus
}
#endif
