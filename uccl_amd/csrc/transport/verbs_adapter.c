/* Real-ibverbs provider for the verbs fabric (verbs_provider.h).
 *
 * Compiled against the system <infiniband/verbs.h> into
 * libuccl_verbs_ib.so ONLY when rdma-core is installed (_build.py probes
 * for the header); the dev/CI image has no RDMA stack, so there this
 * file is build-checked by inspection and the mock provider covers the
 * fabric logic. On an RDMA-capable node this is the data plane of
 * BASELINE config 5 (fp8 AllReduce over RoCE multipath).
 *
 * Semantics map 1:1 onto the reference's fabric usage
 * (/root/reference/collective/rdma/rdma_io.h:128-713, transport.cc:2297
 * ibv_post_send(RDMA_WRITE_WITH_IMM)): RC QPs, GID-routed RoCEv2
 * address handles, MRs with LOCAL_WRITE|REMOTE_WRITE, CQ polling.
 */

#include <errno.h>
#include <infiniband/verbs.h>
#include <stdlib.h>
#include <string.h>

#include "verbs_provider.h"

struct UvImpl {
  struct ibv_context* ctx;
  struct ibv_pd* pd;
  uint8_t port;
  uint8_t gid_index;
  struct ibv_port_attr port_attr;
  union ibv_gid gid;
};

static int uv_open(UvProvider* p, char const* dev_hint) {
  int num = 0;
  struct ibv_device** list = ibv_get_device_list(&num);
  if (!list || num == 0) return -ENODEV;
  struct ibv_context* ctx = NULL;
  for (int i = 0; i < num; ++i) {
    if (dev_hint && *dev_hint &&
        strcmp(ibv_get_device_name(list[i]), dev_hint) != 0)
      continue;
    ctx = ibv_open_device(list[i]);
    if (ctx) break;
  }
  ibv_free_device_list(list);
  if (!ctx) return -ENODEV;
  struct UvImpl* im = calloc(1, sizeof(*im));
  im->ctx = ctx;
  im->port = 1;
  if (ibv_query_port(ctx, im->port, &im->port_attr) != 0) {
    ibv_close_device(ctx);
    free(im);
    return -EIO;
  }
  /* RoCEv2 default: gid index 3 is the conventional v2 IPv4 slot, but
   * fall back to 0 when the table is short (IB link / exotic setups). */
  im->gid_index = (im->port_attr.link_layer == IBV_LINK_LAYER_ETHERNET &&
                   im->port_attr.gid_tbl_len > 3)
                      ? 3
                      : 0;
  {
    char const* gi = getenv("UCCL_VERBS_GID_INDEX");
    if (gi && *gi) im->gid_index = (uint8_t)atoi(gi);
  }
  if (ibv_query_gid(ctx, im->port, im->gid_index, &im->gid) != 0)
    memset(&im->gid, 0, sizeof(im->gid));
  im->pd = ibv_alloc_pd(ctx);
  if (!im->pd) {
    ibv_close_device(ctx);
    free(im);
    return -ENOMEM;
  }
  p->impl = im;
  return 0;
}

static void uv_close(UvProvider* p) {
  struct UvImpl* im = p->impl;
  if (!im) return;
  if (im->pd) ibv_dealloc_pd(im->pd);
  if (im->ctx) ibv_close_device(im->ctx);
  free(im);
  p->impl = NULL;
}

static int uv_query_addr(UvProvider* p, UvAddr* out) {
  struct UvImpl* im = p->impl;
  memcpy(out->gid, im->gid.raw, 16);
  out->lid = im->port_attr.lid;
  out->port = im->port;
  out->gid_index = im->gid_index;
  return 0;
}

static UvMr* uv_reg_mr(UvProvider* p, void* addr, size_t len) {
  struct UvImpl* im = p->impl;
  return (UvMr*)ibv_reg_mr(im->pd, addr, len,
                           IBV_ACCESS_LOCAL_WRITE | IBV_ACCESS_REMOTE_WRITE);
}

static uint32_t uv_mr_lkey(UvMr* mr) { return ((struct ibv_mr*)mr)->lkey; }
static uint32_t uv_mr_rkey(UvMr* mr) { return ((struct ibv_mr*)mr)->rkey; }
static void uv_dereg_mr(UvMr* mr) { ibv_dereg_mr((struct ibv_mr*)mr); }

static UvCq* uv_create_cq(UvProvider* p, int depth) {
  struct UvImpl* im = p->impl;
  return (UvCq*)ibv_create_cq(im->ctx, depth, NULL, NULL, 0);
}

static void uv_destroy_cq(UvCq* cq) { ibv_destroy_cq((struct ibv_cq*)cq); }

static int uv_poll_cq(UvCq* cq, int max, UvCompletion* out) {
  struct ibv_wc wc[64];
  if (max > 64) max = 64;
  int n = ibv_poll_cq((struct ibv_cq*)cq, max, wc);
  for (int i = 0; i < n; ++i) {
    out[i].wr_id = wc[i].wr_id;
    out[i].status = (int32_t)wc[i].status;
    out[i].byte_len = wc[i].byte_len;
    out[i].qp_num = wc[i].qp_num;
    out[i].imm = 0;
    switch (wc[i].opcode) {
      case IBV_WC_SEND:
        out[i].opcode = UV_WC_SEND;
        break;
      case IBV_WC_RDMA_WRITE:
        out[i].opcode = UV_WC_WRITE;
        break;
      case IBV_WC_RECV_RDMA_WITH_IMM:
        out[i].opcode = UV_WC_RECV_IMM;
        out[i].imm = ntohl(wc[i].imm_data);
        break;
      default:
        out[i].opcode = UV_WC_RECV;
        break;
    }
  }
  return n;
}

static UvQp* uv_create_qp(UvProvider* p, UvCq* send_cq, UvCq* recv_cq,
                          int max_send_wr, int max_recv_wr) {
  struct UvImpl* im = p->impl;
  struct ibv_qp_init_attr attr;
  memset(&attr, 0, sizeof(attr));
  attr.send_cq = (struct ibv_cq*)send_cq;
  attr.recv_cq = (struct ibv_cq*)recv_cq;
  attr.qp_type = IBV_QPT_RC;
  attr.cap.max_send_wr = max_send_wr;
  attr.cap.max_recv_wr = max_recv_wr;
  attr.cap.max_send_sge = 1;
  attr.cap.max_recv_sge = 1;
  return (UvQp*)ibv_create_qp(im->pd, &attr);
}

static uint32_t uv_qp_num(UvQp* qp) { return ((struct ibv_qp*)qp)->qp_num; }

static int uv_connect_qp(UvQp* uqp, UvAddr const* peer, uint32_t peer_qpn) {
  struct ibv_qp* qp = (struct ibv_qp*)uqp;
  struct ibv_qp_attr a;

  memset(&a, 0, sizeof(a));
  a.qp_state = IBV_QPS_INIT;
  a.pkey_index = 0;
  a.port_num = peer->port ? peer->port : 1;
  a.qp_access_flags = IBV_ACCESS_LOCAL_WRITE | IBV_ACCESS_REMOTE_WRITE;
  if (ibv_modify_qp(qp, &a,
                    IBV_QP_STATE | IBV_QP_PKEY_INDEX | IBV_QP_PORT |
                        IBV_QP_ACCESS_FLAGS) != 0)
    return -1;

  memset(&a, 0, sizeof(a));
  a.qp_state = IBV_QPS_RTR;
  a.path_mtu = IBV_MTU_4096;
  a.dest_qp_num = peer_qpn;
  a.rq_psn = 0;
  a.max_dest_rd_atomic = 1;
  a.min_rnr_timer = 12;
  a.ah_attr.port_num = peer->port ? peer->port : 1;
  a.ah_attr.dlid = peer->lid;
  a.ah_attr.sl = 0;
  a.ah_attr.src_path_bits = 0;
  /* RoCE: GID-routed global address handle */
  int any_gid = 0;
  for (int i = 0; i < 16; ++i) any_gid |= peer->gid[i];
  if (any_gid) {
    a.ah_attr.is_global = 1;
    memcpy(a.ah_attr.grh.dgid.raw, peer->gid, 16);
    a.ah_attr.grh.sgid_index = peer->gid_index;
    a.ah_attr.grh.hop_limit = 64;
  }
  if (ibv_modify_qp(qp, &a,
                    IBV_QP_STATE | IBV_QP_AV | IBV_QP_PATH_MTU |
                        IBV_QP_DEST_QPN | IBV_QP_RQ_PSN |
                        IBV_QP_MAX_DEST_RD_ATOMIC |
                        IBV_QP_MIN_RNR_TIMER) != 0)
    return -1;

  memset(&a, 0, sizeof(a));
  a.qp_state = IBV_QPS_RTS;
  a.timeout = 14;
  a.retry_cnt = 7;
  a.rnr_retry = 7;
  a.sq_psn = 0;
  a.max_rd_atomic = 1;
  if (ibv_modify_qp(qp, &a,
                    IBV_QP_STATE | IBV_QP_TIMEOUT | IBV_QP_RETRY_CNT |
                        IBV_QP_RNR_RETRY | IBV_QP_SQ_PSN |
                        IBV_QP_MAX_QP_RD_ATOMIC) != 0)
    return -1;
  return 0;
}

static void uv_destroy_qp(UvQp* qp) { ibv_destroy_qp((struct ibv_qp*)qp); }

static int uv_post_recv(UvQp* qp, uint64_t wr_id, void* addr, uint32_t len,
                        UvMr* mr) {
  struct ibv_sge sge = {(uintptr_t)addr, len, uv_mr_lkey(mr)};
  struct ibv_recv_wr wr, *bad = NULL;
  memset(&wr, 0, sizeof(wr));
  wr.wr_id = wr_id;
  wr.sg_list = &sge;
  wr.num_sge = 1;
  return ibv_post_recv((struct ibv_qp*)qp, &wr, &bad);
}

static int uv_post_send(UvQp* qp, uint64_t wr_id, void const* addr,
                        uint32_t len, UvMr* mr) {
  struct ibv_sge sge = {(uintptr_t)addr, len, uv_mr_lkey(mr)};
  struct ibv_send_wr wr, *bad = NULL;
  memset(&wr, 0, sizeof(wr));
  wr.wr_id = wr_id;
  wr.sg_list = &sge;
  wr.num_sge = 1;
  wr.opcode = IBV_WR_SEND;
  wr.send_flags = IBV_SEND_SIGNALED;
  return ibv_post_send((struct ibv_qp*)qp, &wr, &bad);
}

static int uv_post_write_imm(UvQp* qp, uint64_t wr_id, void const* laddr,
                             uint32_t len, UvMr* lmr, uint64_t raddr,
                             uint32_t rkey, uint32_t imm) {
  struct ibv_sge sge = {(uintptr_t)laddr, len, uv_mr_lkey(lmr)};
  struct ibv_send_wr wr, *bad = NULL;
  memset(&wr, 0, sizeof(wr));
  wr.wr_id = wr_id;
  wr.sg_list = &sge;
  wr.num_sge = 1;
  wr.opcode = IBV_WR_RDMA_WRITE_WITH_IMM;
  wr.send_flags = IBV_SEND_SIGNALED;
  wr.imm_data = htonl(imm);
  wr.wr.rdma.remote_addr = raddr;
  wr.wr.rdma.rkey = rkey;
  return ibv_post_send((struct ibv_qp*)qp, &wr, &bad);
}

UvProvider* uccl_verbs_provider(void) {
  UvProvider* p = calloc(1, sizeof(*p));
  p->open = uv_open;
  p->close = uv_close;
  p->query_addr = uv_query_addr;
  p->reg_mr = uv_reg_mr;
  p->mr_lkey = uv_mr_lkey;
  p->mr_rkey = uv_mr_rkey;
  p->dereg_mr = uv_dereg_mr;
  p->create_cq = uv_create_cq;
  p->destroy_cq = uv_destroy_cq;
  p->poll_cq = uv_poll_cq;
  p->create_qp = uv_create_qp;
  p->qp_num = uv_qp_num;
  p->connect_qp = uv_connect_qp;
  p->destroy_qp = uv_destroy_qp;
  p->post_recv = uv_post_recv;
  p->post_send = uv_post_send;
  p->post_write_imm = uv_post_write_imm;
  return p;
}
