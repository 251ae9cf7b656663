// Mock verbs provider (verbs_provider.h): an in-process software
// loopback with real RC-verbs semantics, used by the CPU test tier to
// exercise the whole VerbsFabric logic without an RDMA NIC:
//   - QPs are wired by qp_num through a process-global registry
//   - post_send delivers into the peer's posted recv buffer (RNR-queues
//     when none is posted, like an RC QP retrying)
//   - post_write_imm PLACES the payload at raddr after validating
//     {rkey, range} against the target's registered MRs (the NIC's
//     bounds check), consumes one peer recv WR, and completes with
//     UV_WC_RECV_IMM {imm, byte_len}
//   - UCCL_MOCK_VERBS_DROP_PCT silently drops write deliveries (the
//     sender still sees success, like a lossy UC plane) so the
//     SACK/RTO machinery above gets exercised end-to-end
//
// Built into libuccl_verbs_mock.so; tests select it with
// UCCL_TP_FABRIC=verbs UCCL_VERBS_PROVIDER=<path>.

#include <cstdlib>
#include <cstring>
#include <deque>
#include <map>
#include <mutex>
#include <unordered_map>
#include <vector>

#include "verbs_provider.h"

namespace {

struct MockCq {
  std::deque<UvCompletion> q;
};

struct MockMr {
  void* addr;
  size_t len;
  uint32_t key;
};

struct RecvWr {
  uint64_t wr_id;
  void* addr;
  uint32_t len;
};

struct PendingSend {  // RNR-queued SEND waiting for a peer recv post
  uint64_t wr_id;
  std::vector<char> data;
  uint32_t src_qpn;
};

struct MockQp {
  uint32_t qpn;
  uint32_t peer_qpn = 0;
  bool connected = false;
  MockCq* send_cq;
  MockCq* recv_cq;
  std::deque<RecvWr> recvs;
  std::deque<PendingSend> rnr_queue;
};

struct Registry {
  std::mutex mu;
  std::unordered_map<uint32_t, MockQp*> qps;
  std::unordered_map<uint32_t, MockMr*> mrs;  // by rkey
  uint32_t next_qpn = 100;
  uint32_t next_key = 1000;
  int drop_pct = -1;
  uint64_t drop_seq = 0;

  static Registry& get() {
    static Registry r;
    return r;
  }

  bool should_drop() {
    // re-read each call: tests flip the knob between endpoints within
    // one process (the .so and this registry live process-long)
    char const* e = getenv("UCCL_MOCK_VERBS_DROP_PCT");
    drop_pct = e ? atoi(e) : 0;
    if (drop_pct <= 0) return false;
    // deterministic hash sequence (reproducible like reliable.cpp's)
    uint64_t h = ++drop_seq;
    h ^= h >> 33;
    h *= 0xff51afd7ed558ccdULL;
    h ^= h >> 33;
    return static_cast<int>(h % 100) < drop_pct;
  }
};

void push_wc(MockCq* cq, uint64_t wr_id, int opcode, uint32_t imm,
             uint32_t byte_len, uint32_t qpn) {
  cq->q.push_back(UvCompletion{wr_id, 0, opcode, imm, byte_len, qpn});
}

// deliver a SEND payload into a posted recv on `dst` (caller holds mu)
bool deliver_send(MockQp* dst, char const* data, size_t len,
                  uint64_t /*src_wr*/) {
  if (dst->recvs.empty()) return false;
  RecvWr rw = dst->recvs.front();
  dst->recvs.pop_front();
  size_t const n = len < rw.len ? len : rw.len;
  if (n) memcpy(rw.addr, data, n);
  push_wc(dst->recv_cq, rw.wr_id, UV_WC_RECV, 0,
          static_cast<uint32_t>(len), dst->qpn);
  return true;
}

int m_open(UvProvider*, char const*) { return 0; }
void m_close(UvProvider*) {}
int m_query_addr(UvProvider*, UvAddr* out) {
  memset(out, 0, sizeof(*out));
  out->port = 1;
  return 0;
}

UvMr* m_reg_mr(UvProvider*, void* addr, size_t len) {
  auto& r = Registry::get();
  std::lock_guard<std::mutex> g(r.mu);
  auto* mr = new MockMr{addr, len, r.next_key++};
  r.mrs[mr->key] = mr;
  return reinterpret_cast<UvMr*>(mr);
}

uint32_t m_mr_lkey(UvMr* mr) { return reinterpret_cast<MockMr*>(mr)->key; }
uint32_t m_mr_rkey(UvMr* mr) { return reinterpret_cast<MockMr*>(mr)->key; }

void m_dereg_mr(UvMr* mr) {
  auto& r = Registry::get();
  std::lock_guard<std::mutex> g(r.mu);
  auto* m = reinterpret_cast<MockMr*>(mr);
  r.mrs.erase(m->key);
  delete m;
}

UvCq* m_create_cq(UvProvider*, int) {
  return reinterpret_cast<UvCq*>(new MockCq());
}

void m_destroy_cq(UvCq* cq) { delete reinterpret_cast<MockCq*>(cq); }

int m_poll_cq(UvCq* ucq, int max, UvCompletion* out) {
  auto& r = Registry::get();
  std::lock_guard<std::mutex> g(r.mu);
  auto* cq = reinterpret_cast<MockCq*>(ucq);
  int n = 0;
  while (n < max && !cq->q.empty()) {
    out[n++] = cq->q.front();
    cq->q.pop_front();
  }
  return n;
}

UvQp* m_create_qp(UvProvider*, UvCq* scq, UvCq* rcq, int, int) {
  auto& r = Registry::get();
  std::lock_guard<std::mutex> g(r.mu);
  auto* qp = new MockQp();
  qp->qpn = r.next_qpn++;
  qp->send_cq = reinterpret_cast<MockCq*>(scq);
  qp->recv_cq = reinterpret_cast<MockCq*>(rcq);
  r.qps[qp->qpn] = qp;
  return reinterpret_cast<UvQp*>(qp);
}

uint32_t m_qp_num(UvQp* qp) { return reinterpret_cast<MockQp*>(qp)->qpn; }

int m_connect_qp(UvQp* uqp, UvAddr const*, uint32_t peer_qpn) {
  auto& r = Registry::get();
  std::lock_guard<std::mutex> g(r.mu);
  auto* qp = reinterpret_cast<MockQp*>(uqp);
  qp->peer_qpn = peer_qpn;
  qp->connected = true;
  return 0;
}

void m_destroy_qp(UvQp* uqp) {
  auto& r = Registry::get();
  std::lock_guard<std::mutex> g(r.mu);
  auto* qp = reinterpret_cast<MockQp*>(uqp);
  r.qps.erase(qp->qpn);
  delete qp;
}

int m_post_recv(UvQp* uqp, uint64_t wr_id, void* addr, uint32_t len,
                UvMr*) {
  auto& r = Registry::get();
  std::lock_guard<std::mutex> g(r.mu);
  auto* qp = reinterpret_cast<MockQp*>(uqp);
  qp->recvs.push_back(RecvWr{wr_id, addr, len});
  // drain any RNR-queued sends now that a recv exists
  while (!qp->rnr_queue.empty() && !qp->recvs.empty()) {
    PendingSend ps = std::move(qp->rnr_queue.front());
    qp->rnr_queue.pop_front();
    deliver_send(qp, ps.data.data(), ps.data.size(), ps.wr_id);
    auto sit = r.qps.find(ps.src_qpn);
    if (sit != r.qps.end())
      push_wc(sit->second->send_cq, ps.wr_id, UV_WC_SEND, 0, 0,
              ps.src_qpn);
  }
  return 0;
}

int m_post_send(UvQp* uqp, uint64_t wr_id, void const* addr, uint32_t len,
                UvMr*) {
  auto& r = Registry::get();
  std::lock_guard<std::mutex> g(r.mu);
  auto* qp = reinterpret_cast<MockQp*>(uqp);
  if (!qp->connected) return -1;
  auto pit = r.qps.find(qp->peer_qpn);
  if (pit == r.qps.end()) return -1;
  MockQp* dst = pit->second;
  if (deliver_send(dst, static_cast<char const*>(addr), len, wr_id)) {
    push_wc(qp->send_cq, wr_id, UV_WC_SEND, 0, 0, qp->qpn);
  } else {
    // RNR: hold until the peer posts a recv (RC retry semantics)
    PendingSend ps;
    ps.wr_id = wr_id;
    ps.data.assign(static_cast<char const*>(addr),
                   static_cast<char const*>(addr) + len);
    ps.src_qpn = qp->qpn;
    dst->rnr_queue.push_back(std::move(ps));
  }
  return 0;
}

int m_post_write_imm(UvQp* uqp, uint64_t wr_id, void const* laddr,
                     uint32_t len, UvMr*, uint64_t raddr, uint32_t rkey,
                     uint32_t imm) {
  auto& r = Registry::get();
  std::lock_guard<std::mutex> g(r.mu);
  auto* qp = reinterpret_cast<MockQp*>(uqp);
  if (!qp->connected) return -1;
  auto pit = r.qps.find(qp->peer_qpn);
  if (pit == r.qps.end()) return -1;
  MockQp* dst = pit->second;
  // sender-side completion always fires (like UC with a lossy wire when
  // drops are injected below)
  push_wc(qp->send_cq, wr_id, UV_WC_WRITE, 0, 0, qp->qpn);
  if (r.should_drop()) return 0;
  // the "NIC" bounds check: rkey must name an MR containing the range
  auto mit = r.mrs.find(rkey);
  if (mit == r.mrs.end()) return 0;  // bad rkey: dropped (QP would error)
  MockMr* mr = mit->second;
  auto const base = reinterpret_cast<uint64_t>(mr->addr);
  if (raddr < base || raddr + len > base + mr->len) return 0;
  if (len) memcpy(reinterpret_cast<void*>(raddr), laddr, len);
  if (dst->recvs.empty()) return 0;  // no recv WR: IMM dropped (RNR-ish)
  RecvWr rw = dst->recvs.front();
  dst->recvs.pop_front();
  push_wc(dst->recv_cq, rw.wr_id, UV_WC_RECV_IMM, imm, len, dst->qpn);
  return 0;
}

}  // namespace

extern "C" UvProvider* uccl_verbs_provider(void) {
  auto* p = new UvProvider();
  p->open = m_open;
  p->close = m_close;
  p->query_addr = m_query_addr;
  p->reg_mr = m_reg_mr;
  p->mr_lkey = m_mr_lkey;
  p->mr_rkey = m_mr_rkey;
  p->dereg_mr = m_dereg_mr;
  p->create_cq = m_create_cq;
  p->destroy_cq = m_destroy_cq;
  p->poll_cq = m_poll_cq;
  p->create_qp = m_create_qp;
  p->qp_num = m_qp_num;
  p->connect_qp = m_connect_qp;
  p->destroy_qp = m_destroy_qp;
  p->post_recv = m_post_recv;
  p->post_send = m_post_send;
  p->post_write_imm = m_post_write_imm;
  return p;
}
