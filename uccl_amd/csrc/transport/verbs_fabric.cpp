// VerbsFabric — ibverbs RC data plane for the multipath reliable
// transport (fabric.h).
//
// Design (mirrors the reference's verbs fabric,
// /root/reference/collective/rdma/transport.cc:2228-2306 + rdma_io.h):
//   - per flow: `num_paths` RC data QPs (the spray set) + 1 ctrl QP
//   - receiver FIFO rendezvous: recv_msg posts a buffer ->
//     post_recv_window registers it and advertises {msg_id, addr, rkey,
//     cap} to the sender over the ctrl QP; chunking is gated on that
//     advert (tx_ready)
//   - chunks are RDMA_WRITE_WITH_IMM straight into the advertised
//     window at their byte offset; IMM packs {rid:8, csn:24}; a one-off
//     MSG_BEGIN ctrl frame carries {msg_id, msg_bytes} so the receiver
//     can synthesize full ChunkDescs from CQEs
//   - the reliable layer's ACK/SACK frames ride the ctrl QP unchanged
//
// The provider seam (verbs_provider.h) supplies the actual verbs calls:
// the real adapter (verbs_adapter.c, dlopen'd) on RDMA nodes, the mock
// loopback provider in the CPU test tier.

#include <dlfcn.h>

#include <atomic>
#include <chrono>
#include <cstring>
#include <deque>
#include <map>
#include <mutex>
#include <thread>
#include <unordered_map>
#include <vector>

#include "../core/env.h"
#include "../core/log.h"
#include "fabric.h"
#include "verbs_provider.h"

namespace uccl {
namespace transport {

namespace {

constexpr int kCtrlSlots = 64;       // in-flight ctrl sends per flow
constexpr int kCtrlBuf = 768;        // bytes per ctrl slot / recv
constexpr int kDataRecvRing = 512;   // pre-posted recvs per data QP
constexpr int kMaxInflightWr = 256;  // write WRs outstanding per QP

enum VCtrlKind : uint8_t {
  kVcReliable = 0,  // opaque reliable-layer frame (ACK/SACK)
  kVcWindow = 1,    // receiver window advertisement
  kVcBegin = 2,     // sender message begin {msg_id, msg_bytes}
};

struct VWin {
  uint8_t kind;
  uint8_t pad[7];
  uint64_t msg_id;
  uint64_t addr;
  uint64_t cap;
  uint32_t rkey;
  uint32_t pad2;
};

struct VBegin {
  uint8_t kind;
  uint8_t pad[7];
  uint64_t msg_id;
  uint64_t msg_bytes;
};

struct MdQp {
  UvAddr addr;
  uint32_t ctrl_qpn;
  uint32_t nqp;
  uint32_t qpn[64];
};

// IMM layout: {rid:8 | csn:24}. csn is the low 24 bits of the protocol's
// 32-bit chunk sequence number, re-extended receiver-side.
inline uint32_t pack_imm(uint64_t msg_id, uint32_t csn) {
  return (static_cast<uint32_t>(msg_id & 0xff) << 24) | (csn & 0xffffff);
}

inline uint32_t extend24(uint32_t last, uint32_t low) {
  uint32_t cand = (last & 0xff000000u) | low;
  if (cand + (1u << 23) < last) cand += 1u << 24;
  else if (cand > last + (1u << 23) && cand >= (1u << 24)) cand -= 1u << 24;
  return cand;
}

struct CtrlSlot {
  char buf[kCtrlBuf];
  bool busy = false;
};

struct FlowRes {
  std::vector<UvQp*> data;
  UvQp* ctrl = nullptr;
  bool connector = false;
  bool connected = false;

  // ctrl tx slots (registered once, wr_id = slot index | flow tag)
  std::unique_ptr<CtrlSlot[]> tx_slots;
  UvMr* tx_slots_mr = nullptr;
  // ctrl rx ring
  std::unique_ptr<char[]> rx_ring;
  UvMr* rx_ring_mr = nullptr;

  std::vector<int> inflight_wr;  // per data QP

  // sender side: advertised windows + MR cache for message payloads
  std::unordered_map<uint64_t, VWin> wins;
  std::map<std::pair<void const*, size_t>, UvMr*> payload_mrs;

  // receiver side
  std::unordered_map<uint64_t, UvMr*> rx_win_mrs;  // msg_id -> MR
  std::unordered_map<uint8_t, VBegin> begins;      // rid -> {msg_id, bytes}
  uint32_t last_rx_csn = 0;
};

class VerbsFabric final : public Fabric {
 public:
  VerbsFabric(int num_paths, size_t chunk_bytes)
      : np_(num_paths), chunk_bytes_(chunk_bytes) {
    (void)chunk_bytes_;
    std::string so = env_str("UCCL_VERBS_PROVIDER", "libuccl_verbs_ib.so");
    handle_ = dlopen(so.c_str(), RTLD_NOW | RTLD_GLOBAL);
    if (!handle_)
      throw std::runtime_error(std::string("dlopen ") + so + ": " +
                               dlerror());
    auto fn = reinterpret_cast<uccl_verbs_provider_fn>(
        dlsym(handle_, "uccl_verbs_provider"));
    if (!fn) throw std::runtime_error("provider entry symbol missing");
    prov_ = fn();
    if (!prov_ || prov_->open(prov_, env_str("UCCL_VERBS_DEV", "").c_str()))
      throw std::runtime_error("no RDMA device");
    send_cq_ = prov_->create_cq(prov_, 4096);
    recv_cq_ = prov_->create_cq(prov_, 4096);
    if (!send_cq_ || !recv_cq_) throw std::runtime_error("cq alloc failed");
  }

  ~VerbsFabric() override {
    {
      std::lock_guard<std::mutex> g(mu_);
      for (auto& [flow, fr] : flows_) teardown(fr);
      flows_.clear();
    }
    if (send_cq_) prov_->destroy_cq(send_cq_);
    if (recv_cq_) prov_->destroy_cq(recv_cq_);
    if (prov_) prov_->close(prov_);
    if (handle_) dlclose(handle_);
  }

  int num_paths() const override { return np_; }

  std::string create_flow(uint64_t flow, bool connector) override {
    std::lock_guard<std::mutex> g(mu_);
    if (flows_.count(flow))
      throw std::runtime_error(
          "verbs fabric: duplicate flow (in-process self-connect is not "
          "supported on the verbs plane; use the udp fabric)");
    FlowRes fr;
    fr.connector = connector;
    fr.ctrl = prov_->create_qp(prov_, send_cq_, recv_cq_, kCtrlSlots * 2,
                               kCtrlSlots * 2);
    if (!fr.ctrl) throw std::runtime_error("ctrl qp alloc failed");
    for (int i = 0; i < np_; ++i) {
      UvQp* qp = prov_->create_qp(prov_, send_cq_, recv_cq_, kMaxInflightWr,
                                  kDataRecvRing);
      if (!qp) throw std::runtime_error("data qp alloc failed");
      fr.data.push_back(qp);
      fr.inflight_wr.push_back(0);
    }
    fr.tx_slots.reset(new CtrlSlot[kCtrlSlots]);
    fr.tx_slots_mr = prov_->reg_mr(prov_, fr.tx_slots.get(),
                                   sizeof(CtrlSlot) * kCtrlSlots);
    fr.rx_ring.reset(new char[kCtrlBuf * kCtrlSlots * 2]);
    fr.rx_ring_mr =
        prov_->reg_mr(prov_, fr.rx_ring.get(), kCtrlBuf * kCtrlSlots * 2);
    if (!fr.tx_slots_mr || !fr.rx_ring_mr)
      throw std::runtime_error("ctrl mr reg failed");

    MdQp md{};
    prov_->query_addr(prov_, &md.addr);
    md.ctrl_qpn = prov_->qp_num(fr.ctrl);
    md.nqp = static_cast<uint32_t>(np_);
    for (int i = 0; i < np_; ++i) md.qpn[i] = prov_->qp_num(fr.data[i]);

    qp_flow_[md.ctrl_qpn] = {flow, -1};
    for (int i = 0; i < np_; ++i) qp_flow_[md.qpn[i]] = {flow, i};
    flows_.emplace(flow, std::move(fr));
    return std::string(reinterpret_cast<char const*>(&md), sizeof(md));
  }

  int install_peer(uint64_t flow, std::string const& md) override {
    if (md.size() != sizeof(MdQp)) return 0;
    MdQp peer{};
    memcpy(&peer, md.data(), sizeof(peer));
    if (peer.nqp < 1 || peer.nqp > 64) return 0;
    std::lock_guard<std::mutex> g(mu_);
    auto it = flows_.find(flow);
    if (it == flows_.end()) return 0;
    FlowRes& fr = it->second;
    int const paths = std::min(np_, static_cast<int>(peer.nqp));
    if (prov_->connect_qp(fr.ctrl, &peer.addr, peer.ctrl_qpn)) return 0;
    for (int i = 0; i < paths; ++i)
      if (prov_->connect_qp(fr.data[i], &peer.addr, peer.qpn[i])) return 0;
    // pre-post the ctrl rx ring and the zero-byte data recvs that
    // RDMA_WRITE_WITH_IMM consumes
    for (int i = 0; i < kCtrlSlots * 2; ++i)
      prov_->post_recv(fr.ctrl, 1000 + i, fr.rx_ring.get() + i * kCtrlBuf,
                       kCtrlBuf, fr.rx_ring_mr);
    for (int p = 0; p < paths; ++p)
      for (int i = 0; i < kDataRecvRing; ++i)
        prov_->post_recv(fr.data[p], 0, nullptr, 0, fr.rx_ring_mr);
    fr.connected = true;
    return paths;
  }

  void remove_peer(uint64_t flow) override {
    std::lock_guard<std::mutex> g(mu_);
    auto it = flows_.find(flow);
    if (it == flows_.end()) return;
    teardown(it->second);
    flows_.erase(it);
  }

  bool tx_ready(uint64_t flow, uint64_t msg_id) override {
    std::lock_guard<std::mutex> g(mu_);
    auto it = flows_.find(flow);
    return it != flows_.end() && it->second.wins.count(msg_id) != 0;
  }

  bool post_chunk(uint64_t flow, int path, ChunkDesc const& d,
                  void const* payload) override {
    std::lock_guard<std::mutex> g(mu_);
    auto it = flows_.find(flow);
    if (it == flows_.end() || !it->second.connected) return false;
    FlowRes& fr = it->second;
    auto wit = fr.wins.find(d.msg_id);
    if (wit == fr.wins.end()) return false;  // rendezvous not complete
    if (d.off + d.len > wit->second.cap) return false;
    path %= static_cast<int>(fr.data.size());
    if (fr.inflight_wr[path] >= kMaxInflightWr - 1) {
      drain_send_cq_locked();
      if (fr.inflight_wr[path] >= kMaxInflightWr - 1) return false;
    }
    // first chunk of the message (and any retransmit of it): tell the
    // receiver {msg_id, msg_bytes} so it can synthesize ChunkDescs
    if (d.off == 0) {
      VBegin b{};
      b.kind = kVcBegin;
      b.msg_id = d.msg_id;
      b.msg_bytes = d.msg_bytes;
      send_ctrl_locked(fr, flow, &b, sizeof(b));
    }
    void const* base = static_cast<char const*>(payload) - d.off;
    UvMr* mr = payload_mr_locked(fr, base, d.msg_bytes);
    if (!mr) return false;
    if (prov_->post_write_imm(fr.data[path], /*wr_id=*/flow, payload, d.len,
                              mr, wit->second.addr + d.off,
                              wit->second.rkey,
                              pack_imm(d.msg_id, d.csn)) != 0)
      return false;
    ++fr.inflight_wr[path];
    return true;
  }

  void post_ctrl(uint64_t flow, int path, void const* frame,
                 size_t len) override {
    (void)path;  // all ctrl traffic rides the one ctrl QP
    std::lock_guard<std::mutex> g(mu_);
    auto it = flows_.find(flow);
    if (it == flows_.end() || !it->second.connected) return;
    char buf[kCtrlBuf];
    if (len + 1 > sizeof(buf)) return;
    buf[0] = kVcReliable;
    memcpy(buf + 1, frame, len);
    send_ctrl_locked(it->second, flow, buf, len + 1);
  }

  void post_recv_window(uint64_t flow, uint64_t msg_id, void* buf,
                        size_t cap) override {
    std::lock_guard<std::mutex> g(mu_);
    auto it = flows_.find(flow);
    if (it == flows_.end() || !it->second.connected) return;
    FlowRes& fr = it->second;
    UvMr* mr = prov_->reg_mr(prov_, buf, cap ? cap : 1);
    if (!mr) return;
    // windows post in order; anything 64+ behind is long complete
    if (fr.rx_win_mrs.size() > 64) {
      auto oldest = fr.rx_win_mrs.begin();
      for (auto i = fr.rx_win_mrs.begin(); i != fr.rx_win_mrs.end(); ++i)
        if (i->first < oldest->first) oldest = i;
      prov_->dereg_mr(oldest->second);
      fr.rx_win_mrs.erase(oldest);
    }
    fr.rx_win_mrs[msg_id] = mr;
    VWin w{};
    w.kind = kVcWindow;
    w.msg_id = msg_id;
    w.addr = reinterpret_cast<uint64_t>(buf);
    w.cap = cap;
    w.rkey = prov_->mr_rkey(mr);
    send_ctrl_locked(fr, flow, &w, sizeof(w));
  }

  int poll(std::function<void(FabricEvent const&)> const& cb,
           int timeout_ms) override {
    using clock = std::chrono::steady_clock;
    auto const deadline =
        clock::now() + std::chrono::milliseconds(timeout_ms);
    int delivered = 0;
    woken_.store(false, std::memory_order_relaxed);
    while (true) {
      // Collect events under the lock, deliver after releasing it
      // (the reliable layer takes its own mutex inside cb).
      std::vector<FabricEvent> evs;
      std::vector<std::vector<char>> bufs;
      {
        std::lock_guard<std::mutex> g(mu_);
        drain_send_cq_locked();
        UvCompletion wc[64];
        int n = prov_->poll_cq(recv_cq_, 64, wc);
        for (int i = 0; i < n; ++i) handle_rx_locked(wc[i], evs, bufs);
      }
      for (auto& ev : evs) cb(ev);
      delivered += static_cast<int>(evs.size());
      if (delivered || woken_.load(std::memory_order_relaxed) ||
          clock::now() >= deadline)
        break;
      std::this_thread::sleep_for(std::chrono::microseconds(50));
    }
    return delivered;
  }

  void wake() override { woken_.store(true, std::memory_order_relaxed); }

 private:
  void teardown(FlowRes& fr) {
    for (auto& [k, mr] : fr.payload_mrs) prov_->dereg_mr(mr);
    for (auto& [k, mr] : fr.rx_win_mrs) prov_->dereg_mr(mr);
    if (fr.tx_slots_mr) prov_->dereg_mr(fr.tx_slots_mr);
    if (fr.rx_ring_mr) prov_->dereg_mr(fr.rx_ring_mr);
    for (UvQp* q : fr.data) {
      qp_flow_.erase(prov_->qp_num(q));
      prov_->destroy_qp(q);
    }
    if (fr.ctrl) {
      qp_flow_.erase(prov_->qp_num(fr.ctrl));
      prov_->destroy_qp(fr.ctrl);
    }
  }

  UvMr* payload_mr_locked(FlowRes& fr, void const* base, size_t bytes) {
    auto key = std::make_pair(base, bytes);
    auto it = fr.payload_mrs.find(key);
    if (it != fr.payload_mrs.end()) return it->second;
    UvMr* mr =
        prov_->reg_mr(prov_, const_cast<void*>(base), bytes ? bytes : 1);
    if (mr) fr.payload_mrs[key] = mr;
    if (fr.payload_mrs.size() > 256) {
      // simple cap: drop an arbitrary cold entry (registration is cheap
      // relative to the 64KB+ chunks this plane is sized for)
      auto victim = fr.payload_mrs.begin();
      if (victim->second != mr) {
        prov_->dereg_mr(victim->second);
        fr.payload_mrs.erase(victim);
      }
    }
    return mr;
  }

  void send_ctrl_locked(FlowRes& fr, uint64_t flow, void const* frame,
                        size_t len) {
    int slot = -1;
    for (int attempt = 0; attempt < 10000; ++attempt) {
      for (int i = 0; i < kCtrlSlots; ++i)
        if (!fr.tx_slots[i].busy) {
          slot = i;
          break;
        }
      if (slot >= 0) break;
      drain_send_cq_locked();
    }
    if (slot < 0) return;  // pathological backlog: drop (acks are loss-ok)
    memcpy(fr.tx_slots[slot].buf, frame, len);
    fr.tx_slots[slot].busy = true;
    // wr_id encodes {flow, slot} so the send CQE can release the slot
    uint64_t const wr_id = (flow << 8) | static_cast<uint64_t>(slot);
    if (prov_->post_send(fr.ctrl, wr_id, fr.tx_slots[slot].buf,
                         static_cast<uint32_t>(len), fr.tx_slots_mr) != 0)
      fr.tx_slots[slot].busy = false;
  }

  void drain_send_cq_locked() {
    UvCompletion wc[64];
    int n;
    while ((n = prov_->poll_cq(send_cq_, 64, wc)) > 0) {
      for (int i = 0; i < n; ++i) {
        auto qit = qp_flow_.find(wc[i].qp_num);
        if (qit == qp_flow_.end()) continue;
        auto fit = flows_.find(qit->second.first);
        if (fit == flows_.end()) continue;
        FlowRes& fr = fit->second;
        if (wc[i].opcode == UV_WC_SEND) {
          int const slot = static_cast<int>(wc[i].wr_id & 0xff);
          if (slot >= 0 && slot < kCtrlSlots) fr.tx_slots[slot].busy = false;
        } else if (wc[i].opcode == UV_WC_WRITE) {
          int const path = qit->second.second;
          if (path >= 0 && fr.inflight_wr[path] > 0)
            --fr.inflight_wr[path];
        }
      }
    }
  }

  void handle_rx_locked(UvCompletion const& wc,
                        std::vector<FabricEvent>& evs,
                        std::vector<std::vector<char>>& bufs) {
    auto qit = qp_flow_.find(wc.qp_num);
    if (qit == qp_flow_.end()) return;
    uint64_t const flow = qit->second.first;
    int const path = qit->second.second;
    auto fit = flows_.find(flow);
    if (fit == flows_.end()) return;
    FlowRes& fr = fit->second;

    if (path < 0) {
      // ctrl QP: locate the ring buffer by wr_id and repost it
      int const idx = static_cast<int>(wc.wr_id - 1000);
      if (idx < 0 || idx >= kCtrlSlots * 2) return;
      char* buf = fr.rx_ring.get() + idx * kCtrlBuf;
      uint8_t const kind = wc.byte_len ? static_cast<uint8_t>(buf[0]) : 0xff;
      if (kind == kVcWindow && wc.byte_len >= sizeof(VWin)) {
        VWin w{};
        memcpy(&w, buf, sizeof(w));
        fr.wins[w.msg_id] = w;
      } else if (kind == kVcBegin && wc.byte_len >= sizeof(VBegin)) {
        VBegin b{};
        memcpy(&b, buf, sizeof(b));
        fr.begins[static_cast<uint8_t>(b.msg_id & 0xff)] = b;
      } else if (kind == kVcReliable && wc.byte_len >= 1) {
        bufs.emplace_back(buf + 1, buf + wc.byte_len);
        FabricEvent ev{};
        ev.kind = FabricEvent::kCtrl;
        ev.flow = flow;
        ev.path = 0;
        ev.ctrl = bufs.back().data();
        ev.ctrl_len = wc.byte_len - 1;
        evs.push_back(ev);
      }
      prov_->post_recv(fr.ctrl, wc.wr_id, buf, kCtrlBuf, fr.rx_ring_mr);
      return;
    }

    if (wc.opcode != UV_WC_RECV_IMM) return;
    // data QP: the write is already PLACED; synthesize the chunk event
    prov_->post_recv(fr.data[path], 0, nullptr, 0, fr.rx_ring_mr);
    uint8_t const rid = static_cast<uint8_t>(wc.imm >> 24);
    uint32_t const csn = extend24(fr.last_rx_csn, wc.imm & 0xffffff);
    fr.last_rx_csn = csn;
    auto bit = fr.begins.find(rid);
    if (bit == fr.begins.end()) return;  // BEGIN lost: RTO will re-drive
    FabricEvent ev{};
    ev.kind = FabricEvent::kChunk;
    ev.flow = flow;
    ev.path = path;
    ev.desc = ChunkDesc{flow,
                        bit->second.msg_id,
                        bit->second.msg_bytes,
                        /*off=*/0,  // placed by the NIC; offset consumed
                        wc.byte_len,
                        csn,
                        /*ts_ns=*/0};
    ev.payload = nullptr;  // no inline payload: zero-copy placement
    evs.push_back(ev);
  }

  int np_;
  size_t chunk_bytes_;
  void* handle_ = nullptr;
  UvProvider* prov_ = nullptr;
  UvCq* send_cq_ = nullptr;
  UvCq* recv_cq_ = nullptr;
  std::mutex mu_;
  std::unordered_map<uint64_t, FlowRes> flows_;
  std::unordered_map<uint32_t, std::pair<uint64_t, int>> qp_flow_;
  std::atomic<bool> woken_{false};
};

}  // namespace

std::unique_ptr<Fabric> make_verbs_fabric(int num_paths,
                                          size_t chunk_bytes) {
  return std::make_unique<VerbsFabric>(num_paths, chunk_bytes);
}

}  // namespace transport
}  // namespace uccl
