#include "endpoint.h"
#include "rccl_plane.h"

#include <hip/hip_runtime.h>
#include <unistd.h>

#include <chrono>
#include <stdexcept>
#include <cstring>

#include "../core/env.h"
#include "../core/log.h"
#include "../core/trace.h"
#include "../core/net.h"
#include "../transport/reliable.h"

namespace uccl {
namespace p2p {

namespace {

constexpr uint64_t kAdvertMagic = 0x75636361647665ULL;  // "uccladve"
constexpr size_t kStagingBytes = 8ull << 20;

enum Op : uint64_t {
  kHello = 1,
  kSendData = 2,     // a=bytes                               + payload
  kSendIpc = 3,      // a=bytes, d=token                      + {handle,off}
  kWriteData = 4,    // a=mr, b=off, c=bytes, d=token         + payload
  kWriteIpc = 5,     // a=mr, b=off, c=bytes, d=token         + {handle,off}
  kWriteAck = 6,     // d=token
  kReadReq = 7,      // a=mr, b=off, c=bytes, d=token
  kReadResp = 8,     // a=bytes, d=token                      + payload
  kIpcDone = 9,      // d=token
  // RCCL-as-transport plane (UCCL_P2P_TRANSPORT=rccl): headers ride the
  // TCP channel for matching/ordering; payloads move device-to-device
  // as ncclSend/ncclRecv on the connection's 2-rank comm
  kSendRccl = 10,    // a=bytes
  kWriteRccl = 11,   // a=mr, b=off, c=bytes, d=token
  kReadReqRccl = 12, // a=mr, b=off, c=bytes, d=token
};

struct MsgHdr {
  uint64_t op;
  uint64_t a, b, c, d;
};

struct IpcBlob {
  hipIpcMemHandle_t handle;
  uint64_t offset;
  int device;
  int src_pid;
};

struct Meta {
  char ip[48];
  uint16_t port;
  int gpu;
  int pid;
  char host[64];
  uint16_t tp_len;   // multipath rendezvous metadata (0 = tcp-only peer)
  char tp_md[64];
};

bool is_gpu(int device) { return device >= 0; }

bool p2p_rccl() {
  static bool v = uccl::env_str("UCCL_P2P_TRANSPORT", "tcp") ==
                  std::string("rccl");
  return v;
}

bool p2p_multipath() {
  static bool v = uccl::env_str("UCCL_P2P_TRANSPORT", "tcp") ==
                  std::string("multipath");
  return v;
}

constexpr size_t kMsgChunk = 8ull << 20;  // payload message granule

}  // namespace

struct Endpoint::RxItem {
  // Inline host data, an IPC descriptor to copy from, or an RCCL-plane
  // payload to be received directly into the consumer's buffer.
  std::vector<char> data;
  bool ipc = false;
  bool rccl = false;
  IpcBlob blob{};
  uint64_t token = 0;
  size_t bytes = 0;
};

struct Endpoint::Conn {
  uint64_t id = 0;
  int fd = -1;
  uccl::transport::TransportEndpoint* tp = nullptr;  // multipath mode
  uint64_t flow = 0;
  std::string peer_ip;
  int peer_gpu = -1;
  int peer_pid = -1;
  bool same_host = false;
  std::thread rx;
  std::mutex tx_mu;
  std::atomic<bool> alive{true};

  // in-order two-sided queue
  std::deque<std::shared_ptr<RxItem>> rxq;
  std::mutex rx_mu;
  std::condition_variable rx_cv;

  // per-direction FIFO tickets for ASYNC two-sided ops: the worker pool
  // may start tasks out of submission order, but sends must hit the wire
  // (and recvs must pop rxq) in the order the caller issued them
  std::mutex ord_mu;
  std::condition_variable ord_cv;
  uint64_t tx_ticket = 0, tx_serving = 0;
  uint64_t rx_ticket = 0, rx_serving = 0;

  // token -> completion latch (acks, read responses)
  std::mutex tok_mu;
  std::condition_variable tok_cv;
  std::unordered_map<uint64_t, std::shared_ptr<RxItem>> completed;
  std::atomic<uint64_t> next_token{1};

  // IPC handle cache: src (pid, base-handle bytes) -> mapped ptr
  std::unordered_map<std::string, void*> ipc_cache;
  std::mutex ipc_mu;

  // RCCL-as-transport plane state (UCCL_P2P_TRANSPORT=rccl)
  void* rccl_comm = nullptr;
  int rccl_rank = -1;  // 0 = acceptor, 1 = connector; peer = 1 - rank
  hipStream_t rccl_tx = nullptr;  // sends and recvs ride separate
  hipStream_t rccl_rx = nullptr;  // streams: no bidirectional deadlock

  // one logical message = header + chunked payload; over the multipath
  // plane each piece is a discrete reliable message in the same order
  void send_msg(MsgHdr const& h, void const* payload = nullptr,
                size_t payload_bytes = 0) {
    std::lock_guard<std::mutex> g(tx_mu);
    if (flow) {
      tp->send_msg(flow, &h, sizeof(h));
      for (size_t off = 0; off < payload_bytes; off += kMsgChunk) {
        size_t const n = std::min(kMsgChunk, payload_bytes - off);
        tp->send_msg(flow, static_cast<char const*>(payload) + off, n);
      }
    } else {
      net::send_all(fd, &h, sizeof(h));
      if (payload_bytes) net::send_all(fd, payload, payload_bytes);
    }
  }

  bool recv_hdr(MsgHdr* h) {
    if (flow) {
      try {
        tp->recv_msg(flow, h, sizeof(*h));
        return true;
      } catch (std::exception const&) {
        return false;
      }
    }
    return net::recv_all(fd, h, sizeof(*h));
  }

  bool recv_payload(void* buf, size_t bytes) {
    if (flow) {
      try {
        for (size_t off = 0; off < bytes; off += kMsgChunk) {
          size_t const n = std::min(kMsgChunk, bytes - off);
          tp->recv_msg(flow, static_cast<char*>(buf) + off, n);
        }
        return true;
      } catch (std::exception const&) {
        return false;
      }
    }
    return net::recv_all(fd, buf, bytes);
  }
};

Endpoint::Endpoint(int gpu, int num_workers) : gpu_(gpu) {
  char host[256] = {0};
  gethostname(host, sizeof(host) - 1);
  host_id_ = std::string(host) + ":" + net::local_ip();
  listen_fd_ = net::listen_on(&port_);
  listener_ = std::thread([this] {
    while (!stop_) {
      int fd = ::accept(listen_fd_, nullptr, nullptr);
      if (fd < 0) {
        if (stop_) break;
        continue;
      }
      int one = 1;
      setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
      auto c = std::make_shared<Conn>();
      c->fd = fd;
      c->id = next_conn_++;
      // passive side: expect hello, reply hello
      MsgHdr h{};
      if (!net::recv_all(fd, &h, sizeof(h)) || h.op != kHello) {
        ::close(fd);
        continue;
      }
      std::vector<char> hostbuf(h.b);
      net::recv_all(fd, hostbuf.data(), hostbuf.size());
      c->peer_gpu = static_cast<int>(h.a);
      c->peer_pid = static_cast<int>(h.c);
      std::string peer_host(hostbuf.begin(), hostbuf.end());
      c->same_host = (peer_host == host_id_);
      MsgHdr r{kHello, static_cast<uint64_t>(gpu_), host_id_.size(),
               static_cast<uint64_t>(getpid()), 0};
      net::send_all(fd, &r, sizeof(r));
      net::send_all(fd, host_id_.data(), host_id_.size());
      if (tp_ && h.d) {
        // connector opens a flow tagged with its hello nonce
        std::unique_lock<std::mutex> lk(tp_mu_);
        bool ok = tp_cv_.wait_for(lk, std::chrono::seconds(30), [&] {
          return tp_flows_.count(h.d) || stop_.load();
        });
        if (!ok || stop_) {
          ::close(fd);
          continue;
        }
        c->tp = tp_.get();
        c->flow = tp_flows_[h.d];
        tp_flows_.erase(h.d);
      }
      if (p2p_rccl()) setup_rccl(*c, /*acceptor=*/true);
      {
        std::lock_guard<std::mutex> g(conn_mu_);
        conns_[c->id] = c;
        accepted_.push_back(c->id);
      }
      c->rx = std::thread([this, c] { rx_loop(c); });
      accept_cv_.notify_all();
    }
  });
  if (p2p_multipath()) {
    tp_ = std::make_unique<uccl::transport::TransportEndpoint>(
        static_cast<int>(env_int("UCCL_P2P_PATHS", 8)),
        static_cast<size_t>(env_int("UCCL_P2P_TP_CHUNK", 16384)));
    tp_acceptor_ = std::thread([this] {
      try {
        while (!stop_) {
          uint64_t tag = 0;
          uint64_t flow = tp_->accept(&tag);
          {
            std::lock_guard<std::mutex> g(tp_mu_);
            tp_flows_[tag] = flow;
          }
          tp_cv_.notify_all();
        }
      } catch (std::exception const&) {
      }
    });
  }
  for (int i = 0; i < std::max(1, num_workers); ++i)
    workers_.emplace_back([this] { worker_loop(); });
}

hipStream_t Endpoint::copy_stream() {
  std::lock_guard<std::mutex> g(copy_mu_);
  if (!copy_stream_) {
    hipStream_t s = nullptr;
    UCCL_CHECK_HIP(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
    copy_stream_ = s;
  }
  return static_cast<hipStream_t>(copy_stream_);
}

Endpoint::~Endpoint() {
  {
    std::lock_guard<std::mutex> g(task_mu_);  // lost-wakeup guard
    stop_ = true;
  }
  if (tp_) tp_->shutdown();
  ::shutdown(listen_fd_, SHUT_RDWR);
  ::close(listen_fd_);
  task_cv_.notify_all();
  // Kill connections BEFORE joining workers: a worker blocked in
  // do_recv/do_write on a cv would otherwise never wake and the join
  // would hang forever (advisor r1).
  {
    std::lock_guard<std::mutex> g(conn_mu_);
    for (auto& [id, c] : conns_) {
      {
        std::lock_guard<std::mutex> g1(c->rx_mu);
        std::lock_guard<std::mutex> g2(c->tok_mu);
        std::lock_guard<std::mutex> g3(c->ord_mu);
        c->alive = false;
      }
      ::shutdown(c->fd, SHUT_RDWR);
      if (c->flow && tp_) tp_->close_flow(c->flow);
      c->rx_cv.notify_all();
      c->tok_cv.notify_all();
      c->ord_cv.notify_all();
    }
  }
  for (auto& w : workers_)
    if (w.joinable()) w.join();
  if (listener_.joinable()) listener_.join();
  if (tp_acceptor_.joinable()) tp_acceptor_.join();
  {
    std::lock_guard<std::mutex> g(conn_mu_);
    for (auto& [id, c] : conns_) {
      if (c->rx.joinable()) c->rx.join();
      if (c->rccl_comm) {
        RcclPlane::get().comm_destroy(c->rccl_comm);
        c->rccl_comm = nullptr;
      }
      if (c->rccl_tx) (void)hipStreamDestroy(c->rccl_tx);
      if (c->rccl_rx) (void)hipStreamDestroy(c->rccl_rx);
      ::close(c->fd);
      for (auto& [k, p] : c->ipc_cache) (void)hipIpcCloseMemHandle(p);
    }
    conns_.clear();
  }
  if (staging_) (void)hipHostFree(staging_);
  if (copy_stream_)
    (void)hipStreamDestroy(static_cast<hipStream_t>(copy_stream_));
}

std::string Endpoint::metadata() const {
  Meta m{};
  std::string ip = net::local_ip();
  strncpy(m.ip, ip.c_str(), sizeof(m.ip) - 1);
  m.port = port_;
  m.gpu = gpu_;
  m.pid = static_cast<int>(getpid());
  strncpy(m.host, host_id_.c_str(), sizeof(m.host) - 1);
  if (tp_) {
    std::string md = tp_->metadata();
    m.tp_len = static_cast<uint16_t>(md.size());
    memcpy(m.tp_md, md.data(), std::min(md.size(), sizeof(m.tp_md)));
  }
  return std::string(reinterpret_cast<char*>(&m), sizeof(m));
}

// RCCL-as-transport bootstrap over the fresh TCP connection: exchange
// availability flags, ship the uniqueId acceptor->connector, then the
// collective 2-rank comm init (reference parity: p2p/nccl backend,
// nccl_endpoint.h:58). Any failure leaves rccl_comm null and the
// connection on the default plane.
void Endpoint::setup_rccl(Conn& c, bool acceptor) {
  uint8_t mine = (gpu_ >= 0 && RcclPlane::available()) ? 1 : 0;
  uint8_t theirs = 0;
  net::send_all(c.fd, &mine, 1);
  if (!net::recv_all(c.fd, &theirs, 1)) return;
  if (!mine || !theirs) {
    UCCL_LOG_WARN << "p2p rccl plane unavailable; using default plane";
    return;
  }
  RcclPlane::UniqueId id{};
  auto& plane = RcclPlane::get();
  if (acceptor) {
    if (!plane.create_unique_id(&id)) return;
    net::send_all(c.fd, id.data, sizeof(id.data));
  } else {
    if (!net::recv_all(c.fd, id.data, sizeof(id.data))) return;
  }
  c.rccl_rank = acceptor ? 0 : 1;
  c.rccl_comm = plane.comm_init(c.rccl_rank, id, gpu_);
  if (!c.rccl_comm) {
    c.rccl_rank = -1;
    return;
  }
  UCCL_CHECK_HIP(hipStreamCreateWithFlags(&c.rccl_tx,
                                          hipStreamNonBlocking));
  UCCL_CHECK_HIP(hipStreamCreateWithFlags(&c.rccl_rx,
                                          hipStreamNonBlocking));
  UCCL_LOG_INFO << "p2p conn " << c.id << " using the RCCL data plane";
}

uint64_t Endpoint::connect(const std::string& remote_metadata) {
  UCCL_CHECK(remote_metadata.size() == sizeof(Meta)) << "bad metadata blob";
  Meta m{};
  memcpy(&m, remote_metadata.data(), sizeof(m));
  std::string ip = m.ip;
  if (std::string(m.host) == host_id_) ip = "127.0.0.1";
  int fd = net::connect_to(ip, m.port);
  auto c = std::make_shared<Conn>();
  c->fd = fd;
  c->id = next_conn_++;
  c->peer_ip = ip;
  // nonce identifies this connection's multipath flow on the peer side
  uint64_t const nonce =
      tp_ ? ((static_cast<uint64_t>(getpid()) << 32) | c->id) : 0;
  MsgHdr h{kHello, static_cast<uint64_t>(gpu_), host_id_.size(),
           static_cast<uint64_t>(getpid()), nonce};
  net::send_all(fd, &h, sizeof(h));
  net::send_all(fd, host_id_.data(), host_id_.size());
  MsgHdr r{};
  UCCL_CHECK(net::recv_all(fd, &r, sizeof(r)) && r.op == kHello)
      << "hello handshake failed";
  std::vector<char> hostbuf(r.b);
  net::recv_all(fd, hostbuf.data(), hostbuf.size());
  c->peer_gpu = static_cast<int>(r.a);
  c->peer_pid = static_cast<int>(r.c);
  c->same_host = (std::string(hostbuf.begin(), hostbuf.end()) == host_id_);
  if (tp_ && m.tp_len) {
    c->tp = tp_.get();
    c->flow = tp_->connect(std::string(m.tp_md, m.tp_len), nonce);
  }
  if (p2p_rccl()) setup_rccl(*c, /*acceptor=*/false);
  {
    std::lock_guard<std::mutex> g(conn_mu_);
    conns_[c->id] = c;
  }
  c->rx = std::thread([this, c] { rx_loop(c); });
  return c->id;
}

uint64_t Endpoint::accept() {
  std::unique_lock<std::mutex> lk(conn_mu_);
  accept_cv_.wait(lk, [this] { return !accepted_.empty() || stop_; });
  UCCL_CHECK(!accepted_.empty()) << "endpoint shut down";
  uint64_t id = accepted_.front();
  accepted_.pop_front();
  return id;
}

std::shared_ptr<Endpoint::Conn> Endpoint::conn(uint64_t id) {
  std::lock_guard<std::mutex> g(conn_mu_);
  auto it = conns_.find(id);
  UCCL_CHECK(it != conns_.end()) << "unknown conn " << id;
  return it->second;
}

uint64_t Endpoint::reg(void* ptr, size_t bytes, int device) {
  MR mr{next_mr_++, ptr, bytes, device};
  std::lock_guard<std::mutex> g(mr_mu_);
  mrs_[mr.id] = mr;
  return mr.id;
}

void Endpoint::dereg(uint64_t mr_id) {
  std::lock_guard<std::mutex> g(mr_mu_);
  mrs_.erase(mr_id);
}

// ---------------------------------------------------------------------------
// rx loop: demultiplex inbound messages
// ---------------------------------------------------------------------------

void Endpoint::rx_loop(std::shared_ptr<Conn> c) {
  if (gpu_ >= 0) (void)hipSetDevice(gpu_);
  try {
    rx_loop_body(c);
  } catch (std::exception const& e) {
    // a malformed frame or failed check kills THIS connection, not the
    // process: mark it dead so blocked ops fail (pending-ops contract)
    UCCL_LOG_WARN << "p2p rx loop failed: " << e.what();
    {
      std::lock_guard<std::mutex> g1(c->rx_mu);
      std::lock_guard<std::mutex> g2(c->tok_mu);
      std::lock_guard<std::mutex> g3(c->ord_mu);
      c->alive = false;
    }
    c->rx_cv.notify_all();
    c->tok_cv.notify_all();
    c->ord_cv.notify_all();
  }
}

void Endpoint::rx_loop_body(std::shared_ptr<Conn> c) {
  while (c->alive && !stop_) {
    MsgHdr h{};
    if (!c->recv_hdr(&h)) break;
    switch (h.op) {
      case kSendData: {
        auto item = std::make_shared<RxItem>();
        item->bytes = h.a;
        item->data.resize(h.a);
        c->recv_payload(item->data.data(), h.a);
        {
          std::lock_guard<std::mutex> g(c->rx_mu);
          c->rxq.push_back(item);
        }
        c->rx_cv.notify_all();
        break;
      }
      case kSendIpc: {
        auto item = std::make_shared<RxItem>();
        item->bytes = h.a;
        item->ipc = true;
        item->token = h.d;
        c->recv_payload(&item->blob, sizeof(IpcBlob));
        {
          std::lock_guard<std::mutex> g(c->rx_mu);
          c->rxq.push_back(item);
        }
        c->rx_cv.notify_all();
        break;
      }
      case kWriteData: {
        MR mr;
        {
          std::lock_guard<std::mutex> g(mr_mu_);
          auto it = mrs_.find(h.a);
          UCCL_CHECK(it != mrs_.end()) << "write to unknown mr " << h.a;
          mr = it->second;
        }
        UCCL_CHECK(h.b + h.c <= mr.bytes) << "write overflows mr";
        char* dst = static_cast<char*>(mr.ptr) + h.b;
        if (!is_gpu(mr.device)) {
          c->recv_payload(dst, h.c);
        } else {
          std::vector<char> tmp(h.c);
          c->recv_payload(tmp.data(), h.c);
          hipStream_t cs = copy_stream();
          UCCL_CHECK_HIP(hipMemcpyAsync(dst, tmp.data(), h.c,
                                        hipMemcpyHostToDevice, cs));
          UCCL_CHECK_HIP(hipStreamSynchronize(cs));
        }
        if (h.d) c->send_msg(MsgHdr{kWriteAck, 0, 0, 0, h.d});
        break;
      }
      case kWriteIpc: {
        IpcBlob blob{};
        c->recv_payload(&blob, sizeof(blob));
        MR mr;
        {
          std::lock_guard<std::mutex> g(mr_mu_);
          auto it = mrs_.find(h.a);
          UCCL_CHECK(it != mrs_.end()) << "ipc write to unknown mr " << h.a;
          mr = it->second;
        }
        UCCL_CHECK(h.b + h.c <= mr.bytes) << "ipc write overflows mr";
        void* src_base = open_ipc(*c, &blob, blob.device);
        hipStream_t cs = copy_stream();
        UCCL_CHECK_HIP(hipMemcpyAsync(
            static_cast<char*>(mr.ptr) + h.b,
            static_cast<char*>(src_base) + blob.offset, h.c,
            hipMemcpyDeviceToDevice, cs));
        UCCL_CHECK_HIP(hipStreamSynchronize(cs));
        if (h.d) c->send_msg(MsgHdr{kIpcDone, 0, 0, 0, h.d});
        break;
      }
      case kReadReq: {
        MR mr;
        {
          std::lock_guard<std::mutex> g(mr_mu_);
          auto it = mrs_.find(h.a);
          UCCL_CHECK(it != mrs_.end()) << "read of unknown mr " << h.a;
          mr = it->second;
        }
        UCCL_CHECK(h.b + h.c <= mr.bytes) << "read overflows mr";
        char const* src = static_cast<char const*>(mr.ptr) + h.b;
        MsgHdr resp{kReadResp, h.c, 0, 0, h.d};
        if (!is_gpu(mr.device)) {
          c->send_msg(resp, src, h.c);
        } else {
          std::vector<char> tmp(h.c);
          hipStream_t cs = copy_stream();
          UCCL_CHECK_HIP(hipMemcpyAsync(tmp.data(), src, h.c,
                                        hipMemcpyDeviceToHost, cs));
          UCCL_CHECK_HIP(hipStreamSynchronize(cs));
          c->send_msg(resp, tmp.data(), h.c);
        }
        break;
      }
      case kReadResp: {
        auto item = std::make_shared<RxItem>();
        item->bytes = h.a;
        item->data.resize(h.a);
        c->recv_payload(item->data.data(), h.a);
        {
          std::lock_guard<std::mutex> g(c->tok_mu);
          c->completed[h.d] = item;
        }
        c->tok_cv.notify_all();
        break;
      }
      case kSendRccl: {
        auto item = std::make_shared<RxItem>();
        item->bytes = h.a;
        item->rccl = true;
        {
          std::lock_guard<std::mutex> g(c->rx_mu);
          c->rxq.push_back(item);
        }
        c->rx_cv.notify_all();
        break;
      }
      case kWriteRccl: {
        MR mr;
        {
          std::lock_guard<std::mutex> g(mr_mu_);
          auto it = mrs_.find(h.a);
          UCCL_CHECK(it != mrs_.end()) << "rccl write to unknown mr";
          mr = it->second;
        }
        UCCL_CHECK(h.b + h.c <= mr.bytes) << "rccl write overflows mr";
        UCCL_CHECK(c->rccl_comm) << "rccl frame on non-rccl conn";
        char* dst = static_cast<char*>(mr.ptr) + h.b;
        if (is_gpu(mr.device)) {
          UCCL_CHECK(RcclPlane::get().recv(c->rccl_comm, dst, h.c,
                                           1 - c->rccl_rank, c->rccl_rx))
              << "rccl recv failed";
        } else {
          // host MR: land in a temp device buffer, then DtoH
          void* tmp = nullptr;
          UCCL_CHECK_HIP(hipMalloc(&tmp, h.c));
          bool ok = RcclPlane::get().recv(c->rccl_comm, tmp, h.c,
                                          1 - c->rccl_rank, c->rccl_rx);
          if (ok)
            UCCL_CHECK_HIP(hipMemcpy(dst, tmp, h.c,
                                     hipMemcpyDeviceToHost));
          (void)hipFree(tmp);
          UCCL_CHECK(ok) << "rccl recv failed";
        }
        if (h.d) c->send_msg(MsgHdr{kWriteAck, 0, 0, 0, h.d});
        break;
      }
      case kReadReqRccl: {
        MR mr;
        {
          std::lock_guard<std::mutex> g(mr_mu_);
          auto it = mrs_.find(h.a);
          UCCL_CHECK(it != mrs_.end()) << "rccl read of unknown mr";
          mr = it->second;
        }
        UCCL_CHECK(h.b + h.c <= mr.bytes) << "rccl read overflows mr";
        UCCL_CHECK(c->rccl_comm) << "rccl frame on non-rccl conn";
        char* src = static_cast<char*>(mr.ptr) + h.b;
        if (is_gpu(mr.device)) {
          UCCL_CHECK(RcclPlane::get().send(c->rccl_comm, src, h.c,
                                           1 - c->rccl_rank, c->rccl_tx))
              << "rccl send failed";
        } else {
          void* tmp = nullptr;
          UCCL_CHECK_HIP(hipMalloc(&tmp, h.c));
          UCCL_CHECK_HIP(hipMemcpy(tmp, src, h.c,
                                   hipMemcpyHostToDevice));
          bool ok = RcclPlane::get().send(c->rccl_comm, tmp, h.c,
                                          1 - c->rccl_rank, c->rccl_tx);
          (void)hipFree(tmp);
          UCCL_CHECK(ok) << "rccl send failed";
        }
        break;
      }
      case kWriteAck:
      case kIpcDone: {
        {
          std::lock_guard<std::mutex> g(c->tok_mu);
          c->completed[h.d] = std::make_shared<RxItem>();
        }
        c->tok_cv.notify_all();
        break;
      }
      default:
        UCCL_LOG_ERROR << "unknown p2p op " << h.op;
        {
          std::lock_guard<std::mutex> g(c->ord_mu);
          c->alive = false;
        }
        c->ord_cv.notify_all();
        return;
    }
  }
}

void* Endpoint::open_ipc(Conn& c, const void* blob_bytes, int src_device) {
  auto const* blob = static_cast<IpcBlob const*>(blob_bytes);
  std::string key(reinterpret_cast<char const*>(&blob->handle),
                  sizeof(blob->handle));
  key += std::to_string(blob->src_pid);
  std::lock_guard<std::mutex> g(c.ipc_mu);
  auto it = c.ipc_cache.find(key);
  if (it != c.ipc_cache.end()) return it->second;
  void* p = nullptr;
  UCCL_CHECK_HIP(hipIpcOpenMemHandle(&p, blob->handle,
                                     hipIpcMemLazyEnablePeerAccess));
  c.ipc_cache[key] = p;
  return p;
}

// ---------------------------------------------------------------------------
// data-plane ops
// ---------------------------------------------------------------------------

static bool ipc_enabled() {
  static bool v = env_bool("UCCL_P2P_ENABLE_IPC", true);
  return v;
}

void Endpoint::do_send(Conn& c, void const* ptr, size_t bytes, int device) {
  trace::Span span("p2p", "send");
  OpTimer ot__(st_send_, bytes);
  if (c.rccl_comm && is_gpu(device)) {
    // RCCL plane: header over TCP for matching order; payload moves
    // device-to-device as ncclSend (blocks until the peer's recv posts)
    c.send_msg(MsgHdr{kSendRccl, bytes, 0, 0, 0});
    UCCL_CHECK(RcclPlane::get().send(c.rccl_comm, ptr, bytes,
                                     1 - c.rccl_rank, c.rccl_tx))
        << "rccl plane send failed";
    return;
  }
  if (is_gpu(device) && c.same_host && ipc_enabled()) {
    // one-copy IPC path: ship {handle, offset}; receiver DtoD-copies
    IpcBlob blob{};
    void* base = nullptr;
    size_t base_sz = 0;
    UCCL_CHECK_HIP(hipMemGetAddressRange(
        reinterpret_cast<hipDeviceptr_t*>(&base), &base_sz,
        reinterpret_cast<hipDeviceptr_t>(const_cast<void*>(ptr))));
    UCCL_CHECK_HIP(hipIpcGetMemHandle(&blob.handle, base));
    blob.offset = static_cast<char const*>(ptr) - static_cast<char*>(base);
    blob.device = device;
    blob.src_pid = static_cast<int>(getpid());
    uint64_t token = c.next_token++;
    c.send_msg(MsgHdr{kSendIpc, bytes, 0, 0, token}, &blob, sizeof(blob));
    // wait for receiver's copy (src must stay valid until then)
    std::unique_lock<std::mutex> lk(c.tok_mu);
    c.tok_cv.wait(lk, [&] { return c.completed.count(token) || !c.alive; });
    bool const completed = c.completed.count(token) != 0;
    c.completed.erase(token);
    UCCL_CHECK(completed) << "connection died before IPC send completed";
    return;
  }
  if (!is_gpu(device)) {
    c.send_msg(MsgHdr{kSendData, bytes, 0, 0, 0}, ptr, bytes);
    return;
  }
  // GPU over the wire: pinned staging chunks (kMsgChunk-sized so the
  // multipath plane's message framing mirrors the receiver's reads)
  std::lock_guard<std::mutex> sg(staging_mu_);
  if (!staging_) {
    UCCL_CHECK_HIP(hipSetDevice(gpu_ >= 0 ? gpu_ : device));
    UCCL_CHECK_HIP(hipHostMalloc(&staging_, kMsgChunk));
    staging_bytes_ = kMsgChunk;
  }
  std::lock_guard<std::mutex> g(c.tx_mu);
  MsgHdr h{kSendData, bytes, 0, 0, 0};
  if (c.flow) {
    c.tp->send_msg(c.flow, &h, sizeof(h));
  } else {
    net::send_all(c.fd, &h, sizeof(h));
  }
  hipStream_t cs = copy_stream();
  for (size_t off = 0; off < bytes; off += staging_bytes_) {
    size_t n = std::min(staging_bytes_, bytes - off);
    UCCL_CHECK_HIP(hipMemcpyAsync(staging_,
                                  static_cast<char const*>(ptr) + off, n,
                                  hipMemcpyDeviceToHost, cs));
    UCCL_CHECK_HIP(hipStreamSynchronize(cs));
    if (c.flow) {
      c.tp->send_msg(c.flow, staging_, n);
    } else {
      net::send_all(c.fd, staging_, n);
    }
  }
}

void Endpoint::copy_to_user(RxItem& item, void* dst, size_t bytes,
                            int device) {
  UCCL_CHECK(item.bytes <= bytes)
      << "recv buffer too small: " << bytes << " < " << item.bytes;
  if (!is_gpu(device)) {
    memcpy(dst, item.data.data(), item.bytes);
  } else {
    hipStream_t cs = copy_stream();
    UCCL_CHECK_HIP(hipMemcpyAsync(dst, item.data.data(), item.bytes,
                                  hipMemcpyHostToDevice, cs));
    UCCL_CHECK_HIP(hipStreamSynchronize(cs));
  }
}

void Endpoint::do_recv(Conn& c, void* ptr, size_t bytes, int device) {
  trace::Span span("p2p", "recv");
  OpTimer ot__(st_recv_, bytes);
  std::shared_ptr<RxItem> item;
  {
    std::unique_lock<std::mutex> lk(c.rx_mu);
    c.rx_cv.wait(lk, [&] { return !c.rxq.empty() || !c.alive || stop_; });
    // closing a connection with a pending recv is a normal event (e.g.
    // teardown of a notify drainer): fail the op, don't abort the process
    if (c.rxq.empty())
      throw std::runtime_error("p2p connection closed during recv");
    item = c.rxq.front();
    c.rxq.pop_front();
  }
  if (item->rccl) {
    UCCL_CHECK(item->bytes <= bytes) << "recv buffer too small";
    UCCL_CHECK(c.rccl_comm) << "rccl frame on non-rccl conn";
    if (is_gpu(device)) {
      UCCL_CHECK(RcclPlane::get().recv(c.rccl_comm, ptr, item->bytes,
                                       1 - c.rccl_rank, c.rccl_rx))
          << "rccl plane recv failed";
    } else {
      void* tmp = nullptr;
      UCCL_CHECK_HIP(hipMalloc(&tmp, item->bytes));
      bool ok = RcclPlane::get().recv(c.rccl_comm, tmp, item->bytes,
                                      1 - c.rccl_rank, c.rccl_rx);
      if (ok)
        UCCL_CHECK_HIP(hipMemcpy(ptr, tmp, item->bytes,
                                 hipMemcpyDeviceToHost));
      (void)hipFree(tmp);
      UCCL_CHECK(ok) << "rccl plane recv failed";
    }
    return;
  }
  if (item->ipc) {
    UCCL_CHECK(is_gpu(device)) << "IPC send into host recv buffer";
    UCCL_CHECK(item->bytes <= bytes) << "recv buffer too small";
    void* src_base = open_ipc(c, &item->blob, item->blob.device);
    hipStream_t cs = copy_stream();
    UCCL_CHECK_HIP(hipMemcpyAsync(
        ptr, static_cast<char*>(src_base) + item->blob.offset, item->bytes,
        hipMemcpyDeviceToDevice, cs));
    UCCL_CHECK_HIP(hipStreamSynchronize(cs));
    c.send_msg(MsgHdr{kIpcDone, 0, 0, 0, item->token});
  } else {
    copy_to_user(*item, ptr, bytes, device);
  }
}

void Endpoint::do_write(Conn& c, void const* ptr, size_t bytes, int device,
                        Advert ad) {
  trace::Span span("p2p", "write");
  OpTimer ot__(st_write_, bytes);
  UCCL_CHECK(bytes <= ad.bytes) << "write larger than advertised window";
  uint64_t token = c.next_token++;
  if (c.rccl_comm && is_gpu(device)) {
    c.send_msg(MsgHdr{kWriteRccl, ad.mr_id, ad.offset, bytes, token});
    UCCL_CHECK(RcclPlane::get().send(c.rccl_comm, ptr, bytes,
                                     1 - c.rccl_rank, c.rccl_tx))
        << "rccl plane send failed";
  } else if (is_gpu(device) && c.same_host && ipc_enabled()) {
    IpcBlob blob{};
    void* base = nullptr;
    size_t base_sz = 0;
    UCCL_CHECK_HIP(hipMemGetAddressRange(
        reinterpret_cast<hipDeviceptr_t*>(&base), &base_sz,
        reinterpret_cast<hipDeviceptr_t>(const_cast<void*>(ptr))));
    UCCL_CHECK_HIP(hipIpcGetMemHandle(&blob.handle, base));
    blob.offset = static_cast<char const*>(ptr) - static_cast<char*>(base);
    blob.device = device;
    blob.src_pid = static_cast<int>(getpid());
    c.send_msg(MsgHdr{kWriteIpc, ad.mr_id, ad.offset, bytes, token}, &blob,
               sizeof(blob));
  } else if (!is_gpu(device)) {
    c.send_msg(MsgHdr{kWriteData, ad.mr_id, ad.offset, bytes, token}, ptr,
               bytes);
  } else {
    std::vector<char> tmp(bytes);
    hipStream_t cs = copy_stream();
    UCCL_CHECK_HIP(hipMemcpyAsync(tmp.data(), ptr, bytes,
                                  hipMemcpyDeviceToHost, cs));
    UCCL_CHECK_HIP(hipStreamSynchronize(cs));
    c.send_msg(MsgHdr{kWriteData, ad.mr_id, ad.offset, bytes, token},
               tmp.data(), bytes);
  }
  std::unique_lock<std::mutex> lk(c.tok_mu);
  c.tok_cv.wait(lk, [&] { return c.completed.count(token) || !c.alive; });
  bool const completed = c.completed.count(token) != 0;
  c.completed.erase(token);
  UCCL_CHECK(completed) << "connection died before write was acknowledged";
}

void Endpoint::do_read(Conn& c, void* ptr, size_t bytes, int device,
                       Advert ad) {
  trace::Span span("p2p", "read");
  OpTimer ot__(st_read_, bytes);
  UCCL_CHECK(bytes <= ad.bytes) << "read larger than advertised window";
  uint64_t token = c.next_token++;
  if (c.rccl_comm && is_gpu(device)) {
    // rccl plane: the owner's rx loop ncclSends from the MR; our recv
    // completing IS the data arrival
    c.send_msg(MsgHdr{kReadReqRccl, ad.mr_id, ad.offset, bytes, token});
    UCCL_CHECK(RcclPlane::get().recv(c.rccl_comm, ptr, bytes,
                                     1 - c.rccl_rank, c.rccl_rx))
        << "rccl plane recv failed";
    return;
  }
  c.send_msg(MsgHdr{kReadReq, ad.mr_id, ad.offset, bytes, token});
  std::shared_ptr<RxItem> item;
  {
    std::unique_lock<std::mutex> lk(c.tok_mu);
    c.tok_cv.wait(lk, [&] { return c.completed.count(token) || !c.alive; });
    item = c.completed[token];
    c.completed.erase(token);
  }
  UCCL_CHECK(item) << "read failed (connection closed)";
  copy_to_user(*item, ptr, bytes, device);
}

// ---------------------------------------------------------------------------
// public (sync wrappers + async submission)
// ---------------------------------------------------------------------------

static Advert parse_advert(const std::string& s) {
  Advert ad{};
  UCCL_CHECK(s.size() == sizeof(Advert)) << "bad advert blob";
  memcpy(&ad, s.data(), sizeof(ad));
  UCCL_CHECK(ad.magic == kAdvertMagic) << "bad advert magic";
  return ad;
}

std::string Endpoint::advertise(uint64_t mr_id, uint64_t offset,
                                uint64_t bytes) {
  {
    std::lock_guard<std::mutex> g(mr_mu_);
    auto it = mrs_.find(mr_id);
    UCCL_CHECK(it != mrs_.end()) << "advertise of unknown mr";
    UCCL_CHECK(offset + bytes <= it->second.bytes) << "advert out of range";
  }
  Advert ad{kAdvertMagic, mr_id, offset, bytes};
  return std::string(reinterpret_cast<char*>(&ad), sizeof(ad));
}

void Endpoint::send(uint64_t cid, void const* p, size_t n, int dev) {
  do_send(*conn(cid), p, n, dev);
}
void Endpoint::recv(uint64_t cid, void* p, size_t n, int dev) {
  do_recv(*conn(cid), p, n, dev);
}
void Endpoint::write(uint64_t cid, void const* p, size_t n, int dev,
                     const std::string& ad) {
  do_write(*conn(cid), p, n, dev, parse_advert(ad));
}
void Endpoint::read(uint64_t cid, void* p, size_t n, int dev,
                    const std::string& ad) {
  do_read(*conn(cid), p, n, dev, parse_advert(ad));
}

uint64_t Endpoint::submit(std::function<void()> fn) {
  auto status = std::make_shared<std::atomic<int>>(0);
  uint64_t id = next_xfer_++;
  {
    std::lock_guard<std::mutex> g(xfer_mu_);
    xfers_[id] = status;
  }
  {
    std::lock_guard<std::mutex> g(task_mu_);
    tasks_.push_back([fn = std::move(fn), status] {
      try {
        fn();
        status->store(1);
      } catch (std::exception const& e) {
        UCCL_LOG_WARN << "async p2p op failed: " << e.what();
        status->store(2);  // completed-with-failure
      }
    });
  }
  task_cv_.notify_one();
  return id;
}

void Endpoint::worker_loop() {
  if (gpu_ >= 0) (void)hipSetDevice(gpu_);
  while (true) {
    std::function<void()> task;
    {
      std::unique_lock<std::mutex> lk(task_mu_);
      task_cv_.wait(lk, [this] { return !tasks_.empty() || stop_; });
      if (stop_ && tasks_.empty()) return;
      task = std::move(tasks_.front());
      tasks_.pop_front();
    }
    task();
  }
}

uint64_t Endpoint::send_async(uint64_t cid, void const* p, size_t n,
                              int dev) {
  auto c = conn(cid);
  uint64_t tk;
  {
    std::lock_guard<std::mutex> g(c->ord_mu);
    tk = c->tx_ticket++;
  }
  return submit([this, c, p, n, dev, tk] {
    {
      std::unique_lock<std::mutex> lk(c->ord_mu);
      c->ord_cv.wait(lk, [&] { return c->tx_serving == tk || !c->alive; });
    }
    // serving must advance even if the op throws, or every later ticket
    // on this connection would wait forever
    struct Adv {
      Conn* c;
      ~Adv() {
        {
          std::lock_guard<std::mutex> g(c->ord_mu);
          ++c->tx_serving;
        }
        c->ord_cv.notify_all();
      }
    } adv{c.get()};
    do_send(*c, p, n, dev);
  });
}
uint64_t Endpoint::recv_async(uint64_t cid, void* p, size_t n, int dev) {
  auto c = conn(cid);
  uint64_t tk;
  {
    std::lock_guard<std::mutex> g(c->ord_mu);
    tk = c->rx_ticket++;
  }
  return submit([this, c, p, n, dev, tk] {
    {
      std::unique_lock<std::mutex> lk(c->ord_mu);
      c->ord_cv.wait(lk, [&] { return c->rx_serving == tk || !c->alive; });
    }
    struct Adv {
      Conn* c;
      ~Adv() {
        {
          std::lock_guard<std::mutex> g(c->ord_mu);
          ++c->rx_serving;
        }
        c->ord_cv.notify_all();
      }
    } adv{c.get()};
    do_recv(*c, p, n, dev);
  });
}
uint64_t Endpoint::write_async(uint64_t cid, void const* p, size_t n, int dev,
                               const std::string& ad) {
  auto c = conn(cid);
  Advert a = parse_advert(ad);
  return submit([this, c, p, n, dev, a] { do_write(*c, p, n, dev, a); });
}
uint64_t Endpoint::read_async(uint64_t cid, void* p, size_t n, int dev,
                              const std::string& ad) {
  auto c = conn(cid);
  Advert a = parse_advert(ad);
  return submit([this, c, p, n, dev, a] { do_read(*c, p, n, dev, a); });
}

bool Endpoint::poll_async(uint64_t xfer_id) {
  std::lock_guard<std::mutex> g(xfer_mu_);
  auto it = xfers_.find(xfer_id);
  UCCL_CHECK(it != xfers_.end()) << "unknown transfer " << xfer_id;
  if (it->second->load() != 0) {  // 1 = ok, 2 = failed (warning logged)
    xfers_.erase(it);
    return true;
  }
  return false;
}

std::map<std::string, Endpoint::OpStat> Endpoint::stats() {
  std::map<std::string, OpStat> out;
  auto fill = [&](char const* name, OpRec& r) {
    OpStat st;
    st.calls = r.calls.load(std::memory_order_relaxed);
    st.bytes = r.bytes.load(std::memory_order_relaxed);
    st.p50_us = r.lat.percentile_us(50);
    st.p99_us = r.lat.percentile_us(99);
    out[name] = st;
  };
  fill("send", st_send_);
  fill("recv", st_recv_);
  fill("write", st_write_);
  fill("read", st_read_);
  return out;
}

void Endpoint::close_conn(uint64_t conn_id) {
  std::shared_ptr<Conn> c;
  {
    std::lock_guard<std::mutex> g(conn_mu_);
    auto it = conns_.find(conn_id);
    if (it == conns_.end()) return;
    c = it->second;
    conns_.erase(it);
  }
  {
    std::lock_guard<std::mutex> g1(c->rx_mu);
    std::lock_guard<std::mutex> g2(c->tok_mu);
    std::lock_guard<std::mutex> g3(c->ord_mu);
    c->alive = false;
  }
  ::shutdown(c->fd, SHUT_RDWR);
  if (c->flow && tp_) tp_->close_flow(c->flow);
  c->rx_cv.notify_all();
  c->tok_cv.notify_all();
  c->ord_cv.notify_all();
  if (c->rx.joinable()) c->rx.join();
  ::close(c->fd);
  std::lock_guard<std::mutex> g2(c->ipc_mu);
  for (auto& [k, p] : c->ipc_cache) (void)hipIpcCloseMemHandle(p);
  c->ipc_cache.clear();
}

int Endpoint::num_conns() {
  std::lock_guard<std::mutex> g(conn_mu_);
  return static_cast<int>(conns_.size());
}

}  // namespace p2p
}  // namespace uccl
