// UdpFabric — the portable datagram fabric under the multipath reliable
// transport: N UDP sockets = N paths, chunk payloads carried inline.
// This is the extracted round-1 wire plane of reliable.cpp, now behind
// the Fabric seam so the verbs plane can slot in beside it.

#include <arpa/inet.h>
#include <netinet/in.h>
#include <poll.h>
#include <sys/socket.h>
#include <unistd.h>

#include <array>
#include <cstring>
#include <mutex>
#include <unordered_map>
#include <vector>

#include "../core/env.h"
#include "../core/log.h"
#include "../core/net.h"
#include "fabric.h"

namespace uccl {
namespace transport {

namespace {

constexpr uint32_t kFabMagic = 0x55434654;  // "UCFT"
enum FabKind : uint32_t { kFabData = 1, kFabCtrl = 2, kFabWake = 3 };

struct DataWire {
  uint32_t magic;
  uint32_t kind;
  uint64_t flow;
  uint64_t msg_id;
  uint64_t msg_bytes;
  uint64_t off;
  uint32_t len;
  uint32_t csn;
  uint64_t ts_ns;
};

struct CtrlWire {
  uint32_t magic;
  uint32_t kind;
  uint64_t flow;
};

struct MdWire {
  char ip[48];
  int32_t n;
  uint16_t ports[64];
};

struct PeerPaths {
  int n = 0;
  std::array<sockaddr_in, 64> addr;
};

class UdpFabric final : public Fabric {
 public:
  UdpFabric(int num_paths, size_t chunk_bytes)
      : np_(num_paths), chunk_bytes_(chunk_bytes) {
    UCCL_CHECK(chunk_bytes_ <= 60000) << "chunk must fit a UDP datagram";
    for (int i = 0; i < np_; ++i) {
      int s = ::socket(AF_INET, SOCK_DGRAM, 0);
      UCCL_CHECK(s >= 0) << "udp socket";
      int sz = 16 << 20;
      // FORCE variants bypass net.core.{r,w}mem_max when running as
      // root — without them loopback drops under bursts and
      // masquerades as loss
      if (setsockopt(s, SOL_SOCKET, SO_RCVBUFFORCE, &sz, sizeof(sz)) != 0)
        setsockopt(s, SOL_SOCKET, SO_RCVBUF, &sz, sizeof(sz));
      if (setsockopt(s, SOL_SOCKET, SO_SNDBUFFORCE, &sz, sizeof(sz)) != 0)
        setsockopt(s, SOL_SOCKET, SO_SNDBUF, &sz, sizeof(sz));
      sockaddr_in a{};
      a.sin_family = AF_INET;
      a.sin_addr.s_addr = htonl(INADDR_ANY);
      UCCL_CHECK(::bind(s, reinterpret_cast<sockaddr*>(&a), sizeof(a)) == 0);
      socklen_t al = sizeof(a);
      getsockname(s, reinterpret_cast<sockaddr*>(&a), &al);
      socks_.push_back(s);
      ports_.push_back(ntohs(a.sin_port));
    }
    wake_fd_ = ::socket(AF_INET, SOCK_DGRAM, 0);
    sockaddr_in a{};
    a.sin_family = AF_INET;
    a.sin_addr.s_addr = htonl(INADDR_ANY);
    UCCL_CHECK(::bind(wake_fd_, reinterpret_cast<sockaddr*>(&a),
                      sizeof(a)) == 0);
    socklen_t al = sizeof(a);
    getsockname(wake_fd_, reinterpret_cast<sockaddr*>(&a), &al);
    wake_port_ = ntohs(a.sin_port);
  }

  ~UdpFabric() override {
    for (int s : socks_) ::close(s);
    ::close(wake_fd_);
  }

  int num_paths() const override { return np_; }

  std::string create_flow(uint64_t flow, bool connector) override {
    // UDP paths are endpoint-global sockets; per-flow state is only the
    // peer address table filled by install_peer.
    (void)flow;
    (void)connector;
    MdWire m{};
    std::string ip = net::local_ip();
    strncpy(m.ip, ip.c_str(), sizeof(m.ip) - 1);
    m.n = np_;
    for (int i = 0; i < np_; ++i) m.ports[i] = ports_[i];
    return std::string(reinterpret_cast<char const*>(&m), sizeof(m));
  }

  int install_peer(uint64_t flow, std::string const& md) override {
    if (md.size() != sizeof(MdWire)) return 0;
    MdWire m{};
    memcpy(&m, md.data(), sizeof(m));
    if (m.n < 1 || m.n > 64) return 0;
    PeerPaths pp;
    pp.n = std::min(np_, static_cast<int>(m.n));
    for (int i = 0; i < pp.n; ++i) {
      sockaddr_in a{};
      a.sin_family = AF_INET;
      a.sin_port = htons(m.ports[i]);
      inet_pton(AF_INET, m.ip, &a.sin_addr);
      pp.addr[i] = a;
    }
    std::lock_guard<std::mutex> g(pmu_);
    peers_[flow] = pp;
    return pp.n;
  }

  void remove_peer(uint64_t flow) override {
    std::lock_guard<std::mutex> g(pmu_);
    peers_.erase(flow);
  }

  bool post_chunk(uint64_t flow, int path, ChunkDesc const& d,
                  void const* payload) override {
    sockaddr_in to{};
    if (!peer_addr(flow, path, &to)) return false;
    DataWire h{kFabMagic, kFabData, d.flow,  d.msg_id, d.msg_bytes,
               d.off,     d.len,    d.csn,   d.ts_ns};
    char buf[sizeof(DataWire) + 65536];
    memcpy(buf, &h, sizeof(h));
    if (d.len) memcpy(buf + sizeof(h), payload, d.len);
    ssize_t const n = sendto(socks_[path], buf, sizeof(h) + d.len, 0,
                             reinterpret_cast<sockaddr*>(&to), sizeof(to));
    // ENOBUFS and friends = local transient loss: report so the caller
    // accounts a send failure (the RTO machinery covers recovery)
    return n == static_cast<ssize_t>(sizeof(h) + d.len);
  }

  void post_ctrl(uint64_t flow, int path, void const* frame,
                 size_t len) override {
    sockaddr_in to{};
    if (!peer_addr(flow, path, &to)) return;
    char buf[sizeof(CtrlWire) + 512];
    UCCL_CHECK(len <= 512) << "oversized ctrl frame";
    CtrlWire h{kFabMagic, kFabCtrl, flow};
    memcpy(buf, &h, sizeof(h));
    memcpy(buf + sizeof(h), frame, len);
    (void)sendto(socks_[path], buf, sizeof(h) + len, 0,
                 reinterpret_cast<sockaddr*>(&to), sizeof(to));
  }

  int poll(std::function<void(FabricEvent const&)> const& cb,
           int timeout_ms) override {
    return poll_shard(0, 1, cb, timeout_ms);
  }

  // Engine-sharded drain with recvmmsg batching: one syscall pulls up
  // to kBatch datagrams per socket (the reference's CQ batch polling
  // role, uc_handle_completion). Each engine owns sockets
  // i % nshards == shard; the wake socket belongs to shard 0.
  int poll_shard(int shard, int nshards,
                 std::function<void(FabricEvent const&)> const& cb,
                 int timeout_ms) override {
    constexpr int kBatch = 64;
    size_t const slot = sizeof(DataWire) + 65536 + 64;
    thread_local std::vector<char> bufs;
    if (bufs.size() < kBatch * slot) bufs.resize(kBatch * slot);

    std::vector<pollfd> pfds;
    std::vector<int> paths;  // original path index per pollfd
    for (int i = shard; i < static_cast<int>(socks_.size()); i += nshards) {
      pfds.push_back({socks_[i], POLLIN, 0});
      paths.push_back(i);
    }
    if (shard == 0) {
      pfds.push_back({wake_fd_, POLLIN, 0});
      paths.push_back(-1);
    }
    if (pfds.empty()) {
      struct timespec ts {0, timeout_ms * 1000000L};
      nanosleep(&ts, nullptr);
      return 0;
    }
    (void)::poll(pfds.data(), pfds.size(), timeout_ms);

    mmsghdr msgs[kBatch];
    iovec iov[kBatch];
    int delivered = 0;
    for (size_t i = 0; i < pfds.size(); ++i) {
      while (true) {
        for (int k = 0; k < kBatch; ++k) {
          iov[k] = {bufs.data() + k * slot, slot};
          memset(&msgs[k].msg_hdr, 0, sizeof(msghdr));
          msgs[k].msg_hdr.msg_iov = &iov[k];
          msgs[k].msg_hdr.msg_iovlen = 1;
        }
        int n = recvmmsg(pfds[i].fd, msgs, kBatch, MSG_DONTWAIT, nullptr);
        if (n <= 0) break;
        for (int k = 0; k < n; ++k) {
          char const* buf = bufs.data() + k * slot;
          size_t const len = msgs[k].msg_len;
          if (len < sizeof(CtrlWire)) continue;
          auto const* cw = reinterpret_cast<CtrlWire const*>(buf);
          if (cw->magic != kFabMagic || cw->kind == kFabWake) continue;
          FabricEvent ev{};
          ev.path = paths[i] >= 0 ? paths[i] : 0;
          if (cw->kind == kFabData && len >= sizeof(DataWire)) {
            auto const* h = reinterpret_cast<DataWire const*>(buf);
            // wire len must match the datagram (truncation guard)
            if (sizeof(DataWire) + h->len != len) continue;
            ev.kind = FabricEvent::kChunk;
            ev.flow = h->flow;
            ev.desc = ChunkDesc{h->flow, h->msg_id, h->msg_bytes,
                                h->off,  h->len,    h->csn,     h->ts_ns};
            ev.payload = buf + sizeof(DataWire);
          } else if (cw->kind == kFabCtrl) {
            ev.kind = FabricEvent::kCtrl;
            ev.flow = cw->flow;
            ev.ctrl = buf + sizeof(CtrlWire);
            ev.ctrl_len = len - sizeof(CtrlWire);
          } else {
            continue;
          }
          cb(ev);
          ++delivered;
        }
        if (n < kBatch) break;
      }
    }
    return delivered;
  }

  void wake() override {
    sockaddr_in a{};
    a.sin_family = AF_INET;
    a.sin_port = htons(wake_port_);
    inet_pton(AF_INET, "127.0.0.1", &a.sin_addr);
    CtrlWire w{kFabMagic, kFabWake, 0};
    (void)sendto(wake_fd_, &w, sizeof(w), 0,
                 reinterpret_cast<sockaddr*>(&a), sizeof(a));
  }

 private:
  bool peer_addr(uint64_t flow, int path, sockaddr_in* out) {
    std::lock_guard<std::mutex> g(pmu_);
    auto it = peers_.find(flow);
    if (it == peers_.end() || it->second.n == 0) return false;
    *out = it->second.addr[path % it->second.n];
    return true;
  }

  int np_;
  size_t chunk_bytes_;
  std::vector<int> socks_;
  std::vector<uint16_t> ports_;
  int wake_fd_ = -1;
  uint16_t wake_port_ = 0;
  std::mutex pmu_;
  std::unordered_map<uint64_t, PeerPaths> peers_;
};

}  // namespace

std::unique_ptr<Fabric> make_udp_fabric(int num_paths, size_t chunk_bytes) {
  return std::make_unique<UdpFabric>(num_paths, chunk_bytes);
}

std::unique_ptr<Fabric> make_fabric(int num_paths, size_t chunk_bytes) {
  std::string const kind = env_str("UCCL_TP_FABRIC", "udp");
  if (kind == "verbs") {
    try {
      return make_verbs_fabric(num_paths, chunk_bytes);
    } catch (std::exception const& e) {
      UCCL_LOG_WARN << "verbs fabric unavailable (" << e.what()
                    << "); falling back to udp";
    }
  }
  return make_udp_fabric(num_paths, chunk_bytes);
}

}  // namespace transport
}  // namespace uccl
