// Fabric seam under the multipath reliable transport.
//
// The reliability / spraying / congestion-control layers in reliable.cpp
// are fabric-agnostic (the same split as the reference's engine vs
// rdma_io layering, collective/rdma/rdma_io.h:128-713 under
// transport.cc). A Fabric moves one chunk or one control frame between
// peers over N parallel paths:
//
//   UdpFabric   — N UDP sockets (= N paths), payload carried inline in
//                 the datagram; runs on any host, supports the CPU test
//                 tier and deterministic loss injection.
//   VerbsFabric — ibverbs RC QP pool (one data QP per path + one ctrl
//                 QP), RDMA_WRITE_WITH_IMM straight into the
//                 receiver-advertised message window (FIFO rendezvous),
//                 IMM = {RID:8, CSN:24}; payload is PLACED by the NIC,
//                 so chunk events carry no inline payload. Verbs calls
//                 go through the dlopen'd provider seam
//                 (verbs_provider.h) like the reference's ibverbs_dl.cc.
//
// Selection: UCCL_TP_FABRIC=udp|verbs (default udp; verbs falls back to
// udp with a warning when no RDMA device is present).
#pragma once

#include <cstddef>
#include <cstdint>
#include <ctime>
#include <functional>
#include <memory>
#include <string>

namespace uccl {
namespace transport {

// Wire description of one chunk (the protocol header fields the fabric
// must deliver alongside — or encode into — the payload transfer).
struct ChunkDesc {
  uint64_t flow;
  uint64_t msg_id;
  uint64_t msg_bytes;
  uint64_t off;
  uint32_t len;
  uint32_t csn;
  uint64_t ts_ns;
};

struct FabricEvent {
  enum Kind { kChunk, kCtrl } kind;
  uint64_t flow;
  int path;
  // kChunk: desc always valid; payload is the inline bytes (UDP) or
  // nullptr when the fabric already placed the data at its destination
  // (verbs RDMA write).
  ChunkDesc desc;
  char const* payload = nullptr;
  // kCtrl: opaque control frame (the reliable layer's ACK/SACK format)
  char const* ctrl = nullptr;
  size_t ctrl_len = 0;
};

class Fabric {
 public:
  virtual ~Fabric() = default;

  virtual int num_paths() const = 0;

  // Allocate per-flow fabric resources (verbs: the QP pool) and return
  // the addressing blob the peer needs (ports for UDP; GID+QPNs for
  // verbs). Exchanged over the TCP ctrl channel at flow setup.
  // `connector` distinguishes the two sides of a self-connection.
  virtual std::string create_flow(uint64_t flow, bool connector) = 0;

  // Install the peer side of `flow` from its metadata blob (verbs:
  // transition the QPs to RTS toward the peer's QPNs). Returns the
  // number of usable paths (min of both sides), or 0 on failure.
  virtual int install_peer(uint64_t flow, std::string const& md) = 0;
  virtual void remove_peer(uint64_t flow) = 0;

  // Sender: may the head message start/continue chunking? UDP: always.
  // Verbs: true once the receiver's window advertisement for msg_id has
  // arrived (FIFO rendezvous).
  virtual bool tx_ready(uint64_t flow, uint64_t msg_id) { return true; }

  // Sender: transfer one chunk over `path`. `payload` points at the
  // user's message bytes (off/len per desc). Returns false on a
  // transient would-block (caller re-pumps later).
  virtual bool post_chunk(uint64_t flow, int path, ChunkDesc const& desc,
                          void const* payload) = 0;

  // Either side: send an opaque control frame (ACK/SACK/credit).
  virtual void post_ctrl(uint64_t flow, int path, void const* frame,
                         size_t len) = 0;

  // Receiver: a buffer was posted for msg_id (recv_msg). Rendezvous
  // fabrics advertise {msg_id, addr, rkey, cap} to the peer so chunk
  // writes can be placed; UDP needs nothing.
  virtual void post_recv_window(uint64_t flow, uint64_t msg_id, void* buf,
                                size_t cap) {}

  // Drain inbound events; calls cb for each. Blocks up to timeout_ms
  // when idle. Returns the number of events delivered.
  virtual int poll(std::function<void(FabricEvent const&)> const& cb,
                   int timeout_ms) = 0;

  // Sharded variant for multi-engine progress: engine `shard` of
  // `nshards` drains its slice of the paths (UDP: sockets i where
  // i % nshards == shard). Default: shard 0 gets everything, the rest
  // idle-sleep (single-queue fabrics like the verbs CQ pair).
  virtual int poll_shard(int shard, int nshards,
                         std::function<void(FabricEvent const&)> const& cb,
                         int timeout_ms) {
    if (shard == 0 || nshards <= 1) return poll(cb, timeout_ms);
    struct timespec ts {0, timeout_ms * 1000000L};
    nanosleep(&ts, nullptr);
    return 0;
  }

  // Unblock a concurrent poll() (e.g. at shutdown or when new TX work
  // arrives and the progress thread is sleeping in poll).
  virtual void wake() = 0;
};

// Factory: UCCL_TP_FABRIC=udp|verbs. `num_paths`/`chunk_bytes` bound the
// per-path resources.
std::unique_ptr<Fabric> make_udp_fabric(int num_paths, size_t chunk_bytes);
std::unique_ptr<Fabric> make_verbs_fabric(int num_paths, size_t chunk_bytes);
std::unique_ptr<Fabric> make_fabric(int num_paths, size_t chunk_bytes);

}  // namespace transport
}  // namespace uccl
