// Software multipath reliable transport.
//
// Parity role: the reference's collective/rdma transport core
// (SURVEY.md §2.2 — chunking, multipath QP spraying, CSN sequencing,
// SACK-bitmap selective repeat, dup-ACK fast retransmit, RTO, Timely
// congestion control; transport.cc:2228-3457) re-implemented from
// scratch over a pluggable packet fabric. The shipped fabric is UDP
// (N sockets = N paths), which runs on any host and supports
// deterministic loss injection for tests; an ibverbs UC fabric slots in
// behind the same Fabric interface on RDMA-capable nodes (the spraying /
// reliability / CC layers are fabric-agnostic by design, like the
// reference's engine split).
//
// Protocol:
//   DATA  {flow, msg_id, msg_bytes, offset, len, csn, ts_ns}
//   ACK   {flow, cum_csn, sack[2]x64 bits (csn in [cum, cum+128)), ts_echo}
// CSN is per-flow-direction and monotonic across messages; the receiver
// reassembles by (msg_id, offset) and acks by csn. The sender keeps an
// in-flight window bounded by a Timely-style RTT-gradient congestion
// window, fast-retransmits on 3 dup-SACKs, and falls back to RTO.
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <deque>
#include <map>
#include <memory>
#include <mutex>
#include <string>
#include <thread>
#include <unordered_map>
#include <vector>

namespace uccl {
namespace transport {

struct Stats {
  uint64_t data_sent = 0;
  uint64_t data_recv = 0;
  uint64_t acks_sent = 0;
  uint64_t acks_recv = 0;
  uint64_t retransmits = 0;
  uint64_t rto_retransmits = 0;
  uint64_t injected_drops = 0;
  uint64_t dup_recv = 0;    // non-fresh data arrivals (peer resent an
                            // already-received chunk: ack loss signal)
  uint64_t send_fail = 0;   // fabric post failures (e.g. sendto error)
  uint64_t msgs_sent = 0;
  uint64_t msgs_recv = 0;
  double srtt_us = 0;
  double cwnd = 0;
  double rtt_p50_us = 0;  // from the log-bucket latency recorder
  double rtt_p99_us = 0;
};

class TransportEndpoint {
 public:
  // num_paths UDP sockets are opened per endpoint; chunks spray across the
  // cartesian pairing (our path i -> peer path i).
  explicit TransportEndpoint(int num_paths = 8, size_t chunk_bytes = 8192);
  ~TransportEndpoint();

  std::string metadata() const;  // {ip, ctrl_port}
  // `tag` identifies the connecting peer to the acceptor (e.g. its rank).
  uint64_t connect(const std::string& remote_metadata, uint64_t tag = 0);
  uint64_t accept(uint64_t* peer_tag = nullptr);

  // Blocking reliable message ops (in-order per flow).
  void send_msg(uint64_t flow, void const* ptr, size_t bytes);
  void recv_msg(uint64_t flow, void* ptr, size_t bytes);

  // Async posting (the reference's post-then-poll-CQ discipline,
  // collective/rdma/transport.cc post_send/uc_poll_cq): enqueue without
  // waiting for acks. `ptr` must stay valid until flush_sends() returns.
  // Message order on the flow is the posting order, identical to
  // blocking sends — receivers cannot tell the difference.
  void send_msg_async(uint64_t flow, void const* ptr, size_t bytes);
  // Wait until every message posted on `flow` (async or blocking) has
  // been fully acked; throws if the flow failed.
  void flush_sends(uint64_t flow);

  Stats stats() const;

  // Unblock all pending send/recv (they throw std::runtime_error) and stop
  // the progress machinery; safe to call before destruction while other
  // threads are still blocked in recv_msg.
  void shutdown();

  // Fail one flow: pending and future send/recv on it throw. (The analog
  // of closing one connection while the endpoint stays up.)
  void close_flow(uint64_t flow);

 private:
  struct Flow;
  struct Impl;
  std::unique_ptr<Impl> impl_;
};

}  // namespace transport
}  // namespace uccl
