// P2P transfer engine: NIXL-style initiator/target endpoint.
//
// Parity role: the reference's p2p/engine.h:243 Endpoint facade
// (connect/accept, MR registration, blocking + async send/recv,
// one-sided read/write with receiver-advertised descriptors, same-node
// GPU-IPC fast path, proxy worker threads) — re-designed for MI355X:
//   - same-host GPU transfers ride HIP IPC + hipMemcpy DtoD over xGMI
//     (one copy, no staging), with an IPC-handle cache per (pid, base)
//   - cross-host traffic uses a length-prefixed TCP data plane with
//     pinned-host staging for GPU memory (the RDMA multipath transport
//     slots in behind the same ops when NICs exist; see csrc/transport/)
//   - a small worker pool executes async transfers; completion is polled
//     via transfer ids, mirroring the reference's poll_async
//     (p2p/engine.cc:2267)
#pragma once

#include <hip/hip_runtime.h>

#include "../core/latency.h"

#include <atomic>
#include <chrono>
#include <map>
#include <condition_variable>
#include <cstdint>
#include <deque>
#include <functional>
#include <memory>
#include <mutex>
#include <string>
#include <thread>
#include <unordered_map>
#include <vector>

namespace uccl {
namespace transport {
class TransportEndpoint;
}
namespace p2p {

struct MR {
  uint64_t id;
  void* ptr;
  size_t bytes;
  int device;  // -1 = host memory
};

// Receiver-advertised one-sided descriptor (the reference's 64-byte
// FifoItem, p2p/util/common.h:123).
struct Advert {
  uint64_t magic;
  uint64_t mr_id;
  uint64_t offset;
  uint64_t bytes;
};

class Endpoint {
 public:
  explicit Endpoint(int gpu, int num_workers = 2);
  ~Endpoint();

  // --- rendezvous ---
  std::string metadata() const;          // serialized {ip, port, gpu, ...}
  uint64_t connect(const std::string& remote_metadata);
  uint64_t accept();                     // blocks for an inbound connection

  // --- memory registration ---
  uint64_t reg(void* ptr, size_t bytes, int device);
  void dereg(uint64_t mr_id);

  // --- two-sided ---
  void send(uint64_t conn_id, void const* ptr, size_t bytes, int device);
  void recv(uint64_t conn_id, void* ptr, size_t bytes, int device);
  uint64_t send_async(uint64_t conn_id, void const* ptr, size_t bytes,
                      int device);
  uint64_t recv_async(uint64_t conn_id, void* ptr, size_t bytes, int device);

  // --- one-sided ---
  std::string advertise(uint64_t mr_id, uint64_t offset, uint64_t bytes);
  void write(uint64_t conn_id, void const* ptr, size_t bytes, int device,
             const std::string& advert);
  void read(uint64_t conn_id, void* ptr, size_t bytes, int device,
            const std::string& advert);
  uint64_t write_async(uint64_t conn_id, void const* ptr, size_t bytes,
                       int device, const std::string& advert);
  uint64_t read_async(uint64_t conn_id, void* ptr, size_t bytes, int device,
                      const std::string& advert);

  bool poll_async(uint64_t xfer_id);  // true once complete (then forgets it)

  // per-op-family counters + latency percentiles (reference parity:
  // engine stats threads / proxy timing getters)
  struct OpStat {
    uint64_t calls = 0;
    uint64_t bytes = 0;
    double p50_us = 0, p99_us = 0;
  };
  std::map<std::string, OpStat> stats();

  int num_conns();
  // tear down one connection (reference parity: p2p remove_remote_endpoint,
  // p2p/engine.cc:2208); pending ops on it fail
  void close_conn(uint64_t conn_id);

 private:
  struct Conn;
  struct RxItem;

  void rx_loop(std::shared_ptr<Conn> c);
  void rx_loop_body(std::shared_ptr<Conn> c);
  void worker_loop();
  uint64_t submit(std::function<void()> fn);
  void do_send(Conn& c, void const* ptr, size_t bytes, int device);
  void do_recv(Conn& c, void* ptr, size_t bytes, int device);
  void do_write(Conn& c, void const* ptr, size_t bytes, int device,
                Advert ad);
  void do_read(Conn& c, void* ptr, size_t bytes, int device, Advert ad);
  void copy_to_user(RxItem& item, void* dst, size_t bytes, int device);
  void* open_ipc(Conn& c, const void* handle_bytes, int src_device);
  void setup_rccl(Conn& c, bool acceptor);
  std::shared_ptr<Conn> conn(uint64_t id);

  int gpu_;
  std::string host_id_;
  uint16_t port_ = 0;
  int listen_fd_ = -1;
  std::atomic<bool> stop_{false};

  std::thread listener_;
  std::vector<std::thread> workers_;
  std::deque<std::function<void()>> tasks_;
  std::mutex task_mu_;
  std::condition_variable task_cv_;

  std::mutex conn_mu_;
  std::unordered_map<uint64_t, std::shared_ptr<Conn>> conns_;
  std::deque<uint64_t> accepted_;
  std::condition_variable accept_cv_;
  std::atomic<uint64_t> next_conn_{1};

  std::mutex mr_mu_;
  std::unordered_map<uint64_t, MR> mrs_;
  std::atomic<uint64_t> next_mr_{1};

  std::mutex xfer_mu_;
  std::unordered_map<uint64_t, std::shared_ptr<std::atomic<int>>> xfers_;
  std::atomic<uint64_t> next_xfer_{1};

  // per-op observability
  struct OpRec {
    std::atomic<uint64_t> calls{0};
    std::atomic<uint64_t> bytes{0};
    LatencyHist lat;
    void add(uint64_t n, double us) {
      calls.fetch_add(1, std::memory_order_relaxed);
      bytes.fetch_add(n, std::memory_order_relaxed);
      lat.record_us(us);
    }
  };
  OpRec st_send_, st_recv_, st_write_, st_read_;
  struct OpTimer {
    OpRec& r;
    size_t n;
    std::chrono::steady_clock::time_point t0;
    OpTimer(OpRec& rr, size_t nn)
        : r(rr), n(nn), t0(std::chrono::steady_clock::now()) {}
    ~OpTimer() {
      r.add(n, std::chrono::duration<double, std::micro>(
                   std::chrono::steady_clock::now() - t0)
                   .count());
    }
  };

  // pinned staging for GPU<->TCP
  void* staging_ = nullptr;
  size_t staging_bytes_ = 0;
  std::mutex staging_mu_;

  // dedicated non-blocking stream for IPC DtoD copies: hipMemcpy DtoD is
  // async w.r.t. the host, so completion must be an explicit stream sync
  // before acking the sender (and must not ride the legacy null stream)
  hipStream_t copy_stream();
  void* copy_stream_ = nullptr;
  std::mutex copy_mu_;

  // optional multipath data plane (UCCL_P2P_TRANSPORT=multipath): conns
  // carry their bytes as reliable-transport messages instead of raw TCP
  // (the TCP socket stays as the control/handshake channel)
  std::unique_ptr<transport::TransportEndpoint> tp_;
  std::thread tp_acceptor_;
  std::mutex tp_mu_;
  std::condition_variable tp_cv_;
  std::map<uint64_t, uint64_t> tp_flows_;  // hello nonce -> flow
};

}  // namespace p2p
}  // namespace uccl
