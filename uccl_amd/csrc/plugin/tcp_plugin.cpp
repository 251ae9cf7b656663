// librccl-net-uccl.so — RCCL network plugin (net ABI v6).
//
// Parity role: the reference's collective/rdma/nccl_plugin.cc vtable
// (:85-633) — the drop-in transport under stock RCCL, carrying NCCL's
// traffic over the software multipath reliable transport
// (csrc/transport/: chunking, path spraying, SACK selective repeat,
// Timely/Swift CC) exactly like the reference's plugin rides its
// UcclRDMAEngine. Two data planes, selected by UCCL_NET_TRANSPORT:
//   multipath (default) — TransportEndpoint flows (message-oriented)
//   tcp               — plain stream sockets (NCCL-socket-transport-like)
//
// Pointer support: NCCL_PTR_HOST always; NCCL_PTR_CUDA when built with
// UCCL_NET_HIP (the shipped build) — registered device MRs are staged
// through per-comm pinned bounce buffers with hipMemcpy, mirroring the
// reference's GPU regMr+staging (nccl_plugin.cc:472-594). iflush stays a
// no-op because recv completion is only signalled after the HtoD copy
// has synchronized (visibility is implied, no GDR in this fabric).
//
// When built without UCCL_NET_HIP the plugin has no HIP dependency and
// is fully CPU-testable (the dlopen harness drives the vtable over both
// data planes with host pointers).

#include <fcntl.h>
#include <unistd.h>
#include <cstdio>
#include <poll.h>

#include <atomic>
#include <chrono>
#include <map>
#include <condition_variable>
#include <cstring>
#include <deque>
#include <memory>
#include <mutex>
#include <thread>
#include <vector>

#ifdef UCCL_NET_HIP
#include <hip/hip_runtime.h>
#endif

#include "../core/env.h"
#include "../core/log.h"
#include "../core/net.h"
#include "../transport/reliable.h"
#include "net_plugin_abi.h"

namespace {

using uccl::net::recv_all;
using uccl::net::send_all;

ncclDebugLogger_t g_log = nullptr;

#define PLOG(lvl, ...) \
  if (g_log) g_log(lvl, ~0ul, __FILE__, __LINE__, __VA_ARGS__)

struct Handle {
  char ip[48];
  uint16_t port;
};

// first bytes on every tcp-plane connection: rejects strays (e.g. another
// process's connect-retry landing on a recycled ephemeral port)
constexpr uint64_t kPluginCookie = 0x7563636c2d6e6574ULL;  // "uccl-net"

// Registered memory record. type is NCCL_PTR_HOST or NCCL_PTR_CUDA; device
// is resolved at registration so staging copies can set the right context.
struct Mr {
  int type = NCCL_PTR_HOST;
  int device = 0;
};

// One request may cover n grouped recvs (p_irecv n>1): done counts down.
struct Request {
  std::atomic<int> pending{1};
  std::atomic<int> error{0};
  int n = 1;
  int sizes[8] = {0};
  bool recv = false;
  bool done() const { return pending.load(std::memory_order_acquire) == 0; }
  void complete_one() { pending.fetch_sub(1, std::memory_order_release); }
};

struct Frame {
  uint64_t bytes;
  int tag;
  std::vector<char> data;  // rx side buffer when recv not yet posted
};

struct PostedRecv {
  void* data;
  int cap;
  int tag;
  Request* req;
  int slot;       // index into req->sizes for grouped recvs
  Mr mr;          // destination memory kind
};

struct SendOp {
  void const* data;
  int size;
  int tag;
  Request* req;
  Mr mr;          // source memory kind
};

bool use_multipath() {
  static bool v = uccl::env_str("UCCL_NET_TRANSPORT", "multipath") !=
                  std::string("tcp");
  return v;
}

static bool net_debug() {
  static bool v = uccl::env_bool("UCCL_NET_DEBUG", false);
  return v;
}
#define NET_DBG(fmt, ...) \
  do { \
    if (net_debug()) \
      fprintf(stderr, "[uccl-net %d] " fmt "\n", getpid(), ##__VA_ARGS__); \
  } while (0)

// --- multipath fabric singleton -------------------------------------------
// One TransportEndpoint per process; RCCL's many listenComms are
// distinguished by a nonce carried in the connect tag, so a global
// acceptor thread can route inbound flows to the right listener queue.
struct MpFabric {
  // process-lifetime singleton: the endpoint is intentionally leaked so
  // static-destruction order can never race the detached acceptor thread
  uccl::transport::TransportEndpoint& tp =
      *new uccl::transport::TransportEndpoint(
          static_cast<int>(uccl::env_int("UCCL_NET_PATHS", 8)),
          static_cast<size_t>(uccl::env_int("UCCL_NET_CHUNK", 16384)));
  std::mutex mu;
  std::condition_variable cv;
  std::map<uint64_t, std::deque<uint64_t>> queues;  // nonce -> flows
  std::atomic<uint64_t> next_nonce{1};
  std::thread acceptor;
  MpFabric() {
    acceptor = std::thread([this] {
      try {
        while (true) {
          uint64_t tag = 0;
          uint64_t flow = tp.accept(&tag);
          NET_DBG("fabric accept flow=%llx nonce=%llu",
                  (unsigned long long)flow, (unsigned long long)tag);
          std::lock_guard<std::mutex> g(mu);
          queues[tag].push_back(flow);
          cv.notify_all();
        }
      } catch (std::exception const&) {
      }
    });
    acceptor.detach();  // process-lifetime singleton
  }
  static MpFabric& get() {
    static MpFabric f;
    return f;
  }
};

struct MpHandle {
  uint64_t nonce;
  uint16_t md_len;
  char md[96];
};

struct ListenComm {
  int fd = -1;        // tcp mode
  uint64_t nonce = 0;  // multipath mode
};

#ifdef UCCL_NET_HIP
// Grow-to-fit pinned bounce buffer for device-MR staging (per comm, per
// direction; the worker thread owns it so no locking is needed).
struct Staging {
  char* buf = nullptr;
  size_t cap = 0;
  ~Staging() {
    if (buf) (void)hipHostFree(buf);
  }
  char* ensure(size_t n) {
    if (n > cap) {
      if (buf) (void)hipHostFree(buf);
      size_t c = 4096;
      while (c < n) c <<= 1;
      if (hipHostMalloc(&buf, c) != hipSuccess) {
        buf = nullptr;
        cap = 0;
        return nullptr;
      }
      cap = c;
    }
    return buf;
  }
};
#endif

struct Comm {
  int fd = -1;        // tcp mode
  uint64_t flow = 0;  // multipath mode (0 = tcp)
  bool sender = false;
  std::atomic<bool> alive{true};
  std::thread worker;

  // sender
  std::deque<SendOp> sendq;
  // receiver
  std::deque<PostedRecv> posted;
  std::deque<Frame> unmatched;

#ifdef UCCL_NET_HIP
  Staging tx_staging, rx_staging;
#endif

  std::mutex mu;
  std::condition_variable cv;

  ~Comm();
};

// copy `bytes` from a host buffer into pr.data (host or device MR)
bool place_payload(Comm* c, PostedRecv const& pr, char const* src,
                   uint64_t bytes) {
  (void)c;
  if (!bytes) return true;
#ifdef UCCL_NET_HIP
  if (pr.mr.type == NCCL_PTR_CUDA) {
    (void)hipSetDevice(pr.mr.device);
    return hipMemcpy(pr.data, src, bytes, hipMemcpyHostToDevice) ==
           hipSuccess;
  }
#endif
  memcpy(pr.data, src, bytes);
  return true;
}

struct WireHdr {
  uint64_t bytes;
  int32_t tag;
  int32_t pad;
};

Comm::~Comm() {
  {
    // flag must flip under the mutex or a worker between its predicate
    // check and cv sleep misses the wake forever (lost wakeup)
    std::lock_guard<std::mutex> g(mu);
    alive = false;
  }
  cv.notify_all();
  if (fd >= 0) ::shutdown(fd, SHUT_RDWR);
  if (flow) MpFabric::get().tp.close_flow(flow);  // unblock recv_msg
  if (worker.joinable()) worker.join();
  if (fd >= 0) ::close(fd);
}

// channel ops spanning both data planes (message framing == byte framing:
// the tcp path streams the same {hdr}{payload} sequence the multipath
// path sends as two discrete reliable messages)
void chan_send(Comm* c, void const* buf, size_t n) {
  if (c->flow) {
    NET_DBG("send flow=%llx n=%zu", (unsigned long long)c->flow, n);
    MpFabric::get().tp.send_msg(c->flow, buf, n);
  } else {
    send_all(c->fd, buf, n);
  }
}

bool chan_recv(Comm* c, void* buf, size_t n) {
  if (c->flow) {
    NET_DBG("recv post flow=%llx n=%zu", (unsigned long long)c->flow, n);
    try {
      MpFabric::get().tp.recv_msg(c->flow, buf, n);
      NET_DBG("recv done flow=%llx n=%zu", (unsigned long long)c->flow, n);
      return true;
    } catch (std::exception const&) {
      return false;
    }
  }
  return recv_all(c->fd, buf, n);
}

void tx_loop(Comm* c) {
  while (c->alive) {
    SendOp op;
    {
      std::unique_lock<std::mutex> lk(c->mu);
      // bounded wait: re-check the predicate on a 50ms tick. Under heavy
      // CPU oversubscription we observed rare stalls with an op queued
      // despite the notify; the periodic re-check bounds any missed wake
      // at 50ms (the reference's engines busy-poll with adaptive sleep
      // for the same robustness, p2p/util/adaptive_sleeper.h).
      while (c->sendq.empty() && c->alive)
#ifdef UCCL_SAN_NO_TIMED_WAIT
        // TSan build only: this toolchain's libtsan does not intercept
        // pthread_cond_clockwait (libstdc++'s wait_for), so the in-wait
        // unlock is invisible and every timed wait reports a false
        // "double lock"/race (verified with a 20-line repro). The
        // production bounded wait stays timed.
        c->cv.wait(lk);
#else
        c->cv.wait_for(lk, std::chrono::milliseconds(50));
#endif
      if (!c->alive) return;
      op = c->sendq.front();
      c->sendq.pop_front();
    }
    void const* payload = op.data;
#ifdef UCCL_NET_HIP
    if (op.mr.type == NCCL_PTR_CUDA && op.size) {
      (void)hipSetDevice(op.mr.device);
      char* st = c->tx_staging.ensure(op.size);
      if (!st || hipMemcpy(st, op.data, op.size, hipMemcpyDeviceToHost) !=
                     hipSuccess) {
        PLOG(NCCL_LOG_WARN, "uccl-net: DtoH staging failed (%d B)", op.size);
        op.req->error.store(1, std::memory_order_relaxed);
        op.req->complete_one();
        continue;
      }
      payload = st;
    }
#endif
    WireHdr h{static_cast<uint64_t>(op.size), op.tag, 0};
    chan_send(c, &h, sizeof(h));
    if (op.size) chan_send(c, payload, op.size);
    op.req->sizes[0] = op.size;
    op.req->complete_one();
  }
}

constexpr uint64_t kMaxWireBytes = 1ull << 31;  // frame sanity cap

void rx_loop(Comm* c) {
  while (c->alive) {
    WireHdr h{};
    if (!chan_recv(c, &h, sizeof(h))) return;
    if (h.bytes > kMaxWireBytes) {
      PLOG(NCCL_LOG_WARN, "uccl-net: bogus frame size %llu; closing",
           (unsigned long long)h.bytes);
      return;
    }
    // try to match a posted recv by tag
    PostedRecv pr{};
    bool matched = false;
    {
      std::lock_guard<std::mutex> g(c->mu);
      for (auto it = c->posted.begin(); it != c->posted.end(); ++it) {
        if (it->tag == h.tag) {
          pr = *it;
          c->posted.erase(it);
          matched = true;
          break;
        }
      }
    }
    if (matched) {
      if (pr.cap < static_cast<int>(h.bytes)) {
        // Peer sent more than the posted capacity: drain the frame so the
        // stream stays in sync and fail the REQUEST (RCCL aborts cleanly
        // via p_test), instead of silently killing the connection.
        PLOG(NCCL_LOG_WARN, "uccl-net: recv overflow tag=%d %lu > %d",
             h.tag, (unsigned long)h.bytes, pr.cap);
        std::vector<char> sink(h.bytes);
        if (h.bytes) chan_recv(c, sink.data(), h.bytes);
        pr.req->error.store(1, std::memory_order_relaxed);
        pr.req->complete_one();
        continue;
      }
      bool ok = true;
#ifdef UCCL_NET_HIP
      if (pr.mr.type == NCCL_PTR_CUDA) {
        char* st = c->rx_staging.ensure(h.bytes ? h.bytes : 1);
        if (!st) {
          ok = false;
        } else {
          if (h.bytes) chan_recv(c, st, h.bytes);
          ok = place_payload(c, pr, st, h.bytes);
        }
      } else
#endif
      {
        if (h.bytes) chan_recv(c, pr.data, h.bytes);
      }
      if (!ok) pr.req->error.store(1, std::memory_order_relaxed);
      pr.req->sizes[pr.slot] = static_cast<int>(h.bytes);
      pr.req->complete_one();
    } else {
      Frame f;
      f.bytes = h.bytes;
      f.tag = h.tag;
      f.data.resize(h.bytes);
      if (h.bytes) chan_recv(c, f.data.data(), h.bytes);
      // re-check posted under the lock: between the first posted scan
      // (above, which missed) and this push, p_irecv may have run —
      // it saw an empty `unmatched` and parked its recv in `posted`.
      // Without this second scan the frame and the posted recv would
      // each sit in their queue forever (observed as a rare multi-pair
      // hang under load).
      std::lock_guard<std::mutex> g(c->mu);
      bool late_match = false;
      for (auto it = c->posted.begin(); it != c->posted.end(); ++it) {
        if (it->tag == f.tag) {
          PostedRecv pr2 = *it;
          c->posted.erase(it);
          late_match = true;
          if (pr2.cap >= static_cast<int>(f.bytes)) {
            if (!place_payload(c, pr2, f.data.data(), f.bytes))
              pr2.req->error.store(1, std::memory_order_relaxed);
            pr2.req->sizes[pr2.slot] = static_cast<int>(f.bytes);
            pr2.req->complete_one();
          } else {
            PLOG(NCCL_LOG_WARN,
                 "uccl-net: recv overflow tag=%d %lu > %d (req failed)",
                 f.tag, (unsigned long)f.bytes, pr2.cap);
            pr2.req->error.store(1, std::memory_order_relaxed);
            pr2.req->complete_one();
          }
          break;
        }
      }
      if (!late_match) c->unmatched.push_back(std::move(f));
    }
  }
}

// ---------------------------------------------------------------------------
// vtable implementation
// ---------------------------------------------------------------------------

ncclResult_t p_init(ncclDebugLogger_t logfn) {
  g_log = logfn;
  PLOG(NCCL_LOG_INFO, "uccl-net plugin init (%s data plane)",
       use_multipath() ? "multipath" : "tcp");
  return ncclSuccess;
}

int num_vdevs() {
  static int n = [] {
    long v = uccl::env_int("UCCL_NET_NDEV", 1);
    return static_cast<int>(v < 1 ? 1 : (v > 4 ? 4 : v));
  }();
  return n;
}

ncclResult_t p_devices(int* ndev) {
  *ndev = num_vdevs();
  return ncclSuccess;
}

ncclResult_t p_getProperties(int dev, ncclNetProperties_v6_t* props) {
  static char names[4][8] = {"uccl0", "uccl1", "uccl2", "uccl3"};
  static char pci[] = "";
  if (dev < 0 || dev >= num_vdevs()) return ncclInvalidArgument;
  memset(props, 0, sizeof(*props));
  props->name = names[dev];
  props->pciPath = pci;
  props->guid = 0x75636331 + dev;
  props->ptrSupport = NCCL_PTR_HOST;
#ifdef UCCL_NET_HIP
  props->ptrSupport |= NCCL_PTR_CUDA;
#endif
  props->speed = 100000;
  props->port = 0;
  props->latency = 20.0f;
  props->maxComms = 65536;
  props->maxRecvs = 4;
  return ncclSuccess;
}

ncclResult_t p_listen(int dev, void* opaque, void** listenComm) {
  if (use_multipath()) {
    auto& f = MpFabric::get();
    auto* lc = new ListenComm();
    lc->nonce = f.next_nonce.fetch_add(1);
    {
      std::lock_guard<std::mutex> g(f.mu);
      f.queues[lc->nonce];  // create the routing queue
    }
    MpHandle h{};
    h.nonce = lc->nonce;
    std::string md = f.tp.metadata();
    h.md_len = static_cast<uint16_t>(md.size());
    memcpy(h.md, md.data(), std::min(md.size(), sizeof(h.md)));
    static_assert(sizeof(MpHandle) <= NCCL_NET_HANDLE_MAXSIZE, "handle");
    memcpy(opaque, &h, sizeof(h));
    *listenComm = lc;
    return ncclSuccess;
  }
  auto* lc = new ListenComm();
  uint16_t port = 0;
  lc->fd = uccl::net::listen_on(&port);
  int flags = fcntl(lc->fd, F_GETFL, 0);
  fcntl(lc->fd, F_SETFL, flags | O_NONBLOCK);
  Handle h{};
  std::string ip = uccl::net::local_ip();
  strncpy(h.ip, ip.c_str(), sizeof(h.ip) - 1);
  h.port = port;
  static_assert(sizeof(Handle) <= NCCL_NET_HANDLE_MAXSIZE, "handle size");
  memcpy(opaque, &h, sizeof(h));
  *listenComm = lc;
  return ncclSuccess;
}

ncclResult_t p_connect(int dev, void* opaque, void** sendComm) {
  if (use_multipath()) {
    MpHandle h{};
    memcpy(&h, opaque, sizeof(h));
    try {
      uint64_t flow = MpFabric::get().tp.connect(
          std::string(h.md, h.md_len), h.nonce);
      NET_DBG("connect flow=%llx nonce=%llu", (unsigned long long)flow,
              (unsigned long long)h.nonce);
      auto* c = new Comm();
      c->flow = flow;
      c->sender = true;
      c->worker = std::thread(tx_loop, c);
      *sendComm = c;
      return ncclSuccess;
    } catch (std::exception const&) {
      return ncclSystemError;
    }
  }
  Handle h{};
  memcpy(&h, opaque, sizeof(h));
  // non-blocking contract: attempt one quick connect; if not ready yet,
  // return sendComm=NULL so RCCL retries
  int fd = ::socket(AF_INET, SOCK_STREAM, 0);
  if (fd < 0) return ncclSystemError;
  sockaddr_in addr{};
  addr.sin_family = AF_INET;
  addr.sin_port = htons(h.port);
  inet_pton(AF_INET, h.ip, &addr.sin_addr);
  if (::connect(fd, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)) != 0) {
    ::close(fd);
    *sendComm = nullptr;  // retry later
    return ncclSuccess;
  }
  int one = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
  uint64_t cookie = kPluginCookie;
  send_all(fd, &cookie, sizeof(cookie));
  auto* c = new Comm();
  c->fd = fd;
  c->sender = true;
  c->worker = std::thread(tx_loop, c);
  *sendComm = c;
  return ncclSuccess;
}

ncclResult_t p_accept(void* listenComm, void** recvComm) {
  auto* lc = static_cast<ListenComm*>(listenComm);
  if (use_multipath()) {
    auto& f = MpFabric::get();
    uint64_t flow = 0;
    {
      std::lock_guard<std::mutex> g(f.mu);
      auto& q = f.queues[lc->nonce];
      if (q.empty()) {
        *recvComm = nullptr;  // not ready; RCCL retries
        return ncclSuccess;
      }
      flow = q.front();
      q.pop_front();
    }
    auto* c = new Comm();
    c->flow = flow;
    c->worker = std::thread(rx_loop, c);
    *recvComm = c;
    return ncclSuccess;
  }
  int fd = ::accept(lc->fd, nullptr, nullptr);
  if (fd < 0) {
    *recvComm = nullptr;  // not ready; RCCL retries
    return ncclSuccess;
  }
  int one = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
  // validate the cookie (bounded wait) — drop stray connections
  timeval tv{2, 0};
  setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
  uint64_t cookie = 0;
  if (!recv_all(fd, &cookie, sizeof(cookie)) || cookie != kPluginCookie) {
    ::close(fd);
    *recvComm = nullptr;
    return ncclSuccess;
  }
  timeval tv0{0, 0};
  setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv0, sizeof(tv0));
  auto* c = new Comm();
  c->fd = fd;
  c->worker = std::thread(rx_loop, c);
  *recvComm = c;
  return ncclSuccess;
}

ncclResult_t p_regMr(void* comm, void* data, int size, int type,
                     void** mhandle) {
  (void)comm;
  (void)size;
  if (type == NCCL_PTR_HOST) {
    *mhandle = nullptr;  // null handle = host memory
    return ncclSuccess;
  }
#ifdef UCCL_NET_HIP
  if (type == NCCL_PTR_CUDA) {
    auto* mr = new Mr();
    mr->type = NCCL_PTR_CUDA;
    hipPointerAttribute_t attr{};
    if (hipPointerGetAttributes(&attr, data) == hipSuccess)
      mr->device = attr.device;
    *mhandle = mr;
    return ncclSuccess;
  }
#endif
  (void)data;
  return ncclInternalError;
}

ncclResult_t p_regMrDmaBuf(void*, void*, size_t, int, uint64_t, int,
                           void**) {
  return ncclInternalError;  // no DMA engine in this fabric
}

ncclResult_t p_deregMr(void*, void* mhandle) {
  delete static_cast<Mr*>(mhandle);  // null-safe (host MRs)
  return ncclSuccess;
}

Mr mr_of(void* mhandle) {
  return mhandle ? *static_cast<Mr*>(mhandle) : Mr{};
}

ncclResult_t p_isend(void* sendComm, void* data, int size, int tag,
                     void* mhandle, void** request) {
  auto* c = static_cast<Comm*>(sendComm);
  auto* r = new Request();
  {
    std::lock_guard<std::mutex> g(c->mu);
    c->sendq.push_back(SendOp{data, size, tag, r, mr_of(mhandle)});
  }
  c->cv.notify_one();
  *request = r;
  return ncclSuccess;
}

ncclResult_t p_irecv(void* recvComm, int n, void** data, int* sizes,
                     int* tags, void** mhandles, void** request) {
  if (n < 1 || n > 8) return ncclInternalError;
  auto* c = static_cast<Comm*>(recvComm);
  auto* r = new Request();
  r->recv = true;
  r->n = n;
  r->pending.store(n, std::memory_order_relaxed);
  {
    std::lock_guard<std::mutex> g(c->mu);
    for (int i = 0; i < n; ++i) {
      Mr const mr = mr_of(mhandles ? mhandles[i] : nullptr);
      // match an already-arrived frame first
      bool hit = false;
      for (auto it = c->unmatched.begin(); it != c->unmatched.end(); ++it) {
        if (it->tag == tags[i]) {
          hit = true;
          if (static_cast<int>(it->bytes) > sizes[i]) {
            r->error.store(1, std::memory_order_relaxed);
          } else {
            PostedRecv pr{data[i], sizes[i], tags[i], r, i, mr};
            if (!place_payload(c, pr, it->data.data(), it->bytes))
              r->error.store(1, std::memory_order_relaxed);
            r->sizes[i] = static_cast<int>(it->bytes);
          }
          r->complete_one();
          c->unmatched.erase(it);
          break;
        }
      }
      if (!hit)
        c->posted.push_back(PostedRecv{data[i], sizes[i], tags[i], r, i, mr});
    }
  }
  *request = r;
  return ncclSuccess;
}

ncclResult_t p_iflush(void*, int, void**, int*, void**, void** request) {
  // recv completion is only reported after the (synchronous) HtoD staging
  // copy, so there is never un-flushed GPU data to wait on
  *request = nullptr;
  return ncclSuccess;
}

ncclResult_t p_test(void* request, int* done, int* sizes) {
  auto* r = static_cast<Request*>(request);
  if (r->done()) {
    if (r->error.load(std::memory_order_relaxed)) {
      delete r;
      return ncclInternalError;
    }
    *done = 1;
    if (sizes)
      for (int i = 0; i < r->n; ++i) sizes[i] = r->sizes[i];
    delete r;
  } else {
    *done = 0;
  }
  return ncclSuccess;
}

ncclResult_t p_closeSend(void* comm) {
  delete static_cast<Comm*>(comm);
  return ncclSuccess;
}
ncclResult_t p_closeRecv(void* comm) {
  delete static_cast<Comm*>(comm);
  return ncclSuccess;
}
ncclResult_t p_closeListen(void* comm) {
  auto* lc = static_cast<ListenComm*>(comm);
  if (lc->fd >= 0) ::close(lc->fd);
  delete lc;
  return ncclSuccess;
}

}  // namespace

extern "C" {
__attribute__((visibility("default"))) ncclNet_v6_t ncclNetPlugin_v6 = {
    "uccl",       p_init,     p_devices,   p_getProperties,
    p_listen,     p_connect,  p_accept,    p_regMr,
    p_regMrDmaBuf, p_deregMr, p_isend,     p_irecv,
    p_iflush,     p_test,     p_closeSend, p_closeRecv,
    p_closeListen,
};
}
