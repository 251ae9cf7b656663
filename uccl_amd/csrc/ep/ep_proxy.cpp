#include "ep_proxy.h"

#include <pthread.h>
#include <sched.h>
#include <unistd.h>

#include <chrono>

#include <cstring>
#include <thread>

#include "../core/env.h"
#include "../core/log.h"

namespace uccl {
namespace ep {

namespace {
constexpr uint32_t kDisp = 1, kComb = 2, kCombDone = 3, kBar = 4, kAtom = 5,
                   kCons = 6, kDispB = 7;
constexpr size_t kStageBytes = 16ull << 20;  // pinned staging chunk

// CPU pinning for proxy threads (parity: the reference pins proxies per
// NUMA node / NIC, ep/src/proxy.cpp:168-214). Without libnuma in this
// image, pin to explicit cores: UCCL_EP_PROXY_CORES="c0,c1,.." or
// default to the HIGHEST cores (kernels/apps start filling from 0).
void pin_proxy_thread(int slot) {
  std::string const spec = env_str("UCCL_EP_PROXY_CORES", "");
  int core = -1;
  if (!spec.empty()) {
    int idx = 0;
    size_t pos = 0;
    while (pos < spec.size()) {
      size_t nxt = spec.find(',', pos);
      if (nxt == std::string::npos) nxt = spec.size();
      if (idx == slot) {
        core = atoi(spec.substr(pos, nxt - pos).c_str());
        break;
      }
      pos = nxt + 1;
      ++idx;
    }
  } else {
    int const hw = static_cast<int>(std::thread::hardware_concurrency());
    if (hw >= 16) core = hw - 1 - (slot % (hw / 4));
  }
  if (core < 0) return;
  cpu_set_t set;
  CPU_ZERO(&set);
  CPU_SET(core, &set);
  (void)pthread_setaffinity_np(pthread_self(), sizeof(set), &set);
}
}  // namespace

int EpProxy::flow_peer(uint64_t flow) const {
  for (int r = 0; r < v_.world; ++r)
    if (flows_[r] == flow) return r;
  return 0;
}

// A flow mutex held for seconds means another lane is wedged mid-ship;
// log instead of blocking invisibly (a silent block here produced
// "last expert never shipped" stalls that were unattributable).
// hipStreamSynchronize that logs instead of blocking invisibly: a lane
// sync stuck for seconds means the copy engine / device is wedged.
static void lane_sync(hipStream_t stream, char const* who) {
  auto const t0 = std::chrono::steady_clock::now();
  for (;;) {
    hipError_t const e = hipStreamQuery(stream);
    if (e == hipSuccess) return;
    if (e != hipErrorNotReady) {
      UCCL_CHECK_HIP(e);
      return;
    }
    auto const el = std::chrono::duration<double>(
                        std::chrono::steady_clock::now() - t0)
                        .count();
    if (el > 5.0) {
      UCCL_LOG_WARN << "ep proxy lane sync stalled " << el << "s (" << who
                    << ")";
      (void)hipStreamSynchronize(stream);
      return;
    }
    usleep(50);
  }
}

struct TimedFlowLock {
  std::timed_mutex& m;
  TimedFlowLock(std::timed_mutex& mu, char const* who) : m(mu) {
    while (!m.try_lock_for(std::chrono::seconds(5)))
      UCCL_LOG_WARN << "ep proxy flow mutex contended 5s (" << who << ")";
  }
  ~TimedFlowLock() { m.unlock(); }
};

EpProxy::Lane::Lane(int device, size_t bytes) {
  UCCL_CHECK_HIP(hipSetDevice(device));
  UCCL_CHECK_HIP(hipStreamCreateWithFlags(&stream, hipStreamNonBlocking));
  UCCL_CHECK_HIP(hipHostMalloc(&buf, bytes));
  UCCL_CHECK_HIP(hipHostMalloc(&buf2, bytes));
}

EpProxy::Lane::~Lane() {
  if (buf) (void)hipHostFree(buf);
  if (buf2) (void)hipHostFree(buf2);
  if (stream) (void)hipStreamDestroy(stream);
}

EpProxy::EpProxy(const EpView& view, void* heap, D2HRing* ring_host,
                 int device)
    : v_(view), heap_(heap), ring_(ring_host), device_(device) {
  tp_ = std::make_unique<transport::TransportEndpoint>(
      static_cast<int>(env_int("UCCL_EP_PROXY_PATHS", 8)),
      static_cast<size_t>(env_int("UCCL_EP_PROXY_CHUNK", 16384)));
  flows_.resize(v_.world, 0);
  for (int r = 0; r < v_.world; ++r)
    flow_mu_.emplace_back(new std::timed_mutex());
  stage_bytes_ = kStageBytes;
}

EpProxy::~EpProxy() {
  // DRAIN before shutdown: a rank may destroy its buffer as soon as ITS
  // combine completed, while peers still await the returns this proxy is
  // shipping asynchronously (comb_tx_loop) or dispatch rows still queued
  // in the D2H ring. Killing the transport mid-ship strands those peers
  // in their device waits (observed at 256-expert forced-proxy: the
  // faster rank's close() trapped the slower rank).
  {
    std::unique_lock<std::mutex> lk(mu_);
    cv_.wait_for(lk, std::chrono::seconds(20),
                 [this] { return comb_q_.empty() && !comb_busy_; });
  }
  if (ring_) {
    for (int i = 0; i < 20000; ++i) {  // up to ~20s
      uint64_t const h = __atomic_load_n(
          const_cast<uint64_t*>(&ring_->head), __ATOMIC_ACQUIRE);
      uint64_t const t = __atomic_load_n(
          const_cast<uint64_t*>(&ring_->tail), __ATOMIC_ACQUIRE);
      if (h == t) break;
      usleep(1000);
    }
  }
  {
    std::lock_guard<std::mutex> g(mu_);  // lost-wakeup guard
    stop_ = true;
  }
  cv_.notify_all();
  if (tp_) tp_->shutdown();
  if (ring_thread_.joinable()) ring_thread_.join();
  if (comb_thread_.joinable()) comb_thread_.join();
  for (auto& t : rx_threads_)
    if (t.joinable()) t.join();
  tp_.reset();
}

void EpProxy::establish_flows(const std::vector<std::string>& tp_md,
                              uint32_t proxy_mask) {
  proxy_mask_ = proxy_mask;
  // higher rank connects to lower rank; lower rank accepts and matches by
  // the connector's tag (= its rank)
  int expected_accepts = 0;
  for (int r = 0; r < v_.world; ++r) {
    if (!((proxy_mask_ >> r) & 1u)) continue;
    if (r < v_.rank) {
      flows_[r] = tp_->connect(tp_md[r], static_cast<uint64_t>(v_.rank));
    } else {
      ++expected_accepts;
    }
  }
  for (int i = 0; i < expected_accepts; ++i) {
    uint64_t tag = 0;
    uint64_t flow = tp_->accept(&tag);
    UCCL_CHECK(tag < static_cast<uint64_t>(v_.world)) << "bad flow tag";
    flows_[tag] = flow;
  }
}

void EpProxy::start() {
  ring_thread_ = std::thread([this] { ring_loop(); });
  comb_thread_ = std::thread([this] { comb_tx_loop(); });
  for (int r = 0; r < v_.world; ++r)
    if ((proxy_mask_ >> r) & 1u)
      rx_threads_.emplace_back([this, r] { rx_loop(r); });
}

// D2H-copy `count` rows from dev_rows and stream them over `flow` in
// staging-sized chunks, then the metas.
void EpProxy::ship_rows(Lane& lane, uint64_t flow, WireHdr const& h,
                        void const* dev_rows,
                        uint32_t const* dev_metas_or_null,
                        std::vector<uint32_t> const* host_metas,
                        size_t row_bytes, void const* dev_scales,
                        size_t scale_row_bytes) {
  TimedFlowLock guard(*flow_mu_[flow_peer(flow)], "ship_rows");
  if (!row_bytes) row_bytes = static_cast<size_t>(v_.hidden) * v_.elem_size;
  size_t const total = h.count * row_bytes;
  // control messages post async (acked while the payload streams);
  // flush_sends() before return bounds every lifetime to this call
  tp_->send_msg_async(flow, &h, sizeof(h));
  // double-buffered pipeline: the D2H copy of chunk i+1 runs while the
  // transport ships chunk i (the reference's batched-posting role,
  // proxy.cpp:1203 post_gpu_commands_mixed)
  void* bufs[2] = {lane.buf, lane.buf2};
  size_t off = 0;
  int cur = 0;
  if (total) {
    size_t const n0 = std::min(stage_bytes_, total);
    UCCL_CHECK_HIP(hipMemcpyAsync(bufs[cur], dev_rows, n0,
                                  hipMemcpyDeviceToHost, lane.stream));
    lane_sync(lane.stream, "ship first d2h");
  }
  while (off < total) {
    size_t const n = std::min(stage_bytes_, total - off);
    size_t const next_off = off + n;
    if (next_off < total) {
      size_t const n1 = std::min(stage_bytes_, total - next_off);
      UCCL_CHECK_HIP(hipMemcpyAsync(
          bufs[cur ^ 1], static_cast<char const*>(dev_rows) + next_off, n1,
          hipMemcpyDeviceToHost, lane.stream));
    }
    tp_->send_msg(flow, bufs[cur], n);
    if (next_off < total) lane_sync(lane.stream, "ship next d2h");
    off = next_off;
    cur ^= 1;
  }
  if (dev_scales && h.count) {
    // fp8 egress: second contiguous stream of per-row scales
    size_t const stotal = h.count * scale_row_bytes;
    size_t soff = 0;
    while (soff < stotal) {
      size_t const n = std::min(stage_bytes_, stotal - soff);
      UCCL_CHECK_HIP(hipMemcpyAsync(
          lane.buf, static_cast<char const*>(dev_scales) + soff, n,
          hipMemcpyDeviceToHost, lane.stream));
      lane_sync(lane.stream, "ship scales d2h");
      tp_->send_msg(flow, lane.buf, n);
      soff += n;
    }
  }
  std::vector<uint32_t> metas;
  if (host_metas) {
    metas = *host_metas;
  } else {
    metas.resize(h.count);
    UCCL_CHECK_HIP(hipMemcpyAsync(metas.data(), dev_metas_or_null,
                                  h.count * sizeof(uint32_t),
                                  hipMemcpyDeviceToHost, lane.stream));
    lane_sync(lane.stream, "ship metas d2h");
  }
  if (h.count)
    tp_->send_msg_async(flow, metas.data(), h.count * sizeof(uint32_t));
  tp_->flush_sends(flow);
}

// Batched dispatch shipping: one wire transaction for a contiguous run
// of per-expert egress spans headed to the same peer (hdr, (le,count)
// pairs, one chunked row stream, one meta block) instead of
// 3*n_experts blocking sends. This is the latency lever for many-expert
// shapes: per-message round trips dominated the 256-expert generation
// time before batching.
void EpProxy::ship_batch(Lane& lane, uint64_t flow, uint32_t seq,
                         uint64_t row0, std::vector<uint32_t> const& les,
                         std::vector<uint32_t> const& cnts,
                         uint64_t total_rows) {
  TimedFlowLock guard(*flow_mu_[flow_peer(flow)], "ship_batch");
  size_t const row_bytes =
      v_.disp_fp8 ? static_cast<size_t>(v_.hidden)
                  : static_cast<size_t>(v_.hidden) * v_.elem_size;
  size_t const total = total_rows * row_bytes;
  WireHdr h{kDispB, seq, les[0], static_cast<uint32_t>(v_.rank),
            total_rows, les.size()};
  tp_->send_msg_async(flow, &h, sizeof(h));
  std::vector<uint32_t> pairs(2 * les.size());
  for (size_t i = 0; i < les.size(); ++i) {
    pairs[2 * i] = les[i];
    pairs[2 * i + 1] = cnts[i];
  }
  tp_->send_msg_async(flow, pairs.data(), pairs.size() * sizeof(uint32_t));
  char const* dev_rows = v_.disp_fp8
                             ? egress_x_fp8(heap_, v_, row0)
                             : egress_row(heap_, v_, row0);
  void* bufs[2] = {lane.buf, lane.buf2};
  size_t off = 0;
  int cur = 0;
  if (total) {
    size_t const n0 = std::min(stage_bytes_, total);
    UCCL_CHECK_HIP(hipMemcpyAsync(bufs[cur], dev_rows, n0,
                                  hipMemcpyDeviceToHost, lane.stream));
    lane_sync(lane.stream, "batch first d2h");
  }
  while (off < total) {
    size_t const n = std::min(stage_bytes_, total - off);
    size_t const next_off = off + n;
    if (next_off < total) {
      size_t const n1 = std::min(stage_bytes_, total - next_off);
      UCCL_CHECK_HIP(hipMemcpyAsync(bufs[cur ^ 1], dev_rows + next_off, n1,
                                    hipMemcpyDeviceToHost, lane.stream));
    }
    tp_->send_msg(flow, bufs[cur], n);
    if (next_off < total) lane_sync(lane.stream, "batch next d2h");
    off = next_off;
    cur ^= 1;
  }
  if (v_.disp_fp8 && total_rows) {
    size_t const srow = (static_cast<size_t>(v_.hidden) / 128) * 4;
    size_t const stotal = total_rows * srow;
    char const* dev_scales = reinterpret_cast<char const*>(
        egress_scale_fp8(heap_, v_, row0));
    size_t soff = 0;
    while (soff < stotal) {
      size_t const n = std::min(stage_bytes_, stotal - soff);
      UCCL_CHECK_HIP(hipMemcpyAsync(lane.buf, dev_scales + soff, n,
                                    hipMemcpyDeviceToHost, lane.stream));
      lane_sync(lane.stream, "batch scales d2h");
      tp_->send_msg(flow, lane.buf, n);
      soff += n;
    }
  }
  if (total_rows) {
    std::vector<uint32_t> metas(total_rows);
    UCCL_CHECK_HIP(hipMemcpyAsync(metas.data(),
                                  egress_meta(heap_, v_, row0),
                                  total_rows * sizeof(uint32_t),
                                  hipMemcpyDeviceToHost, lane.stream));
    lane_sync(lane.stream, "batch metas d2h");
    tp_->send_msg_async(flow, metas.data(), total_rows * sizeof(uint32_t));
  }
  tp_->flush_sends(flow);
}

int EpProxy::num_proxy_peers() const {
  int n = 0;
  for (int r = 0; r < v_.world; ++r)
    if ((proxy_mask_ >> r) & 1u) ++n;
  return n;
}

void EpProxy::write_sync_flag(Lane& lane, int idx, uint64_t seq) {
  UCCL_CHECK_HIP(hipMemcpyAsync(sync_ptr(heap_, v_, idx), &seq, sizeof(seq),
                                hipMemcpyHostToDevice, lane.stream));
  UCCL_CHECK_HIP(hipStreamSynchronize(lane.stream));
}

// one arrival (self or a peer's kBar message) for barrier `seq`; when
// self + every proxy peer have arrived, publish the device flag
void EpProxy::handle_barrier_arrival(Lane& lane, uint64_t seq) {
  bool complete = false;
  {
    std::lock_guard<std::mutex> g(bar_mu_);
    int const need = 1 + num_proxy_peers();
    if (++bar_seen_[seq] >= need) {
      bar_seen_.erase(seq);
      complete = true;
    }
  }
  if (complete) write_sync_flag(lane, 0, seq);
}

void EpProxy::ring_loop() {
  (void)hipSetDevice(device_);
  pin_proxy_thread(0);
  Lane lane(device_, stage_bytes_);
  uint64_t head = 0;
  uint64_t idle_us = 0;
  while (!stop_) {
    uint64_t tail =
        __atomic_load_n(const_cast<uint64_t*>(&ring_->tail),
                        __ATOMIC_ACQUIRE);
    if (head == tail) {
      usleep(20);
      // idle diagnostic: if a peer later reports missing tail entries,
      // "ring idle at H" here proves the GPU publish never pushed them
      // (vs the loop being wedged processing one)
      idle_us += 20;
      if (idle_us >= 5'000'000) {
        idle_us = 0;
        static bool const dbg = env_bool("UCCL_TP_DEBUG_RTO", false);
        if (dbg)
          UCCL_LOG_WARN << "ep proxy ring idle: head=tail=" << head;
      }
      continue;
    }
    idle_us = 0;
    while (head != tail) {
      TransferCmd c;
      memcpy(&c, const_cast<TransferCmd*>(&ring_->cmds[head % kRingSlots]),
             sizeof(c));
      auto const t0 = std::chrono::steady_clock::now();
      try {
        switch (static_cast<CmdOp>(c.op)) {
          case CmdOp::kDispatchWrite: {
            int const e = static_cast<int>(c.a);
            int const dst = e / v_.local_experts;
            // BATCH: fold the maximal run of queued dispatch cmds for
            // the same (dst, seq) with contiguous egress spans into one
            // wire transaction — the reference proxy's batched posting
            // (proxy.cpp:1203 post_gpu_commands_mixed). The publish
            // kernel emits experts in pfx order, so a generation's
            // cmds per dst are contiguous and this collapses
            // 3*local_experts blocking sends into ~3.
            std::vector<uint32_t> les{static_cast<uint32_t>(
                e % v_.local_experts)};
            std::vector<uint32_t> cnts{static_cast<uint32_t>(c.c)};
            uint64_t row0 = c.b;
            uint64_t rows = c.c;
            uint64_t scan = head + 1;
            while (scan != tail) {
              TransferCmd n;
              memcpy(&n,
                     const_cast<TransferCmd*>(
                         &ring_->cmds[scan % kRingSlots]),
                     sizeof(n));
              if (static_cast<CmdOp>(n.op) != CmdOp::kDispatchWrite ||
                  n.seq32 != c.seq32 ||
                  static_cast<int>(n.a) / v_.local_experts != dst ||
                  n.b != row0 + rows)
                break;
              les.push_back(
                  static_cast<uint32_t>(n.a % v_.local_experts));
              cnts.push_back(static_cast<uint32_t>(n.c));
              rows += n.c;
              ++scan;
            }
            if (les.size() > 1) {
              ship_batch(lane, flows_[dst], c.seq32, row0, les, cnts,
                         rows);
              head = scan - 1;  // ++head below completes the batch
            } else {
              WireHdr h{kDisp, c.seq32, les[0],
                        static_cast<uint32_t>(v_.rank), c.c, 0};
              if (v_.disp_fp8)
                ship_rows(lane, flows_[dst], h,
                          egress_x_fp8(heap_, v_, c.b),
                          egress_meta(heap_, v_, c.b), nullptr,
                          static_cast<size_t>(v_.hidden),
                          egress_scale_fp8(heap_, v_, c.b),
                          (static_cast<size_t>(v_.hidden) / 128) * 4);
              else
                ship_rows(lane, flows_[dst], h, egress_row(heap_, v_, c.b),
                          egress_meta(heap_, v_, c.b), nullptr);
            }
            break;
          }
          case CmdOp::kAtomicAdd: {
            // a = dst rank, b = heap byte offset, c = value
            int const dst = static_cast<int>(c.a);
            WireHdr h{kAtom, c.seq32, 0, static_cast<uint32_t>(v_.rank),
                      c.c, c.b};
            TimedFlowLock guard(*flow_mu_[dst], "atomic");
            tp_->send_msg(flows_[dst], &h, sizeof(h));
            break;
          }
          case CmdOp::kBarrier: {
            // announce to every proxy peer, count self
            uint64_t const seq = c.a;
            WireHdr h{kBar, static_cast<uint32_t>(seq), 0,
                      static_cast<uint32_t>(v_.rank), seq, 0};
            for (int r = 0; r < v_.world; ++r) {
              if (!((proxy_mask_ >> r) & 1u)) continue;
              TimedFlowLock guard(*flow_mu_[r], "barrier");
              tp_->send_msg_async(flows_[r], &h, sizeof(h));
            }
            for (int r = 0; r < v_.world; ++r)
              if ((proxy_mask_ >> r) & 1u) tp_->flush_sends(flows_[r]);
            handle_barrier_arrival(lane, seq);
            break;
          }
          case CmdOp::kConsume: {
            // relay "generation a consumed" to every proxied peer so
            // their next dispatch_send's consume gate can open. Posted
            // async to ALL peers, then flushed once — a serial blocking
            // send per peer paid one transport round trip each.
            WireHdr h{kCons, c.seq32, 0, static_cast<uint32_t>(v_.rank),
                      c.a, 0};
            for (int r = 0; r < v_.world; ++r) {
              if (!((proxy_mask_ >> r) & 1u)) continue;
              TimedFlowLock guard(*flow_mu_[r], "consume");
              tp_->send_msg_async(flows_[r], &h, sizeof(h));
            }
            for (int r = 0; r < v_.world; ++r)
              if ((proxy_mask_ >> r) & 1u) tp_->flush_sends(flows_[r]);
            break;
          }
          case CmdOp::kQuiet: {
            // ring cmds execute in FIFO order and ship_rows is
            // synchronous, so reaching this cmd means every prior
            // transfer has been fully handed to the reliable transport
            // (the reference's quiet_cq role, proxy.cpp:1341)
            write_sync_flag(lane, 1, c.a);
            break;
          }
          default:
            break;
        }
      } catch (std::exception const& e) {
        if (!stop_)
          UCCL_LOG_ERROR << "ep proxy ring_loop died: " << e.what();
        return;
      }
      {
        // slow-command diagnostic: ship_rows should complete in ms; a
        // multi-second command pins down WHICH ring entry wedged when a
        // peer later times out waiting for the generation's tail
        auto const t1 = std::chrono::steady_clock::now();
        double const s = std::chrono::duration<double>(t1 - t0).count();
        if (s > 3.0)
          UCCL_LOG_WARN << "ep proxy ring cmd slow: op " << c.op << " seq "
                        << c.seq32 << " a " << c.a << " took " << s << "s";
      }
      // publish head ONLY after the command is fully shipped: the GPU
      // gates the next dispatch's egress reuse on head==tail
      // (k_ep_ring_wait_empty), so early publication would let the copy
      // kernel clobber rows mid-ship (observed as corrupt meta ->
      // out-of-bounds combine writes)
      ++head;
      __atomic_store_n(const_cast<uint64_t*>(&ring_->head), head,
                       __ATOMIC_RELEASE);
    }
  }
}

void EpProxy::comb_tx_loop() {
  (void)hipSetDevice(device_);
  pin_proxy_thread(1);
  Lane lane(device_, stage_bytes_);
  while (true) {
    CombTask task;
    {
      std::unique_lock<std::mutex> lk(mu_);
      cv_.wait(lk, [this] { return !comb_q_.empty() || stop_; });
      if (stop_ && comb_q_.empty()) return;
      task = std::move(comb_q_.front());
      comb_q_.pop_front();
      comb_busy_ = true;
    }
    (void)hipEventSynchronize(task.ready);
    size_t const row_bytes = static_cast<size_t>(v_.hidden) * v_.elem_size;
    try {
      for (int src = 0; src < v_.world; ++src) {
        if (!((proxy_mask_ >> src) & 1u)) continue;
        // GATHER-BATCH: per-expert return blocks are strided in
        // expert_out, so D2H-gather groups of them into pinned staging
        // and ship each group as ONE wire message (scatter at the
        // receiver is meta-driven, so it needs no per-expert framing).
        // Collapses 3*local_experts blocking sends per peer into
        // ~2*ceil(rows/groupcap) — the combine half of the reference's
        // batched posting.
        size_t const group_rows_cap = std::min(
            static_cast<size_t>(v_.max_tokens),  // rx ingress slice
            stage_bytes_ / row_bytes);
        size_t const slot0 = static_cast<size_t>(src) * v_.max_tokens;
        int le = 0;
        while (le < v_.local_experts) {
          {
            // an expert block too large for one staging group ships
            // alone through the chunked per-expert path
            size_t const cnt0 = static_cast<size_t>(
                task.counts[static_cast<size_t>(le) * v_.world + src]);
            if (cnt0 > group_rows_cap) {
              WireHdr h{kComb, static_cast<uint32_t>(task.seq),
                        static_cast<uint32_t>(le),
                        static_cast<uint32_t>(v_.rank), cnt0};
              char const* blk =
                  static_cast<char const*>(task.expert_out) +
                  (static_cast<size_t>(le) * v_.world * v_.max_tokens +
                   slot0) *
                      row_bytes;
              ship_rows(lane, flows_[src], h, blk,
                        disp_meta_ptr(heap_, v_, le, slot0), nullptr);
              ++le;
              continue;
            }
          }
          size_t rows = 0;
          std::vector<uint32_t> metas;
          metas.reserve(group_rows_cap);  // no realloc under async D2H
          int const le0 = le;
          while (le < v_.local_experts) {
            size_t const cnt = static_cast<size_t>(
                task.counts[static_cast<size_t>(le) * v_.world + src]);
            if (rows + cnt > group_rows_cap) break;
            if (cnt) {
              char const* blk =
                  static_cast<char const*>(task.expert_out) +
                  (static_cast<size_t>(le) * v_.world * v_.max_tokens +
                   slot0) *
                      row_bytes;
              UCCL_CHECK_HIP(hipMemcpyAsync(
                  static_cast<char*>(lane.buf) + rows * row_bytes, blk,
                  cnt * row_bytes, hipMemcpyDeviceToHost, lane.stream));
              size_t const m0 = metas.size();
              metas.resize(m0 + cnt);
              UCCL_CHECK_HIP(hipMemcpyAsync(
                  metas.data() + m0, disp_meta_ptr(heap_, v_, le, slot0),
                  cnt * sizeof(uint32_t), hipMemcpyDeviceToHost,
                  lane.stream));
              rows += cnt;
            }
            ++le;
          }
          if (!rows) continue;  // all-empty group: nothing to carry
          lane_sync(lane.stream, "comb gather d2h");
          WireHdr h{kComb, static_cast<uint32_t>(task.seq),
                    static_cast<uint32_t>(le0),
                    static_cast<uint32_t>(v_.rank), rows};
          TimedFlowLock guard(*flow_mu_[src], "comb_batch");
          tp_->send_msg_async(flows_[src], &h, sizeof(h));
          tp_->send_msg_async(flows_[src], lane.buf, rows * row_bytes);
          tp_->send_msg_async(flows_[src], metas.data(),
                              rows * sizeof(uint32_t));
          // flush bounds h/metas/lane.buf lifetimes before group reuse
          tp_->flush_sends(flows_[src]);
        }
        WireHdr done{kCombDone, static_cast<uint32_t>(task.seq), 0,
                     static_cast<uint32_t>(v_.rank), 0};
        {
          TimedFlowLock guard(*flow_mu_[src], "comb_done");
          tp_->send_msg(flows_[src], &done, sizeof(done));
        }
      }
    } catch (std::exception const& e) {
      if (!stop_)
        UCCL_LOG_ERROR << "ep proxy comb_tx_loop died: " << e.what();
      {
        std::lock_guard<std::mutex> g(mu_);
        comb_busy_ = false;
      }
      cv_.notify_all();
      return;
    }
    (void)hipEventDestroy(task.ready);
    {
      std::lock_guard<std::mutex> g(mu_);
      comb_busy_ = false;
    }
    cv_.notify_all();
  }
}

void EpProxy::enqueue_combine(void const* expert_out, uint64_t seq,
                              hipEvent_t ready, int const* counts) {
  CombTask t;
  t.expert_out = expert_out;
  t.seq = seq;
  t.ready = ready;
  t.counts = counts;
  {
    std::lock_guard<std::mutex> g(mu_);
    comb_q_.push_back(std::move(t));
  }
  cv_.notify_one();
}

// scatter kernel for combine ingress (defined in ep_kernels.hip)
void launch_ep_comb_scatter(const EpView& v, size_t row0, size_t count,
                            hipStream_t s);

void EpProxy::rx_loop(int peer) {
  (void)hipSetDevice(device_);
  pin_proxy_thread(2 + peer);
  Lane lane(device_, stage_bytes_);
  uint64_t const flow = flows_[peer];
  size_t const row_bytes = static_cast<size_t>(v_.hidden) * v_.elem_size;
  std::vector<uint32_t> metas;
  try {
    while (!stop_) {
      WireHdr h{};
      tp_->recv_msg(flow, &h, sizeof(h));
      if (h.kind == kBar) {
        handle_barrier_arrival(lane, h.count);
        continue;
      }
      if (h.kind == kCons) {
        // peer consumed generation h.count of OUR egress: open our next
        // dispatch_send's consume gate for that peer
        uint64_t const seq = h.count;
        UCCL_CHECK_HIP(hipMemcpyAsync(consumed_ptr(heap_, v_, peer), &seq,
                                      sizeof(seq), hipMemcpyHostToDevice,
                                      lane.stream));
        lane_sync(lane.stream, "consume h2d");
        continue;
      }
      if (h.kind == kAtom) {
        // serialize RMW on the heap word (rx threads are the only
        // writers of these offsets; the mutex covers multi-peer adds)
        std::lock_guard<std::mutex> g(atomic_mu_);
        uint64_t cur = 0;
        char* addr = static_cast<char*>(heap_) + h.aux;
        UCCL_CHECK_HIP(hipMemcpyAsync(&cur, addr, sizeof(cur),
                                      hipMemcpyDeviceToHost, lane.stream));
        UCCL_CHECK_HIP(hipStreamSynchronize(lane.stream));
        cur += h.count;
        UCCL_CHECK_HIP(hipMemcpyAsync(addr, &cur, sizeof(cur),
                                      hipMemcpyHostToDevice, lane.stream));
        UCCL_CHECK_HIP(hipStreamSynchronize(lane.stream));
        continue;
      }
      if (h.kind == kCombDone) {
        // all combine payloads from `peer` for this seq have been
        // scattered (lane.stream is in-order); publish the flag
        UCCL_CHECK_HIP(hipStreamSynchronize(lane.stream));
        // NB: must NOT use the legacy default stream here — it would wait
        // for the peer's spinning wait-kernel (deadlock). lane.stream is
        // non-blocking.
        uint64_t const seq = h.seq32;
        UCCL_CHECK_HIP(hipMemcpyAsync(comb_flag_ptr(heap_, v_, peer), &seq,
                                      sizeof(seq), hipMemcpyHostToDevice,
                                      lane.stream));
        UCCL_CHECK_HIP(hipStreamSynchronize(lane.stream));
        continue;
      }
      if (h.kind == kDispB) {
        // batched dispatch: (le,count) pairs, then one contiguous row
        // stream split across the per-expert slot regions, then metas.
        // Every wire-controlled value is bounded before it indexes the
        // heap (same discipline as the transport's handle_data).
        size_t const nles = h.aux;
        UCCL_CHECK(nles >= 1 &&
                   nles <= static_cast<size_t>(v_.local_experts))
            << "dispatch batch nles " << nles;
        std::vector<uint32_t> pairs(2 * nles);
        tp_->recv_msg(flow, pairs.data(), pairs.size() * sizeof(uint32_t));
        // per-le destination spans and cumulative row boundaries
        std::vector<uint64_t> bound(nles + 1, 0);
        for (size_t i = 0; i < nles; ++i) {
          UCCL_CHECK(pairs[2 * i] <
                         static_cast<uint32_t>(v_.local_experts) &&
                     pairs[2 * i + 1] <=
                         static_cast<uint32_t>(v_.max_tokens))
              << "dispatch batch pair (" << pairs[2 * i] << ","
              << pairs[2 * i + 1] << ")";
          bound[i + 1] = bound[i] + pairs[2 * i + 1];
        }
        UCCL_CHECK(bound[nles] == h.count)
            << "dispatch batch rows " << h.count << " != " << bound[nles];
        // receive a contiguous stream of per-row records of size `rsz`
        // and split it across the per-expert destination spans
        auto recv_split = [&](size_t rsz, auto dst_of_le) {
          size_t const total = h.count * rsz;
          size_t off = 0;
          size_t li = 0;
          while (off < total) {
            size_t const n = std::min(stage_bytes_, total - off);
            tp_->recv_msg(flow, lane.buf, n);
            size_t done = 0;
            while (done < n) {
              size_t const gpos = off + done;
              while (li + 1 < nles && gpos >= bound[li + 1] * rsz) ++li;
              size_t const le_end = bound[li + 1] * rsz;
              size_t const span = std::min(n - done, le_end - gpos);
              char* dst = dst_of_le(pairs[2 * li]) +
                          (gpos - bound[li] * rsz);
              UCCL_CHECK_HIP(hipMemcpyAsync(
                  dst, static_cast<char*>(lane.buf) + done, span,
                  hipMemcpyHostToDevice, lane.stream));
              done += span;
            }
            lane_sync(lane.stream, "batch rows h2d");
            off += n;
          }
        };
        size_t const slot0b = static_cast<size_t>(h.src) * v_.max_tokens;
        size_t const rb =
            v_.disp_fp8 ? static_cast<size_t>(v_.hidden) : row_bytes;
        recv_split(rb, [&](uint32_t le) {
          return disp_x_ptr(heap_, v_, le, slot0b);
        });
        if (v_.disp_fp8)
          recv_split((static_cast<size_t>(v_.hidden) / 128) * 4,
                     [&](uint32_t le) {
                       return reinterpret_cast<char*>(
                           disp_scale_ptr(heap_, v_, le, slot0b));
                     });
        metas.resize(h.count);
        if (h.count)
          tp_->recv_msg(flow, metas.data(), h.count * sizeof(uint32_t));
        for (size_t i = 0; i < nles; ++i) {
          uint32_t const le = pairs[2 * i];
          uint64_t const cnt = pairs[2 * i + 1];
          if (cnt) {
            UCCL_CHECK_HIP(hipMemcpyAsync(
                disp_meta_ptr(heap_, v_, le,
                              static_cast<size_t>(h.src) * v_.max_tokens),
                metas.data() + bound[i], cnt * sizeof(uint32_t),
                hipMemcpyHostToDevice, lane.stream));
          }
          uint64_t const tagged =
              (static_cast<uint64_t>(h.seq32) << 32) | cnt;
          UCCL_CHECK_HIP(hipMemcpyAsync(
              disp_count_ptr(heap_, v_, le, h.src), &tagged,
              sizeof(tagged), hipMemcpyHostToDevice, lane.stream));
          lane_sync(lane.stream, "batch tag h2d");
        }
        continue;
      }
      bool const fp8_disp = v_.disp_fp8 && h.kind == kDisp;
      size_t const rb = fp8_disp ? static_cast<size_t>(v_.hidden)
                                 : row_bytes;
      size_t const total = h.count * rb;
      // destination in device memory
      size_t const ing0 = static_cast<size_t>(peer) * v_.max_tokens;
      char* dev_dst = (h.kind == kDisp)
                          ? disp_x_ptr(heap_, v_, h.le,
                                       static_cast<size_t>(h.src) *
                                           v_.max_tokens)
                          : ingress_row(heap_, v_, ing0);
      for (size_t off = 0; off < total; off += stage_bytes_) {
        size_t const n = std::min(stage_bytes_, total - off);
        tp_->recv_msg(flow, lane.buf, n);
        UCCL_CHECK_HIP(hipMemcpyAsync(dev_dst + off, lane.buf, n,
                                      hipMemcpyHostToDevice, lane.stream));
        UCCL_CHECK_HIP(hipStreamSynchronize(lane.stream));
      }
      if (fp8_disp && h.count) {
        // second stream: per-row scales into the contiguous scale span
        size_t const srow = (static_cast<size_t>(v_.hidden) / 128) * 4;
        size_t const stotal = h.count * srow;
        char* sdst = reinterpret_cast<char*>(disp_scale_ptr(
            heap_, v_, h.le, static_cast<size_t>(h.src) * v_.max_tokens));
        for (size_t off = 0; off < stotal; off += stage_bytes_) {
          size_t const n = std::min(stage_bytes_, stotal - off);
          tp_->recv_msg(flow, lane.buf, n);
          UCCL_CHECK_HIP(hipMemcpyAsync(sdst + off, lane.buf, n,
                                        hipMemcpyHostToDevice,
                                        lane.stream));
          lane_sync(lane.stream, "disp scales h2d");
        }
      }
      metas.resize(h.count);
      if (h.count)
        tp_->recv_msg(flow, metas.data(), h.count * sizeof(uint32_t));
      if (h.kind == kDisp) {
        UCCL_CHECK_HIP(hipMemcpyAsync(
            disp_meta_ptr(heap_, v_, h.le,
                          static_cast<size_t>(h.src) * v_.max_tokens),
            metas.data(), h.count * sizeof(uint32_t),
            hipMemcpyHostToDevice, lane.stream));
        UCCL_CHECK_HIP(hipStreamSynchronize(lane.stream));
        uint64_t const tagged =
            (static_cast<uint64_t>(h.seq32) << 32) | h.count;
        UCCL_CHECK_HIP(hipMemcpyAsync(disp_count_ptr(heap_, v_, h.le, h.src),
                                      &tagged, sizeof(tagged),
                                      hipMemcpyHostToDevice, lane.stream));
        UCCL_CHECK_HIP(hipStreamSynchronize(lane.stream));
      } else {  // kComb: scatter ingress rows into comb_x cells by meta
        if (h.count) {
          UCCL_CHECK_HIP(hipMemcpyAsync(ingress_meta(heap_, v_, ing0),
                                        metas.data(),
                                        h.count * sizeof(uint32_t),
                                        hipMemcpyHostToDevice, lane.stream));
          launch_ep_comb_scatter(v_, ing0, h.count, lane.stream);
          UCCL_CHECK_HIP(hipStreamSynchronize(lane.stream));
        }
      }
    }
  } catch (std::exception const& e) {
    if (!stop_)
      UCCL_LOG_ERROR << "ep proxy rx_loop(" << peer
                     << ") died: " << e.what();
  }
}

}  // namespace ep
}  // namespace uccl
