// Planner / lowering / spray executor (see ukernel.h for the design).

#include "ukernel.h"

#include "../core/trace.h"

#include <algorithm>
#include <cassert>
#include <chrono>
#include <condition_variable>
#include <cstring>
#include <deque>
#include <mutex>
#include <sstream>
#include <stdexcept>
#include <thread>

namespace uccl {
namespace uk {

namespace {
constexpr uint64_t kMaxFlags = 1u << 16;

char const* op_name(Op op) {
  switch (op) {
    case Op::kCopy: return "copy";
    case Op::kReduce: return "reduce";
    case Op::kPut: return "put";
    case Op::kSignal: return "signal";
    case Op::kWait: return "wait";
  }
  return "?";
}
}  // namespace

std::string ChunkGraph::dump() const {
  std::ostringstream os;
  for (size_t i = 0; i < tasks.size(); ++i) {
    auto const& t = tasks[i];
    os << i << ": " << op_name(t.op) << " rank" << t.rank;
    if (t.op == Op::kSignal || t.op == Op::kWait) {
      os << " flag=" << t.flag;
    } else {
      os << " r" << t.src.rank << ":s" << int(t.src.space) << "+"
         << t.src.offset << " -> r" << t.dst.rank << ":s" << int(t.dst.space)
         << "+" << t.dst.offset << " bytes=" << t.bytes;
    }
    if (!t.deps.empty()) {
      os << " deps[";
      for (size_t d = 0; d < t.deps.size(); ++d)
        os << (d ? "," : "") << t.deps[d];
      os << "]";
    }
    os << "\n";
  }
  return os.str();
}

// ---------------------------------------------------------------------------
// Planners
// ---------------------------------------------------------------------------

ChunkGraph plan_sendrecv_spray(Topology const& topo, int src, int dst,
                               uint64_t nbytes, uint64_t chunk_bytes) {
  if (src == dst) throw std::invalid_argument("src == dst");
  ChunkGraph g;
  g.world = topo.world;
  g.scratch_bytes = nbytes;  // relay staging, chunk-offset addressed

  // candidate paths: direct, then one relay per other rank; capacity of a
  // relay path is its bottleneck link
  struct Path {
    int relay;  // -1 = direct
    double cap;
    uint64_t assigned = 0;
  };
  std::vector<Path> paths;
  paths.push_back({-1, topo.weight(src, dst)});
  for (int v = 0; v < topo.world; ++v) {
    if (v == src || v == dst) continue;
    double cap = std::min(topo.weight(src, v), topo.weight(v, dst));
    if (cap > 0.0) paths.push_back({v, cap});
  }

  for (uint64_t off = 0; off < nbytes; off += chunk_bytes) {
    uint64_t const len = std::min(chunk_bytes, nbytes - off);
    // greedy: the path whose normalized finish time grows least
    size_t best = 0;
    double best_t = 1e300;
    for (size_t p = 0; p < paths.size(); ++p) {
      if (paths[p].cap <= 0.0) continue;
      double t = double(paths[p].assigned + len) / paths[p].cap;
      if (t < best_t) {
        best_t = t;
        best = p;
      }
    }
    paths[best].assigned += len;
    int const relay = paths[best].relay;
    if (relay < 0) {
      Task t;
      t.op = Op::kPut;
      t.rank = src;
      t.src = {src, Space::kInput, off};
      t.dst = {dst, Space::kOutput, off};
      t.bytes = len;
      g.add(std::move(t));
    } else {
      Task t1;
      t1.op = Op::kPut;
      t1.rank = src;
      t1.src = {src, Space::kInput, off};
      t1.dst = {relay, Space::kScratch, off};
      t1.bytes = len;
      int const a = g.add(std::move(t1));
      Task t2;
      t2.op = Op::kPut;
      t2.rank = relay;
      t2.src = {relay, Space::kScratch, off};
      t2.dst = {dst, Space::kOutput, off};
      t2.bytes = len;
      t2.deps = {a};
      g.add(std::move(t2));
    }
  }
  return g;
}

ChunkGraph plan_allreduce_rsag(Topology const& topo, uint64_t nbytes,
                               uint64_t elem_bytes, uint64_t chunk_bytes) {
  int const world = topo.world;
  ChunkGraph g;
  g.world = world;
  uint64_t const elems = nbytes / elem_bytes;
  uint64_t const shard_elems = (elems + world - 1) / world;
  uint64_t const shard = shard_elems * elem_bytes;
  g.scratch_bytes = shard * world;  // one inbound slot per source rank

  for (int s = 0; s < world; ++s) {
    uint64_t const base = uint64_t(s) * shard;
    if (base >= nbytes) break;
    uint64_t const slen = std::min(shard, nbytes - base);
    for (uint64_t c = 0; c < slen; c += chunk_bytes) {
      uint64_t const len = std::min(chunk_bytes, slen - c);
      // owner seeds its output with its own input chunk
      Task seed;
      seed.op = Op::kCopy;
      seed.rank = s;
      seed.src = {s, Space::kInput, base + c};
      seed.dst = {s, Space::kOutput, base + c};
      seed.bytes = len;
      int last = g.add(std::move(seed));
      // fixed source order (0..world-1, skipping owner) => bitwise
      // reproducible, matching the collective engine's reduction rule
      for (int r = 0; r < world; ++r) {
        if (r == s) continue;
        Task put;
        put.op = Op::kPut;
        put.rank = r;
        put.src = {r, Space::kInput, base + c};
        put.dst = {s, Space::kScratch, uint64_t(r) * shard + c};
        put.bytes = len;
        int const p = g.add(std::move(put));
        Task red;
        red.op = Op::kReduce;
        red.rank = s;
        red.src = {s, Space::kScratch, uint64_t(r) * shard + c};
        red.dst = {s, Space::kOutput, base + c};
        red.bytes = len;
        red.deps = {p, last};
        last = g.add(std::move(red));
      }
      // all-gather the reduced chunk
      for (int r = 0; r < world; ++r) {
        if (r == s) continue;
        Task ag;
        ag.op = Op::kPut;
        ag.rank = s;
        ag.src = {s, Space::kOutput, base + c};
        ag.dst = {r, Space::kOutput, base + c};
        ag.bytes = len;
        ag.deps = {last};
        g.add(std::move(ag));
      }
    }
  }
  return g;
}

ChunkGraph plan_allreduce_oneshot(Topology const& topo, uint64_t nbytes,
                                  uint64_t elem_bytes) {
  (void)elem_bytes;
  int const world = topo.world;
  ChunkGraph g;
  g.world = world;
  g.scratch_bytes = nbytes * world;

  // every rank pushes its payload into a per-source slot on every peer,
  // every rank reduces its slots locally (one-shot fullmesh)
  std::vector<std::vector<int>> puts(world);  // [dst] -> put task per src
  for (int r = 0; r < world; ++r) puts[r].resize(world, -1);
  for (int src = 0; src < world; ++src) {
    for (int dst = 0; dst < world; ++dst) {
      if (src == dst) continue;
      Task put;
      put.op = Op::kPut;
      put.rank = src;
      put.src = {src, Space::kInput, 0};
      put.dst = {dst, Space::kScratch, uint64_t(src) * nbytes};
      put.bytes = nbytes;
      puts[dst][src] = g.add(std::move(put));
    }
  }
  for (int r = 0; r < world; ++r) {
    Task seed;
    seed.op = Op::kCopy;
    seed.rank = r;
    seed.src = {r, Space::kInput, 0};
    seed.dst = {r, Space::kOutput, 0};
    seed.bytes = nbytes;
    int last = g.add(std::move(seed));
    for (int src = 0; src < world; ++src) {
      if (src == r) continue;
      Task red;
      red.op = Op::kReduce;
      red.rank = r;
      red.src = {r, Space::kScratch, uint64_t(src) * nbytes};
      red.dst = {r, Space::kOutput, 0};
      red.bytes = nbytes;
      red.deps = {puts[r][src], last};
      last = g.add(std::move(red));
    }
  }
  return g;
}

ChunkGraph plan_broadcast(Topology const& topo, int root, uint64_t nbytes,
                          uint64_t chunk_bytes) {
  int const world = topo.world;
  ChunkGraph g;
  g.world = world;
  g.scratch_bytes = uint64_t(world) * nbytes;  // per-destination relay slots

  Task self;
  self.op = Op::kCopy;
  self.rank = root;
  self.src = {root, Space::kInput, 0};
  self.dst = {root, Space::kOutput, 0};
  self.bytes = nbytes;
  g.add(std::move(self));

  for (int dst = 0; dst < world; ++dst) {
    if (dst == root) continue;
    // same spray routing as sendrecv, with relay slots keyed by dst
    ChunkGraph sub = plan_sendrecv_spray(topo, root, dst, nbytes, chunk_bytes);
    int const base = static_cast<int>(g.tasks.size());
    for (auto t : sub.tasks) {
      if (t.src.space == Space::kScratch)
        t.src.offset += uint64_t(dst) * nbytes;
      if (t.dst.space == Space::kScratch)
        t.dst.offset += uint64_t(dst) * nbytes;
      for (auto& d : t.deps) d += base;
      g.add(std::move(t));
    }
  }
  return g;
}

ChunkGraph plan_allgather(Topology const& topo, uint64_t nbytes,
                          uint64_t chunk_bytes) {
  int const world = topo.world;
  ChunkGraph g;
  g.world = world;
  g.scratch_bytes = 0;
  for (int src = 0; src < world; ++src) {
    uint64_t const base = uint64_t(src) * nbytes;
    for (uint64_t c = 0; c < nbytes; c += chunk_bytes) {
      uint64_t const len = std::min(chunk_bytes, nbytes - c);
      Task self;
      self.op = Op::kCopy;
      self.rank = src;
      self.src = {src, Space::kInput, c};
      self.dst = {src, Space::kOutput, base + c};
      self.bytes = len;
      g.add(std::move(self));
      for (int dst = 0; dst < world; ++dst) {
        if (dst == src) continue;
        Task put;
        put.op = Op::kPut;
        put.rank = src;
        put.src = {src, Space::kInput, c};
        put.dst = {dst, Space::kOutput, base + c};
        put.bytes = len;
        g.add(std::move(put));
      }
    }
  }
  return g;
}

ChunkGraph plan_reducescatter(Topology const& topo, uint64_t shard_bytes,
                              uint64_t elem_bytes, uint64_t chunk_bytes) {
  (void)elem_bytes;
  int const world = topo.world;
  ChunkGraph g;
  g.world = world;
  g.scratch_bytes = shard_bytes * world;
  for (int owner = 0; owner < world; ++owner) {
    uint64_t const base = uint64_t(owner) * shard_bytes;
    for (uint64_t c = 0; c < shard_bytes; c += chunk_bytes) {
      uint64_t const len = std::min(chunk_bytes, shard_bytes - c);
      Task seed;
      seed.op = Op::kCopy;
      seed.rank = owner;
      seed.src = {owner, Space::kInput, base + c};
      seed.dst = {owner, Space::kOutput, c};
      seed.bytes = len;
      int last = g.add(std::move(seed));
      for (int r = 0; r < world; ++r) {
        if (r == owner) continue;
        Task put;
        put.op = Op::kPut;
        put.rank = r;
        put.src = {r, Space::kInput, base + c};
        put.dst = {owner, Space::kScratch, uint64_t(r) * shard_bytes + c};
        put.bytes = len;
        int const p = g.add(std::move(put));
        Task red;
        red.op = Op::kReduce;
        red.rank = owner;
        red.src = {owner, Space::kScratch, uint64_t(r) * shard_bytes + c};
        red.dst = {owner, Space::kOutput, c};
        red.bytes = len;
        red.deps = {p, last};
        last = g.add(std::move(red));
      }
    }
  }
  return g;
}

ChunkGraph plan_alltoall(Topology const& topo, uint64_t seg_bytes,
                         uint64_t chunk_bytes) {
  int const world = topo.world;
  ChunkGraph g;
  g.world = world;
  for (int src = 0; src < world; ++src) {
    for (int dst = 0; dst < world; ++dst) {
      for (uint64_t c = 0; c < seg_bytes; c += chunk_bytes) {
        uint64_t const len = std::min(chunk_bytes, seg_bytes - c);
        Task t;
        t.op = src == dst ? Op::kCopy : Op::kPut;
        t.rank = src;
        t.src = {src, Space::kInput, uint64_t(dst) * seg_bytes + c};
        t.dst = {dst, Space::kOutput, uint64_t(src) * seg_bytes + c};
        t.bytes = len;
        g.add(std::move(t));
      }
    }
  }
  return g;
}

// ---------------------------------------------------------------------------
// Cost model
// ---------------------------------------------------------------------------

double estimate_us(ChunkGraph const& g, Topology const& topo,
                   double link_gbps, double local_gbps, double overhead_us) {
  size_t const n = g.tasks.size();
  int const world = g.world;
  // topological order (deps are acyclic; lower() or planners guarantee it)
  std::vector<int> indeg(n, 0);
  std::vector<std::vector<int>> out(n);
  for (size_t i = 0; i < n; ++i)
    for (int d : g.tasks[i].deps) {
      indeg[i]++;
      out[d].push_back(static_cast<int>(i));
    }
  std::deque<int> q;
  for (size_t i = 0; i < n; ++i)
    if (!indeg[i]) q.push_back(static_cast<int>(i));

  std::vector<double> finish(n, 0.0);
  std::vector<double> link_free(size_t(world) * world, 0.0);
  std::vector<double> rank_free(world, 0.0);
  double total = 0.0;
  size_t seen = 0;
  while (!q.empty()) {
    int const i = q.front();
    q.pop_front();
    ++seen;
    Task const& t = g.tasks[i];
    double ready = 0.0;
    for (int d : t.deps) ready = std::max(ready, finish[d]);
    double dur = overhead_us;
    double* res = nullptr;
    if (t.op == Op::kPut) {
      double const w = topo.weight(t.src.rank, t.dst.rank);
      double const rate = link_gbps * (w > 0 ? w : 1e-3);  // GB/s
      dur = overhead_us + t.bytes / rate * 1e-3;           // B/(GB/s)=ns
      res = &link_free[size_t(t.src.rank) * world + t.dst.rank];
    } else if (t.op == Op::kCopy || t.op == Op::kReduce) {
      double const mult = t.op == Op::kReduce ? 2.0 : 1.0;  // rd+rd+wr vs rd+wr
      dur = overhead_us + mult * t.bytes / local_gbps * 1e-3;
      res = &rank_free[t.rank];
    }
    double const start = res ? std::max(ready, *res) : ready;
    finish[i] = start + dur;
    if (res) *res = finish[i];
    total = std::max(total, finish[i]);
    for (int s2 : out[i])
      if (--indeg[s2] == 0) q.push_back(s2);
  }
  if (seen != n) throw std::invalid_argument("estimate: graph has a cycle");
  return total;
}

ChunkGraph plan_allreduce_auto(Topology const& topo, uint64_t nbytes,
                               uint64_t elem_bytes, uint64_t chunk_bytes) {
  ChunkGraph one = plan_allreduce_oneshot(topo, nbytes, elem_bytes);
  ChunkGraph rsag = plan_allreduce_rsag(topo, nbytes, elem_bytes,
                                        chunk_bytes);
  double const t1 = estimate_us(one, topo);
  double const t2 = estimate_us(rsag, topo);
  return lower(t1 <= t2 ? one : rsag);
}

// ---------------------------------------------------------------------------
// Lowering
// ---------------------------------------------------------------------------

ChunkGraph lower(ChunkGraph const& g) {
  size_t const n = g.tasks.size();
  // cycle check (Kahn)
  {
    std::vector<int> indeg(n, 0);
    std::vector<std::vector<int>> out(n);
    for (size_t i = 0; i < n; ++i)
      for (int d : g.tasks[i].deps) {
        if (d < 0 || size_t(d) >= n)
          throw std::invalid_argument("dep out of range");
        indeg[i]++;
        out[d].push_back(static_cast<int>(i));
      }
    std::deque<int> q;
    for (size_t i = 0; i < n; ++i)
      if (!indeg[i]) q.push_back(static_cast<int>(i));
    size_t seen = 0;
    while (!q.empty()) {
      int const u = q.front();
      q.pop_front();
      ++seen;
      for (int v : out[u])
        if (--indeg[v] == 0) q.push_back(v);
    }
    if (seen != n) throw std::invalid_argument("chunk graph has a cycle");
  }

  ChunkGraph out;
  out.world = g.world;
  out.scratch_bytes = g.scratch_bytes;
  uint64_t next_flag = 0;
  std::vector<int> remap(n);
  // A kPut's payload lands on dst.rank, so a same-"rank" successor that
  // READS the data on dst.rank still needs a flag. We treat an edge as
  // cross-rank when the consumer executes on a different rank than the
  // producer OR the producer is a put whose destination rank differs from
  // its executing rank and the consumer runs on that destination.
  for (size_t i = 0; i < n; ++i) {
    Task t = g.tasks[i];
    std::vector<int> new_deps;
    for (int d : t.deps) {
      Task const& p = g.tasks[d];
      bool cross = p.rank != t.rank;
      if (!cross) {
        new_deps.push_back(remap[d]);
        continue;
      }
      if (next_flag >= kMaxFlags) throw std::runtime_error("flag overflow");
      uint64_t const f = next_flag++;
      Task sig;
      sig.op = Op::kSignal;
      sig.rank = p.rank;
      sig.flag = f;
      sig.peer = t.rank;
      sig.deps = {remap[d]};
      int const si = out.add(std::move(sig));
      Task wait;
      wait.op = Op::kWait;
      wait.rank = t.rank;
      wait.flag = f;
      wait.deps = {si};
      new_deps.push_back(out.add(std::move(wait)));
    }
    t.deps = std::move(new_deps);
    remap[i] = out.add(std::move(t));
  }
  return out;
}

// ---------------------------------------------------------------------------
// Executor
// ---------------------------------------------------------------------------

ExecStats execute(ChunkGraph const& g, Backend& backend) {
  size_t const n = g.tasks.size();
  int const world = g.world;
  ExecStats stats;
  stats.link_bytes.assign(size_t(world) * world, 0);

  std::vector<std::atomic<int>> indeg(n);
  std::vector<std::vector<int>> out(n);
  for (size_t i = 0; i < n; ++i) {
    indeg[i].store(static_cast<int>(g.tasks[i].deps.size()),
                   std::memory_order_relaxed);
    for (int d : g.tasks[i].deps) out[d].push_back(static_cast<int>(i));
  }

  // one shared mutex/cv for all rank queues: a host executor is not
  // lock-contention-bound, and the single-lock design makes the
  // shutdown handshake trivially race-free (validated under TSan by
  // tools/run_sanitizers.sh)
  std::mutex mu;
  std::condition_variable cv;
  std::vector<std::deque<int>> queues(world);
  size_t done = 0;  // guarded by mu
  std::atomic<uint64_t> tasks_run{0}, wait_requeues{0};
  std::vector<std::atomic<uint64_t>> link_bytes(size_t(world) * world);
  for (auto& b : link_bytes) b.store(0, std::memory_order_relaxed);

  auto push = [&](int ti) {
    {
      std::lock_guard<std::mutex> lk(mu);
      queues[g.tasks[ti].rank].push_back(ti);
    }
    cv.notify_all();
  };
  for (size_t i = 0; i < n; ++i)
    if (g.tasks[i].deps.empty()) push(static_cast<int>(i));

  auto worker = [&](int rank) {
    int backoff = 0;
    for (;;) {
      int ti = -1;
      {
        std::unique_lock<std::mutex> lk(mu);
        cv.wait(lk, [&] { return !queues[rank].empty() || done >= n; });
        if (queues[rank].empty()) return;  // all tasks complete
        ti = queues[rank].front();
        queues[rank].pop_front();
      }
      Task const& t = g.tasks[ti];
      if (t.op == Op::kWait && !backend.poll(t.flag)) {
        // deferred re-queue: let other ready chunks on this rank run
        wait_requeues.fetch_add(1, std::memory_order_relaxed);
        {
          std::lock_guard<std::mutex> lk(mu);
          queues[rank].push_back(ti);
        }
        if (++backoff > 64) {
          std::this_thread::sleep_for(std::chrono::microseconds(50));
          backoff = 0;
        }
        continue;
      }
      backoff = 0;
      {
        trace::Span span("ukernel", op_name(t.op));
        switch (t.op) {
          case Op::kCopy: backend.copy(t); break;
          case Op::kReduce: backend.reduce(t); break;
          case Op::kPut:
            backend.put(t);
            link_bytes[size_t(t.src.rank) * world + t.dst.rank].fetch_add(
                t.bytes, std::memory_order_relaxed);
            break;
          case Op::kSignal: backend.signal(t.flag); break;
          case Op::kWait: break;  // poll already succeeded
        }
      }
      tasks_run.fetch_add(1, std::memory_order_relaxed);
      for (int s : out[ti])
        if (indeg[s].fetch_sub(1, std::memory_order_acq_rel) == 1) push(s);
      {
        std::lock_guard<std::mutex> lk(mu);
        if (++done == n) cv.notify_all();
      }
    }
  };

  std::vector<std::thread> ths;
  ths.reserve(world);
  for (int r = 0; r < world; ++r) ths.emplace_back(worker, r);
  for (auto& th : ths) th.join();

  stats.tasks_run = tasks_run.load();
  stats.wait_requeues = wait_requeues.load();
  for (size_t i = 0; i < link_bytes.size(); ++i)
    stats.link_bytes[i] = link_bytes[i].load();
  return stats;
}

// ---------------------------------------------------------------------------
// HostBackend
// ---------------------------------------------------------------------------

HostBackend::HostBackend(int world, uint64_t in_bytes, uint64_t out_bytes,
                         uint64_t scratch_bytes)
    : link_bytes(size_t(world) * world),
      world_(world),
      in_bytes_(in_bytes),
      out_bytes_(out_bytes),
      scratch_bytes_(scratch_bytes),
      flags_(kMaxFlags) {
  for (auto& b : link_bytes) b.store(0, std::memory_order_relaxed);
  for (auto& f : flags_) f.store(0, std::memory_order_relaxed);
  auto elems = [](uint64_t b) { return (b + 3) / 4; };
  in_.resize(world);
  out_.resize(world);
  scratch_.resize(world);
  for (int r = 0; r < world; ++r) {
    in_[r].assign(elems(in_bytes), 0.f);
    out_[r].assign(elems(out_bytes), 0.f);
    scratch_[r].assign(elems(scratch_bytes), 0.f);
  }
}

float* HostBackend::input(int rank) { return in_[rank].data(); }
float* HostBackend::output(int rank) { return out_[rank].data(); }

float* HostBackend::resolve(BufRef const& b) {
  std::vector<float>* v = nullptr;
  uint64_t cap = 0;
  switch (b.space) {
    case Space::kInput: v = &in_[b.rank]; cap = in_bytes_; break;
    case Space::kOutput: v = &out_[b.rank]; cap = out_bytes_; break;
    case Space::kScratch: v = &scratch_[b.rank]; cap = scratch_bytes_; break;
  }
  if (b.offset % 4 != 0 || b.offset > cap)
    throw std::out_of_range("ukernel: buffer ref outside its space");
  return v->data() + b.offset / 4;
}

void HostBackend::copy(Task const& t) {
  std::memcpy(resolve(t.dst), resolve(t.src), t.bytes);
}

void HostBackend::reduce(Task const& t) {
  float* d = resolve(t.dst);
  float const* s = resolve(t.src);
  for (uint64_t i = 0; i < t.bytes / 4; ++i) d[i] += s[i];
}

void HostBackend::put(Task const& t) {
  std::memcpy(resolve(t.dst), resolve(t.src), t.bytes);
  link_bytes[size_t(t.src.rank) * world_ + t.dst.rank].fetch_add(
      t.bytes, std::memory_order_relaxed);
}

void HostBackend::signal(uint64_t flag) {
  flags_[flag].fetch_add(1, std::memory_order_release);
}

bool HostBackend::poll(uint64_t flag) {
  return flags_[flag].load(std::memory_order_acquire) > 0;
}

}  // namespace uk
}  // namespace uccl
