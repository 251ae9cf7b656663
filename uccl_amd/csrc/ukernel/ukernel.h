// Chunk-graph collective planner + spray executor.
//
// The uccl_amd analog of the reference's experimental "ukernel" layer
// (reference: experimental/ukernel/src/ccl/{algo/chunk_graph.cc, lower.cc,
// executor.cc} — planner → lowering → SprayExecutor over pluggable
// backends). Re-designed for the MI355X fabric instead of translated:
//
//   * topology is a single-hop xGMI fullmesh (7 point-to-point links per
//     GPU, no switch), so the planning problem is not ring construction
//     but PATH SELECTION: a pairwise transfer is bound by ONE link
//     (~153 GB/s); splitting it into chunks and relaying some through
//     other GPUs ("spray") aggregates up to all 7 links;
//   * cross-rank dependencies lower to explicit SIGNAL/WAIT task pairs —
//     the same flag-round discipline the collective engine uses (see
//     csrc/collective/kernels.hip k_signal_wait);
//   * the executor is backend-agnostic: the HostBackend runs the whole
//     plan on CPU memory with per-link byte accounting (this is how the
//     reference unit-tests its executor too: mock backends,
//     single-process), and a device backend can bind the same Task
//     stream to HIP copies/kernels.
//
// Everything here is host-side C++ — no GPU required to plan, lower, or
// (with HostBackend) execute.

#pragma once

#include <atomic>
#include <cstdint>
#include <functional>
#include <memory>
#include <string>
#include <vector>

namespace uccl {
namespace uk {

// ---------------------------------------------------------------------------
// Topology: world of fullmesh-connected ranks. Link (r, p) is the direct
// xGMI lane from r to p (r != p). Relay path r -> v -> p uses two links.
// ---------------------------------------------------------------------------
struct Topology {
  int world = 1;
  // per-link relative capacity (uniform on MI355X xGMI; kept as a knob so
  // tests can model a degraded link and verify the planner shifts load)
  std::vector<double> link_weight;  // world*world entries, [src*world+dst]

  explicit Topology(int w) : world(w), link_weight(w * w, 1.0) {
    for (int r = 0; r < w; ++r) link_weight[r * w + r] = 0.0;
  }
  double weight(int src, int dst) const { return link_weight[src * world + dst]; }
};

// ---------------------------------------------------------------------------
// Buffer addressing: abstract (rank, space, offset) — resolved by the
// backend. Spaces mirror the collective engine's heap layout.
// ---------------------------------------------------------------------------
enum class Space : uint8_t { kInput = 0, kOutput = 1, kScratch = 2 };

struct BufRef {
  int rank = 0;
  Space space = Space::kInput;
  uint64_t offset = 0;
};

enum class Op : uint8_t {
  kCopy = 0,    // intra-rank: dst <- src            (rank = src.rank = dst.rank)
  kReduce = 1,  // intra-rank: dst <- dst + src
  kPut = 2,     // cross-rank: dst@dst.rank <- src@src.rank over link (src->dst)
  kSignal = 3,  // raise flag `flag` visible to rank dst.rank
  kWait = 4,    // block rank `rank` until flag `flag` >= 1
};

struct Task {
  Op op = Op::kCopy;
  int rank = 0;  // executing rank
  BufRef src, dst;
  uint64_t bytes = 0;
  uint64_t flag = 0;     // for kSignal/kWait
  int peer = -1;         // for kSignal: rank that will observe the flag
  // dependency edges (indices into ChunkGraph::tasks)
  std::vector<int> deps;
};

// A chunk DAG. Planner emits only data ops (kCopy/kReduce/kPut) with
// arbitrary cross-rank edges; lower() materializes signal/wait pairs for
// every cross-rank edge and returns a schedulable graph.
struct ChunkGraph {
  int world = 1;
  std::vector<Task> tasks;
  uint64_t scratch_bytes = 0;  // per-rank scratch the plan needs

  int add(Task t) {
    tasks.push_back(std::move(t));
    return static_cast<int>(tasks.size()) - 1;
  }
  std::string dump() const;
};

// ---------------------------------------------------------------------------
// Planners. nbytes is the full payload per rank; elem_bytes the reduction
// element width (HostBackend reduces in fp32).
// ---------------------------------------------------------------------------

// Pairwise transfer src -> dst sprayed over the direct link plus 1-hop
// relays through every other rank, chunk assignment proportional to path
// capacity (bottleneck link weight).
ChunkGraph plan_sendrecv_spray(Topology const& topo, int src, int dst,
                               uint64_t nbytes, uint64_t chunk_bytes);

// Reduce-scatter + all-gather allreduce: shard s is reduced at rank s
// (every rank PUTs its shard-s chunk into owner scratch; owner reduces),
// then the owner pushes the result to every rank. One-hop fullmesh — the
// natural large-message algorithm on xGMI (see k_twoshot_* kernels).
ChunkGraph plan_allreduce_rsag(Topology const& topo, uint64_t nbytes,
                               uint64_t elem_bytes, uint64_t chunk_bytes);

// One-shot fullmesh allreduce: every rank pulls all peers' payloads into
// scratch and reduces locally (small/medium messages).
ChunkGraph plan_allreduce_oneshot(Topology const& topo, uint64_t nbytes,
                                  uint64_t elem_bytes);

// Broadcast root -> all, relay-sprayed like sendrecv per destination.
ChunkGraph plan_broadcast(Topology const& topo, int root, uint64_t nbytes,
                          uint64_t chunk_bytes);

// All-gather: rank r's nbytes input becomes slot r of every rank's
// world*nbytes output (fullmesh push, chunked).
ChunkGraph plan_allgather(Topology const& topo, uint64_t nbytes,
                          uint64_t chunk_bytes);

// Reduce-scatter: rank r's output is the sum over ranks of shard r of
// their world*shard_bytes inputs (fullmesh push + owner reduce).
ChunkGraph plan_reducescatter(Topology const& topo, uint64_t shard_bytes,
                              uint64_t elem_bytes, uint64_t chunk_bytes);

// All-to-all: segment j of rank i's input becomes segment i of rank j's
// output (all-pairs push, chunked).
ChunkGraph plan_alltoall(Topology const& topo, uint64_t seg_bytes,
                         uint64_t chunk_bytes);

// ---------------------------------------------------------------------------
// Cost model: deterministic list-scheduling estimate of a plan's
// completion time. Resources: every directed link serializes its kPuts at
// link_gbps x topo.weight; every rank serializes local kCopy/kReduce at
// local_gbps; signal/wait cost overhead_us each. This is what lets the
// planner CHOOSE an algorithm per (topology, size) instead of hard
// thresholds (the reference planner's topology-aware role).
// ---------------------------------------------------------------------------
double estimate_us(ChunkGraph const& g, Topology const& topo,
                   double link_gbps = 150.0, double local_gbps = 1500.0,
                   double overhead_us = 4.0);

// Build + lower the better allreduce plan for this size on this topology
// (one-shot vs RS+AG by estimated time).
ChunkGraph plan_allreduce_auto(Topology const& topo, uint64_t nbytes,
                               uint64_t elem_bytes = 4,
                               uint64_t chunk_bytes = 1 << 20);

// ---------------------------------------------------------------------------
// Lowering: validate the DAG, convert every cross-rank dependency edge
// into an explicit kSignal/kWait pair, and assign unique flags.
// Returns the lowered graph (topologically executable by rank workers).
// ---------------------------------------------------------------------------
ChunkGraph lower(ChunkGraph const& g);

// ---------------------------------------------------------------------------
// Executor. One worker thread per rank drains that rank's ready queue;
// finishing a task decrements successors' indegrees (cross-rank edges no
// longer exist after lower() — they became signal/wait). A kWait whose
// flag is not yet raised is REQUEUED (deferred), which is what lets a
// single worker make progress on other chunks meanwhile — the spray
// behavior the reference's SprayExecutor implements with its deferred
// re-queue (executor.cc).
// ---------------------------------------------------------------------------
class Backend {
 public:
  virtual ~Backend() = default;
  virtual void copy(Task const& t) = 0;
  virtual void reduce(Task const& t) = 0;
  virtual void put(Task const& t) = 0;
  virtual void signal(uint64_t flag) = 0;
  virtual bool poll(uint64_t flag) = 0;
};

struct ExecStats {
  uint64_t tasks_run = 0;
  uint64_t wait_requeues = 0;
  // bytes moved per directed link [src*world+dst]
  std::vector<uint64_t> link_bytes;
};

ExecStats execute(ChunkGraph const& lowered, Backend& backend);

// ---------------------------------------------------------------------------
// HostBackend: every rank's Input/Output/Scratch spaces are host buffers;
// reductions are fp32. Used by the CPU test tier; also the reference's
// approach to executor unit testing (mock backends, single process).
// ---------------------------------------------------------------------------
class HostBackend : public Backend {
 public:
  HostBackend(int world, uint64_t in_bytes, uint64_t out_bytes,
              uint64_t scratch_bytes);
  float* input(int rank);
  float* output(int rank);
  uint64_t in_bytes() const { return in_bytes_; }

  void copy(Task const& t) override;
  void reduce(Task const& t) override;
  void put(Task const& t) override;
  void signal(uint64_t flag) override;
  bool poll(uint64_t flag) override;

  // per-link byte counters (filled by execute() as well; kept here so the
  // backend can be inspected directly)
  std::vector<std::atomic<uint64_t>> link_bytes;

 private:
  float* resolve(BufRef const& b);
  int world_;
  uint64_t in_bytes_, out_bytes_, scratch_bytes_;
  std::vector<std::vector<float>> in_, out_, scratch_;
  std::vector<std::atomic<uint32_t>> flags_;
};

}  // namespace uk
}  // namespace uccl
