// Sanitizer harness for the codec and the chunk-graph planner/executor
// (built with ASan/UBSan/TSan by tools/run_sanitizers.sh).

#include <cassert>
#include <cstdio>
#include <cstdint>
#include <cstring>
#include <random>
#include <vector>

#include "../uccl_amd/csrc/p2p/compress.h"
#include "../uccl_amd/csrc/ukernel/ukernel.h"

using namespace uccl;

static void codec_pass(std::mt19937_64& rng) {
  for (int trial = 0; trial < 40; ++trial) {
    size_t const n = rng() % 300000;
    int const elem = (trial % 2) ? 2 : 4;
    int const strat = trial % 3;
    std::vector<uint8_t> data(n);
    for (auto& b : data) b = static_cast<uint8_t>(rng());
    std::string frame =
        p2p::comp::compress(data.data(), n, elem, elem == 2 ? 2 : 0, strat);
    std::vector<uint8_t> out(n + 16);
    size_t const got =
        p2p::comp::decompress(frame.data(), frame.size(), out.data(), n);
    assert(got == n && memcmp(out.data(), data.data(), n) == 0);
    // mutate the frame; decompress must throw or produce bounded output
    if (!frame.empty()) {
      std::string bad = frame;
      for (int m = 0; m < 20; ++m)
        bad[rng() % bad.size()] = static_cast<char>(rng());
      try {
        p2p::comp::decompress(bad.data(), bad.size(), out.data(), n);
      } catch (std::exception const&) {
      }
      // truncation
      try {
        p2p::comp::decompress(frame.data(), rng() % (frame.size() + 1),
                              out.data(), n);
      } catch (std::exception const&) {
      }
    }
  }
}

static void ukernel_pass(std::mt19937_64& rng) {
  for (int trial = 0; trial < 12; ++trial) {
    int const world = 2 + static_cast<int>(rng() % 7);
    uint64_t const elems = 1 + rng() % 4000;
    uint64_t const chunk = (64 + rng() % 2048) * 4;
    uk::Topology topo(world);
    uk::ChunkGraph g;
    switch (trial % 5) {
      case 0: g = uk::plan_allreduce_rsag(topo, elems * 4, 4, chunk); break;
      case 1: g = uk::plan_allreduce_oneshot(topo, elems * 4, 4); break;
      case 2: g = uk::plan_sendrecv_spray(topo, 0, world - 1, elems * 4,
                                          chunk); break;
      case 3: g = uk::plan_alltoall(topo, elems * 4, chunk); break;
      default: g = uk::plan_reducescatter(topo, elems * 4, 4, chunk);
    }
    uk::ChunkGraph low = uk::lower(g);
    uint64_t const in_b =
        trial % 5 == 3 || trial % 5 == 4 ? elems * 4 * world : elems * 4;
    uint64_t const out_b =
        trial % 5 == 3 ? elems * 4 * world : elems * 4 * (world + 1);
    uk::HostBackend hb(world, in_b, out_b, low.scratch_bytes);
    uk::ExecStats st = uk::execute(low, hb);
    assert(st.tasks_run == low.tasks.size());
    (void)uk::estimate_us(g, topo);
  }
}

int main() {
  std::mt19937_64 rng(20260912);
  codec_pass(rng);
  ukernel_pass(rng);
  printf("SAN CODEC+UKERNEL OK\n");
  return 0;
}
