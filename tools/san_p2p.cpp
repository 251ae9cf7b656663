// Sanitizer harness for the p2p endpoint host paths (TCP plane, no GPU):
// connect/accept, concurrent two-sided + async batches with FIFO
// verification, one-sided write/read, notify-style traffic, teardown.

#include <cassert>
#include <cstdio>
#include <cstring>
#include <string>
#include <thread>
#include <vector>

#include "../uccl_amd/csrc/p2p/endpoint.h"

using uccl::p2p::Endpoint;

int main() {
  setenv("UCCL_P2P_ENABLE_IPC", "0", 1);
  for (int round = 0; round < 4; ++round) {
    // alternate data planes: plain TCP and the multipath reliable engine
    setenv("UCCL_P2P_TRANSPORT", round % 2 ? "multipath" : "tcp", 1);
    // buffers BEFORE endpoints: destruction is reverse order, and the
    // endpoints' rx threads (joined in ~Endpoint) must die before any
    // memory they served one-sided reads from
    std::vector<char> window(1 << 16, 0);
    std::vector<char> payload(1 << 16);
    Endpoint a(-1, 3), b(-1, 3);
    uint64_t cb = 0;
    std::thread acc([&] { cb = b.accept(); });
    uint64_t ca = a.connect(b.metadata());
    acc.join();

    // concurrent async batches, FIFO per direction
    constexpr int kMsgs = 6;
    std::vector<std::vector<char>> src(kMsgs), dst(kMsgs);
    std::vector<uint64_t> sids, rids;
    for (int i = 0; i < kMsgs; ++i) {
      size_t const n = 512 + 9173 * i;
      src[i].assign(n, static_cast<char>(i + round));
      dst[i].assign(n, 0);
      rids.push_back(b.recv_async(cb, dst[i].data(), n, -1));
    }
    for (int i = 0; i < kMsgs; ++i)
      sids.push_back(a.send_async(ca, src[i].data(), src[i].size(), -1));
    for (auto id : sids)
      while (!a.poll_async(id)) std::this_thread::yield();
    for (auto id : rids)
      while (!b.poll_async(id)) std::this_thread::yield();
    for (int i = 0; i < kMsgs; ++i)
      assert(memcmp(src[i].data(), dst[i].data(), src[i].size()) == 0);

    // one-sided against an advertised window, concurrent with sends
    uint64_t const mr = b.reg(window.data(), window.size(), -1);
    std::string const ad = b.advertise(mr, 0, window.size());
    for (size_t i = 0; i < payload.size(); ++i)
      payload[i] = static_cast<char>(i * 13 + round);
    std::thread w([&] { a.write(ca, payload.data(), payload.size(), -1, ad); });
    std::vector<char> extra(2048, 'x'), got(2048);
    std::thread s2([&] { a.send(ca, extra.data(), extra.size(), -1); });
    std::thread r2([&] { b.recv(cb, got.data(), got.size(), -1); });
    w.join();
    s2.join();
    r2.join();
    assert(memcmp(got.data(), extra.data(), extra.size()) == 0);
    std::vector<char> back(1 << 16, 0);
    a.read(ca, back.data(), back.size(), -1, ad);
    assert(memcmp(back.data(), payload.data(), payload.size()) == 0);
    b.dereg(mr);
    printf("round %d ok\n", round);
  }
  printf("SAN P2P OK\n");
  return 0;
}
