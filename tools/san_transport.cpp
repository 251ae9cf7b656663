// Sanitizer harness for the reliable transport: a pair of endpoints in
// one process, concurrent bidirectional traffic across several flows.
// Built by tools/run_sanitizers.sh with -fsanitize=thread (and address),
// which the pytest tier cannot do through the torch extension.

#include <cassert>
#include <cstdio>
#include <cstring>
#include <string>
#include <thread>
#include <vector>

#include "../uccl_amd/csrc/transport/reliable.h"

using uccl::transport::TransportEndpoint;

int main() {
  setenv("UCCL_TP_CWND_MAX", "256", 0);
  for (int round = 0; round < 6; ++round) {
    // alternate CC modes and loss across rounds
    setenv("UCCL_TP_CC", round % 3 == 0 ? "timely"
                         : round % 3 == 1 ? "swift" : "eqds", 1);
    setenv("UCCL_TP_LOSS_PCT", round % 2 ? "3" : "0", 1);
    // exercise the paced path + timing wheel under the sanitizers
    setenv("UCCL_TP_PACE_MBPS", round % 3 == 1 ? "200" : "0", 1);
    TransportEndpoint a(2, 4096), b(2, 4096);
    uint64_t fb = 0;
    std::thread acc([&] { fb = b.accept(nullptr); });
    uint64_t fa = a.connect(b.metadata(), 1);
    acc.join();

    // concurrent bidirectional messages on the same flow pair
    std::vector<std::thread> ths;
    constexpr int kMsgs = 5;
    std::vector<std::vector<char>> a2b(kMsgs), b2a(kMsgs), ra(kMsgs),
        rb(kMsgs);
    for (int i = 0; i < kMsgs; ++i) {
      size_t const n = 1000 + 37013 * i;
      a2b[i].assign(n, static_cast<char>(i + 1));
      b2a[i].assign(n, static_cast<char>(0x40 + i));
      rb[i].resize(n);
      ra[i].resize(n);
    }
    ths.emplace_back([&] {
      for (int i = 0; i < kMsgs; ++i)
        a.send_msg(fa, a2b[i].data(), a2b[i].size());
    });
    ths.emplace_back([&] {
      for (int i = 0; i < kMsgs; ++i)
        b.recv_msg(fb, rb[i].data(), rb[i].size());
    });
    ths.emplace_back([&] {
      for (int i = 0; i < kMsgs; ++i)
        b.send_msg(fb, b2a[i].data(), b2a[i].size());
    });
    ths.emplace_back([&] {
      for (int i = 0; i < kMsgs; ++i)
        a.recv_msg(fa, ra[i].data(), ra[i].size());
    });
    for (auto& t : ths) t.join();
    for (int i = 0; i < kMsgs; ++i) {
      assert(memcmp(rb[i].data(), a2b[i].data(), a2b[i].size()) == 0);
      assert(memcmp(ra[i].data(), b2a[i].data(), b2a[i].size()) == 0);
    }
    printf("round %d ok (cc=%s)\n", round, getenv("UCCL_TP_CC"));
  }
  printf("SAN TRANSPORT OK\n");
  return 0;
}
