// Standalone harness that dlopens librccl-net-uccl.so and drives the v6
// vtable end-to-end in one process (listen/connect/accept, tag-matched
// isend/irecv both orders, test-completion polling). Exercises exactly the
// calls RCCL's proxy makes, without needing a multi-GPU RCCL job.

#include <dlfcn.h>

#ifdef UCCL_NET_HIP_TEST
#include <hip/hip_runtime.h>
#endif

#include <cassert>
#include <cstdio>
#include <cstring>
#include <thread>
#include <vector>

#include "net_plugin_abi.h"

static void logger(ncclDebugLogLevel, unsigned long, const char*, int,
                   const char*, ...) {}

static void wait_req(ncclNet_v6_t* net, void* req, int* size) {
  int done = 0, sz = 0;
  while (!done) assert(net->test(req, &done, &sz) == ncclSuccess);
  if (size) *size = sz;
}

int main(int argc, char** argv) {
  const char* path = argc > 1 ? argv[1] : "librccl-net-uccl.so";
  void* so = dlopen(path, RTLD_NOW);
  if (!so) {
    fprintf(stderr, "dlopen failed: %s\n", dlerror());
    return 1;
  }
  auto* net = static_cast<ncclNet_v6_t*>(dlsym(so, "ncclNetPlugin_v6"));
  assert(net && "missing ncclNetPlugin_v6");
  assert(net->init(logger) == ncclSuccess);
  int ndev = 0;
  assert(net->devices(&ndev) == ncclSuccess && ndev >= 1);
  ncclNetProperties_v6_t props{};
  assert(net->getProperties(0, &props) == ncclSuccess);
  printf("dev %s speed %d maxRecvs %d\n", props.name, props.speed,
         props.maxRecvs);

  char handle[NCCL_NET_HANDLE_MAXSIZE] = {};
  void *lc = nullptr, *sc = nullptr, *rc = nullptr;
  assert(net->listen(0, handle, &lc) == ncclSuccess);
  while (!sc) assert(net->connect(0, handle, &sc) == ncclSuccess);
  while (!rc) assert(net->accept(lc, &rc) == ncclSuccess);

  // --- ordered send -> recv (tag 7) ---------------------------------------
  std::vector<char> src(1 << 20), dst(1 << 20, 0);
  for (size_t i = 0; i < src.size(); ++i) src[i] = static_cast<char>(i * 13);
  void *sreq = nullptr, *rreq = nullptr;
  void* mh = nullptr;
  assert(net->regMr(sc, src.data(), src.size(), NCCL_PTR_HOST, &mh) ==
         ncclSuccess);
  assert(net->isend(sc, src.data(), src.size(), 7, mh, &sreq) ==
         ncclSuccess);
  void* datas[1] = {dst.data()};
  int sizes[1] = {static_cast<int>(dst.size())};
  int tags[1] = {7};
  void* mhs[1] = {nullptr};
  assert(net->irecv(rc, 1, datas, sizes, tags, mhs, &rreq) == ncclSuccess);
  int got = 0;
  wait_req(net, sreq, nullptr);
  wait_req(net, rreq, &got);
  assert(got == static_cast<int>(src.size()));
  assert(memcmp(src.data(), dst.data(), src.size()) == 0);
  printf("ordered send/recv OK\n");

  // --- out-of-order tags: send tag 2 then 1; post recv 1 first ------------
  char a[64] = "tag-one-payload", b[64] = "tag-two-payload";
  char ra[64] = {0}, rb[64] = {0};
  void *s1 = nullptr, *s2 = nullptr, *r1 = nullptr, *r2 = nullptr;
  assert(net->isend(sc, b, sizeof(b), 2, nullptr, &s2) == ncclSuccess);
  wait_req(net, s2, nullptr);  // frame lands unmatched on rx side
  void* d1[1] = {ra};
  int z1[1] = {64};
  int t1[1] = {1};
  assert(net->irecv(rc, 1, d1, z1, t1, mhs, &r1) == ncclSuccess);
  assert(net->isend(sc, a, sizeof(a), 1, nullptr, &s1) == ncclSuccess);
  wait_req(net, s1, nullptr);
  wait_req(net, r1, &got);
  assert(got == 64 && strcmp(ra, "tag-one-payload") == 0);
  void* d2[1] = {rb};
  int z2[1] = {64};
  int t2[1] = {2};
  assert(net->irecv(rc, 1, d2, z2, t2, mhs, &r2) == ncclSuccess);
  wait_req(net, r2, &got);
  assert(got == 64 && strcmp(rb, "tag-two-payload") == 0);
  printf("tag matching OK\n");

  // --- zero-byte message ---------------------------------------------------
  void *s0 = nullptr, *r0 = nullptr;
  assert(net->isend(sc, a, 0, 3, nullptr, &s0) == ncclSuccess);
  void* d0[1] = {ra};
  int z0[1] = {0};
  int t0[1] = {3};
  assert(net->irecv(rc, 1, d0, z0, t0, mhs, &r0) == ncclSuccess);
  wait_req(net, s0, nullptr);
  wait_req(net, r0, &got);
  assert(got == 0);
  printf("zero-byte OK\n");

  // --- flush (host: immediate) --------------------------------------------
  void* freq = reinterpret_cast<void*>(0x1);
  assert(net->iflush(rc, 1, datas, sizes, mhs, &freq) == ncclSuccess);
  assert(freq == nullptr);

  assert(net->closeSend(sc) == ncclSuccess);
  assert(net->closeRecv(rc) == ncclSuccess);
  assert(net->closeListen(lc) == ncclSuccess);

  // --- concurrent multi-pair: 3 listeners, interleaved traffic ------------
  // (exercises listener demultiplexing — on the multipath plane, the
  // nonce-routed flow queues of the shared fabric endpoint)
  {
    std::vector<std::thread> ths;
    for (int pair = 0; pair < 3; ++pair) {
      ths.emplace_back([net, pair] {
        char hdl[NCCL_NET_HANDLE_MAXSIZE] = {};
        void *plc = nullptr, *psc = nullptr, *prc = nullptr;
        assert(net->listen(0, hdl, &plc) == ncclSuccess);
        std::thread acc([&] {
          while (!prc) assert(net->accept(plc, &prc) == ncclSuccess);
        });
        while (!psc) assert(net->connect(0, hdl, &psc) == ncclSuccess);
        acc.join();
        for (int it = 0; it < 5; ++it) {
          std::vector<char> sbuf(100000 + 1000 * pair,
                                 static_cast<char>(pair * 31 + it));
          std::vector<char> rbuf(sbuf.size());
          void *sr = nullptr, *rr = nullptr;
          int const tag = pair * 100 + it;
          assert(net->isend(psc, sbuf.data(),
                            static_cast<int>(sbuf.size()), tag, nullptr,
                            &sr) == ncclSuccess);
          void* d[1] = {rbuf.data()};
          int z[1] = {static_cast<int>(rbuf.size())};
          int t[1] = {tag};
          void* mh[1] = {nullptr};
          assert(net->irecv(prc, 1, d, z, t, mh, &rr) == ncclSuccess);
          wait_req(net, sr, nullptr);
          int got = 0;
          wait_req(net, rr, &got);
          assert(got == static_cast<int>(sbuf.size()));
          assert(memcmp(sbuf.data(), rbuf.data(), sbuf.size()) == 0);
        }
        assert(net->closeSend(psc) == ncclSuccess);
        assert(net->closeRecv(prc) == ncclSuccess);
        assert(net->closeListen(plc) == ncclSuccess);
      });
    }
    for (auto& t : ths) t.join();
    printf("multi-pair OK\n");
  }

  // --- rapid-fire unmatched/posted interleave stress ----------------------
  // regression for the rx cross-match race: when a frame arrives in the
  // window between irecv's unmatched scan and its posted push, the rx
  // loop must deliver the late match. Alternate posting order and sizes
  // across many quick iterations to drive both paths.
  {
    char hdl[NCCL_NET_HANDLE_MAXSIZE] = {};
    void *plc = nullptr, *psc = nullptr, *prc = nullptr;
    assert(net->listen(0, hdl, &plc) == ncclSuccess);
    std::thread acc([&] {
      while (!prc) assert(net->accept(plc, &prc) == ncclSuccess);
    });
    while (!psc) assert(net->connect(0, hdl, &psc) == ncclSuccess);
    acc.join();
    for (int it = 0; it < 300; ++it) {
      int const n = 64 + (it * 97) % 4000;
      std::vector<char> sb(n, static_cast<char>(it));
      std::vector<char> rb(n, 0);
      void *sr = nullptr, *rr = nullptr;
      void* d[1] = {rb.data()};
      int z[1] = {n};
      int t[1] = {it};
      void* mh[1] = {nullptr};
      if (it & 1) {
        // send first: the frame usually lands before the recv is posted
        assert(net->isend(psc, sb.data(), n, it, nullptr, &sr) ==
               ncclSuccess);
        wait_req(net, sr, nullptr);
        assert(net->irecv(prc, 1, d, z, t, mh, &rr) == ncclSuccess);
      } else {
        assert(net->irecv(prc, 1, d, z, t, mh, &rr) == ncclSuccess);
        assert(net->isend(psc, sb.data(), n, it, nullptr, &sr) ==
               ncclSuccess);
        wait_req(net, sr, nullptr);
      }
      int got = 0;
      wait_req(net, rr, &got);
      assert(got == n && memcmp(sb.data(), rb.data(), n) == 0);
    }
    assert(net->closeSend(psc) == ncclSuccess);
    assert(net->closeRecv(prc) == ncclSuccess);
    assert(net->closeListen(plc) == ncclSuccess);
    printf("interleave stress OK\n");
  }

  // --- grouped irecv (n=2, one request) -----------------------------------
  {
    char hdl[NCCL_NET_HANDLE_MAXSIZE] = {};
    void *plc = nullptr, *psc = nullptr, *prc = nullptr;
    assert(net->listen(0, hdl, &plc) == ncclSuccess);
    std::thread acc([&] {
      while (!prc) assert(net->accept(plc, &prc) == ncclSuccess);
    });
    while (!psc) assert(net->connect(0, hdl, &psc) == ncclSuccess);
    acc.join();
    std::vector<char> x1(4096, 'a'), x2(8192, 'b');
    std::vector<char> y1(4096, 0), y2(8192, 0);
    void *gs1 = nullptr, *gs2 = nullptr, *gr = nullptr;
    assert(net->isend(psc, x1.data(), 4096, 21, nullptr, &gs1) ==
           ncclSuccess);
    assert(net->isend(psc, x2.data(), 8192, 22, nullptr, &gs2) ==
           ncclSuccess);
    void* d[2] = {y1.data(), y2.data()};
    int z[2] = {4096, 8192};
    int t[2] = {21, 22};
    void* m2[2] = {nullptr, nullptr};
    assert(net->irecv(prc, 2, d, z, t, m2, &gr) == ncclSuccess);
    wait_req(net, gs1, nullptr);
    wait_req(net, gs2, nullptr);
    int done = 0, szs[2] = {0, 0};
    while (!done) assert(net->test(gr, &done, szs) == ncclSuccess);
    assert(szs[0] == 4096 && szs[1] == 8192);
    assert(memcmp(x1.data(), y1.data(), 4096) == 0);
    assert(memcmp(x2.data(), y2.data(), 8192) == 0);
    assert(net->closeSend(psc) == ncclSuccess);
    assert(net->closeRecv(prc) == ncclSuccess);
    assert(net->closeListen(plc) == ncclSuccess);
    printf("grouped irecv OK\n");
  }

#ifdef UCCL_NET_HIP_TEST
  // --- device-MR staging (NCCL_PTR_CUDA) ----------------------------------
  // The exact calls RCCL's proxy makes when ptrSupport advertises CUDA:
  // regMr(device ptr) -> isend/irecv with the device MR handle.
  {
    int devcount = 0;
    if (hipGetDeviceCount(&devcount) == hipSuccess && devcount > 0) {
      (void)hipSetDevice(0);
      size_t const n = 3 << 20;
      std::vector<char> hsrc(n), hdst(n, 0);
      for (size_t i = 0; i < n; ++i) hsrc[i] = static_cast<char>(i * 31);
      void *gsrc = nullptr, *gdst = nullptr;
      assert(hipMalloc(&gsrc, n) == hipSuccess);
      assert(hipMalloc(&gdst, n) == hipSuccess);
      assert(hipMemcpy(gsrc, hsrc.data(), n, hipMemcpyHostToDevice) ==
             hipSuccess);
      char hdl[NCCL_NET_HANDLE_MAXSIZE] = {};
      void *plc = nullptr, *psc = nullptr, *prc = nullptr;
      assert(net->listen(0, hdl, &plc) == ncclSuccess);
      std::thread acc([&] {
        while (!prc) assert(net->accept(plc, &prc) == ncclSuccess);
      });
      while (!psc) assert(net->connect(0, hdl, &psc) == ncclSuccess);
      acc.join();
      assert(net->getProperties(0, &props) == ncclSuccess);
      assert(props.ptrSupport & NCCL_PTR_CUDA);
      void *smh = nullptr, *rmh = nullptr;
      assert(net->regMr(psc, gsrc, n, NCCL_PTR_CUDA, &smh) == ncclSuccess);
      assert(net->regMr(prc, gdst, n, NCCL_PTR_CUDA, &rmh) == ncclSuccess);
      void *sr = nullptr, *rr = nullptr;
      assert(net->isend(psc, gsrc, static_cast<int>(n), 42, smh, &sr) ==
             ncclSuccess);
      void* d[1] = {gdst};
      int z[1] = {static_cast<int>(n)};
      int t[1] = {42};
      void* mh[1] = {rmh};
      assert(net->irecv(prc, 1, d, z, t, mh, &rr) == ncclSuccess);
      wait_req(net, sr, nullptr);
      int got = 0;
      wait_req(net, rr, &got);
      assert(got == static_cast<int>(n));
      assert(hipMemcpy(hdst.data(), gdst, n, hipMemcpyDeviceToHost) ==
             hipSuccess);
      assert(memcmp(hsrc.data(), hdst.data(), n) == 0);
      assert(net->deregMr(psc, smh) == ncclSuccess);
      assert(net->deregMr(prc, rmh) == ncclSuccess);
      assert(net->closeSend(psc) == ncclSuccess);
      assert(net->closeRecv(prc) == ncclSuccess);
      assert(net->closeListen(plc) == ncclSuccess);
      (void)hipFree(gsrc);
      (void)hipFree(gdst);
      printf("CUDA-MR staging OK\n");
    } else {
      printf("CUDA-MR SKIP (no GPU)\n");
    }
  }
#endif

  printf("PLUGIN HARNESS OK\n");
  return 0;
}
