#include "c_api.h"

#include <cstring>
#include <deque>
#include <exception>
#include <memory>
#include <mutex>
#include <thread>
#include <unordered_map>
#include <vector>

#include "endpoint.h"

using uccl::p2p::Endpoint;

struct uccl_engine {
  Endpoint ep;
  uccl_engine(int gpu, int nw) : ep(gpu, nw) {}

  // notify drainers: one blocking-recv thread per conn, feeding a queue
  struct NotifyQ {
    std::thread th;
    std::mutex mu;
    std::deque<std::vector<char>> q;
    bool dead = false;
  };
  std::mutex nmu;
  std::unordered_map<uint64_t, std::unique_ptr<NotifyQ>> notify;
  ~uccl_engine() {
    // close each notify conn so its drainer's blocking recv throws, then
    // join — a detached drainer would race the queue's destruction
    std::lock_guard<std::mutex> g(nmu);
    for (auto& [id, nq] : notify) {
      try {
        ep.close_conn(id);
      } catch (std::exception const&) {
      }
      if (nq->th.joinable()) nq->th.join();
    }
  }
};

static constexpr size_t kNotifyFrame = 4096;

extern "C" {

uccl_engine_t* uccl_engine_create(int gpu, int num_workers) {
  try {
    return new uccl_engine(gpu, num_workers);
  } catch (std::exception const&) {
    return nullptr;
  }
}

void uccl_engine_destroy(uccl_engine_t* e) { delete e; }

int uccl_engine_metadata(uccl_engine_t* e, void* buf, size_t cap) {
  std::string md = e->ep.metadata();
  if (md.size() > cap) return -1;
  memcpy(buf, md.data(), md.size());
  return static_cast<int>(md.size());
}

uint64_t uccl_engine_connect(uccl_engine_t* e, void const* md, size_t len) {
  try {
    return e->ep.connect(std::string(static_cast<char const*>(md), len));
  } catch (std::exception const&) {
    return 0;
  }
}

uint64_t uccl_engine_accept(uccl_engine_t* e) {
  try {
    return e->ep.accept();
  } catch (std::exception const&) {
    return 0;
  }
}

uint64_t uccl_engine_reg(uccl_engine_t* e, void* ptr, size_t bytes,
                         int device) {
  return e->ep.reg(ptr, bytes, device);
}

void uccl_engine_dereg(uccl_engine_t* e, uint64_t mr) { e->ep.dereg(mr); }

#define WRAP(expr)                 \
  try {                            \
    expr;                          \
    return 0;                      \
  } catch (std::exception const&) { \
    return -1;                     \
  }

int uccl_engine_send(uccl_engine_t* e, uint64_t conn, void const* ptr,
                     size_t bytes, int device) {
  WRAP(e->ep.send(conn, ptr, bytes, device));
}

int uccl_engine_recv(uccl_engine_t* e, uint64_t conn, void* ptr,
                     size_t bytes, int device) {
  WRAP(e->ep.recv(conn, ptr, bytes, device));
}

int uccl_engine_advertise(uccl_engine_t* e, uint64_t mr, uint64_t offset,
                          uint64_t bytes, void* buf, size_t cap) {
  try {
    std::string ad = e->ep.advertise(mr, offset, bytes);
    if (ad.size() > cap) return -1;
    memcpy(buf, ad.data(), ad.size());
    return static_cast<int>(ad.size());
  } catch (std::exception const&) {
    return -1;
  }
}

int uccl_engine_write(uccl_engine_t* e, uint64_t conn, void const* ptr,
                      size_t bytes, int device, void const* advert,
                      size_t advert_len) {
  WRAP(e->ep.write(conn, ptr, bytes, device,
                   std::string(static_cast<char const*>(advert),
                               advert_len)));
}

int uccl_engine_read(uccl_engine_t* e, uint64_t conn, void* ptr,
                     size_t bytes, int device, void const* advert,
                     size_t advert_len) {
  WRAP(e->ep.read(conn, ptr, bytes, device,
                  std::string(static_cast<char const*>(advert),
                              advert_len)));
}

uint64_t uccl_engine_write_async(uccl_engine_t* e, uint64_t conn,
                                 void const* ptr, size_t bytes, int device,
                                 void const* advert, size_t advert_len) {
  try {
    return e->ep.write_async(conn, ptr, bytes, device,
                             std::string(static_cast<char const*>(advert),
                                         advert_len));
  } catch (std::exception const&) {
    return 0;
  }
}

uint64_t uccl_engine_read_async(uccl_engine_t* e, uint64_t conn, void* ptr,
                                size_t bytes, int device,
                                void const* advert, size_t advert_len) {
  try {
    return e->ep.read_async(conn, ptr, bytes, device,
                            std::string(static_cast<char const*>(advert),
                                        advert_len));
  } catch (std::exception const&) {
    return 0;
  }
}

int uccl_engine_poll(uccl_engine_t* e, uint64_t xfer) {
  try {
    return e->ep.poll_async(xfer) ? 1 : 0;
  } catch (std::exception const&) {
    return -1;
  }
}


int uccl_engine_notify(uccl_engine_t* e, uint64_t conn, void const* data,
                       size_t len) {
  if (len > kNotifyFrame - 16) return -1;
  std::vector<char> frame(kNotifyFrame, 0);
  uint64_t const n = len;
  memcpy(frame.data(), &n, 8);
  if (len) memcpy(frame.data() + 16, data, len);
  try {
    e->ep.send(conn, frame.data(), frame.size(), -1);
    return 0;
  } catch (std::exception const&) {
    return -1;
  }
}

int uccl_engine_notify_poll(uccl_engine_t* e, uint64_t conn, void* buf,
                            size_t cap) {
  uccl_engine::NotifyQ* nq;
  {
    std::lock_guard<std::mutex> g(e->nmu);
    auto& slot = e->notify[conn];
    if (!slot) {
      slot.reset(new uccl_engine::NotifyQ());
      auto* raw = slot.get();
      Endpoint* ep = &e->ep;
      raw->th = std::thread([raw, ep, conn] {
        std::vector<char> frame(kNotifyFrame);
        while (true) {
          try {
            ep->recv(conn, frame.data(), frame.size(), -1);
          } catch (std::exception const&) {
            std::lock_guard<std::mutex> g2(raw->mu);
            raw->dead = true;
            return;
          }
          uint64_t n = 0;
          memcpy(&n, frame.data(), 8);
          if (n > kNotifyFrame - 16) continue;  // corrupt; drop
          std::vector<char> msg(frame.begin() + 16,
                                frame.begin() + 16 + n);
          std::lock_guard<std::mutex> g2(raw->mu);
          raw->q.push_back(std::move(msg));
        }
      });
    }
    nq = slot.get();
  }
  std::lock_guard<std::mutex> g(nq->mu);
  if (nq->q.empty()) return nq->dead ? -1 : 0;
  auto& m = nq->q.front();
  if (m.size() > cap) return -1;
  int const n = static_cast<int>(m.size());
  if (n) memcpy(buf, m.data(), n);
  nq->q.pop_front();
  return n;
}

}  // extern "C"
