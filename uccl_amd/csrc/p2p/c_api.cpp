#include "c_api.h"

#include <cstring>
#include <exception>

#include "endpoint.h"

using uccl::p2p::Endpoint;

struct uccl_engine {
  Endpoint ep;
  uccl_engine(int gpu, int nw) : ep(gpu, nw) {}
};

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

}  // extern "C"
