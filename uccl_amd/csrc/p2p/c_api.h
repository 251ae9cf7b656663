// Flat C API over the P2P engine, for NIXL-style integrations that cannot
// use the Python/nanobind surface. Parity role: the reference's
// p2p/uccl_engine.h:43-315 uccl_engine_* API.
#pragma once

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct uccl_engine uccl_engine_t;

// lifecycle --------------------------------------------------------------
uccl_engine_t* uccl_engine_create(int gpu, int num_workers);
void uccl_engine_destroy(uccl_engine_t* e);

// rendezvous -------------------------------------------------------------
// Writes this endpoint's metadata blob into buf (cap bytes); returns the
// blob length, or -1 if cap is too small.
int uccl_engine_metadata(uccl_engine_t* e, void* buf, size_t cap);
// Returns a connection id > 0, or 0 on failure.
uint64_t uccl_engine_connect(uccl_engine_t* e, void const* md, size_t len);
uint64_t uccl_engine_accept(uccl_engine_t* e);

// memory registration ----------------------------------------------------
uint64_t uccl_engine_reg(uccl_engine_t* e, void* ptr, size_t bytes,
                         int device /* -1 = host */);
void uccl_engine_dereg(uccl_engine_t* e, uint64_t mr);

// two-sided (blocking) ---------------------------------------------------
int uccl_engine_send(uccl_engine_t* e, uint64_t conn, void const* ptr,
                     size_t bytes, int device);
int uccl_engine_recv(uccl_engine_t* e, uint64_t conn, void* ptr,
                     size_t bytes, int device);

// one-sided --------------------------------------------------------------
// Serializes an advert for [offset, offset+bytes) of mr into buf (>=32B);
// returns blob length or -1.
int uccl_engine_advertise(uccl_engine_t* e, uint64_t mr, uint64_t offset,
                          uint64_t bytes, void* buf, size_t cap);
int uccl_engine_write(uccl_engine_t* e, uint64_t conn, void const* ptr,
                      size_t bytes, int device, void const* advert,
                      size_t advert_len);
int uccl_engine_read(uccl_engine_t* e, uint64_t conn, void* ptr,
                     size_t bytes, int device, void const* advert,
                     size_t advert_len);

// async ------------------------------------------------------------------
uint64_t uccl_engine_write_async(uccl_engine_t* e, uint64_t conn,
                                 void const* ptr, size_t bytes, int device,
                                 void const* advert, size_t advert_len);
uint64_t uccl_engine_read_async(uccl_engine_t* e, uint64_t conn, void* ptr,
                                size_t bytes, int device,
                                void const* advert, size_t advert_len);
// 1 = done, 0 = pending, -1 = unknown id
int uccl_engine_poll(uccl_engine_t* e, uint64_t xfer);

// --- notify messages (reference: NotifyMsg, p2p/util/common.h:78) ---------
// Small out-of-band messages riding the same connection. len <= 4080.
// A connection used for notifies must not also be used for raw
// uccl_engine_recv (the notify drainer owns the inbound stream).
// 0 = sent, -1 = error
int uccl_engine_notify(uccl_engine_t* e, uint64_t conn, void const* data,
                       size_t len);
// >=0 = message length copied into buf, 0 with no message = none pending,
// -1 = error. Non-blocking.
int uccl_engine_notify_poll(uccl_engine_t* e, uint64_t conn, void* buf,
                            size_t cap);

#ifdef __cplusplus
}
#endif
