#include "trace.h"

#include <atomic>
#include <chrono>
#include <cstdlib>
#include <mutex>
#include <sstream>
#include <thread>
#include <vector>

#include "ring.h"

namespace uccl {
namespace trace {

namespace {
struct Ev {
  char const* cat;
  char const* name;  // must be string literals / static storage
  char ph;
  int64_t arg;
  uint64_t ts_us;
  uint64_t tid;
};

// bounded lock-free recording (core/ring.h MPMC): engines on hot paths
// never take a lock to emit an event; the ring is allocated lazily on
// first enable so a disabled process pays one pointer load
constexpr size_t kRingCap = 1u << 18;

std::atomic<bool> g_on{[] {
  char const* e = std::getenv("UCCL_TRACE");
  return e && e[0] == '1';
}()};
std::atomic<MpmcRing<Ev>*> g_ring{nullptr};
std::mutex g_mu;  // guards dump/clear (drain side), not recording
std::atomic<uint64_t> g_dropped{0};

MpmcRing<Ev>* ring() {
  MpmcRing<Ev>* r = g_ring.load(std::memory_order_acquire);
  if (r) return r;
  std::lock_guard<std::mutex> lk(g_mu);
  r = g_ring.load(std::memory_order_acquire);
  if (!r) {
    r = new MpmcRing<Ev>(kRingCap);
    g_ring.store(r, std::memory_order_release);
  }
  return r;
}

uint64_t now_us() {
  return std::chrono::duration_cast<std::chrono::microseconds>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

uint64_t tid() {
  return std::hash<std::thread::id>{}(std::this_thread::get_id()) & 0xffff;
}
}  // namespace

bool enabled() { return g_on.load(std::memory_order_relaxed); }
void set_enabled(bool on) { g_on.store(on, std::memory_order_relaxed); }

void event(char const* cat, char const* name, char ph, int64_t arg) {
  if (!enabled()) return;
  Ev e{cat, name, ph, arg, now_us(), tid()};
  if (!ring()->push(e)) g_dropped.fetch_add(1, std::memory_order_relaxed);
}

std::string dump_json() {
  std::lock_guard<std::mutex> lk(g_mu);
  std::ostringstream os;
  os << "{\"traceEvents\":[";
  MpmcRing<Ev>* r = g_ring.load(std::memory_order_acquire);
  bool first = true;
  Ev e;
  while (r && r->pop(&e)) {
    if (!first) os << ",";
    first = false;
    os << "{\"cat\":\"" << e.cat << "\",\"name\":\"" << e.name
       << "\",\"ph\":\"" << e.ph << "\",\"ts\":" << e.ts_us
       << ",\"pid\":1,\"tid\":" << e.tid;
    if (e.ph == 'C')
      os << ",\"args\":{\"value\":" << e.arg << "}";
    else if (e.arg)
      os << ",\"args\":{\"arg\":" << e.arg << "}";
    os << "}";
  }
  os << "],\"meta\":{\"dropped\":" << g_dropped.load() << "}}";
  return os.str();
}

void clear() {
  std::lock_guard<std::mutex> lk(g_mu);
  MpmcRing<Ev>* r = g_ring.load(std::memory_order_acquire);
  Ev e;
  while (r && r->pop(&e)) {
  }
  g_dropped.store(0, std::memory_order_relaxed);
}

size_t num_events() {
  MpmcRing<Ev>* r = g_ring.load(std::memory_order_acquire);
  return r ? r->size_approx() : 0;
}

}  // namespace trace
}  // namespace uccl
