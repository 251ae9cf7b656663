#include "trace.h"

#include <atomic>
#include <chrono>
#include <cstdlib>
#include <mutex>
#include <sstream>
#include <thread>
#include <vector>

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

constexpr size_t kMaxEvents = 1u << 20;

std::atomic<bool> g_on{[] {
  char const* e = std::getenv("UCCL_TRACE");
  return e && e[0] == '1';
}()};
std::mutex g_mu;
std::vector<Ev> g_events;
std::atomic<uint64_t> g_dropped{0};

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
  uint64_t const ts = now_us();
  uint64_t const t = tid();
  std::lock_guard<std::mutex> lk(g_mu);
  if (g_events.size() >= kMaxEvents) {
    g_dropped.fetch_add(1, std::memory_order_relaxed);
    return;
  }
  g_events.push_back({cat, name, ph, arg, ts, t});
}

std::string dump_json() {
  std::lock_guard<std::mutex> lk(g_mu);
  std::ostringstream os;
  os << "{\"traceEvents\":[";
  for (size_t i = 0; i < g_events.size(); ++i) {
    auto const& e = g_events[i];
    if (i) os << ",";
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
  g_events.clear();
  g_dropped.store(0, std::memory_order_relaxed);
}

size_t num_events() {
  std::lock_guard<std::mutex> lk(g_mu);
  return g_events.size();
}

}  // namespace trace
}  // namespace uccl
