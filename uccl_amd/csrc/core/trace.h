// Chrome-trace event recorder (about://tracing / Perfetto JSON).
//
// Parity role: the reference's NPKit event tracing
// (experimental/lite/lite-collective/core/npkit.cc). Host-side here:
// engines record begin/end spans and instant events into a bounded
// in-memory ring; dump_json() emits the Trace Event Format. Enabled by
// UCCL_TRACE=1 (or programmatically); disabled recording is one
// relaxed-atomic load.

#pragma once

#include <cstdint>
#include <string>

namespace uccl {
namespace trace {

bool enabled();
void set_enabled(bool on);

// ph: 'B' begin, 'E' end, 'i' instant, 'C' counter (value via arg)
void event(char const* cat, char const* name, char ph, int64_t arg = 0);

// RAII span; no-op when tracing is off at construction
class Span {
 public:
  Span(char const* cat, char const* name) : cat_(cat), name_(name) {
    if ((on_ = enabled())) event(cat_, name_, 'B');
  }
  ~Span() {
    if (on_) event(cat_, name_, 'E');
  }
  Span(Span const&) = delete;
  Span& operator=(Span const&) = delete;

 private:
  char const* cat_;
  char const* name_;
  bool on_ = false;
};

// drains the recorded events into Trace Event Format JSON
std::string dump_json();
void clear();
size_t num_events();

}  // namespace trace
}  // namespace uccl
