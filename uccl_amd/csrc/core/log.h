// Logging + runtime checks for the uccl_amd native core.
// Mirrors the role of the reference's include/util/debug.h (UCCL_LOG /
// UCCL_CHECK, env-set level) with a clean MI355X-native implementation.
#pragma once

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <sstream>
#include <stdexcept>
#include <string>

namespace uccl {

enum class LogLevel : int { kDebug = 0, kInfo = 1, kWarn = 2, kError = 3 };

inline LogLevel log_level() {
  static LogLevel lvl = [] {
    const char* e = std::getenv("UCCL_LOG_LEVEL");
    if (!e) return LogLevel::kWarn;
    if (!strcasecmp(e, "debug")) return LogLevel::kDebug;
    if (!strcasecmp(e, "info")) return LogLevel::kInfo;
    if (!strcasecmp(e, "warn") || !strcasecmp(e, "warning"))
      return LogLevel::kWarn;
    return LogLevel::kError;
  }();
  return lvl;
}

class CheckError : public std::runtime_error {
 public:
  explicit CheckError(std::string const& m) : std::runtime_error(m) {}
};

class LogMessage {
 public:
  LogMessage(LogLevel lvl, const char* file, int line, bool fatal = false)
      : lvl_(lvl), fatal_(fatal) {
    const char* base = strrchr(file, '/');
    ss_ << "[uccl " << tag(lvl) << " " << (base ? base + 1 : file) << ":"
        << line << "] ";
  }
  ~LogMessage() noexcept(false) {
    if (fatal_) {
      fprintf(stderr, "%s\n", ss_.str().c_str());
      fflush(stderr);
      // throw, don't abort: API-reachable failures surface as Python /
      // C++ exceptions; an unhandled throw in a worker thread still
      // terminates, matching the old abort for truly-fatal contexts
      throw CheckError(ss_.str());
    }
    if (lvl_ >= log_level()) {
      fprintf(stderr, "%s\n", ss_.str().c_str());
    }
  }
  std::ostringstream& stream() { return ss_; }

 private:
  static const char* tag(LogLevel l) {
    switch (l) {
      case LogLevel::kDebug: return "DBG";
      case LogLevel::kInfo: return "INF";
      case LogLevel::kWarn: return "WRN";
      default: return "ERR";
    }
  }
  std::ostringstream ss_;
  LogLevel lvl_;
  bool fatal_;
};

}  // namespace uccl

#define UCCL_LOG_DEBUG \
  ::uccl::LogMessage(::uccl::LogLevel::kDebug, __FILE__, __LINE__).stream()
#define UCCL_LOG_INFO \
  ::uccl::LogMessage(::uccl::LogLevel::kInfo, __FILE__, __LINE__).stream()
#define UCCL_LOG_WARN \
  ::uccl::LogMessage(::uccl::LogLevel::kWarn, __FILE__, __LINE__).stream()
#define UCCL_LOG_ERROR \
  ::uccl::LogMessage(::uccl::LogLevel::kError, __FILE__, __LINE__).stream()

#define UCCL_CHECK(cond)                                                     \
  if (!(cond))                                                               \
  ::uccl::LogMessage(::uccl::LogLevel::kError, __FILE__, __LINE__, true)     \
      .stream()                                                              \
      << "CHECK failed: " #cond " "

#define UCCL_CHECK_HIP(expr)                                                 \
  do {                                                                       \
    hipError_t _e = (expr);                                                  \
    if (_e != hipSuccess)                                                    \
      ::uccl::LogMessage(::uccl::LogLevel::kError, __FILE__, __LINE__, true) \
              .stream()                                                      \
          << "HIP error: " << hipGetErrorString(_e) << " @ " #expr;          \
  } while (0)
