// Env-var driven configuration, the uccl_amd analog of the reference's
// UCCL_PARAM system (collective/rdma/param.h:16-44): lazy-cached lookups,
// zero mandatory configuration.
#pragma once

#include <cstdint>
#include <cstdlib>
#include <string>

namespace uccl {

inline int64_t env_int(const char* name, int64_t dflt) {
  const char* e = std::getenv(name);
  if (!e || !*e) return dflt;
  return strtoll(e, nullptr, 0);
}

inline std::string env_str(const char* name, const std::string& dflt) {
  const char* e = std::getenv(name);
  return (e && *e) ? std::string(e) : dflt;
}

inline bool env_bool(const char* name, bool dflt) {
  const char* e = std::getenv(name);
  if (!e || !*e) return dflt;
  return !(e[0] == '0' || e[0] == 'n' || e[0] == 'N' || e[0] == 'f' ||
           e[0] == 'F');
}

// UCCL_<X> with NCCL_<X> / RCCL_<X> fallback (the reference's param.h
// honors NCCL aliases, param.h:31-44): pass the UCCL-prefixed name.
inline const char* env_aliased(const char* uccl_name) {
  const char* e = std::getenv(uccl_name);
  if (e && *e) return e;
  std::string tail(uccl_name);
  if (tail.rfind("UCCL_", 0) == 0) tail = tail.substr(5);
  std::string n = "NCCL_" + tail;
  if ((e = std::getenv(n.c_str())) && *e) return e;
  n = "RCCL_" + tail;
  if ((e = std::getenv(n.c_str())) && *e) return e;
  return nullptr;
}

inline std::string env_str_aliased(const char* uccl_name,
                                   const std::string& dflt) {
  const char* e = env_aliased(uccl_name);
  return e ? std::string(e) : dflt;
}

// Declares a lazily-cached env parameter accessor:  uccl_param_Foo()
#define UCCL_PARAM(Name, Env, Default)              \
  inline int64_t uccl_param_##Name() {              \
    static int64_t v = ::uccl::env_int(Env, Default); \
    return v;                                       \
  }

}  // namespace uccl
