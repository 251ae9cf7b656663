// Minimal TCP helpers for out-of-band control and the socket data plane.
// Parity role: the reference's include/util/net.h (listen/connect helpers,
// send_message/receive_message, get_oob_ip), rewritten from scratch.
#pragma once

#include <arpa/inet.h>
#include <ifaddrs.h>

#include "env.h"
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cerrno>
#include <cstdint>
#include <cstring>
#include <string>

#include "log.h"

namespace uccl {
namespace net {

inline int listen_on(uint16_t* port /*in-out; 0 = ephemeral*/) {
  int fd = ::socket(AF_INET, SOCK_STREAM, 0);
  UCCL_CHECK(fd >= 0) << "socket: " << strerror(errno);
  int one = 1;
  setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
  sockaddr_in addr{};
  addr.sin_family = AF_INET;
  addr.sin_addr.s_addr = htonl(INADDR_ANY);
  addr.sin_port = htons(*port);
  UCCL_CHECK(::bind(fd, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)) == 0)
      << "bind port " << *port << ": " << strerror(errno);
  UCCL_CHECK(::listen(fd, 128) == 0) << "listen: " << strerror(errno);
  socklen_t len = sizeof(addr);
  UCCL_CHECK(::getsockname(fd, reinterpret_cast<sockaddr*>(&addr), &len) == 0);
  *port = ntohs(addr.sin_port);
  return fd;
}

inline int connect_to(const std::string& ip, uint16_t port,
                      int timeout_ms = 20000) {
  int fd = ::socket(AF_INET, SOCK_STREAM, 0);
  UCCL_CHECK(fd >= 0) << "socket: " << strerror(errno);
  sockaddr_in addr{};
  addr.sin_family = AF_INET;
  addr.sin_port = htons(port);
  UCCL_CHECK(inet_pton(AF_INET, ip.c_str(), &addr.sin_addr) == 1)
      << "bad ip " << ip;
  int tries = timeout_ms / 50 + 1;
  while (tries-- > 0) {
    if (::connect(fd, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)) == 0) {
      int one = 1;
      setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
      return fd;
    }
    if (errno != ECONNREFUSED && errno != ETIMEDOUT && errno != EINTR) break;
    usleep(50 * 1000);
    ::close(fd);
    fd = ::socket(AF_INET, SOCK_STREAM, 0);
  }
  UCCL_CHECK(false) << "connect " << ip << ":" << port << " failed: "
                    << strerror(errno);
  return -1;
}

inline void send_all(int fd, void const* buf, size_t n) {
  auto* p = static_cast<char const*>(buf);
  while (n > 0) {
    ssize_t r = ::send(fd, p, n, MSG_NOSIGNAL);
    if (r < 0 && errno == EINTR) continue;
    UCCL_CHECK(r > 0) << "send: " << strerror(errno);
    p += r;
    n -= static_cast<size_t>(r);
  }
}

// returns false on clean EOF at a message boundary
inline bool recv_all(int fd, void* buf, size_t n) {
  auto* p = static_cast<char*>(buf);
  bool first = true;
  while (n > 0) {
    ssize_t r = ::recv(fd, p, n, 0);
    if (r < 0 && errno == EINTR) continue;
    if (r == 0 && first) return false;
    UCCL_CHECK(r > 0) << "recv: " << strerror(errno);
    first = false;
    p += r;
    n -= static_cast<size_t>(r);
  }
  return true;
}

// Best-effort non-loopback IPv4 of this host (for metadata blobs).
// UCCL_SOCKET_IFNAME (NCCL_/RCCL_SOCKET_IFNAME honored as aliases, like
// the reference's param system) pins the interface by name prefix.
inline std::string local_ip() {
  std::string result = "127.0.0.1";
  ifaddrs* ifs = nullptr;
  if (getifaddrs(&ifs) != 0) return result;
  std::string want = env_str_aliased("UCCL_SOCKET_IFNAME", "");
  for (ifaddrs* it = ifs; it; it = it->ifa_next) {
    if (!it->ifa_addr || it->ifa_addr->sa_family != AF_INET) continue;
    if (!want.empty() &&
        std::string(it->ifa_name).rfind(want, 0) != 0)
      continue;
    auto* sin = reinterpret_cast<sockaddr_in*>(it->ifa_addr);
    char buf[INET_ADDRSTRLEN];
    inet_ntop(AF_INET, &sin->sin_addr, buf, sizeof(buf));
    std::string ip(buf);
    if (ip != "127.0.0.1" || !want.empty()) {
      result = ip;
      break;
    }
  }
  freeifaddrs(ifs);
  return result;
}

}  // namespace net
}  // namespace uccl
