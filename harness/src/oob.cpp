// SPDX-License-Identifier: MIT
#include "rocp2p_oob.h"

#include <arpa/inet.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstdlib>
#include <cstring>
#include <sstream>
#include <stdexcept>

namespace rocp2p {

std::string kv_encode(const KvMap& kv) {
  std::ostringstream os;
  bool first = true;
  for (auto& [k, v] : kv) {
    if (!first) os << ' ';
    os << k << '=' << v;
    first = false;
  }
  os << '\n';
  return os.str();
}

KvMap kv_decode(const std::string& line) {
  KvMap kv;
  std::istringstream is(line);
  std::string tok;
  while (is >> tok) {
    auto eq = tok.find('=');
    if (eq != std::string::npos)
      kv[tok.substr(0, eq)] = tok.substr(eq + 1);
  }
  return kv;
}

OobSocket::~OobSocket() {
  if (fd_ >= 0) close(fd_);
}

void OobSocket::send_kv(const KvMap& kv) {
  std::string line = kv_encode(kv);
  const char* p = line.data();
  size_t left = line.size();
  while (left) {
    ssize_t n = ::send(fd_, p, left, 0);
    if (n <= 0) throw std::runtime_error("oob send failed");
    p += n;
    left -= (size_t)n;
  }
}

KvMap OobSocket::recv_kv() {
  for (;;) {
    auto nl = rxbuf_.find('\n');
    if (nl != std::string::npos) {
      std::string line = rxbuf_.substr(0, nl);
      rxbuf_.erase(0, nl + 1);
      return kv_decode(line);
    }
    char buf[512];
    ssize_t n = ::recv(fd_, buf, sizeof(buf), 0);
    if (n <= 0) throw std::runtime_error("oob peer closed");
    rxbuf_.append(buf, (size_t)n);
  }
}

OobServer::OobServer(int port) {
  lfd_ = ::socket(AF_INET, SOCK_STREAM, 0);
  if (lfd_ < 0) throw std::runtime_error("oob socket failed");
  int one = 1;
  setsockopt(lfd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
  sockaddr_in a{};
  a.sin_family = AF_INET;
  // Listen on all interfaces so the documented two-host client/server
  // mode actually accepts remote clients (ib_write_bw server shape);
  // ROCP2P_OOB_BIND=127.0.0.1 restricts to loopback.
  a.sin_addr.s_addr = htonl(INADDR_ANY);
  if (const char* b = getenv("ROCP2P_OOB_BIND")) {
    if (inet_pton(AF_INET, b, &a.sin_addr) != 1)
      throw std::runtime_error("ROCP2P_OOB_BIND: bad IPv4 address");
  }
  a.sin_port = htons((uint16_t)port);
  if (bind(lfd_, (sockaddr*)&a, sizeof(a)) != 0)
    throw std::runtime_error("oob bind failed");
  socklen_t len = sizeof(a);
  getsockname(lfd_, (sockaddr*)&a, &len);
  port_ = ntohs(a.sin_port);
  if (listen(lfd_, 1) != 0) throw std::runtime_error("oob listen failed");
}

OobServer::~OobServer() {
  if (lfd_ >= 0) close(lfd_);
}

OobSocket* OobServer::accept_one() {
  int fd = ::accept(lfd_, nullptr, nullptr);
  if (fd < 0) throw std::runtime_error("oob accept failed");
  auto* s = new OobSocket();
  s->fd_ = fd;
  return s;
}

OobSocket* oob_connect(const std::string& host, int port) {
  int fd = ::socket(AF_INET, SOCK_STREAM, 0);
  if (fd < 0) throw std::runtime_error("oob socket failed");
  sockaddr_in a{};
  a.sin_family = AF_INET;
  a.sin_port = htons((uint16_t)port);
  if (inet_pton(AF_INET, host.c_str(), &a.sin_addr) != 1) {
    close(fd);
    throw std::runtime_error("oob bad host (use a dotted IPv4)");
  }
  if (connect(fd, (sockaddr*)&a, sizeof(a)) != 0) {
    close(fd);
    throw std::runtime_error("oob connect failed");
  }
  auto* s = new OobSocket();
  s->fd_ = fd;
  return s;
}

}  // namespace rocp2p
