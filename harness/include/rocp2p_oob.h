// SPDX-License-Identifier: MIT
// Minimal TCP out-of-band bootstrap for two-endpoint runs (the C++
// mirror of rocnrdma_amd/transport/oob.py).  Wire format: one
// newline-terminated line of space-separated key=value tokens per
// message — enough for QP/MR exchange, no JSON dependency.
#pragma once
#include <cstdint>
#include <map>
#include <string>

namespace rocp2p {

using KvMap = std::map<std::string, std::string>;

std::string kv_encode(const KvMap& kv);
KvMap kv_decode(const std::string& line);

class OobSocket {
 public:
  ~OobSocket();
  void send_kv(const KvMap& kv);
  KvMap recv_kv();  // throws on EOF
  int fd_ = -1;
  std::string rxbuf_;
};

class OobServer {
 public:
  explicit OobServer(int port);  // 0 = ephemeral
  ~OobServer();
  int port() const { return port_; }
  OobSocket* accept_one();  // caller owns

 private:
  int lfd_ = -1;
  int port_ = 0;
};

OobSocket* oob_connect(const std::string& host, int port);  // caller owns

}  // namespace rocp2p
