// SPDX-License-Identifier: MIT
// Native transport interface of the rocp2p_bw harness — RDMA write/read
// semantics over pluggable backends (same contract as the Python layer
// rocnrdma_amd/transport/base.py; this is the reference-grade C++
// implementation the reference repo assumed existed in the form of
// OFED perftest — SURVEY.md §4).
#pragma once
#include <cstdint>
#include <cstddef>
#include <memory>
#include <string>

namespace rocp2p {

enum class Direction { Write, Read };

struct TransportConfig {
  size_t msg_bytes = 64ull << 20;
  size_t region_bytes = 1ull << 30;
  Direction dir = Direction::Write;
  int device_index = 0;      // GPU ordinal (hip/verbs-gpu backends)
  int num_streams = 2;       // hip stream engine
  size_t inflight = 0;       // 0 = backend-chosen
  size_t chain = 0;          // verbs: WRs per doorbell (0 = auto)
  std::string engine = "auto";  // hip: auto|kernel|stream
  std::string verbs_mr = "auto";  // verbs: auto|peer|dmabuf|host
  bool wc_staging = false;   // hip: write-combined pinned staging
};

class Transport {
 public:
  virtual ~Transport() = default;
  virtual const char* name() const = 0;
  // enqueue messages [start, start+n) (dst offset = (i % msgs_per_region)
  // * msg_bytes, staging slot = i % inflight)
  virtual void post_many(uint64_t start, uint64_t n) = 0;
  virtual void flush() = 0;
  // full-region pattern transfer + receiver-side verification;
  // returns mismatching 8-byte words (0 = intact).  Never timed.
  virtual uint64_t integrity_check(uint64_t seed) = 0;

  size_t msg_bytes() const { return cfg_.msg_bytes; }
  size_t region_bytes() const { return cfg_.region_bytes; }
  size_t inflight() const { return inflight_; }
  size_t msgs_per_region() const { return cfg_.region_bytes / cfg_.msg_bytes; }
  Direction dir() const { return cfg_.dir; }

 protected:
  explicit Transport(const TransportConfig& cfg) : cfg_(cfg) {}
  TransportConfig cfg_;
  size_t inflight_ = 0;
};

// Factory.  Throws std::runtime_error with a actionable message when a
// backend is unavailable on this host.
std::unique_ptr<Transport> make_transport(const std::string& name,
                                          const TransportConfig& cfg);

bool hip_available();
bool verbs_runtime_available();  // libibverbs loads AND >=1 device

// Client/server mode (ib_write_bw shape): the target registers the
// region and answers OOB control ops; the client performs one-sided
// ops against it.  Implemented in verbs_backend.cpp; throws on
// verbs-less builds.
std::unique_ptr<Transport> make_verbs_client(const TransportConfig& cfg,
                                             const std::string& host,
                                             int port);
int run_verbs_target(const TransportConfig& cfg, int port,
                     void (*announce)(int));

}  // namespace rocp2p
