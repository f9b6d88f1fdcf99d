// SPDX-License-Identifier: MIT
// CPU loopback backend — BASELINE config 1 ("host-malloc ibv_reg_mr +
// ib_write_bw loopback") with no hardware; also the CI-tier backend.
#include <cstring>
#include <stdexcept>
#include <vector>

#include "rocp2p_transport.h"
#include "../../rocnrdma_amd/ops/csrc/p2p_pattern.h"

namespace rocp2p {

class FakeTransport final : public Transport {
 public:
  explicit FakeTransport(const TransportConfig& cfg) : Transport(cfg) {
    if (cfg.region_bytes % cfg.msg_bytes)
      throw std::runtime_error("region must be a multiple of msg size");
    inflight_ = cfg.inflight ? cfg.inflight : 8;
    if (inflight_ > msgs_per_region()) inflight_ = msgs_per_region();
    staging_.resize(inflight_ * cfg.msg_bytes);
    region_.resize(cfg.region_bytes);
  }

  const char* name() const override { return "fake"; }

  void post_many(uint64_t start, uint64_t n) override {
    for (uint64_t i = start; i < start + n; i++) {
      uint8_t* slot = &staging_[(i % inflight_) * cfg_.msg_bytes];
      uint8_t* dst = &region_[(i % msgs_per_region()) * cfg_.msg_bytes];
      if (cfg_.dir == Direction::Write)
        memcpy(dst, slot, cfg_.msg_bytes);
      else
        memcpy(slot, dst, cfg_.msg_bytes);
    }
  }

  void flush() override {}

  uint64_t integrity_check(uint64_t seed) override {
    const size_t words_per_msg = cfg_.msg_bytes / 8;
    uint64_t bad = 0;
    if (cfg_.dir == Direction::Write) {
      for (size_t m = 0; m < msgs_per_region(); m++) {
        uint64_t* slot = reinterpret_cast<uint64_t*>(
            &staging_[(m % inflight_) * cfg_.msg_bytes]);
        for (size_t w = 0; w < words_per_msg; w++)
          slot[w] = rocp2p_pattern_word(seed, m * words_per_msg + w);
        post_many(m, 1);
      }
      flush();
      const uint64_t* r = reinterpret_cast<const uint64_t*>(region_.data());
      for (size_t w = 0; w < cfg_.region_bytes / 8; w++)
        bad += (r[w] != rocp2p_pattern_word(seed, w));
      return bad;
    }
    // read: pattern the region, pull message by message
    uint64_t* r = reinterpret_cast<uint64_t*>(region_.data());
    for (size_t w = 0; w < cfg_.region_bytes / 8; w++)
      r[w] = rocp2p_pattern_word(seed, w);
    for (size_t m = 0; m < msgs_per_region(); m++) {
      post_many(m, 1);
      flush();
      const uint64_t* slot = reinterpret_cast<const uint64_t*>(
          &staging_[(m % inflight_) * cfg_.msg_bytes]);
      for (size_t w = 0; w < words_per_msg; w++)
        bad += (slot[w] != rocp2p_pattern_word(seed, m * words_per_msg + w));
    }
    return bad;
  }

 private:
  std::vector<uint8_t> staging_;
  std::vector<uint8_t> region_;
};

std::unique_ptr<Transport> make_fake_transport(const TransportConfig& cfg) {
  return std::make_unique<FakeTransport>(cfg);
}

}  // namespace rocp2p
