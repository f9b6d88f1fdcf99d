// SPDX-License-Identifier: MIT
// rocp2p_bw — native bandwidth harness (the perftest-style tool the
// reference assumed its users would bring: README.md:67 "IB Verbs
// interface must be used ..."; here it is, with GPU-only fallbacks).
//
//   rocp2p_bw [--transport fake|hip|verbs] [--msg BYTES] [--region BYTES]
//             [--dir write|read] [--secs S] [--gpus N] [--sweep]
//             [--engine auto|kernel|stream] [--mr auto|peer|dmabuf|host]
//             [--seed S] [--json]
//
// --gpus N fans out one transport ("QP") per GPU on N worker threads and
// reports per-GPU + aggregate bandwidth (BASELINE configs 4-5).
#include <algorithm>
#include <atomic>
#include <chrono>
#include <condition_variable>
#include <mutex>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <stdexcept>
#include <string>
#include <thread>
#include <vector>

#include "rocp2p_transport.h"

#ifdef ROCNR_FAKEVERBS_PEER
#include "../../module/shim/peer_glue.h"
#endif

namespace rocp2p {
std::unique_ptr<Transport> make_fake_transport(const TransportConfig&);
std::unique_ptr<Transport> make_hip_transport(const TransportConfig&);
std::unique_ptr<Transport> make_verbs_transport(const TransportConfig&);

std::unique_ptr<Transport> make_transport(const std::string& name,
                                          const TransportConfig& cfg) {
  if (name == "fake") return make_fake_transport(cfg);
  if (name == "hip") return make_hip_transport(cfg);
  if (name == "verbs") return make_verbs_transport(cfg);
  if (name == "auto") {
    if (verbs_runtime_available()) return make_verbs_transport(cfg);
    if (hip_available()) return make_hip_transport(cfg);
    return make_fake_transport(cfg);
  }
  throw std::runtime_error("unknown transport: " + name);
}
}  // namespace rocp2p

using namespace rocp2p;
using clk = std::chrono::steady_clock;

struct Result {
  double gbps = 0, secs = 0;
  uint64_t msgs = 0, bad = ~0ull;
  double lat_min = 0, lat_p50 = 0, lat_p99 = 0, lat_max = 0;  // us
};

// ib_write_lat analog: one message at a time, completion-to-completion
static Result run_lat(Transport& tp, int iters, uint64_t seed,
                      bool integrity) {
  Result r;
  tp.post_many(0, 1);
  tp.flush();
  std::vector<double> us;
  us.reserve(iters);
  for (int i = 0; i < iters; i++) {
    auto t0 = clk::now();
    tp.post_many(i, 1);
    tp.flush();
    us.push_back(
        std::chrono::duration<double, std::micro>(clk::now() - t0).count());
  }
  std::sort(us.begin(), us.end());
  r.msgs = iters;
  r.lat_min = us.front();
  r.lat_p50 = us[us.size() / 2];
  r.lat_p99 = us[(size_t)(us.size() * 0.99)];
  r.lat_max = us.back();
  r.secs = 0;
  if (integrity) r.bad = tp.integrity_check(seed);
  return r;
}

// bench.py step mode: W untimed warmup steps then EXACTLY K timed
// steps, one full region pass per step (the driver's steps/warmup
// contract, executed inside the native data plane).
static Result run_steps(Transport& tp, int steps, int warmup,
                        uint64_t seed, bool integrity) {
  Result r;
  uint64_t mps = tp.msgs_per_region();
  for (int w = 0; w < warmup; w++) {
    tp.post_many((uint64_t)w * mps, mps);
    tp.flush();
  }
  auto t0 = clk::now();
  for (int s = 0; s < steps; s++) {
    tp.post_many((uint64_t)s * mps, mps);
    tp.flush();
  }
  r.secs = std::chrono::duration<double>(clk::now() - t0).count();
  r.msgs = (uint64_t)steps * mps;
  r.gbps = (double)r.msgs * tp.msg_bytes() / r.secs / 1e9;
  if (integrity) r.bad = tp.integrity_check(seed);
  return r;
}

static Result run_point(Transport& tp, double secs, uint64_t seed,
                        bool integrity) {
  Result r;
  // warmup: one bounded region pass
  uint64_t warm = std::min<uint64_t>(tp.msgs_per_region(),
                                     std::max<uint64_t>(64, tp.inflight()));
  tp.post_many(0, warm);
  tp.flush();

  uint64_t burst = std::max<uint64_t>(8, tp.inflight());
  uint64_t posted = 0;
  auto t0 = clk::now();
  auto t_end = t0 + std::chrono::duration_cast<clk::duration>(
                        std::chrono::duration<double>(secs));
  while (clk::now() < t_end) {
    tp.post_many(posted, burst);
    posted += burst;
    tp.flush();
  }
  r.secs = std::chrono::duration<double>(clk::now() - t0).count();
  r.msgs = posted;
  r.gbps = (double)posted * tp.msg_bytes() / r.secs / 1e9;
  if (integrity) r.bad = tp.integrity_check(seed);
  return r;
}

int main(int argc, char** argv) {
  std::string transport = "auto";
  TransportConfig cfg;
  double secs = 1.0;
  int gpus = 1;
  bool sweep = false, json = false, integrity = true, bidir = false;
  int lat_iters = 0;
  int steps = 0, step_warmup = 0;
  int device_index = 0;
  int serve_port = -1;
  bool remote_selftest = false;
  std::string connect_to;
  uint64_t seed = 0xC0FFEE;

  for (int i = 1; i < argc; i++) {
    std::string a = argv[i];
    auto next = [&]() -> std::string {
      if (i + 1 >= argc) throw std::runtime_error("missing value for " + a);
      return argv[++i];
    };
    if (a == "--transport") transport = next();
    else if (a == "--msg") cfg.msg_bytes = strtoull(next().c_str(), 0, 0);
    else if (a == "--region")
      cfg.region_bytes = strtoull(next().c_str(), 0, 0);
    else if (a == "--dir") {
      std::string d = next();
      if (d == "bidir") bidir = true;
      cfg.dir = d == "read" ? Direction::Read : Direction::Write;
    }
    else if (a == "--secs") secs = atof(next().c_str());
    else if (a == "--gpus") gpus = atoi(next().c_str());
    else if (a == "--engine") cfg.engine = next();
    else if (a == "--mr") cfg.verbs_mr = next();
    else if (a == "--streams") cfg.num_streams = atoi(next().c_str());
    else if (a == "--inflight") cfg.inflight = strtoull(next().c_str(), 0, 0);
    else if (a == "--chain") cfg.chain = strtoull(next().c_str(), 0, 0);
    else if (a == "--seed") seed = strtoull(next().c_str(), 0, 0);
    else if (a == "--wc") cfg.wc_staging = true;
    else if (a == "--serve") serve_port = atoi(next().c_str());
    else if (a == "--connect") connect_to = next();
    else if (a == "--remote-selftest") remote_selftest = true;
    else if (a == "--peer-revoke-selftest") {
#ifdef ROCNR_FAKEVERBS_PEER
      // SURVEY §3.4 through the verbs surface: register a peer MR via
      // the real bridge, move data, then the producer frees the
      // memory under the live MR — the invalidation must tear the MR
      // down (bridge free_cb -> IB-core invalidate) and subsequent
      // posts must fail as remote-access errors, with deregistration
      // still idempotent.
      try {
        cfg.verbs_mr = "peer";
        cfg.msg_bytes = 64 << 10;
        cfg.region_bytes = 8 << 20;
        auto tp = make_verbs_transport(cfg);
        tp->post_many(0, 16);
        tp->flush();
        rocnr_glue_revoke_last();  // GPU frees the region now
        bool failed = false;
        try {
          tp->post_many(16, 1);
          tp->flush();
        } catch (const std::exception&) {
          failed = true;
        }
        if (!failed) {
          fprintf(stderr, "post after revoke unexpectedly succeeded\n");
          return 1;
        }
        tp.reset();  // dereg path after invalidation: must be a no-op
        printf("{\"mode\":\"peer-revoke-selftest\",\"result\":\"ok\"}\n");
        return 0;
      } catch (const std::exception& e) {
        fprintf(stderr, "peer-revoke-selftest: %s\n", e.what());
        return 3;
      }
#else
      fprintf(stderr, "--peer-revoke-selftest needs the fullstack build\n");
      return 2;
#endif
    }
    else if (a == "--lat") lat_iters = atoi(next().c_str());
    else if (a == "--steps") steps = atoi(next().c_str());
    else if (a == "--warmup") step_warmup = atoi(next().c_str());
    else if (a == "--device") device_index = atoi(next().c_str());
    else if (a == "--sweep") sweep = true;
    else if (a == "--json") json = true;
    else if (a == "--no-integrity") integrity = false;
    else {
      fprintf(stderr, "unknown arg %s\n", a.c_str());
      return 2;
    }
  }

  if (steps > 0) {
    // bench.py step mode: one transport, one JSON line with the timed
    // seconds so the caller can aggregate across ranks.
    try {
      cfg.device_index = device_index;
      auto tp = make_transport(transport, cfg);
      Result r = run_steps(*tp, steps, step_warmup, seed, integrity);
      const char* ok =
          !integrity ? "skipped" : (r.bad == 0 ? "ok" : "FAILED");
      printf("{\"transport\":\"%s\",\"mode\":\"steps\",\"msg_bytes\":%zu,"
             "\"region_bytes\":%zu,\"mr\":\"%s\",\"steps\":%d,"
             "\"warmup\":%d,\"secs\":%.6f,\"msgs\":%llu,\"gbps\":%.3f,"
             "\"integrity\":\"%s\"}\n",
             tp->name(), cfg.msg_bytes, cfg.region_bytes,
             cfg.verbs_mr.c_str(), steps, step_warmup, r.secs,
             (unsigned long long)r.msgs, r.gbps, ok);
      return (integrity && r.bad) ? 1 : 0;
    } catch (const std::exception& e) {
      fprintf(stderr, "steps: %s\n", e.what());
      return 3;
    }
  }

  if (serve_port >= 0) {
    // passive target (ib_write_bw server shape); blocks until the
    // client says bye
    try {
      run_verbs_target(cfg, serve_port, [](int p) {
        printf("listening on port %d\n", p);
        fflush(stdout);  // piped consumers (tests) wait on this line
      });
    } catch (const std::exception& e) {
      fprintf(stderr, "serve: %s\n", e.what());
      return 3;
    }
    return 0;
  }

  if (remote_selftest || !connect_to.empty()) try {
    // client/server data plane through the verbs backend.  selftest:
    // both endpoints in THIS process over real TCP loopback (runs
    // under the fake-verbs CI layer; on an HCA host it exercises the
    // real NIC end to end on one box).
    std::unique_ptr<Transport> tp;
    std::thread server_thread;
    if (remote_selftest) {
      static std::atomic<int> s_port{0};
      static std::mutex s_mu;
      static std::condition_variable s_cv;
      TransportConfig scfg = cfg;
      server_thread = std::thread([scfg] {
        try {
          run_verbs_target(scfg, 0, [](int p) {
            {
              std::lock_guard<std::mutex> g(s_mu);
              s_port.store(p);
            }
            s_cv.notify_all();
          });
        } catch (const std::exception& e) {
          fprintf(stderr, "selftest server: %s\n", e.what());
          {
            std::lock_guard<std::mutex> g(s_mu);
            s_port.store(-1);  // unblock the client side
          }
          s_cv.notify_all();
        }
      });
      {
        std::unique_lock<std::mutex> lk(s_mu);
        s_cv.wait(lk, [] { return s_port.load() != 0; });
      }
      if (s_port.load() < 0) {
        server_thread.join();
        throw std::runtime_error("selftest server failed to start");
      }
      tp = make_verbs_client(cfg, "127.0.0.1", s_port.load());
    } else {
      auto colon = connect_to.rfind(':');
      if (colon == std::string::npos) {
        fprintf(stderr, "--connect needs HOST:PORT\n");
        return 2;
      }
      tp = make_verbs_client(cfg, connect_to.substr(0, colon),
                             atoi(connect_to.c_str() + colon + 1));
    }
    Result r = lat_iters ? run_lat(*tp, lat_iters, seed, integrity)
                         : run_point(*tp, secs, seed, integrity);
    const char* ok = !integrity ? "skipped" : (r.bad == 0 ? "ok" : "FAILED");
    printf("{\"transport\":\"%s\",\"msg_bytes\":%zu,\"mode\":\"%s\","
           "\"gbps\":%.3f,\"msgs\":%llu,\"remote_integrity\":\"%s\"}\n",
           tp->name(), cfg.msg_bytes,
           remote_selftest ? "remote-selftest" : "client", r.gbps,
           (unsigned long long)r.msgs, ok);
    tp.reset();  // sends nothing; dtor closes OOB -> server exits loop
    if (server_thread.joinable()) server_thread.join();
    return (integrity && r.bad) ? 1 : 0;
  } catch (const std::exception& e) {
    fprintf(stderr, "client: %s\n", e.what());
    return 3;
  }

  std::vector<size_t> sizes =
      sweep ? std::vector<size_t>{4ull << 10, 64ull << 10, 1ull << 20,
                                  16ull << 20, 64ull << 20}
            : std::vector<size_t>{cfg.msg_bytes};

  if (!json)
    printf("%12s %6s %4s %10s %12s %10s\n", "msg", "dir", "gpu", "GB/s",
           "msgs/s", "integrity");
  int rc = 0;
  for (size_t msg : sizes) {
    TransportConfig c = cfg;
    c.msg_bytes = msg;
    c.region_bytes = std::max(cfg.region_bytes / msg, (size_t)1) * msg;

    // --dir bidir: two workers per GPU (write + read concurrently) —
    // full-duplex PCIe, the ib_write_bw --bidirectional analog
    int workers = bidir ? 2 * gpus : gpus;
    std::vector<Result> res(workers);
    std::vector<std::string> names(workers);
    std::vector<std::thread> ths;
    std::vector<std::string> errs(workers);
    for (int w = 0; w < workers; w++) {
      ths.emplace_back([&, w] {
        try {
          TransportConfig cg = c;
          cg.device_index = bidir ? w / 2 : w;
          if (bidir)
            cg.dir = (w & 1) ? Direction::Read : Direction::Write;
          auto tp = make_transport(transport, cg);
          names[w] = tp->name();
          res[w] = lat_iters
                       ? run_lat(*tp, lat_iters, seed + w, integrity)
                       : run_point(*tp, secs, seed + w, integrity);
        } catch (const std::exception& e) {
          errs[w] = e.what();
        }
      });
    }
    for (auto& t : ths) t.join();

    double agg = 0;
    uint64_t bad = 0, msgs = 0;
    for (int w = 0; w < workers; w++) {
      if (!errs[w].empty()) {
        fprintf(stderr, "worker %d: %s\n", w, errs[w].c_str());
        return 3;
      }
      agg += res[w].gbps;
      msgs += res[w].msgs;
      if (integrity) bad += res[w].bad;
    }
    const char* ok = !integrity ? "skipped" : (bad == 0 ? "ok" : "FAILED");
    if (lat_iters) {
      // latency lines (per GPU 0; multi-gpu latency is per-worker)
      if (json)
        printf("{\"transport\":\"%s\",\"msg_bytes\":%zu,\"mode\":\"lat\","
               "\"iters\":%d,\"us_min\":%.2f,\"us_p50\":%.2f,"
               "\"us_p99\":%.2f,\"us_max\":%.2f,\"integrity\":\"%s\"}\n",
               names[0].c_str(), msg, lat_iters, res[0].lat_min,
               res[0].lat_p50, res[0].lat_p99, res[0].lat_max, ok);
      else
        printf("%12zu lat us: min %.2f p50 %.2f p99 %.2f max %.2f  %s\n",
               msg, res[0].lat_min, res[0].lat_p50, res[0].lat_p99,
               res[0].lat_max, ok);
      if (integrity && bad) rc = 1;
      continue;
    }
    if (json) {
      printf("{\"transport\":\"%s\",\"msg_bytes\":%zu,\"direction\":\"%s\","
             "\"gpus\":%d,\"gbps\":%.3f,\"msgs_per_s\":%.0f,"
             "\"integrity\":\"%s\"}\n",
             names[0].c_str(), msg,
             bidir ? "bidir" : (c.dir == Direction::Write ? "write" : "read"),
             gpus, agg, msgs / res[0].secs, ok);
    } else {
      printf("%12zu %6s %4d %10.3f %12.0f %10s\n", msg,
             bidir ? "bidir" : (c.dir == Direction::Write ? "write" : "read"),
             gpus, agg, msgs / res[0].secs, ok);
    }
    if (integrity && bad) rc = 1;
  }
  return rc;
}
