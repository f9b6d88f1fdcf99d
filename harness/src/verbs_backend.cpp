// SPDX-License-Identifier: MIT
// IB-verbs backend — the REAL PeerDirect data path: ibv_reg_mr on a
// hipMalloc pointer (dispatched to the rocp2p bridge by the IB core's
// peer-memory probe) and one-sided RDMA WRITE/READ through a loopback
// RC QP pair, perftest-style.  MR modes:
//   peer   — ibv_reg_mr(pd, hipMalloc ptr, ...): requires rocp2p.ko
//            (the product under test);
//   dmabuf — ibv_reg_dmabuf_mr over hipMemGetHandleForAddressRange:
//            the modern kernel-module-free path, used as cross-check
//            (SURVEY.md §5 "dmabuf is a validation path, not the design");
//   host   — plain host memory (BASELINE config 1 loopback).
//
// This pool ships no rdma-core, so the implementation is compile-gated
// on <infiniband/verbs.h>; on verbs-less hosts a stub throws with an
// actionable message and verbs_runtime_available() (dlopen probe)
// reports false.  The gated code follows the stable documented verbs
// API; it has NOT run against real hardware from this pool — treat the
// first run on an HCA host as a bring-up step (docs/RUNBOOK.md).
#include <dlfcn.h>

#include <cstring>
#include <stdexcept>
#include <string>

#include "rocp2p_transport.h"

namespace rocp2p {

bool verbs_runtime_available() {
  void* h = dlopen("libibverbs.so.1", RTLD_NOW | RTLD_LOCAL);
  if (!h) return false;
  using get_list_t = void** (*)(int*);
  using free_list_t = void (*)(void**);
  auto get_list = (get_list_t)dlsym(h, "ibv_get_device_list");
  auto free_list = (free_list_t)dlsym(h, "ibv_free_device_list");
  bool ok = false;
  if (get_list) {
    int n = 0;
    void** devs = get_list(&n);
    ok = devs && n > 0;
    if (devs && free_list) free_list(devs);
  }
  dlclose(h);
  return ok;
}

}  // namespace rocp2p

#if defined(__has_include)
#if __has_include(<infiniband/verbs.h>)
#define ROCP2P_HAVE_VERBS 1
#endif
#endif

#ifdef ROCP2P_HAVE_VERBS
#include <infiniband/verbs.h>
#include <hip/hip_runtime.h>

#include <vector>

#include "../../rocnrdma_amd/ops/csrc/p2p_kernels.h"
#include "../../rocnrdma_amd/ops/csrc/p2p_pattern.h"

namespace rocp2p {

#define VB_THROW(cond, msg)                                          \
  do {                                                               \
    if (!(cond)) throw std::runtime_error(std::string("verbs: ") + msg); \
  } while (0)

class VerbsTransport final : public Transport {
 public:
  explicit VerbsTransport(const TransportConfig& cfg) : Transport(cfg) {
    if (cfg.region_bytes % cfg.msg_bytes)
      throw std::runtime_error("region must be a multiple of msg size");
    inflight_ = cfg.inflight ? cfg.inflight : 64;
    if (inflight_ > msgs_per_region()) inflight_ = msgs_per_region();

    int ndev = 0;
    ibv_device** devs = ibv_get_device_list(&ndev);
    VB_THROW(devs && ndev > 0, "no IB devices");
    ctx_ = ibv_open_device(devs[0]);
    ibv_free_device_list(devs);
    VB_THROW(ctx_, "ibv_open_device failed");

    pd_ = ibv_alloc_pd(ctx_);
    VB_THROW(pd_, "ibv_alloc_pd failed");
    cq_ = ibv_create_cq(ctx_, 2 * (int)inflight_ + 16, nullptr, nullptr, 0);
    VB_THROW(cq_, "ibv_create_cq failed");

    // staging: host-pinned (hip) so the same buffers work for GPU paths
    mr_mode_ = cfg.verbs_mr;
    if (mr_mode_ == "auto") mr_mode_ = hip_available() ? "peer" : "host";
    if (mr_mode_ == "host") {
      staging_ = (uint8_t*)aligned_alloc(4096, inflight_ * cfg.msg_bytes);
      region_host_ = (uint8_t*)aligned_alloc(4096, cfg.region_bytes);
      region_ptr_ = region_host_;
    } else {
      VB_THROW(hipSetDevice(cfg.device_index) == hipSuccess, "hipSetDevice");
      VB_THROW(hipHostMalloc((void**)&staging_,
                             inflight_ * cfg.msg_bytes, 0) == hipSuccess,
               "hipHostMalloc staging");
      VB_THROW(hipMalloc((void**)&region_gpu_, cfg.region_bytes) ==
                   hipSuccess,
               "hipMalloc region");
      region_ptr_ = region_gpu_;
    }

    staging_mr_ = ibv_reg_mr(pd_, staging_, inflight_ * cfg.msg_bytes,
                             IBV_ACCESS_LOCAL_WRITE);
    VB_THROW(staging_mr_, "ibv_reg_mr(staging) failed");

    int acc = IBV_ACCESS_LOCAL_WRITE | IBV_ACCESS_REMOTE_WRITE |
              IBV_ACCESS_REMOTE_READ;
    if (mr_mode_ == "dmabuf") {
#ifdef IBV_ACCESS_RELAXED_ORDERING
      // optional; not required for correctness
#endif
      int fd = -1;
      // export the HBM range as a dmabuf (ROCm >= 5.7)
      VB_THROW(hipMemGetHandleForAddressRange(
                   &fd, region_gpu_, cfg.region_bytes,
                   hipMemRangeHandleTypeDmaBufFd, 0) == hipSuccess,
               "hipMemGetHandleForAddressRange(dmabuf) failed");
      region_mr_ = ibv_reg_dmabuf_mr(pd_, 0, cfg.region_bytes,
                                     (uint64_t)region_gpu_, fd, acc);
      VB_THROW(region_mr_, "ibv_reg_dmabuf_mr failed");
    } else {
      // peer mode: the IB core's peer-memory probe must dispatch this
      // GPU VA to the rocp2p bridge; failure here on a GPU pointer
      // means the bridge is not loaded/registered (RUNBOOK.md).
      region_mr_ = ibv_reg_mr(pd_, region_ptr_, cfg.region_bytes, acc);
      VB_THROW(region_mr_,
               mr_mode_ == "peer"
                   ? "ibv_reg_mr(GPU VA) failed — rocp2p bridge loaded?"
                   : "ibv_reg_mr(region) failed");
    }

    qp_send_ = make_qp();
    qp_recv_ = make_qp();
    connect_loopback(qp_send_, qp_recv_);
    connect_loopback(qp_recv_, qp_send_);
  }

  ~VerbsTransport() override {
    if (qp_send_) ibv_destroy_qp(qp_send_);
    if (qp_recv_) ibv_destroy_qp(qp_recv_);
    if (region_mr_) ibv_dereg_mr(region_mr_);
    if (staging_mr_) ibv_dereg_mr(staging_mr_);
    if (cq_) ibv_destroy_cq(cq_);
    if (pd_) ibv_dealloc_pd(pd_);
    if (ctx_) ibv_close_device(ctx_);
    if (region_host_) free(region_host_);
    if (region_gpu_) hipFree(region_gpu_);
    if (mr_mode_ == "host") free(staging_);
    else if (staging_) hipHostFree(staging_);
  }

  const char* name() const override { return "verbs"; }

  void post_many(uint64_t start, uint64_t n) override {
    for (uint64_t i = start; i < start + n; i++) {
      if (outstanding_ >= inflight_) drain(1);
      ibv_sge sge;
      sge.addr = (uint64_t)(staging_ + (i % inflight_) * cfg_.msg_bytes);
      sge.length = (uint32_t)cfg_.msg_bytes;
      sge.lkey = staging_mr_->lkey;
      ibv_send_wr wr;
      memset(&wr, 0, sizeof(wr));
      wr.wr_id = i;
      wr.sg_list = &sge;
      wr.num_sge = 1;
      wr.opcode = cfg_.dir == Direction::Write ? IBV_WR_RDMA_WRITE
                                               : IBV_WR_RDMA_READ;
      wr.send_flags = IBV_SEND_SIGNALED;
      wr.wr.rdma.remote_addr =
          (uint64_t)region_ptr_ + (i % msgs_per_region()) * cfg_.msg_bytes;
      wr.wr.rdma.rkey = region_mr_->rkey;
      ibv_send_wr* bad = nullptr;
      VB_THROW(ibv_post_send(qp_send_, &wr, &bad) == 0, "ibv_post_send");
      outstanding_++;
    }
  }

  void flush() override { drain(outstanding_); }

  uint64_t integrity_check(uint64_t seed) override {
    const size_t words_per_msg = cfg_.msg_bytes / 8;
    const bool gpu = region_gpu_ != nullptr;
    uint64_t bad = 0;
    if (cfg_.dir == Direction::Write) {
      for (size_t base = 0; base < msgs_per_region(); base += inflight_) {
        size_t burst = std::min(inflight_, msgs_per_region() - base);
        for (size_t m = base; m < base + burst; m++) {
          uint64_t* slot = reinterpret_cast<uint64_t*>(
              staging_ + (m % inflight_) * cfg_.msg_bytes);
          for (size_t w = 0; w < words_per_msg; w++)
            slot[w] = rocp2p_pattern_word(seed, m * words_per_msg + w);
        }
        post_many(base, burst);
        flush();
      }
      if (gpu) {
        unsigned long long* d_bad = nullptr;
        VB_THROW(hipMalloc((void**)&d_bad, 8) == hipSuccess, "hipMalloc");
        hipMemset(d_bad, 0, 8);
        VB_THROW(rocp2p_verify(region_gpu_, cfg_.region_bytes, seed, d_bad,
                               0) == hipSuccess,
                 "verify kernel");
        unsigned long long h_bad = 0;
        hipMemcpy(&h_bad, d_bad, 8, hipMemcpyDeviceToHost);
        hipFree(d_bad);
        return h_bad;
      }
      const uint64_t* r = reinterpret_cast<const uint64_t*>(region_host_);
      for (size_t w = 0; w < cfg_.region_bytes / 8; w++)
        bad += (r[w] != rocp2p_pattern_word(seed, w));
      return bad;
    }
    // read direction: pattern the region, RDMA_READ back, verify host side
    if (gpu) {
      VB_THROW(rocp2p_fill(region_gpu_, cfg_.region_bytes, seed, 0) ==
                   hipSuccess,
               "fill kernel");
      hipDeviceSynchronize();
    } else {
      uint64_t* r = reinterpret_cast<uint64_t*>(region_host_);
      for (size_t w = 0; w < cfg_.region_bytes / 8; w++)
        r[w] = rocp2p_pattern_word(seed, w);
    }
    for (size_t base = 0; base < msgs_per_region(); base += inflight_) {
      size_t burst = std::min(inflight_, msgs_per_region() - base);
      post_many(base, burst);
      flush();
      for (size_t m = base; m < base + burst; m++) {
        const uint64_t* slot = reinterpret_cast<const uint64_t*>(
            staging_ + (m % inflight_) * cfg_.msg_bytes);
        for (size_t w = 0; w < words_per_msg; w++)
          bad += (slot[w] != rocp2p_pattern_word(seed, m * words_per_msg + w));
      }
    }
    return bad;
  }

 private:
  ibv_qp* make_qp() {
    ibv_qp_init_attr a;
    memset(&a, 0, sizeof(a));
    a.send_cq = cq_;
    a.recv_cq = cq_;
    a.qp_type = IBV_QPT_RC;
    a.cap.max_send_wr = (uint32_t)(2 * inflight_ + 8);
    a.cap.max_recv_wr = 16;
    a.cap.max_send_sge = 1;
    a.cap.max_recv_sge = 1;
    ibv_qp* qp = ibv_create_qp(pd_, &a);
    VB_THROW(qp, "ibv_create_qp failed");
    return qp;
  }

  void connect_loopback(ibv_qp* qp, ibv_qp* peer) {
    ibv_port_attr pattr;
    VB_THROW(ibv_query_port(ctx_, 1, &pattr) == 0, "ibv_query_port");

    ibv_qp_attr at;
    memset(&at, 0, sizeof(at));
    at.qp_state = IBV_QPS_INIT;
    at.pkey_index = 0;
    at.port_num = 1;
    at.qp_access_flags = IBV_ACCESS_LOCAL_WRITE | IBV_ACCESS_REMOTE_WRITE |
                         IBV_ACCESS_REMOTE_READ;
    VB_THROW(ibv_modify_qp(qp, &at,
                           IBV_QP_STATE | IBV_QP_PKEY_INDEX | IBV_QP_PORT |
                               IBV_QP_ACCESS_FLAGS) == 0,
             "modify->INIT");

    memset(&at, 0, sizeof(at));
    at.qp_state = IBV_QPS_RTR;
    at.path_mtu = pattr.active_mtu;
    at.dest_qp_num = peer->qp_num;
    at.rq_psn = 0;
    at.max_dest_rd_atomic = 4;
    at.min_rnr_timer = 12;
    at.ah_attr.port_num = 1;
    if (pattr.link_layer == IBV_LINK_LAYER_ETHERNET) {
      union ibv_gid gid;
      VB_THROW(ibv_query_gid(ctx_, 1, gid_index_, &gid) == 0,
               "ibv_query_gid");
      at.ah_attr.is_global = 1;
      at.ah_attr.grh.dgid = gid;
      at.ah_attr.grh.sgid_index = (uint8_t)gid_index_;
      at.ah_attr.grh.hop_limit = 1;
    } else {
      at.ah_attr.is_global = 0;
      at.ah_attr.dlid = pattr.lid;
    }
    VB_THROW(ibv_modify_qp(qp, &at,
                           IBV_QP_STATE | IBV_QP_AV | IBV_QP_PATH_MTU |
                               IBV_QP_DEST_QPN | IBV_QP_RQ_PSN |
                               IBV_QP_MAX_DEST_RD_ATOMIC |
                               IBV_QP_MIN_RNR_TIMER) == 0,
             "modify->RTR");

    memset(&at, 0, sizeof(at));
    at.qp_state = IBV_QPS_RTS;
    at.sq_psn = 0;
    at.timeout = 14;
    at.retry_cnt = 7;
    at.rnr_retry = 7;
    at.max_rd_atomic = 4;
    VB_THROW(ibv_modify_qp(qp, &at,
                           IBV_QP_STATE | IBV_QP_SQ_PSN | IBV_QP_TIMEOUT |
                               IBV_QP_RETRY_CNT | IBV_QP_RNR_RETRY |
                               IBV_QP_MAX_QP_RD_ATOMIC) == 0,
             "modify->RTS");
  }

  void drain(size_t at_least) {
    ibv_wc wc[16];
    size_t done = 0;
    while (done < at_least && outstanding_ > 0) {
      int n = ibv_poll_cq(cq_, 16, wc);
      VB_THROW(n >= 0, "ibv_poll_cq");
      for (int i = 0; i < n; i++)
        VB_THROW(wc[i].status == IBV_WC_SUCCESS,
                 std::string("completion error: ") +
                     ibv_wc_status_str(wc[i].status));
      done += n;
      outstanding_ -= n;
    }
  }

  ibv_context* ctx_ = nullptr;
  ibv_pd* pd_ = nullptr;
  ibv_cq* cq_ = nullptr;
  ibv_qp* qp_send_ = nullptr;
  ibv_qp* qp_recv_ = nullptr;
  ibv_mr* staging_mr_ = nullptr;
  ibv_mr* region_mr_ = nullptr;
  uint8_t* staging_ = nullptr;
  uint8_t* region_host_ = nullptr;
  uint8_t* region_gpu_ = nullptr;
  uint8_t* region_ptr_ = nullptr;
  std::string mr_mode_;
  size_t outstanding_ = 0;
  int gid_index_ = 1;  // RoCEv2 default; override via env if needed
};

std::unique_ptr<Transport> make_verbs_transport(const TransportConfig& cfg) {
  return std::make_unique<VerbsTransport>(cfg);
}

}  // namespace rocp2p

#else  // !ROCP2P_HAVE_VERBS

namespace rocp2p {

std::unique_ptr<Transport> make_verbs_transport(const TransportConfig&) {
  throw std::runtime_error(
      "verbs backend compiled out: <infiniband/verbs.h> (rdma-core) was "
      "not present at build time. Rebuild on an HCA-equipped host.");
}

}  // namespace rocp2p

#endif
